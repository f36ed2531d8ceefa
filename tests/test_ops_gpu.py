"""GPU numerics tests: each HIP/CDNA4 kernel vs a plain fp32 torch reference.

Mirrors the reference's per-op parity tests (tests/unit/ops/* in
microsoft/DeepSpeed): same-op comparison within dtype tolerance. Every test
here requires an MI355X and the in-tree extension (no eager fallback).
"""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu


def _ext():
    from deepspeed_amd.ops import get_ext, has_ext
    assert has_ext(), "HIP extension must be built on the GPU box"
    return get_ext()


def _rtol_atol(dtype):
    if dtype == torch.float32:
        return 1e-5, 1e-5
    return 2e-2, 2e-2  # bf16/fp16: one-ulp-ish on normalized values


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float16, torch.float32])
@pytest.mark.parametrize("rows,H", [(512, 4096), (33, 1000), (2048, 8192)])
def test_rms_norm_fwd_bwd(dtype, rows, H):
    from deepspeed_amd.ops import rms_norm
    _ext()
    torch.manual_seed(0)
    x = torch.randn(rows, H, device="cuda", dtype=dtype, requires_grad=True)
    w = torch.randn(H, device="cuda", dtype=dtype, requires_grad=True)
    y = rms_norm(x, w, eps=1e-6)

    xf = x.detach().float().requires_grad_(True)
    wf = w.detach().float().requires_grad_(True)
    inv = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-6)
    ref = xf * inv * wf

    rtol, atol = _rtol_atol(dtype)
    torch.testing.assert_close(y.float(), ref, rtol=rtol, atol=atol)

    dy = torch.randn_like(y)
    y.backward(dy)
    ref.backward(dy.float())
    torch.testing.assert_close(x.grad.float(), xf.grad, rtol=rtol,
                               atol=atol * 4)
    # dw accumulates over rows in fp32: scale tolerance with sqrt(rows)
    torch.testing.assert_close(w.grad.float(), wf.grad, rtol=5e-2,
                               atol=0.5 * rows ** 0.5 * atol)


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
@pytest.mark.parametrize("bias", [True, False])
def test_layer_norm_fwd_bwd(dtype, bias):
    from deepspeed_amd.ops import layer_norm
    _ext()
    torch.manual_seed(1)
    rows, H = 768, 2048
    x = torch.randn(rows, H, device="cuda", dtype=dtype, requires_grad=True)
    w = torch.randn(H, device="cuda", dtype=dtype, requires_grad=True)
    b = torch.randn(H, device="cuda", dtype=dtype, requires_grad=True) if bias else None
    y = layer_norm(x, w, b, eps=1e-5)

    xf = x.detach().float().requires_grad_(True)
    wf = w.detach().float().requires_grad_(True)
    bf = b.detach().float().requires_grad_(True) if bias else None
    ref = torch.nn.functional.layer_norm(xf, (H,), wf, bf, 1e-5)

    rtol, atol = _rtol_atol(dtype)
    torch.testing.assert_close(y.float(), ref, rtol=rtol, atol=atol)

    dy = torch.randn_like(y)
    y.backward(dy)
    ref.backward(dy.float())
    torch.testing.assert_close(x.grad.float(), xf.grad, rtol=rtol, atol=atol * 4)
    torch.testing.assert_close(w.grad.float(), wf.grad, rtol=5e-2,
                               atol=0.5 * rows ** 0.5 * atol)
    if bias:
        torch.testing.assert_close(b.grad.float(), bf.grad, rtol=5e-2,
                                   atol=0.5 * rows ** 0.5 * atol)


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
def test_rope_fwd_bwd(dtype):
    from deepspeed_amd.ops import apply_rope, rope_tables
    from deepspeed_amd.ops.rope import _torch_rope
    _ext()
    torch.manual_seed(2)
    B, S, Hh, D = 2, 512, 8, 128
    cos, sin = rope_tables(D, 2048, device="cuda")
    x = torch.randn(B, S, Hh, D, device="cuda", dtype=dtype, requires_grad=True)
    y = apply_rope(x, cos, sin)
    ref = _torch_rope(x.detach().float(), cos, sin)
    rtol, atol = _rtol_atol(dtype)
    torch.testing.assert_close(y.float(), ref.float(), rtol=rtol, atol=atol)
    dy = torch.randn_like(y)
    y.backward(dy)
    dref = _torch_rope(dy.float(), cos, sin, backward=True)
    torch.testing.assert_close(x.grad.float(), dref.float(), rtol=rtol, atol=atol)


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
@pytest.mark.parametrize("fn_name", ["swiglu", "geglu"])
def test_gated_act_fwd_bwd(dtype, fn_name):
    import deepspeed_amd.ops as ops
    _ext()
    fn = getattr(ops, fn_name)
    torch.manual_seed(3)
    N = 4096 * 128 + 7
    g = torch.randn(N, device="cuda", dtype=dtype, requires_grad=True)
    u = torch.randn(N, device="cuda", dtype=dtype, requires_grad=True)
    y = fn(g, u)

    gf = g.detach().float().requires_grad_(True)
    uf = u.detach().float().requires_grad_(True)
    if fn_name == "swiglu":
        ref = torch.nn.functional.silu(gf) * uf
    else:
        ref = torch.nn.functional.gelu(gf, approximate="tanh") * uf
    rtol, atol = _rtol_atol(dtype)
    torch.testing.assert_close(y.float(), ref, rtol=rtol, atol=atol)
    dy = torch.randn_like(y)
    y.backward(dy)
    ref.backward(dy.float())
    torch.testing.assert_close(g.grad.float(), gf.grad, rtol=rtol, atol=atol * 2)
    torch.testing.assert_close(u.grad.float(), uf.grad, rtol=rtol, atol=atol * 2)


@pytest.mark.parametrize("wd", [0.0, 0.1])
def test_fused_adam_vs_torch(wd):
    """HIP fused Adam (fp32 flat path) vs torch.optim.AdamW, 10 steps."""
    from deepspeed_amd.ops import FusedAdam
    _ext()
    torch.manual_seed(4)
    p_ref = torch.randn(3 * 4096 + 5, device="cuda", requires_grad=True)
    p_hip = p_ref.detach().clone().requires_grad_(True)
    opt_ref = torch.optim.AdamW([p_ref], lr=1e-3, betas=(0.9, 0.95),
                                eps=1e-8, weight_decay=wd)
    opt_hip = FusedAdam([p_hip], lr=1e-3, betas=(0.9, 0.95), eps=1e-8,
                        weight_decay=wd, adam_w_mode=True)
    for i in range(10):
        g = torch.randn_like(p_ref)
        p_ref.grad = g.clone()
        p_hip.grad = g.clone()
        opt_ref.step()
        opt_hip.step()
    torch.testing.assert_close(p_hip, p_ref, rtol=1e-5, atol=1e-6)


def test_fused_adam_bf16_master():
    """bf16 params keep an fp32 master; trajectory must match fp32 AdamW
    then cast."""
    from deepspeed_amd.ops import FusedAdam
    _ext()
    torch.manual_seed(5)
    base = torch.randn(8192, device="cuda")
    p_ref = base.clone().requires_grad_(True)
    p_hip = base.bfloat16().requires_grad_(True)
    # align masters: FusedAdam initializes master from the bf16 value
    p_ref.data = p_hip.detach().float()
    opt_ref = torch.optim.AdamW([p_ref], lr=1e-3, betas=(0.9, 0.95),
                                eps=1e-8, weight_decay=0.0)
    opt_hip = FusedAdam([p_hip], lr=1e-3, betas=(0.9, 0.95), eps=1e-8,
                        weight_decay=0.0)
    for _ in range(5):
        g = torch.randn_like(p_ref)
        p_ref.grad = g.clone()
        p_hip.grad = g.bfloat16()
        opt_ref.step()
        opt_hip.step()
    # bf16 grads vs fp32 grads diverge slightly; loose tolerance
    torch.testing.assert_close(p_hip.float(), p_ref, rtol=3e-2, atol=3e-2)


@pytest.mark.gpu
@pytest.mark.parametrize("causal", [True, False])
@pytest.mark.parametrize("shape", [(2, 256, 8, 2), (1, 512, 32, 8),
                                   (2, 4096, 32, 8)])
def test_flash_attn_fwd_numerics(causal, shape):
    """Hand-written MFMA flash fwd vs fp32 SDPA reference (GQA, causal)."""
    from deepspeed_amd.ops.attention import flash_attn_fwd, sdpa_reference
    B, S, H, Hkv = shape
    torch.manual_seed(0)
    q = torch.randn(B, S, H, 128, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, S, Hkv, 128, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, S, Hkv, 128, device="cuda", dtype=torch.bfloat16)
    with torch.no_grad():
        out = flash_attn_fwd(q, k, v, causal=causal)
        ref = sdpa_reference(q, k, v, causal=causal)
    err = (out.float() - ref.float()).abs()
    rel = err.max() / ref.float().abs().max()
    assert torch.isfinite(out.float()).all()
    assert rel < 3e-2, f"max rel err {rel}"
    assert err.mean() < 5e-3, f"mean err {err.mean()}"


@pytest.mark.gpu
def test_flash_attn_perf_smoke():
    """Measure flash fwd vs torch SDPA on the bench shape (informational:
    v2 is numerics-verified at 130 TF; the remaining guide-ladder steps —
    KVBLK=64, async staging, defer-max — are round-2 work, so the model
    path keeps SDPA unless DS_AMD_FLASH=1)."""
    import time
    from deepspeed_amd.ops.attention import flash_attn_fwd
    import torch.nn.functional as F
    B, S, H, Hkv, D = 8, 4096, 32, 8, 128
    q = torch.randn(B, S, H, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, S, Hkv, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, S, Hkv, D, device="cuda", dtype=torch.bfloat16)

    def time_fn(fn, n=10):
        for _ in range(3):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(n):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / n

    with torch.no_grad():
        t_flash = time_fn(lambda: flash_attn_fwd(q, k, v, True))
        qt, kt, vt = (t.transpose(1, 2) for t in (q, k, v))
        t_sdpa = time_fn(lambda: F.scaled_dot_product_attention(
            qt, kt, vt, is_causal=True, enable_gqa=True))
    flops = 4 * B * H * S * S * D / 2  # causal
    print(f"\nflash: {t_flash*1e3:.2f} ms ({flops/t_flash/1e12:.0f} TF) "
          f"sdpa: {t_sdpa*1e3:.2f} ms ({flops/t_sdpa/1e12:.0f} TF)")
    assert t_flash < 0.1, "flash fwd pathologically slow"  # sanity only


@pytest.mark.gpu
@pytest.mark.skipif(os.environ.get("DS_AMD_FLASH_BWD_TEST") != "1",
                    reason="flash bwd GPU validation pending (round-2); "
                           "set DS_AMD_FLASH_BWD_TEST=1 to run")
@pytest.mark.parametrize("shape", [(1, 4, 2, 256), (2, 8, 2, 512)])
def test_flash_attn_bwd_numerics(shape):
    """Hand-written MFMA flash bwd vs autograd SDPA (BHSD, causal, GQA)."""
    import torch.nn.functional as F
    from deepspeed_amd.ops.attention import flash_attn_bwd
    B, H, Hkv, S = shape
    D = 128
    torch.manual_seed(0)
    q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    v = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    rep = H // Hkv
    o = F.scaled_dot_product_attention(
        q, k.repeat_interleave(rep, 1), v.repeat_interleave(rep, 1),
        is_causal=True)
    do = torch.randn_like(o)
    o.backward(do)

    scale = 1.0 / D ** 0.5
    st = (q.detach().float() @ k.detach().float()
          .repeat_interleave(rep, 1).transpose(-1, -2)) * scale
    mask = torch.triu(torch.ones(S, S, dtype=torch.bool, device="cuda"), 1)
    lse = torch.logsumexp(st.masked_fill(mask, -float("inf")), dim=-1)

    dq, dk, dv = flash_attn_bwd(q.detach(), k.detach(), v.detach(),
                                o.detach(), do, lse, causal=True)
    for got, want, name in ((dq, q.grad, "dq"), (dk, k.grad, "dk"),
                            (dv, v.grad, "dv")):
        rel = (got.float() - want.float()).abs().max() / \
            want.float().abs().max()
        assert rel < 5e-2, (name, rel)


@pytest.mark.gpu
@pytest.mark.skipif(os.environ.get("DS_AMD_FLASH_BWD_TEST") != "1",
                    reason="flash bwd GPU validation pending (round-2)")
def test_flash_attn_func_autograd():
    """End-to-end differentiable flash attention vs SDPA autograd (BSHD)."""
    from deepspeed_amd.ops.attention import flash_attn_func, sdpa_reference
    B, S, H, Hkv, D = 2, 512, 8, 2, 128
    torch.manual_seed(1)
    q = torch.randn(B, S, H, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn(B, S, Hkv, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    v = torch.randn(B, S, Hkv, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    o = flash_attn_func(q, k, v, causal=True)
    do = torch.randn_like(o)
    o.backward(do)
    got = (q.grad.clone(), k.grad.clone(), v.grad.clone())

    q2 = q.detach().clone().requires_grad_(True)
    k2 = k.detach().clone().requires_grad_(True)
    v2 = v.detach().clone().requires_grad_(True)
    ref = sdpa_reference(q2, k2, v2, causal=True)
    ref.backward(do)
    for g, w, n in zip(got, (q2.grad, k2.grad, v2.grad), "qkv"):
        rel = (g.float() - w.float()).abs().max() / w.float().abs().max()
        assert rel < 5e-2, (n, rel)


@pytest.mark.gpu
@pytest.mark.skipif(os.environ.get("DS_AMD_FUSED_CE_TEST") != "1",
                    reason="fused CE GPU validation pending (round-2); "
                           "set DS_AMD_FUSED_CE_TEST=1 to run")
def test_fused_cross_entropy_numerics():
    """Fused bf16 CE (loss + dlogits) vs fp32 torch cross_entropy."""
    from deepspeed_amd.ops.cross_entropy import _FusedCE
    torch.manual_seed(0)
    N, V = 1024, 32000
    logits = (torch.randn(N, V, device="cuda") * 3).bfloat16() \
        .requires_grad_(True)
    labels = torch.randint(0, V, (N,), device="cuda")
    labels[::7] = -100  # ignore rows

    loss_sum, count = _FusedCE.apply(logits, labels, -100)
    (loss_sum / count).backward()

    l2 = logits.detach().float().requires_grad_(True)
    ref = torch.nn.functional.cross_entropy(l2, labels, ignore_index=-100,
                                            reduction="sum")
    (ref / count).backward()
    assert abs(loss_sum.item() - ref.item()) / ref.item() < 2e-2
    err = (logits.grad.float() - l2.grad).abs().max()
    assert err < 1e-3, err


def test_block_sparse_attention_gpu_numerics():
    """Gather-path block-sparse attention on device vs fp32 masked SDPA
    (batched hipBLASLt GEMMs + index_reduce segment softmax)."""
    import torch.nn.functional as F
    from deepspeed_amd.ops.sparse_attention import (FixedSparsityConfig,
                                                    block_sparse_attention)
    torch.manual_seed(0)
    B, H, S, D, bs = 2, 4, 512, 64, 64
    q, k, v = (torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
               for _ in range(3))
    layout = FixedSparsityConfig(H, block=bs, num_local_blocks=4,
                                 attention="unidirectional").make_layout(S)
    layout[..., 0] |= layout.sum(-1) == 0
    got = block_sparse_attention(q, k, v, layout, bs)
    mask = layout.repeat_interleave(bs, 1).repeat_interleave(bs, 2)
    want = F.scaled_dot_product_attention(
        q.float(), k.float(), v.float(),
        attn_mask=mask.unsqueeze(0).cuda())
    torch.testing.assert_close(got.float(), want, atol=2e-2, rtol=2e-2)


def test_evoformer_attention_gpu():
    from deepspeed_amd.ops.evoformer import DS4Sci_EvoformerAttention
    torch.manual_seed(1)
    B, N, S, H, D = 1, 4, 64, 4, 32
    Q, K, V = (torch.randn(B, N, S, H, D, device="cuda",
                           dtype=torch.bfloat16) for _ in range(3))
    bias1 = torch.randn(B, N, 1, 1, S, device="cuda")
    bias2 = torch.randn(B, 1, H, S, S, device="cuda")
    out = DS4Sci_EvoformerAttention(Q, K, V, [bias1, bias2])
    assert out.shape == (B, N, S, H, D) and out.dtype == torch.bfloat16
    ref = DS4Sci_EvoformerAttention(Q.float().cpu(), K.float().cpu(),
                                    V.float().cpu(),
                                    [bias1.cpu(), bias2.cpu()])
    torch.testing.assert_close(out.float().cpu(), ref, atol=2e-2, rtol=2e-2)


@pytest.mark.gpu
def test_fused_lion_gpu_matches_torch():
    """GPU fused Lion (csrc/optim.hip) vs the torch composition."""
    from deepspeed_amd.ops.lion import Lion
    torch.manual_seed(0)
    p_gpu = torch.randn(4097, device="cuda", requires_grad=True)
    p_cpu = p_gpu.detach().cpu().clone().requires_grad_(True)
    og, oc = Lion([p_gpu], lr=1e-2, weight_decay=0.01), \
        Lion([p_cpu], lr=1e-2, weight_decay=0.01)
    for _ in range(3):
        g = torch.randn(4097)
        p_gpu.grad = g.cuda()
        p_cpu.grad = g
        og.step()
        oc.step()
    torch.testing.assert_close(p_gpu.cpu(), p_cpu, rtol=1e-5, atol=1e-6)


@pytest.mark.gpu
def test_fused_lamb_gpu_matches_torch():
    """GPU fused LAMB (2-phase kernel) vs the torch composition."""
    from deepspeed_amd.ops.lamb import FusedLamb
    torch.manual_seed(1)
    p_gpu = torch.randn(3000, device="cuda", requires_grad=True)
    p_cpu = p_gpu.detach().cpu().clone().requires_grad_(True)
    og = FusedLamb([p_gpu], lr=1e-2, weight_decay=0.01)
    oc = FusedLamb([p_cpu], lr=1e-2, weight_decay=0.01)
    for _ in range(3):
        g = torch.randn(3000)
        p_gpu.grad = g.cuda()
        p_cpu.grad = g
        og.step()
        oc.step()
    torch.testing.assert_close(p_gpu.cpu(), p_cpu, rtol=1e-4, atol=1e-5)


@pytest.mark.gpu
def test_transpose_bf16_kernel():
    """Tiled transpose kernel vs torch permute for both wrapper layouts."""
    from deepspeed_amd.ops.attention import _vt_from_bshd, _t_last2_bhsd
    torch.manual_seed(0)
    v = torch.randn(2, 160, 3, 128, device="cuda", dtype=torch.bfloat16)
    got = _vt_from_bshd(v)
    want = v.permute(0, 2, 3, 1).contiguous()
    assert torch.equal(got, want)
    x = torch.randn(2, 4, 256, 128, device="cuda", dtype=torch.bfloat16)
    got = _t_last2_bhsd(x)
    want = x.transpose(-1, -2).contiguous()
    assert torch.equal(got, want)


@pytest.mark.gpu
def test_fused_softmax_gpu():
    """Masked/alibi/causal fused softmax vs the torch reference path."""
    from deepspeed_amd.ops.softmax_dropout import fused_softmax
    torch.manual_seed(0)
    B, H, Sq, Sk = 2, 4, 33, 65
    x = torch.randn(B, H, Sq, Sk, device="cuda", dtype=torch.bfloat16)
    mask = torch.where(torch.rand(B, 1, 1, Sk, device="cuda") > 0.2,
                       0.0, float("-inf")).bfloat16()
    slopes = torch.tensor([2.0 ** (-i) for i in range(H)], device="cuda")
    for kwargs in ({"causal": True}, {"mask": mask.expand(B, H, Sq, Sk)},
                   {"alibi_slopes": slopes, "causal": True},
                   {"mask": mask.expand(B, H, Sq, Sk),
                    "alibi_slopes": slopes}):
        got = fused_softmax(x, heads=H, scale=0.3, **kwargs)
        want = fused_softmax(x.cpu(), heads=H, scale=0.3,
                             **{k: v.cpu() if torch.is_tensor(v) else v
                                for k, v in kwargs.items()})
        torch.testing.assert_close(got.float().cpu(), want.float(),
                                   rtol=2e-2, atol=2e-2)


@pytest.mark.gpu
def test_fused_dropout_gpu():
    """Dropout statistics, determinism by seed, and mask-exact backward."""
    from deepspeed_amd.ops.softmax_dropout import fused_bias_dropout_residual
    torch.manual_seed(0)
    x = torch.randn(512, 256, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    bias = torch.randn(256, device="cuda", dtype=torch.bfloat16,
                       requires_grad=True)
    res = torch.randn_like(x)
    y = fused_bias_dropout_residual(x, bias, res, ratio=0.3, seed=7)
    y2 = fused_bias_dropout_residual(x, bias, res, ratio=0.3, seed=7)
    assert torch.equal(y, y2)  # seed-deterministic
    y.sum().backward()
    # the backward reveals the exact keep mask (grad = 1/(1-p) on kept)
    kept = x.grad.float() != 0
    rate = kept.float().mean().item()
    assert abs(rate - 0.7) < 0.03, rate
    torch.testing.assert_close(
        x.grad.float()[kept],
        torch.full_like(x.grad.float()[kept], 1 / 0.7), rtol=1e-2,
        atol=1e-3)
    # forward values: kept scaled by 1/(1-p) (+res), dropped = res
    ref = torch.where(kept, (x.detach() + bias.detach()).float() / 0.7,
                      torch.zeros(())) + res.float()
    torch.testing.assert_close(y.float(), ref, rtol=3e-2, atol=3e-2)


@pytest.mark.gpu
def test_paged_decode_attention_gpu():
    """Split-S paged flash-decode vs dense SDPA over the gathered blocks."""
    from deepspeed_amd.ops.paged_attention import paged_decode_attention
    torch.manual_seed(0)
    n, H, Hkv, D, BS = 3, 8, 2, 128, 16
    lens = torch.tensor([70, 33, 128], dtype=torch.int32, device="cuda")
    max_blocks = 8
    nb = n * max_blocks
    kpool = torch.randn(nb, Hkv, BS, D, device="cuda", dtype=torch.bfloat16)
    vpool = torch.randn_like(kpool)
    # each seq uses a scrambled set of blocks
    g = torch.Generator().manual_seed(3)
    perm = torch.randperm(nb, generator=g)
    table = perm[:n * max_blocks].view(n, max_blocks).int().cuda()
    q = torch.randn(n, H, D, device="cuda", dtype=torch.bfloat16)
    got = paged_decode_attention(q, kpool, vpool, table, lens, splits=4)

    # reference: gather each sequence's kv and run fp32 attention
    for i in range(n):
        L = int(lens[i])
        blocks = table[i][: (L + BS - 1) // BS].long()
        k = kpool[blocks].permute(1, 0, 2, 3).reshape(Hkv, -1, D)[:, :L]
        v = vpool[blocks].permute(1, 0, 2, 3).reshape(Hkv, -1, D)[:, :L]
        rep = H // Hkv
        k = k.repeat_interleave(rep, 0).float()
        v = v.repeat_interleave(rep, 0).float()
        qi = q[i].float().unsqueeze(1)                     # [H, 1, D]
        s = (qi @ k.transpose(-1, -2)) / D ** 0.5
        ref = (torch.softmax(s, -1) @ v).squeeze(1)        # [H, D]
        rel = (got[i].float() - ref).abs().max() / ref.abs().max()
        assert rel < 2e-2, (i, rel)


@pytest.mark.gpu
def test_nhwc_bias_add_gpu():
    """Fused channels-last bias-add variants vs the torch fallback
    (reference csrc/spatial/csrc/opt_bias_add.cu surface)."""
    from deepspeed_amd.ops.spatial import nhwc_bias_add
    torch.manual_seed(0)
    for dt in (torch.bfloat16, torch.float16):
        rows, C = 1024, 320
        act = torch.randn(rows, C, device="cuda", dtype=dt)
        bias = torch.randn(C, device="cuda", dtype=dt)
        other = torch.randn(rows, C, device="cuda", dtype=dt)
        obias = torch.randn(C, device="cuda", dtype=dt)
        for args in ((act, bias), (act, bias, other),
                     (act, bias, other, obias)):
            got = nhwc_bias_add(*args)
            want = nhwc_bias_add(*(a.cpu() for a in args)).to(dt)
            torch.testing.assert_close(got.float().cpu(), want.float(),
                                       rtol=2e-2, atol=2e-2)


@pytest.mark.gpu
def test_token_gather_scatter_gpu():
    """Row-coalesced token gather/scatter kernels vs torch gather/scatter
    (random-LTD path), forward and backward."""
    from deepspeed_amd.ops.token_ops import token_gather, token_scatter
    torch.manual_seed(0)
    B, S, K, D = 3, 64, 23, 256
    for dt in (torch.bfloat16, torch.float32):
        x = torch.randn(B, S, D, device="cuda", dtype=dt, requires_grad=True)
        idx = torch.stack([torch.randperm(S, device="cuda")[:K].sort().values
                           for _ in range(B)]).int()
        sub = token_gather(x, idx)
        g = idx.long().unsqueeze(-1).expand(B, K, D)
        want_sub = x.gather(1, g)
        assert torch.equal(sub, want_sub)
        y = token_scatter(x.detach().requires_grad_(), 2 * sub.detach(), idx)
        want_y = x.detach().scatter(1, g, 2 * sub.detach())
        assert torch.equal(y, want_y)
        # backward through gather
        loss = (sub.float() ** 2).sum()
        loss.backward()
        xr = x.detach().clone().requires_grad_()
        (xr.gather(1, g).float() ** 2).sum().backward()
        torch.testing.assert_close(x.grad, xr.grad)
