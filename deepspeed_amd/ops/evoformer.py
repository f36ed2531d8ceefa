"""EvoformerAttention (DeepSpeed4Science) — MSA/pair attention with up to
two additive biases (reference: deepspeed/ops/deepspeed4science/
evoformer_attn.py DS4Sci_EvoformerAttention + csrc/deepspeed4science/
evoformer_attn kernels).

Shapes follow the reference contract exactly:
  Q, K, V : [B, N, S, H, D]   (batch, rows/MSA seqs, seq, heads, head_dim)
  bias1   : broadcastable to [B, N, 1, 1, S]   (residue mask bias)
  bias2   : broadcastable to [B, 1, H, S, S]   (pair bias)
  out     : [B, N, S, H, D]
Softmax over the last key axis at 1/sqrt(D) scaling, computed in fp32.

MI355X status: composed from torch ops (the permutes collapse into the
hipBLASLt batched-GEMM calls and the softmax is MIOpen's); the dedicated
fused MFMA kernel follows the flash-attention ladder in
ops/csrc/attention.hip and is scheduled after its backward validates —
the op keeps this call signature either way."""

import math
from typing import List, Optional

import torch


def DS4Sci_EvoformerAttention(Q: torch.Tensor, K: torch.Tensor,
                              V: torch.Tensor,
                              biases: Optional[List[Optional[torch.Tensor]]]
                              = None) -> torch.Tensor:
    assert Q.dim() == 5, f"expected [B,N,S,H,D], got {tuple(Q.shape)}"
    B, N, S, H, D = Q.shape
    scale = 1.0 / math.sqrt(D)
    # [B,N,S,H,D] -> [B,N,H,S,D] so the GEMMs batch over (B,N,H)
    q = Q.permute(0, 1, 3, 2, 4)
    k = K.permute(0, 1, 3, 2, 4)
    v = V.permute(0, 1, 3, 2, 4)
    logits = torch.matmul(q.float(), k.float().transpose(-1, -2)) * scale
    if biases:
        for b in biases:
            if b is None:
                continue
            # bias axes are [B, N|1, H|1, S|1, S] — broadcast onto logits
            logits = logits + b.float()
    probs = torch.softmax(logits, dim=-1)
    out = torch.matmul(probs, v.float())
    return out.permute(0, 1, 3, 2, 4).to(Q.dtype)


class EvoformerAttention(torch.nn.Module):
    """Module wrapper matching the reference's functional op."""

    def forward(self, q, k, v, biases=None):
        return DS4Sci_EvoformerAttention(q, k, v, biases)
