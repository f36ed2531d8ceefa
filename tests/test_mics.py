"""MiCS hierarchical sharding tests (reference contract:
tests/unit/runtime/zero/test_mics_*.py): shard-in-subgroup + replica
gradient all-reduce must match plain training."""

import torch

from .common import run_distributed


def _mics_worker(rank, world):
    import deepspeed_amd
    from deepspeed_amd.models import LlamaForCausalLM, llama_tiny

    torch.manual_seed(23)
    model = LlamaForCausalLM(llama_tiny())
    config = {
        "train_micro_batch_size_per_gpu": 2,
        "zero_optimization": {"stage": 3, "mics_shard_size": 1,
                              "overlap_comm": False},
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}},
    }
    engine, opt, _, _ = deepspeed_amd.initialize(model=model, config=config)
    assert opt.world_size == 1 and opt.replica_world == world

    # reference: plain single-process training on the same data
    torch.manual_seed(23)
    ref = LlamaForCausalLM(llama_tiny())
    opt_ref = torch.optim.AdamW(ref.parameters(), lr=1e-3)

    torch.manual_seed(77)  # same data on every rank -> DDP == single proc
    for _ in range(3):
        ids = torch.randint(0, 512, (2, 32))
        loss = engine(ids, labels=ids)
        engine.backward(loss)
        engine.step()

        l2 = ref(ids, labels=ids)
        l2.backward()
        opt_ref.step()
        opt_ref.zero_grad()
        assert abs(loss.item() - l2.item()) < 1e-4

    fp32 = opt.get_full_state_dict(dtype=torch.float32)
    if rank == 0:
        for n, p in ref.named_parameters():
            torch.testing.assert_close(fp32[n].float(), p.detach(),
                                       rtol=1e-4, atol=2e-4), n

    # replicas agree
    import torch.distributed as td
    for u in opt.units[:3]:
        peers = [torch.empty_like(u.shard) for _ in range(world)]
        td.all_gather(peers, u.shard)
        assert torch.equal(peers[0], peers[1])


def test_mics_shard1_replica2():
    run_distributed(_mics_worker, world_size=2)
