"""Universal checkpointing — convert ZeRO-sharded optimizer checkpoints to
a world-size-independent per-parameter layout and load them elastically
(reference: deepspeed/checkpoint/ds_to_universal.py main :469).

Flow:
  1. engine.save_checkpoint(dir, tag)  -> per-rank zero_pp_rank_* files,
     each embedding a layout manifest (bucket spans + param names).
  2. ds_to_universal(dir, tag)         -> <dir>/<tag>_universal/
     universal_optim_states.pt holding {param, exp_avg, exp_avg_sq}[name]
     as full fp32 tensors (merged across the saved DP world, offline, no
     model needed).
  3. a NEW job at ANY DP world size:
     engine.load_checkpoint(dir, tag, load_universal=True) -> each rank
     slices its own shard spans out of the universal tensors.
"""

import glob
import os
import re

import torch


def _load_zero_files(ckpt_dir):
    files = glob.glob(os.path.join(ckpt_dir, "zero_pp_rank_*_optim_states.pt"))
    if not files:
        raise FileNotFoundError(f"no zero_pp_rank_* files in {ckpt_dir}")

    def dp_rank(f):
        return int(re.search(r"zero_pp_rank_(\d+)_", os.path.basename(f))
                   .group(1))
    files.sort(key=dp_rank)
    return [torch.load(f, map_location="cpu",
                       weights_only=False)["optimizer_state_dict"]
            for f in files]


def ds_to_universal(load_dir, tag=None, out_dir=None):
    """Merge per-rank ZeRO shards into per-param fp32 universal state.
    Returns the universal checkpoint directory."""
    if tag is None:
        with open(os.path.join(load_dir, "latest")) as f:
            tag = f.read().strip()
    ckpt_dir = os.path.join(load_dir, tag)
    states = _load_zero_files(ckpt_dir)
    layout = states[0].get("layout")
    assert layout is not None, \
        "checkpoint has no layout manifest (saved by an older version?)"

    out = {"param": {}, "exp_avg": {}, "exp_avg_sq": {}, "step": 0}
    # Partition (file, bucket) pairs by the bucket's PARAM IDENTITY.
    # With expert parallelism, the same bucket slot on different EP ranks
    # holds DIFFERENT experts under distinct @epR-decorated names; with
    # pipeline parallelism, different stages have entirely different
    # layouts (even different bucket COUNTS — never index other files by
    # this file's bucket number). Each identity assembles its own full
    # flat from its own members' shards and emits its per-param entries.
    by_identity = {}
    for sd in states:
        for lb in sd["layout"]:
            key = tuple(n for n, *_ in lb["params"])
            by_identity.setdefault(key, []).append((sd, lb))
    for members in by_identity.values():
        _, b0 = members[0]
        pg_world = b0["pg_world"]
        shards = {}
        for sd, lb in members:
            gi = lb["group_idx"]
            mo, ss = lb["master_offset"], lb["shard_size"]
            flat = sd["fp32_flat_groups"][gi]
            base = sd["base_optimizer_state"]["state"].get(gi, {})
            shards[lb["pg_rank"]] = {
                "param": flat[mo:mo + ss],
                "exp_avg": base.get("exp_avg",
                                    torch.zeros(ss))[mo:mo + ss],
                "exp_avg_sq": base.get("exp_avg_sq",
                                       torch.zeros(ss))[mo:mo + ss],
            }
            step = base.get("step", 0)
            out["step"] = int(step.item() if torch.is_tensor(step)
                              else step)
        assert len(shards) == pg_world, \
            f"bucket {b0['params'][0]}: found {len(shards)} shards, " \
            f"expected {pg_world}"
        for kind in ("param", "exp_avg", "exp_avg_sq"):
            full = torch.cat([shards[r][kind] for r in range(pg_world)])
            for name, off, numel, shape in b0["params"]:
                if name is None:
                    continue
                out[kind][name] = full[off:off + numel].view(shape).clone()

    out_dir = out_dir or os.path.join(load_dir, f"{tag}_universal")
    os.makedirs(out_dir, exist_ok=True)
    torch.save(out, os.path.join(out_dir, "universal_optim_states.pt"))
    return out_dir


def load_universal(path_or_dir):
    p = path_or_dir
    if os.path.isdir(p):
        p = os.path.join(p, "universal_optim_states.pt")
    return torch.load(p, map_location="cpu", weights_only=False)


def main(argv=None):
    import argparse
    ap = argparse.ArgumentParser(description="ZeRO -> universal checkpoint")
    ap.add_argument("--input_folder", required=True)
    ap.add_argument("--tag", default=None)
    ap.add_argument("--output_folder", default=None)
    args = ap.parse_args(argv)
    out = ds_to_universal(args.input_folder, args.tag, args.output_folder)
    print(f"universal checkpoint written to {out}")


if __name__ == "__main__":
    main()
