"""Launcher — one process per GPU (reference: deepspeed/launcher/runner.py
main :419 + launch.py :133).

Usage (CLI installed as ``dsamd`` / ``python -m deepspeed_amd.launcher.runner``):

    dsamd --num_gpus 8 train.py --deepspeed_config ds.json

Single MI355X node is the primary target: spawn ``num_gpus`` child
processes with RANK/LOCAL_RANK/WORLD_SIZE/MASTER_ADDR/MASTER_PORT set, one
rank per GPU over RCCL. Multi-node runs use ``--hostfile`` with ssh
(pdsh-style fan-out, reference multinode_runner.py) — each remote node
re-invokes this module with a node rank offset.
"""

import argparse
import os
import signal
import subprocess
import sys
import time
from collections import OrderedDict

from ..utils.logging import logger


def parse_hostfile(path):
    """Lines of ``hostname slots=N`` (reference runner.py:_parse_hostfile)."""
    hosts = OrderedDict()
    with open(path) as f:
        for line in f:
            line = line.split("#")[0].strip()
            if not line:
                continue
            parts = line.split()
            name = parts[0]
            slots = 1
            for p in parts[1:]:
                if p.startswith("slots="):
                    slots = int(p.split("=")[1])
            hosts[name] = slots
    return hosts


def parse_args(args=None):
    p = argparse.ArgumentParser(
        description="deepspeed_amd launcher (one process per MI355X GPU)")
    p.add_argument("--num_gpus", "--num-gpus", type=int, default=-1,
                   help="GPUs on this node (-1 = all visible)")
    p.add_argument("--num_nodes", type=int, default=1)
    p.add_argument("--launcher", type=str, default="ssh",
                   choices=["ssh", "pdsh", "slurm", "openmpi", "mpich",
                            "impi"],
                   help="multinode backend (reference multinode_runner.py)")
    p.add_argument("--node_rank", type=int, default=0)
    p.add_argument("--hostfile", type=str, default=None)
    p.add_argument("--master_addr", type=str, default="127.0.0.1")
    p.add_argument("--master_port", type=int, default=29500)
    p.add_argument("--include", type=str, default="",
                   help="e.g. 'localhost:0,2,4' to pin specific GPUs")
    p.add_argument("--module", action="store_true",
                   help="run user_script as a python module (-m)")
    p.add_argument("--no_python", action="store_true")
    p.add_argument("--max_restarts", type=int, default=0,
                   help="elastic agent: on worker failure, restart the "
                        "whole local group up to N times (workers resume "
                        "from their latest checkpoint; DSAMD_RESTART_COUNT "
                        "in the env tells them which attempt this is)")
    p.add_argument("user_script", type=str)
    p.add_argument("user_args", nargs=argparse.REMAINDER)
    return p.parse_args(args)


def device_count():
    try:
        import torch
        return max(torch.cuda.device_count(), 1)
    except Exception:
        return 1


def launch_local(args, local_gpu_ids=None, restart_count=0):
    n_local = args.num_gpus if args.num_gpus > 0 else device_count()
    if local_gpu_ids:
        n_local = len(local_gpu_ids)
    world_size = n_local * args.num_nodes

    procs = []
    for local_rank in range(n_local):
        env = dict(os.environ)
        env["RANK"] = str(args.node_rank * n_local + local_rank)
        env["LOCAL_RANK"] = str(local_rank)
        env["WORLD_SIZE"] = str(world_size)
        env["LOCAL_WORLD_SIZE"] = str(n_local)
        env["MASTER_ADDR"] = args.master_addr
        env["MASTER_PORT"] = str(args.master_port)
        env["DSAMD_RESTART_COUNT"] = str(restart_count)
        env["TORCHELASTIC_RESTART_COUNT"] = str(restart_count)
        if local_gpu_ids:
            env["HIP_VISIBLE_DEVICES"] = str(local_gpu_ids[local_rank])
            env["LOCAL_RANK"] = "0"
        cmd = []
        if not args.no_python:
            cmd += [sys.executable, "-u"]
            if args.module:
                cmd += ["-m"]
        cmd.append(args.user_script)
        cmd += args.user_args
        procs.append(subprocess.Popen(cmd, env=env))
        logger.info(f"launched rank {env['RANK']} (local {local_rank}): "
                    f"{' '.join(cmd)}")

    def _terminate(signum, frame):
        for p in procs:
            if p.poll() is None:
                p.terminate()
        sys.exit(1)

    signal.signal(signal.SIGINT, _terminate)
    signal.signal(signal.SIGTERM, _terminate)

    # fail fast: if any rank dies, kill the rest (reference launch.py sigkill
    # handler semantics)
    exit_code = 0
    alive = set(range(len(procs)))
    while alive:
        time.sleep(0.5)
        for i in list(alive):
            rc = procs[i].poll()
            if rc is None:
                continue
            alive.discard(i)
            if rc != 0:
                exit_code = rc
                logger.error(f"rank process {i} exited with {rc}; "
                             f"terminating remaining ranks")
                for j in alive:
                    if procs[j].poll() is None:
                        procs[j].terminate()
                alive.clear()
                break
    return exit_code


def launch_multinode(args, hosts):
    """ssh fan-out: re-invoke this runner on every host with its node_rank."""
    first = next(iter(hosts))
    procs = []
    for node_rank, (host, slots) in enumerate(hosts.items()):
        inner = [sys.executable, "-m", "deepspeed_amd.launcher.runner",
                 "--num_gpus", str(slots),
                 "--num_nodes", str(len(hosts)),
                 "--node_rank", str(node_rank),
                 "--master_addr", args.master_addr if
                 args.master_addr != "127.0.0.1" else first,
                 "--master_port", str(args.master_port)]
        if args.module:
            inner.append("--module")
        inner.append(args.user_script)
        inner += args.user_args
        if host in ("localhost", "127.0.0.1"):
            procs.append(subprocess.Popen(inner))
        else:
            cwd = os.getcwd()
            remote = f"cd {cwd} && " + " ".join(inner)
            procs.append(subprocess.Popen(["ssh", host, remote]))
    rc = 0
    for p in procs:
        rc = p.wait() or rc
    return rc


def main(argv=None):
    args = parse_args(argv)
    if args.hostfile:
        hosts = parse_hostfile(args.hostfile)
        if len(hosts) > 1:
            if args.launcher != "ssh":
                from .multinode_runner import build_runner
                runner = build_runner(args.launcher, args, hosts)
                if not runner.backend_exists():
                    raise SystemExit(
                        f"launcher backend '{args.launcher}' not found on "
                        "PATH")
                user_cmd = ([args.user_script] if not args.module else
                            ["-m", args.user_script]) + args.user_args
                return sys.exit(runner.run(user_cmd))
            return sys.exit(launch_multinode(args, hosts))
        args.num_gpus = args.num_gpus if args.num_gpus > 0 \
            else next(iter(hosts.values()))
    local_ids = None
    if args.include:
        # 'host:0,2' -> [0, 2]
        part = args.include.split("@")[-1]
        if ":" in part:
            local_ids = [int(x) for x in part.split(":")[1].split(",")]
    # elastic agent loop (reference: torch-elastic restart-all semantics +
    # deepspeed elastic training): a failed worker kills the group; the
    # whole group relaunches on a fresh rendezvous port and the application
    # resumes from its latest checkpoint.
    attempt = 0
    while True:
        rc = launch_local(args, local_ids, restart_count=attempt)
        if rc == 0 or attempt >= args.max_restarts:
            sys.exit(rc)
        attempt += 1
        args.master_port += 1   # stale TCPStore may linger on the old port
        logger.warning(f"worker group failed (rc={rc}); elastic restart "
                       f"{attempt}/{args.max_restarts} on port "
                       f"{args.master_port}")


if __name__ == "__main__":
    main()
