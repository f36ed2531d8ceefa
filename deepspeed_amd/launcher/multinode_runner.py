"""Multinode runner backends (reference: deepspeed/launcher/
multinode_runner.py — PDSH / OpenMPI / MVAPICH / SLURM / MPICH / IMPI
runners). Each runner turns (hosts, per-node command) into ONE cluster
launch command; `get_cmd` is separated from execution so parse-level tests
run without a cluster (the reference's own test strategy,
tests/unit/launcher/)."""

import os
import shutil
import subprocess
import sys
from typing import Dict, List


class MultiNodeRunner:
    name = "base"

    def __init__(self, args, hosts: Dict[str, int]):
        self.args = args
        self.hosts = hosts
        self.exports = dict(self._env_exports())

    def _env_exports(self):
        # reference .deepspeed_env passthrough (launcher/runner.py:38)
        keep = ("NCCL_", "RCCL_", "HSA_", "HIP_", "ROCR_", "PYTHONPATH",
                "DS_AMD_", "MASTER_", "GLOO_")
        for k, v in os.environ.items():
            if any(k.startswith(p) for p in keep):
                yield k, v
        env_file = os.path.join(os.path.expanduser("~"), ".deepspeed_env")
        if os.path.exists(env_file):
            with open(env_file) as f:
                for line in f:
                    if "=" in line:
                        k, v = line.strip().split("=", 1)
                        yield k, v

    def backend_exists(self) -> bool:
        raise NotImplementedError

    def get_cmd(self, user_cmd: List[str]) -> List[str]:
        raise NotImplementedError

    def run(self, user_cmd: List[str]) -> int:
        return subprocess.call(self.get_cmd(user_cmd))

    @property
    def total_slots(self):
        return sum(self.hosts.values())


class PDSHRunner(MultiNodeRunner):
    """pdsh fan-out: one per-node launcher invocation per host."""

    name = "pdsh"

    def backend_exists(self):
        return shutil.which("pdsh") is not None

    def get_cmd(self, user_cmd):
        env = "".join(f"export {k}={v}; " for k, v in self.exports.items())
        hostlist = ",".join(self.hosts.keys())
        first = next(iter(self.hosts))
        node_cmds = []
        for node_rank, (host, slots) in enumerate(self.hosts.items()):
            inner = [sys.executable, "-m", "deepspeed_amd.launcher.runner",
                     "--num_gpus", str(slots),
                     "--num_nodes", str(len(self.hosts)),
                     "--node_rank", str(node_rank),
                     "--master_addr", first,
                     "--master_port", str(self.args.master_port)] + user_cmd
            node_cmds.append((host, " ".join(inner)))
        # pdsh runs ONE command string; node_rank is derived per host via
        # %n is not portable -> use a case switch on hostname
        case = "case $(hostname) in "
        for host, cmd in node_cmds:
            case += f"{host}) {env}cd {os.getcwd()}; {cmd};; "
        case += "esac"
        return ["pdsh", "-R", "ssh", "-w", hostlist, case]


class SlurmRunner(MultiNodeRunner):
    name = "slurm"

    def backend_exists(self):
        return shutil.which("srun") is not None

    def get_cmd(self, user_cmd):
        cmd = ["srun", "--ntasks", str(self.total_slots),
               "--nodes", str(len(self.hosts)),
               "--ntasks-per-node", str(next(iter(self.hosts.values())))]
        for k, v in self.exports.items():
            cmd += [f"--export=ALL,{k}={v}"]
        return cmd + [sys.executable, "-u"] + user_cmd


class OpenMPIRunner(MultiNodeRunner):
    name = "openmpi"

    def backend_exists(self):
        return shutil.which("mpirun") is not None

    def get_cmd(self, user_cmd):
        hostfile_args = []
        for host, slots in self.hosts.items():
            hostfile_args += ["--host", f"{host}:{slots}"]
        cmd = ["mpirun", "-n", str(self.total_slots)] + hostfile_args + \
            ["--mca", "btl", "^openib", "--mca", "btl_tcp_if_include",
             "eth0"]
        for k, v in self.exports.items():
            cmd += ["-x", f"{k}={v}"]
        return cmd + [sys.executable, "-u"] + user_cmd


class MPICHRunner(MultiNodeRunner):
    name = "mpich"

    def backend_exists(self):
        return shutil.which("mpiexec") is not None

    def get_cmd(self, user_cmd):
        hosts = ",".join(f"{h}:{s}" for h, s in self.hosts.items())
        cmd = ["mpiexec", "-n", str(self.total_slots), "-hosts", hosts]
        for k, v in self.exports.items():
            cmd += ["-genv", k, str(v)]
        return cmd + [sys.executable, "-u"] + user_cmd


class IMPIRunner(MPICHRunner):
    name = "impi"

    def get_cmd(self, user_cmd):
        cmd = super().get_cmd(user_cmd)
        # Intel MPI: per-rank pinning off, launcher handles local ranks
        return cmd[:1] + ["-ppn", str(next(iter(self.hosts.values())))] + \
            cmd[1:]


RUNNERS = {r.name: r for r in
           (PDSHRunner, SlurmRunner, OpenMPIRunner, MPICHRunner, IMPIRunner)}


def build_runner(launcher: str, args, hosts):
    if launcher not in RUNNERS:
        raise ValueError(f"unknown launcher '{launcher}' "
                         f"(have {sorted(RUNNERS)})")
    return RUNNERS[launcher](args, hosts)
