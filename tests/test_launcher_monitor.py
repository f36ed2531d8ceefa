"""Launcher + monitor tests (reference contract:
tests/unit/launcher/test_ds_arguments.py, tests/unit/monitor/test_monitor.py).
"""

import os
import subprocess
import sys

import torch


def test_hostfile_parse(tmp_path):
    from deepspeed_amd.launcher.runner import parse_hostfile
    hf = tmp_path / "hostfile"
    hf.write_text("nodeA slots=8\n# comment\nnodeB slots=4  # trailing\n\n")
    hosts = parse_hostfile(str(hf))
    assert hosts == {"nodeA": 8, "nodeB": 4}


def test_launcher_spawns_ranks(tmp_path):
    """End-to-end: the runner must set RANK/LOCAL_RANK/WORLD_SIZE."""
    script = tmp_path / "probe.py"
    script.write_text(
        "import os\n"
        f"open(os.path.join({str(tmp_path)!r}, 'r' + os.environ['RANK']),"
        " 'w').write(' '.join(\n"
        "    [os.environ['RANK'], os.environ['LOCAL_RANK'],"
        " os.environ['WORLD_SIZE']]))\n")
    out = subprocess.run(
        [sys.executable, "-m", "deepspeed_amd.launcher.runner",
         "--num_gpus", "2", "--master_port", "29871", str(script)],
        capture_output=True, text=True, timeout=120,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert out.returncode == 0, out.stdout + out.stderr
    assert (tmp_path / "r0").read_text() == "0 0 2"
    assert (tmp_path / "r1").read_text() == "1 1 2"


def test_launcher_propagates_failure(tmp_path):
    script = tmp_path / "boom.py"
    script.write_text("import os, sys; sys.exit(3 if os.environ['RANK'] == '1' else 0)\n")
    out = subprocess.run(
        [sys.executable, "-m", "deepspeed_amd.launcher.runner",
         "--num_gpus", "2", str(script)],
        capture_output=True, text=True, timeout=120,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert out.returncode == 3


def test_csv_monitor(tmp_path):
    from deepspeed_amd.monitor.monitor import CsvMonitor, MonitorMaster
    from deepspeed_amd.config import MonitorConfig
    m = CsvMonitor(output_path=str(tmp_path), job_name="j")
    m.write_events([("Train/loss", 1.5, 1), ("Train/loss", 1.2, 2),
                    ("Train/lr", 0.1, 1)])
    m.close()
    loss = (tmp_path / "j" / "Train_loss.csv").read_text().strip().splitlines()
    assert loss == ["1,1.5", "2,1.2"]

    mm = MonitorMaster(MonitorConfig(
        enabled=True,
        csv_monitor={"enabled": True, "output_path": str(tmp_path),
                     "job_name": "k"}))
    mm.write_events([("a/b", 3.0, 7)])
    assert (tmp_path / "k" / "a_b.csv").read_text().strip() == "7,3.0"


def test_engine_writes_monitor_events(tmp_path):
    from .common import run_local

    def worker(rank, world, tmp=str(tmp_path)):
        import deepspeed_amd
        from deepspeed_amd.models import GPT2ForCausalLM, gpt2_tiny
        model = GPT2ForCausalLM(gpt2_tiny())
        engine, _, _, _ = deepspeed_amd.initialize(model=model, config={
            "train_micro_batch_size_per_gpu": 1,
            "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}},
            "monitor_config": {"enabled": True,
                               "csv_monitor": {"enabled": True,
                                               "output_path": tmp,
                                               "job_name": "m"}}})
        ids = torch.randint(0, 128, (1, 16))
        for _ in range(2):
            loss = engine(ids, labels=ids)
            engine.backward(loss)
            engine.step()

    run_local(worker)
    files = {f.name for f in (tmp_path / "m").iterdir()}
    assert "Train_loss.csv" in files and "Train_lr.csv" in files
    rows = (tmp_path / "m" / "Train_loss.csv").read_text().strip().splitlines()
    assert len(rows) == 2


def test_top_level_writer_blocks_and_disabled_fallbacks(tmp_path):
    """Reference accepts tensorboard/wandb/comet blocks at ds_config top
    level; unavailable writers (wandb/comet not installed) degrade to a
    warning, never an exception, and CSV still engages as fallback."""
    from deepspeed_amd.config import Config
    from deepspeed_amd.monitor.monitor import MonitorMaster, CsvMonitor
    cfg = Config({
        "train_micro_batch_size_per_gpu": 1,
        "wandb": {"enabled": True, "project": "x"},
        "comet": {"enabled": True, "project": "x"},
        "csv_monitor": {"enabled": True, "output_path": str(tmp_path),
                        "job_name": "j"},
    })
    assert cfg.monitor.wandb["enabled"]
    assert cfg.monitor.comet["enabled"]
    mm = MonitorMaster(cfg.monitor)
    assert any(isinstance(m, CsvMonitor) for m in mm.monitors)
    mm.write_events([("a/b", 1.0, 0)])
    assert (tmp_path / "j" / "a_b.csv").exists()


def test_launcher_elastic_restart(tmp_path):
    """Elastic agent: a worker failure restarts the whole group (fresh
    rendezvous port, DSAMD_RESTART_COUNT bumped); the relaunched attempt
    succeeds and the job exits 0."""
    script = tmp_path / "flaky.py"
    script.write_text(
        "import os, sys\n"
        "attempt = os.environ['DSAMD_RESTART_COUNT']\n"
        f"open(os.path.join({str(tmp_path)!r},"
        " 'a' + attempt + '_r' + os.environ['RANK']), 'w')"
        ".write(os.environ['MASTER_PORT'])\n"
        "sys.exit(5 if (attempt == '0' and os.environ['RANK'] == '1')"
        " else 0)\n")
    out = subprocess.run(
        [sys.executable, "-m", "deepspeed_amd.launcher.runner",
         "--num_gpus", "2", "--max_restarts", "2",
         "--master_port", "29881", str(script)],
        capture_output=True, text=True, timeout=120,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert out.returncode == 0, out.stdout + out.stderr
    # both attempts ran, second on a bumped port
    assert (tmp_path / "a0_r1").read_text() == "29881"
    assert (tmp_path / "a1_r0").read_text() == "29882"
    assert (tmp_path / "a1_r1").exists()


def test_launcher_elastic_restarts_exhausted(tmp_path):
    """A persistently failing worker exhausts max_restarts and the final
    exit code propagates."""
    script = tmp_path / "always_fails.py"
    script.write_text("import os, sys; sys.exit(7)\n")
    out = subprocess.run(
        [sys.executable, "-m", "deepspeed_amd.launcher.runner",
         "--num_gpus", "1", "--max_restarts", "1",
         "--master_port", "29891", str(script)],
        capture_output=True, text=True, timeout=120,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert out.returncode == 7
    assert "elastic restart 1/1" in out.stdout + out.stderr


def test_multinode_runner_commands():
    """Parse-level runner command construction (reference
    tests/unit/launcher: command assembly without a cluster)."""
    import argparse
    from deepspeed_amd.launcher.multinode_runner import (build_runner,
                                                         RUNNERS)
    args = argparse.Namespace(master_port=29500, module=False)
    hosts = {"node1": 8, "node2": 8}
    user = ["train.py", "--lr", "1e-4"]

    slurm = build_runner("slurm", args, hosts)
    cmd = slurm.get_cmd(user)
    assert cmd[0] == "srun" and "--ntasks" in cmd and "16" in cmd
    assert "train.py" in cmd

    ompi = build_runner("openmpi", args, hosts)
    cmd = ompi.get_cmd(user)
    assert cmd[0] == "mpirun" and "-n" in cmd and "16" in cmd
    assert "node1:8" in cmd

    mpich = build_runner("mpich", args, hosts)
    cmd = mpich.get_cmd(user)
    assert cmd[0] == "mpiexec" and "node1:8,node2:8" in cmd

    impi = build_runner("impi", args, hosts)
    cmd = impi.get_cmd(user)
    assert "-ppn" in cmd

    pdsh = build_runner("pdsh", args, hosts)
    cmd = pdsh.get_cmd(user)
    assert cmd[0] == "pdsh" and "node1,node2" in cmd
    assert "node_rank" in " ".join(cmd) or "--node_rank" in cmd[-1]

    assert set(RUNNERS) == {"pdsh", "slurm", "openmpi", "mpich", "impi"}

    try:
        build_runner("bogus", args, hosts)
        assert False
    except ValueError:
        pass


def test_bin_cli_entry_points():
    """bin/ scripts exist, are executable, and --launcher parses."""
    import os
    import stat
    from deepspeed_amd.launcher.runner import parse_args
    root = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "bin")
    for name in ("deepspeed", "ds_report", "ds_io", "ds_nvme_tune",
                 "ds_bench"):
        p = os.path.join(root, name)
        assert os.path.exists(p), p
        assert os.stat(p).st_mode & stat.S_IXUSR
    a = parse_args(["--launcher", "slurm", "--num_gpus", "4", "x.py"])
    assert a.launcher == "slurm" and a.user_script == "x.py"


def test_engine_monitor_csv_and_comms_logger(tmp_path):
    """Engine-integrated observability: monitor csv files appear with
    Train/ events, and the comms logger records collectives (ws2)."""
    from .common import run_distributed
    run_distributed(_monitored_worker, world_size=2,
                    args=(str(tmp_path),))


def _monitored_worker(rank, world, tmp):
    import os
    import deepspeed_amd
    import deepspeed_amd.comm as dcomm

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.fc = torch.nn.Linear(8, 1)

        def forward(self, x, labels=None):
            return torch.nn.functional.mse_loss(self.fc(x).float(),
                                                labels.float())

    eng, _, _, _ = deepspeed_amd.initialize(model=M(), config={
        "train_micro_batch_size_per_gpu": 2,
        "steps_per_print": 1,
        "csv_monitor": {"enabled": True, "output_path": tmp,
                        "job_name": "t"},
        "comms_logger": {"enabled": True},
        "zero_optimization": {"stage": 2, "overlap_comm": False},
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}}})
    for _ in range(2):
        loss = eng(torch.randn(2, 8).to(eng.device),
                   labels=torch.randn(2, 1).to(eng.device))
        eng.backward(loss)
        eng.step()
    cl = dcomm.get_comms_logger()
    assert cl is not None and cl.records, cl.records if cl else None
    if rank == 0:
        files = []
        for root, _, fs in os.walk(tmp):
            files += [os.path.join(root, f) for f in fs if f.endswith(".csv")]
        assert files, f"no csv monitor output under {tmp}"
        assert any("Train_lr" in os.path.basename(f) for f in files)
