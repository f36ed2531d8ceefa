"""Experiment monitoring — event writers behind one fan-out master
(reference: deepspeed/monitor/monitor.py MonitorMaster :30 + tensorboard /
csv / wandb writers).

Events are ``(tag, value, global_step)`` tuples, identical to the reference
contract. Offline-first: CSV always works; TensorBoard/W&B engage only if
their packages are importable.
"""

import csv
import os
from typing import List, Tuple

from ..utils.logging import logger

Event = Tuple[str, float, int]


class Monitor:
    def write_events(self, events: List[Event]):
        raise NotImplementedError


class CsvMonitor(Monitor):
    """One CSV file per tag under ``output_path/job_name``."""

    def __init__(self, output_path="ds_csv", job_name="run"):
        self.dir = os.path.join(output_path, job_name)
        os.makedirs(self.dir, exist_ok=True)
        self._files = {}

    def _writer(self, tag):
        if tag not in self._files:
            safe = tag.replace("/", "_")
            f = open(os.path.join(self.dir, f"{safe}.csv"), "a", newline="")
            self._files[tag] = (f, csv.writer(f))
        return self._files[tag]

    def write_events(self, events: List[Event]):
        for tag, value, step in events:
            f, w = self._writer(tag)
            w.writerow([step, float(value)])
            f.flush()

    def close(self):
        for f, _ in self._files.values():
            f.close()
        self._files.clear()


class TensorBoardMonitor(Monitor):
    def __init__(self, output_path="ds_tb", job_name="run"):
        from torch.utils.tensorboard import SummaryWriter  # may raise
        self.writer = SummaryWriter(log_dir=os.path.join(output_path, job_name))

    def write_events(self, events: List[Event]):
        for tag, value, step in events:
            self.writer.add_scalar(tag, value, step)
        self.writer.flush()


class WandbMonitor(Monitor):
    def __init__(self, project=None, team=None, group=None):
        import wandb  # may raise
        self.wandb = wandb
        wandb.init(project=project, entity=team, group=group)

    def write_events(self, events: List[Event]):
        for tag, value, step in events:
            self.wandb.log({tag: value}, step=step)


class CometMonitor(Monitor):
    """Comet experiment writer (reference monitor/comet.py CometMonitor)."""

    def __init__(self, project=None, workspace=None, api_key=None,
                 experiment_name=None, experiment_key=None, mode=None,
                 online=None, samples_log_interval=100):
        import comet_ml  # may raise
        kwargs = {}
        if mode:
            kwargs["mode"] = mode
        if online is not None:
            kwargs["online"] = online
        self.experiment = comet_ml.start(
            api_key=api_key, project=project, workspace=workspace,
            experiment_key=experiment_key, **kwargs)
        if experiment_name:
            self.experiment.set_name(experiment_name)

    def write_events(self, events: List[Event]):
        for tag, value, step in events:
            self.experiment.log_metric(tag, value, step=step)


class MonitorMaster(Monitor):
    """Fans events out to every enabled writer (reference monitor.py:30)."""

    def __init__(self, config):
        self.monitors: List[Monitor] = []
        tb = dict(getattr(config, "tensorboard", {}) or {})
        if tb.pop("enabled", False):
            try:
                self.monitors.append(TensorBoardMonitor(**tb))
            except Exception as e:  # tensorboard not installed
                logger.warning(f"tensorboard writer disabled: {e}")
        wb = dict(getattr(config, "wandb", {}) or {})
        if wb.pop("enabled", False):
            try:
                self.monitors.append(WandbMonitor(**wb))
            except Exception as e:  # wandb not installed / offline
                logger.warning(f"wandb writer disabled: {e}")
        cm = dict(getattr(config, "comet", {}) or {})
        if cm.pop("enabled", False):
            try:
                self.monitors.append(CometMonitor(**cm))
            except Exception as e:  # comet_ml not installed / offline
                logger.warning(f"comet writer disabled: {e}")
        csv_cfg = dict(getattr(config, "csv_monitor", {}) or {})
        if csv_cfg.pop("enabled", False) or not self.monitors:
            self.monitors.append(CsvMonitor(**csv_cfg))

    def write_events(self, events: List[Event]):
        for m in self.monitors:
            m.write_events(events)
