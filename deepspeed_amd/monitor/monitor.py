"""Metric sinks: TensorBoard / CSV, rank-0 gated.

Parity: reference `deepspeed/monitor/monitor.py:13` (Monitor ABC),
`:30` (MonitorMaster), `csv_monitor.py:12`, `tensorboard.py:13`.
"""
import os

from .. import comm as dist


class Monitor:
    def __init__(self, config):
        self.config = config

    def write_events(self, event_list):
        raise NotImplementedError


class CSVMonitor(Monitor):
    def __init__(self, config):
        super().__init__(config)
        self.enabled = config.enabled and dist.get_rank() == 0
        self._files = {}
        if self.enabled:
            self.out = os.path.join(config.output_path or "csv_monitor",
                                    config.job_name)
            os.makedirs(self.out, exist_ok=True)

    def write_events(self, event_list):
        if not self.enabled:
            return
        for name, value, step in event_list:
            fname = os.path.join(self.out, name.replace("/", "_") + ".csv")
            first = not os.path.exists(fname)
            with open(fname, "a") as f:
                if first:
                    f.write("step,value\n")
                f.write(f"{step},{value}\n")


class TensorBoardMonitor(Monitor):
    def __init__(self, config):
        super().__init__(config)
        self.enabled = config.enabled and dist.get_rank() == 0
        self.writer = None
        if self.enabled:
            try:
                from torch.utils.tensorboard import SummaryWriter
                self.writer = SummaryWriter(
                    log_dir=os.path.join(config.output_path or "tb_logs",
                                         config.job_name))
            except ImportError:
                self.enabled = False

    def write_events(self, event_list):
        if not self.enabled or self.writer is None:
            return
        for name, value, step in event_list:
            self.writer.add_scalar(name, value, step)
        self.writer.flush()


class CometMonitor(Monitor):
    """Comet sink (ref deepspeed/monitor/comet.py:23); no-ops when
    comet_ml is not installed."""

    def __init__(self, config):
        super().__init__(config)
        self.enabled = getattr(config, "enabled", False) and \
            dist.get_rank() == 0
        self._exp = None
        if self.enabled:
            try:
                import comet_ml
                self._exp = comet_ml.Experiment(
                    project_name=getattr(config, "project", "dsamd"))
            except ImportError:
                self.enabled = False

    def write_events(self, event_list):
        if not self.enabled or self._exp is None:
            return
        for name, value, step in event_list:
            self._exp.log_metric(name, value, step=step)


class MonitorMaster(Monitor):
    def __init__(self, ds_config):
        self.monitors = []
        if getattr(ds_config, "csv_monitor", None) and ds_config.csv_monitor.enabled:
            self.monitors.append(CSVMonitor(ds_config.csv_monitor))
        if getattr(ds_config, "tensorboard", None) and ds_config.tensorboard.enabled:
            self.monitors.append(TensorBoardMonitor(ds_config.tensorboard))
        wb = getattr(ds_config, "wandb", None)
        if wb and (wb.get("enabled") if isinstance(wb, dict)
                   else getattr(wb, "enabled", False)):
            from types import SimpleNamespace
            cfg = SimpleNamespace(**wb) if isinstance(wb, dict) else wb
            self.monitors.append(WandbMonitor(cfg))
        cm = getattr(ds_config, "comet", None)
        if cm and (cm.get("enabled") if isinstance(cm, dict)
                   else getattr(cm, "enabled", False)):
            from types import SimpleNamespace
            cfg = SimpleNamespace(**cm) if isinstance(cm, dict) else cm
            self.monitors.append(CometMonitor(cfg))
        self.enabled = len(self.monitors) > 0

    def write_events(self, event_list):
        for m in self.monitors:
            m.write_events(event_list)


class WandbMonitor(Monitor):
    """Weights & Biases sink (ref deepspeed/monitor/wandb.py:12)."""

    def __init__(self, config):
        super().__init__(config)
        self.enabled = getattr(config, "enabled", False) and \
            dist.get_rank() == 0
        self._wandb = None
        if self.enabled:
            try:
                import wandb
                self._wandb = wandb
                wandb.init(project=getattr(config, "project", "dsamd"),
                           group=getattr(config, "group", None),
                           name=getattr(config, "job_name", None))
            except ImportError:
                self.enabled = False

    def write_events(self, event_list):
        if not self.enabled or self._wandb is None:
            return
        for name, value, step in event_list:
            self._wandb.log({name: value}, step=step)
