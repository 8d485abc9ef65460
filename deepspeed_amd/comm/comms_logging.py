"""Per-collective size/latency/bandwidth logging.

Parity: reference `deepspeed/utils/comms_logging.py:67` (CommsLogger) and
`deepspeed/comm/comm.py:439` (log_summary). algbw = payload/time; busbw
scales by the collective's wire factor on an N-rank ring.
"""
import math
from collections import defaultdict

from ..utils.logging import log_dist


def _bw_factor(op_name, n):
    """Bus-bandwidth correction factor (NCCL-tests convention)."""
    if n <= 1:
        return 1.0
    if "all_reduce" in op_name:
        return 2 * (n - 1) / n
    if "all_gather" in op_name or "reduce_scatter" in op_name:
        return (n - 1) / n
    if "all_to_all" in op_name:
        return (n - 1) / n
    return 1.0


class CommsLogger:
    def __init__(self):
        self.enabled = False
        self.verbose = False
        self.prof_all = True
        self.debug = False
        self.prof_ops = []
        self.comms_dict = defaultdict(lambda: defaultdict(lambda: [0, [], []]))

    def configure(self, enabled=False, verbose=False, prof_all=True,
                  debug=False, prof_ops=None):
        self.enabled = enabled
        self.verbose = verbose
        self.prof_all = prof_all
        self.debug = debug
        self.prof_ops = prof_ops or []

    def append(self, op_name, size_bytes, latency_s, async_op=False):
        if self.prof_ops and op_name not in self.prof_ops:
            return
        rec = self.comms_dict[op_name][size_bytes]
        rec[0] += 1
        rec[1].append(latency_s)
        try:
            import torch.distributed as dist
            n = dist.get_world_size() if dist.is_initialized() else 1
        except Exception:
            n = 1
        if latency_s > 0:
            algbw = size_bytes / latency_s / 1e9
            rec[2].append(algbw * _bw_factor(op_name, n))
        if self.verbose:
            log_dist(f"comm op: {op_name} | size {size_bytes} B | "
                     f"{latency_s*1000:.3f} ms", ranks=[0])

    def log_all(self, show_straggler=False):
        """Summary table; with show_straggler, also reports per-op
        straggler delay = avg(rank-max latency) - avg(own latency)
        across the world (ref comms_logging get_bw + log_summary
        straggler effect)."""
        if show_straggler:
            import torch
            import torch.distributed as dist
            if dist.is_initialized() and dist.get_world_size() > 1:
                log_dist("straggler effect (max-over-ranks minus own "
                         "mean latency):", ranks=[0])
                for op_name, sizes in sorted(self.comms_dict.items()):
                    own = [sum(l) / max(len(l), 1)
                           for _, (c, l, b) in sorted(sizes.items())]
                    t = torch.tensor(own, dtype=torch.float64)
                    mx = t.clone()
                    dist.all_reduce(mx, op=dist.ReduceOp.MAX)
                    delay = (mx - t).mean().item() * 1000 if len(own) \
                        else 0.0
                    log_dist(f"  {op_name}: {delay:.3f} ms", ranks=[0])
        for op_name, sizes in sorted(self.comms_dict.items()):
            log_dist(f"Op: {op_name}", ranks=[0])
            log_dist(f"{'size(B)':>14} {'count':>8} {'avg lat(ms)':>12} "
                     f"{'busbw(GB/s)':>12}", ranks=[0])
            for size, (count, lats, bws) in sorted(sizes.items()):
                avg_lat = sum(lats) / max(len(lats), 1) * 1000
                avg_bw = sum(bws) / max(len(bws), 1) if bws else float("nan")
                log_dist(f"{size:>14} {count:>8} {avg_lat:>12.3f} "
                         f"{avg_bw if not math.isnan(avg_bw) else 0:>12.2f}",
                         ranks=[0])

    def reset(self):
        self.comms_dict.clear()


_default = CommsLogger()


def get_default_logger():
    return _default
