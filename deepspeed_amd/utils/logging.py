"""Rank-aware logging.

Parity: reference `deepspeed/utils/logging.py:113` (`logger`, `log_dist`).
"""
import logging
import os
import sys

_FMT = "[%(asctime)s] [%(levelname)s] [%(name)s:%(lineno)d] %(message)s"


def _create_logger(name="dsamd", level=logging.INFO):
    lg = logging.getLogger(name)
    lg.setLevel(level)
    lg.propagate = False
    if not lg.handlers:
        h = logging.StreamHandler(stream=sys.stdout)
        h.setFormatter(logging.Formatter(_FMT, datefmt="%Y-%m-%d %H:%M:%S"))
        lg.addHandler(h)
    return lg


logger = _create_logger()

if os.environ.get("DSAMD_LOG_LEVEL"):
    logger.setLevel(os.environ["DSAMD_LOG_LEVEL"].upper())


def _get_rank():
    try:
        import torch.distributed as dist
        if dist.is_available() and dist.is_initialized():
            return dist.get_rank()
    except Exception:
        pass
    return int(os.environ.get("RANK", 0))


def log_dist(message, ranks=None, level=logging.INFO):
    """Log `message` only on the listed ranks (None or [-1] = all ranks)."""
    rank = _get_rank()
    if ranks is None or -1 in ranks or rank in ranks:
        logger.log(level, f"[Rank {rank}] {message}")


def warning_once(message, _seen=set()):
    if message not in _seen:
        _seen.add(message)
        logger.warning(message)
