"""Alias of the process-group factory at the reference's import path
(deepspeed.utils.groups). The implementation lives in comm/groups.py."""
from ..comm.groups import *  # noqa: F401,F403
from ..comm import groups as _g

def __getattr__(name):  # keep module-level state (lazy globals) live
    return getattr(_g, name)
