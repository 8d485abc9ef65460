from .comm import *  # noqa: F401,F403
