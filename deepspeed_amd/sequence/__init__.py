from .layer import DistributedAttention, single_all_to_all  # noqa: F401
