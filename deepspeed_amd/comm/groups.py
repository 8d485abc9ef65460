"""Process-group factory: DP / SP / EP / expert-DP / TP / PP groups.

Parity: reference `deepspeed/utils/groups.py:61-916`
(`_create_expert_and_data_parallel:304`, SP getters @799-854).

Topology convention on an 8×MI355X node (fully-connected xGMI): rank order
is [node-major]; SP/TP/EP groups are built from *adjacent* ranks so their
all-to-alls stay intra-node (all 7 links of each GPU engaged), and DP groups
stride across.
"""
import torch.distributed as torch_dist

from . import comm as dist

_WORLD_GROUP = None
_DATA_PARALLEL_GROUP = None
_SEQUENCE_PARALLEL_GROUP = None
_SEQUENCE_DATA_PARALLEL_GROUP = None
_EXPERT_PARALLEL_GROUP = {}     # name -> group
_EXPERT_DATA_PARALLEL_GROUP = {}
_MODEL_PARALLEL_GROUP = None
_TENSOR_PARALLEL_GROUP = None
_mpu = None


def reset_groups():
    global _WORLD_GROUP, _DATA_PARALLEL_GROUP, _SEQUENCE_PARALLEL_GROUP
    global _SEQUENCE_DATA_PARALLEL_GROUP, _MODEL_PARALLEL_GROUP
    global _TENSOR_PARALLEL_GROUP, _mpu
    _WORLD_GROUP = None
    _DATA_PARALLEL_GROUP = None
    _SEQUENCE_PARALLEL_GROUP = None
    _SEQUENCE_DATA_PARALLEL_GROUP = None
    _EXPERT_PARALLEL_GROUP.clear()
    _EXPERT_DATA_PARALLEL_GROUP.clear()
    _MODEL_PARALLEL_GROUP = None
    _TENSOR_PARALLEL_GROUP = None
    _mpu = None


def set_mpu(mpu):
    """Install an external model-parallel unit (Megatron-style)."""
    global _mpu
    _mpu = mpu


def _ensure_dist():
    assert dist.is_initialized(), "dist not initialized"


def _get_world_group():
    global _WORLD_GROUP
    _ensure_dist()
    if _WORLD_GROUP is None:
        _WORLD_GROUP = torch_dist.group.WORLD
    return _WORLD_GROUP


# -- data parallel -----------------------------------------------------------

def get_data_parallel_group():
    if _mpu is not None and hasattr(_mpu, "get_data_parallel_group"):
        return _mpu.get_data_parallel_group()
    if _DATA_PARALLEL_GROUP is not None:
        return _DATA_PARALLEL_GROUP
    return _get_world_group()


def get_data_parallel_world_size():
    return dist.get_world_size(get_data_parallel_group())


def get_data_parallel_rank():
    return dist.get_rank(get_data_parallel_group())


# -- sequence parallel (Ulysses) --------------------------------------------

def initialize_sequence_parallel(sp_size):
    """Partition WORLD into [DP × SP] with SP over adjacent (intra-node) ranks."""
    global _SEQUENCE_PARALLEL_GROUP, _DATA_PARALLEL_GROUP
    _ensure_dist()
    world = dist.get_world_size()
    assert world % sp_size == 0, f"world {world} % sp {sp_size} != 0"
    rank = dist.get_rank()
    # SP groups: adjacent ranks (one node => intra-node a2a over xGMI)
    for start in range(0, world, sp_size):
        ranks = list(range(start, start + sp_size))
        grp = dist.new_group(ranks)
        if rank in ranks:
            _SEQUENCE_PARALLEL_GROUP = grp
    # DP groups stride across SP
    for offset in range(sp_size):
        ranks = list(range(offset, world, sp_size))
        grp = dist.new_group(ranks)
        if rank in ranks:
            _DATA_PARALLEL_GROUP = grp
    return _SEQUENCE_PARALLEL_GROUP


def get_sequence_parallel_group():
    assert _SEQUENCE_PARALLEL_GROUP is not None, "SP not initialized"
    return _SEQUENCE_PARALLEL_GROUP


def sequence_parallel_is_initialized():
    return _SEQUENCE_PARALLEL_GROUP is not None


def get_sequence_parallel_world_size():
    return dist.get_world_size(get_sequence_parallel_group())


def get_sequence_parallel_rank():
    return dist.get_rank(get_sequence_parallel_group())


def get_sequence_data_parallel_group():
    """ZeRO partitions across DP×SP (reference engine.py:2410,2497)."""
    if _SEQUENCE_PARALLEL_GROUP is None:
        return get_data_parallel_group()
    return _get_world_group()


# -- expert parallel (MoE) ---------------------------------------------------

def _get_expert_parallel_ranks(world_size, ep_size):
    """EP groups over adjacent ranks; expert-DP strides.

    world=8, ep=4 -> EP groups [0-3],[4-7]; expert-DP [0,4],[1,5],...
    """
    ep_groups = [list(range(i, i + ep_size)) for i in range(0, world_size, ep_size)]
    edp_groups = [list(range(off, world_size, ep_size)) for off in range(ep_size)]
    return ep_groups, edp_groups


def create_expert_and_data_parallel(ep_size):
    _ensure_dist()
    world = dist.get_world_size()
    assert world % ep_size == 0, f"world {world} % ep {ep_size} != 0"
    name = f"ep_size_{ep_size}"
    if name in _EXPERT_PARALLEL_GROUP:
        return
    rank = dist.get_rank()
    ep_groups, edp_groups = _get_expert_parallel_ranks(world, ep_size)
    for ranks in ep_groups:
        grp = dist.new_group(ranks)
        if rank in ranks:
            _EXPERT_PARALLEL_GROUP[name] = grp
    for ranks in edp_groups:
        grp = dist.new_group(ranks)
        if rank in ranks:
            _EXPERT_DATA_PARALLEL_GROUP[name] = grp


def get_expert_parallel_group(name):
    return _EXPERT_PARALLEL_GROUP[name]


def get_expert_data_parallel_group(name):
    return _EXPERT_DATA_PARALLEL_GROUP[name]


def get_expert_parallel_group_dict():
    return _EXPERT_PARALLEL_GROUP


def get_expert_parallel_world_size(name):
    return dist.get_world_size(get_expert_parallel_group(name))


def get_expert_parallel_rank(name):
    return dist.get_rank(get_expert_parallel_group(name))


def get_expert_data_parallel_rank(name):
    return dist.get_rank(get_expert_data_parallel_group(name))


# -- model/tensor parallel ---------------------------------------------------

def get_model_parallel_group():
    if _mpu is not None and hasattr(_mpu, "get_model_parallel_group"):
        return _mpu.get_model_parallel_group()
    return _MODEL_PARALLEL_GROUP


def get_model_parallel_world_size():
    g = get_model_parallel_group()
    return dist.get_world_size(g) if g is not None else 1


def get_model_parallel_rank():
    g = get_model_parallel_group()
    return dist.get_rank(g) if g is not None else 0


def initialize_tensor_parallel(tp_size):
    global _TENSOR_PARALLEL_GROUP, _MODEL_PARALLEL_GROUP, _DATA_PARALLEL_GROUP
    _ensure_dist()
    world = dist.get_world_size()
    assert world % tp_size == 0
    rank = dist.get_rank()
    for start in range(0, world, tp_size):
        ranks = list(range(start, start + tp_size))
        grp = dist.new_group(ranks)
        if rank in ranks:
            _TENSOR_PARALLEL_GROUP = grp
            _MODEL_PARALLEL_GROUP = grp
    for offset in range(tp_size):
        ranks = list(range(offset, world, tp_size))
        grp = dist.new_group(ranks)
        if rank in ranks:
            _DATA_PARALLEL_GROUP = grp
    return _TENSOR_PARALLEL_GROUP


def get_tensor_parallel_group():
    if _TENSOR_PARALLEL_GROUP is None and _mpu is not None:
        # Megatron-style mpu passed to initialize(): its model-parallel
        # group IS the tensor-parallel group
        for name in ("get_tensor_model_parallel_group",
                     "get_model_parallel_group"):
            if hasattr(_mpu, name):
                return getattr(_mpu, name)()
    return _TENSOR_PARALLEL_GROUP


def get_tensor_parallel_world_size():
    g = get_tensor_parallel_group()
    return dist.get_world_size(g) if g is not None else 1


def get_tensor_parallel_rank():
    g = get_tensor_parallel_group()
    return dist.get_rank(g) if g is not None else 0


# -- MiCS: shard groups (adjacent, intra-node) x replica groups (strided) ---

def initialize_mics(shard_size):
    """Returns (shard_group, replica_group) for MiCS-style ZeRO-3:
    params shard over `shard_size` adjacent ranks (intra-node xGMI
    all-gathers); gradients additionally average across replica groups."""
    _ensure_dist()
    world = dist.get_world_size()
    assert world % shard_size == 0, f"world {world} % shard {shard_size}"
    rank = dist.get_rank()
    shard_group = None
    replica_group = None
    for start in range(0, world, shard_size):
        ranks = list(range(start, start + shard_size))
        grp = dist.new_group(ranks)
        if rank in ranks:
            shard_group = grp
    for offset in range(shard_size):
        ranks = list(range(offset, world, shard_size))
        grp = dist.new_group(ranks)
        if rank in ranks:
            replica_group = grp
    return shard_group, replica_group
