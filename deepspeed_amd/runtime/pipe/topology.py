"""Pipeline process topology: [pipe x data] grid.

Parity: reference `runtime/pipe/topology.py:12` (ProcessTopology),
`:251` (PipelineParallelGrid). rank = pipe * dp_world + data, so DP groups
are adjacent ranks (intra-node xGMI collectives) and p2p crosses groups.
"""
from ... import comm as dist


class PipelineParallelGrid:
    def __init__(self, num_stages, world_size=None):
        world_size = world_size or dist.get_world_size()
        assert world_size % num_stages == 0, \
            f"world {world_size} % stages {num_stages} != 0"
        self.pipe_parallel_size = num_stages
        self.data_parallel_size = world_size // num_stages
        self.global_rank = dist.get_rank()
        dp = self.data_parallel_size

        self.stage_id = self.global_rank // dp
        self.data_parallel_id = self.global_rank % dp

        self._dp_group = None
        self._pp_group = None
        # DP groups: ranks within a stage
        for s in range(num_stages):
            ranks = list(range(s * dp, (s + 1) * dp))
            grp = dist.new_group(ranks)
            if self.global_rank in ranks:
                self._dp_group = grp
                self._dp_ranks = ranks
        # PP groups: same data index across stages
        for d in range(dp):
            ranks = list(range(d, world_size, dp))
            grp = dist.new_group(ranks)
            if self.global_rank in ranks:
                self._pp_group = grp
                self._pp_ranks = ranks

    def get_stage_id(self):
        return self.stage_id

    def get_data_parallel_group(self):
        return self._dp_group

    def get_data_parallel_rank(self):
        return self.data_parallel_id

    def get_data_parallel_world_size(self):
        return self.data_parallel_size

    def get_pipe_parallel_group(self):
        return self._pp_group

    def get_pipe_parallel_rank(self):
        return self.stage_id

    def get_pipe_parallel_world_size(self):
        return self.pipe_parallel_size

    def stage_to_global(self, stage_id):
        """Global rank of `stage_id` with this rank's data index."""
        return stage_id * self.data_parallel_size + self.data_parallel_id

    def is_first_stage(self):
        return self.stage_id == 0

    def is_last_stage(self):
        return self.stage_id == self.pipe_parallel_size - 1

    # model-parallel API expected by the engine (no TP inside PP yet)
    def get_model_parallel_group(self):
        return None

    def get_model_parallel_rank(self):
        return 0

    def get_model_parallel_world_size(self):
        return 1


class ProcessTopology:
    """Generic cartesian process topology (ref runtime/pipe/topology.py:12).

    Maps ranks <-> named-axis coordinates, row-major in axis order:
    ProcessTopology(axes=['pipe','data'], dims=[2,4]) puts data innermost.
    """

    def __init__(self, axes, dims):
        assert len(axes) == len(dims)
        self.axes = list(axes)
        self.dims = list(dims)

    def world_size(self):
        n = 1
        for d in self.dims:
            n *= d
        return n

    def get_dim(self, axis):
        return self.dims[self.axes.index(axis)]

    def get_coord(self, rank):
        coords = {}
        for axis, dim in zip(reversed(self.axes), reversed(self.dims)):
            coords[axis] = rank % dim
            rank //= dim
        from collections import namedtuple
        Coord = namedtuple("Coord", self.axes)
        return Coord(**coords)

    def get_rank(self, **coords):
        rank = 0
        for axis, dim in zip(self.axes, self.dims):
            c = coords[axis]
            assert 0 <= c < dim, f"{axis}={c} out of range {dim}"
            rank = rank * dim + c
        return rank

    def get_axis_list(self, axis, idx):
        """All ranks whose `axis` coordinate equals idx."""
        return [r for r in range(self.world_size())
                if getattr(self.get_coord(r), axis) == idx]

    def get_axis_comm_lists(self, axis):
        """Rank lists that differ only along `axis` (comm groups)."""
        lists = {}
        for r in range(self.world_size()):
            c = self.get_coord(r)
            key = tuple(v for a, v in zip(self.axes, c) if a != axis)
            lists.setdefault(key, []).append(r)
        return list(lists.values())

    def filter_match(self, **filters):
        out = []
        for r in range(self.world_size()):
            c = self.get_coord(r)
            if all(getattr(c, a) == v for a, v in filters.items()):
                out.append(r)
        return out


class PipeDataParallelTopology(ProcessTopology):
    def __init__(self, num_pp, num_dp):
        super().__init__(axes=["pipe", "data"], dims=[num_pp, num_dp])


class PipeModelDataParallelTopology(ProcessTopology):
    def __init__(self, num_pp, num_mp, num_dp):
        super().__init__(axes=["pipe", "data", "model"],
                         dims=[num_pp, num_dp, num_mp])
