"""PipelineModule: layer-list model partitioned across stages.

Parity: reference `runtime/pipe/module.py:86` (PipelineModule),
`:30` (LayerSpec), `:396` (_partition_layers uniform/parameters).
"""
import re

import torch

from ... import comm as dist
from ...utils.logging import log_dist
from ..utils import partition_balanced, partition_uniform
from .topology import PipelineParallelGrid


class LayerSpec:
    """Deferred layer construction: built only on the owning stage."""

    def __init__(self, typename, *module_args, **module_kwargs):
        self.typename = typename
        self.module_args = module_args
        self.module_kwargs = module_kwargs
        if not issubclass(typename, torch.nn.Module):
            raise RuntimeError("LayerSpec requires an nn.Module subclass")

    def build(self):
        return self.typename(*self.module_args, **self.module_kwargs)

    def param_count(self):
        """Estimate without building (builds once on meta if needed)."""
        with torch.device("meta"):
            try:
                m = self.build()
                return sum(p.numel() for p in m.parameters())
            except Exception:
                return 0


class TiedLayerSpec(LayerSpec):
    def __init__(self, key, typename, *module_args, forward_fn=None,
                 tied_weight_attr="weight", **module_kwargs):
        super().__init__(typename, *module_args, **module_kwargs)
        self.key = key
        self.forward_fn = forward_fn
        self.tied_weight_attr = tied_weight_attr


class PipelineModule(torch.nn.Module):
    def __init__(self, layers, num_stages=None, topology=None,
                 loss_fn=None, partition_method="parameters",
                 activation_checkpoint_interval=0, seed_layers=False):
        super().__init__()
        if num_stages is None and topology is None:
            raise RuntimeError("must provide num_stages or topology")
        self.loss_fn = loss_fn
        self.activation_checkpoint_interval = activation_checkpoint_interval
        self._layer_specs = list(layers)

        if not dist.is_initialized():
            dist.init_distributed()
        self._grid = topology or PipelineParallelGrid(num_stages)
        self.num_stages = self._grid.pipe_parallel_size
        self.stage_id = self._grid.get_stage_id()

        self._partition_layers(partition_method)
        self._build()

    # -- partitioning -------------------------------------------------------
    def _layer_weights(self, method):
        specs = self._layer_specs
        if method == "uniform":
            return None
        weights = []
        for s in specs:
            if isinstance(s, LayerSpec):
                weights.append(max(s.param_count(), 1))
            elif isinstance(s, torch.nn.Module):
                weights.append(max(sum(p.numel() for p in s.parameters()), 1))
            else:
                weights.append(1)
        return weights

    def _partition_layers(self, method):
        n = len(self._layer_specs)
        method = method.lower()
        if method in ("uniform",):
            self.parts = partition_uniform(n, self.num_stages)
        elif method in ("parameters", "best"):
            self.parts = partition_balanced(self._layer_weights(method),
                                            self.num_stages)
        elif method.startswith("type:"):
            regex = method.split(":", 1)[1]
            weights = [1 if re.search(regex, str(getattr(
                s, "typename", type(s)).__name__), re.IGNORECASE) else 0
                for s in self._layer_specs]
            self.parts = partition_balanced(weights, self.num_stages)
        else:
            raise ValueError(f"unknown partition method {method}")
        log_dist(f"pipeline partition bounds: {self.parts}", ranks=[0])

    def _build(self):
        start, stop = self.parts[self.stage_id], self.parts[self.stage_id + 1]
        self._local_start = start
        self._local_stop = stop
        self.forward_funcs = []
        mods = torch.nn.ModuleDict()
        tied_local = {}   # key -> module instance (same-stage weight share)
        for i, spec in enumerate(self._layer_specs[start:stop]):
            idx = start + i
            if isinstance(spec, TiedLayerSpec):
                if spec.key in tied_local:
                    mod = tied_local[spec.key]  # share the instance
                else:
                    mod = spec.build()
                    tied_local[spec.key] = mod
                    mods[str(idx)] = mod
                if spec.forward_fn is not None:
                    fn = spec.forward_fn
                    self.forward_funcs.append(
                        (lambda m, f: (lambda x: f(m, x)))(mod, fn))
                else:
                    self.forward_funcs.append(mod)
            elif isinstance(spec, LayerSpec):
                mod = spec.build()
                mods[str(idx)] = mod
                self.forward_funcs.append(mod)
            elif isinstance(spec, torch.nn.Module):
                mods[str(idx)] = spec
                self.forward_funcs.append(spec)
            else:  # plain callable (e.g. lambda reshaping)
                self.forward_funcs.append(spec)
        self.stage_modules = mods
        self._tied_local = tied_local
        self._setup_tied_comms()

    def _stage_of_layer(self, idx):
        for st in range(self.num_stages):
            if self.parts[st] <= idx < self.parts[st + 1]:
                return st
        return self.num_stages - 1

    def _setup_tied_comms(self):
        """Cross-stage weight tying (ref module.py tied_comms +
        engine tied-grad allreduce): stages holding the same TiedLayerSpec
        key form a process group; weights broadcast from the lowest
        owner at init and grads all-reduce (SUM) after each backward so
        every replica takes identical optimizer steps."""
        self.tied_comms = {}
        keys = {}
        for idx, spec in enumerate(self._layer_specs):
            if isinstance(spec, TiedLayerSpec):
                keys.setdefault(spec.key, []).append(idx)
        for key in sorted(keys):
            stages = sorted({self._stage_of_layer(i) for i in keys[key]})
            if len(stages) <= 1:
                continue
            # every rank must take part in new_group creation
            for dp in range(self._grid.data_parallel_size):
                ranks = [st * self._grid.data_parallel_size + dp
                         for st in stages]
                grp = dist.new_group(ranks)
                if dist.get_rank() in ranks:
                    my_group = grp
            if self.stage_id in stages:
                spec = self._layer_specs[keys[key][0]]
                mod = self._tied_local[key]
                weight = getattr(mod, spec.tied_weight_attr)
                self.tied_comms[key] = (my_group, weight,
                                        min(stages), stages)
                # initial sync: lowest owner stage broadcasts
                src = (min(stages) * self._grid.data_parallel_size
                       + self._grid.data_parallel_id)
                dist.broadcast(weight.data, src=src, group=my_group)

    def allreduce_tied_weight_gradients(self):
        for key, (grp, weight, _, _) in self.tied_comms.items():
            if weight.grad is not None:
                dist.all_reduce(weight.grad, group=grp)

    def forward(self, x):
        def run(funcs, inp):
            for f in funcs:
                inp = f(inp)
            return inp

        if self.activation_checkpoint_interval > 0 and self.training:
            k = self.activation_checkpoint_interval
            for i in range(0, len(self.forward_funcs), k):
                chunk = self.forward_funcs[i:i + k]
                if torch.is_tensor(x) and x.requires_grad:
                    x = torch.utils.checkpoint.checkpoint(
                        lambda inp, c=chunk: run(c, inp), x,
                        use_reentrant=False)
                else:
                    x = run(chunk, x)
            return x
        return run(self.forward_funcs, x)

    def mpu(self):
        return self._grid

    def num_pipeline_stages(self):
        return self.num_stages
