"""Compression: QAT weight/activation quantization + layer reduction.

Parity: reference `compression/compress.py`, `basic_layer.py`
(LinearLayer_Compress), `helper.py`. `init_compression` walks the module
tree and swaps Linears for QAT-capable ones per config; `redundancy_clean`
makes quantization permanent.
"""
import re

import torch
import torch.nn.functional as F

from .utils.logging import log_dist


class _FakeQuant(torch.autograd.Function):
    """Straight-through symmetric fake quantization."""

    @staticmethod
    def forward(ctx, x, bits):
        qmax = 2 ** (bits - 1) - 1
        amax = x.abs().amax().clamp(min=1e-8)
        scale = amax / qmax
        return torch.clamp(torch.round(x / scale), -qmax, qmax) * scale

    @staticmethod
    def backward(ctx, g):
        return g, None


def fake_quantize(x, bits=8):
    return _FakeQuant.apply(x, bits)


class LinearLayerCompress(torch.nn.Linear):
    """Linear with optional QAT weight/activation quantization and
    row/head pruning masks."""

    def __init__(self, in_features, out_features, bias=True):
        super().__init__(in_features, out_features, bias=bias)
        self.weight_quant_enabled = False
        self.weight_bits = 8
        self.act_quant_enabled = False
        self.act_bits = 8
        self.register_buffer("prune_mask", None)

    @classmethod
    def from_linear(cls, linear):
        m = cls(linear.in_features, linear.out_features,
                bias=linear.bias is not None)
        m.weight.data.copy_(linear.weight.data)
        if linear.bias is not None:
            m.bias.data.copy_(linear.bias.data)
        return m.to(linear.weight.dtype)

    def enable_weight_quantization(self, bits=8):
        self.weight_quant_enabled = True
        self.weight_bits = bits

    def enable_activation_quantization(self, bits=8):
        self.act_quant_enabled = True
        self.act_bits = bits

    def enable_row_pruning(self, ratio, method="l1"):
        """Zero whole output rows by L1 importance (structured)."""
        imp = self.weight.data.abs().sum(dim=1)
        k = int(imp.numel() * ratio)
        if k > 0:
            thresh = imp.kthvalue(k).values
            mask = (imp > thresh).to(self.weight.dtype)
            self.prune_mask = mask.unsqueeze(1).expand_as(self.weight) \
                .contiguous()

    def enable_head_pruning(self, ratio, num_heads):
        """Zero whole attention heads (blocks of output rows) by L2
        importance — for the concatenated q/k/v/o projections."""
        O = self.weight.shape[0]
        assert O % num_heads == 0
        hs = O // num_heads
        imp = self.weight.data.reshape(num_heads, hs, -1) \
            .float().pow(2).sum(dim=(1, 2)).sqrt()
        k = int(num_heads * ratio)
        if k > 0:
            thresh = imp.kthvalue(k).values
            mask = (imp > thresh).to(self.weight.dtype)
            self.prune_mask = mask.view(num_heads, 1, 1).expand(
                num_heads, hs, self.weight.shape[1]) \
                .reshape_as(self.weight).contiguous()

    def enable_sparse_pruning(self, ratio, method="l1"):
        w = self.weight.data.abs()
        k = int(w.numel() * ratio)
        if k > 0:
            thresh = w.reshape(-1).kthvalue(k).values
            self.prune_mask = (w > thresh).to(self.weight.dtype)

    def forward(self, x):
        w = self.weight
        if self.prune_mask is not None:
            w = w * self.prune_mask
        if self.weight_quant_enabled:
            w = fake_quantize(w.float(), self.weight_bits).to(x.dtype)
        if self.act_quant_enabled:
            x = fake_quantize(x.float(), self.act_bits).to(w.dtype)
        return F.linear(x, w, self.bias)


def init_compression(model, compression_config):
    """Swap matching Linears for compressible ones; apply settings.

    Config (subset of reference schema):
      {"weight_quantization": {"shared_parameters": {"enabled": true},
          "different_groups": {"wq1": {
              "params": {"target_bits": 8},
              "modules": ["attention.*", ".*mlp.*"]}}},
       "sparse_pruning": {...}}
    """
    wq = compression_config.get("weight_quantization", {})
    sp = compression_config.get("sparse_pruning", {})
    aq = compression_config.get("activation_quantization", {})
    rp = compression_config.get("row_pruning", {})
    hp = compression_config.get("head_pruning", {})
    n = 0
    for name, module in list(model.named_modules()):
        for child_name, child in list(module.named_children()):
            if type(child) is torch.nn.Linear or \
                    isinstance(child, LinearLayerCompress):
                full = f"{name}.{child_name}" if name else child_name
                # already-compressible modules take new techniques in
                # place (scheduler activates methods at different steps)
                new = child if isinstance(child, LinearLayerCompress) \
                    else None
                for gname, g in wq.get("different_groups", {}).items():
                    if any(re.search(pat, full)
                           for pat in g.get("modules", [".*"])):
                        new = LinearLayerCompress.from_linear(child)
                        new.enable_weight_quantization(
                            g.get("params", {}).get("target_bits", 8))
                        break
                for gname, g in sp.get("different_groups", {}).items():
                    if any(re.search(pat, full)
                           for pat in g.get("modules", [".*"])):
                        if new is None:
                            new = LinearLayerCompress.from_linear(child)
                        new.enable_sparse_pruning(
                            g.get("params", {}).get("dense_ratio", 0.5))
                        break
                for gname, g in aq.get("different_groups", {}).items():
                    if any(re.search(pat, full)
                           for pat in g.get("modules", [".*"])):
                        if new is None:
                            new = LinearLayerCompress.from_linear(child)
                        new.enable_activation_quantization(
                            g.get("params", {}).get("bits", 8))
                        break
                for gname, g in rp.get("different_groups", {}).items():
                    if any(re.search(pat, full)
                           for pat in g.get("modules", [".*"])):
                        if new is None:
                            new = LinearLayerCompress.from_linear(child)
                        new.enable_row_pruning(
                            1.0 - g.get("params", {}).get("dense_ratio",
                                                          0.5))
                        break
                for gname, g in hp.get("different_groups", {}).items():
                    if any(re.search(pat, full)
                           for pat in g.get("modules", [".*"])):
                        if new is None:
                            new = LinearLayerCompress.from_linear(child)
                        new.enable_head_pruning(
                            1.0 - g.get("params", {}).get("dense_ratio",
                                                          0.5),
                            g.get("params", {}).get("num_heads", 1))
                        break
                if new is not None:
                    setattr(module, child_name, new)
                    n += 1
    log_dist(f"init_compression: converted {n} linears", ranks=[0])
    return model


def redundancy_clean(model, compression_config=None):
    """Make quantization/pruning permanent (bake into weights)."""
    for module in model.modules():
        if isinstance(module, LinearLayerCompress):
            if module.prune_mask is not None:
                module.weight.data.mul_(module.prune_mask)
                module.prune_mask = None
            if module.weight_quant_enabled:
                module.weight.data.copy_(
                    fake_quantize(module.weight.data.float(),
                                  module.weight_bits)
                    .to(module.weight.dtype))
                module.weight_quant_enabled = False
    return model


def apply_layer_reduction(model, layer_reduction_config):
    """Student-model layer reduction (ref compression layer_reduction):
    keep `keep_number` layers of the ModuleList at `module_name_prefix`,
    selecting the listed `teacher_layer` indices."""
    cfg = layer_reduction_config
    prefix = cfg["module_name_prefix"]
    keep = cfg.get("teacher_layer",
                   list(range(cfg["keep_number"])))
    holder = model
    for part in prefix.split("."):
        holder = getattr(holder, part)
    assert isinstance(holder, torch.nn.ModuleList), prefix
    new_layers = torch.nn.ModuleList([holder[i] for i in keep])
    parent = model
    parts = prefix.split(".")
    for part in parts[:-1]:
        parent = getattr(parent, part)
    setattr(parent, parts[-1], new_layers)
    log_dist(f"layer_reduction: kept {len(new_layers)} layers {keep}",
             ranks=[0])
    return model


class CompressionScheduler:
    """Step-scheduled compression (ref compression/scheduler.py:12):
    each technique activates once training reaches its
    shared_parameters.schedule_offset (and deactivates at
    schedule_offset_end if given), instead of from step 0."""

    METHODS = ("weight_quantization", "activation_quantization",
               "sparse_pruning", "row_pruning", "head_pruning")

    def __init__(self, model, compression_config):
        self.model = model
        self.config = dict(compression_config or {})
        self.training_steps = 0
        self._activated = set()
        self._deactivated = set()
        # convert eagerly only methods with offset 0 (classic behavior)
        self.step(0)

    def _offsets(self, method):
        sp = self.config.get(method, {}).get("shared_parameters", {})
        return (sp.get("schedule_offset", 0),
                sp.get("schedule_offset_end", None),
                sp.get("enabled", True))

    def step(self, increment=1):
        self.training_steps += increment
        for m in self.METHODS:
            if m not in self.config:
                continue
            off, off_end, enabled = self._offsets(m)
            if not enabled:
                continue
            if m not in self._activated and self.training_steps >= off:
                init_compression(self.model, {m: self.config[m]})
                self._activated.add(m)
                log_dist(f"compression: {m} enabled at step "
                         f"{self.training_steps}", ranks=[0])
            if off_end is not None and m in self._activated and \
                    m not in self._deactivated and \
                    self.training_steps >= off_end:
                for mod in self.model.modules():
                    if isinstance(mod, LinearLayerCompress):
                        if m == "weight_quantization":
                            mod.weight_quant_enabled = False
                        elif m == "activation_quantization":
                            mod.act_quant_enabled = False
                        else:
                            mod.prune_mask = None
                self._deactivated.add(m)
                log_dist(f"compression: {m} disabled at step "
                         f"{self.training_steps}", ranks=[0])

    def state_dict(self):
        return {"training_steps": self.training_steps,
                "activated": sorted(self._activated),
                "deactivated": sorted(self._deactivated)}

    def load_state_dict(self, sd):
        self.training_steps = sd["training_steps"]
        self._activated = set(sd["activated"])
        self._deactivated = set(sd["deactivated"])
