"""Inference weight quantization at injection/load time.

Parity: reference `runtime/weight_quantizer.py` (WeightQuantization:
group-wise symmetric int-N of checkpoint weights, with extra grouping
for MLP matrices and a model-level entry point). Storage and forward
reuse `linear/quantization.py` (group-wise int8/fp8 bytes + scales,
dequant-on-the-fly) — on MI355X the win is HBM footprint (half the
weight bytes -> bigger models / KV caches per 288 GB), compute stays
bf16 MFMA.
"""
import torch

from ..linear.quantization import QuantizedLinear


class WeightQuantization:
    def __init__(self, mlp_extra_grouping=True, mp_size=1):
        self.mlp_extra_grouping = mlp_extra_grouping
        self.mp_size = mp_size
        self.scales = {}

    @torch.no_grad()
    def quantize_data(self, data, quantize_bits=8, groups=1):
        """Symmetric group-wise fake-int quantization of one tensor:
        returns (int8 tensor of quantized levels, fp32 scales[groups])."""
        assert 2 <= quantize_bits <= 8
        flat = data.float().reshape(groups, -1)
        qmax = (1 << (quantize_bits - 1)) - 1
        absmax = flat.abs().amax(1, keepdim=True).clamp_min(1e-8)
        scale = absmax / qmax
        q = (flat / scale).round().clamp(-qmax - 1, qmax).to(torch.int8)
        return q.reshape(data.shape), scale.squeeze(1)

    @torch.no_grad()
    def dequantize_data(self, qdata, scale, groups=1):
        flat = qdata.float().reshape(groups, -1)
        return (flat * scale.reshape(groups, 1)).reshape(qdata.shape)

    def _groups_for(self, name, shape, quantize_groups):
        g = quantize_groups
        if self.mlp_extra_grouping and any(
                t in name for t in ("mlp", "fc", "up_proj", "down_proj",
                                    "gate_proj", "h_to_4h", "4h_to_h")):
            g *= 2  # MLP matrices carry wider dynamic range per row
        while shape.numel() % g:
            g //= 2
        return max(g, 1)

    @torch.no_grad()
    def sd_quantize(self, sd, quantize_bits=8, quantize_groups=64):
        """Quantize every 2-D weight in a state dict; scales recorded in
        self.scales keyed by parameter name."""
        out = {}
        for name, t in sd.items():
            if torch.is_tensor(t) and t.dim() == 2 and \
                    name.endswith("weight"):
                g = self._groups_for(name, t, quantize_groups)
                q, s = self.quantize_data(t, quantize_bits, g)
                out[name] = q
                self.scales[name] = (s, g)
            else:
                out[name] = t
        return out

    @torch.no_grad()
    def sd_dequantize(self, sd, dtype=torch.bfloat16):
        out = {}
        for name, t in sd.items():
            if name in self.scales:
                s, g = self.scales[name]
                out[name] = self.dequantize_data(t, s, g).to(dtype)
            else:
                out[name] = t
        return out

    def model_quantize(self, model, quantize_bits=8, groups=64,
                       exclude=("lm_head", "embed")):
        """Swap every eligible nn.Linear for a dequant-on-the-fly
        QuantizedLinear (int8 bytes + group scales resident)."""
        assert quantize_bits == 8, \
            "live-module path stores int8; use sd_quantize for other widths"
        n = 0
        for parent in model.modules():
            for cname, child in list(parent.named_children()):
                if isinstance(child, torch.nn.Linear) and not any(
                        e in cname for e in exclude):
                    gs = max(child.weight.numel() // max(groups, 1), 1)
                    setattr(parent, cname,
                            QuantizedLinear(child, quantization="int8",
                                            group_size=gs))
                    n += 1
        return model, n
