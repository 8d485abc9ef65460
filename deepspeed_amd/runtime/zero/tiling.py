"""TiledLinear: tile a huge Linear so ZeRO-3 can partition/release the
inactive tiles (ref runtime/zero/tiling.py:32).

Each (out_split, in_split) tile is its own Linear module, so under
ZeRO-3 every tile's parameters shard independently and only the tile
being computed is gathered — peak memory for an [I, O] layer drops from
I*O to I*O/(in_splits*out_splits) plus one tile's activations.
"""
import torch


def _split_sizes(n, parts):
    base = n // parts
    sizes = [base] * parts
    for i in range(n - base * parts):
        sizes[i] += 1
    return sizes


class TiledLinear(torch.nn.Module):
    def __init__(self, in_features, out_features, bias=True, in_splits=1,
                 out_splits=1, input_is_already_split=False,
                 combine_out_splits=True, linear_cls=torch.nn.Linear,
                 init_linear=None, **kwargs):
        super().__init__()
        assert in_splits >= 1 and out_splits >= 1
        self.in_features = in_features
        self.out_features = out_features
        self.in_splits = in_splits
        self.out_splits = out_splits
        self.input_is_already_split = input_is_already_split
        self.combine_out_splits = combine_out_splits
        self.in_parts = _split_sizes(in_features, in_splits)
        self.out_parts = _split_sizes(out_features, out_splits)
        self.linears = torch.nn.ModuleList()
        for o, on in enumerate(self.out_parts):
            row = torch.nn.ModuleList()
            for i, inn in enumerate(self.in_parts):
                # bias only once per output tile (applied on the last
                # in-split so partial sums stay bias-free)
                row.append(linear_cls(inn, on,
                                      bias=bias and i == in_splits - 1,
                                      **kwargs))
            self.linears.append(row)
        if init_linear is not None:
            self.copy_params_from(init_linear)

    @torch.no_grad()
    def copy_params_from(self, other):
        """Load weights/bias from a plain nn.Linear of the full shape."""
        assert other.weight.shape == (self.out_features, self.in_features)
        o0 = 0
        for o, on in enumerate(self.out_parts):
            i0 = 0
            for i, inn in enumerate(self.in_parts):
                self.linears[o][i].weight.copy_(
                    other.weight[o0:o0 + on, i0:i0 + inn])
                if self.linears[o][i].bias is not None and \
                        other.bias is not None:
                    self.linears[o][i].bias.copy_(other.bias[o0:o0 + on])
                i0 += inn
            o0 += on

    def forward(self, x):
        if self.input_is_already_split:
            xs = x
        else:
            xs = torch.split(x, self.in_parts, dim=-1)
        outs = []
        for o in range(self.out_splits):
            acc = None
            for i in range(self.in_splits):
                y = self.linears[o][i](xs[i])
                acc = y if acc is None else acc + y
            outs.append(acc)
        if self.combine_out_splits:
            return torch.cat(outs, dim=-1)
        return outs


class TiledLinearReturnBias(TiledLinear):
    """Variant for Megatron-style linears returning (out, bias): partial
    outputs sum, the returned bias passes through once."""

    def forward(self, x):
        if self.input_is_already_split:
            xs = x
        else:
            xs = torch.split(x, self.in_parts, dim=-1)
        outs, biases = [], []
        for o in range(self.out_splits):
            acc, bias = None, None
            for i in range(self.in_splits):
                r = self.linears[o][i](xs[i])
                y, b = r if isinstance(r, tuple) else (r, None)
                acc = y if acc is None else acc + y
                if b is not None:
                    bias = b
            outs.append(acc)
            biases.append(bias)
        if self.combine_out_splits:
            out = torch.cat(outs, dim=-1)
            bias = torch.cat(biases, dim=-1) \
                if all(b is not None for b in biases) else None
            return out, bias
        return outs, biases
