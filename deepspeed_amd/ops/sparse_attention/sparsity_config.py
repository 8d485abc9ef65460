"""Block-sparsity layout configs for sparse attention.

Parity: reference `deepspeed/ops/sparse_attention/sparsity_config.py`
(SparsityConfig and the Dense/Fixed/Variable/BigBird/BSLongformer/
LocalSlidingWindow variants). Each config produces a [heads, nq, nk]
0/1 block layout over `block`-sized tiles of the sequence; the attention
module turns that into an additive mask.
"""
import random

import torch


class SparsityConfig:
    def __init__(self, num_heads, block=16, different_layout_per_head=False):
        self.num_heads = num_heads
        self.block = block
        self.different_layout_per_head = different_layout_per_head

    def setup_layout(self, seq_len):
        if seq_len % self.block != 0:
            raise ValueError(
                f"seq len {seq_len} must be divisible by block {self.block}")
        n = seq_len // self.block
        return torch.zeros(self.num_heads, n, n, dtype=torch.int64)

    def check_and_propagate_first_head_layout(self, layout):
        if not self.different_layout_per_head:
            layout[1:] = layout[0]
        return layout

    def make_layout(self, seq_len):
        raise NotImplementedError


class DenseSparsityConfig(SparsityConfig):
    def make_layout(self, seq_len):
        layout = self.setup_layout(seq_len)
        layout[:] = 1
        return layout


class FixedSparsityConfig(SparsityConfig):
    """Local blocks of `num_local_blocks` + `num_global_blocks` global
    columns chosen from the end of each local window (GPT-3-style)."""

    def __init__(self, num_heads, block=16, different_layout_per_head=False,
                 num_local_blocks=4, num_global_blocks=1,
                 attention="bidirectional", horizontal_global_attention=False,
                 num_different_global_patterns=1):
        super().__init__(num_heads, block, different_layout_per_head)
        self.num_local_blocks = num_local_blocks
        self.num_global_blocks = num_global_blocks
        self.attention = attention
        self.horizontal_global_attention = horizontal_global_attention
        self.num_different_global_patterns = num_different_global_patterns

    def make_layout(self, seq_len):
        layout = self.setup_layout(seq_len)
        n = layout.shape[1]
        for h in range(self.num_heads):
            # local windows
            for start in range(0, n, self.num_local_blocks):
                end = min(start + self.num_local_blocks, n)
                for i in range(start, end):
                    hi = (i + 1) if self.attention == "unidirectional" else end
                    layout[h, i, start:hi] = 1
            # global columns: last num_global_blocks of each window
            pat = h % self.num_different_global_patterns \
                if self.different_layout_per_head else 0
            for start in range(0, n, self.num_local_blocks):
                end = min(start + self.num_local_blocks, n)
                first = max(start,
                            end - (pat + 1) * self.num_global_blocks)
                g0 = first
                g1 = min(end, first + self.num_global_blocks)
                if self.attention == "unidirectional":
                    layout[h, end:, g0:g1] = 1
                else:
                    layout[h, :, g0:g1] = 1
                    if self.horizontal_global_attention:
                        layout[h, g0:g1, :] = 1
        if self.attention == "unidirectional":
            layout = torch.tril(layout)
        return self.check_and_propagate_first_head_layout(layout)


class VariableSparsityConfig(SparsityConfig):
    """Custom local window sizes + explicit global block indices."""

    def __init__(self, num_heads, block=16, different_layout_per_head=False,
                 num_random_blocks=0, local_window_blocks=None,
                 global_block_indices=None, global_block_end_indices=None,
                 attention="bidirectional",
                 horizontal_global_attention=False):
        super().__init__(num_heads, block, different_layout_per_head)
        self.num_random_blocks = num_random_blocks
        self.local_window_blocks = local_window_blocks or [4]
        self.global_block_indices = global_block_indices or [0]
        self.global_block_end_indices = global_block_end_indices
        self.attention = attention
        self.horizontal_global_attention = horizontal_global_attention

    def make_layout(self, seq_len):
        layout = self.setup_layout(seq_len)
        n = layout.shape[1]
        for h in range(self.num_heads):
            start = 0
            wi = 0
            while start < n:
                w = self.local_window_blocks[
                    min(wi, len(self.local_window_blocks) - 1)]
                end = min(start + w, n)
                for i in range(start, end):
                    hi = (i + 1) if self.attention == "unidirectional" else end
                    layout[h, i, start:hi] = 1
                start = end
                wi += 1
            ends = self.global_block_end_indices
            for gi, g in enumerate(self.global_block_indices):
                if g >= n:
                    continue
                g1 = min(ends[gi], n) if ends else g + 1
                if self.attention == "unidirectional":
                    layout[h, g:, g:g1] = 1
                else:
                    layout[h, :, g:g1] = 1
                    if self.horizontal_global_attention:
                        layout[h, g:g1, :] = 1
            rng = random.Random(h)
            for i in range(n):
                for _ in range(self.num_random_blocks):
                    layout[h, i, rng.randrange(n)] = 1
        if self.attention == "unidirectional":
            layout = torch.tril(layout)
        return self.check_and_propagate_first_head_layout(layout)


class BigBirdSparsityConfig(SparsityConfig):
    """random + sliding-window + global blocks (BigBird)."""

    def __init__(self, num_heads, block=16, different_layout_per_head=False,
                 num_random_blocks=1, num_sliding_window_blocks=3,
                 num_global_blocks=1, attention="bidirectional"):
        super().__init__(num_heads, block, different_layout_per_head)
        self.num_random_blocks = num_random_blocks
        self.num_sliding_window_blocks = num_sliding_window_blocks
        self.num_global_blocks = num_global_blocks
        self.attention = attention

    def make_layout(self, seq_len):
        layout = self.setup_layout(seq_len)
        n = layout.shape[1]
        w = self.num_sliding_window_blocks // 2
        for h in range(self.num_heads):
            for i in range(n):
                layout[h, i, max(0, i - w):min(n, i + w + 1)] = 1
            g = min(self.num_global_blocks, n)
            layout[h, :, :g] = 1
            layout[h, :g, :] = 1
            rng = random.Random(h)
            for i in range(n):
                for _ in range(self.num_random_blocks):
                    layout[h, i, rng.randrange(n)] = 1
        if self.attention == "unidirectional":
            layout = torch.tril(layout)
        return self.check_and_propagate_first_head_layout(layout)


class BSLongformerSparsityConfig(SparsityConfig):
    """sliding window + explicit global block indices (Longformer)."""

    def __init__(self, num_heads, block=16, different_layout_per_head=False,
                 num_sliding_window_blocks=3, global_block_indices=None,
                 global_block_end_indices=None, attention="bidirectional"):
        super().__init__(num_heads, block, different_layout_per_head)
        self.num_sliding_window_blocks = num_sliding_window_blocks
        self.global_block_indices = global_block_indices or [0]
        self.global_block_end_indices = global_block_end_indices
        self.attention = attention

    def make_layout(self, seq_len):
        layout = self.setup_layout(seq_len)
        n = layout.shape[1]
        w = self.num_sliding_window_blocks // 2
        for h in range(self.num_heads):
            for i in range(n):
                layout[h, i, max(0, i - w):min(n, i + w + 1)] = 1
            ends = self.global_block_end_indices
            for gi, g in enumerate(self.global_block_indices):
                if g >= n:
                    continue
                g1 = min(ends[gi], n) if ends else g + 1
                layout[h, :, g:g1] = 1
                layout[h, g:g1, :] = 1
        if self.attention == "unidirectional":
            layout = torch.tril(layout)
        return self.check_and_propagate_first_head_layout(layout)


class LocalSlidingWindowSparsityConfig(SparsityConfig):
    def __init__(self, num_heads, block=16, num_sliding_window_blocks=3,
                 attention="unidirectional"):
        super().__init__(num_heads, block, False)
        self.num_sliding_window_blocks = num_sliding_window_blocks
        self.attention = attention

    def make_layout(self, seq_len):
        layout = self.setup_layout(seq_len)
        n = layout.shape[1]
        w = self.num_sliding_window_blocks // 2
        for h in range(self.num_heads):
            for i in range(n):
                layout[h, i, max(0, i - w):min(n, i + w + 1)] = 1
        if self.attention == "unidirectional":
            layout = torch.tril(layout)
        return layout
