"""hipGraph capture for the launch-bound decode loop.

Reference: core/transformer/cuda_graphs.py (create_cudagraphs / graphed
module wrappers).  On MI355X the single-token decode step is dominated by
kernel-launch latency — a few hundred tiny kernels for one token — so the
whole step is captured once into a hipGraph (`torch.cuda.CUDAGraph` is the
hipGraph API on ROCm) and replayed per token.

Graph replay requires every tensor SHAPE and kernel sequence to be fixed;
only tensor CONTENTS may change.  Decode breaks that in two places, both
solved here:

* KV cache append — normally indexed by a Python int offset.
  :class:`GraphDecodeContext` keeps the offset as a device tensor and
  appends with ``index_copy_``, advancing the offset *inside* the graph,
  so replays self-advance.
* Attention over a growing prefix — normally ``k[:cur_len]``.  The graphed
  step always attends over the full ``max_sequence_length`` window with a
  padding mask computed from the device offset (invalid tail masked), so
  shapes never change.

Usage::

    ctx = GraphDecodeContext(max_batch_size=b, max_sequence_length=L)
    ...prefill eagerly with inference_context=ctx...
    step = GraphedDecodeStep(model, ctx, batch_size=b)  # captures
    logits = step(tokens_1col, pos_1col)                # replay per token
"""

from __future__ import annotations

import torch

from .inference_params import InferenceParams


class GraphDecodeContext(InferenceParams):
    """Inference KV-cache whose append path is hipGraph-replayable."""

    is_graph_context = True
    window = None   # None = whole cache (class default survives __new__)

    def __init__(self, max_batch_size: int, max_sequence_length: int):
        super().__init__(max_batch_size, max_sequence_length)
        self.device = torch.device("cuda")
        # device-resident sequence length; single source of truth while
        # inside a captured graph
        self.cur_len = torch.zeros(1, dtype=torch.long, device=self.device)
        self._arange = torch.arange(max_sequence_length, device=self.device)
        self.graph_mode = False  # True once shapes must stay fixed
        # attention window for the CURRENT capture/replay: a graph
        # captured at window W only ever reads cache[:W], so short
        # contexts stop paying for the full max-length window
        self.window = max_sequence_length

    # -- offset bookkeeping -------------------------------------------------
    @property
    def sequence_len_offset(self):
        # int view for the eager prefill path
        return self._offset_int

    @sequence_len_offset.setter
    def sequence_len_offset(self, v):
        self._offset_int = int(v)
        if hasattr(self, "cur_len"):
            self.cur_len.fill_(int(v))

    def increment_sequence_len_offset(self, n: int):
        self._offset_int += n
        # cur_len is advanced inside the graph in graph_mode; keep the two
        # in sync only when eager
        if not self.graph_mode:
            self.cur_len.fill_(self._offset_int)

    # -- cache update -------------------------------------------------------
    def update_kv_cache(self, layer_number: int, key: torch.Tensor,
                        value: torch.Tensor):
        s_new, b, ng, hd = key.shape
        if layer_number not in self.key_value_memory_dict:
            assert not self.graph_mode, "cache must be allocated pre-capture"
            k_cache = torch.zeros(self.max_sequence_length, b, ng, hd,
                                  dtype=key.dtype, device=key.device)
            v_cache = torch.zeros(self.max_sequence_length, b, ng,
                                  value.shape[-1], dtype=value.dtype,
                                  device=value.device)
            self.key_value_memory_dict[layer_number] = (k_cache, v_cache)
        k_cache, v_cache = self.key_value_memory_dict[layer_number]
        # graph-safe append: positions come from the device offset
        idx = self._arange[:s_new] + self.cur_len
        k_cache.index_copy_(0, idx, key)
        v_cache.index_copy_(0, idx, value)
        # fixed-shape window slice (= whole cache unless bucketed);
        # DotProductAttention masks the tail
        w = self.window or k_cache.shape[0]
        return k_cache[:w], v_cache[:w]

    def decode_padding_mask(self, sq: int, batch: int) -> torch.Tensor:
        """[b, 1, sq, window] bool, True = masked.  Valid keys for query
        row r (global position cur_len + r) are positions <= cur_len + r."""
        pos_q = (self.cur_len + self._arange[:sq]).view(1, 1, sq, 1)
        w = self.window or self.max_sequence_length
        pos_k = self._arange[:w].view(1, 1, 1, -1)
        return (pos_k > pos_q).expand(batch, 1, sq, -1)


class BucketedGraphedDecodeStep:
    """Lengths-bucketed graph set: one captured graph per power-of-two
    attention window, selected by the current offset — long-context
    decode stops attending over the full max-length window (the cost
    that shrank the round-1 2.3x win as context grew)."""

    MIN_BUCKET = 256

    def __init__(self, model, ctx: GraphDecodeContext, batch_size: int):
        self.model = model
        self.ctx = ctx
        self.batch_size = batch_size
        self.graphs = {}

    @staticmethod
    def _bucket_for(needed: int, max_len: int, min_bucket: int) -> int:
        w = min_bucket
        while w < needed:
            w *= 2
        return min(w, max_len)

    @torch.no_grad()
    def __call__(self, tokens: torch.Tensor, pos: torch.Tensor):
        needed = self.ctx._offset_int + 1
        w = self._bucket_for(needed, self.ctx.max_sequence_length,
                             self.MIN_BUCKET)
        if w not in self.graphs:
            self.ctx.window = w
            # all buckets share ONE memory pool: capturing a second
            # graph against a fresh pool after replaying the first
            # faults the allocator (observed as a device abort)
            pool = next(iter(self.graphs.values())).graph.pool() \
                if self.graphs else None
            self.graphs[w] = GraphedDecodeStep(self.model, self.ctx,
                                               self.batch_size, pool=pool)
        self.ctx.window = w
        return self.graphs[w](tokens, pos)


class GraphedDecodeStep:
    """Capture model(tokens, pos) single-token decode into one hipGraph."""

    def __init__(self, model, ctx: GraphDecodeContext, batch_size: int,
                 warmup_iters: int = 3, pool=None):
        self.model = model
        self.ctx = ctx
        device = ctx.device
        self.static_tokens = torch.zeros(batch_size, 1, dtype=torch.long,
                                         device=device)
        self.static_pos = torch.zeros(batch_size, 1, dtype=torch.long,
                                      device=device)

        ctx.graph_mode = True
        start_len = int(ctx.cur_len.item())
        # warmup on a side stream (allocator + RCCL lazy init outside capture)
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(warmup_iters):
                ctx.cur_len.fill_(start_len)
                self.model(self.static_tokens, self.static_pos, None,
                           inference_context=ctx)
        torch.cuda.current_stream().wait_stream(side)

        ctx.cur_len.fill_(start_len)
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph, pool=pool):
            self.static_logits = self.model(
                self.static_tokens, self.static_pos, None,
                inference_context=ctx)
            ctx.cur_len.add_(1)  # replays self-advance
        ctx.cur_len.fill_(start_len)

    @torch.no_grad()
    def __call__(self, tokens: torch.Tensor, pos: torch.Tensor):
        self.static_tokens.copy_(tokens)
        self.static_pos.copy_(pos)
        self.graph.replay()
        self.ctx._offset_int += 1
        return self.static_logits
