"""hipGraph-captured single-token decoding.

The batch-1 decode step is launch-bound on MI355X (b1: ~370 kernel launches
per token ≈ 7.4 ms/token eager). With the graph-mode KV cache (device-side
write cursor, full static buffers, cursor-masked attention) the whole
layer stack is shape-static, so one decode step is captured once into a
hipGraph and replayed per token: the host loop only updates the input-token
buffer and samples.

Eager fallback everywhere capture is unsupported; numerics equivalence is
tested in tests/test_ops_gpu.py (graph vs eager greedy decode).
"""

from __future__ import annotations

from typing import List, Optional

import torch


class GraphedDecoder:
    """Capture-once, replay-per-token decode for a DeepSeekTransformer."""

    def __init__(self, model, max_context: int, batch: int = 1):
        self.model = model
        self.device = next(model.parameters()).device
        self.max_context = max_context
        self.batch = batch
        self.caches = model.make_kv_caches(max_len=max_context,
                                           graph_mode=True)
        self._graph: Optional[torch.cuda.CUDAGraph] = None
        self._in_tok: Optional[torch.Tensor] = None
        self._out_logits: Optional[torch.Tensor] = None

    @property
    def seq_len(self) -> int:
        return self.caches[0].seq_len

    def reset(self):
        """Clear the caches for a new prompt (the captured graph keeps
        referencing the same buffers, so capture survives resets)."""
        for c in self.caches:
            c._len = 0
            if c.pos_dev is not None:
                c.pos_dev.zero_()

    @torch.no_grad()
    def prefill(self, ids: torch.Tensor) -> torch.Tensor:
        """Run the prompt eagerly (fills the static caches); returns logits
        of the last position."""
        # pre-size the RoPE tables so capture never rebuilds them
        self.model.rotary.get(self.max_context, self.device)
        logits, _, _ = self.model(ids, kv_caches=self.caches)
        return logits[:, -1]

    @torch.no_grad()
    def _capture(self):
        self._in_tok = torch.zeros(self.batch, 1, dtype=torch.long,
                                   device=self.device)
        # warmup on a side stream (required before capture)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                logits, _, _ = self.model(self._in_tok, kv_caches=self.caches)
                self._rollback(1)          # undo the warmup advance
        torch.cuda.current_stream().wait_stream(s)

        self._graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self._graph):
            logits, _, _ = self.model(self._in_tok, kv_caches=self.caches)
            self._out_logits = logits[:, -1].float()
        # stream capture records but does NOT execute: only the host-side
        # _len mirror advanced during the captured python pass — undo it
        for c in self.caches:
            c._len -= 1

    def _rollback(self, n: int):
        """Undo n EXECUTED single-token advances (device cursor + mirror)."""
        for c in self.caches:
            c.pos_dev.sub_(n)
            c._len -= n

    def ensure_captured(self):
        # A prompt that exactly fills max_context leaves the cache cursor at
        # capacity; the warmup/capture single-token pass would then
        # index_copy_ at idx == capacity (a sticky device-side assert on
        # GPU). Skip capture -- the decode loop stops before any step.
        if self.seq_len >= self.max_context:
            return
        if self._graph is None:
            self._capture()

    @torch.no_grad()
    def step(self, token_id: torch.Tensor) -> torch.Tensor:
        """token_id: [B] or [B,1] long on device -> logits [B, V] fp32."""
        if self.seq_len >= self.max_context:
            raise RuntimeError(
                f"KV cache full (seq_len={self.seq_len} == max_context); "
                "cannot decode further")
        if self._graph is None:
            self._capture()
        self._in_tok.copy_(token_id.view(self.batch, 1))
        self._graph.replay()
        for c in self.caches:
            c._len += 1          # host mirror (device cursor moved in-graph)
        return self._out_logits
