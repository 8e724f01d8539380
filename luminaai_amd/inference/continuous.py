"""Token-level continuous batching for serving.

Sequences join and leave the running decode batch between steps (vLLM-style
iteration-level scheduling, sized for one MI355X): a fixed pool of
`max_batch` slots shares one batched KV cache; a new request prefill goes
through `KVCache.row_view` into a free slot, left-padded so its prompt ends
at the batch's shared write column (RoPE attention depends only on relative
positions, so the per-row shift is exact); every `step()` decodes ONE token
for all active slots in a single batched forward with a cache-wide
attention mask (transformer.forward width pos_offset+S) that hides each
row's pad columns and retired history.

The reference has no serving path at all (Chat.py is a REPL); this plus
inference/server.py is the production serving stack.
"""

from __future__ import annotations

import threading
import time
from typing import Callable, Dict, List, Optional

import torch

from .engine import GenerationConfig


class _Slot:
    __slots__ = ("active", "cfg", "stops", "recent", "out", "cb",
                 "done", "result")

    def __init__(self):
        self.active = False


class _RowsView:
    """A compacted multi-row facade over a batched KVCache: forward work
    runs only on the ACTIVE rows (batch len(rows), not max_batch). Appends
    scatter into the parent rows at the shared cursor; reads gather the
    selected rows' prefixes (the attention has to stream those bytes
    anyway)."""

    graph_mode = False
    pos_dev = None

    def __init__(self, parent, rows: List[int]):
        self.parent = parent
        self.rows = torch.as_tensor(rows, dtype=torch.long,
                                    device=parent.k.device
                                    if parent.k is not None else None)

    @property
    def seq_len(self) -> int:
        return self.parent.seq_len

    def append(self, k: torch.Tensor, v: torch.Tensor):
        p = self.parent
        cur = p.seq_len
        S = k.shape[1]
        p.k[self.rows, cur:cur + S] = k
        p.v[self.rows, cur:cur + S] = v
        p._len = cur + S
        return (p.k[self.rows, :cur + S], p.v[self.rows, :cur + S])


class ContinuousBatchingEngine:
    """Iteration-level scheduler over one shared batched KV cache."""

    def __init__(self, model, tokenizer, max_batch: int = 8,
                 max_len: int = 2048, device: Optional[torch.device] = None):
        self.model = model.eval()
        self.tokenizer = tokenizer
        self.device = device or next(model.parameters()).device
        self.max_batch = max_batch
        self.max_len = max_len
        self.caches = model.make_kv_caches(max_len=max_len)
        self.slots = [_Slot() for _ in range(max_batch)]
        # occupancy[b, c] == 1 iff column c of row b holds a live token
        self.occupancy = torch.zeros(max_batch, max_len, dtype=torch.long,
                                     device=self.device)
        self.last_logits = torch.zeros(max_batch, 1, device=self.device)
        self._pad = tokenizer.pad_token_id
        self.stats = {"admitted": 0, "finished": 0, "steps": 0,
                      "tokens_out": 0}

    # --------------------------------------------------------------- state
    @property
    def cursor(self) -> int:
        return self.caches[0].seq_len

    def free_slot(self) -> Optional[int]:
        for i, s in enumerate(self.slots):
            if not s.active:
                return i
        return None

    def n_active(self) -> int:
        return sum(1 for s in self.slots if s.active)

    def can_admit(self, prompt_len: int) -> bool:
        """A join is gap-free only if the prompt fits inside the already-
        decoded column range (start = cursor - P), so existing rows keep
        contiguous positions. An empty batch resets the cursor, so any
        prompt fits there; otherwise a long prompt becomes admissible as
        the batch decodes forward (cursor grows past P)."""
        if self.free_slot() is None:
            return False
        if self.n_active() == 0:
            return prompt_len + 1 < self.max_len
        return prompt_len <= self.cursor and self.cursor + 1 < self.max_len

    # --------------------------------------------------------------- admit
    @torch.no_grad()
    def admit(self, prompt_ids: List[int], cfg: Optional[GenerationConfig]
              = None, stream_callback: Optional[Callable[[int], None]]
              = None) -> int:
        """Prefill one request into a free slot of the LIVE batch; the next
        step() decodes its first token together with everyone else's.
        Returns the slot id."""
        cfg = cfg or GenerationConfig()
        b = self.free_slot()
        if b is None:
            raise RuntimeError("no free slot — poll can_admit first")
        ids = prompt_ids[-min(cfg.max_context, self.max_len - 1):] \
            or [self._pad]
        P = len(ids)
        if self.n_active() == 0 and self.cursor > 0:
            # empty batch: reclaim the whole cache
            for c in self.caches:
                c.truncate(0)
            self.occupancy.zero_()
        cur = self.cursor
        if not self.can_admit(P):
            raise RuntimeError("cannot admit now — poll can_admit first")
        # the prompt ENDS at the shared write column (start = cur - P), so
        # every row's live columns stay contiguous and RoPE positions are a
        # uniform per-row shift; other rows' history at those columns is
        # hidden by the per-row occupancy mask.
        start = max(0, cur - P)
        x = torch.tensor([ids], dtype=torch.long, device=self.device)
        views = []
        for c in self.caches:
            c.ensure_batch(self.max_batch, like=torch.zeros(
                1, 1, self.model.config.num_kv_heads,
                self.model.config.hidden_size
                // self.model.config.num_heads,
                dtype=next(self.model.parameters()).dtype,
                device=self.device))
            views.append(c.row_view(b, start))
        row_mask = torch.zeros(1, start + P, dtype=torch.long,
                               device=self.device)
        row_mask[0, start:] = 1
        logits, _, _ = self.model(x, attention_mask=row_mask,
                                  kv_caches=views)
        # row content between start+P and the shared cursor stays masked
        self.occupancy[b].zero_()
        self.occupancy[b, start:start + P] = 1
        self.last_logits = self.last_logits if \
            self.last_logits.shape[-1] == logits.shape[-1] else \
            torch.zeros(self.max_batch, logits.shape[-1],
                        device=self.device)
        self.last_logits[b] = logits[0, -1].float()
        # a row admitted into an EMPTY region shorter than the batch cursor
        # leaves columns [start+P, cur) dead for this row (occupancy 0)
        s = self.slots[b]
        s.active = True
        s.cfg = cfg
        s.stops = set(cfg.stop_token_ids) | {self.tokenizer.eos_token_id}
        s.recent = list(ids)
        s.out = []
        s.cb = stream_callback
        s.done = threading.Event()
        s.result = None
        self.stats["admitted"] += 1
        # if this prefill advanced the shared cursor (first request, or the
        # longest row so far), other rows' dead columns stay masked
        for c, vw in zip(self.caches, views):
            c._len = max(c.seq_len, vw.seq_len)
        return b

    # ---------------------------------------------------------------- step
    @torch.no_grad()
    def step(self) -> List[int]:
        """Decode ONE token for every active slot (single batched forward).
        Returns slot ids that finished this step."""
        from .engine import sample_token
        if self.n_active() == 0:
            return []
        cur = self.cursor
        finished: List[int] = []
        tok_of: Dict[int, int] = {}
        for b, s in enumerate(self.slots):
            if not s.active:
                continue
            nid = sample_token(self.last_logits[b].clone(), s.cfg,
                               s.recent[-s.cfg.rep_window:])
            if nid in s.stops or len(s.out) >= s.cfg.max_new_tokens:
                finished.append(b)
                continue
            s.out.append(nid)
            s.recent.append(nid)
            tok_of[b] = nid
            if s.cb is not None:
                try:
                    s.cb(nid)
                except Exception:  # noqa: BLE001
                    pass
        for b in finished:
            self._retire(b)
        live = [b for b, s in enumerate(self.slots) if s.active]
        if not live:
            return finished
        if cur + 1 >= self.max_len:
            for b in live:
                self._retire(b)
            return finished + live
        # compacted forward: batch = len(live), not max_batch — a lone
        # request pays B=1 compute
        step_ids = torch.tensor([[tok_of[b]] for b in live],
                                dtype=torch.long, device=self.device)
        mask = torch.cat([self.occupancy[live, :cur],
                          torch.ones(len(live), 1, dtype=torch.long,
                                     device=self.device)], dim=1)
        views = [c if len(live) == self.max_batch else _RowsView(c, live)
                 for c in self.caches]
        logits, _, _ = self.model(step_ids, attention_mask=mask,
                                  kv_caches=views)
        for i, b in enumerate(live):
            self.occupancy[b, cur] = 1
            self.last_logits[b] = logits[i, -1].float()
        self.stats["steps"] += 1
        self.stats["tokens_out"] += len(tok_of)
        return finished

    def _retire(self, b: int):
        s = self.slots[b]
        s.active = False
        s.result = list(s.out)
        self.occupancy[b].zero_()
        self.stats["finished"] += 1
        if s.done is not None:
            s.done.set()

    # ------------------------------------------------------------- convenience
    @torch.no_grad()
    def run_to_completion(self, requests: List, configs=None,
                          admit_schedule: Optional[List[int]] = None
                          ) -> List[List[int]]:
        """Drive admit/step until every request finishes. admit_schedule[i]
        = the step index at which request i arrives (tests the mid-flight
        join path); default admits everything up front."""
        n = len(requests)
        cfgs = configs if isinstance(configs, list) else [configs] * n
        sched = admit_schedule or [0] * n
        slot_of: Dict[int, int] = {}
        results: List[Optional[List[int]]] = [None] * n
        t = 0
        while None in results:
            for i in range(n):
                if results[i] is None and i not in slot_of \
                        and sched[i] <= t and self.can_admit(len(requests[i])):
                    slot_of[i] = self.admit(requests[i],
                                            cfgs[i] or GenerationConfig())
            self.step()
            for i, b in list(slot_of.items()):
                s = self.slots[b]
                if not s.active and s.result is not None:
                    results[i] = s.result
                    del slot_of[i]
            t += 1
            if t > 10 * self.max_len:
                raise RuntimeError("continuous batching did not converge")
        return results  # type: ignore[return-value]
