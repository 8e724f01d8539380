"""Incremental-decoding generation engine.

Rebuild of the reference GenerationEngine (/root/reference/Src/Main_Scripts/
Chat.py:355-465) with the O(L²)-per-token flaw fixed: the reference re-ran the
full forward for every generated token even though its attention supported a
KV cache (Chat.py:381, model.py:694-702). Here the prompt is prefilled once
and each new token does a single-position forward against per-layer KV caches.
Sampling: repetition penalty over the last `rep_window` tokens, temperature,
top-k, top-p nucleus, greedy; 4 named sampling modes (Chat.py:60-85)."""

from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional

import torch

SAMPLING_MODES: Dict[str, Dict] = {
    "greedy": {"temperature": 0.0, "top_k": 0, "top_p": 1.0},
    "standard": {"temperature": 0.8, "top_k": 50, "top_p": 0.9},
    "creative": {"temperature": 1.1, "top_k": 100, "top_p": 0.95},
    "precise": {"temperature": 0.3, "top_k": 20, "top_p": 0.85},
}


@dataclass
class GenerationConfig:
    max_new_tokens: int = 256
    temperature: float = 0.8
    top_k: int = 50
    top_p: float = 0.9
    repetition_penalty: float = 1.1
    rep_window: int = 50
    stop_token_ids: List[int] = field(default_factory=list)
    max_context: int = 2048

    @classmethod
    def from_mode(cls, mode: str, **overrides) -> "GenerationConfig":
        kw = dict(SAMPLING_MODES.get(mode, SAMPLING_MODES["standard"]))
        kw.update(overrides)
        return cls(**kw)


def sample_token(logits: torch.Tensor, cfg: GenerationConfig,
                 recent: List[int]) -> int:
    """One token from processed logits: repetition penalty over `recent`
    (reference Chat.py:392-398), temperature, top-k, top-p nucleus, greedy
    at temperature 0. Mutates `logits` (callers pass a clone or a buffer
    they are done with)."""
    if cfg.repetition_penalty != 1.0 and recent:
        idx = torch.tensor(sorted(set(recent)), device=logits.device)
        sel = logits[idx]
        logits[idx] = torch.where(sel > 0, sel / cfg.repetition_penalty,
                                  sel * cfg.repetition_penalty)
    if cfg.temperature <= 0.0:
        return int(logits.argmax())
    logits = logits / cfg.temperature
    if cfg.top_k and cfg.top_k > 0:
        kth = torch.topk(logits, min(cfg.top_k, logits.numel())).values[-1]
        logits = logits.masked_fill(logits < kth, float("-inf"))
    if cfg.top_p < 1.0:
        sorted_logits, sorted_idx = torch.sort(logits, descending=True)
        probs = torch.softmax(sorted_logits, dim=-1)
        cum = torch.cumsum(probs, dim=-1)
        cut = cum - probs > cfg.top_p       # keep first token above p
        sorted_logits = sorted_logits.masked_fill(cut, float("-inf"))
        logits = torch.full_like(logits, float("-inf")).scatter(
            0, sorted_idx, sorted_logits)
    probs = torch.softmax(logits, dim=-1)
    return int(torch.multinomial(probs, 1))


class GenerationEngine:
    def __init__(self, model, tokenizer, device: Optional[torch.device] = None):
        self.model = model
        self.tokenizer = tokenizer
        self.device = device or next(model.parameters()).device
        self.stats = {"tokens_generated": 0, "time_in_generate": 0.0,
                      "prefill_tokens": 0}

    @torch.no_grad()
    def generate(self, prompt_ids: List[int],
                 config: Optional[GenerationConfig] = None,
                 stream_callback: Optional[Callable[[int], None]] = None,
                 use_graph: Optional[bool] = None) -> List[int]:
        """Returns the generated token ids (not including the prompt).
        On GPU the single-token steps replay a captured hipGraph by default
        (graph_decode.py); pass use_graph=False for the eager loop."""
        cfg = config or GenerationConfig()
        t0 = time.perf_counter()
        self.model.eval()
        stop = set(cfg.stop_token_ids) | {self.tokenizer.eos_token_id}

        ids = prompt_ids[-cfg.max_context:] or [self.tokenizer.pad_token_id]
        x = torch.tensor([ids], dtype=torch.long, device=self.device)
        if use_graph is None:
            use_graph = self.device.type == "cuda"

        dec = None
        if use_graph:
            # prefer the fused 5-kernel decode path (inference/fused_decode
            # .py: 2x the graphed eager decoder) for dense/MoD models
            try:
                from .fused_decode import FusedDecoder, can_fuse_decode
                if can_fuse_decode(self.model):
                    if getattr(self, "_fused_dec", None) is None or \
                            self._fused_dec.max_context != cfg.max_context:
                        self._fused_dec = FusedDecoder(self.model,
                                                       cfg.max_context)
                    fd = self._fused_dec
                    fd.reset()
                    last_logits = fd.prefill(x)[0].float()

                    class _FDAdapter:
                        seq_len = property(lambda s: fd.seq_len)

                        @staticmethod
                        def step(t):
                            return fd.step(t.view(1)).float().view(1, -1)

                    dec = _FDAdapter()
                    seq_len = lambda: fd.seq_len  # noqa: E731
            except Exception:  # noqa: BLE001
                self._fused_dec = None
                dec = None
        if use_graph and dec is None:
            try:
                from .graph_decode import GraphedDecoder
                if getattr(self, "_graph_dec", None) is None or \
                        self._graph_dec.max_context != cfg.max_context:
                    self._graph_dec = GraphedDecoder(self.model,
                                                     cfg.max_context)
                dec = self._graph_dec
                dec.reset()
                last_logits = dec.prefill(x)[0].float()
                dec.ensure_captured()      # capture now; fall back on failure
                seq_len = lambda: dec.seq_len  # noqa: E731
            except Exception:  # noqa: BLE001 — e.g. un-capturable MoE ops
                self._graph_dec = None
                dec = None
                use_graph = False
        if not use_graph:
            caches = self.model.make_kv_caches(max_len=cfg.max_context)
            logits, _, _ = self.model(x, kv_caches=caches)
            last_logits = logits[0, -1].float()
            seq_len = lambda: caches[0].seq_len  # noqa: E731

        self.stats["prefill_tokens"] += len(ids)
        generated: List[int] = []
        recent: List[int] = list(ids)

        for _ in range(cfg.max_new_tokens):
            next_id = self._sample(last_logits, cfg, recent[-cfg.rep_window:])
            if next_id in stop:
                break
            generated.append(next_id)
            recent.append(next_id)
            if stream_callback is not None:
                stream_callback(next_id)
            if seq_len() >= cfg.max_context:
                break
            # ---- single-token incremental step
            step = torch.tensor([[next_id]], dtype=torch.long,
                                device=self.device)
            if dec is not None:
                last_logits = dec.step(step)[0]
            else:
                logits, _, _ = self.model(step, kv_caches=caches)
                last_logits = logits[0, -1].float()

        self.stats["tokens_generated"] += len(generated)
        self.stats["time_in_generate"] += time.perf_counter() - t0
        return generated

    @torch.no_grad()
    def generate_speculative(self, prompt_ids: List[int], draft_model,
                             config: Optional[GenerationConfig] = None,
                             draft_k: int = 4) -> List[int]:
        """Greedy speculative decoding: a cheap draft model proposes
        draft_k tokens, the target verifies them in ONE forward (sequence
        of k+1 tokens — MFMA-shaped work instead of k GEMV chains). Output
        is EXACTLY the target's greedy decode for any draft; acceptance is
        argmax equality. With temperature > 0 the classic acceptance-
        rejection scheme runs instead (accept x ~ q with prob
        min(1, p(x)/q(x)); on reject resample from norm(max(p-q, 0))),
        which preserves the target's sampling distribution exactly.
        Rejections rewind the KV caches via KVCache.truncate."""
        cfg = config or GenerationConfig(temperature=0.0)
        greedy = cfg.temperature == 0.0
        self.model.eval()
        draft_model.eval()
        stop = set(cfg.stop_token_ids) | {self.tokenizer.eos_token_id}
        ids = prompt_ids[-cfg.max_context:] or [self.tokenizer.pad_token_id]
        cap = len(ids) + cfg.max_new_tokens + draft_k + 2
        ct = self.model.make_kv_caches(max_len=cap)
        cd = draft_model.make_kv_caches(max_len=cap)
        x = torch.tensor([ids], dtype=torch.long, device=self.device)
        lt, _, _ = self.model(x, kv_caches=ct)
        draft_model(x, kv_caches=cd)
        if greedy:
            nxt = int(lt[0, -1].argmax())
        else:
            nxt = int(self._filtered_probs(
                lt[0, -1].float(), cfg,
                ids[-cfg.rep_window:]).multinomial(1))
        if not greedy:
            return self._speculative_sampling_loop(
                draft_model, cfg, draft_k, ids, ct, cd, nxt, stop)

        generated: List[int] = []
        self.stats["prefill_tokens"] += len(ids)
        t0 = time.perf_counter()
        while len(generated) < cfg.max_new_tokens and nxt not in stop:
            generated.append(nxt)
            if len(generated) >= cfg.max_new_tokens:
                break
            # draft first consumes whatever confirmed history it is missing
            # (>= 1 token: at least the newest `nxt`), then proposes k
            confirmed = ids + generated
            missing = confirmed[cd[0].seq_len:]
            ld, _, _ = draft_model(
                torch.tensor([missing], dtype=torch.long,
                             device=self.device), kv_caches=cd)
            cur = int(ld[0, -1].argmax())
            proposal: List[int] = [cur]
            for _ in range(draft_k - 1):
                ld, _, _ = draft_model(
                    torch.tensor([[cur]], dtype=torch.long,
                                 device=self.device), kv_caches=cd)
                cur = int(ld[0, -1].argmax())
                proposal.append(cur)
            # target verifies the whole block in one forward
            block = torch.tensor([[nxt] + proposal], dtype=torch.long,
                                 device=self.device)
            lt, _, _ = self.model(block, kv_caches=ct)
            preds = lt[0].argmax(-1).tolist()     # len k+1
            j = 0
            while j < len(proposal) and preds[j] == proposal[j] \
                    and proposal[j] not in stop \
                    and len(generated) + j + 1 < cfg.max_new_tokens:
                j += 1
            generated.extend(proposal[:j])
            nxt = preds[j]
            # rewind to the accepted prefix (truncate is a no-op when the
            # cache holds fewer tokens, e.g. the draft after a full accept)
            conf = len(ids) + len(generated)
            for c in ct:
                c.truncate(conf)
            for c in cd:
                c.truncate(conf)
        self.stats["tokens_generated"] += len(generated)
        self.stats["time_in_generate"] += time.perf_counter() - t0
        return generated

    def _speculative_sampling_loop(self, draft_model, cfg, draft_k, ids,
                                   ct, cd, nxt, stop) -> List[int]:
        """Acceptance-rejection speculative decoding (temperature > 0)."""
        generated: List[int] = []
        t0 = time.perf_counter()
        self.stats["prefill_tokens"] += len(ids)
        W = cfg.rep_window
        while len(generated) < cfg.max_new_tokens and nxt not in stop:
            generated.append(nxt)
            if len(generated) >= cfg.max_new_tokens:
                break
            confirmed = ids + generated
            missing = confirmed[cd[0].seq_len:]
            ld, _, _ = draft_model(
                torch.tensor([missing], dtype=torch.long,
                             device=self.device), kv_caches=cd)
            proposal: List[int] = []
            qs: List[torch.Tensor] = []
            for step in range(draft_k):
                q = self._filtered_probs(ld[0, -1].float(), cfg,
                                         (confirmed + proposal)[-W:])
                tok = int(q.multinomial(1))
                proposal.append(tok)
                qs.append(q)
                if step < draft_k - 1:
                    ld, _, _ = draft_model(
                        torch.tensor([[tok]], dtype=torch.long,
                                     device=self.device), kv_caches=cd)
            block = torch.tensor([[nxt] + proposal], dtype=torch.long,
                                 device=self.device)
            lt, _, _ = self.model(block, kv_caches=ct)
            j = 0
            nxt = None
            while j < len(proposal):
                if len(generated) + j + 1 >= cfg.max_new_tokens:
                    break
                p = self._filtered_probs(lt[0, j].float(), cfg,
                                         (confirmed + proposal[:j])[-W:])
                xj = proposal[j]
                ratio = float(p[xj]) / max(float(qs[j][xj]), 1e-20)
                if float(torch.rand(())) < ratio:
                    if xj in stop:          # accepted stop token ends it
                        nxt = xj
                        break
                    j += 1
                    continue
                resid = (p - qs[j]).clamp_min(0)
                tot = float(resid.sum())
                nxt = int((resid / tot).multinomial(1)) if tot > 0 \
                    else int(p.multinomial(1))
                break
            generated.extend(proposal[:j])
            if nxt is None:
                # no rejection token chosen: sample the next token from the
                # target's distribution at the accepted position (j ==
                # len(proposal) is the all-accepted bonus token)
                p = self._filtered_probs(lt[0, j].float(), cfg,
                                         (confirmed + proposal[:j])[-W:])
                nxt = int(p.multinomial(1))
            conf = len(ids) + len(generated)
            for c in ct:
                c.truncate(conf)
            for c in cd:
                c.truncate(conf)
        self.stats["tokens_generated"] += len(generated)
        self.stats["time_in_generate"] += time.perf_counter() - t0
        return generated

    @torch.no_grad()
    def generate_batch(self, prompts: List[List[int]],
                       configs=None) -> List[List[int]]:
        """Batched decoding over left-padded prompts with a shared KV cache.
        configs: one GenerationConfig for all rows, or a list (per-request
        sampling params — the serving batcher mixes requests freely). RoPE
        attention scores depend only on relative positions, so the uniform
        per-row pad shift is exact; pad slots are masked via the cache-wide
        attention mask (transformer.forward, width pos_offset+S)."""
        B = len(prompts)
        if configs is None:
            configs = GenerationConfig()
        cfgs = configs if isinstance(configs, list) else [configs] * B
        assert len(cfgs) == B
        t0 = time.perf_counter()
        self.model.eval()
        max_ctx = max(c.max_context for c in cfgs)
        prompts = [p[-max_ctx:] or [self.tokenizer.pad_token_id]
                   for p in prompts]
        Smax = max(len(p) for p in prompts)
        max_len = min(max_ctx, Smax + max(c.max_new_tokens for c in cfgs))
        pad = self.tokenizer.pad_token_id
        ids = torch.full((B, Smax), pad, dtype=torch.long)
        mask = torch.zeros(B, Smax, dtype=torch.long)
        for i, p in enumerate(prompts):
            ids[i, Smax - len(p):] = torch.tensor(p, dtype=torch.long)
            mask[i, Smax - len(p):] = 1
        ids = ids.to(self.device)
        mask = mask.to(self.device)
        caches = self.model.make_kv_caches(max_len=max_len)
        logits, _, _ = self.model(ids, attention_mask=mask, kv_caches=caches)
        last = logits[:, -1].float()

        stops = [set(c.stop_token_ids) | {self.tokenizer.eos_token_id}
                 for c in cfgs]
        outs: List[List[int]] = [[] for _ in range(B)]
        recent: List[List[int]] = [list(p) for p in prompts]
        finished = [False] * B
        for _ in range(max(c.max_new_tokens for c in cfgs)):
            next_ids = []
            for b in range(B):
                if finished[b]:
                    next_ids.append(pad)
                    continue
                nid = self._sample(last[b], cfgs[b],
                                   recent[b][-cfgs[b].rep_window:])
                if nid in stops[b] or len(outs[b]) >= cfgs[b].max_new_tokens:
                    finished[b] = True
                    next_ids.append(pad)
                    continue
                outs[b].append(nid)
                recent[b].append(nid)
                next_ids.append(nid)
                if len(outs[b]) >= cfgs[b].max_new_tokens:
                    finished[b] = True
            if all(finished) or caches[0].seq_len >= max_len:
                break
            step = torch.tensor(next_ids, dtype=torch.long,
                                device=self.device).view(B, 1)
            col = torch.tensor([0 if finished[b] else 1 for b in range(B)],
                               dtype=torch.long, device=self.device).view(B, 1)
            mask = torch.cat([mask, col], dim=1)
            logits, _, _ = self.model(step, attention_mask=mask,
                                      kv_caches=caches)
            last = logits[:, -1].float()
        self.stats["prefill_tokens"] += sum(len(p) for p in prompts)
        self.stats["tokens_generated"] += sum(len(o) for o in outs)
        self.stats["time_in_generate"] += time.perf_counter() - t0
        return outs

    def generate_text(self, prompt: str, config: Optional[GenerationConfig] = None,
                      stream: bool = False) -> str:
        ids = self.tokenizer.encode(prompt)
        out: List[str] = []

        def cb(tok_id):
            piece = self.tokenizer.decode([tok_id])
            out.append(piece)
            if stream:
                print(piece, end="", flush=True)

        self.generate(ids, config, stream_callback=cb)
        return "".join(out)

    # ------------------------------------------------------------------
    def _sample(self, logits: torch.Tensor, cfg: GenerationConfig,
                recent: List[int]) -> int:
        return sample_token(logits, cfg, recent)

    def _filtered_probs(self, logits: torch.Tensor, cfg: GenerationConfig,
                        recent: List[int]) -> torch.Tensor:
        """The sampling distribution _sample draws from (rep penalty +
        temperature + top-k + top-p), as an explicit probability vector —
        needed by speculative acceptance sampling. temperature > 0 only."""
        logits = logits.clone()
        if cfg.repetition_penalty != 1.0 and recent:
            idx = torch.tensor(sorted(set(recent)), device=logits.device)
            sel = logits[idx]
            logits[idx] = torch.where(sel > 0, sel / cfg.repetition_penalty,
                                      sel * cfg.repetition_penalty)
        logits = logits / cfg.temperature
        if cfg.top_k and cfg.top_k > 0:
            kth = torch.topk(logits, min(cfg.top_k, logits.numel())).values[-1]
            logits = logits.masked_fill(logits < kth, float("-inf"))
        if cfg.top_p < 1.0:
            sorted_logits, sorted_idx = torch.sort(logits, descending=True)
            probs = torch.softmax(sorted_logits, dim=-1)
            cum = torch.cumsum(probs, dim=-1)
            cut = cum - probs > cfg.top_p
            sorted_logits = sorted_logits.masked_fill(cut, float("-inf"))
            logits = torch.full_like(logits, float("-inf")).scatter(
                0, sorted_idx, sorted_logits)
        return torch.softmax(logits, dim=-1)

    def get_stats(self) -> Dict:
        t = self.stats["time_in_generate"]
        return dict(self.stats,
                    tokens_per_sec=self.stats["tokens_generated"] / t if t else 0.0)
