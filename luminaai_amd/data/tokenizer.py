"""Conversation tokenizer with ChatML-style special tokens.

API-compatible rebuild of the reference ConversationTokenizer
(/root/reference/Src/Main_Scripts/core/tokenizer.py:36-616): tiktoken
cl100k_base base vocabulary when available, with 13 special tokens appended
after the base vocab and the total padded to a multiple of 128. This
environment has no tiktoken wheel and no network, so a byte-level BPE-free
fallback (256 byte tokens) provides the same API; vocab ids and special-token
layout stay identical in shape so checkpoints keep their geometry.
"""

from __future__ import annotations

import threading
from functools import lru_cache
from typing import Dict, List, Optional

try:
    import tiktoken
    _HAS_TIKTOKEN = True
except ImportError:
    tiktoken = None
    _HAS_TIKTOKEN = False

SPECIAL_TOKEN_NAMES = [
    "<|im_start|>", "<|im_end|>", "<|user|>", "<|assistant|>", "<|system|>",
    "<|human|>", "<|ai|>", "<|bot|>", "<|thought|>", "<|tool|>", "<|error|>",
    "<|truncated|>", "<|endoftext|>",
]

ROLE_TOKENS = {
    "user": "<|user|>", "assistant": "<|assistant|>", "system": "<|system|>",
    "human": "<|human|>", "ai": "<|ai|>", "bot": "<|bot|>",
}


class _ByteBackend:
    """Offline fallback: UTF-8 bytes as tokens (ids 0..255)."""

    n_vocab = 256

    def encode(self, text: str) -> List[int]:
        return list(text.encode("utf-8", errors="replace"))

    def decode(self, ids: List[int]) -> str:
        return bytes(i for i in ids if 0 <= i < 256).decode("utf-8", errors="replace")


class ConversationTokenizer:
    """Thread-safe tokenizer with conversation encoding + loss weighting."""

    def __init__(self, base: str = "cl100k_base", max_length: int = 2048,
                 assistant_loss_weight: float = 2.0):
        self._lock = threading.RLock()
        if _HAS_TIKTOKEN:
            self._backend = tiktoken.get_encoding(base)
            self.backend_name = base
        else:
            self._backend = _ByteBackend()
            self.backend_name = "byte_fallback"
        self.base_vocab_size = self._backend.n_vocab
        self.special_tokens: Dict[str, int] = {
            name: self.base_vocab_size + i for i, name in enumerate(SPECIAL_TOKEN_NAMES)
        }
        self._reverse_special = {v: k for k, v in self.special_tokens.items()}
        self.pad_token_id = 0
        self.eos_token_id = self.special_tokens["<|im_end|>"]
        raw = self.base_vocab_size + len(self.special_tokens)
        self.vocab_size = ((raw + 127) // 128) * 128  # pad to x128
        self.max_length = max_length
        self.assistant_loss_weight = assistant_loss_weight
        self._cache: Dict[str, List[int]] = {}
        self._stats = {"encoded_texts": 0, "encoded_tokens": 0, "cache_hits": 0}

    # ------------------------------------------------------------------
    def encode(self, text: str, use_cache: bool = True) -> List[int]:
        if use_cache and len(text) < 256:
            with self._lock:
                hit = self._cache.get(text)
                if hit is not None:
                    self._stats["cache_hits"] += 1
                    return list(hit)
        ids = self._backend.encode(text)
        with self._lock:
            self._stats["encoded_texts"] += 1
            self._stats["encoded_tokens"] += len(ids)
            if use_cache and len(text) < 256 and len(self._cache) < 8192:
                self._cache[text] = list(ids)
        return ids

    def decode(self, ids: List[int]) -> str:
        out: List[str] = []
        run: List[int] = []
        for i in ids:
            if i in self._reverse_special:
                if run:
                    out.append(self._backend.decode(run))
                    run = []
                out.append(self._reverse_special[i])
            elif i < self.base_vocab_size:
                run.append(i)
        if run:
            out.append(self._backend.decode(run))
        return "".join(out)

    # ------------------------------------------------------------------
    def encode_message(self, role: str, content: str) -> List[int]:
        """<|im_start|><role>content<|im_end|> (reference tokenizer.py:251-414)."""
        role_tok = ROLE_TOKENS.get(role, ROLE_TOKENS["user"])
        return ([self.special_tokens["<|im_start|>"], self.special_tokens[role_tok]]
                + self.encode(content)
                + [self.special_tokens["<|im_end|>"]])

    def encode_conversation(self, conversation, max_length: Optional[int] = None,
                            return_loss_weights: bool = False):
        """conversation: {"messages": [{"role", "content"}, ...]} or a list of
        messages. Returns ids (+ per-token loss weights: 0 on specials/pads,
        assistant_loss_weight on assistant content, 1.0 otherwise)."""
        msgs = conversation.get("messages", conversation) \
            if isinstance(conversation, dict) else conversation
        max_length = max_length or self.max_length
        ids: List[int] = []
        weights: List[float] = []
        for m in msgs:
            role = m.get("role", "user")
            content = m.get("content", m.get("text", ""))
            tok = self.encode_message(role, content)
            w_content = self.assistant_loss_weight if role in ("assistant", "ai", "bot") else 1.0
            w = [0.0, 0.0] + [w_content] * (len(tok) - 3) + [0.0]
            ids.extend(tok)
            weights.extend(w)
        if len(ids) > max_length:
            ids = ids[:max_length - 1] + [self.special_tokens["<|truncated|>"]]
            weights = weights[:max_length - 1] + [0.0]
        if return_loss_weights:
            return ids, weights
        return ids

    def batch_encode(self, texts: List[str]) -> List[List[int]]:
        return [self.encode(t) for t in texts]

    def get_stats(self) -> Dict:
        with self._lock:
            return dict(self._stats, cache_size=len(self._cache),
                        vocab_size=self.vocab_size, backend=self.backend_name)
