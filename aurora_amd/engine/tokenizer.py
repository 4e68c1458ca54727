"""Native tokenizers.

The reference counts/encodes tokens with rented tiktoken models
(reference: server/chat/backend/agent/utils/llm_usage_tracker.py:6,
utils/chat_context_manager.py:13).  Here tokenization is in-process:

* ByteTokenizer — reversible byte-level vocab (256 bytes + specials).  The
  default in this checkpoint-free environment: every model is random-init,
  so a learned merge table adds nothing, and byte-level makes constrained
  JSON decoding exact.
* HFTokenizer — wraps a `tokenizers` JSON file when a real checkpoint's
  tokenizer is available (production path).
"""
from __future__ import annotations

from typing import List, Optional, Sequence


class Tokenizer:
    bos_id: int
    eos_id: int
    pad_id: int

    def encode(self, text: str, add_bos: bool = False) -> List[int]:
        raise NotImplementedError

    def decode(self, ids: Sequence[int]) -> str:
        raise NotImplementedError

    def token_bytes(self, token_id: int) -> bytes:
        """Raw bytes a token contributes to the output (specials -> b'')."""
        raise NotImplementedError

    @property
    def vocab_size(self) -> int:
        raise NotImplementedError

    def count(self, text: str) -> int:
        return len(self.encode(text))

    def first_byte_buckets(self) -> dict:
        """byte -> list of token ids whose byte expansion starts with it.
        Lets grammar-constrained sampling gather candidates for a known
        legal next byte instead of scanning the whole vocabulary."""
        cache = getattr(self, "_fb_buckets", None)
        if cache is None:
            cache = {}
            for t in range(self.vocab_size):
                b = self.token_bytes(t)
                if b:
                    cache.setdefault(b[0], []).append(t)
            self._fb_buckets = cache
        return cache


class ByteTokenizer(Tokenizer):
    BOS = 256
    EOS = 257
    PAD = 258
    N_SPECIAL = 3

    def __init__(self):
        self.bos_id = self.BOS
        self.eos_id = self.EOS
        self.pad_id = self.PAD

    @property
    def vocab_size(self) -> int:
        return 256 + self.N_SPECIAL

    def encode(self, text: str, add_bos: bool = False) -> List[int]:
        ids = list(text.encode("utf-8"))
        return ([self.BOS] + ids) if add_bos else ids

    def decode(self, ids: Sequence[int]) -> str:
        return bytes(i for i in ids if i < 256).decode("utf-8", errors="replace")

    def token_bytes(self, token_id: int) -> bytes:
        return bytes([token_id]) if token_id < 256 else b""


class HFTokenizer(Tokenizer):
    def __init__(self, path: str):
        from tokenizers import Tokenizer as _HT  # offline wheelhouse package

        self._t = _HT.from_file(path)
        self.bos_id = self._t.token_to_id("<|begin_of_text|>") or self._t.token_to_id("<s>") or 0
        self.eos_id = self._t.token_to_id("<|end_of_text|>") or self._t.token_to_id("</s>") or 1
        self.pad_id = self.eos_id
        self._byte_cache: dict[int, bytes] = {}

    @property
    def vocab_size(self) -> int:
        return self._t.get_vocab_size()

    def encode(self, text: str, add_bos: bool = False) -> List[int]:
        ids = self._t.encode(text, add_special_tokens=False).ids
        return ([self.bos_id] + ids) if add_bos else ids

    def decode(self, ids: Sequence[int]) -> str:
        return self._t.decode(list(ids), skip_special_tokens=True)

    def token_bytes(self, token_id: int) -> bytes:
        b = self._byte_cache.get(token_id)
        if b is None:
            b = self._t.decode([token_id], skip_special_tokens=True).encode("utf-8")
            self._byte_cache[token_id] = b
        return b


def get_tokenizer(path: Optional[str] = None) -> Tokenizer:
    if path:
        return HFTokenizer(path)
    return ByteTokenizer()
