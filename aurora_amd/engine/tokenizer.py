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

        def first_id(names: Sequence[str], default: int) -> int:
            # explicit None checks: `or` would skip a special at id 0
            for n in names:
                i = self._t.token_to_id(n)
                if i is not None:
                    return i
            return default

        self.bos_id = first_id(("<|begin_of_text|>", "<s>", "<bos>"), 0)
        self.eos_id = first_id(
            ("<|end_of_text|>", "<|eot_id|>", "</s>", "<eos>",
             "<|endoftext|>", "<|im_end|>"), 1)
        self.pad_id = self.eos_id
        self._byte_cache: dict[int, bytes] = {}
        # ids whose tokens are control/special markers (no text contribution)
        self._special_ids = set()
        added = getattr(self._t, "get_added_tokens_decoder", None)
        if added is not None:
            for i, at in added().items():
                if getattr(at, "special", True):
                    self._special_ids.add(i)
        else:  # older tokenizers: fall back to the declared specials
            for n in ("<s>", "</s>", "<unk>", "<pad>", "<|begin_of_text|>",
                      "<|end_of_text|>", "<|eot_id|>", "<|endoftext|>"):
                i = self._t.token_to_id(n)
                if i is not None:
                    self._special_ids.add(i)

    @property
    def vocab_size(self) -> int:
        return self._t.get_vocab_size()

    def encode(self, text: str, add_bos: bool = False) -> List[int]:
        ids = self._t.encode(text, add_special_tokens=False).ids
        return ([self.bos_id] + ids) if add_bos else ids

    def decode(self, ids: Sequence[int]) -> str:
        return self._t.decode(list(ids), skip_special_tokens=True)

    @staticmethod
    def _bytelevel_table() -> dict:
        """GPT-2/ByteLevel char -> raw byte map (the printable-unicode
        encoding byte-level BPE vocabs store their merges in)."""
        bs = (list(range(33, 127)) + list(range(161, 173))
              + list(range(174, 256)))
        cs = list(bs)
        n = 0
        for b in range(256):
            if b not in bs:
                bs.append(b)
                cs.append(256 + n)
                n += 1
        return {chr(c): b for b, c in zip(bs, cs)}

    _BL_TABLE: Optional[dict] = None

    def token_bytes(self, token_id: int) -> bytes:
        """Raw output bytes of one token, derived from the VOCAB string —
        decoding a lone token would mangle codepoints whose UTF-8 bytes
        are split across neighbouring tokens."""
        b = self._byte_cache.get(token_id)
        if b is None:
            s = self._t.id_to_token(token_id)
            if s is None or token_id in self._special_ids:
                b = b""  # special / control token contributes no text
            elif s.startswith("<0x") and s.endswith(">") and len(s) == 6:
                b = bytes([int(s[3:5], 16)])  # sentencepiece byte token
            else:
                if HFTokenizer._BL_TABLE is None:
                    HFTokenizer._BL_TABLE = self._bytelevel_table()
                tbl = HFTokenizer._BL_TABLE
                if all(ch in tbl for ch in s):
                    b = bytes(tbl[ch] for ch in s)  # byte-level BPE vocab
                else:
                    # plain-text vocab (sentencepiece "▁word" style)
                    b = s.replace("▁", " ").encode("utf-8")
            self._byte_cache[token_id] = b
        return b


def get_tokenizer(path: Optional[str] = None) -> Tokenizer:
    if path:
        return HFTokenizer(path)
    return ByteTokenizer()
