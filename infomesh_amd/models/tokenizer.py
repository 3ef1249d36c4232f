"""Deterministic hashing tokenizer (offline — no vocab files in image).

Token ids are stable hashes into the model's vocab, so random-init
models see a consistent id space across processes/shards. Word + CJK
character tokenization; specials follow BERT conventions.
"""
from __future__ import annotations

import re
from dataclasses import dataclass

from ..hashing import hash64

PAD, CLS, SEP, UNK = 0, 1, 2, 3
N_SPECIAL = 4

_TOKEN_RE = re.compile(
    r"[A-Za-z0-9_]+|[一-鿿㐀-䶿぀-ヿ가-힯]"
    r"|[^\sA-Za-z0-9_]", re.UNICODE)


def tokenize(text: str) -> list[str]:
    return _TOKEN_RE.findall(text.lower())


@dataclass(frozen=True)
class HashTokenizer:
    vocab_size: int = 30522

    def token_id(self, token: str) -> int:
        return N_SPECIAL + hash64(token) % (self.vocab_size - N_SPECIAL)

    def encode(self, text: str, max_len: int = 512,
               add_special: bool = True) -> list[int]:
        ids = [self.token_id(t) for t in tokenize(text)]
        if add_special:
            ids = [CLS] + ids[: max_len - 2] + [SEP]
        else:
            ids = ids[:max_len]
        return ids

    def encode_pair(self, a: str, b: str, max_len: int = 512) -> list[int]:
        """[CLS] a [SEP] b [SEP] with proportional truncation
        (cross-encoder query/passage input)."""
        ta = [self.token_id(t) for t in tokenize(a)]
        tb = [self.token_id(t) for t in tokenize(b)]
        budget = max_len - 3
        a_keep = min(len(ta), max(budget // 4, budget - len(tb)))
        b_keep = budget - a_keep
        return [CLS] + ta[:a_keep] + [SEP] + tb[:b_keep] + [SEP]

    def encode_batch(self, texts: list[str], max_len: int = 512
                     ) -> tuple[list[list[int]], list[int]]:
        """Returns padded id lists and true lengths."""
        encoded = [self.encode(t, max_len) for t in texts]
        lens = [len(e) for e in encoded]
        width = max(lens) if lens else 1
        return [e + [PAD] * (width - len(e)) for e in encoded], lens
