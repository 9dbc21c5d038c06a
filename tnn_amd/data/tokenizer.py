"""GPT-2 vocab.bin tokenizer (reference include/tokenizer/tokenizer.hpp:11-68).

File format: u32 token count, then per token {u32 length, bytes}.
Decode-only, like the reference (encoding is done offline with tiktoken).
We additionally provide a greedy longest-match encoder so inference demos
work without tiktoken.
"""

from __future__ import annotations

import struct
from typing import List


class Tokenizer:
    def __init__(self):
        self.tokens: List[bytes] = []
        self._index = {}

    def load(self, path: str) -> "Tokenizer":
        with open(path, "rb") as f:
            (count,) = struct.unpack("<I", f.read(4))
            for _ in range(count):
                (ln,) = struct.unpack("<I", f.read(4))
                self.tokens.append(f.read(ln))
        self._index = {t: i for i, t in enumerate(self.tokens)}
        return self

    def save(self, path: str):
        with open(path, "wb") as f:
            f.write(struct.pack("<I", len(self.tokens)))
            for t in self.tokens:
                f.write(struct.pack("<I", len(t)))
                f.write(t)

    @property
    def vocab_size(self) -> int:
        return len(self.tokens)

    def decode(self, ids) -> str:
        return b"".join(self.tokens[i] for i in ids).decode("utf-8", errors="replace")

    def encode(self, text: str) -> List[int]:
        """Greedy longest-match (sufficient for demos; not BPE-exact)."""
        data = text.encode("utf-8")
        out, i = [], 0
        max_len = max((len(t) for t in self.tokens), default=1)
        while i < len(data):
            for ln in range(min(max_len, len(data) - i), 0, -1):
                tid = self._index.get(data[i:i + ln])
                if tid is not None:
                    out.append(tid)
                    i += ln
                    break
            else:
                i += 1  # unknown byte: skip
        return out
