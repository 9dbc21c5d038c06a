"""Offline dataset preparation: local text -> vocab.bin + train.bin
(analogs of the reference's python/openwebtext.py + export_vocab.py,
reference python/openwebtext.py:1-40, export_vocab.py:5-20 — those pull
HF datasets + tiktoken; this environment has no network, so the same
artifacts are built from local text with an in-tree byte-level BPE).

    python tools/prepare_tokens.py build-vocab corpus1.txt ... \
        --out vocab.bin [--vocab-size 4096]
    python tools/prepare_tokens.py tokenize corpus1.txt ... \
        --vocab vocab.bin --out train.bin

Formats match the reference exactly: vocab.bin = <u32 count> then
(<u32 len><bytes>)*; train.bin = flat uint16 token ids, one EOT token
appended per document (data/tokenizer.py reads both).
"""

from __future__ import annotations

import argparse
import collections
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tnn_amd.data.tokenizer import Tokenizer


def train_bpe(texts, vocab_size: int) -> Tokenizer:
    """Byte-level BPE: 256 byte tokens + learned merges + <eot>."""
    tok = Tokenizer()
    tok.tokens = [bytes([i]) for i in range(256)]
    # word-split keeps merges inside whitespace-delimited chunks (the
    # gpt2 regex's effect, simplified): count words once, merge on the
    # unique-word histogram instead of the full stream
    words = collections.Counter()
    for t in texts:
        for w in t.encode("utf-8").split():
            words[bytes(w)] += 1
    seqs = {w: [bytes([b]) for b in w] for w in words}
    while len(tok.tokens) < vocab_size - 1:
        pairs = collections.Counter()
        for w, seq in seqs.items():
            c = words[w]
            for a, b in zip(seq, seq[1:]):
                pairs[(a, b)] += c
        if not pairs:
            break
        (a, b), n = pairs.most_common(1)[0]
        if n < 2:
            break
        merged = a + b
        tok.tokens.append(merged)
        for w, seq in seqs.items():
            out, i = [], 0
            while i < len(seq):
                if i + 1 < len(seq) and seq[i] == a and seq[i + 1] == b:
                    out.append(merged)
                    i += 2
                else:
                    out.append(seq[i])
                    i += 1
            seqs[w] = out
    tok.tokens.append(b"<|endoftext|>")
    tok._index = {t: i for i, t in enumerate(tok.tokens)}
    return tok


def main():
    p = argparse.ArgumentParser()
    sub = p.add_subparsers(dest="cmd", required=True)
    bv = sub.add_parser("build-vocab")
    bv.add_argument("files", nargs="+")
    bv.add_argument("--out", default="vocab.bin")
    bv.add_argument("--vocab-size", type=int, default=4096)
    tk = sub.add_parser("tokenize")
    tk.add_argument("files", nargs="+")
    tk.add_argument("--vocab", required=True)
    tk.add_argument("--out", default="train.bin")
    args = p.parse_args()

    if args.cmd == "build-vocab":
        texts = [open(f, encoding="utf-8", errors="replace").read()
                 for f in args.files]
        tok = train_bpe(texts, args.vocab_size)
        tok.save(args.out)
        print(f"vocab.bin: {tok.vocab_size} tokens -> {args.out}")
    else:
        tok = Tokenizer().load(args.vocab)
        eot = tok.vocab_size - 1
        n = 0
        with open(args.out, "wb") as f:
            for fn in args.files:
                ids = tok.encode(open(fn, encoding="utf-8",
                                      errors="replace").read())
                ids.append(eot)
                arr = np.asarray(ids, dtype=np.uint16)
                f.write(arr.tobytes())
                n += len(arr)
        print(f"{n} tokens -> {args.out} "
              f"(uint16, OpenWebTextLoader-compatible)")


if __name__ == "__main__":
    main()
