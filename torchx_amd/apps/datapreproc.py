"""Data-prep app: pack a text corpus into a binary token file for the
trainer (parity: torchx/examples/apps/datapreproc — a single-process data
transform launched via ``utils.python``; this one produces the packed
uint32 token file the reference trainer consumes).

Offline-friendly: with ``--synthetic N`` it generates N documents of
pseudo-text instead of reading an input corpus.
"""

from __future__ import annotations

import argparse
import random
import sys

import numpy as np


def tokenize(text: str, vocab_size: int) -> "np.ndarray":
    """Byte-pair-free toy tokenizer: stable hash of whitespace tokens."""
    ids = [hash(w) % (vocab_size - 2) + 2 for w in text.split()]
    return np.asarray(ids, dtype=np.uint32)


def main(argv=None) -> int:
    p = argparse.ArgumentParser(description="pack text into token shards")
    p.add_argument("--input", type=str, default=None,
                   help="fsspec path of a text file (one doc per line)")
    p.add_argument("--synthetic", type=int, default=0,
                   help="generate N synthetic documents instead")
    p.add_argument("--output", type=str, required=True,
                   help="fsspec path of the packed uint32 token file")
    p.add_argument("--vocab-size", type=int, default=128256)
    p.add_argument("--eos-id", type=int, default=1)
    args = p.parse_args(argv)

    import fsspec

    if args.synthetic:
        rng = random.Random(0)
        words = [f"w{i}" for i in range(1000)]
        docs = (
            " ".join(rng.choices(words, k=rng.randint(16, 128)))
            for _ in range(args.synthetic)
        )
    elif args.input:
        def _read():
            with fsspec.open(args.input, "r") as f:
                yield from (ln.strip() for ln in f if ln.strip())

        docs = _read()
    else:
        print("one of --input/--synthetic is required", file=sys.stderr)
        return 2

    total = 0
    with fsspec.open(args.output, "wb") as out:
        for doc in docs:
            ids = tokenize(doc, args.vocab_size)
            out.write(ids.tobytes())
            out.write(np.asarray([args.eos_id], dtype=np.uint32).tobytes())
            total += len(ids) + 1
    print(f"wrote {total} tokens to {args.output}", flush=True)
    return 0


if __name__ == "__main__":
    sys.exit(main())
