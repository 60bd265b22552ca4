"""Build a char-level dataset from text already present in the image
(no network — the reference's shakespeare download is not reachable).

Mirrors data_prep/prepare_shakespeare.py's contract (reference
data/shakespeare_char/prepare.py:12-61): train.bin/val.bin uint16 char
ids + meta.pkl {vocab_size, stoi, itos}, 90/10 split. Corpus: Python
stdlib sources — public text, ~11 MB, deterministic file order. Used for
the end-to-end loss-curve acceptance run (train a real model on real
bytes to convergence on the GPU).
"""
from __future__ import annotations

import argparse
import glob
import os
import pickle

import numpy as np


def build(out_dir: str, pattern: str, limit_mb: float):
    files = sorted(glob.glob(pattern, recursive=True))
    parts, total = [], 0
    for f in files:
        try:
            t = open(f, encoding="utf-8", errors="ignore").read()
        except OSError:
            continue
        parts.append(t)
        total += len(t)
        if total > limit_mb * 1e6:
            break
    data = "".join(parts)
    # printable-ASCII alphabet (+\n\t): stable small vocab like char-level
    # shakespeare (65); others map to a single OOV glyph
    keep = sorted(set(chr(c) for c in range(32, 127)) | {"\n", "\t"})
    alphabet = keep + ["\x00"]  # OOV marker last
    stoi = {c: i for i, c in enumerate(alphabet)}
    oov = len(alphabet) - 1
    ids = np.frombuffer(data.encode("ascii", errors="replace"), dtype=np.uint8)
    lut = np.full(256, oov, dtype=np.uint16)
    for c, i in stoi.items():
        lut[ord(c)] = i
    tok = lut[ids]
    n = len(tok)
    tr, va = tok[: int(n * 0.9)], tok[int(n * 0.9):]
    os.makedirs(out_dir, exist_ok=True)
    tr.tofile(os.path.join(out_dir, "train.bin"))
    va.tofile(os.path.join(out_dir, "val.bin"))
    with open(os.path.join(out_dir, "meta.pkl"), "wb") as f:
        pickle.dump({"vocab_size": len(alphabet), "stoi": stoi,
                     "itos": {i: c for c, i in stoi.items()}}, f)
    print(f"{n/1e6:.1f}M chars, vocab {len(alphabet)}, "
          f"train {len(tr)/1e6:.1f}M / val {len(va)/1e6:.1f}M -> {out_dir}")


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--out", default="data/local_char")
    p.add_argument("--pattern", default="/usr/lib/python3.10/**/*.py")
    p.add_argument("--limit-mb", type=float, default=20.0)
    args = p.parse_args()
    build(args.out, args.pattern, args.limit_mb)
