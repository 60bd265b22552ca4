"""Character-level Shakespeare -> train.bin/val.bin + meta.pkl.

Parity with reference data/shakespeare_char/prepare.py: download
input.txt, 65-char vocab, 90/10 split, uint16 bins, meta.pkl with
{vocab_size, stoi, itos}.

Usage: python -m midgpt_amd.data_prep.prepare_shakespeare [--out DIR]
"""
from __future__ import annotations

import argparse
import os
import pickle

import numpy as np

URL = ("https://raw.githubusercontent.com/karpathy/char-rnn/master/data/"
       "tinyshakespeare/input.txt")


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--out", default="data/shakespeare_char")
    args = p.parse_args()
    os.makedirs(args.out, exist_ok=True)
    path = os.path.join(args.out, "input.txt")
    if not os.path.exists(path):
        import requests
        with open(path, "w") as f:
            f.write(requests.get(URL, timeout=60).text)
    data = open(path).read()
    chars = sorted(set(data))
    stoi = {c: i for i, c in enumerate(chars)}
    itos = {i: c for i, c in enumerate(chars)}
    ids = np.array([stoi[c] for c in data], dtype=np.uint16)
    n = int(0.9 * len(ids))
    ids[:n].tofile(os.path.join(args.out, "train.bin"))
    ids[n:].tofile(os.path.join(args.out, "val.bin"))
    with open(os.path.join(args.out, "meta.pkl"), "wb") as f:
        pickle.dump({"vocab_size": len(chars), "stoi": stoi, "itos": itos}, f)
    print(f"vocab {len(chars)}, train {n}, val {len(ids) - n}")


if __name__ == "__main__":
    main()
