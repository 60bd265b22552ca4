"""OpenWebText -> train.bin/val.bin uint16 token streams.

Parity with reference data/openwebtext/prepare.py: HF openwebtext dataset,
0.05% val split (seed 2357), GPT-2 BPE via tiktoken ``encode_ordinary`` +
EOT appended per document, concatenated into uint16 memmap bins written in
1024 batches. Requires network + the ``datasets``/``tiktoken`` packages
(one-time host-side job; the training benchmark uses synthetic data).

Usage: python -m midgpt_amd.data_prep.prepare_openwebtext [--out DIR]
"""
from __future__ import annotations

import argparse
import os

import numpy as np


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--out", default="data/openwebtext")
    p.add_argument("--num_proc", type=int, default=8)
    args = p.parse_args()

    import tiktoken
    from datasets import load_dataset

    enc = tiktoken.get_encoding("gpt2")
    dataset = load_dataset("openwebtext", num_proc=args.num_proc)
    split = dataset["train"].train_test_split(test_size=0.0005, seed=2357,
                                              shuffle=True)
    split["val"] = split.pop("test")

    def process(example):
        ids = enc.encode_ordinary(example["text"])
        ids.append(enc.eot_token)
        return {"ids": ids, "len": len(ids)}

    tokenized = split.map(process, remove_columns=["text"],
                          desc="tokenizing", num_proc=args.num_proc)

    os.makedirs(args.out, exist_ok=True)
    for name, dset in tokenized.items():
        arr_len = int(np.sum(dset["len"], dtype=np.uint64))
        path = os.path.join(args.out, f"{name}.bin")
        arr = np.memmap(path, dtype=np.uint16, mode="w+", shape=(arr_len,))
        total_batches = 1024
        idx = 0
        for b in range(total_batches):
            batch = dset.shard(num_shards=total_batches, index=b,
                               contiguous=True).with_format("numpy")
            batch_ids = np.concatenate(batch["ids"])
            arr[idx:idx + len(batch_ids)] = batch_ids
            idx += len(batch_ids)
        arr.flush()
        print(f"{path}: {arr_len} tokens")


if __name__ == "__main__":
    main()
