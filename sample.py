"""Sampling / inference — contract parity with the reference sample.py:
``python sample.py --ckpt_dir=RUNDIR [--start=STR] [--num_samples N]
[--max_new_tokens M] [--temperature T]``

Loads config.json + the latest checkpoint from the rundir, rebuilds the
model, and generates. Improvement over the reference (which re-runs a full
forward per new token, sample.py:68-95): an incremental KV cache decode.
"""
from __future__ import annotations

import argparse
import os
import pickle

import torch

from midgpt_amd.config import ExperimentConfig
from midgpt_amd.generate import generate
from midgpt_amd.models.gpt import GPT
from midgpt_amd.utils import checkpoint as ckpt


def load_model(ckpt_dir: str, device: torch.device) -> tuple[GPT, ExperimentConfig]:
    with open(os.path.join(ckpt_dir, "config.json")) as f:
        config = ExperimentConfig.from_json(f.read())
    model = GPT(config.model_config)
    state = ckpt.load_full_state(ckpt_dir)
    if state is not None:
        master = state["master"]
        sd = {}
        for pm in state["manifest"]["params"]:
            n, shape = pm["numel"], pm["shape"]
            sd[pm["name"]] = master[pm["offset"]:pm["offset"] + n].view(shape)
        # strict=False only because rope sin/cos buffers are non-persistent
        missing, unexpected = model.load_state_dict(sd, strict=False)
        assert not unexpected, f"unexpected checkpoint params: {unexpected}"
        print(f"loaded checkpoint step {state['step']}")
    else:
        print("no checkpoint found; using random init")
    model = model.to(device)
    if device.type == "cuda":
        model = model.to(torch.bfloat16)
        # rope tables stay fp32
        model.rope_sin = model.rope_sin.float()
        model.rope_cos = model.rope_cos.float()
    model.eval()
    return model, config


def load_model_and_tokenizer(ckpt_dir: str, device):
    """Model + encode/decode fns for a rundir (tokenizer: char meta.pkl
    if the config's data_dir has one, else tiktoken GPT-2 — reference
    sample.py:143-159)."""
    device = torch.device(device) if not isinstance(device, torch.device) \
        else device
    model, config = load_model(ckpt_dir, device)
    meta_path = os.path.join(config.data_dir, "meta.pkl")
    if os.path.exists(meta_path):
        with open(meta_path, "rb") as f:
            meta = pickle.load(f)
        encode = lambda s: [meta["stoi"][c] for c in s]  # noqa: E731
        decode = lambda t: "".join(meta["itos"][i] for i in t)  # noqa: E731
    else:
        try:
            import tiktoken
        except ImportError as e:
            raise SystemExit(
                "no meta.pkl in the config's data_dir and tiktoken is not "
                "installed (offline image): token-id I/O only — rerun with "
                "a char-level dataset or install tiktoken") from e
        enc = tiktoken.get_encoding("gpt2")
        encode = lambda s: enc.encode(s, allowed_special={"<|endoftext|>"})  # noqa: E731
        decode = enc.decode
    return model, encode, decode, config


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--ckpt_dir", required=True)
    p.add_argument("--start", default="\n")
    p.add_argument("--num_samples", type=int, default=3)
    p.add_argument("--max_new_tokens", type=int, default=200)
    p.add_argument("--temperature", type=float, default=0.8)
    p.add_argument("--seed", type=int, default=None)
    args = p.parse_args()

    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    model, encode, decode, config = load_model_and_tokenizer(args.ckpt_dir,
                                                             device)

    gen = torch.Generator(device="cpu")
    if args.seed is not None:
        gen.manual_seed(args.seed)
    prompt = torch.tensor(encode(args.start), dtype=torch.int64, device=device)
    prompt = prompt.unsqueeze(0).expand(args.num_samples, -1)
    out = generate(model, prompt, args.max_new_tokens,
                   temperature=args.temperature, generator=gen)
    for i in range(args.num_samples):
        print(decode(out[i].tolist()))
        print("---------------")


if __name__ == "__main__":
    main()
