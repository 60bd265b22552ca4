"""Loss-curve acceptance run (VERDICT round-1 item 7): train the
shakespeare_char-shaped model end-to-end on REAL local text (char-level,
scripts/make_local_char_data.py) on the GPU, printing the loss/eval
trajectory, then sample from the trained model.

Run (GPU box):
  python scripts/make_local_char_data.py
  python scripts/loss_curve_run.py --steps 3000 --out gpurun_out/losscurve
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from midgpt_amd.config import load_config  # noqa: E402
from midgpt_amd.train import train  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=3000)
    p.add_argument("--out", default="gpurun_out/losscurve")
    p.add_argument("--data", default="data/local_char")
    p.add_argument("--config", default="shakespeare_char")
    p.add_argument("--batch", type=int, default=None)
    p.add_argument("--debug", action="store_true",
                   help="1-batch evals (CPU smoke)")
    args = p.parse_args()

    config = load_config(args.config)
    config.data_dir = args.data
    if args.batch:
        config.batch_size = args.batch
        config.g_accum_iters = 1
    config.model_config.vocab_size = 98  # local_char alphabet
    config.max_steps = args.steps
    config.eval_interval = max(1, args.steps // (12 if args.steps >= 1000 else 6))
    config.rundir = args.out
    config.seed = 1234
    config.debug = args.debug
    os.makedirs(args.out, exist_ok=True)
    with open(os.path.join(args.out, "config.json"), "w") as f:
        f.write(config.to_json())
    train(config)

    # sample from the trained model (reference sample.py parity path)
    import pickle

    from midgpt_amd.generate import generate
    from midgpt_amd.models.gpt import GPT
    from midgpt_amd.parallel.engine import ShardedAdamW
    from midgpt_amd.utils import checkpoint as ckpt
    meta = pickle.load(open(os.path.join(args.data, "meta.pkl"), "rb"))
    dev = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    model = GPT(config.model_config).to(dev)
    engine = ShardedAdamW(model, compute_dtype=torch.float32, zero=False,
                          device=dev)
    state = ckpt.load_full_state(args.out)
    assert state is not None, "no checkpoint written"
    engine.load_state_full(state["master"], state["m"], state["v"],
                           state["step_count"])
    start = "def main():\n"
    idx = torch.tensor([[meta["stoi"][c] for c in start]], device=dev)
    out = generate(model, idx, 400, temperature=0.8,
                   generator=torch.Generator().manual_seed(0))
    text = "".join(meta["itos"][int(t)] for t in out[0])
    print("=== SAMPLE ===")
    print(text)
    with open(os.path.join(args.out, "sample.txt"), "w") as f:
        f.write(text)


if __name__ == "__main__":
    main()
