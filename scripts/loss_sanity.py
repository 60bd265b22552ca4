"""120-step training sanity on synthetic data: loss must fall well below
the uniform-vocab baseline (validates the full HIP fwd/bwd/optimizer path)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from midgpt_amd.config import load_config
from midgpt_amd.data import synthetic_batch
from midgpt_amd.train import build_engine
from midgpt_amd.utils.lr import warmup_cosine_lr

config = load_config("openwebtext")
config.synthetic_data = True
torch.manual_seed(0)
model, engine = build_engine(config, torch.device("cuda", 0))
mc = config.model_config
g = torch.Generator().manual_seed(7)
x, y = synthetic_batch(mc.vocab_size, mc.block_size, 32, 1, device="cuda", generator=g)
losses = []
t0 = time.perf_counter()
for it in range(120):
    lr = warmup_cosine_lr(it, peak_lr=1e-3, warmup_steps=20, decay_steps=1000, min_lr=1e-5)
    loss = model.loss(x[0], y[0])
    loss.backward()
    engine.microstep_end()
    engine.step(lr)
    losses.append(float(loss.detach()))
print(f"steps/s {120/(time.perf_counter()-t0):.2f}")
print("loss[0:3]", [round(v,3) for v in losses[:3]])
print("loss[-3:]", [round(v,3) for v in losses[-3:]])
assert losses[-1] < losses[0] - 3.0, "loss did not fall enough (overfit memorization expected)"
print("LOSS SANITY OK")
