"""Checkpoint-under-load burn-in: train 1.5B with saves every 50 steps,
then restart the process state from the latest checkpoint and verify the
loss continues (not restarts)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from midgpt_amd.config import load_config
from midgpt_amd.data import synthetic_batch
from midgpt_amd.train import build_engine
from midgpt_amd.utils import checkpoint as ckpt
from midgpt_amd.utils.lr import warmup_cosine_lr

RUNDIR = "gpurun_out/burnin_run"
STEPS = int(sys.argv[1]) if len(sys.argv) > 1 else 200

config = load_config("openwebtext_xl")
config.synthetic_data = True
config.remat = False
torch.manual_seed(0)
model, engine = build_engine(config, torch.device("cuda", 0))
mc = config.model_config
g = torch.Generator().manual_seed(7)
batches = [synthetic_batch(mc.vocab_size, mc.block_size, 32, 1, device="cuda",
                           generator=g) for _ in range(4)]
mngr = ckpt.CheckpointManager(RUNDIR, save_interval=50)
state = ckpt.load_full_state(RUNDIR)
first = 0
if state is not None:
    engine.load_state_full(state["master"], state["m"], state["v"],
                           state["step_count"])
    first = state["step"] + 1
    print(f"resumed at {first}")
losses = []
for it in range(first, first + STEPS):
    lr = warmup_cosine_lr(it, peak_lr=1e-3, warmup_steps=100,
                          decay_steps=25000, min_lr=1e-5)
    x, y = batches[it % 4]
    loss = model.loss(x[0], y[0])
    loss.backward()
    engine.microstep_end()
    engine.step(lr)
    if mngr.should_save(it):
        mngr.save(it, engine)
    if it % 50 == 0:
        losses.append(float(loss.detach()))
        print(f"step {it}: loss {losses[-1]:.4f}", flush=True)
mngr.wait()
print(f"done at step {first + STEPS - 1}; latest ckpt "
      f"{ckpt.latest_step(RUNDIR)}")
if first > 0:
    assert losses[0] < 7.0, f"resume lost progress: first loss {losses[0]}"
    print("RESUME CONTINUITY OK")
