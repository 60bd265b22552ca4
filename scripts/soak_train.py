"""Short real train() soak on the GPU: exercises the round-2 loop
machinery end-to-end at 124M scale — pinned prefetch, backward-hook
buckets (world=1 pipeline), async copy-stream checkpointing, eval on
both splits, resume — on synthetic data."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from midgpt_amd.config import load_config
from midgpt_amd.train import train
from midgpt_amd.utils import checkpoint as ckpt

cfg = load_config("openwebtext")
cfg.synthetic_data = True
cfg.batch_size = 32
cfg.g_accum_iters = 2
cfg.max_steps = 40
cfg.eval_interval = 20
cfg.rundir = "gpurun_out/soak124"
cfg.seed = 11
train(cfg)
assert ckpt.latest_step(cfg.rundir) == 39, ckpt.latest_step(cfg.rundir)
cfg.max_steps = 45
train(cfg)  # resume path
assert ckpt.latest_step(cfg.rundir) == 44
print("SOAK OK")
