"""300-step 1.5B stability run: checks for memory growth, faults, loss
divergence over a longer horizon than the bench."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from midgpt_amd.config import load_config
from midgpt_amd.data import synthetic_batch
from midgpt_amd.train import build_engine
from midgpt_amd.utils.lr import warmup_cosine_lr

config = load_config("openwebtext_xl")
config.synthetic_data = True
config.remat = False
torch.manual_seed(0)
model, engine = build_engine(config, torch.device("cuda", 0))
mc = config.model_config
g = torch.Generator().manual_seed(7)
batches = [synthetic_batch(mc.vocab_size, mc.block_size, 32, 1, device="cuda",
                           generator=g) for _ in range(4)]
t0 = time.perf_counter()
mem0 = None
for it in range(1000):
    lr = warmup_cosine_lr(it, peak_lr=1e-3, warmup_steps=100,
                          decay_steps=25000, min_lr=1e-5)
    x, y = batches[it % 4]
    loss = model.loss(x[0], y[0])
    loss.backward()
    engine.microstep_end()
    engine.step(lr)
    if it == 20:
        torch.cuda.synchronize()
        mem0 = torch.cuda.memory_allocated()
    if it % 250 == 249:
        torch.cuda.synchronize()
        lv = float(loss.detach())
        mem = torch.cuda.memory_allocated()
        print(f"step {it}: loss {lv:.4f} mem {mem/2**30:.2f} GiB "
              f"({(time.perf_counter()-t0)/(it+1)*1000:.0f} ms/step)", flush=True)
        assert lv == lv and lv < 12.0, f"loss diverged: {lv}"
mem_end = torch.cuda.memory_allocated()
growth = (mem_end - mem0) / 2**20
print(f"memory growth after step 20: {growth:.1f} MiB")
assert growth < 256, f"memory leak suspected: {growth} MiB"
print("STABILITY OK")
