"""50-step 7B@4096 no-remat soak: memory stability at 224 GB."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from midgpt_amd.config import load_config
from midgpt_amd.data import synthetic_batch
from midgpt_amd.train import build_engine
from midgpt_amd.utils.lr import warmup_cosine_lr

config = load_config("llama7b_4k")
config.synthetic_data = True
config.remat = False
torch.manual_seed(0)
model, engine = build_engine(config, torch.device("cuda", 0))
mc = config.model_config
g = torch.Generator().manual_seed(7)
batches = [synthetic_batch(mc.vocab_size, mc.block_size, 4, 1, device="cuda",
                           generator=g) for _ in range(2)]
mem0 = None
t0 = time.perf_counter()
for it in range(50):
    lr = warmup_cosine_lr(it, peak_lr=3e-4, warmup_steps=2000,
                          decay_steps=25000, min_lr=3e-5)
    for gi in range(4):
        x, y = batches[(it * 4 + gi) % 2]
        loss = model.loss(x[0], y[0])
        loss.backward()
        engine.microstep_end()
    engine.step(lr, 4)
    if it == 5:
        torch.cuda.synchronize()
        mem0 = torch.cuda.max_memory_allocated()
    if it % 10 == 9:
        torch.cuda.synchronize()
        lv = float(loss.detach())
        print(f"step {it}: loss {lv:.4f} max_mem {torch.cuda.max_memory_allocated()/2**30:.1f} GiB "
              f"({(time.perf_counter()-t0)/(it+1)*1000:.0f} ms/step)", flush=True)
        assert lv == lv and lv < 12.5, lv
growth = (torch.cuda.max_memory_allocated() - mem0) / 2**30
print(f"max-mem growth after step 5: {growth:.2f} GiB")
assert growth < 8, growth
print("SOAK7B OK")
