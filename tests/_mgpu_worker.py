"""torchrun worker for the multi-GPU RCCL loss-equivalence test.

Launched by tests/test_multigpu.py as
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N
      --master-addr 127.0.0.1 tests/_mgpu_worker.py
Each rank trains the tiny GPT for 3 steps at world=N (ZeRO over RCCL) in
fp32 (MIDGPT_FORCE_REF — this test targets the ENGINE + collectives, the
HIP kernels have their own 1-GPU numerics tests); rank 0 also runs the
single-process reference on the full global batch and asserts the
assembled master matches. Exit code 0 = pass on every rank.
"""
import os
import sys

os.environ["MIDGPT_FORCE_REF"] = "1"

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402
import torch.distributed as dist  # noqa: E402

from midgpt_amd.config import GPTConfig  # noqa: E402
from midgpt_amd.models.gpt import GPT  # noqa: E402
from midgpt_amd.parallel import dist as pdist  # noqa: E402
from midgpt_amd.parallel.engine import ShardedAdamW  # noqa: E402

TINY = GPTConfig(block_size=16, vocab_size=37, n_layer=2, n_head=2,
                 n_embd=32, dropout=0.0)


def make_batch(world):
    g = torch.Generator().manual_seed(42)
    x = torch.randint(0, 37, (4 * world, 16), generator=g)
    y = torch.randint(0, 37, (4 * world, 16), generator=g)
    return x, y


def train(device, rank, world, steps=3):
    torch.manual_seed(0)
    model = GPT(TINY).to(device)
    engine = ShardedAdamW(model, compute_dtype=torch.float32, zero=True,
                          device=device)
    x, y = make_batch(world)
    xs = x[rank * 4:(rank + 1) * 4].to(device)
    ys = y[rank * 4:(rank + 1) * 4].to(device)
    losses = []
    for _ in range(steps):
        loss = model.loss(xs, ys)
        loss.backward()
        engine.microstep_end()
        engine.step(1e-3)
        losses.append(float(loss))
    return engine, losses


def main():
    rank, world, device = pdist.init_distributed()
    assert world > 1, "run under torchrun with nproc > 1"
    engine, losses = train(device, rank, world)
    full = torch.zeros(engine.padded, device=device)
    engine._scatter_shard(engine.master, full)
    dist.all_reduce(full)
    full = full[:engine.total].cpu()

    if rank == 0:
        # single-process reference on the full global batch: build a
        # world-1 engine without touching the live process group by
        # monkeypatching the size helpers for the constructor only.
        import midgpt_amd.parallel.dist as pd
        orig_rank, orig_ws = pd.get_rank, pd.get_world_size
        pd.get_rank = lambda: 0
        pd.get_world_size = lambda: 1
        try:
            torch.manual_seed(0)
            ref_model = GPT(TINY).to(device)
            ref_engine = ShardedAdamW(ref_model, compute_dtype=torch.float32,
                                      zero=True, device=device)
        finally:
            pd.get_rank, pd.get_world_size = orig_rank, orig_ws
        x, y = make_batch(world)
        x, y = x.to(device), y.to(device)
        for _ in range(3):
            ref_model.loss(x, y).backward()
            ref_engine.microstep_end()
            ref_engine.step(1e-3)
        ref = ref_engine.master[:ref_engine.total].cpu()
        err = float((full - ref).abs().max())
        assert err < 5e-5, f"world={world} master mismatch: max abs err {err}"
        print(f"MULTIGPU-EQUIV-OK world={world} max_abs_err={err:.2e} "
              f"losses={losses}")
    dist.barrier()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
