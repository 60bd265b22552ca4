"""Training runtime: train(config) — the MI355X-native equivalent of the
reference's src/train.py:127-225.

Flow parity (reference call stack):
  init distributed -> data load / per-process split -> engine (flat bf16
  weights + ZeRO fp32 master) -> restore latest checkpoint -> hot loop
  { eval every eval_interval | get_batch -> H2D -> G microsteps
    (fwd+bwd with per-block remat, bf16 compute, reduce-scatter per
    microstep) -> fused AdamW step -> log } -> final checkpoint wait.
"""
from __future__ import annotations

import json
import os
import time

import torch

from midgpt_amd.config import ExperimentConfig
from midgpt_amd.data import BatchLoader
from midgpt_amd.models.gpt import GPT, count_params
from midgpt_amd.parallel import dist as pdist
from midgpt_amd.parallel.engine import ShardedAdamW
from midgpt_amd.utils import checkpoint as ckpt
from midgpt_amd.utils.lr import warmup_cosine_lr
from midgpt_amd.utils.prefetch import DevicePrefetcher

DTYPES = {"float32": torch.float32, "bfloat16": torch.bfloat16}


def build_engine(config: ExperimentConfig, device: torch.device,
                 generator: torch.Generator | None = None
                 ) -> tuple[GPT, ShardedAdamW]:
    model = GPT(config.model_config, generator)
    model = model.to(device)
    model.remat = config.remat
    compute_dtype = DTYPES[config.compute_dtype]
    if device.type == "cpu" and compute_dtype == torch.bfloat16:
        # CPU path runs fp32 (bf16 CPU matmuls are slow and the CPU tier is
        # a correctness tier); GPU honors the config.
        compute_dtype = torch.float32
    engine = ShardedAdamW(
        model, compute_dtype=compute_dtype, zero=config.shard_model,
        beta1=config.beta1, beta2=config.beta2, eps=config.adam_eps,
        weight_decay=config.weight_decay, peak_lr=config.learning_rate,
        grad_clip=config.grad_clip, device=device)
    return model, engine


@torch.no_grad()
def evaluate(model: GPT, loader: BatchLoader, split: str, batch_size: int,
             device: torch.device, n_batches: int = 200) -> float:
    """Mean loss on ``split`` over fixed-size batches, all-reduced (mean)
    across ranks so every rank sees the global estimate (reference
    src/train.py:107-117,195-196 evaluates train AND val data)."""
    model.eval()
    total = 0.0
    for _ in range(n_batches):
        x, y = loader.batch(split, batch_size, 1)
        x, y = x[0].to(device), y[0].to(device)
        total += float(model.loss(x, y))
    model.train()
    mean = total / n_batches
    world = pdist.get_world_size()
    if world > 1:
        t = torch.tensor([mean], dtype=torch.float64,
                         device=device if device.type == "cuda" else "cpu")
        pdist.all_reduce_(t)
        mean = float(t[0]) / world
    return mean


def train(config: ExperimentConfig):
    rank, world, device = pdist.init_distributed()
    # identical seed on every rank for MODEL INIT (data-parallel replicas
    # must start from the same weights); re-seed per rank below for
    # data/dropout streams.
    base_seed = 1234 if config.seed is None else config.seed
    torch.manual_seed(base_seed)
    gen = torch.Generator().manual_seed(base_seed)

    model, engine = build_engine(config, device, gen)
    torch.manual_seed(base_seed + rank)
    if pdist.is_main():
        print(f"params: {count_params(model)/1e6:.1f}M  world={world} "
              f"device={device} zero={engine.zero}")

    assert config.batch_size % world == 0, "global batch must divide world size"
    local_bs = config.batch_size // world

    loader = BatchLoader(config.data_dir, config.model_config.vocab_size,
                         config.model_config.block_size, rank, world,
                         synthetic=config.synthetic_data, seed=config.seed)

    mngr = None
    first_step = 0
    if config.rundir:
        mngr = ckpt.CheckpointManager(
            config.rundir, 0 if config.debug else config.eval_interval)
        state = ckpt.load_full_state(config.rundir)
        if state is not None:
            engine.load_state_full(state["master"], state["m"], state["v"],
                                   state["step_count"])
            first_step = state["step"] + 1
            if pdist.is_main():
                print(f"resumed from step {state['step']}")

    eval_batches = 1 if config.debug else 200
    t0 = time.perf_counter()
    pbar = None
    if pdist.is_main():
        try:
            from tqdm import tqdm
            pbar = tqdm(total=config.max_steps, initial=first_step,
                        dynamic_ncols=True)
        except ImportError:
            pbar = None
    # one-batch-ahead pinned H2D prefetch on a copy stream (C5):
    # the host-side loader + hipMemcpyAsync of step it+1 overlap step it's
    # device compute; the compute stream only waits on the copy event.
    prefetch = DevicePrefetcher(device)
    pending = prefetch.start(*loader.batch("train", local_bs,
                                           config.g_accum_iters))
    for it in range(first_step, config.max_steps):
        if it % config.eval_interval == 0:
            tl = evaluate(model, loader, "train", local_bs, device, eval_batches)
            vl = evaluate(model, loader, "val", local_bs, device, eval_batches)
            if pdist.is_main():
                print(f"step {it}: loss/train {tl:.4f} loss/val {vl:.4f}")
            log_metrics(config, it, {"loss/train": tl, "loss/val": vl})

        lr = warmup_cosine_lr(it, peak_lr=config.learning_rate,
                              warmup_steps=config.warmup_steps,
                              decay_steps=config.lr_decay_steps,
                              min_lr=config.min_lr)
        x, y = prefetch.wait(pending)

        def one_step():
            nonlocal pending
            losses = []
            for g in range(config.g_accum_iters):
                loss = model.loss(x[g], y[g])
                loss.backward()
                engine.microstep_end()
                if g == 0 and it + 1 < config.max_steps:
                    # next step's batch: host gather + async H2D, hidden
                    # under the remaining microsteps' device work
                    pending = prefetch.start(
                        *loader.batch("train", local_bs, config.g_accum_iters))
                losses.append(loss.detach())
            engine.step(lr, config.g_accum_iters)
            # single host sync per step (keeps the device queue deep
            # through the microstep loop)
            return float(torch.stack(losses).mean())

        if config.debug and it == first_step and config.rundir:
            # reference parity: --debug traces step 0 (src/train.py:205-211);
            # here a torch.profiler (rocprof-sdk-backed) chrome trace.
            from torch.profiler import ProfilerActivity, profile
            acts = [ProfilerActivity.CPU]
            if device.type == "cuda":
                acts.append(ProfilerActivity.CUDA)
            with profile(activities=acts) as prof:
                loss_step = one_step()
                if device.type == "cuda":
                    torch.cuda.synchronize()
            prof.export_chrome_trace(
                os.path.join(config.rundir, f"trace_step0_rank{rank}.json"))
        else:
            loss_step = one_step()

        if it % 20 == 0:
            # global-batch mean at log points (C3: the reference's logged
            # loss is the cross-device mean; collective only when logging)
            if world > 1:
                t = torch.tensor([loss_step], dtype=torch.float64,
                                 device=device if device.type == "cuda"
                                 else "cpu")
                pdist.all_reduce_(t)
                loss_step = float(t[0]) / world
            log_metrics(config, it, {"loss/optimized": loss_step})
        if mngr is not None and mngr.should_save(it):
            mngr.save(it, engine)
        if pdist.is_main():
            dt = time.perf_counter() - t0
            thpt = (it - first_step + 1) * config.batch_size * \
                config.g_accum_iters / max(dt, 1e-9)
            if pbar is not None:
                pbar.update(1)
                pbar.set_postfix(loss=f"{loss_step:.4f}", lr=f"{lr:.2e}",
                                 thpt=f"{thpt:.1f} seq/s")
            elif it % 10 == 0:
                print(f"step {it}: loss {loss_step:.4f} lr {lr:.2e} "
                      f"thpt {thpt:.1f} seq/s", flush=True)
    if pbar is not None:
        pbar.close()
    if mngr is not None:
        mngr.save(config.max_steps - 1, engine)
        mngr.wait()


_WANDB = None


def log_metrics(config: ExperimentConfig, step: int, metrics: dict):
    """wandb if available and configured (proc 0 only), else no-op."""
    global _WANDB
    if not pdist.is_main():
        return
    if _WANDB is None:
        _WANDB = False
        if os.environ.get("WANDB_API_KEY"):
            try:
                import wandb
                run_id = None
                idfile = os.path.join(config.rundir, "wandb_id.txt")
                if config.rundir and os.path.exists(idfile):
                    run_id = open(idfile).read().strip()
                r = wandb.init(project="midgpt", id=run_id, resume="allow",
                               config=json.loads(config.to_json()))
                if config.rundir:
                    with open(idfile, "w") as f:
                        f.write(r.id)
                _WANDB = True
            except Exception:
                _WANDB = False
    if _WANDB:
        import wandb
        wandb.log(metrics, step=step)
