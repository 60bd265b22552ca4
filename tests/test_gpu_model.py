"""GPU end-to-end: model forward/backward on HIP kernels vs CPU reference;
one engine step; extension-presence guard."""

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from midgpt_amd import ops
    assert ops.have_ext()

from midgpt_amd.config import ExperimentConfig, GPTConfig
from midgpt_amd.data import synthetic_batch
from midgpt_amd.models.gpt import GPT
from midgpt_amd.parallel.engine import ShardedAdamW

SMALL = GPTConfig(block_size=128, vocab_size=512, n_layer=2, n_head=2,
                  n_embd=128, dropout=0.0)


def test_model_gpu_loss_matches_cpu_reference():
    torch.manual_seed(0)
    model = GPT(SMALL)
    x = torch.randint(0, 512, (2, 128))
    y = torch.randint(0, 512, (2, 128))
    loss_cpu = float(model.loss(x, y).detach())
    gm = GPT(SMALL)
    gm.load_state_dict(model.state_dict())
    gm = gm.to("cuda").to(torch.bfloat16)
    gm.rope_sin = gm.rope_sin.float()
    gm.rope_cos = gm.rope_cos.float()
    loss_gpu = float(gm.loss(x.cuda(), y.cuda()).detach())
    assert abs(loss_gpu - loss_cpu) < 0.05 * abs(loss_cpu) + 0.05, \
        (loss_gpu, loss_cpu)


def test_model_gpu_grads_match_cpu_reference():
    torch.manual_seed(1)
    model = GPT(SMALL)
    x = torch.randint(0, 512, (2, 128))
    y = torch.randint(0, 512, (2, 128))
    model.loss(x, y).backward()
    cpu_grads = {n: p.grad.clone() for n, p in model.named_parameters()}
    gm = GPT(SMALL)
    gm.load_state_dict({k: v for k, v in model.state_dict().items()})
    gm = gm.to("cuda").to(torch.bfloat16)
    gm.rope_sin = gm.rope_sin.float()
    gm.rope_cos = gm.rope_cos.float()
    gm.loss(x.cuda(), y.cuda()).backward()
    for n, p in gm.named_parameters():
        g1 = cpu_grads[n].float()
        g2 = p.grad.detach().cpu().float()
        rel = (g1 - g2).norm() / (g1.norm() + 1e-9)
        assert rel < 0.15, (n, float(rel))


def test_engine_step_gpu_trains():
    torch.manual_seed(2)
    model = GPT(SMALL).to("cuda")
    model.remat = True
    engine = ShardedAdamW(model, compute_dtype=torch.bfloat16, zero=False,
                          peak_lr=1e-2)
    x, y = synthetic_batch(512, 128, 8, 1, device="cuda")
    losses = []
    for _ in range(20):
        loss = model.loss(x[0], y[0])
        loss.backward()
        engine.microstep_end()
        engine.step(5e-3)
        losses.append(float(loss.detach()))
    assert losses[-1] < losses[0] - 0.3, losses


def test_native_extension_is_the_executing_path():
    """The dispatch layer must raise rather than silently fall back when a
    GPU tensor hits an op with the extension masked out."""
    import midgpt_amd.ops as O
    saved = O._C
    try:
        O._C = None
        x = torch.randn(4, 64, device="cuda", dtype=torch.bfloat16)
        with pytest.raises(RuntimeError):
            O.rmsnorm(x, None, 1e-6)
    finally:
        O._C = saved


def test_checkpoint_roundtrip_gpu(tmp_path):
    from midgpt_amd.utils import checkpoint as ckpt
    torch.manual_seed(3)
    model = GPT(SMALL).to("cuda")
    engine = ShardedAdamW(model, compute_dtype=torch.bfloat16, zero=False)
    x, y = synthetic_batch(512, 128, 4, 1, device="cuda")
    for _ in range(2):
        model.loss(x[0], y[0]).backward()
        engine.microstep_end()
        engine.step(1e-3)
    mngr = ckpt.CheckpointManager(str(tmp_path), save_interval=1)
    mngr.save(2, engine)
    mngr.wait()
    state = ckpt.load_full_state(str(tmp_path))
    model2 = GPT(SMALL).to("cuda")
    engine2 = ShardedAdamW(model2, compute_dtype=torch.bfloat16, zero=False)
    engine2.load_state_full(state["master"], state["m"], state["v"],
                            state["step_count"])
    assert torch.allclose(engine2.master.cpu(), engine.master.cpu())
    l1 = float(model.loss(x[0], y[0]).detach())
    l2 = float(model2.loss(x[0], y[0]).detach())
    assert abs(l1 - l2) < 1e-3


def test_generate_gpu():
    from midgpt_amd.generate import generate
    torch.manual_seed(4)
    model = GPT(SMALL).to("cuda").to(torch.bfloat16)
    model.rope_sin = model.rope_sin.float()
    model.rope_cos = model.rope_cos.float()
    idx = torch.randint(0, 512, (2, 8), device="cuda")
    out = generate(model, idx, 16, temperature=0.8)
    assert out.shape == (2, 24)
    assert int(out.max()) < 512


def test_train_entrypoint_gpu(tmp_path):
    from midgpt_amd.config import ExperimentConfig
    from midgpt_amd.train import train
    cfg = ExperimentConfig(
        rundir=str(tmp_path), data_dir="", learning_rate=1e-3, batch_size=4,
        warmup_steps=2, min_lr=1e-4, lr_decay_steps=50, max_steps=3,
        beta2=0.95, weight_decay=1e-4, eval_interval=2,
        param_dtype="float32", compute_dtype="bfloat16", g_accum_iters=2,
        shard_model=False, model_config=SMALL, debug=True,
        synthetic_data=True, seed=5)
    train(cfg)
    assert (tmp_path / "trace_step0_rank0.json").exists()


def _two_proc_gpu_worker(rank, init_file, out_q):
    import torch.distributed as dist
    import os
    os.environ["WORLD_SIZE"] = "2"
    os.environ["RANK"] = str(rank)
    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=2)
    torch.manual_seed(0)
    model = GPT(SMALL).to("cuda:0")  # both ranks share the one GPU
    engine = ShardedAdamW(model, compute_dtype=torch.bfloat16, zero=True)
    g = torch.Generator().manual_seed(42)
    x, y = synthetic_batch(512, 128, 8, 1, generator=g)
    xs = x[0, rank * 4:(rank + 1) * 4].cuda()
    ys = y[0, rank * 4:(rank + 1) * 4].cuda()
    for _ in range(3):
        model.loss(xs, ys).backward()
        engine.microstep_end()
        engine.step(1e-3)
    torch.cuda.synchronize()
    full = torch.zeros(engine.padded)
    engine._scatter_shard(engine.master.cpu(), full)
    dist.all_reduce(full)  # assemble the full master across shards (gloo)
    out_q.put((rank, full[:engine.total].numpy().copy()))
    dist.destroy_process_group()


def test_two_process_one_gpu_zero_engine(tmp_path):
    """2 ranks sharing cuda:0 over gloo: exercises the CUDA engine's
    overlap pipeline + collectives + ZeRO sharding end-to-end in a real
    multi-process setting (RCCL needs 2 devices; gloo does not)."""
    import torch.multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    init_file = str(tmp_path / "pg_gpu")
    procs = [ctx.Process(target=_two_proc_gpu_worker, args=(r, init_file, q))
             for r in range(2)]
    for p in procs:
        p.start()
    res = {}
    for _ in range(2):
        r, full = q.get()
        res[r] = torch.from_numpy(full)
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0
    # single-process reference on the full batch of 8
    torch.manual_seed(0)
    model = GPT(SMALL).to("cuda:0")
    engine = ShardedAdamW(model, compute_dtype=torch.bfloat16, zero=False)
    g = torch.Generator().manual_seed(42)
    x, y = synthetic_batch(512, 128, 8, 1, generator=g)
    xs, ys = x[0].cuda(), y[0].cuda()
    for _ in range(3):
        model.loss(xs, ys).backward()
        engine.microstep_end()
        engine.step(1e-3)
    ref = engine.master.cpu()
    for r, full in res.items():
        rel = (full - ref[:full.numel()]).norm() / ref[:full.numel()].norm()
        assert rel < 5e-3, (r, float(rel))


def test_training_trajectory_tracks_cpu_reference():
    """20 steps of the tiny model: the GPU bf16 HIP path's loss curve must
    track the CPU fp32 reference path (systematic kernel-numerics drift
    beyond per-op tolerances would separate the curves)."""
    def run(device, dtype):
        torch.manual_seed(0)
        model = GPT(SMALL).to(device)
        engine = ShardedAdamW(model, compute_dtype=dtype, zero=False,
                              peak_lr=1e-2)
        g = torch.Generator().manual_seed(1)
        x, y = synthetic_batch(512, 128, 8, 1, generator=g)
        xs, ys = x[0].to(device), y[0].to(device)
        losses = []
        for _ in range(20):
            loss = model.loss(xs, ys)
            loss.backward()
            engine.microstep_end()
            engine.step(3e-3)
            losses.append(float(loss.detach()))
        return losses

    gpu = run("cuda", torch.bfloat16)
    cpu = run("cpu", torch.float32)
    diffs = [abs(a - b) for a, b in zip(gpu, cpu)]
    assert max(diffs[:10]) < 0.08, (gpu[:10], cpu[:10])
    assert abs(gpu[-1] - cpu[-1]) < 0.3, (gpu[-1], cpu[-1])


def test_train_from_real_bin_with_async_checkpoint(tmp_path):
    """Full train() on GPU from actual uint16 .bin files (pinned-prefetch
    H2D path + async copy-stream checkpoint snapshot), then resume."""
    import numpy as np
    from midgpt_amd.config import ExperimentConfig
    from midgpt_amd.train import train
    from midgpt_amd.utils import checkpoint as ckpt
    rng = np.random.default_rng(0)
    rng.integers(0, 512, 200_000).astype(np.uint16).tofile(
        tmp_path / "train.bin")
    rng.integers(0, 512, 20_000).astype(np.uint16).tofile(
        tmp_path / "val.bin")
    cfg = ExperimentConfig(
        rundir=str(tmp_path / "run"), data_dir=str(tmp_path),
        learning_rate=1e-3, batch_size=8, warmup_steps=2, min_lr=1e-4,
        lr_decay_steps=50, max_steps=6, beta2=0.95, weight_decay=1e-4,
        eval_interval=3, param_dtype="float32", compute_dtype="bfloat16",
        g_accum_iters=2, shard_model=False, model_config=SMALL, seed=3)
    train(cfg)
    step = ckpt.latest_step(cfg.rundir)
    assert step == 5, step
    state = ckpt.load_full_state(cfg.rundir)
    assert torch.isfinite(state["master"]).all()
    # resume path: continues from the checkpoint without error
    cfg.max_steps = 8
    train(cfg)
    assert ckpt.latest_step(cfg.rundir) == 7
