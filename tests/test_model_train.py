"""Model semantics, engine, checkpoint, data, schedule — CPU tests."""
import math
import os

import numpy as np
import pytest
import torch

from midgpt_amd.config import ExperimentConfig, GPTConfig, load_config
from midgpt_amd.data import BatchLoader, get_batch, synthetic_batch
from midgpt_amd.models.gpt import GPT, count_params
from midgpt_amd.parallel.engine import ShardedAdamW
from midgpt_amd.utils import checkpoint as ckpt
from midgpt_amd.utils.lr import warmup_cosine_lr

TINY = GPTConfig(block_size=16, vocab_size=37, n_layer=2, n_head=2,
                 n_embd=32, dropout=0.0)


def tiny_config(tmpdir="", **kw):
    base = dict(rundir=str(tmpdir), data_dir="", learning_rate=1e-3,
                batch_size=4, warmup_steps=2, min_lr=1e-4, lr_decay_steps=50,
                max_steps=10, beta2=0.95, weight_decay=1e-4, eval_interval=100,
                param_dtype="float32", compute_dtype="float32",
                g_accum_iters=1, shard_model=False, model_config=TINY,
                synthetic_data=True, seed=7)
    base.update(kw)
    return ExperimentConfig(**base)


def test_config_presets_load():
    for name in ["shakespeare_char", "openwebtext", "openwebtext_mh",
                 "openwebtext_xl", "llama7b_4k"]:
        c = load_config(name)
        assert c.model_config.n_embd % c.model_config.n_head == 0


def test_config_json_roundtrip():
    c = load_config("openwebtext_xl")
    c2 = ExperimentConfig.from_json(c.to_json())
    assert c2 == c


def test_model_forward_shapes_and_loss():
    torch.manual_seed(0)
    model = GPT(TINY)
    x = torch.randint(0, 37, (3, 16))
    logits = model(x)
    assert logits.shape == (3, 16, 37)
    y = torch.randint(0, 37, (3, 16))
    loss = model.loss(x, y)
    # random init ~ roughly uniform over vocab (tied head adds variance)
    assert abs(float(loss.detach()) - math.log(37)) < 1.5


def test_lm_head_tied_at_init_untied_after():
    model = GPT(TINY)
    assert torch.equal(model.wte, model.lm_head.weight)
    assert model.wte.data_ptr() != model.lm_head.weight.data_ptr()


def test_count_params_excludes_head():
    model = GPT(TINY)
    total = sum(p.numel() for p in model.parameters())
    assert count_params(model) == total - 37 * 32


def test_remat_matches_no_remat():
    torch.manual_seed(1)
    model = GPT(TINY)
    x = torch.randint(0, 37, (2, 16))
    y = torch.randint(0, 37, (2, 16))
    model.remat = False
    l1 = model.loss(x, y)
    l1.backward()
    g1 = {n: p.grad.clone() for n, p in model.named_parameters()}
    model.zero_grad()
    model.remat = True
    l2 = model.loss(x, y)
    l2.backward()
    assert torch.allclose(l1, l2)
    for n, p in model.named_parameters():
        assert torch.allclose(g1[n], p.grad, atol=1e-6), n


def test_engine_training_reduces_loss():
    torch.manual_seed(2)
    model = GPT(TINY)
    engine = ShardedAdamW(model, compute_dtype=torch.float32, zero=False,
                          beta2=0.95, weight_decay=1e-4, peak_lr=1e-2)
    x, y = synthetic_batch(37, 16, 8, 1,
                           generator=torch.Generator().manual_seed(3))
    losses = []
    for it in range(30):
        loss = model.loss(x[0], y[0])
        loss.backward()
        engine.microstep_end()
        engine.step(1e-2)
        losses.append(float(loss))
    assert losses[-1] < losses[0] - 0.5, losses


def test_engine_flat_views_alias_params():
    model = GPT(TINY)
    engine = ShardedAdamW(model, compute_dtype=torch.float32, zero=False)
    p = next(model.parameters())
    assert p.data_ptr() >= engine.flat_w.data_ptr()
    engine.flat_w.zero_()
    assert float(p.abs().sum()) == 0.0


def test_grad_accumulation_equivalence():
    """G microsteps of batch B == one step of batch G*B (loss mean scaling)."""
    torch.manual_seed(4)
    x, y = synthetic_batch(37, 16, 8, 1,
                           generator=torch.Generator().manual_seed(5))

    def run(g_accum):
        torch.manual_seed(6)
        model = GPT(TINY)
        engine = ShardedAdamW(model, compute_dtype=torch.float32, zero=False)
        if g_accum == 1:
            loss = model.loss(x[0], y[0])
            loss.backward()
            engine.microstep_end()
        else:
            for g in range(g_accum):
                sl = slice(g * 4, (g + 1) * 4)
                loss = model.loss(x[0][sl], y[0][sl])
                loss.backward()
                engine.microstep_end()
        engine.step(1e-3, g_accum)
        return engine.master.clone()

    m1 = run(1)
    m2 = run(2)
    assert torch.allclose(m1, m2, atol=1e-6)


def test_lr_schedule_matches_optax_shape():
    peak, mn = 1e-3, 1e-5
    assert warmup_cosine_lr(0, peak_lr=peak, warmup_steps=100,
                            decay_steps=1000, min_lr=mn) == 0.0
    assert warmup_cosine_lr(50, peak_lr=peak, warmup_steps=100,
                            decay_steps=1000, min_lr=mn) == pytest.approx(peak / 2)
    assert warmup_cosine_lr(100, peak_lr=peak, warmup_steps=100,
                            decay_steps=1000, min_lr=mn) == pytest.approx(peak)
    mid = warmup_cosine_lr(550, peak_lr=peak, warmup_steps=100,
                           decay_steps=1000, min_lr=mn)
    assert mid == pytest.approx((peak + mn) / 2)
    assert warmup_cosine_lr(5000, peak_lr=peak, warmup_steps=100,
                            decay_steps=1000, min_lr=mn) == mn


def test_get_batch_shapes_and_shift(tmp_path):
    data = np.arange(1000, dtype=np.uint16)
    rng = np.random.default_rng(0)
    x, y = get_batch(data, 8, 4, 2, rng)
    assert x.shape == (2, 4, 8) and y.shape == (2, 4, 8)
    assert torch.equal(x[0, 0, 1:], y[0, 0, :-1])


def test_batchloader_synthetic():
    bl = BatchLoader("/nonexistent", 37, 16, synthetic=True, seed=1)
    x, y = bl.batch("train", 4, 2)
    assert x.shape == (2, 4, 16)
    assert x.max() < 37


def test_checkpoint_roundtrip(tmp_path):
    torch.manual_seed(7)
    model = GPT(TINY)
    engine = ShardedAdamW(model, compute_dtype=torch.float32, zero=False)
    x, y = synthetic_batch(37, 16, 4, 1,
                           generator=torch.Generator().manual_seed(8))
    for _ in range(3):
        model.loss(x[0], y[0]).backward()
        engine.microstep_end()
        engine.step(1e-3)
    mngr = ckpt.CheckpointManager(str(tmp_path), save_interval=1)
    mngr.save(3, engine)
    mngr.wait()
    assert ckpt.latest_step(str(tmp_path)) == 3
    state = ckpt.load_full_state(str(tmp_path))
    model2 = GPT(TINY)
    engine2 = ShardedAdamW(model2, compute_dtype=torch.float32, zero=False)
    engine2.load_state_full(state["master"], state["m"], state["v"],
                            state["step_count"])
    assert torch.allclose(engine2.master, engine.master)
    assert torch.allclose(engine2.m, engine.m)
    assert engine2.step_count == engine.step_count
    # identical forward after restore
    l1 = model.loss(x[0], y[0])
    l2 = model2.loss(x[0], y[0])
    assert torch.allclose(l1, l2, atol=1e-6)


def test_checkpoint_max_to_keep(tmp_path):
    model = GPT(TINY)
    engine = ShardedAdamW(model, compute_dtype=torch.float32, zero=False)
    mngr = ckpt.CheckpointManager(str(tmp_path), save_interval=1, max_to_keep=1)
    mngr.save(1, engine)
    mngr.wait()
    mngr.save(2, engine)
    mngr.wait()
    dirs = [d for d in os.listdir(tmp_path) if d.startswith("ckpt_")]
    assert dirs == ["ckpt_0000002"]


def test_generate_prompt_longer_than_block():
    """A prompt with T0 > block_size must condition on its trailing window
    (reference sample.py crops to block_size) instead of raising."""
    torch.manual_seed(12)
    from midgpt_amd.generate import generate
    model = GPT(TINY)  # block_size 16
    idx = torch.randint(0, 37, (1, 25))
    out = generate(model, idx, 5, temperature=0.0)
    assert out.shape == (1, 30)
    assert torch.equal(out[:, :25], idx)
    cur = idx.clone()
    for _ in range(5):
        logits = model(cur[:, -16:])
        cur = torch.cat([cur, logits[:, -1].argmax(-1)[:, None]], dim=1)
    assert torch.equal(out, cur)


def test_evaluate_reads_requested_split(tmp_path):
    """loss/train must come from train.bin, loss/val from val.bin
    (reference src/train.py:195-196 evaluates both datasets)."""
    from midgpt_amd.train import evaluate
    # distinct constant token streams per split
    np.full(4000, 3, dtype=np.uint16).tofile(tmp_path / "train.bin")
    np.full(4000, 11, dtype=np.uint16).tofile(tmp_path / "val.bin")
    bl = BatchLoader(str(tmp_path), 37, 16, seed=0)

    class TokenMeanModel:
        def eval(self):
            pass

        def train(self):
            pass

        def loss(self, x, y):
            return x.float().mean()

    m = TokenMeanModel()
    dev = torch.device("cpu")
    assert evaluate(m, bl, "train", 2, dev, n_batches=3) == pytest.approx(3.0)
    assert evaluate(m, bl, "val", 2, dev, n_batches=3) == pytest.approx(11.0)


def test_train_entrypoint_runs(tmp_path):
    from midgpt_amd.train import train
    cfg = tiny_config(tmp_path, max_steps=3, eval_interval=2, debug=True)
    train(cfg)


def test_generate_runs():
    from midgpt_amd.generate import generate
    torch.manual_seed(9)
    model = GPT(TINY)
    idx = torch.randint(0, 37, (2, 5))
    out = generate(model, idx, 10, temperature=1.0,
                   generator=torch.Generator().manual_seed(0))
    assert out.shape == (2, 15)
    assert out.max() < 37


def test_generate_cache_matches_full_forward():
    """KV-cache decode must match argmax decode via full forwards."""
    torch.manual_seed(10)
    from midgpt_amd.generate import generate
    model = GPT(TINY)
    idx = torch.randint(0, 37, (1, 4))
    out = generate(model, idx, 6, temperature=0.0)
    # reference: greedy with full forward each step
    cur = idx.clone()
    for _ in range(6):
        logits = model(cur[:, -16:])
        nxt = logits[:, -1].argmax(-1)
        cur = torch.cat([cur, nxt[:, None]], dim=1)
    assert torch.equal(out, cur)


def test_generate_beyond_block_size():
    """Generation crossing block_size exercises the cropped-window fallback
    (reference behavior: recompute the full window per token)."""
    torch.manual_seed(11)
    from midgpt_amd.generate import generate
    model = GPT(TINY)  # block_size 16
    idx = torch.randint(0, 37, (1, 10))
    out = generate(model, idx, 12, temperature=0.0)
    assert out.shape == (1, 22)
    # greedy reference: full forward on the cropped window each step
    cur = idx.clone()
    for _ in range(12):
        logits = model(cur[:, -16:])
        cur = torch.cat([cur, logits[:, -1].argmax(-1)[:, None]], dim=1)
    assert torch.equal(out, cur)


def test_device_prefetcher_cpu_passthrough():
    from midgpt_amd.utils.prefetch import DevicePrefetcher
    pre = DevicePrefetcher(torch.device("cpu"))
    x = torch.arange(12).reshape(3, 4)
    y = x + 1
    h = pre.start(x, y)
    xd, yd = pre.wait(h)
    assert torch.equal(xd, x) and torch.equal(yd, y)


def test_engine_manifest_matches_named_parameters():
    """Checkpoint manifest order/offsets == named_parameters traversal
    (the reference's flat-leaves contract, src/train.py:215)."""
    model = GPT(TINY)
    engine = ShardedAdamW(model, compute_dtype=torch.float32, zero=False)
    man = engine.named_param_manifest()
    off = 0
    for entry, (name, p) in zip(man, model.named_parameters()):
        assert entry["name"] == name
        assert entry["offset"] == off
        assert entry["numel"] == p.numel()
        assert tuple(entry["shape"]) == tuple(p.shape)
        off += p.numel()
    assert off == engine.total


def test_config_from_json_ignores_unknown_fields():
    """Forward compatibility: a rundir config.json written by a newer
    version (extra fields) must still load."""
    c = load_config("openwebtext_xl")
    import json as _json
    d = _json.loads(c.to_json())
    d["future_field"] = 123
    d["model_config"]["future_model_field"] = "x"
    c2 = ExperimentConfig.from_json(_json.dumps(d))
    assert c2.model_config.n_embd == c.model_config.n_embd


def test_zero_flag_collapses_at_world_one():
    """shard_model=True at world 1 must behave as the replicated engine
    (full master, no padding surprises)."""
    model = GPT(TINY)
    engine = ShardedAdamW(model, compute_dtype=torch.float32, zero=True)
    assert not engine.zero  # collapses: zero needs world > 1
    assert engine.master.numel() == engine.total


def test_prepare_shakespeare_offline(tmp_path, monkeypatch):
    """The char data-prep pipeline runs offline when input.txt exists
    (reference data/shakespeare_char/prepare.py contract)."""
    import pickle
    import sys
    (tmp_path / "input.txt").write_text("hello world\n" * 200)
    monkeypatch.setattr(sys, "argv", ["prep", "--out", str(tmp_path)])
    from midgpt_amd.data_prep import prepare_shakespeare
    prepare_shakespeare.main()
    meta = pickle.load(open(tmp_path / "meta.pkl", "rb"))
    train = np.fromfile(tmp_path / "train.bin", dtype=np.uint16)
    val = np.fromfile(tmp_path / "val.bin", dtype=np.uint16)
    assert meta["vocab_size"] == len(set("hello world\n"))
    assert len(train) + len(val) == 12 * 200
    # decode round-trip
    text = "".join(meta["itos"][i] for i in train[:12])
    assert text == "hello world\n"
