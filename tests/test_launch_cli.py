"""End-to-end CLI tests: launch.py and sample.py as subprocesses (CPU)."""
import json
import os
import subprocess
import sys


REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

TINY_CFG = '''
from midgpt_amd.config import ExperimentConfig, GPTConfig
config = ExperimentConfig(
    rundir="", data_dir="", learning_rate=1e-3, batch_size=4,
    warmup_steps=2, min_lr=1e-4, lr_decay_steps=50, max_steps=4,
    beta2=0.95, weight_decay=1e-4, eval_interval=2,
    param_dtype="float32", compute_dtype="float32", g_accum_iters=1,
    shard_model=False, synthetic_data=True, seed=11,
    model_config=GPTConfig(block_size=16, vocab_size=37, n_layer=2,
                           n_head=2, n_embd=32, dropout=0.0))
'''


def test_launch_and_sample_cli(tmp_path):
    # install a temporary config preset
    cfg_path = os.path.join(REPO, "midgpt_amd", "configs", "_tiny_ci.py")
    with open(cfg_path, "w") as f:
        f.write(TINY_CFG)
    rundir = str(tmp_path / "run")
    try:
        r = subprocess.run(
            [sys.executable, "launch.py", "--config=_tiny_ci",
             f"--rundir={rundir}"],
            cwd=REPO, capture_output=True, text=True, timeout=300)
        assert r.returncode == 0, r.stderr[-2000:]
        # rundir contract: config.json + a committed checkpoint
        cfg = json.load(open(os.path.join(rundir, "config.json")))
        assert cfg["model_config"]["vocab_size"] == 37
        ckpts = [d for d in os.listdir(rundir) if d.startswith("ckpt_")]
        assert ckpts, os.listdir(rundir)
        # resume: second launch starts past the saved step and exits quickly
        r2 = subprocess.run(
            [sys.executable, "launch.py", "--config=_tiny_ci",
             f"--rundir={rundir}"],
            cwd=REPO, capture_output=True, text=True, timeout=300)
        assert r2.returncode == 0, r2.stderr[-2000:]
        assert "resumed from step" in (r2.stdout + r2.stderr)
        # sample.py needs a tokenizer: char meta or tiktoken (offline) —
        # point data_dir at a meta.pkl we fabricate
        import pickle
        meta_dir = tmp_path / "data"
        meta_dir.mkdir()
        chars = [chr(ord('a') + i % 26) for i in range(37)]
        with open(meta_dir / "meta.pkl", "wb") as f:
            pickle.dump({"vocab_size": 37,
                         "stoi": {c: i for i, c in enumerate(chars)},
                         "itos": {i: c for i, c in enumerate(chars)}}, f)
        cfg["data_dir"] = str(meta_dir)
        with open(os.path.join(rundir, "config.json"), "w") as f:
            json.dump(cfg, f)
        r3 = subprocess.run(
            [sys.executable, "sample.py", f"--ckpt_dir={rundir}",
             "--start=ab", "--num_samples=1", "--max_new_tokens=8",
             "--seed=0"],
            cwd=REPO, capture_output=True, text=True, timeout=300)
        assert r3.returncode == 0, r3.stderr[-2000:]
        assert "---------------" in r3.stdout
    finally:
        os.remove(cfg_path)


def test_serve_endpoint(tmp_path):
    """serve.py: health + generation over a tiny trained rundir."""
    import json
    import pickle

    import numpy as np

    from midgpt_amd.train import train
    from tests.test_model_train import tiny_config

    # char "dataset" metadata so the server picks the char tokenizer
    data_dir = tmp_path / "data"
    data_dir.mkdir()
    alphabet = [chr(97 + i) for i in range(37)]
    with open(data_dir / "meta.pkl", "wb") as f:
        pickle.dump({"vocab_size": 37,
                     "stoi": {c: i for i, c in enumerate(alphabet)},
                     "itos": {i: c for i, c in enumerate(alphabet)}}, f)
    cfg = tiny_config(tmp_path / "run", max_steps=2, eval_interval=10,
                      data_dir=str(data_dir))
    train(cfg)
    # launch.py normally freezes config.json; do it here for the loader
    with open(tmp_path / "run" / "config.json", "w") as f:
        f.write(cfg.to_json())

    from starlette.testclient import TestClient

    from serve import build_app
    app = build_app(str(tmp_path / "run"), device="cpu")
    client = TestClient(app)
    r = client.get("/healthz")
    assert r.status_code == 200 and r.json()["status"] == "ok"
    r = client.post("/generate", json={"prompt": "ab", "max_new_tokens": 5,
                                       "num_samples": 2, "seed": 0,
                                       "temperature": 1.0})
    assert r.status_code == 200
    samples = r.json()["samples"]
    assert len(samples) == 2
    assert all(s.startswith("ab") and len(s) == 7 for s in samples)


def test_distributed_flag_requires_torchrun():
    """--distributed without WORLD_SIZE must fail with a clear error."""
    import subprocess
    import sys
    env = dict(os.environ)
    env.pop("WORLD_SIZE", None)
    r = subprocess.run(
        [sys.executable, "launch.py", "--config", "shakespeare_char",
         "--distributed"],
        capture_output=True, text=True, env=env,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert r.returncode != 0
    assert "torchrun" in r.stderr


def test_broadcast_str_single_process():
    from midgpt_amd.parallel.dist import broadcast_str
    assert broadcast_str("runs/abc") == "runs/abc"


def test_prefetcher_shape_change():
    import torch

    from midgpt_amd.utils.prefetch import DevicePrefetcher
    pre = DevicePrefetcher(torch.device("cpu"))
    for shape in [(2, 4), (3, 5), (2, 4)]:
        x = torch.ones(shape)
        xd, yd = pre.wait(pre.start(x, x + 1))
        assert xd.shape == torch.Size(shape)
