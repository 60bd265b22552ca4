"""Data-loader smoke (reference scripts/test_data.py parity): batch
assembly from a memmap bin must sustain training-feeding rates."""
import time

import numpy as np

from midgpt_amd.data import get_batch


def test_get_batch_throughput(tmp_path):
    data = np.random.randint(0, 50304, size=2_000_000).astype(np.uint16)
    p = tmp_path / "train.bin"
    data.tofile(p)
    arr = np.memmap(p, dtype=np.uint16, mode="r")
    rng = np.random.default_rng(0)
    t0 = time.perf_counter()
    n = 20
    for _ in range(n):
        x, y = get_batch(arr, 1024, 16, 1, rng)
    dt = time.perf_counter() - t0
    rate = n * 16 * 1024 / dt  # tokens/s assembled
    # generous bound: even slow CI boxes assemble >2M tok/s; a 1.5B step
    # consumes ~0.1M tokens/s per GPU
    assert rate > 2e6, f"batch assembly too slow: {rate:.0f} tok/s"
    assert x.shape == (1, 16, 1024)
