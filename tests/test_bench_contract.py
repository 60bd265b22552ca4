"""The driver depends on bench.py's exact output contract: one JSON line
from rank 0 with specific keys. Run it end-to-end (CPU, tiny config) and
validate the schema."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    r = subprocess.run(
        [sys.executable, "bench.py", "--gpus", "1", "--steps", "2",
         "--warmup", "1", "--config", "shakespeare_char",
         "--local-batch", "4"],
        cwd=REPO, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [ln for ln in r.stdout.strip().splitlines() if ln.startswith("{")]
    assert len(lines) == 1, r.stdout
    d = json.loads(lines[0])
    for key, typ in [("metric", str), ("value", (int, float)), ("unit", str),
                     ("n_gpus", int), ("steps", int), ("warmup", int),
                     ("ms_per_step", (int, float)), ("higher_is_better", bool),
                     ("scaling", str), ("vs_baseline", (int, float)),
                     ("dtype", str), ("data", str), ("config", dict)]:
        assert key in d, key
        assert isinstance(d[key], typ), (key, type(d[key]))
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["warmup"] == 1
    assert d["data"] == "synthetic"
    assert d["scaling"] == "weak"
    assert d["higher_is_better"] is True
    for k in ["model", "global_batch", "seq_len", "parallelism"]:
        assert k in d["config"], k
