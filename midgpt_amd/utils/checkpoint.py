"""Checkpointing: async, sharded, rundir-contract-compatible.

Layout (parity with the reference rundir contract, launch.py:56-67 and
src/train.py:139-145, 214-215):

    rundir/
      config.json            # frozen ExperimentConfig
      wandb_id.txt           # persisted run id (crash-safe resume)
      ckpt_0001000/
        manifest.json        # ordered named-parameter manifest + shard map
        rank00000.pt         # this rank's fp32 master/m/v shard
        ...
        DONE                 # commit marker written after all shards land

max_to_keep=1, save every eval_interval steps, async (background thread
after a device->host copy), resume at latest_step()+1 — all reference
semantics. The saved object is the fp32 master + Adam state (the model's
bf16 weights are derived), keyed by the ordered manifest — the named
equivalent of the reference's flat leaf lists (src/train.py:215).
"""
from __future__ import annotations

import json
import os
import shutil
import threading

import torch

from midgpt_amd.parallel import dist as pdist


def ckpt_dir(rundir: str, step: int) -> str:
    return os.path.join(rundir, f"ckpt_{step:07d}")


def latest_step(rundir: str) -> int | None:
    if not os.path.isdir(rundir):
        return None
    steps = []
    for d in os.listdir(rundir):
        if d.startswith("ckpt_") and os.path.exists(os.path.join(rundir, d, "DONE")):
            try:
                steps.append(int(d.split("_")[1]))
            except ValueError:
                pass
    return max(steps) if steps else None


class CheckpointManager:
    def __init__(self, rundir: str, save_interval: int, max_to_keep: int = 1):
        self.rundir = rundir
        self.save_interval = save_interval
        self.max_to_keep = max_to_keep
        self._thread: threading.Thread | None = None

    def should_save(self, step: int) -> bool:
        return self.save_interval > 0 and step > 0 and step % self.save_interval == 0

    def save(self, step: int, engine, extra: dict | None = None):
        """Async save: D2H snapshot on a side copy stream (pinned staging),
        file write in a background thread. The engine's next optimizer
        step waits on the snapshot event instead of the train loop
        blocking here."""
        self.wait()
        shard = engine.state_shard()
        evt = None
        on_gpu = any(torch.is_tensor(v) and v.is_cuda for v in shard.values())
        if on_gpu:
            if not hasattr(self, "_copy_stream"):
                self._copy_stream = torch.cuda.Stream()
                self._pinned = {}
            host = {}
            with torch.cuda.stream(self._copy_stream):
                self._copy_stream.wait_stream(torch.cuda.current_stream())
                for k, v in shard.items():
                    if torch.is_tensor(v):
                        buf = self._pinned.get(k)
                        if buf is None or buf.shape != v.shape or \
                                buf.dtype != v.dtype:
                            buf = torch.empty_like(v, device="cpu",
                                                   pin_memory=True)
                            self._pinned[k] = buf
                        buf.copy_(v.detach(), non_blocking=True)
                        host[k] = buf
                    else:
                        host[k] = v
                evt = torch.cuda.Event()
                evt.record(self._copy_stream)
            # master/m/v must not be overwritten before the copy drains:
            # the engine waits on this event at its next step()
            engine.defer_until(evt)
        else:
            host = {k: (v.detach().to("cpu", copy=True)
                        if torch.is_tensor(v) else v)
                    for k, v in shard.items()}
        if extra:
            host["extra"] = extra
        manifest = {
            "step": step,
            "world": pdist.get_world_size(),
            "zero": engine.zero,
            "total": engine.total,
            "padded": engine.padded,
            "shard_size": engine.shard_size,
            "params": engine.named_param_manifest(),
        }
        rank = pdist.get_rank()
        # In replicated (ddp) mode every rank holds the full state: rank 0
        # writes alone. In zero mode every rank writes its shard.
        self._nshards = pdist.get_world_size() if engine.zero else 1
        write_shard = engine.zero or rank == 0
        d = ckpt_dir(self.rundir, step)

        def _write():
            if evt is not None:
                evt.synchronize()  # pinned staging buffers fully landed
            os.makedirs(d, exist_ok=True)
            if write_shard:
                torch.save(host, os.path.join(d, f"rank{rank:05d}.pt"))
            if rank == 0:
                with open(os.path.join(d, "manifest.json"), "w") as f:
                    json.dump(manifest, f, indent=2)

        self._thread = threading.Thread(target=_write, daemon=True)
        self._thread.start()
        # Commit marker + GC happen on the next wait() via finalize.
        self._pending = (d, step)

    def wait(self):
        """Block until the in-flight save finishes; commit and GC."""
        if self._thread is not None:
            self._thread.join()
            self._thread = None
            d, step = self._pending
            pdist.barrier()  # all ranks' shards on disk
            if pdist.is_main():
                nshards = len([f for f in os.listdir(d) if f.startswith("rank")])
                if nshards == self._nshards:
                    with open(os.path.join(d, "DONE"), "w") as f:
                        f.write(str(step))
                self._gc(keep_step=step)
            pdist.barrier()

    def _gc(self, keep_step: int):
        steps = sorted(
            int(x.split("_")[1]) for x in os.listdir(self.rundir)
            if x.startswith("ckpt_") and
            os.path.exists(os.path.join(self.rundir, x, "DONE")))
        for s in steps[:-self.max_to_keep]:
            shutil.rmtree(ckpt_dir(self.rundir, s), ignore_errors=True)


def load_full_state(rundir: str, step: int | None = None):
    """Assemble the FULL fp32 (master, m, v, step_count) from shard files —
    resharding-safe restore for any new world size."""
    if step is None:
        step = latest_step(rundir)
        if step is None:
            return None
    d = ckpt_dir(rundir, step)
    with open(os.path.join(d, "manifest.json")) as f:
        manifest = json.load(f)
    padded = manifest["padded"]
    master = torch.zeros(padded, dtype=torch.float32)
    m = torch.zeros(padded, dtype=torch.float32)
    v = torch.zeros(padded, dtype=torch.float32)
    step_count = 0
    for fn in sorted(os.listdir(d)):
        if not fn.startswith("rank"):
            continue
        sh = torch.load(os.path.join(d, fn), map_location="cpu", weights_only=False)
        if "pieces" in sh:
            # bucketed shard: shard[po:po+n] lives at full[fo:fo+n]
            for fo, n, po in sh["pieces"]:
                master[fo:fo + n] = sh["master"][po:po + n]
                m[fo:fo + n] = sh["m"][po:po + n]
                v[fo:fo + n] = sh["v"][po:po + n]
        else:
            o, n = sh["shard_off"], sh["shard_size"]
            master[o:o + n] = sh["master"]
            m[o:o + n] = sh["m"]
            v[o:o + n] = sh["v"]
        step_count = sh["step_count"]
    return {"step": step, "master": master, "m": m, "v": v,
            "step_count": step_count, "manifest": manifest}
