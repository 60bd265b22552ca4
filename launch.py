"""Launcher — user contract parity with the reference launch.py:
``python launch.py --config=<name> [--rundir=DIR] [--debug] [--distributed]``

(--distributed replaces the reference's --multihost: here it means "I was
launched under torchrun, one process per GPU over RCCL"; single-process
runs need no flag.)

Writes config.json to the rundir for exact reload by sample.py
(reference launch.py:56-57).
"""
from __future__ import annotations

import argparse
import datetime
import os

from midgpt_amd.config import load_config
from midgpt_amd.parallel import dist as pdist
from midgpt_amd.train import train


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--config", required=True, help="config module name, e.g. openwebtext_xl")
    p.add_argument("--rundir", default=None, help="resume/run directory (created if missing)")
    p.add_argument("--debug", action="store_true")
    p.add_argument("--distributed", action="store_true",
                   help="multi-process launch (torchrun); reference --multihost")
    args = p.parse_args()

    config = load_config(args.config)
    config.debug = args.debug

    world = int(os.environ.get("WORLD_SIZE", "1"))
    if args.distributed and world <= 1:
        p.error("--distributed requires a torchrun launch (WORLD_SIZE unset); "
                "use: python -m torch.distributed.run --nproc-per-node N "
                "--master-addr 127.0.0.1 launch.py ...")

    rundir = args.rundir
    if rundir is None:
        # Every rank must agree on the generated rundir, or checkpoint
        # shards scatter across per-rank timestamp dirs (reference
        # launch.py asserts multihost runs prespecify rundir): broadcast
        # rank 0's generated name over the process group.
        stamp = datetime.datetime.now().strftime("%Y%m%d_%H%M%S")
        rundir = os.path.join("runs", f"{args.config}_{stamp}")
        if world > 1:
            pdist.init_distributed()
            rundir = pdist.broadcast_str(rundir)
    config.rundir = rundir

    # rank 0 creates the rundir and freezes the config
    if int(os.environ.get("RANK", "0")) == 0:
        os.makedirs(rundir, exist_ok=True)
        with open(os.path.join(rundir, "config.json"), "w") as f:
            f.write(config.to_json())
    print(config.to_json() if int(os.environ.get("RANK", "0")) == 0 else "")

    train(config)
    pdist.barrier()


if __name__ == "__main__":
    main()
