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

    rundir = args.rundir
    if rundir is None:
        stamp = datetime.datetime.now().strftime("%Y%m%d_%H%M%S")
        rundir = os.path.join("runs", f"{args.config}_{stamp}")
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
