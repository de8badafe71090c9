#!/usr/bin/env python3
"""Benchmark harness — runs the BASELINE.json configs (the reference's
benchmark/ + KWOK-study analog).

  python benchmark/run.py --list
  python benchmark/run.py --config 2          # 1k nodes / 10k pods
  python benchmark/run.py --config 4 --gpus 8 # sharded across 8 GPUs

Each config shells out to the repo-root bench.py (the driver contract)
with the matching inventory shape and prints its JSON line.
"""

from __future__ import annotations

import argparse
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

# BASELINE.json "configs" (index 0 is the CPU-plumbing demo)
CONFIGS = {
    0: {"desc": "example job gang-schedules end-to-end (control plane demo)",
        "cmd": [sys.executable, os.path.join(ROOT, "examples", "demo.py")]},
    1: {"desc": "1k nodes / 10k pods, allocate+gang+drf, 1 GPU",
        "bench": ["--nodes", "1000", "--jobs", "1000", "--pods-per-job", "10"]},
    2: {"desc": "10k nodes / 100k pods, binpack+nodeorder scoring, 1 GPU",
        "bench": ["--nodes", "10000", "--jobs", "10000",
                  "--pods-per-job", "10"]},
    3: {"desc": "10k nodes / 100k pods, queue proportion, sharded N GPUs",
        "bench": ["--nodes", "10000", "--jobs", "10000",
                  "--pods-per-job", "10"]},
    4: {"desc": "50k nodes / 1M pods, task-topology + numaaware, "
             "8 GPUs (also meaningful at 1)",
        "bench": ["--nodes", "50000", "--jobs", "50000",
                  "--pods-per-job", "20", "--steps", "1",
                  "--extra-plugins", "task-topology,numaaware"]},
}


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", type=int, default=2)
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=None)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--shard-mode", default="hard")
    ap.add_argument("--cpu", action="store_true")
    ap.add_argument("--list", action="store_true")
    args = ap.parse_args()

    if args.list:
        for i, c in CONFIGS.items():
            print(f"  {i}: {c['desc']}")
        return 0

    cfg = CONFIGS[args.config]
    if "cmd" in cfg:
        return subprocess.call(cfg["cmd"], cwd=ROOT)

    bench = [os.path.join(ROOT, "bench.py")] + cfg["bench"]
    if args.steps is not None:
        bench += ["--steps", str(args.steps)]
    bench += ["--warmup", str(args.warmup),
              "--shard-mode", args.shard_mode]
    if args.cpu:
        bench.append("--cpu")
    if args.gpus > 1:
        cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
               f"--nproc-per-node={args.gpus}", "--master-addr", "127.0.0.1",
               "--master-port", "29510"] + bench + ["--gpus", str(args.gpus)]
    else:
        cmd = [sys.executable] + bench
    print("+", " ".join(cmd), file=sys.stderr)
    return subprocess.call(cmd, cwd=ROOT)


if __name__ == "__main__":
    sys.exit(main())
