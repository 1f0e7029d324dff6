#!/usr/bin/env python3
"""Benchmark grid driver (reference laboratory/benchmark/benchmark.py:
{WDL, DeepFM, xDeepFM} x dim {9, 64} x GPUs {1..8} grid).

Runs bench.py per cell as a subprocess (torchrun for n>1) and prints a
summary table + writes JSON lines.

    python scripts/benchmark_grid.py --gpus 1 --steps 50
    python scripts/benchmark_grid.py --gpus 1,8 --models deepfm,wdl
"""

import argparse
import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_cell(model, dim, n_gpus, steps, warmup, batch):
    args = [f"--model={model}", f"--dim={dim}", f"--gpus={n_gpus}",
            f"--steps={steps}", f"--warmup={warmup}", f"--batch={batch}"]
    if n_gpus == 1:
        cmd = [sys.executable, "bench.py"] + args
    else:
        cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
               f"--nproc-per-node={n_gpus}", "--master-addr=127.0.0.1",
               "--master-port=29617", "bench.py"] + args
    env = dict(os.environ, PYTHONPATH=ROOT)
    out = subprocess.run(cmd, cwd=ROOT, env=env, capture_output=True,
                         text=True, timeout=1200)
    for line in reversed(out.stdout.splitlines()):
        line = line.strip()
        if line.startswith("{") and '"metric"' in line:
            return json.loads(line)
    raise RuntimeError(f"{model} dim{dim} x{n_gpus}: no result line\n"
                       f"{out.stdout[-2000:]}\n{out.stderr[-2000:]}")


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--models", default="deepfm,wdl,xdeepfm")
    p.add_argument("--dims", default="9,64")
    p.add_argument("--gpus", default="1")
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--batch", type=int, default=4096)
    p.add_argument("--out", default="gpurun_out/grid.jsonl")
    args = p.parse_args()

    cells = [(m, int(d), int(g))
             for m in args.models.split(",")
             for d in args.dims.split(",")
             for g in args.gpus.split(",")]
    os.makedirs(os.path.dirname(args.out) or ".", exist_ok=True)
    results = []
    with open(args.out, "w") as f:
        for model, dim, g in cells:
            try:
                r = run_cell(model, dim, g, args.steps, args.warmup,
                             args.batch)
            except Exception as e:  # noqa: BLE001
                print(f"FAIL {model} dim{dim} x{g}: {e}", file=sys.stderr)
                continue
            results.append(r)
            f.write(json.dumps(r) + "\n")
            f.flush()
            print(f"{model:8s} dim{dim:<3d} x{g}  "
                  f"{r['value'] / 1e6:8.2f}M samples/s  "
                  f"{r['ms_per_step']:7.3f} ms/step")
    print(f"\n{len(results)}/{len(cells)} cells -> {args.out}")


if __name__ == "__main__":
    main()
