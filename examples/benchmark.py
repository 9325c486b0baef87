#!/usr/bin/env python3
"""Multi-GPU scaling sweep (reference example/image-classification/
benchmark.py): runs the flagship bench at each GPU count via
torch.distributed.run (one rank per GPU over RCCL) and reports the curve."""
import argparse
import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_one(n, args):
    bench = os.path.join(ROOT, "bench.py")
    passthru = ["--gpus", str(n), "--steps", str(args.steps),
                "--warmup", str(args.warmup), "--batch-size", str(args.batch_size),
                "--network", args.network, "--num-layers", str(args.num_layers),
                "--dtype", args.dtype]
    if n == 1:
        cmd = [sys.executable, bench] + passthru
    else:
        cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
               f"--nproc-per-node={n}", "--master-addr", "127.0.0.1",
               "--master-port", str(29500 + n), bench] + passthru
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=1800)
    for line in reversed(out.stdout.splitlines()):
        line = line.strip()
        if line.startswith("{"):
            return json.loads(line)
    raise RuntimeError(f"no bench output for n={n}:\n{out.stdout}\n{out.stderr}")


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpu-counts", type=str, default="1,2,4,8")
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--batch-size", type=int, default=1024)
    ap.add_argument("--network", type=str, default="resnet")
    ap.add_argument("--num-layers", type=int, default=50)
    ap.add_argument("--dtype", type=str, default="bfloat16")
    args = ap.parse_args()

    base = None
    for n in (int(x) for x in args.gpu_counts.split(",")):
        r = run_one(n, args)
        if base is None:
            base = r["value"]
        eff = r["value"] / (base * n) * 100.0
        print(f"gpus={n:2d}  {r['value']:10.1f} images/sec  "
              f"({r['ms_per_step']:.1f} ms/step, weak-scaling eff {eff:.1f}%)")
