#!/usr/bin/env python3
"""Wgrad NT per-shape micro: TF/s + implied HBM traffic, for the roofline
question 'is the 20%-of-step wgrad family bandwidth-bound or schedule-bound?'
Run under rocprofv3 --pmc FETCH_SIZE,WRITE_SIZE to get true fetched bytes."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from dtmx.ops.hip import require_ext

ext = require_ext()
DEV = "cuda:0"


def timeit(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main(bs=1024):
    torch.manual_seed(0)
    # (C_in, K_out, H, stride, pad) resnet50 3x3 representatives + stem-adjacent
    shapes = [(128, 128, 28, 1, 1), (256, 256, 14, 1, 1), (512, 512, 7, 1, 1),
              (64, 64, 56, 1, 1)]
    for (C, Ko, H, st, pa) in shapes:
        x = torch.randn(bs, C, H, H, dtype=torch.bfloat16, device=DEV).contiguous(
            memory_format=torch.channels_last)
        dy = torch.randn(bs, Ko, H, H, dtype=torch.bfloat16, device=DEV).contiguous(
            memory_format=torch.channels_last)
        s = timeit(lambda: ext.conv_wgrad(x, dy, 3, 3, st, pa))
        fl = 2.0 * bs * H * H * Ko * C * 9
        # minimum HBM traffic: dy once, x once, dw out (tiny)
        min_gb = (dy.numel() + x.numel()) * 2 / 1e9
        print(f"wgrad3x3 bs{bs} {C}->{Ko} {H}^2: {fl/s/1e12:7.1f} TF  "
              f"{s*1e3:7.3f} ms  min-traffic {min_gb:5.2f} GB -> "
              f"{min_gb/s/1e3:5.2f} TB/s floor", flush=True)


def triad():
    """bf16 RW-mix ceilings for the BN elementwise kernels: what does THIS
    box sustain on d=a+b (2R+1W, the bn_bwd_dx mix) and b=a*s (1R+1W, the
    bn_apply mix)? Quantifies how close 4.9-5.0 TB/s is to attainable."""
    n = 1024 * 256 * 56 * 56  # the bn micro shape's element count
    a = torch.randn(n, dtype=torch.bfloat16, device=DEV)
    b = torch.randn_like(a)
    d = torch.empty_like(a)
    s = timeit(lambda: torch.add(a, b, out=d), iters=20)
    print(f"triad d=a+b  (2R+1W): {3 * n * 2 / s / 1e12:5.2f} TB/s")
    s = timeit(lambda: torch.mul(a, 1.5, out=d), iters=20)
    print(f"scale b=a*s  (1R+1W): {2 * n * 2 / s / 1e12:5.2f} TB/s")


if __name__ == "__main__":
    if len(sys.argv) > 1 and sys.argv[1] == "triad":
        triad()
    else:
        main(int(sys.argv[1]) if len(sys.argv) > 1 else 1024)
