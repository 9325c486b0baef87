#!/usr/bin/env python3
"""256x128 8-phase probe (gemm256n) vs the production route for
N=128-class dense shapes (linear_fwd routes those to the 128^2 TN
kernel). Correctness: fp32-ref + integer-exact, then per-shape TF."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from dtmx.ops.hip import require_ext

ext = require_ext()
DEV = "cuda:0"


def timeit(fn, iters=20, warmup=6):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    torch.manual_seed(0)
    a = torch.randn(512, 512, dtype=torch.bfloat16, device=DEV)
    b = torch.randn(128, 512, dtype=torch.bfloat16, device=DEV)
    ref = a.float() @ b.float().T
    err = (ext.gemm256n_nt(a, b).float() - ref).abs().max().item() / ref.abs().max().item()
    print(f"refcheck 512x128x512: rel err {err:.4f}")
    assert err < 0.05
    ai = torch.randint(-3, 4, (770, 640), device=DEV).to(torch.bfloat16)
    bi = torch.randint(-3, 4, (130, 640), device=DEV).to(torch.bfloat16)
    exact = (ext.gemm256n_nt(ai, bi).double() ==
             (ai.double() @ bi.double().T).to(torch.bfloat16).double()).all().item()
    print(f"integer-exact 770x130x640 (tails): {exact}")
    assert exact
    # fp16 smoke
    ah = torch.randn(512, 512, dtype=torch.float16, device=DEV)
    bh = torch.randn(128, 512, dtype=torch.float16, device=DEV)
    errh = (ext.gemm256n_nt(ah, bh).float() - ah.float() @ bh.float().T).abs().max().item()
    assert errh < 2.0, errh

    # l2-class conv-as-GEMM shapes (M = N*H*W at bs1024/4096, N=128,
    # K = C*3*3 or C) + one tall FC-ish shape
    for M, N, K in [(802816, 128, 1152), (802816, 128, 512),
                    (200704, 128, 1152), (100352, 128, 2304),
                    (8192, 128, 4096)]:
        a = torch.randn(M, K, dtype=torch.bfloat16, device=DEV)
        b = torch.randn(N, K, dtype=torch.bfloat16, device=DEV)
        fl = 2.0 * M * N * K
        s0 = timeit(lambda: ext.linear_fwd(a, b, None))
        s1 = timeit(lambda: ext.gemm256n_nt(a, b))
        print(f"{M}x{N}x{K}: routed {fl/s0/1e12:6.1f} TF  {s0*1e3:7.3f} ms | "
              f"g256n {fl/s1/1e12:6.1f} TF  {s1*1e3:7.3f} ms  "
              f"({s0/s1:5.2f}x)", flush=True)


if __name__ == "__main__":
    main()
