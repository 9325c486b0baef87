#!/usr/bin/env python3
"""Kernel microbenchmarks: per-shape TF/s for the dtmx GEMM/conv kernels
(within-run A/B harness for kernel tuning; cdna_hip_programming.md §5.4)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from dtmx.ops.hip import require_ext

ext = require_ext()
DEV = "cuda:0"


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def tf(flops, sec):
    return flops / sec / 1e12


def nhwc(t):
    return t.to(DEV).contiguous(memory_format=torch.channels_last)


def main():
    torch.manual_seed(0)
    # 0) 256^2 8-phase kernel: correctness then A/B vs the 128^2 kernel
    a = torch.randn(512, 512, dtype=torch.bfloat16, device=DEV)
    b = torch.randn(512, 512, dtype=torch.bfloat16, device=DEV)
    ref = (a.float() @ b.float().T)
    got = ext.gemm256_nt(a, b).float()
    err = (got - ref).abs().max().item() / ref.abs().max().item()
    print(f"gemm256 refcheck 512^3: rel max err {err:.4f}")
    a = torch.randint(-3, 4, (768, 640), device=DEV).to(torch.bfloat16)
    b = torch.randint(-3, 4, (520, 640), device=DEV).to(torch.bfloat16)
    got = ext.gemm256_nt(a, b).double()
    ref = (a.double() @ b.double().T)
    exact = (got == ref.to(torch.bfloat16).double()).all().item()
    print(f"gemm256 integer-exact 768x520x640 (tails): {exact}")

    # 1) square dense GEMM peak of the kernel
    for M, N, K in [(4096, 4096, 4096), (8192, 8192, 8192)]:
        a = torch.randn(M, K, dtype=torch.bfloat16, device=DEV)
        b = torch.randn(N, K, dtype=torch.bfloat16, device=DEV)
        s = timeit(lambda: ext.linear_fwd(a, b, None))
        print(f"linear_fwd(routed) {M}x{N}x{K}: {tf(2*M*N*K, s):7.1f} TF  {s*1e3:.3f} ms")
        s = timeit(lambda: ext.gemm256_nt(a, b))
        print(f"gemm256_direct     {M}x{N}x{K}: {tf(2*M*N*K, s):7.1f} TF  {s*1e3:.3f} ms")

    # 2) resnet-shaped dense (same GEMM the conv would do, no gather)
    for M, N, K in [(401408, 64, 576), (401408, 128, 256), (100352, 256, 1152),
                    (25088, 512, 2304), (6272, 2048, 512)]:
        a = torch.randn(M, K, dtype=torch.bfloat16, device=DEV)
        b = torch.randn(N, K, dtype=torch.bfloat16, device=DEV)
        s = timeit(lambda: ext.linear_fwd(a, b, None))
        print(f"skinny {M}x{N}x{K}: {tf(2*M*N*K, s):7.1f} TF  {s*1e3:.3f} ms")

    # 3) conv fwd/dgrad/wgrad on key resnet-50 bs128 layers
    cases = [
        ("l1.conv2 3x3 64->64 56^2 s1", 128, 64, 56, 64, 3, 1, 1),
        ("l1.conv3 1x1 64->256 56^2", 128, 64, 56, 256, 1, 1, 0),
        ("l2.conv2 3x3 128->128 28^2", 128, 128, 28, 128, 3, 1, 1),
        ("l3.conv2 3x3 256->256 14^2", 128, 256, 14, 256, 3, 1, 1),
        ("l4.conv2 3x3 512->512 7^2", 128, 512, 7, 512, 3, 1, 1),
        ("l2.ds 1x1 256->512 s2 28^2", 128, 256, 56, 512, 1, 2, 0),
    ]
    for name, Nb, C, H, K, R, stride, pad in cases:
        x = torch.randn(Nb, C, H, H, dtype=torch.bfloat16)
        w = torch.randn(K, C, R, R, dtype=torch.bfloat16) * 0.1
        xd, wd = nhwc(x), nhwc(w)
        P = (H + 2 * pad - R) // stride + 1
        dy = nhwc(torch.randn(Nb, K, P, P, dtype=torch.bfloat16))
        flops = 2.0 * Nb * P * P * K * R * R * C
        sf = timeit(lambda: ext.conv_fwd(xd, wd, stride, pad))
        sd = timeit(lambda: ext.conv_dgrad(dy, wd, stride, pad, H, H))
        sw = timeit(lambda: ext.conv_wgrad(xd, dy, R, R, stride, pad))
        print(f"{name}: fwd {tf(flops,sf):6.1f} TF ({sf*1e3:.3f} ms)  "
              f"dgrad {tf(flops,sd):6.1f} TF ({sd*1e3:.3f})  "
              f"wgrad {tf(flops,sw):6.1f} TF ({sw*1e3:.3f})")




def bn_micro():
    """Per-kernel GB/s of the BN elementwise/reduction passes (roofline is
    ~6.3 TB/s; anything far below is a tuning target, not 'memory-bound')."""
    torch.manual_seed(0)
    N, C, H = 1024, 256, 56
    x = torch.randn(N, C, H, H, dtype=torch.bfloat16, device=DEV).contiguous(
        memory_format=torch.channels_last)
    dy = torch.randn_like(x)
    y = torch.relu(x)
    gamma = torch.randn(C, dtype=torch.bfloat16, device=DEV)
    beta = torch.randn(C, dtype=torch.bfloat16, device=DEV)
    rm = torch.zeros(C, device=DEV)
    rv = torch.ones(C, device=DEV)
    nbytes = x.numel() * 2

    s = timeit(lambda: ext.bn_fwd_train(x, gamma, beta, rm, rv, 0.9, 1e-5,
                                        True, None, None, None), iters=10)
    print(f"bn_fwd_train (stats+apply, ~3T): {3 * nbytes / s / 1e12:.2f} TB/s  {s*1e3:.3f} ms")
    out = ext.bn_fwd_train(x, gamma, beta, rm, rv, 0.9, 1e-5, True, None,
                           None, None)
    yv, mean, invstd = out
    s = timeit(lambda: ext.bn_bwd(x, dy, gamma, mean, invstd, True, yv, False),
               iters=10)
    print(f"bn_bwd (stats+dx, ~7T):          {7 * nbytes / s / 1e12:.2f} TB/s  {s*1e3:.3f} ms")
    tdb = mean.clone(); tdg = invstd.clone()
    s = timeit(lambda: ext.bn_bwd_dx_presummed(x, dy, yv, mean, invstd, gamma,
                                               tdb, tdg, N * H * H, False,
                                               False), iters=10)
    print(f"bn_bwd_dx (2R+1W):               {3 * nbytes / s / 1e12:.2f} TB/s  {s*1e3:.3f} ms")
    s = timeit(lambda: ext.bn_local_sums(x), iters=10)
    print(f"bn_stats (1R):                   {1 * nbytes / s / 1e12:.2f} TB/s  {s*1e3:.3f} ms")




def conv_routed():
    """Big-batch conv shapes that route the GATHER providers through
    gemm256f — per-shape A/B vs DTMX_DISABLE_GEMM256=1."""
    import os
    for (C, Ko, H, s, p, bs) in [(256, 256, 14, 1, 1, 512),
                                 (512, 512, 7, 1, 1, 1024),
                                 (128, 128, 28, 1, 1, 512)]:
        x = nhwc(torch.randn(bs, C, H, H, dtype=torch.bfloat16))
        w = nhwc(torch.randn(Ko, C, 3, 3, dtype=torch.bfloat16) * 0.05)
        dy = nhwc(torch.randn(bs, Ko, H, H, dtype=torch.bfloat16))
        P = H
        fl = 2 * bs * P * P * Ko * C * 9
        sf = timeit(lambda: ext.conv_fwd(x, w, s, p), iters=10)
        sd = timeit(lambda: ext.conv_dgrad(dy, w, s, p, H, H, None), iters=10)
        print(f"conv3x3 bs{bs} {C}->{Ko} {H}^2: fwd {tf(fl, sf):6.1f} TF  "
              f"dgrad {tf(fl, sd):6.1f} TF")




if __name__ == "__main__":
    if os.environ.get("DTMX_BK_ONLY_CONV") == "1":
        conv_routed()
    else:
        main()
        bn_micro()
