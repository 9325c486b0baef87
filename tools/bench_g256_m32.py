#!/usr/bin/env python3
"""A/B the gemm256 32x32x16-MFMA probe variant (DTMX_G256_M32=1) against
the production 16x16x32 schedule. Env is latched once per process, so run
this twice (wrapper re-execs itself with the env set)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from dtmx.ops.hip import require_ext


def timeit(fn, iters=30, warmup=8):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def run(tag):
    ext = require_ext()
    torch.manual_seed(0)
    dev = "cuda:0"
    a = torch.randn(512, 512, dtype=torch.bfloat16, device=dev)
    b = torch.randn(512, 512, dtype=torch.bfloat16, device=dev)
    ref = a.float() @ b.float().T
    got = ext.gemm256_nt(a, b).float()
    err = (got - ref).abs().max().item() / ref.abs().max().item()
    ai = torch.randint(-3, 4, (768, 640), device=dev).to(torch.bfloat16)
    bi = torch.randint(-3, 4, (520, 640), device=dev).to(torch.bfloat16)
    exact = (ext.gemm256_nt(ai, bi).double() ==
             (ai.double() @ bi.double().T).to(torch.bfloat16).double()).all().item()
    print(f"[{tag}] refcheck err {err:.4f}  integer-exact {exact}")
    assert err < 0.05 and exact, f"{tag}: CORRECTNESS FAIL"
    for M, N, K in [(4096, 4096, 4096), (8192, 8192, 8192),
                    (2048, 512, 4096), (16384, 2048, 1024)]:
        a = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
        b = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
        s = timeit(lambda: ext.gemm256_nt(a, b))
        print(f"[{tag}] {M}x{N}x{K}: {2*M*N*K/s/1e12:7.1f} TF  {s*1e3:.3f} ms",
              flush=True)
    # fp16 smoke: the f16 builtin path
    a = torch.randn(1024, 1024, dtype=torch.float16, device=dev)
    b = torch.randn(1024, 1024, dtype=torch.float16, device=dev)
    errh = (ext.gemm256_nt(a, b).float() - a.float() @ b.float().T).abs().max().item()
    print(f"[{tag}] fp16 1024^3 abs err {errh:.3f}")
    assert errh < 2.0


if __name__ == "__main__":
    import subprocess
    if len(sys.argv) > 1:
        run(sys.argv[1])
    else:
        for tag, env in [("base", {}), ("m32", {"DTMX_G256_M32": "1"})]:
            e = dict(os.environ, **env)
            r = subprocess.run([sys.executable, __file__, tag], env=e)
            if r.returncode:
                sys.exit(r.returncode)
