"""One-off probe: compare conv_dgrad / conv_wgrad against torch fp32 for a
shape list (inception-v3 sweep) to localize a numerics regression."""
import sys

import torch
import torch.nn.functional as F

sys.path.insert(0, ".")
from dtmx.ops.hip import require_ext  # noqa: E402

SHAPES = [
    ((3, 299, 299), (32, 3, 3, 3), 2, 0),
    ((32, 147, 147), (64, 32, 3, 3), 1, 1),
    ((32, 149, 149), (32, 32, 3, 3), 1, 0),
    ((48, 35, 35), (64, 48, 5, 5), 1, 2),
    ((64, 35, 35), (96, 64, 3, 3), 1, 1),
    ((64, 73, 73), (80, 64, 1, 1), 1, 0),
    ((80, 73, 73), (192, 80, 3, 3), 1, 0),
    ((96, 35, 35), (96, 96, 3, 3), 2, 0),
    ((160, 17, 17), (192, 160, 3, 3), 1, 1),
    ((192, 17, 17), (192, 192, 3, 3), 2, 0),
    ((192, 35, 35), (32, 192, 1, 1), 1, 0),
    ((288, 35, 35), (384, 288, 3, 3), 2, 0),
    ((448, 8, 8), (384, 448, 3, 3), 1, 1),
    ((768, 17, 17), (128, 768, 1, 1), 1, 0),
    ((1280, 8, 8), (448, 1280, 1, 1), 1, 0),
    ((2048, 8, 8), (192, 2048, 1, 1), 1, 0),
]

ext = require_ext()
B = 4
bad = 0
for (cin_shape, wshape, stride, pad) in SHAPES:
    C, H, W = cin_shape
    K, _, R, S = wshape
    g = torch.Generator().manual_seed(hash((C, H, K, R)) % 2**31)
    x = (torch.randn(B, C, H, W, generator=g) * 0.5).to(torch.bfloat16)
    w = (torch.randn(*wshape, generator=g) * 0.1).to(torch.bfloat16)
    P = (H + 2 * pad - R) // stride + 1
    Q = (W + 2 * pad - S) // stride + 1
    dy = (torch.randn(B, K, P, Q, generator=g) * 0.5).to(torch.bfloat16)

    xr = x.float().requires_grad_(True)
    wr = w.float().requires_grad_(True)
    yr = F.conv2d(xr, wr, stride=stride, padding=pad)
    yr.backward(dy.float())

    xg = x.cuda().contiguous(memory_format=torch.channels_last)
    wg = w.cuda().contiguous(memory_format=torch.channels_last)
    dyg = dy.cuda().contiguous(memory_format=torch.channels_last)

    for name, fn, ref in (
        ("dgrad", lambda: ext.conv_dgrad(dyg, wg, stride, pad, H, W), xr.grad),
        ("wgrad", lambda: ext.conv_wgrad(xg, dyg, R, S, stride, pad), wr.grad),
        ("wgrad2", lambda: ext.conv_wgrad(xg, dyg, R, S, stride, pad), wr.grad),
    ):
        if name == "dgrad" and C % 8:
            continue
        got = fn().float().cpu()
        want = ref
        err = (got - want).abs().max().item()
        rel = err / (want.abs().max().item() + 1e-6)
        status = "OK " if rel < 0.05 else "BAD"
        if status == "BAD":
            bad += 1
        print(f"{status} {name:6s} C={C:4d} HW={H:3d} K={K:4d} RS={R} s={stride} "
              f"p={pad}  maxerr={err:.4f} rel={rel:.4f}")
print("FAILURES:", bad)
