"""Minimal repro: linear_fwd at tiny M with garbage-filled allocator blocks."""
import sys

import torch

sys.path.insert(0, ".")
from dtmx.ops.hip import require_ext  # noqa: E402

ext = require_ext()
torch.manual_seed(0)

for M, N, K in [(4, 100, 2048), (4, 100, 2048), (8, 100, 2048), (4, 1000, 2048),
                (4, 100, 9216), (16, 100, 2048), (128, 100, 2048)]:
    x = (torch.randn(M, K) * 0.5).to(torch.bfloat16).cuda()
    w = (torch.randn(N, K) * 0.05).to(torch.bfloat16).cuda()
    ref = (x.float() @ w.float().T).cpu()
    worst = 0.0
    for it in range(6):
        # churn: fill a block with big garbage, free it
        junk = torch.full((it % 3 + 1, 4 << 20), 3e4, dtype=torch.bfloat16,
                          device="cuda")
        del junk
        y = ext.linear_fwd(x, w, None).float().cpu()
        err = (y - ref).abs().max().item()
        worst = max(worst, err)
    scale = ref.abs().max().item()
    print(f"M={M} N={N} K={K}: worst_err={worst:.4g} (ref max {scale:.3g}) "
          f"{'BAD' if worst > 0.05 * scale + 0.1 else 'ok'}")
