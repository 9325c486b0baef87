"""Bisect the BN-backward fusion divergence: run resnet-18 fwd+bwd under
(A) fusion off, (B) interior seams only, (C) interior+cross, and print
per-parameter relative grad distances vs A."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import dtmx  # noqa: F401
from dtmx.models import get_symbol
from dtmx.ops import fusedblock


def run(fuse, cross):
    os.environ["DTMX_FUSE_BN_BWD"] = fuse
    os.environ["DTMX_FUSE_BN_CROSS"] = cross
    os.environ["DTMX_FUSED_BLOCK"] = "1"
    for k in fusedblock.bnbwd_stats:
        fusedblock.bnbwd_stats[k] = 0
    torch.manual_seed(0)
    net = get_symbol("resnet", num_layers=18, num_classes=10,
                     image_shape="3,32,32")
    net = net.to("cuda").to(torch.bfloat16).to(memory_format=torch.channels_last)
    torch.manual_seed(1)
    x = torch.randn(8, 3, 32, 32, device="cuda").to(torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last)
    label = torch.randint(0, 10, (8,), device="cuda")
    net.zero_grad()
    out = net(x)
    loss = torch.nn.functional.cross_entropy(out.float(), label, reduction="sum")
    loss.backward()
    grads = {n: p.grad.detach().float().clone()
             for n, p in net.named_parameters() if p.grad is not None}
    return loss.item(), grads, dict(fusedblock.bnbwd_stats)


la, ga, _ = run("0", "0")
la2, ga2, _ = run("0", "0")  # run-to-run noise floor (split-K atomics)
lb, gb, sb = run("1", "0")
lc, gc, sc = run("1", "1")
print(f"loss A={la:.4f} A2={la2:.4f} B={lb:.4f} C={lc:.4f}")
print("stats B:", sb, " C:", sc)
print(f"{'param':44s} {'A2-vs-A':>9s} {'B-vs-A':>9s} {'C-vs-A':>9s}")
for n in ga:
    da = ((ga2[n] - ga[n]).norm() / (ga[n].norm() + 1e-6)).item()
    db = ((gb[n] - ga[n]).norm() / (ga[n].norm() + 1e-6)).item()
    dc = ((gc[n] - ga[n]).norm() / (ga[n].norm() + 1e-6)).item()
    flag = " <<<" if max(db, dc) > 3 * max(da, 0.005) else ""
    print(f"{n:44s} {da:9.4f} {db:9.4f} {dc:9.4f}{flag}")
