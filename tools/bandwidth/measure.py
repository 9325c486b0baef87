#!/usr/bin/env python3
"""Communication bandwidth harness (reference tools/bandwidth/measure.py):
creates a model's full gradient key set, repeatedly all-reduces it through
the dtmx collective engine, and reports GB/s per GPU plus numeric error —
the tool for tuning xGMI bucket sizes (DTMX_BUCKET_MB).

Run under torchrun (one rank per GPU):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 tools/bandwidth/measure.py --network resnet
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))))

import torch  # noqa: E402
import torch.distributed as dist  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--network", type=str, default="resnet")
    ap.add_argument("--num-layers", type=int, default=50)
    ap.add_argument("--image-shape", type=str, default="3,224,224")
    ap.add_argument("--num-iters", type=int, default=20)
    ap.add_argument("--bucket-mb", type=float, default=None)
    ap.add_argument("--dtype", type=str, default="bfloat16")
    ap.add_argument("--test-gradient-compression", type=int, default=0)
    args = ap.parse_args()
    if args.bucket_mb:
        os.environ["DTMX_BUCKET_MB"] = str(args.bucket_mb)

    from dtmx.models import get_symbol
    from dtmx.parallel.bucketer import GradBucketer

    backend = "nccl" if torch.cuda.is_available() else "gloo"
    if "RANK" not in os.environ:  # standalone single-process run
        os.environ.setdefault("RANK", "0")
        os.environ.setdefault("WORLD_SIZE", "1")
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29561")
    dist.init_process_group(backend=backend)
    rank = dist.get_rank()
    world = dist.get_world_size()
    if torch.cuda.is_available():
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)))
        device = torch.device("cuda", int(os.environ.get("LOCAL_RANK", rank)))
        dtype = getattr(torch, args.dtype)
    else:
        device = torch.device("cpu")
        dtype = torch.float32

    net = get_symbol(args.network, num_layers=args.num_layers,
                     image_shape=args.image_shape).to(device, dtype)
    bucketer = GradBucketer(list(net.parameters()))
    total_bytes = sum(b.numel() * b.element_size() for b in bucketer.flat)

    def sweep():
        bucketer.zero_grad()
        for p in bucketer.params:  # simulate backward completion order
            p.grad.fill_(1.0)
        for bi in range(len(bucketer.flat)):
            bucketer._works.append(
                (bi, dist.all_reduce(bucketer.flat[bi], op=dist.ReduceOp.SUM, async_op=True))
            )
        bucketer.finish()

    for _ in range(3):
        sweep()
    if device.type == "cuda":
        torch.cuda.synchronize()
    dist.barrier()
    tic = time.time()
    for _ in range(args.num_iters):
        sweep()
    if device.type == "cuda":
        torch.cuda.synchronize()
    dist.barrier()
    elapsed = time.time() - tic

    # ring all-reduce moves 2*(W-1)/W of the payload per GPU
    algo_bytes = total_bytes * 2 * (world - 1) / world
    gbps = algo_bytes * args.num_iters / elapsed / 1e9
    err = max((b - float(world)).abs().max().item() for b in bucketer.flat)
    if rank == 0:
        print(f"network={args.network}-{args.num_layers} world={world} "
              f"payload={total_bytes/1e6:.1f} MB buckets={len(bucketer.flat)} "
              f"time/iter={elapsed/args.num_iters*1e3:.2f} ms "
              f"algo-bandwidth={gbps:.1f} GB/s/GPU max-err={err:.1e}")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
