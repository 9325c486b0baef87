#!/usr/bin/env python3
"""Inference throughput benchmark (reference example/image-classification/
benchmark_score.py — the BASELINE.md inference tables): images/sec per
network per batch size, synthetic data, forward only."""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

import dtmx  # noqa: E402
from dtmx.models import get_symbol  # noqa: E402


def score(network, batch_size, dtype, dev, image_shape=(3, 224, 224), steps=30,
          **net_kwargs):
    net = get_symbol(network, num_classes=1000,
                     image_shape=",".join(map(str, image_shape)), **net_kwargs)
    mod = dtmx.Module(net, context=dev)
    shape = (batch_size,) + image_shape
    mod.bind(data_shapes=[("data", shape)], label_shapes=None,
             for_training=False, dtype=dtype)
    mod.init_params()
    device = dev.torch_device()
    data = torch.randn(shape, dtype=dtype, device=device)
    if device.type == "cuda":
        data = data.contiguous(memory_format=torch.channels_last)
    from dtmx.io import DataBatch

    batch = DataBatch(data=[data], label=None)
    for _ in range(5):
        mod.forward(batch, is_train=False)
    if device.type == "cuda":
        torch.cuda.synchronize()
    tic = time.time()
    for _ in range(steps):
        mod.forward(batch, is_train=False)
    if device.type == "cuda":
        torch.cuda.synchronize()
    return steps * batch_size / (time.time() - tic)


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=str, default="0" if torch.cuda.is_available() else None)
    ap.add_argument("--batch-sizes", type=str, default="1,32,128")
    ap.add_argument("--networks", type=str,
                    default="alexnet,vgg,inception-bn,inception-v3,resnet,resnet-152")
    ap.add_argument("--dtype", type=str, default="bfloat16")
    args = ap.parse_args()
    dev = dtmx.gpu(int(args.gpus.split(",")[0])) if args.gpus else dtmx.cpu()
    dtype = getattr(torch, args.dtype) if dev.torch_device().type == "cuda" else torch.float32

    for net in args.networks.split(","):
        kwargs = {}
        name = net
        if net == "resnet-152":
            name, kwargs = "resnet", {"num_layers": 152}
        elif net == "resnet":
            kwargs = {"num_layers": 50}
        elif net == "vgg":
            kwargs = {"num_layers": 16}
        shape = (3, 299, 299) if "inception" in net else (3, 224, 224)
        for bs in (int(b) for b in args.batch_sizes.split(",")):
            try:
                ips = score(name, bs, dtype, dev, image_shape=shape, **kwargs)
                print(f"network: {net:14s} batch: {bs:4d}  {ips:10.1f} images/sec")
            except Exception as e:  # noqa: BLE001
                print(f"network: {net:14s} batch: {bs:4d}  FAILED: {e}")
