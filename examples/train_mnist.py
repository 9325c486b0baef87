#!/usr/bin/env python3
"""MNIST MLP/LeNet training (reference example/image-classification/
train_mnist.py). BASELINE.json config 1: MLP, kvstore='local', CPU-capable."""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from common import data, fit  # noqa: E402
from dtmx.models import get_symbol  # noqa: E402

if __name__ == "__main__":
    parser = argparse.ArgumentParser(description="train mnist")
    parser.add_argument("--data-dir", type=str, default="data")
    fit.add_fit_args(parser)
    parser.set_defaults(
        network="mlp", num_layers=2, kv_store="local", num_epochs=10,
        lr=0.05, batch_size=64, disp_batches=50, num_classes=10,
        num_examples=8192, image_shape="1,28,28",
    )
    parser.add_argument("--num-classes", type=int, default=10)
    parser.add_argument("--num-examples", type=int, default=8192)
    args = parser.parse_args()
    args.flat = args.network == "mlp"

    if args.network == "mlp":
        net = get_symbol("mlp", num_classes=args.num_classes, input_dim=784)
    else:
        net = get_symbol(args.network, num_classes=args.num_classes)
    fit.fit(args, net, data.get_mnist_iter)
