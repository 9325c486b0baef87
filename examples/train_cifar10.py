#!/usr/bin/env python3
"""CIFAR-10 training (reference example/image-classification/train_cifar10.py)."""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from common import data, fit  # noqa: E402
from dtmx.models import get_symbol  # noqa: E402

if __name__ == "__main__":
    parser = argparse.ArgumentParser(description="train cifar10")
    fit.add_fit_args(parser)
    data.add_data_args(parser)
    parser.set_defaults(
        network="resnet", num_layers=18, kv_store="device",
        num_epochs=300, lr=0.05, lr_step_epochs="200,250",
        batch_size=128, image_shape="3,32,32", num_classes=10,
        num_examples=50000,
    )
    args = parser.parse_args()
    net = get_symbol(args.network, num_layers=args.num_layers,
                     num_classes=args.num_classes, image_shape=args.image_shape)
    fit.fit(args, net, data.get_rec_iter)
