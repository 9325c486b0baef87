#!/usr/bin/env python3
"""ImageNet-scale training (reference example/image-classification/
train_imagenet.py) — the BASELINE.md training-throughput config
(`--benchmark 1` = synthetic data)."""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from common import data, fit  # noqa: E402
from common.data import ETDataIterator  # noqa: E402
from dtmx.models import get_symbol  # noqa: E402

if __name__ == "__main__":
    parser = argparse.ArgumentParser(
        description="train imagenet",
        formatter_class=argparse.ArgumentDefaultsHelpFormatter,
    )
    fit.add_fit_args(parser)
    data.add_data_args(parser)
    parser.set_defaults(
        network="resnet", num_layers=50, kv_store="device",
        num_epochs=80, lr=0.1, lr_step_epochs="30,60",
        batch_size=128, image_shape="3,224,224", dtype="bfloat16",
    )
    args = parser.parse_args()

    net = get_symbol(args.network, num_layers=args.num_layers,
                     num_classes=args.num_classes, image_shape=args.image_shape)
    elastic = os.environ.get("ELASTIC_TRAINING_ENABLED", "0").lower() in ("1", "true")
    if elastic:
        # elastic mode: hand fit() the iterator FACTORY so shards re-derive
        # after membership changes (reference ETDataIterator, fit.py:31-44)
        fit.fit(args, net, lambda a, kv: (ETDataIterator(a, data.get_rec_iter), None))
    else:
        fit.fit(args, net, data.get_rec_iter)
