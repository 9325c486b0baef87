"""Data loaders for the examples (reference example/image-classification/
common/data.py). No network in this environment: real datasets load from
local idx/npz files when present, otherwise deterministic synthetic data of
the same shape is used (stated in logs)."""
from __future__ import annotations

import logging
import os

import numpy as np
import torch

from dtmx.io import DataIter, MNISTIter, NDArrayIter, SyntheticDataIter


def add_data_args(parser):
    data = parser.add_argument_group("Data", "the input data")
    data.add_argument("--data-train", type=str, default=None)
    data.add_argument("--data-val", type=str, default=None)
    data.add_argument("--num-classes", type=int, default=1000)
    data.add_argument("--num-examples", type=int, default=1281167)
    data.add_argument("--image-shape", type=str, default="3,224,224")
    data.add_argument("--data-nthreads", type=int, default=4)
    # augmentation flags (reference common/data.py add_data_aug_args subset)
    data.add_argument("--random-crop", type=int, default=1)
    data.add_argument("--random-mirror", type=int, default=1)
    data.add_argument("--resize", type=int, default=0,
                      help="resize shorter side before crop (0 = off)")
    return data


def _synthetic_pair(args, kv, image_shape, device=None, dtype=torch.float32):
    logging.info("using synthetic data (no dataset files found)")
    train = SyntheticDataIter(
        args.num_classes, (args.batch_size,) + image_shape,
        max_iter=max(1, args.num_examples // args.batch_size // max(1, kv.num_workers)),
        dtype=dtype, device=device,
    )
    val = SyntheticDataIter(args.num_classes, (args.batch_size,) + image_shape,
                            max_iter=8, dtype=dtype, device=device)
    return train, val


def get_rec_iter(args, kv):
    """ImageNet-style iterator, sharded by (kv.rank, kv.num_workers)
    (reference get_rec_iter: shards RecordIO by rank). Synthetic fallback."""
    image_shape = tuple(int(x) for x in args.image_shape.split(","))
    if getattr(args, "benchmark", 0) or not args.data_train:
        return _synthetic_pair(args, kv, image_shape)
    if args.data_train.endswith(".rec") and os.path.exists(args.data_train):
        from dtmx.io import ImageRecordIter

        train = ImageRecordIter(
            args.data_train, image_shape, args.batch_size, shuffle=True,
            part_index=kv.rank, num_parts=kv.num_workers,
            preprocess_threads=args.data_nthreads,
            rand_crop=bool(getattr(args, "random_crop", 1)),
            rand_mirror=bool(getattr(args, "random_mirror", 1)),
            resize=getattr(args, "resize", 0),
        )
        val = None
        if args.data_val and os.path.exists(args.data_val):
            val = ImageRecordIter(
                args.data_val, image_shape, args.batch_size,
                preprocess_threads=args.data_nthreads,
                resize=getattr(args, "resize", 0),
            )
        return train, val
    if args.data_train.endswith(".npz") and os.path.exists(args.data_train):
        blob = np.load(args.data_train)
        train = NDArrayIter({"data": blob["x"]}, {"softmax_label": blob["y"]},
                            args.batch_size, shuffle=True,
                            part_index=kv.rank, num_parts=kv.num_workers)
        val = None
        if args.data_val and os.path.exists(args.data_val):
            vb = np.load(args.data_val)
            val = NDArrayIter({"data": vb["x"]}, {"softmax_label": vb["y"]},
                              args.batch_size)
        return train, val
    return _synthetic_pair(args, kv, image_shape)


def get_mnist_iter(args, kv):
    """MNIST (reference train_mnist.py get_mnist_iter); reads idx files from
    --data-dir when present, synthetic digits otherwise."""
    data_dir = getattr(args, "data_dir", "data")
    img = os.path.join(data_dir, "train-images-idx3-ubyte")
    lab = os.path.join(data_dir, "train-labels-idx1-ubyte")
    flat = getattr(args, "flat", False)
    train = MNISTIter(image=img, label=lab, batch_size=args.batch_size,
                      shuffle=True, flat=flat, part_index=kv.rank,
                      num_parts=kv.num_workers,
                      num_examples=getattr(args, "num_examples", 8192))
    val = MNISTIter(image=img.replace("train", "t10k"),
                    label=lab.replace("train", "t10k"),
                    batch_size=args.batch_size, shuffle=False, flat=flat,
                    seed=99, num_examples=1024)
    return train, val


class ETDataIterator:
    """Elastic data-iterator factory (reference common/fit.py:31-44
    ETDataIterator): re-creates shards for the current (rank, num_workers)
    when the membership changes. Pass as Module.fit(train_data=...)."""

    def __init__(self, args, loader):
        self.args = args
        self.loader = loader
        self.val = None

    def __call__(self, kv):
        train, val = self.loader(self.args, kv)
        self.val = val
        return train
