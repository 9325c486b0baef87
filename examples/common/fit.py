"""Training-loop driver with the reference CLI surface
(reference example/image-classification/common/fit.py:94-356).

dtmx port: same flags, same flow (kvstore creation, LR schedule from
--lr-step-epochs with epoch size recomputed per worker count, checkpointing,
Speedometer) on the dtmx Module/KVStore stack.
"""
from __future__ import annotations

import argparse
import logging
import os
import time

import torch

import dtmx
from dtmx import callback as cb
from dtmx import lr_scheduler as lrs


def add_fit_args(parser: argparse.ArgumentParser):
    train = parser.add_argument_group("Training", "model training")
    train.add_argument("--network", type=str, help="the neural network to use")
    train.add_argument("--num-layers", type=int, help="number of layers (resnet/vgg)")
    train.add_argument("--gpus", type=str, default=None,
                       help="list of gpus to run, e.g. 0 or 0,2,5; empty means cpu")
    train.add_argument("--kv-store", type=str, default="device")
    train.add_argument("--num-epochs", type=int, default=100)
    train.add_argument("--lr", type=float, default=0.1)
    train.add_argument("--lr-factor", type=float, default=0.1)
    train.add_argument("--lr-step-epochs", type=str, default=None)
    train.add_argument("--initializer", type=str, default="default")
    train.add_argument("--optimizer", type=str, default="sgd")
    train.add_argument("--mom", type=float, default=0.9)
    train.add_argument("--wd", type=float, default=0.0001)
    train.add_argument("--batch-size", type=int, default=128)
    train.add_argument("--disp-batches", type=int, default=20)
    train.add_argument("--model-prefix", type=str, default=None)
    train.add_argument("--save-period", type=int, default=1)
    train.add_argument("--load-epoch", type=int, default=None)
    train.add_argument("--top-k", type=int, default=0)
    train.add_argument("--dtype", type=str, default="float32",
                       help="precision: float32, float16 or bfloat16")
    train.add_argument("--warmup-epochs", type=int, default=5)
    train.add_argument("--warmup-strategy", type=str, default="linear")
    train.add_argument("--gc-type", type=str, default="none",
                       help="gradient compression: 2bit or none")
    train.add_argument("--gc-threshold", type=float, default=0.5)
    train.add_argument("--test-io", type=int, default=0)
    train.add_argument("--benchmark", type=int, default=0,
                       help="1 = synthetic data benchmark mode")
    train.add_argument("--profile-worker-suffix", type=str, default="")
    return train


def _get_lr_scheduler(args, kv, epoch_size):
    if not args.lr_step_epochs:
        return args.lr, None
    begin_epoch = args.load_epoch or 0
    step_epochs = [int(l) for l in args.lr_step_epochs.split(",")]
    lr = args.lr
    for s in step_epochs:
        if begin_epoch >= s:
            lr *= args.lr_factor
    steps = [epoch_size * (x - begin_epoch) for x in step_epochs if x - begin_epoch > 0]
    sched = lrs.MultiFactorScheduler(step=steps, factor=args.lr_factor)
    if args.warmup_epochs > 0:
        sched = lrs.WarmupScheduler(lr, args.warmup_epochs * epoch_size, after=sched)
    return lr, sched


def fit(args, network, data_loader, **kwargs):
    """network: an nn.Module (dtmx.models builder output);
    data_loader: fn(args, kv) -> (train_iter, val_iter)."""
    kv = dtmx.kvstore.create(args.kv_store)
    if args.gc_type and args.gc_type != "none":
        kv.set_gradient_compression({"type": args.gc_type, "threshold": args.gc_threshold})

    head = "%(asctime)-15s Node[" + str(kv.rank) + "] %(message)s"
    logging.basicConfig(level=logging.INFO, format=head)
    logging.info("start with arguments %s", args)

    epoch_size = max(1, args.num_examples // args.batch_size // kv.num_workers) \
        if getattr(args, "num_examples", 0) else None

    train, val = data_loader(args, kv)
    if args.test_io:
        tic = time.time()
        for i, batch in enumerate(train):
            if i == 50:
                break
        logging.info("io speed: %.2f samples/sec",
                     50 * args.batch_size / (time.time() - tic))
        return

    if args.gpus:
        devs = [dtmx.gpu(int(i)) for i in args.gpus.split(",")]
    else:
        devs = [dtmx.cpu()]
    dtype = getattr(torch, args.dtype)

    mod = dtmx.Module(network, context=devs)
    data_shape = (args.batch_size,) + tuple(train.provide_data[0].shape[1:])
    mod.bind(data_shapes=[("data", data_shape)],
             label_shapes=[("softmax_label", (args.batch_size,))], dtype=dtype)

    arg_params = aux_params = None
    begin_epoch = 0
    if args.load_epoch is not None and args.model_prefix:
        _, arg_params, aux_params = dtmx.model.load_checkpoint(args.model_prefix,
                                                               args.load_epoch)
        begin_epoch = args.load_epoch

    lr, sched = _get_lr_scheduler(args, kv, epoch_size or 100)
    optimizer_params = {
        "learning_rate": lr,
        "wd": args.wd,
        "lr_scheduler": sched,
    }
    if args.optimizer in ("sgd", "lbsgd"):
        optimizer_params["momentum"] = args.mom

    checkpoint = None
    if args.model_prefix and kv.rank == 0:
        checkpoint = cb.do_checkpoint(args.model_prefix, args.save_period)

    eval_metrics = ["accuracy"]
    if args.top_k > 0:
        eval_metrics.append(dtmx.metric.create("top_k", top_k=args.top_k))

    if args.profile_worker_suffix:
        dtmx.profiler.set_config(filename=f"rank{kv.rank}_{args.profile_worker_suffix}")
        dtmx.profiler.set_state("run")

    mod.fit(
        train,
        eval_data=val,
        eval_metric=eval_metrics,
        kvstore=kv,
        optimizer=args.optimizer,
        optimizer_params=tuple(optimizer_params.items()),
        initializer=dtmx.initializer.create(args.initializer),
        arg_params=arg_params,
        aux_params=aux_params,
        begin_epoch=begin_epoch,
        num_epoch=args.num_epochs,
        batch_end_callback=[cb.Speedometer(args.batch_size, args.disp_batches)],
        epoch_end_callback=checkpoint,
        **kwargs,
    )

    if args.profile_worker_suffix:
        dtmx.profiler.set_state("stop")
    return mod
