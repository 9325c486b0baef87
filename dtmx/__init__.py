"""dtmx: MI355X-native elastic data-parallel training framework with the
capabilities (and Python surface) of the Dynamic Training MXNet fork
(awslabs/dynamic-training-with-apache-mxnet-on-aws). See README.md.

Usage mirrors `import mxnet as mx`:

    import dtmx as mx
    kv = mx.kvstore.create('dist_sync')
    net = mx.models.get_symbol('resnet', num_layers=50)
    mod = mx.mod.Module(net, context=mx.gpu(0))
    mod.bind(...); mod.fit(train_iter, kvstore=kv, ...)
"""
import os as _os

# NaiveEngine analog (reference MXNET_ENGINE_TYPE=NaiveEngine, engine.cc:32-48):
# serialize every kernel launch for race bisection. Must be set before HIP
# initializes, hence here at package import.
if _os.environ.get("DTMX_BLOCKING", "0") == "1":
    _os.environ.setdefault("AMD_SERIALIZE_KERNEL", "3")
    _os.environ.setdefault("HIP_LAUNCH_BLOCKING", "1")

from . import callback, context, initializer, io, lr_scheduler, metric, model, monitor
from . import gluon, kvstore, models, ndarray, optimizer, parallel, profiler, recordio
from . import autograd, image, nd, random, rnn, symbol, visualization
from . import module as mod
from .context import Context, cpu, gpu, num_gpus
from .module import Module

# dtmx.nd is the mx.nd-style op namespace (nd.py); it re-exports the
# byte-compatible save/load/array/zeros from dtmx.ndarray, so the round-1
# `nd = ndarray` alias is subsumed

__version__ = "0.1.0"
