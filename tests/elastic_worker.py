"""Worker body for the elastic integration test (runs as a subprocess)."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

import dtmx
from dtmx.io import NDArrayIter
from dtmx.models import get_symbol


def main():
    out_path = os.environ["ELASTIC_TEST_OUT"]
    if os.environ.get("ELASTIC_TEST_OUT_PER_WID") == "1":
        out_path += "." + os.environ.get("DMLC_WORKER_ID", "w").replace("#", "_")
    epoch_sleep = float(os.environ.get("EPOCH_SLEEP", "0.25"))
    num_epoch = int(os.environ.get("NUM_EPOCH", "8"))
    torch.manual_seed(0)
    np.random.seed(0)

    # ELASTIC_TEST_DEVICE=cuda: run the same scenario on GPU with the RCCL
    # backend (several ranks sharing one device — RCCL permits this), the
    # hardware proof of the elastic comm destroy/re-form path.
    use_cuda = os.environ.get("ELASTIC_TEST_DEVICE") == "cuda"
    ctx = dtmx.gpu(0) if use_cuda else dtmx.cpu()
    net = get_symbol("mlp", num_classes=10, input_dim=32)
    mod = dtmx.Module(net, context=ctx)
    mod.bind(data_shapes=[("data", (8, 32))], label_shapes=[("softmax_label", (8,))])
    kv = dtmx.kvstore.create("dist_sync")

    rng = np.random.RandomState(42)  # identical dataset on every worker
    X = rng.randn(256, 32).astype(np.float32)
    Y = rng.randint(0, 10, 256).astype(np.float32)
    counts = []

    def data_factory(kv_):
        counts.append(kv_.num_workers)
        return NDArrayIter({"data": X}, {"softmax_label": Y}, 8,
                           part_index=kv_.rank, num_parts=kv_.num_workers)

    def epoch_cb(epoch, sym, arg, aux):
        time.sleep(epoch_sleep)

    mod.fit(
        data_factory,
        kvstore=kv,
        num_epoch=num_epoch,
        optimizer_params=(("learning_rate", 0.05), ("momentum", 0.9)),
        elastic_training=True,
        epoch_end_callback=epoch_cb,
    )
    arg, aux = mod.get_params()
    param_sum = float(sum(p.double().sum().item() for p in arg.values()))
    import torch.distributed as tdist

    result = {
        "wid": os.environ.get("DMLC_WORKER_ID"),
        "worker_counts": counts,
        "param_sum": param_sum,
        "final_workers": kv.num_workers,
        "final_rank": kv.rank,
        "backend": tdist.get_backend() if tdist.is_initialized() else None,
        "device": "cuda" if use_cuda else "cpu",
        "generation": getattr(kv._elastic, "version", 0),
    }
    with open(out_path, "w") as f:
        json.dump(result, f)
    kv.close()


if __name__ == "__main__":
    main()
