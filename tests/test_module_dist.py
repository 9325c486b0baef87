"""Distributed Module training (gloo, world 2): bucketed all-reduce overlap
path vs single-process reference — the dist_sync semantics check
(reference tests/nightly/dist_sync_kvstore.py / dist_lenet.py)."""
import json
import multiprocessing as mp
import os

import numpy as np
import pytest
import torch


def _worker(rank, world, port, q):
    os.environ.update(
        RANK=str(rank), WORLD_SIZE=str(world),
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
        DTMX_BACKEND="gloo",
    )
    try:
        import dtmx
        from dtmx.io import NDArrayIter
        from dtmx.models import get_symbol

        torch.manual_seed(0)
        np.random.seed(0)
        net = get_symbol("mlp", num_classes=10, input_dim=16)
        mod = dtmx.Module(net, context=dtmx.cpu())
        mod.bind(data_shapes=[("data", (4, 16))], label_shapes=[("softmax_label", (4,))])
        kv = dtmx.kvstore.create("dist_sync")
        rng = np.random.RandomState(7)
        X = rng.randn(64, 16).astype(np.float32)
        Y = rng.randint(0, 10, 64).astype(np.float32)
        it = NDArrayIter({"data": X}, {"softmax_label": Y}, 4,
                         part_index=kv.rank, num_parts=kv.num_workers)
        mod.fit(it, kvstore=kv, num_epoch=3,
                optimizer_params=(("learning_rate", 0.1), ("momentum", 0.9)))
        arg, _ = mod.get_params()
        psum = float(sum(p.double().sum().item() for p in arg.values()))
        rescale = mod._optimizer.rescale_grad
        kv.close()
        q.put(("ok", rank, {"psum": psum, "rescale": rescale}))
    except Exception:
        import traceback
        q.put(("err", rank, traceback.format_exc()))


def test_dist_module_params_stay_identical():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = torch.randint(20000, 40000, (1,)).item()
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    res = {}
    for _ in procs:
        status, rank, payload = q.get(timeout=180)
        assert status == "ok", payload
        res[rank] = payload
    for p in procs:
        p.join(timeout=30)
    # replicated update invariant
    assert res[0]["psum"] == pytest.approx(res[1]["psum"], rel=0, abs=0)
    # rescale_grad = 1/(B*W) with per-worker batch 4, world 2
    assert res[0]["rescale"] == pytest.approx(1.0 / 8)
