"""dist_async semantics (reference kvstore 'dist_async': no push barrier):
dtmx implements it as one-step-delayed pipelined collectives — step t's
update consumes step t-1's reduced gradients."""
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
        from dtmx.io import DataBatch
        from dtmx.models import get_symbol

        torch.manual_seed(0)
        np.random.seed(0)
        net = get_symbol("mlp", num_classes=4, input_dim=8, )
        mod = dtmx.Module(net, context=dtmx.cpu())
        mod.bind(data_shapes=[("data", (2, 8))], label_shapes=[("softmax_label", (2,))])
        kv = dtmx.kvstore.create("dist_async")
        mod.init_params()
        mod.init_optimizer(kvstore=kv, optimizer_params=(("learning_rate", 0.1),))
        p0 = [p.detach().clone() for p in mod.symbol.parameters()]
        data = torch.randn(2, 8, generator=torch.Generator().manual_seed(rank))
        label = torch.tensor([1.0, 2.0])
        batch = DataBatch(data=[data], label=[label])
        # step 1: delayed grads are zeros -> params unchanged
        mod.forward_backward(batch)
        mod.update()
        unchanged = all(
            torch.equal(a, b) for a, b in zip(p0, mod.symbol.parameters())
        )
        # step 2: uses step-1's reduced grads -> params move
        mod.forward_backward(batch)
        mod.update()
        moved = any(
            not torch.allclose(a, b) for a, b in zip(p0, mod.symbol.parameters())
        )
        psum = float(sum(p.double().sum().item() for p in mod.symbol.parameters()))
        kv.close()
        q.put(("ok", rank, {"unchanged": unchanged, "moved": moved, "psum": psum}))
    except Exception:
        import traceback
        q.put(("err", rank, traceback.format_exc()))


@pytest.mark.timeout(180)
def test_async_one_step_delay():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = torch.randint(20000, 40000, (1,)).item()
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    res = {}
    for _ in procs:
        status, rank, payload = q.get(timeout=120)
        assert status == "ok", payload
        res[rank] = payload
    for p in procs:
        p.join(timeout=30)
    assert res[0]["unchanged"] and res[1]["unchanged"]
    assert res[0]["moved"] and res[1]["moved"]
    # replicated invariant holds in async mode too (same delayed grads)
    assert res[0]["psum"] == pytest.approx(res[1]["psum"], rel=0, abs=0)
