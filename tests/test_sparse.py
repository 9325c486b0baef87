"""Embedding/take + row-sparse gradient path (reference
src/operator/tensor/indexing_op.cu, kvstore_dist.h:452-528 PushRowSparse/
PullRowSparse_, optimizer_op-inl.h:563-676 row-sparse optimizer kernels)."""
import multiprocessing as mp
import os

import numpy as np
import pytest
import torch

import dtmx
from dtmx.ops import functional as DF
from dtmx.ops.layers import Embedding


def test_take_forward_backward_cpu():
    torch.manual_seed(0)
    table = torch.randn(20, 8, requires_grad=True)
    idx = torch.tensor([[1, 3], [3, 19]])
    out = DF.take(table, idx)
    assert out.shape == (2, 2, 8)
    torch.testing.assert_close(out[0, 0], table[1])
    torch.testing.assert_close(out[1, 1], table[19])
    out.sum().backward()
    # row 3 appears twice -> grad 2, rows 1/19 once, others 0
    expect = torch.zeros(20, 8)
    expect[1] = 1
    expect[3] = 2
    expect[19] = 1
    torch.testing.assert_close(table.grad, expect)


def test_take_oob_clips():
    table = torch.arange(12.0).reshape(3, 4)
    out = DF.take(table, torch.tensor([-5, 7]))
    torch.testing.assert_close(out[0], table[0])
    torch.testing.assert_close(out[1], table[2])


def test_embedding_module_trains():
    torch.manual_seed(0)
    emb = Embedding(50, 16)
    idx = torch.randint(0, 50, (8,))
    out = emb(idx)
    loss = out.float().pow(2).sum()
    loss.backward()
    assert emb.weight.grad is not None
    touched = emb.weight.grad.abs().sum(dim=1) > 0
    assert set(idx.tolist()) == set(torch.nonzero(touched).reshape(-1).tolist())


def test_row_sparse_grad_matches_dense():
    torch.manual_seed(1)
    dy = torch.randn(6, 8)
    idx = torch.tensor([2, 2, 5, 0, 5, 5])
    rs = DF.embedding_row_sparse_grad(dy, idx, vocab=10)
    assert rs.rows.tolist() == [0, 2, 5]
    dense = torch.zeros(10, 8)
    dense.index_add_(0, idx, dy)
    torch.testing.assert_close(rs.to_dense(), dense, rtol=1e-5, atol=1e-6)


def test_local_kvstore_row_sparse_push_sgd():
    """Dense-updater fallback: row-sparse push == dense push of the same
    gradient (SGD has no row-local state shortcut)."""
    kv1 = dtmx.kvstore.create("local")
    kv2 = dtmx.kvstore.create("local")
    from dtmx.optimizer import create as opt_create

    for kv in (kv1, kv2):
        kv.set_optimizer(opt_create("sgd", learning_rate=0.5))
    torch.manual_seed(2)
    w = torch.randn(10, 4)
    kv1.init("emb", w.clone())
    kv2.init("emb", w.clone())
    rows = torch.tensor([1, 4])
    vals = torch.randn(2, 4)
    dense = torch.zeros(10, 4)
    dense[rows] = vals
    kv1.push("emb", DF.RowSparse(rows, vals, (10, 4)))
    kv2.push("emb", dense)
    torch.testing.assert_close(kv1.pull("emb")[0], kv2.pull("emb")[0],
                               rtol=1e-5, atol=1e-6)


def test_local_kvstore_row_sparse_push_group_adagrad():
    """Lazy row update: only touched rows move (update_rows path)."""
    from dtmx.optimizer import create as opt_create

    kv = dtmx.kvstore.create("local")
    kv.set_optimizer(opt_create("groupadagrad", learning_rate=0.5))
    torch.manual_seed(3)
    w = torch.randn(10, 4)
    kv.init("emb", w.clone())
    rows = torch.tensor([1, 4])
    vals = torch.randn(2, 4)
    kv.push("emb", DF.RowSparse(rows, vals, (10, 4)))
    got = kv.pull("emb")[0]
    untouched = [i for i in range(10) if i not in (1, 4)]
    torch.testing.assert_close(got[untouched], w[untouched], rtol=0, atol=0)
    assert (got[rows] - w[rows]).abs().sum() > 0


def test_row_sparse_pull_selects_rows():
    kv = dtmx.kvstore.create("local")
    w = torch.arange(40.0).reshape(10, 4)
    kv.init("emb", w.clone())
    out = torch.full((10, 4), 99.0)
    kv.row_sparse_pull("emb", out=out, row_ids=torch.tensor([0, 7]))
    assert out[0].tolist() == w[0].tolist()
    assert out[7].tolist() == w[7].tolist()
    assert out[1].abs().sum() == 0  # unselected rows zeroed


# ------------------------------------------------ distributed (gloo, world 2)

def _dist_worker(rank, world, port, q):
    os.environ.update(
        RANK=str(rank), WORLD_SIZE=str(world),
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
        DTMX_BACKEND="gloo",
    )
    try:
        import dtmx
        from dtmx.optimizer import create as opt_create
        from dtmx.ops import functional as DF

        kv = dtmx.kvstore.create("dist_sync")
        kv.set_optimizer(opt_create("groupadagrad", learning_rate=0.5))
        torch.manual_seed(0)  # same init everywhere
        w = torch.randn(10, 4)
        kv.init("emb", w.clone())
        # each rank touches a different (overlapping) row set
        if rank == 0:
            rows = torch.tensor([1, 4])
            vals = torch.ones(2, 4)
        else:
            rows = torch.tensor([4, 7])
            vals = torch.full((2, 4), 2.0)
        kv.push("emb", DF.RowSparse(rows, vals, (10, 4)))
        got = kv.pull("emb")[0]
        kv.close()
        q.put(("ok", rank, {"w0": w.numpy().tolist(),
                            "got": got.numpy().tolist()}))
    except Exception:
        import traceback
        q.put(("err", rank, traceback.format_exc()))


def _dist_compress_worker(rank, world, port, q):
    os.environ.update(
        RANK=str(rank), WORLD_SIZE=str(world),
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
        DTMX_BACKEND="gloo",
    )
    try:
        import dtmx

        kv = dtmx.kvstore.create("dist_sync")
        kv.set_gradient_compression({"type": "2bit", "threshold": 0.5})
        kv.init("a", torch.zeros(8))
        kv.init("b", torch.zeros(8))
        # 0.3-gradients: quantize to 0 (residual 0.3) on push 1, to 0.5 on
        # push 2 — per-key residuals must not alias (kvstore_dist.h:778)
        for _ in range(2):
            kv.push("a", torch.full((8,), 0.3))
            kv.push("b", torch.full((8,), 0.3))
        a = kv.pull("a")[0].clone()
        b = kv.pull("b")[0].clone()
        kv.close()
        q.put(("ok", rank, {"a": a.tolist(), "b": b.tolist()}))
    except Exception:
        import traceback
        q.put(("err", rank, traceback.format_exc()))


@pytest.mark.timeout(180)
def test_dist_compressed_push_per_key_residuals():
    """2-bit compression through DistKVStore.push on gloo world 2: stored =
    sum over ranks of quantized grads; second push crosses the threshold
    via each key's OWN residual (reference dist_sync_kvstore.py compression
    checks + the per-key residual fix)."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = torch.randint(20000, 40000, (1,)).item()
    procs = [ctx.Process(target=_dist_compress_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    res = {}
    for _ in procs:
        status, rank, payload = q.get(timeout=120)
        assert status == "ok", payload
        res[rank] = payload
    for p in procs:
        p.join(timeout=30)
    assert res[0] == res[1]
    # push 1: both ranks quantize 0.3 -> 0; push 2: residual 0.3+0.3 -> 0.5
    # per rank, summed over 2 ranks = 1.0 per element (no updater: SUM)
    assert res[0]["a"] == [1.0] * 8, res[0]["a"]
    assert res[0]["b"] == [1.0] * 8, res[0]["b"]


@pytest.mark.timeout(180)
def test_dist_row_sparse_push_merges_rows():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = torch.randint(20000, 40000, (1,)).item()
    procs = [ctx.Process(target=_dist_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    res = {}
    for _ in procs:
        status, rank, payload = q.get(timeout=120)
        assert status == "ok", payload
        res[rank] = payload
    for p in procs:
        p.join(timeout=30)
    # replicated result on both ranks
    assert res[0]["got"] == res[1]["got"]
    w0 = np.array(res[0]["w0"])
    got = np.array(res[0]["got"])
    # rows {1,4,7} updated, others untouched; row 4 got BOTH contributions:
    # its history entry is larger, so its |step| differs from row 1's
    changed = np.abs(got - w0).sum(axis=1) > 0
    assert set(np.nonzero(changed)[0].tolist()) == {1, 4, 7}
