"""KVStore semantics (reference tests/python/unittest/test_kvstore.py,
tests/nightly/dist_sync_kvstore.py). Distributed cases run as spawned gloo
subprocesses (world_size 2) — the reference's local-launcher pattern."""
import multiprocessing as mp
import os
import pickle

import pytest
import torch

import dtmx
from dtmx import kvstore as kvs
from dtmx.optimizer import SGD


def test_local_push_pull():
    kv = kvs.create("local")
    kv.init("w", torch.ones(4))
    kv.push("w", torch.full((4,), 2.0))
    out = torch.zeros(4)
    kv.pull("w", out=out)
    assert torch.allclose(out, torch.full((4,), 2.0))


def test_local_multi_value_push_sums():
    kv = kvs.create("device")
    kv.init("w", torch.zeros(3))
    kv.push("w", [torch.ones(3), torch.full((3,), 2.0)])
    out = torch.zeros(3)
    kv.pull("w", out=out)
    assert torch.allclose(out, torch.full((3,), 3.0))


def test_local_updater_applies_sgd():
    kv = kvs.create("local")
    opt = SGD(learning_rate=0.5, rescale_grad=1.0)
    kv.set_optimizer(opt)
    kv.init("w", torch.ones(4))
    kv.push("w", torch.full((4,), 1.0))  # w -= 0.5 * 1
    out = torch.zeros(4)
    kv.pull("w", out=out)
    assert torch.allclose(out, torch.full((4,), 0.5))


def test_aux_key_averaged_not_updated():
    """Keys initialized with exclude_update are averaged (reference
    kvstore_dist_server.h:353-360) even with an optimizer set."""
    kv = kvs.create("local")
    kv.set_optimizer(SGD(learning_rate=100.0))
    kv.init("bn_mean", torch.zeros(4), exclude_update=True)
    kv.push("bn_mean", [torch.ones(4), torch.full((4,), 3.0)])
    out = torch.zeros(4)
    kv.pull("bn_mean", out=out)
    assert torch.allclose(out, torch.full((4,), 2.0))  # (1+3)/2


def test_string_key_namespaces():
    kv = kvs.create("local")
    kv.init("w", torch.zeros(1))
    kv.init("aux", torch.zeros(1), exclude_update=True)
    assert kv._str_key_dict["w"] < kvs.MAX_ALLOWED_KEY_FOR_UPDATE
    assert kv._str_key_dict["aux"] >= kvs.MAX_ALLOWED_KEY_FOR_UPDATE


# ---------------------------------------------------------------- dist (gloo)

def _dist_worker(rank, world, port, fn_name, q):
    os.environ.update(
        RANK=str(rank), WORLD_SIZE=str(world),
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
        DTMX_BACKEND="gloo",
    )
    torch.manual_seed(rank)
    try:
        result = globals()[fn_name](rank, world)
        q.put(("ok", rank, result))
    except Exception as e:  # pragma: no cover
        import traceback
        q.put(("err", rank, traceback.format_exc()))


def run_dist(fn_name, world=2):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = torch.randint(20000, 40000, (1,)).item()
    procs = [ctx.Process(target=_dist_worker, args=(r, world, port, fn_name, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in procs:
        status, rank, payload = q.get(timeout=120)
        assert status == "ok", f"rank {rank} failed:\n{payload}"
        results[rank] = payload
    for p in procs:
        p.join(timeout=30)
    return results


def _check_dist_push_sums(rank, world):
    kv = kvs.create("dist_sync")
    kv.init("w", torch.zeros(4))
    kv.push("w", torch.full((4,), float(rank + 1)))  # no updater: stored = sum
    out = torch.zeros(4)
    kv.pull("w", out=out)
    kv.close()
    return out[0].item()


def _check_dist_updater_identical(rank, world):
    kv = kvs.create("dist_sync")
    kv.set_optimizer(SGD(learning_rate=0.1, rescale_grad=1.0 / world))
    kv.init("w", torch.ones(4))
    kv.push("w", torch.full((4,), float(rank + 1)))  # mean grad = 1.5
    out = torch.zeros(4)
    kv.pull("w", out=out)
    kv.close()
    return out.tolist()


def _check_dist_aux_average(rank, world):
    kv = kvs.create("dist_sync")
    kv.init("bn", torch.zeros(2), exclude_update=True)
    kv.push("bn", torch.full((2,), float(rank)))
    out = torch.zeros(2)
    kv.pull("bn", out=out)
    kv.close()
    return out[0].item()


def test_dist_push_sums():
    res = run_dist("_check_dist_push_sums")
    assert all(v == 3.0 for v in res.values())  # 1+2


def test_dist_updater_identical_on_all_ranks():
    res = run_dist("_check_dist_updater_identical")
    expect = [1.0 - 0.1 * 1.5] * 4
    for v in res.values():
        assert v == pytest.approx(expect)


def test_dist_aux_average():
    res = run_dist("_check_dist_aux_average")
    assert all(v == 0.5 for v in res.values())  # (0+1)/2
