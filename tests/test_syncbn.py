"""SyncBatchNorm: cross-rank statistics (reference contrib/sync_batch_norm.cu).

World-2 gloo check: each rank normalizes its shard of a batch with SyncBN; the
result must match single-process BN over the WHOLE batch (y, dx, running
stats), and the sum of per-rank dgamma/dbeta must equal the full-batch grads
(the kvstore gradient all-reduce supplies that sum in real training).
"""
import os

import torch
import torch.multiprocessing as mp
import torch.nn.functional as F


def _worker(rank, world, port, q):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from dtmx.ops.layers import SyncBatchNorm2dNHWC

        torch.manual_seed(7)  # same full batch on every rank
        X = torch.randn(8, 16, 6, 6)
        DY = torch.randn(8, 16, 6, 6)
        n = X.shape[0] // world
        x = X[rank * n:(rank + 1) * n].clone().requires_grad_(True)

        bn = SyncBatchNorm2dNHWC(16, momentum=0.9)
        bn.train()
        y = bn(x)
        y.backward(DY[rank * n:(rank + 1) * n])
        out = {"y": y.detach(), "dx": x.grad,
               "dgamma": bn.weight.grad, "dbeta": bn.bias.grad,
               "rm": bn.running_mean, "rv": bn.running_var}
        # pickle by value (numpy) — shm-shared tensors die with the producer
        q.put(("ok", rank, {k: v.numpy().copy() for k, v in out.items()}))
    except Exception:  # pragma: no cover
        import traceback
        q.put(("err", rank, traceback.format_exc()))
    finally:
        dist.destroy_process_group()


def test_syncbn_matches_full_batch_bn():
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = torch.randint(20000, 40000, (1,)).item()
    procs = [ctx.Process(target=_worker, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    res = {}
    for _ in procs:
        status, rank, payload = q.get(timeout=120)
        assert status == "ok", f"rank {rank} failed:\n{payload}"
        res[rank] = payload
    for p in procs:
        p.join(timeout=30)
    res = {r: {k: torch.from_numpy(v) for k, v in d.items()}
           for r, d in res.items()}

    # single-process reference over the whole batch (torch momentum = 1 - 0.9)
    torch.manual_seed(7)
    X = torch.randn(8, 16, 6, 6)
    DY = torch.randn(8, 16, 6, 6)
    xf = X.clone().requires_grad_(True)
    g = torch.ones(16, requires_grad=True)
    b = torch.zeros(16, requires_grad=True)
    rm, rv = torch.zeros(16), torch.ones(16)
    yref = F.batch_norm(xf, rm, rv, g, b, True, 0.1, 1e-5)
    yref.backward(DY)

    y = torch.cat([res[0]["y"], res[1]["y"]])
    dx = torch.cat([res[0]["dx"], res[1]["dx"]])
    torch.testing.assert_close(y, yref.detach(), rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(dx, xf.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(res[0]["dgamma"] + res[1]["dgamma"], g.grad,
                               rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(res[0]["dbeta"] + res[1]["dbeta"], b.grad,
                               rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(res[0]["rm"], rm, rtol=1e-4, atol=1e-6)
    torch.testing.assert_close(res[0]["rv"], rv, rtol=1e-4, atol=1e-6)
    torch.testing.assert_close(res[0]["rm"], res[1]["rm"], rtol=0, atol=0)


def test_syncbn_falls_back_without_dist():
    from dtmx.ops.layers import SyncBatchNorm2dNHWC
    bn = SyncBatchNorm2dNHWC(8)
    bn.train()
    x = torch.randn(4, 8, 5, 5)
    y = bn(x)
    ref = F.batch_norm(x, torch.zeros(8), torch.ones(8), torch.ones(8),
                       torch.zeros(8), True, 0.1, 1e-5)
    torch.testing.assert_close(y, ref, rtol=1e-4, atol=1e-5)
