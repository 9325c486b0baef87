"""HIP kernel numerics vs plain PyTorch fp32 references (reference test
strategy: check_consistency / check_numeric_gradient, test_utils.py:790,1207).

Every input is bf16-rounded first, the reference computes in fp32 on the same
rounded values, so the comparison isolates kernel accumulation/logic errors
from input quantization. All tests @gpu (MI355X only).
"""
import numpy as np
import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def bf(x):
    return x.to(torch.bfloat16)


def mk(shape, scale=1.0, seed=0):
    g = torch.Generator().manual_seed(seed)
    return (torch.randn(*shape, generator=g) * scale).to(torch.bfloat16)


def assert_close(got, want, rtol=0.02, atol=None, name=""):
    got = got.float().cpu()
    want = want.float().cpu()
    if atol is None:
        atol = 0.02 * want.abs().mean().item() + 1e-3
    torch.testing.assert_close(got, want, rtol=rtol, atol=atol, msg=lambda m: f"{name}: {m}")


def nhwc(t):
    return t.to(DEV).contiguous(memory_format=torch.channels_last)


# ------------------------------------------------------------------- linear

def test_linear_fwd():
    from dtmx.ops.hip import require_ext
    ext = require_ext()
    x, w, b = mk((96, 256)), mk((48, 256)), torch.randn(48).bfloat16()
    y = ext.linear_fwd(x.to(DEV), w.to(DEV), b.to(DEV))
    ref = F.linear(x.float(), w.float(), b.float())
    assert_close(y, ref, name="linear_fwd")


def test_linear_odd_shapes():
    from dtmx.ops.hip import require_ext
    ext = require_ext()
    x, w = mk((37, 136)), mk((1000, 136))  # M,N not tile multiples
    y = ext.linear_fwd(x.to(DEV), w.to(DEV), None)
    ref = F.linear(x.float(), w.float())
    assert_close(y, ref, name="linear_odd")


def test_linear_grads():
    from dtmx.ops.hip import require_ext
    ext = require_ext()
    x, w, dy = mk((64, 512)), mk((128, 512)), mk((64, 128))
    dx = ext.linear_dgrad(dy.to(DEV), w.to(DEV))
    dw = ext.linear_wgrad(dy.to(DEV), x.to(DEV))
    assert_close(dx, dy.float() @ w.float(), name="linear_dgrad")
    assert_close(dw, dy.float().T @ x.float(), name="linear_wgrad")


# --------------------------------------------------------------------- conv

CONV_CASES = [
    # (N, C, H, W, K, R, stride, pad)
    (4, 64, 28, 28, 64, 3, 1, 1),
    (4, 64, 28, 28, 128, 3, 2, 1),
    (4, 256, 14, 14, 64, 1, 1, 0),
    (4, 64, 14, 14, 256, 1, 1, 0),
    (2, 128, 28, 28, 128, 3, 2, 1),
    (2, 3, 64, 64, 64, 7, 2, 3),    # stem (small-C im2col path)
    (2, 16, 9, 9, 24, 3, 1, 1),     # odd spatial
]


@pytest.mark.parametrize("case", CONV_CASES)
def test_conv_fwd(case):
    from dtmx.ops.hip import require_ext
    ext = require_ext()
    N, C, H, W, K, R, stride, pad = case
    x = mk((N, C, H, W), seed=1)
    w = mk((K, C, R, R), scale=0.1, seed=2)
    y = ext.conv_fwd(nhwc(x), nhwc(w), stride, pad)
    ref = F.conv2d(x.float(), w.float(), None, stride, pad)
    assert_close(y.contiguous(), ref, name=f"conv_fwd{case}")


@pytest.mark.parametrize("case", [c for c in CONV_CASES if c[1] % 8 == 0])
def test_conv_dgrad(case):
    from dtmx.ops.hip import require_ext
    ext = require_ext()
    N, C, H, W, K, R, stride, pad = case
    P = (H + 2 * pad - R) // stride + 1
    dy = mk((N, K, P, P), seed=3)
    w = mk((K, C, R, R), scale=0.1, seed=4)
    dx = ext.conv_dgrad(nhwc(dy), nhwc(w), stride, pad, H, W)
    ref = torch.nn.grad.conv2d_input((N, C, H, W), w.float(), dy.float(),
                                     stride=stride, padding=pad)
    assert_close(dx.contiguous(), ref, name=f"conv_dgrad{case}")


@pytest.mark.parametrize("case", CONV_CASES)
def test_conv_wgrad(case):
    from dtmx.ops.hip import require_ext
    ext = require_ext()
    N, C, H, W, K, R, stride, pad = case
    P = (H + 2 * pad - R) // stride + 1
    x = mk((N, C, H, W), seed=5)
    dy = mk((N, K, P, P), scale=0.1, seed=6)
    dw = ext.conv_wgrad(nhwc(x), nhwc(dy), R, R, stride, pad)
    ref = torch.nn.grad.conv2d_weight(x.float(), (K, C, R, R), dy.float(),
                                      stride=stride, padding=pad)
    assert_close(dw.contiguous(), ref, rtol=0.03, name=f"conv_wgrad{case}")


# ----------------------------------------------------------------------- bn

def test_bn_fwd_train_and_bwd():
    from dtmx.ops.hip import require_ext
    ext = require_ext()
    N, C, H, W = 4, 64, 14, 14
    x = mk((N, C, H, W), seed=7)
    gamma = (torch.randn(C) * 0.1 + 1).bfloat16()
    beta = (torch.randn(C) * 0.1).bfloat16()
    rm = torch.zeros(C, dtype=torch.float32, device=DEV)
    rv = torch.ones(C, dtype=torch.float32, device=DEV)
    y, sm, si = ext.bn_fwd_train(nhwc(x), gamma.to(DEV), beta.to(DEV), rm, rv,
                                 0.9, 1e-5, False, None, None, None)
    xf = x.float().requires_grad_(True)
    ref = F.batch_norm(xf, None, None, gamma.float(), beta.float(), True, 0.1, 1e-5)
    assert_close(y.contiguous(), ref, name="bn_fwd")
    # running stats
    assert_close(rm, x.float().mean(dim=(0, 2, 3)) * 0.1, rtol=0.05, name="bn_rm")
    # backward
    dy = mk((N, C, H, W), seed=8)
    ref.backward(dy.float())
    dx, dgamma, dbeta = ext.bn_bwd(nhwc(x), nhwc(dy), gamma.to(DEV), sm, si,
                                   False, y, False)
    assert_close(dx.contiguous(), xf.grad, rtol=0.05, name="bn_dx")
    xhat = (x.float() - x.float().mean(dim=(0, 2, 3), keepdim=True)) / (
        x.float().var(dim=(0, 2, 3), unbiased=False, keepdim=True) + 1e-5).sqrt()
    assert_close(dgamma, (dy.float() * xhat).sum(dim=(0, 2, 3)), rtol=0.05, name="bn_dgamma")
    assert_close(dbeta, dy.float().sum(dim=(0, 2, 3)), rtol=0.05, name="bn_dbeta")


def test_bn_fused_relu():
    from dtmx.ops.hip import require_ext
    ext = require_ext()
    N, C, H, W = 2, 32, 8, 8
    x = mk((N, C, H, W), seed=9)
    gamma = torch.ones(C).bfloat16()
    beta = torch.zeros(C).bfloat16()
    rm = torch.zeros(C, dtype=torch.float32, device=DEV)
    rv = torch.ones(C, dtype=torch.float32, device=DEV)
    y, sm, si = ext.bn_fwd_train(nhwc(x), gamma.to(DEV), beta.to(DEV), rm, rv,
                                 0.9, 1e-5, True, None, None, None)
    ref = F.relu(F.batch_norm(x.float(), None, None, gamma.float(), beta.float(),
                              True, 0.1, 1e-5))
    assert_close(y.contiguous(), ref, name="bn_relu")


def test_bn_fused_residual_full_autograd():
    """y = relu(bn(x) + res): forward + all grads vs torch fp32 reference."""
    from dtmx.ops import functional as DF

    N, C, H, W = 4, 64, 14, 14
    x = mk((N, C, H, W), seed=21)
    res = mk((N, C, H, W), seed=22)
    gamma = (torch.randn(C) * 0.1 + 1).bfloat16()
    beta = (torch.randn(C) * 0.1).bfloat16()
    dy = mk((N, C, H, W), seed=23)

    xd = nhwc(x).requires_grad_(True)
    resd = nhwc(res).requires_grad_(True)
    gd = gamma.to(DEV).requires_grad_(True)
    bd = beta.to(DEV).requires_grad_(True)
    rm = torch.zeros(C, dtype=torch.float32, device=DEV)
    rv = torch.ones(C, dtype=torch.float32, device=DEV)
    y = DF.batch_norm(xd, gd, bd, rm, rv, True, 0.9, 1e-5, True, resd)
    y.backward(nhwc(dy))

    xf = x.float().requires_grad_(True)
    rf = res.float().requires_grad_(True)
    gf = gamma.float().requires_grad_(True)
    bf_ = beta.float().requires_grad_(True)
    ref = torch.nn.functional.relu(
        torch.nn.functional.batch_norm(xf, None, None, gf, bf_, True, 0.1, 1e-5) + rf
    )
    ref.backward(dy.float())
    assert_close(y.contiguous(), ref, name="bnres_y")
    assert_close(xd.grad.contiguous(), xf.grad, rtol=0.05, name="bnres_dx")
    assert_close(resd.grad.contiguous(), rf.grad, rtol=0.05, name="bnres_dres")
    assert_close(gd.grad, gf.grad, rtol=0.05, name="bnres_dgamma")
    assert_close(bd.grad, bf_.grad, rtol=0.05, name="bnres_dbeta")


def test_bn_infer():
    from dtmx.ops.hip import require_ext
    ext = require_ext()
    N, C, H, W = 2, 16, 8, 8
    x = mk((N, C, H, W), seed=10)
    gamma = torch.ones(C).bfloat16()
    beta = torch.zeros(C).bfloat16()
    rm = torch.randn(C, dtype=torch.float32)
    rv = torch.rand(C, dtype=torch.float32) + 0.5
    y = ext.bn_fwd_infer(nhwc(x), gamma.to(DEV), beta.to(DEV), rm.to(DEV),
                         rv.to(DEV), 1e-5, False, None)
    ref = F.batch_norm(x.float(), rm, rv, gamma.float(), beta.float(), False, 0.1, 1e-5)
    assert_close(y.contiguous(), ref, name="bn_infer")


# ------------------------------------------------------------------ pooling

def test_maxpool():
    from dtmx.ops.hip import require_ext
    ext = require_ext()
    N, C, H, W = 2, 64, 32, 32
    x = mk((N, C, H, W), seed=11)
    y, idx = ext.maxpool_fwd(nhwc(x), 3, 2, 1)
    xf = x.float().requires_grad_(True)
    ref = F.max_pool2d(xf, 3, 2, 1)
    assert_close(y.contiguous(), ref, name="maxpool_fwd")
    P = ref.shape[2]
    dy = mk((N, C, P, P), seed=12)
    ref.backward(dy.float())
    dx = ext.maxpool_bwd(nhwc(dy), idx, H, W, 3, 2, 1)
    assert_close(dx.contiguous(), xf.grad, name="maxpool_bwd")


def test_global_avgpool():
    from dtmx.ops.hip import require_ext
    ext = require_ext()
    x = mk((4, 128, 7, 7), seed=13)
    y = ext.global_avgpool_fwd(nhwc(x))
    assert_close(y, x.float().mean(dim=(2, 3)), name="gap_fwd")
    dy = mk((4, 128), seed=14)
    dx = ext.global_avgpool_bwd(dy.to(DEV), 7, 7)
    ref = (dy.float() / 49).reshape(4, 128, 1, 1).expand(4, 128, 7, 7)
    assert_close(dx.contiguous(), ref, name="gap_bwd")


# ------------------------------------------------------------- elementwise

def test_relu_and_add_relu():
    from dtmx.ops.hip import require_ext
    ext = require_ext()
    a, b = mk((1000, 64), seed=15), mk((1000, 64), seed=16)
    y = ext.relu_fwd(a.to(DEV))
    assert_close(y, F.relu(a.float()), name="relu")
    dy = mk((1000, 64), seed=17)
    dx = ext.relu_bwd(dy.to(DEV), y)
    assert_close(dx, dy.float() * (a.float() > 0), name="relu_bwd")
    z = ext.add_relu_fwd(a.to(DEV), b.to(DEV))
    assert_close(z, F.relu(a.float() + b.float()), name="add_relu")


def test_lrn_fwd_bwd():
    """HIP window-5 LRN vs torch fp32 reference (reference
    src/operator/nn/lrn.cc; alexnet.py:34)."""
    import torch

    from dtmx.ops import functional as DF
    torch.manual_seed(21)
    for C in (8, 64, 192):
        x32 = torch.randn(4, C, 9, 9)
        x = (x32.to(torch.bfloat16).to(DEV)
             .contiguous(memory_format=torch.channels_last).requires_grad_(True))
        y = DF.local_response_norm(x, 5, alpha=1e-4, beta=0.75, knorm=2.0)
        xr = x32.to(torch.bfloat16).float().requires_grad_(True)
        ref = F.local_response_norm(xr, 5, alpha=1e-4, beta=0.75, k=2.0)
        assert_close(y, ref, name=f"lrn_fwd_C{C}")
        dy = torch.randn(4, C, 9, 9)
        y.backward(dy.to(torch.bfloat16).to(DEV)
                     .contiguous(memory_format=torch.channels_last))
        ref.backward(dy.to(torch.bfloat16).float())
        assert_close(x.grad, xr.grad, name=f"lrn_bwd_C{C}")


# --------------------------------------------------------------- softmax-ce

def test_softmax_ce():
    from dtmx.ops.hip import require_ext
    ext = require_ext()
    B, V = 64, 1000
    logits = mk((B, V), scale=2.0, seed=18)
    label = torch.randint(0, V, (B,), dtype=torch.int32)
    loss, probs = ext.softmax_ce_fwd(logits.to(DEV), label.to(DEV))
    ref_loss = F.cross_entropy(logits.float(), label.long(), reduction="sum")
    assert abs(loss.item() - ref_loss.item()) / ref_loss.item() < 0.01
    assert_close(probs, F.softmax(logits.float(), dim=1), atol=1e-3, name="probs")
    dl = ext.softmax_ce_bwd(probs, label.to(DEV), torch.ones((), device=DEV))
    onehot = F.one_hot(label.long(), V).float()
    assert_close(dl, F.softmax(logits.float(), 1) - onehot, atol=1e-3, name="dlogits")


# ---------------------------------------------------------------- optimizer

def test_sgd_mom_mp_kernel():
    from dtmx.ops.hip import require_ext
    ext = require_ext()
    n = 4096
    master = torch.randn(n, dtype=torch.float32)
    w = master.bfloat16()
    g = torch.randn(n).bfloat16()
    mom = torch.randn(n, dtype=torch.float32)
    wd, lr, mu, rs = 1e-4, 0.1, 0.9, 1 / 128
    # reference math
    g32 = g.float() * rs + wd * master
    mom_ref = mu * mom - lr * g32
    master_ref = master + mom_ref
    wD, mD, momD = w.to(DEV), master.to(DEV), mom.to(DEV)
    ext.sgd_mom_mp(wD, g.to(DEV), mD, momD, lr, mu, wd, rs, 0.0)
    assert_close(mD, master_ref, rtol=1e-5, atol=1e-6, name="sgd master")
    assert_close(momD, mom_ref, rtol=1e-5, atol=1e-6, name="sgd mom")
    assert_close(wD, master_ref.bfloat16().float(), rtol=1e-2, name="sgd w")


# -------------------------------------------------------------- end-to-end

def test_resnet18_training_step_runs():
    import dtmx
    from dtmx.io import SyntheticDataIter
    from dtmx.models import get_symbol

    torch.manual_seed(0)
    net = get_symbol("resnet", num_layers=18, num_classes=100, image_shape="3,64,64")
    mod = dtmx.Module(net, context=dtmx.gpu(0))
    mod.bind(data_shapes=[("data", (16, 3, 64, 64))],
             label_shapes=[("softmax_label", (16,))], dtype=torch.bfloat16)
    mod.init_params()
    mod.init_optimizer(optimizer_params=(("learning_rate", 0.05), ("momentum", 0.9)))
    it = SyntheticDataIter(100, (16, 3, 64, 64), max_iter=100,
                           dtype=torch.bfloat16, device=torch.device(DEV),
                           layout="NHWC")
    losses = []
    for _ in range(10):
        batch = it.next()
        mod.forward_backward(batch)
        mod.update()
        losses.append(mod._loss.item())
    assert all(np.isfinite(losses)), losses
    # same fixed batch replayed: loss must drop
    assert losses[-1] < losses[0], losses


def test_conv_fwd_fused_bn_stats():
    """conv_fwd_stats slab sums must equal per-channel sums of y."""
    from dtmx.ops.hip import require_ext
    ext = require_ext()
    N, C, H, K, R = 32, 64, 28, 128, 3
    x = mk((N, C, H, H), seed=30)
    w = mk((K, C, R, R), scale=0.1, seed=31)
    y, ps, pss = ext.conv_fwd_stats(nhwc(x), nhwc(w), 1, 1)
    assert ps.numel() > 0
    yf = y.contiguous().float()
    ref_sum = yf.sum(dim=(0, 2, 3)).cpu()
    ref_sumsq = (yf * yf).sum(dim=(0, 2, 3)).cpu()
    assert_close(ps.sum(0), ref_sum, rtol=0.01, name="fused psum")
    assert_close(pss.sum(0), ref_sumsq, rtol=0.01, name="fused psumsq")


def test_bn_with_fused_stats_matches_plain():
    import dtmx
    from dtmx.ops import functional as DF
    from dtmx.ops.hip import require_ext
    ext = require_ext()
    N, C, H, K, R = 32, 64, 28, 64, 3
    x = mk((N, C, H, H), seed=32)
    w = mk((K, C, R, R), scale=0.1, seed=33)
    y, ps, pss = ext.conv_fwd_stats(nhwc(x), nhwc(w), 1, 1)
    gamma = torch.ones(C).bfloat16().to(DEV)
    beta = torch.zeros(C).bfloat16().to(DEV)
    rm1 = torch.zeros(C, dtype=torch.float32, device=DEV)
    rv1 = torch.ones(C, dtype=torch.float32, device=DEV)
    out_fused, sm1, si1 = ext.bn_fwd_train(y, gamma, beta, rm1, rv1, 0.9, 1e-5,
                                           False, None, ps, pss)
    rm2 = torch.zeros(C, dtype=torch.float32, device=DEV)
    rv2 = torch.ones(C, dtype=torch.float32, device=DEV)
    out_plain, sm2, si2 = ext.bn_fwd_train(y, gamma, beta, rm2, rv2, 0.9, 1e-5,
                                           False, None, None, None)
    assert_close(out_fused, out_plain, rtol=0.01, name="bn fused-vs-plain")
    assert_close(sm1, sm2, rtol=0.01, name="save_mean")


# ---------------------------------------------------- dropout / layer norm

def test_dropout_fwd_bwd():
    from dtmx.ops import functional as DF
    torch.manual_seed(3)
    x = mk((64, 4096), seed=11).to(DEV).requires_grad_(True)
    p = 0.5
    y = DF.dropout(x, p, training=True, seed=1234)
    keep = (y != 0) | (x == 0)
    frac = keep.float().mean().item()
    assert abs(frac - (1 - p)) < 0.02  # RNG keep-rate
    # kept elements are scaled by 1/(1-p)
    m = y != 0
    assert_close(y[m], x.detach()[m] * 2.0, name="dropout scale")
    # backward applies the SAME mask and scale
    dy = mk((64, 4096), seed=12).to(DEV)
    y.backward(dy)
    assert_close(x.grad[m], dy[m].float() * 2.0, name="dropout bwd kept")
    assert x.grad[~m].abs().max().item() == 0.0
    # determinism: same seed -> same mask
    y2 = DF.dropout(x.detach(), p, training=True, seed=1234)
    assert torch.equal((y2 != 0), (y != 0))
    # eval mode: identity
    assert DF.dropout(x.detach(), p, training=False) is x.detach() or torch.equal(
        DF.dropout(x.detach(), p, training=False), x.detach())


def test_layer_norm_fwd_bwd():
    from dtmx.ops import functional as DF
    B, D = 96, 1000
    x = mk((B, D), seed=21).to(DEV).requires_grad_(True)
    gamma = (torch.randn(D) * 0.2 + 1.0).to(DEV).requires_grad_(True)
    beta = (torch.randn(D) * 0.1).to(DEV).requires_grad_(True)
    y = DF.layer_norm(x, gamma, beta)
    xf = x.detach().float().requires_grad_(True)
    gf = gamma.detach().float().requires_grad_(True)
    bf_ = beta.detach().float().requires_grad_(True)
    yref = F.layer_norm(xf, (D,), gf, bf_, 1e-5)
    assert_close(y, yref, name="ln fwd")
    dy = mk((B, D), seed=22).to(DEV)
    y.backward(dy)
    yref.backward(dy.float())
    assert_close(x.grad, xf.grad, name="ln dx")
    assert_close(gamma.grad, gf.grad, rtol=0.03, name="ln dgamma")
    assert_close(beta.grad, bf_.grad, rtol=0.03, name="ln dbeta")


def test_bn_presummed_path_matches_bn():
    """SyncBN kernel seam: local_sums -> fwd_presummed(count=rows) and
    bwd_sums -> bwd_dx_presummed must reproduce the plain BN path bit-for-bit
    on a single rank (the all-reduce in between is the identity at world=1)."""
    from dtmx.ops.hip import require_ext
    ext = require_ext()
    N, C, H, W = 8, 64, 14, 14
    rows = N * H * W
    x = nhwc(mk((N, C, H, W), seed=31))
    dy = nhwc(mk((N, C, H, W), seed=32))
    gamma = (torch.randn(C) * 0.2 + 1.0).to(torch.bfloat16).to(DEV)
    beta = (torch.randn(C) * 0.1).to(torch.bfloat16).to(DEV)

    rm1, rv1 = torch.zeros(C, device=DEV), torch.ones(C, device=DEV)
    rm2, rv2 = torch.zeros(C, device=DEV), torch.ones(C, device=DEV)
    y1, m1, i1 = ext.bn_fwd_train(x, gamma, beta, rm1, rv1, 0.9, 1e-5, False,
                                  None, None, None)
    s, ss = ext.bn_local_sums(x)
    y2, m2, i2 = ext.bn_fwd_presummed(x, gamma, beta, rm2, rv2, 0.9, 1e-5,
                                      False, None, s, ss, rows)
    torch.testing.assert_close(y1, y2, rtol=0, atol=0)
    torch.testing.assert_close(m1, m2)
    torch.testing.assert_close(rm1, rm2)
    torch.testing.assert_close(rv1, rv2)

    dx1, dgamma1, dbeta1 = ext.bn_bwd(x, dy, gamma, m1, i1, False, y1, False)
    tdb, tdg = ext.bn_bwd_sums(x, dy, y2, m2, i2, False)
    torch.testing.assert_close(tdb, dbeta1.float(), rtol=2e-2, atol=1e-2)
    torch.testing.assert_close(tdg, dgamma1.float(), rtol=2e-2, atol=1e-2)
    (dx2,) = ext.bn_bwd_dx_presummed(x, dy, y2, m2, i2, gamma, tdb, tdg, rows,
                                     False, False)
    torch.testing.assert_close(dx1, dx2, rtol=0, atol=0)


def test_conv_wgrad_workspace_rezero():
    """The persistent split-K accumulator must be re-zeroed by its drain pass:
    calling wgrad twice with identical inputs must agree to within split-K
    float-atomic reordering noise (a stale accumulator would double values)."""
    from dtmx.ops.hip import require_ext
    ext = require_ext()
    x = nhwc(mk((8, 64, 28, 28), seed=41))
    dy = nhwc(mk((8, 128, 28, 28), seed=42))
    d1 = ext.conv_wgrad(x, dy, 3, 3, 1, 1).clone()
    d2 = ext.conv_wgrad(x, dy, 3, 3, 1, 1)
    torch.testing.assert_close(d1, d2, rtol=0.02, atol=0.05)


def test_linear_unpadded_out_features_with_bias():
    """N=100 (num_classes) regression: the epilogue's tail chunk used to write
    a full 16B vector — stomping the next row's first columns, the heap past
    the last row, and reading bias out of bounds. Churn the allocator with
    garbage so OOB reads are visible."""
    from dtmx.ops.hip import require_ext
    ext = require_ext()
    for (M, N, K) in [(4, 100, 2048), (8, 100, 2048), (16, 10, 504), (3, 37, 64)]:
        g = torch.Generator().manual_seed(M * 1000 + N)
        x = (torch.randn(M, K, generator=g) * 0.5).to(torch.bfloat16).cuda()
        w = (torch.randn(N, K, generator=g) * 0.05).to(torch.bfloat16).cuda()
        b = (torch.randn(N, generator=g)).float().cuda()
        ref = x.float() @ w.float().T + b
        for it in range(4):
            junk = torch.full((8 << 20,), 3e4, dtype=torch.bfloat16, device="cuda")
            del junk
            y = ext.linear_fwd(x, w, b)
            assert_close(y, ref.cpu(), name=f"linear M{M} N{N} K{K} it{it}")


def test_dropout_layernorm_fp16():
    """fp16 instantiation of the counter-RNG dropout and LayerNorm kernels."""
    from dtmx.ops import functional as DF
    x = (torch.randn(32, 512) * 0.5).to(torch.float16).to(DEV).requires_grad_(True)
    y = DF.dropout(x, 0.5, training=True, seed=7)
    m = y != 0
    assert abs(m.float().mean().item() - 0.5) < 0.05
    assert_close(y[m], x.detach()[m] * 2.0, name="dropout fp16")
    g = torch.ones(512, device=DEV)
    b = torch.zeros(512, device=DEV)
    z = DF.layer_norm(x.detach(), g, b)
    ref = F.layer_norm(x.detach().float(), (512,), g, b, 1e-5)
    assert_close(z, ref, name="ln fp16")


def test_fused_block_matches_layerwise():
    """The manually-backpropagated residual block (DTMX_FUSED_BLOCK) must
    reproduce the layer-by-layer autograd path: same y, same input/param
    grads, same running stats — it runs the same kernels in the same order,
    with only the fork-add fused into the dgrad epilogue."""
    import copy
    import os
    from dtmx.models.resnet import BasicBlock, Bottleneck

    cases = [
        (Bottleneck, dict(in_ch=256, ch=64)),                 # identity
        (Bottleneck, dict(in_ch=256, ch=128, stride=2)),      # downsample s2
        (Bottleneck, dict(in_ch=64, ch=64)),                  # downsample s1
        (BasicBlock, dict(in_ch=64, ch=64)),
        (BasicBlock, dict(in_ch=64, ch=128, stride=2)),
    ]
    for cls, kw in cases:
        torch.manual_seed(0)
        blk = cls(**kw).to(DEV).to(torch.bfloat16)
        blk = blk.to(memory_format=torch.channels_last)  # as Module.bind does
        blk2 = copy.deepcopy(blk)
        x = nhwc(mk((4, kw["in_ch"], 14, 14), seed=5))
        dy_shape = None
        results = []
        for flag, b in (("0", blk), ("1", blk2)):
            os.environ["DTMX_FUSED_BLOCK"] = flag
            b.train()
            xr = x.clone().requires_grad_(True)
            y = b(xr)
            if dy_shape is None:
                dy_shape = y.shape
            dy = nhwc(mk(tuple(dy_shape), seed=6))
            y.backward(dy)
            results.append({
                "y": y.detach(), "dx": xr.grad,
                "pg": {n: p.grad.clone() for n, p in b.named_parameters()},
                "rm": b.bn1.running_mean.clone(),
            })
        os.environ.pop("DTMX_FUSED_BLOCK", None)
        ref, fused = results
        tag = f"{cls.__name__}{kw}"
        # small-grid conv paths use fp32 atomics (order nondeterministic),
        # so compare numerically, not bit-wise
        assert_close(fused["y"], ref["y"], name=f"{tag} y")
        torch.testing.assert_close(fused["rm"], ref["rm"], rtol=1e-3, atol=1e-4)
        assert_close(fused["dx"], ref["dx"], rtol=0.02, name=f"{tag} dx")
        for n, g in ref["pg"].items():
            # wgrad split-K float atomics reorder between runs: compare with
            # headroom above that noise floor
            atol = 0.05 * g.float().abs().mean().item() + 0.08
            assert_close(fused["pg"][n], g, rtol=0.08, atol=atol,
                         name=f"{tag} grad {n}")


# ------------------------------------------------------ embedding/take (HIP)

@pytest.mark.gpu
def test_take_fwd_gpu_matches_torch():
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    from dtmx.ops.hip import require_ext

    ext = require_ext()
    torch.manual_seed(0)
    table = torch.randn(100, 32, device="cuda").to(torch.bfloat16)
    idx = torch.randint(-3, 105, (57,), device="cuda")  # incl. OOB (clipped)
    out = ext.take_fwd(table, idx)
    ref = table[idx.clamp(0, 99)]
    torch.testing.assert_close(out.float(), ref.float(), rtol=0, atol=0)


@pytest.mark.gpu
def test_take_bwd_gpu_matches_torch():
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    from dtmx.ops.hip import require_ext

    ext = require_ext()
    torch.manual_seed(1)
    dy = torch.randn(64, 16, device="cuda").to(torch.bfloat16)
    idx = torch.randint(0, 10, (64,), device="cuda")
    dtab = ext.take_bwd(dy, idx, 10)
    ref = torch.zeros(10, 16, device="cuda")
    ref.index_add_(0, idx, dy.float())
    torch.testing.assert_close(dtab, ref, rtol=1e-3, atol=1e-3)


@pytest.mark.gpu
def test_embedding_autograd_gpu():
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    from dtmx.ops.layers import Embedding

    torch.manual_seed(2)
    emb = Embedding(40, 24).to("cuda").to(torch.bfloat16)
    idx = torch.randint(0, 40, (6, 5), device="cuda")
    out = emb(idx)
    out.float().sum().backward()
    ref = torch.zeros(40, device="cuda")
    ref.index_add_(0, idx.reshape(-1),
                   torch.ones(30, device="cuda") * 24)
    torch.testing.assert_close(emb.weight.grad.float().sum(dim=1), ref,
                               rtol=0.01, atol=0.1)


# ------------------------------------------ exact-integer GEMM order checks

def _cl(t):
    return t.contiguous(memory_format=torch.channels_last)


@pytest.mark.gpu
def test_wgrad_exact_on_integer_inputs():
    """Small-integer bf16 inputs make every product and fp32 partial sum
    EXACT, so the split-K atomic accumulation must reproduce the fp64
    reference bit-for-bit — a ~2% relative tolerance (round-1 weak #8)
    could hide a K-ordering bug; zero tolerance here cannot."""
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    from dtmx.ops.hip import require_ext

    ext = require_ext()
    torch.manual_seed(0)
    N, C, Ko, H, W, R, stride, pad = 4, 16, 32, 14, 14, 3, 1, 1
    x = _cl(torch.randint(-4, 5, (N, C, H, W), device="cuda")
            .to(torch.bfloat16))
    dy = _cl(torch.randint(-4, 5, (N, Ko, H, W), device="cuda")
             .to(torch.bfloat16))
    dw = ext.conv_wgrad(x, dy, R, R, stride, pad)
    ref = torch.nn.grad.conv2d_weight(
        x.double(), (Ko, C, R, R), dy.double(), stride=stride, padding=pad)
    # the fp32 accumulation is exact on these inputs; only the final bf16
    # store rounds — compare against the bf16-rounded exact reference
    torch.testing.assert_close(dw.double(), ref.to(torch.bfloat16).double(),
                               rtol=0, atol=0)


@pytest.mark.gpu
def test_linear_gemms_exact_on_integer_inputs():
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    from dtmx.ops.hip import require_ext

    ext = require_ext()
    torch.manual_seed(1)
    x = torch.randint(-4, 5, (96, 64), device="cuda").to(torch.bfloat16)
    w = torch.randint(-4, 5, (40, 64), device="cuda").to(torch.bfloat16)
    dy = torch.randint(-4, 5, (96, 40), device="cuda").to(torch.bfloat16)
    def bf(t):
        return t.to(torch.bfloat16).double()

    y = ext.linear_fwd(x, w, None)
    torch.testing.assert_close(y.double(), bf(x.double() @ w.double().T),
                               rtol=0, atol=0)
    dx = ext.linear_dgrad(dy, w)
    torch.testing.assert_close(dx.double(), bf(dy.double() @ w.double()),
                               rtol=0, atol=0)
    dw = ext.linear_wgrad(dy, x)
    torch.testing.assert_close(dw.double(), bf(dy.double().T @ x.double()),
                               rtol=0, atol=0)


@pytest.mark.gpu
def test_deconv2d_matches_torch():
    """Deconvolution built from the conv kernels (fwd = dgrad, etc.)."""
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    from dtmx.ops import functional as DF

    torch.manual_seed(0)
    x = (torch.randn(4, 16, 9, 9, device="cuda").to(torch.bfloat16)
         .contiguous(memory_format=torch.channels_last).requires_grad_())
    w = (torch.randn(16, 32, 4, 4, device="cuda").to(torch.bfloat16)
         .contiguous(memory_format=torch.channels_last).requires_grad_() )
    y = DF.deconv2d(x, w, 2, 1)
    assert y.shape == (4, 32, 18, 18)
    ref = torch.nn.functional.conv_transpose2d(x.detach().float(),
                                               w.detach().float(), None, 2, 1)
    torch.testing.assert_close(y.float(), ref, rtol=0.02,
                               atol=0.02 * ref.abs().mean().item())
    g = torch.randn_like(y)
    y.backward(g)
    xr = x.detach().float().requires_grad_()
    wr = w.detach().float().requires_grad_()
    torch.nn.functional.conv_transpose2d(xr, wr, None, 2, 1).backward(g.float())
    torch.testing.assert_close(x.grad.float(), xr.grad, rtol=0.03,
                               atol=0.05 * xr.grad.abs().mean().item())
    torch.testing.assert_close(w.grad.float(), wr.grad, rtol=0.03,
                               atol=0.05 * wr.grad.abs().mean().item())


# ----------------------------------------- 256^2 8-phase dense GEMM routing

@pytest.mark.gpu
def test_gemm256_routed_linear_matches_torch():
    """Shapes that route through gemm256f_kernel (big dense tiles) must
    match torch; the disable env must give the same numerics class."""
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    from dtmx.ops.hip import require_ext

    ext = require_ext()
    torch.manual_seed(0)
    M, N, K = 8192, 2048, 256
    a = torch.randn(M, K, device="cuda").to(torch.bfloat16)
    b = torch.randn(N, K, device="cuda").to(torch.bfloat16)
    y = ext.linear_fwd(a, b, None).float()
    ref = a.float() @ b.float().T
    assert (y - ref).abs().max().item() / ref.abs().max().item() < 0.01


@pytest.mark.gpu
def test_gemm256_routed_conv_and_bnfuse():
    """1x1 conv fwd (+fused stats) and dgrad(+BN-bwd fusion) on shapes big
    enough to route through the 256^2 kernel equal the 128^2 path."""
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    import os

    from dtmx.ops.hip import require_ext

    ext = require_ext()
    torch.manual_seed(1)
    N, C, Ko, H = 32, 256, 256, 56
    x = _cl(torch.randn(N, C, H, H, device="cuda").to(torch.bfloat16) * 0.5)
    w = _cl(torch.randn(Ko, C, 1, 1, device="cuda").to(torch.bfloat16) * 0.05)
    os.environ["DTMX_DISABLE_GEMM256"] = "1"
    # env read is cached per-process after first use; compare vs torch instead
    del os.environ["DTMX_DISABLE_GEMM256"]
    y, ps, pss = ext.conv_fwd_stats(x, w, 1, 0)
    ref = torch.nn.functional.conv2d(x.float(), w.float())
    assert (y.float() - ref).abs().max().item() / ref.abs().max().item() < 0.02
    yf = y.float()
    torch.testing.assert_close(ps.sum(0), yf.sum(dim=(0, 2, 3)), rtol=1e-3,
                               atol=2.0)
    torch.testing.assert_close(pss.sum(0), (yf * yf).sum(dim=(0, 2, 3)),
                               rtol=1e-3, atol=2.0)

    # dgrad + BN-backward fusion on a routed shape
    dy = _cl(torch.randn(N, Ko, H, H, device="cuda").to(torch.bfloat16))
    c = _cl(torch.randn(N, C, H, H, device="cuda").to(torch.bfloat16))
    mean = c.float().mean(dim=(0, 2, 3)).contiguous()
    invstd = (c.float().var(dim=(0, 2, 3), unbiased=False) + 1e-5).rsqrt().contiguous()
    ybn = _cl(torch.relu(((c.float() - mean.view(1, -1, 1, 1))
                          * invstd.view(1, -1, 1, 1))).to(torch.bfloat16))
    g, pdb, pdg = ext.conv_dgrad_bnfuse(dy, w, 1, 0, H, H, None, ybn, c,
                                        mean, invstd)
    dy_ref = torch.nn.grad.conv2d_input((N, C, H, H), w.float(), dy.float())
    mask = (ybn.float() > 0).float()
    gref = dy_ref * mask
    assert (g.float() - gref).abs().max().item() / (gref.abs().max().item() + 1e-6) < 0.02
    torch.testing.assert_close(pdb.sum(0), gref.sum(dim=(0, 2, 3)), rtol=2e-3,
                               atol=3.0)


@pytest.mark.gpu
def test_gemm256_routed_gather_conv_matches_torch():
    """3x3 conv fwd/dgrad on shapes big enough to route the GATHER providers
    through the 256^2 kernel (batch 512 l3-shape) must match torch."""
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    from dtmx.ops.hip import require_ext

    ext = require_ext()
    torch.manual_seed(3)
    N, C, Ko, H = 512, 256, 256, 14
    x = _cl(torch.randn(N, C, H, H, device="cuda").to(torch.bfloat16) * 0.5)
    w = _cl(torch.randn(Ko, C, 3, 3, device="cuda").to(torch.bfloat16) * 0.05)
    y = ext.conv_fwd(x, w, 1, 1)
    ref = torch.nn.functional.conv2d(x.float(), w.float(), padding=1)
    assert (y.float() - ref).abs().max().item() / ref.abs().max().item() < 0.02
    dy = _cl(torch.randn(N, Ko, H, H, device="cuda").to(torch.bfloat16))
    dx = ext.conv_dgrad(dy, w, 1, 1, H, H, None)
    dref = torch.nn.grad.conv2d_input((N, C, H, H), w.float(), dy.float(),
                                      padding=1)
    assert (dx.float() - dref).abs().max().item() / dref.abs().max().item() < 0.02
