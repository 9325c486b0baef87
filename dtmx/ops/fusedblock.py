"""Manually-backpropagated ResNet blocks (residual-join gradient fusion).

Eager autograd sums the two gradients meeting at a residual fork (the block
input feeds both conv1 and the shortcut) with a separate elementwise add —
~5% of the ResNet-50 step at bs1024, pure HBM traffic. Here the whole block
is one autograd.Function whose backward accumulates the shortcut gradient
inside conv1's dgrad epilogue (`conv_dgrad(..., acc=...)`, in place), so the
fork add never materializes. Numerics are identical to the layer-by-layer
path: the same HIP kernels run in the same order with the same operands.

Enabled when DTMX_FUSED_BLOCK=1 (default on for CUDA training; see
models/resnet.py). Reference behavior: the per-op backward of
symbols/resnet.py residual units.
"""
from __future__ import annotations

import os

import torch

from .hip import require_ext


def _cl(t):
    return t.contiguous(memory_format=torch.channels_last)


def _bnbwd_on() -> bool:
    """BN-backward fusion into the dgrad epilogue (conv_dgrad_bnfuse):
    the producing GEMM relu-masks dy and emits per-block (Σg, Σg·xhat)
    slabs, so bn_bwd_stats never re-reads (x, dy, y) from HBM and the dx
    pass drops its y read. Default on; DTMX_FUSE_BN_BWD=0 restores the
    standalone bn_bwd path (numerics-identical, used by the equivalence
    test)."""
    return os.environ.get("DTMX_FUSE_BN_BWD", "1") == "1"


def _bnbwd_cross_on() -> bool:
    """Separate gate for the cross-block seam (debug/bisect aid)."""
    return _bnbwd_on() and os.environ.get("DTMX_FUSE_BN_CROSS", "1") == "1"


def _bnstats_on() -> bool:
    """Forward BN-stats fusion: the producing conv's epilogue emits the
    (sum, sumsq) slabs, so bn_fwd_train skips its whole-tensor stats read.
    No-atomic scheme (round 1's atomicAdd version measured -3.5% and was
    reverted; see EpiBF16FwdStats, gemm_conv.hip)."""
    return os.environ.get("DTMX_FUSE_BN_STATS", "1") == "1"


def _conv_fs(ext, x, w, stride, pad, want):
    """conv forward, optionally with fused BN-stat slabs."""
    if want:
        c, ps, pss = ext.conv_fwd_stats(x, w, stride, pad)
        if ps.numel():
            return c, ps, pss
        return c, None, None
    return ext.conv_fwd(x, w, stride, pad), None, None


# instrumentation: how many BN backwards took each path (read by tests and
# the profiling harness; reset freely)
bnbwd_stats = {"interior": 0, "cross": 0, "cross_emit": 0, "standalone": 0}


def _bn_bwd_from_masked(ext, c, g, mean, invstd, gamma, pdb, pdg):
    """Finish a BN backward whose relu mask + partial sums were produced by
    the upstream GEMM epilogue. Returns (dc, dgamma, dbeta)."""
    dgamma, dbeta, tdb, tdg = ext.bn_bwd_finalize_slabs(pdb, pdg, g)
    rows = c.numel() // c.shape[1]
    dc = ext.bn_bwd_dx_presummed(c, g, g, mean, invstd, gamma, tdb, tdg,
                                 rows, False, False)[0]
    return dc, dgamma, dbeta


class _FusedBottleneck(torch.autograd.Function):
    """conv1(1x1) -> bn1+relu -> conv2(3x3,s) -> bn2+relu -> conv3(1x1)
    -> bn3(+shortcut residual, relu); shortcut = identity or 1x1/s conv+bn."""

    @staticmethod
    def forward(ctx, x, w1, g1, b1, w2, g2, b2, w3, g3, b3, wd, gd, bd,
                stride, bn1, bn2, bn3, bnd, bnprev=None, handle=None):
        ext = require_ext()
        mom, eps = bn1.momentum, bn1.eps
        fs = _bnstats_on()
        c1, ps1, pss1 = _conv_fs(ext, x, w1, 1, 0, fs)
        y1, m1, i1 = ext.bn_fwd_train(c1, g1, b1, bn1.running_mean,
                                      bn1.running_var, mom, eps, True, None,
                                      ps1, pss1)
        c2, ps2, pss2 = _conv_fs(ext, y1, w2, stride, 1, fs)
        y2, m2, i2 = ext.bn_fwd_train(c2, g2, b2, bn2.running_mean,
                                      bn2.running_var, mom, eps, True, None,
                                      ps2, pss2)
        c3, ps3, pss3 = _conv_fs(ext, y2, w3, 1, 0, fs)
        if wd is not None:
            cd, psd, pssd = _conv_fs(ext, x, wd, stride, 0, fs)
            sc, md, idn = ext.bn_fwd_train(cd, gd, bd, bnd.running_mean,
                                           bnd.running_var, mom, eps, False,
                                           None, psd, pssd)
        else:
            cd = sc = x
            md = idn = m1  # placeholders (unused)
        y3, m3, i3 = ext.bn_fwd_train(c3, g3, b3, bn3.running_mean,
                                      bn3.running_var, mom, eps, True, sc,
                                      ps3, pss3)
        ctx.save_for_backward(x, w1, g1, w2, g2, w3, g3, wd, gd,
                              c1, y1, c2, y2, c3, y3, cd, sc,
                              m1, i1, m2, i2, m3, i3, md, idn)
        ctx.stride = stride
        ctx.has_down = wd is not None
        # cross-block BN-backward fusion handle: the NEXT block's backward
        # produces dy3 (= its dx) and can mask+stats bn3 in its dgrad
        # epilogue, given bn3's pre-BN output and saved stats. apply() may
        # rewrap the output tensor, so the handle travels through the
        # `handle` dict and the WRAPPER attaches it to the real output.
        ctx.bn3prev = bnprev
        if handle is not None and _bnbwd_on():
            handle["bnout"] = (c3, m3, i3)
        return y3

    @staticmethod
    def backward(ctx, dy):
        ext = require_ext()
        (x, w1, g1, w2, g2, w3, g3, wd, gd,
         c1, y1, c2, y2, c3, y3, cd, sc,
         m1, i1, m2, i2, m3, i3, md, idn) = ctx.saved_tensors
        stride = ctx.stride
        fuse = _bnbwd_on() and dy.is_cuda
        dy = _cl(dy)
        H, W = x.shape[2], x.shape[3]
        H1, W1 = y1.shape[2], y1.shape[3]
        H2, W2 = y2.shape[2], y2.shape[3]

        # bn3: the upstream block's dgrad epilogue may have pre-masked dy and
        # produced the stat slabs (attribute attached by OUR return below,
        # one block downstream)
        slabs3 = getattr(dy, "_dtmx_bnslabs", None)
        if slabs3 is not None:
            bnbwd_stats["cross"] += 1
            dc3, dg3, db3 = _bn_bwd_from_masked(ext, c3, dy, m3, i3, g3,
                                                slabs3[0], slabs3[1])
            dres = dy  # already relu-masked at the residual join
        else:
            bnbwd_stats["standalone"] += 1
            dc3, dg3, db3, dres = ext.bn_bwd(c3, dy, g3, m3, i3, True, y3, True)
        if fuse:
            dy2, pdb2, pdg2 = ext.conv_dgrad_bnfuse(dc3, w3, 1, 0, H2, W2,
                                                    None, y2, c2, m2, i2)
            dc2, dg2, db2 = _bn_bwd_from_masked(ext, c2, dy2, m2, i2, g2,
                                                pdb2, pdg2)
        else:
            dy2 = ext.conv_dgrad(dc3, w3, 1, 0, H2, W2)
            dc2, dg2, db2 = ext.bn_bwd(c2, dy2, g2, m2, i2, True, y2, False)
        dw3 = ext.conv_wgrad(y2, dc3, 1, 1, 1, 0)
        if fuse:
            dy1, pdb1, pdg1 = ext.conv_dgrad_bnfuse(dc2, w2, stride, 1, H1, W1,
                                                    None, y1, c1, m1, i1)
            dc1, dg1, db1 = _bn_bwd_from_masked(ext, c1, dy1, m1, i1, g1,
                                                pdb1, pdg1)
        else:
            dy1 = ext.conv_dgrad(dc2, w2, stride, 1, H1, W1)
            dc1, dg1, db1 = ext.bn_bwd(c1, dy1, g1, m1, i1, True, y1, False)
        dw2 = ext.conv_wgrad(y1, dc2, 3, 3, stride, 1)
        dw1 = ext.conv_wgrad(x, dc1, 1, 1, 1, 0)
        bnp = ctx.bn3prev  # previous block's (c3, mean, invstd) or None
        if ctx.has_down:
            dcd, dgd, dbd = ext.bn_bwd(cd, dres, gd, md, idn, False, sc, False)
            dwd = ext.conv_wgrad(x, dcd, 1, 1, stride, 0)
            dx = ext.conv_dgrad(dc1, w1, 1, 0, H, W)
            if fuse and bnp is not None and stride == 1:
                # last dx producer fuses the PREVIOUS block's bn3 backward
                dx, pdb, pdg = ext.conv_dgrad_bnfuse(dcd, wd, 1, 0, H, W, dx,
                                                     x, bnp[0], bnp[1], bnp[2])
                bnbwd_stats["cross_emit"] += 1
                dx._dtmx_bnslabs = (pdb, pdg)
            else:
                # shortcut dgrad accumulates into dx in the epilogue
                dx = ext.conv_dgrad(dcd, wd, stride, 0, H, W, acc=dx)
        else:
            if fuse and bnp is not None:
                dx, pdb, pdg = ext.conv_dgrad_bnfuse(dc1, w1, 1, 0, H, W, dres,
                                                     x, bnp[0], bnp[1], bnp[2])
                bnbwd_stats["cross_emit"] += 1
                dx._dtmx_bnslabs = (pdb, pdg)
            else:
                # conv1 dgrad accumulates into the residual grad in place
                dx = ext.conv_dgrad(dc1, w1, 1, 0, H, W, acc=dres)
            dwd = dgd = dbd = None
        return (dx, dw1, dg1, db1, dw2, dg2, db2, dw3, dg3, db3,
                dwd, dgd, dbd, None, None, None, None, None, None, None)


class _FusedBasicBlock(torch.autograd.Function):
    """conv1(3x3,s) -> bn1+relu -> conv2(3x3) -> bn2(+shortcut, relu)."""

    @staticmethod
    def forward(ctx, x, w1, g1, b1, w2, g2, b2, wd, gd, bd, stride,
                bn1, bn2, bnd, bnprev=None, handle=None):
        ext = require_ext()
        mom, eps = bn1.momentum, bn1.eps
        fs = _bnstats_on()
        c1, ps1, pss1 = _conv_fs(ext, x, w1, stride, 1, fs)
        y1, m1, i1 = ext.bn_fwd_train(c1, g1, b1, bn1.running_mean,
                                      bn1.running_var, mom, eps, True, None,
                                      ps1, pss1)
        c2, ps2, pss2 = _conv_fs(ext, y1, w2, 1, 1, fs)
        if wd is not None:
            cd, psd, pssd = _conv_fs(ext, x, wd, stride, 0, fs)
            sc, md, idn = ext.bn_fwd_train(cd, gd, bd, bnd.running_mean,
                                           bnd.running_var, mom, eps, False,
                                           None, psd, pssd)
        else:
            cd = sc = x
            md = idn = m1
        y2, m2, i2 = ext.bn_fwd_train(c2, g2, b2, bn2.running_mean,
                                      bn2.running_var, mom, eps, True, sc,
                                      ps2, pss2)
        ctx.save_for_backward(x, w1, g1, w2, g2, wd, gd, c1, y1, c2, y2, cd,
                              sc, m1, i1, m2, i2, md, idn)
        ctx.stride = stride
        ctx.has_down = wd is not None
        ctx.bnprev = bnprev
        if handle is not None and _bnbwd_on():
            handle["bnout"] = (c2, m2, i2)
        return y2

    @staticmethod
    def backward(ctx, dy):
        ext = require_ext()
        (x, w1, g1, w2, g2, wd, gd, c1, y1, c2, y2, cd, sc,
         m1, i1, m2, i2, md, idn) = ctx.saved_tensors
        stride = ctx.stride
        fuse = _bnbwd_on() and dy.is_cuda
        dy = _cl(dy)
        H, W = x.shape[2], x.shape[3]
        H1, W1 = y1.shape[2], y1.shape[3]

        slabs2 = getattr(dy, "_dtmx_bnslabs", None)
        if slabs2 is not None:
            bnbwd_stats["cross"] += 1
            dc2, dg2, db2 = _bn_bwd_from_masked(ext, c2, dy, m2, i2, g2,
                                                slabs2[0], slabs2[1])
            dres = dy
        else:
            bnbwd_stats["standalone"] += 1
            dc2, dg2, db2, dres = ext.bn_bwd(c2, dy, g2, m2, i2, True, y2, True)
        if fuse:
            dy1, pdb1, pdg1 = ext.conv_dgrad_bnfuse(dc2, w2, 1, 1, H1, W1,
                                                    None, y1, c1, m1, i1)
            dc1, dg1, db1 = _bn_bwd_from_masked(ext, c1, dy1, m1, i1, g1,
                                                pdb1, pdg1)
        else:
            dy1 = ext.conv_dgrad(dc2, w2, 1, 1, H1, W1)
            dc1, dg1, db1 = ext.bn_bwd(c1, dy1, g1, m1, i1, True, y1, False)
        dw2 = ext.conv_wgrad(y1, dc2, 3, 3, 1, 1)
        dw1 = ext.conv_wgrad(x, dc1, 3, 3, stride, 1)
        bnp = ctx.bnprev
        if ctx.has_down:
            dcd, dgd, dbd = ext.bn_bwd(cd, dres, gd, md, idn, False, sc, False)
            dwd = ext.conv_wgrad(x, dcd, 1, 1, stride, 0)
            dx = ext.conv_dgrad(dc1, w1, stride, 1, H, W)
            if fuse and bnp is not None and stride == 1:
                dx, pdb, pdg = ext.conv_dgrad_bnfuse(dcd, wd, 1, 0, H, W, dx,
                                                     x, bnp[0], bnp[1], bnp[2])
                bnbwd_stats["cross_emit"] += 1
                dx._dtmx_bnslabs = (pdb, pdg)
            else:
                dx = ext.conv_dgrad(dcd, wd, stride, 0, H, W, acc=dx)
        else:
            if fuse and bnp is not None:
                dx, pdb, pdg = ext.conv_dgrad_bnfuse(dc1, w1, stride, 1, H, W,
                                                     dres, x, bnp[0], bnp[1],
                                                     bnp[2])
                bnbwd_stats["cross_emit"] += 1
                dx._dtmx_bnslabs = (pdb, pdg)
            else:
                dx = ext.conv_dgrad(dc1, w1, stride, 1, H, W, acc=dres)
            dwd = dgd = dbd = None
        return (dx, dw1, dg1, db1, dw2, dg2, db2, dwd, dgd, dbd,
                None, None, None, None, None, None)


def fused_bottleneck(x, block):
    d = block.downsample
    wd = d[0].weight if d is not None else None
    gd = d[1].weight if d is not None else None
    bd = d[1].bias if d is not None else None
    bnd = d[1] if d is not None else None
    prev = getattr(x, "_dtmx_bnout", None) if _bnbwd_cross_on() else None
    handle = {}
    out = _FusedBottleneck.apply(
        x, block.conv1.weight, block.bn1.weight, block.bn1.bias,
        block.conv2.weight, block.bn2.weight, block.bn2.bias,
        block.conv3.weight, block.bn3.weight, block.bn3.bias,
        wd, gd, bd, block.conv2.stride, block.bn1, block.bn2, block.bn3, bnd,
        prev, handle)
    bnout = handle.get("bnout")
    if bnout is not None:
        out._dtmx_bnout = bnout
    return out


def fused_basic_block(x, block):
    d = block.downsample
    wd = d[0].weight if d is not None else None
    gd = d[1].weight if d is not None else None
    bd = d[1].bias if d is not None else None
    bnd = d[1] if d is not None else None
    prev = getattr(x, "_dtmx_bnout", None) if _bnbwd_cross_on() else None
    handle = {}
    out = _FusedBasicBlock.apply(
        x, block.conv1.weight, block.bn1.weight, block.bn1.bias,
        block.conv2.weight, block.bn2.weight, block.bn2.bias,
        wd, gd, bd, block.conv1.stride, block.bn1, block.bn2, bnd,
        prev, handle)
    bnout = handle.get("bnout")
    if bnout is not None:
        out._dtmx_bnout = bnout
    return out
