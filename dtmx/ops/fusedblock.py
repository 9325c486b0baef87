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

import torch

from .hip import require_ext


def _cl(t):
    return t.contiguous(memory_format=torch.channels_last)


class _FusedBottleneck(torch.autograd.Function):
    """conv1(1x1) -> bn1+relu -> conv2(3x3,s) -> bn2+relu -> conv3(1x1)
    -> bn3(+shortcut residual, relu); shortcut = identity or 1x1/s conv+bn."""

    @staticmethod
    def forward(ctx, x, w1, g1, b1, w2, g2, b2, w3, g3, b3, wd, gd, bd,
                stride, bn1, bn2, bn3, bnd):
        ext = require_ext()
        mom, eps = bn1.momentum, bn1.eps
        c1 = ext.conv_fwd(x, w1, 1, 0)
        y1, m1, i1 = ext.bn_fwd_train(c1, g1, b1, bn1.running_mean,
                                      bn1.running_var, mom, eps, True, None,
                                      None, None)
        c2 = ext.conv_fwd(y1, w2, stride, 1)
        y2, m2, i2 = ext.bn_fwd_train(c2, g2, b2, bn2.running_mean,
                                      bn2.running_var, mom, eps, True, None,
                                      None, None)
        c3 = ext.conv_fwd(y2, w3, 1, 0)
        if wd is not None:
            cd = ext.conv_fwd(x, wd, stride, 0)
            sc, md, idn = ext.bn_fwd_train(cd, gd, bd, bnd.running_mean,
                                           bnd.running_var, mom, eps, False,
                                           None, None, None)
        else:
            cd = sc = x
            md = idn = m1  # placeholders (unused)
        y3, m3, i3 = ext.bn_fwd_train(c3, g3, b3, bn3.running_mean,
                                      bn3.running_var, mom, eps, True, sc,
                                      None, None)
        ctx.save_for_backward(x, w1, g1, w2, g2, w3, g3, wd, gd,
                              c1, y1, c2, y2, c3, y3, cd, sc,
                              m1, i1, m2, i2, m3, i3, md, idn)
        ctx.stride = stride
        ctx.has_down = wd is not None
        return y3

    @staticmethod
    def backward(ctx, dy):
        ext = require_ext()
        (x, w1, g1, w2, g2, w3, g3, wd, gd,
         c1, y1, c2, y2, c3, y3, cd, sc,
         m1, i1, m2, i2, m3, i3, md, idn) = ctx.saved_tensors
        stride = ctx.stride
        dy = _cl(dy)
        H, W = x.shape[2], x.shape[3]
        H1, W1 = y1.shape[2], y1.shape[3]
        H2, W2 = y2.shape[2], y2.shape[3]

        dc3, dg3, db3, dres = ext.bn_bwd(c3, dy, g3, m3, i3, True, y3, True)
        dy2 = ext.conv_dgrad(dc3, w3, 1, 0, H2, W2)
        dw3 = ext.conv_wgrad(y2, dc3, 1, 1, 1, 0)
        dc2, dg2, db2 = ext.bn_bwd(c2, dy2, g2, m2, i2, True, y2, False)
        dy1 = ext.conv_dgrad(dc2, w2, stride, 1, H1, W1)
        dw2 = ext.conv_wgrad(y1, dc2, 3, 3, stride, 1)
        dc1, dg1, db1 = ext.bn_bwd(c1, dy1, g1, m1, i1, True, y1, False)
        dw1 = ext.conv_wgrad(x, dc1, 1, 1, 1, 0)
        if ctx.has_down:
            dcd, dgd, dbd = ext.bn_bwd(cd, dres, gd, md, idn, False, sc, False)
            dwd = ext.conv_wgrad(x, dcd, 1, 1, stride, 0)
            dx = ext.conv_dgrad(dc1, w1, 1, 0, H, W)
            # shortcut dgrad accumulates into dx in the epilogue
            dx = ext.conv_dgrad(dcd, wd, stride, 0, H, W, acc=dx)
        else:
            # conv1 dgrad accumulates into the residual grad in place
            dx = ext.conv_dgrad(dc1, w1, 1, 0, H, W, acc=dres)
            dwd = dgd = dbd = None
        return (dx, dw1, dg1, db1, dw2, dg2, db2, dw3, dg3, db3,
                dwd, dgd, dbd, None, None, None, None, None)


class _FusedBasicBlock(torch.autograd.Function):
    """conv1(3x3,s) -> bn1+relu -> conv2(3x3) -> bn2(+shortcut, relu)."""

    @staticmethod
    def forward(ctx, x, w1, g1, b1, w2, g2, b2, wd, gd, bd, stride,
                bn1, bn2, bnd):
        ext = require_ext()
        mom, eps = bn1.momentum, bn1.eps
        c1 = ext.conv_fwd(x, w1, stride, 1)
        y1, m1, i1 = ext.bn_fwd_train(c1, g1, b1, bn1.running_mean,
                                      bn1.running_var, mom, eps, True, None,
                                      None, None)
        c2 = ext.conv_fwd(y1, w2, 1, 1)
        if wd is not None:
            cd = ext.conv_fwd(x, wd, stride, 0)
            sc, md, idn = ext.bn_fwd_train(cd, gd, bd, bnd.running_mean,
                                           bnd.running_var, mom, eps, False,
                                           None, None, None)
        else:
            cd = sc = x
            md = idn = m1
        y2, m2, i2 = ext.bn_fwd_train(c2, g2, b2, bn2.running_mean,
                                      bn2.running_var, mom, eps, True, sc,
                                      None, None)
        ctx.save_for_backward(x, w1, g1, w2, g2, wd, gd, c1, y1, c2, y2, cd,
                              sc, m1, i1, m2, i2, md, idn)
        ctx.stride = stride
        ctx.has_down = wd is not None
        return y2

    @staticmethod
    def backward(ctx, dy):
        ext = require_ext()
        (x, w1, g1, w2, g2, wd, gd, c1, y1, c2, y2, cd, sc,
         m1, i1, m2, i2, md, idn) = ctx.saved_tensors
        stride = ctx.stride
        dy = _cl(dy)
        H, W = x.shape[2], x.shape[3]
        H1, W1 = y1.shape[2], y1.shape[3]

        dc2, dg2, db2, dres = ext.bn_bwd(c2, dy, g2, m2, i2, True, y2, True)
        dy1 = ext.conv_dgrad(dc2, w2, 1, 1, H1, W1)
        dw2 = ext.conv_wgrad(y1, dc2, 3, 3, 1, 1)
        dc1, dg1, db1 = ext.bn_bwd(c1, dy1, g1, m1, i1, True, y1, False)
        dw1 = ext.conv_wgrad(x, dc1, 3, 3, stride, 1)
        if ctx.has_down:
            dcd, dgd, dbd = ext.bn_bwd(cd, dres, gd, md, idn, False, sc, False)
            dwd = ext.conv_wgrad(x, dcd, 1, 1, stride, 0)
            dx = ext.conv_dgrad(dc1, w1, stride, 1, H, W)
            dx = ext.conv_dgrad(dcd, wd, stride, 0, H, W, acc=dx)
        else:
            dx = ext.conv_dgrad(dc1, w1, stride, 1, H, W, acc=dres)
            dwd = dgd = dbd = None
        return (dx, dw1, dg1, db1, dw2, dg2, db2, dwd, dgd, dbd,
                None, None, None, None)


def fused_bottleneck(x, block):
    d = block.downsample
    wd = d[0].weight if d is not None else None
    gd = d[1].weight if d is not None else None
    bd = d[1].bias if d is not None else None
    bnd = d[1] if d is not None else None
    return _FusedBottleneck.apply(
        x, block.conv1.weight, block.bn1.weight, block.bn1.bias,
        block.conv2.weight, block.bn2.weight, block.bn2.bias,
        block.conv3.weight, block.bn3.weight, block.bn3.bias,
        wd, gd, bd, block.conv2.stride, block.bn1, block.bn2, block.bn3, bnd)


def fused_basic_block(x, block):
    d = block.downsample
    wd = d[0].weight if d is not None else None
    gd = d[1].weight if d is not None else None
    bd = d[1].bias if d is not None else None
    bnd = d[1] if d is not None else None
    return _FusedBasicBlock.apply(
        x, block.conv1.weight, block.bn1.weight, block.bn1.bias,
        block.conv2.weight, block.bn2.weight, block.bn2.bias,
        wd, gd, bd, block.conv1.stride, block.bn1, block.bn2, bnd)
