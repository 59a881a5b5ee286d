"""Fused ResNet bottleneck (GPU training path): one autograd Function spans
conv1→bn1relu→conv2→bn2relu→conv3→bn3(+residual add+relu) and the optional
downsample branch, chaining the HIP primitives manually.

What the fusion buys over per-op autograd (per block, per step):
  - the residual join's add is folded into bn3's apply pass (one fewer full
    activation read+write);
  - the join's BACKWARD sum (conv1-dgrad + skip-grad, a 51M-element
    at::add per block in autograd) happens inside conv1's dgrad epilogue
    (accumulate writer) — no separate pass;
  - autograd bookkeeping for 6+ interior nodes disappears.

Numerics are identical to the unfused path: same kernels, same order.
"""
from __future__ import annotations

import torch

from ._backend import hip_ext

import os as _os

# Conv-epilogue BN-stats fusion: measured net-negative same-box (3124 vs
# 3192 img/s — the epilogue tax on every conv exceeds the removed partials
# pass at these shapes); default OFF, MPIAMD_FUSEBN=1 enables for A/B.
_FUSEBN = _os.environ.get("MPIAMD_FUSEBN", "0") == "1"

_EMPTY = None


def _empty():
    global _EMPTY
    if _EMPTY is None:
        _EMPTY = torch.empty(0)
    return _EMPTY


class BottleneckFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, stride, eps, momentum,
                w1, g1, b1, rm1, rv1,
                w2, g2, b2, rm2, rv2,
                w3, g3, b3, rm3, rv3,
                wd, gd, bd, rmd, rvd):
        ext = hip_ext()
        e = _empty()

        def conv_bn(xin, w, st, pad, g, b, eps_, relu, rm, rv, mom, res):
            """conv with BN stats fused into the GEMM epilogue; the empty-
            slab return (split-K shapes) falls back to the partials pass.
            Returns the relu bitmask too (backward skips the y re-read)."""
            if not _FUSEBN:
                a = ext.conv2d_fwd(xin, w, st, pad)
                y, m, v, mk = ext.bn_fwd_train(a, g, b, eps_, relu, rm, rv,
                                               mom, res)
                return a, y, m, v, mk
            a, slab = ext.conv2d_fwd_bn(xin, w, st, pad)
            if slab.numel() > 0:
                y, m, v, mk = ext.bn_fwd_train_pre(a, slab, g, b, eps_, relu,
                                                   rm, rv, mom, res)
            else:
                y, m, v, mk = ext.bn_fwd_train(a, g, b, eps_, relu, rm, rv,
                                               mom, res)
            return a, y, m, v, mk

        a1, y1, m1, v1, mk1 = conv_bn(x, w1, 1, 0, g1, b1, eps, True, rm1,
                                      rv1, momentum, e)
        a2, y2, m2, v2, mk2 = conv_bn(y1, w2, stride, 1, g2, b2, eps, True,
                                      rm2, rv2, momentum, e)
        has_ds = wd is not None
        if has_ds:
            ad, res, md, vd, _ = conv_bn(x, wd, stride, 0, gd, bd, eps,
                                         False, rmd, rvd, momentum, e)
        else:
            ad = res = md = vd = None
        a3, out, m3, v3, mk3 = conv_bn(y2, w3, 1, 0, g3, b3, eps, True, rm3,
                                       rv3, momentum, res if has_ds else x)
        saved = [x, a1, y1, a2, y2, a3, out, m1, v1, m2, v2, m3, v3,
                 w1, g1, w2, g2, w3, g3, mk1, mk2, mk3]
        if has_ds:
            saved += [ad, md, vd, wd, gd]
        ctx.save_for_backward(*saved)
        ctx.stride = stride
        ctx.has_ds = has_ds
        return out

    @staticmethod
    def backward(ctx, dout):
        ext = hip_ext()
        st = ctx.stride
        (x, a1, y1, a2, y2, a3, out, m1, v1, m2, v2, m3, v3,
         w1, g1, w2, g2, w3, g3, mk1, mk2, mk3) = ctx.saved_tensors[:22]
        if ctx.has_ds:
            ad, md, vd, wd, gd = ctx.saved_tensors[22:]

        dout = dout.contiguous(memory_format=torch.channels_last)
        # The join's ReLU mask (out > 0) is applied INSIDE each consumer
        # (bn_bwd's relu path reads `out` anyway) — the old materialized
        # g = dout·(out>0) was pure data movement (2 extra passes/block).
        dx3, dg3, db3 = ext.bn_bwd(dout, a3, out, g3, m3, v3, True, mk3)
        dw3 = ext.conv2d_wgrad(y2, dx3, 1, 1, 1, 0)
        dy2 = ext.conv2d_dgrad(dx3, w3, y2.shape[2], y2.shape[3], 1, 0)
        dx2, dg2, db2 = ext.bn_bwd(dy2, a2, y2, g2, m2, v2, True, mk2)
        dw2 = ext.conv2d_wgrad(y1, dx2, 3, 3, st, 1)
        dy1 = ext.conv2d_dgrad(dx2, w2, y1.shape[2], y1.shape[3], st, 1)
        dx1, dg1, db1 = ext.bn_bwd(dy1, a1, y1, g1, m1, v1, True, mk1)
        dw1 = ext.conv2d_wgrad(x, dx1, 1, 1, 1, 0)

        dx0 = ext.conv2d_dgrad(dx1, w1, x.shape[2], x.shape[3], 1, 0)
        if ctx.has_ds:
            # mask from the join output (the downsample BN itself has no
            # ReLU): dy_ds = dout·(out>0), applied inside bn_bwd
            # the join mask (out>0) is bn3's mask
            dad, dgd, dbd = ext.bn_bwd(dout, ad, out, gd, md, vd, True, mk3)
            dwd = ext.conv2d_wgrad(x, dad, 1, 1, st, 0)
            skip = ext.conv2d_dgrad(dad, wd, x.shape[2], x.shape[3], st, 0)
            dxt = ext.add_bf16(skip, dx0)
        else:
            dwd = dgd = dbd = None
            # one-pass join: dxt = dx0 + dout·(out>0)
            dxt = ext.add_relu_bwd_add_mask(dout, mk3, dx0)

        f32 = torch.float32
        return (dxt, None, None, None,
                dw1, dg1.to(f32), db1.to(f32), None, None,
                dw2, dg2.to(f32), db2.to(f32), None, None,
                dw3, dg3.to(f32), db3.to(f32), None, None,
                dwd, dgd.to(f32) if dgd is not None else None,
                dbd.to(f32) if dbd is not None else None, None, None)


def fused_bottleneck(x, block):
    """Run `block` (a models.resnet.Bottleneck) through the fused Function."""
    ds = block.downsample
    if ds is not None:
        wd, bnd = ds[0].weight, ds[1]
        args = (wd, bnd.weight, bnd.bias, bnd.running_mean, bnd.running_var)
    else:
        args = (None, None, None, None, None)
    return BottleneckFn.apply(
        x, block.conv2.stride, block.bn1.eps, block.bn1.momentum,
        block.conv1.weight, block.bn1.weight, block.bn1.bias,
        block.bn1.running_mean, block.bn1.running_var,
        block.conv2.weight, block.bn2.weight, block.bn2.bias,
        block.bn2.running_mean, block.bn2.running_var,
        block.conv3.weight, block.bn3.weight, block.bn3.bias,
        block.bn3.running_mean, block.bn3.running_var,
        *args)
