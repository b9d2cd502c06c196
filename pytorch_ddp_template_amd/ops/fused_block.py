"""Fused ResNet BasicBlock — one autograd node for conv1→bn1(relu)→conv2→bn2
(+ optional 1x1 downsample) → add+relu.

Why (measured on MI355X, ResNet-18/CIFAR bf16 step profile):
* the conv epilogue already holds its output tile in registers, so it emits
  the BN per-block partial sums for free (``conv2d_fwd_stats``) — the BN
  stats pass that re-read every conv output disappears;
* BN backward applies the downstream ReLU mask inline (``bn_bwd(y_relu=·)``)
  — no relu_bwd pass;
* the residual add + final ReLU ride in the last BN's normalize epilogue
  (``bn_fwd_ws(addend=·)``) — the separate add_relu pass disappears.
  (The backward-side equivalent — accumulating the skip grad inside the
  conv dgrad epilogue — was measured SLOWER (+200 us/launch of per-lane
  2-B gathers against the MFMA C-layout scatter) and reverted; the skip
  grad uses a plain bandwidth-bound add.)

The unfused module path (ops/modules.py) remains the CPU/eval/odd-shape
reference; tests compare the two.  Reference scope note: the reference
template has no conv/BN at all (these ops come from the
BASELINE.json ResNet configs).
"""

from __future__ import annotations

import os

import torch

from .native import native

# Weight-gradient GEMMs can run on a second HIP stream, co-resident with the
# dgrad/BN-dx chain.  Measured on ResNet-18/CIFAR bf16: the serial stream is
# already 99% kernel-busy and co-scheduling LOST ~4%, so this defaults OFF
# (PDT_FUSED_STREAMS=1 opts in; kept for sparser models / future shapes).
_side = None


def _side_stream():
    global _side
    if _side is None:
        _side = torch.cuda.Stream()
    return _side


def _use_streams():
    return os.environ.get("PDT_FUSED_STREAMS", "0") == "1"


def _flat(t):
    return t.reshape(-1, t.shape[-1])


class _FusedBasicBlockFn(torch.autograd.Function):
    @staticmethod
    def forward(
        ctx, x, w1, g1, b1, w2, g2, b2, wd, gd, bd, bn1, bn2, bnd, stride
    ):
        ext = native()
        x = x.contiguous()
        mom, eps = bn1.momentum, bn1.eps
        t1, ws1 = ext.conv2d_fwd_stats(x, w1, None, stride, 1, False)
        z2, mean1, rstd1 = ext.bn_fwd_ws(
            _flat(t1), g1, b1, ws1, bn1.running_mean, bn1.running_var,
            mom, eps, True,
        )
        z = z2.reshape(t1.shape)
        t2, ws2 = ext.conv2d_fwd_stats(z, w2, None, 1, 1, False)
        if wd is not None:
            td, wsd = ext.conv2d_fwd_stats(x, wd, None, stride, 0, False)
            idn2, meand, rstdd = ext.bn_fwd_ws(
                _flat(td), gd, bd, wsd, bnd.running_mean, bnd.running_var,
                mom, eps, False,
            )
            idn = idn2.reshape(td.shape)
        else:
            td = meand = rstdd = None
            idn = x
        # residual add + relu fused into bn2's normalize pass (one fewer
        # full-tensor read+write than a separate add_relu)
        out2, mean2, rstd2 = ext.bn_fwd_ws(
            _flat(t2), g2, b2, ws2, bn2.running_mean, bn2.running_var,
            mom, eps, True, _flat(idn.contiguous()),
        )
        out = out2.reshape(t2.shape)
        ctx.save_for_backward(
            x, w1, g1, t1, z, mean1, rstd1, w2, g2, t2, mean2, rstd2,
            wd, gd, td, meand, rstdd, out,
        )
        ctx.stride = stride
        return out

    @staticmethod
    def backward(ctx, dy):
        (x, w1, g1, t1, z, mean1, rstd1, w2, g2, t2, mean2, rstd2,
         wd, gd, td, meand, rstdd, out) = ctx.saved_tensors
        stride = ctx.stride
        ext = native()
        dy = dy.contiguous()
        H, W = x.shape[1], x.shape[2]
        Hz, Wz = z.shape[1], z.shape[2]

        # apply the add_relu mask ONCE; bn2/bnd backwards then read two
        # tensors instead of three, and the skip grad is g itself
        g = ext.relu_bwd(dy, out)
        g2d = _flat(g)
        dx2, dg2, db2 = ext.bn_bwd(g2d, _flat(t2), g2, mean2, rstd2, None)
        dx2 = dx2.reshape(t2.shape)
        use_side = _use_streams()
        main = torch.cuda.current_stream()
        side = _side_stream() if use_side else None
        if use_side:
            side.wait_stream(main)
            with torch.cuda.stream(side):
                dw2 = ext.conv2d_wgrad(dx2, z, 1, 1, 3, 3).to(w2.dtype)
        # conv2 backward dgrad stays on the critical path
        dz = ext.conv2d_dgrad(dx2, w2, 1, 1, Hz, Wz)
        if not use_side:
            dw2 = ext.conv2d_wgrad(dx2, z, 1, 1, 3, 3).to(w2.dtype)
        # bn1 backward (its own fused ReLU: z is bn1's relu output)
        dt1, dg1, db1 = ext.bn_bwd(
            _flat(dz), _flat(t1), g1, mean1, rstd1, _flat(z)
        )
        dt1 = dt1.reshape(t1.shape)
        if use_side:
            side.wait_stream(main)
            with torch.cuda.stream(side):
                dw1 = ext.conv2d_wgrad(dt1, x, stride, 1, 3, 3).to(w1.dtype)
        else:
            dw1 = ext.conv2d_wgrad(dt1, x, stride, 1, 3, 3).to(w1.dtype)

        if wd is None:
            # in-place in-house add (the skip grad rides the dgrad output
            # buffer; round 1 used an eager out-of-place torch add here)
            dx = ext.add_(ext.conv2d_dgrad(dt1, w1, stride, 1, H, W), g)
            dwd = dgd = dbd = None
        else:
            dxa = ext.conv2d_dgrad(dt1, w1, stride, 1, H, W)
            dtd, dgd, dbd = ext.bn_bwd(g2d, _flat(td), gd, meand, rstdd, None)
            dtd = dtd.reshape(td.shape)
            if use_side:
                side.wait_stream(main)
                with torch.cuda.stream(side):
                    dwd = ext.conv2d_wgrad(dtd, x, stride, 0, 1, 1).to(wd.dtype)
            else:
                dwd = ext.conv2d_wgrad(dtd, x, stride, 0, 1, 1).to(wd.dtype)
            dx = ext.add_(ext.conv2d_dgrad(dtd, wd, stride, 0, H, W), dxa)
        if use_side:
            # weight grads are consumed (bucket accumulation) on the main
            # stream after this node returns
            main.wait_stream(side)
            for t in (dw2, dw1) + ((dwd,) if wd is not None else ()):
                t.record_stream(main)
        return (dx, dw1, dg1, db1, dw2, dg2, db2, dwd, dgd, dbd,
                None, None, None, None)


def fused_basic_block(x, conv1, bn1, conv2, bn2, convd, bnd, stride):
    """Run a BasicBlock through the fused Function (training, native path)."""
    return _FusedBasicBlockFn.apply(
        x, conv1.weight, bn1.weight, bn1.bias,
        conv2.weight, bn2.weight, bn2.bias,
        convd.weight if convd is not None else None,
        bnd.weight if bnd is not None else None,
        bnd.bias if bnd is not None else None,
        bn1, bn2, bnd, stride,
    )


class _FusedBottleneckFn(torch.autograd.Function):
    """ResNet Bottleneck (1x1 -> 3x3(s) -> 1x1, + optional 1x1(s) downsample)
    as one autograd node, same fusions as the BasicBlock variant."""

    @staticmethod
    def forward(ctx, x, w1, g1, b1, w2, g2, b2, w3, g3, b3, wd, gd, bd,
                bn1, bn2, bn3, bnd, stride):
        ext = native()
        x = x.contiguous()
        mom, eps = bn1.momentum, bn1.eps
        t1, ws1 = ext.conv2d_fwd_stats(x, w1, None, 1, 0, False)
        z1f, mean1, rstd1 = ext.bn_fwd_ws(
            _flat(t1), g1, b1, ws1, bn1.running_mean, bn1.running_var,
            mom, eps, True)
        z1 = z1f.reshape(t1.shape)
        t2, ws2 = ext.conv2d_fwd_stats(z1, w2, None, stride, 1, False)
        z2f, mean2, rstd2 = ext.bn_fwd_ws(
            _flat(t2), g2, b2, ws2, bn2.running_mean, bn2.running_var,
            mom, eps, True)
        z2 = z2f.reshape(t2.shape)
        t3, ws3 = ext.conv2d_fwd_stats(z2, w3, None, 1, 0, False)
        if wd is not None:
            td, wsd = ext.conv2d_fwd_stats(x, wd, None, stride, 0, False)
            idnf, meand, rstdd = ext.bn_fwd_ws(
                _flat(td), gd, bd, wsd, bnd.running_mean, bnd.running_var,
                mom, eps, False)
            idn = idnf.reshape(td.shape)
        else:
            td = meand = rstdd = None
            idn = x
        # residual add + relu fused into bn3's normalize pass
        out3, mean3, rstd3 = ext.bn_fwd_ws(
            _flat(t3), g3, b3, ws3, bn3.running_mean, bn3.running_var,
            mom, eps, True, _flat(idn.contiguous()))
        out = out3.reshape(t3.shape)
        ctx.save_for_backward(
            x, w1, g1, t1, z1, mean1, rstd1, w2, g2, t2, z2, mean2, rstd2,
            w3, g3, t3, mean3, rstd3, wd, gd, td, meand, rstdd, out)
        ctx.stride = stride
        return out

    @staticmethod
    def backward(ctx, dy):
        (x, w1, g1, t1, z1, mean1, rstd1, w2, g2, t2, z2, mean2, rstd2,
         w3, g3, t3, mean3, rstd3, wd, gd, td, meand, rstdd, out) = (
            ctx.saved_tensors)
        stride = ctx.stride
        ext = native()
        dy = dy.contiguous()
        H, W = x.shape[1], x.shape[2]
        H1, W1 = z1.shape[1], z1.shape[2]
        H2, W2 = z2.shape[1], z2.shape[2]

        g = ext.relu_bwd(dy, out)
        g2d = _flat(g)
        dx3, dg3, db3 = ext.bn_bwd(g2d, _flat(t3), g3, mean3, rstd3, None)
        dx3 = dx3.reshape(t3.shape)
        dz2 = ext.conv2d_dgrad(dx3, w3, 1, 0, H2, W2)
        dw3 = ext.conv2d_wgrad(dx3, z2, 1, 0, 1, 1).to(w3.dtype)

        dt2, dg2, db2 = ext.bn_bwd(
            _flat(dz2), _flat(t2), g2, mean2, rstd2, _flat(z2))
        dt2 = dt2.reshape(t2.shape)
        dz1 = ext.conv2d_dgrad(dt2, w2, stride, 1, H1, W1)
        dw2 = ext.conv2d_wgrad(dt2, z1, stride, 1, 3, 3).to(w2.dtype)

        dt1, dg1, db1 = ext.bn_bwd(
            _flat(dz1), _flat(t1), g1, mean1, rstd1, _flat(z1))
        dt1 = dt1.reshape(t1.shape)
        dw1 = ext.conv2d_wgrad(dt1, x, 1, 0, 1, 1).to(w1.dtype)

        if wd is None:
            dx = ext.add_(ext.conv2d_dgrad(dt1, w1, 1, 0, H, W), g)
            dwd = dgd = dbd = None
        else:
            dxa = ext.conv2d_dgrad(dt1, w1, 1, 0, H, W)
            dtd, dgd, dbd = ext.bn_bwd(g2d, _flat(td), gd, meand, rstdd, None)
            dtd = dtd.reshape(td.shape)
            dwd = ext.conv2d_wgrad(dtd, x, stride, 0, 1, 1).to(wd.dtype)
            dx = ext.add_(ext.conv2d_dgrad(dtd, wd, stride, 0, H, W), dxa)
        return (dx, dw1, dg1, db1, dw2, dg2, db2, dw3, dg3, db3,
                dwd, dgd, dbd, None, None, None, None, None)


def fused_bottleneck(x, conv1, bn1, conv2, bn2, conv3, bn3, convd, bnd,
                     stride):
    return _FusedBottleneckFn.apply(
        x, conv1.weight, bn1.weight, bn1.bias,
        conv2.weight, bn2.weight, bn2.bias,
        conv3.weight, bn3.weight, bn3.bias,
        convd.weight if convd is not None else None,
        bnd.weight if bnd is not None else None,
        bnd.bias if bnd is not None else None,
        bn1, bn2, bn3, bnd, stride,
    )
