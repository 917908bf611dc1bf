"""Fused bottleneck block: one autograd Function for the whole
conv1-bn1-relu / conv2-bn2-relu / conv3-bn3(+res)+relu residual block
(identity shortcut or conv+BN downsample shortcut).

Why this exists (MI355X-first): with per-layer autograd Functions, the
gradient w.r.t. the block input arrives from TWO paths — conv1's dgrad and
the shortcut path (``dres`` from the fused bn3 backward, possibly through
the downsample conv's dgrad) — and the autograd engine materialises both
then sums them with an elementwise add (3 HBM passes over an N*C*H*W tensor
per join; ~4% of the ResNet-50 step in the steady-state profile). Owning
the whole block's backward lets the join-add ride conv1's dgrad epilogue
(``conv_backward(..., acc=...)``): the add becomes one extra coalesced read
in a kernel that was writing that tensor anyway. Measured +1% whole-model
step throughput on 1x MI355X.

Numerics match the per-layer path: identical kernels in the identical
order; the only difference is where the (bf16 + bf16) join-add rounds, and
the fused path rounds from the fp32 accumulator (>= the engine-add's
precision). Covered by ``tests/test_conv_kernels.py`` parity tests vs both
the stock fp32 oracle and the unfused HIP path.

Enabled by default on the HIP training path; ``DDLW_FUSED_BLOCK=0``
disables (falls back to per-layer Functions). Reference scope: this fuses
the backward of the reference's K1-K5 kernel chain at
block granularity; the reference itself leaves fusion to XLA.
"""
from __future__ import annotations

import os

import torch

from . import binding, conv_gemm


def fused_block_enabled() -> bool:
    return os.environ.get("DDLW_FUSED_BLOCK", "1") == "1"


def wgrad_stream_enabled() -> bool:
    """Run the weight-gradient chain of the fused block on a side HIP
    stream (wgrads are independent of the dgrad/BN chain). Measured a WASH
    on 1x MI355X (8844 vs 8855 img/s single-stream): the step is ~99%
    kernel-busy and bandwidth-saturated, so co-running the wgrad GEMMs with
    the small finalize kernels does not compress the timeline. Default OFF;
    kept behind DDLW_WGRAD_STREAM=1 for future configs (e.g. smaller
    per-GPU batch where kernels stop saturating the chip)."""
    return os.environ.get("DDLW_WGRAD_STREAM", "0") == "1"


_wgrad_streams = {}


def _wgrad_stream(device) -> "torch.cuda.Stream":
    s = _wgrad_streams.get(device)
    if s is None:
        s = _wgrad_streams[device] = torch.cuda.Stream(device=device)
    return s


def _cl(t: torch.Tensor) -> torch.Tensor:
    return t.contiguous(memory_format=torch.channels_last)


def _fwd_conv(x, w, stride, padding, hip_fwd):
    if hip_fwd:
        return conv_gemm.conv_fwd_kernel(x, w.to(torch.bfloat16), stride, padding)
    return torch.nn.functional.conv2d(x, w.to(x.dtype), None, stride, padding)


def _dgrad_bn_reduce(dy, conv_in, weight, stride, padding, hip_dgrad,
                     x_bn, mask, mean, rstd):
    """conv dgrad + the NEXT BatchNorm backward's (dbeta, dgamma) reduce.
    On the stride-1 hip route the dgrad epilogue emits the reduce partials
    (dy never re-read); otherwise dgrad then the standalone reduce pass.
    DDLW_FUSED_BNB=0 disables (A/B lever). Returns (dx, dbeta, dgamma)."""
    st = stride[0] if isinstance(stride, (tuple, list)) else stride
    if (hip_dgrad and st == 1
            and os.environ.get("DDLW_FUSED_BNB", "1") == "1"):
        dx, parts, np_ = conv_gemm.conv_dgrad_kernel(
            dy, weight.to(torch.bfloat16), conv_in.shape, padding, stride,
            bnb=(x_bn, mask, mean, rstd))
        db, dg = binding.bn_grad_finalize_parts(
            parts[0], parts[1], np_, x_bn.shape[1])
        return dx, db, dg
    dx, _ = conv_gemm.conv_backward(dy, conv_in, weight, stride, padding,
                                    hip_dgrad, False, need_dw=False)
    db, dg = binding.bn_bwd_reduce(dx, mask, x_bn, mean, rstd,
                                   mask is not None)
    return dx, db, dg


def _fwd_conv_bn_stats(x, w, stride, padding, hip_fwd, bn):
    """conv + BN batch statistics. On the hip path the conv epilogue emits
    per-tile partial sums (no separate k_bn_stats read of the conv output);
    otherwise conv then the standalone stats pass. Returns (t, mean, rstd).
    DDLW_FUSED_BN_STATS=0 disables the fusion (A/B lever)."""
    if hip_fwd and os.environ.get("DDLW_FUSED_BN_STATS", "1") == "1":
        t, parts, np_ = conv_gemm.conv_fwd_kernel(
            x, w.to(torch.bfloat16), stride, padding, bn_parts=True)
        rows = t.shape[0] * t.shape[2] * t.shape[3]
        mean, rstd = binding.bn_finalize_parts(
            parts[0], parts[1], np_, rows, t.shape[1], bn.eps, bn.momentum,
            bn.running_mean, bn.running_var)
        return t, mean, rstd
    t = _fwd_conv(x, w, stride, padding, hip_fwd)
    mean, rstd = binding.bn_stats(t, bn.eps, bn.momentum,
                                  bn.running_mean, bn.running_var)
    return t, mean, rstd


class _BottleneckFn(torch.autograd.Function):
    """Training-mode fused bottleneck. ``wd/gd/bd`` are the downsample
    conv/BN parameters (None for identity-shortcut blocks)."""

    @staticmethod
    def forward(ctx, x, w1, g1, b1, w2, g2, b2, w3, g3, b3, wd, gd, bd, block):
        bn1, bn2, bn3 = block.bn1, block.bn2, block.bn3
        stride = block.stride
        mode = os.environ.get("DDLW_CONV", "auto")
        x = _cl(x)

        if wd is not None:
            dsc = block.downsample.conv
            dsbn = block.downsample.bn
            conv_gemm.available(dsc, x, mode)
            rdf, rdd, rdg = getattr(dsc, "_ddlw_route", (False, False, False))
            td, md, sd = _fwd_conv_bn_stats(x, wd, stride, 0, rdf, dsbn)
            res, _ = binding.bn_apply(td, None, md, sd, gd, bd, False)
            dsbn.num_batches_tracked += 1
        else:
            rdd = rdg = False
            td = md = sd = None
            res = x

        conv_gemm.available(block.conv1, x, mode)
        r1f, r1d, r1g = getattr(block.conv1, "_ddlw_route", (False, False, False))
        t1, m1, s1 = _fwd_conv_bn_stats(x, w1, 1, 0, r1f, bn1)
        a1, mask1 = binding.bn_apply(t1, None, m1, s1, g1, b1, True)

        conv_gemm.available(block.conv2, a1, mode)
        r2f, r2d, r2g = getattr(block.conv2, "_ddlw_route", (False, False, False))
        t2, m2, s2 = _fwd_conv_bn_stats(a1, w2, stride, 1, r2f, bn2)
        a2, mask2 = binding.bn_apply(t2, None, m2, s2, g2, b2, True)

        conv_gemm.available(block.conv3, a2, mode)
        r3f, r3d, r3g = getattr(block.conv3, "_ddlw_route", (False, False, False))
        t3, m3, s3 = _fwd_conv_bn_stats(a2, w3, 1, 0, r3f, bn3)
        out, mask3 = binding.bn_apply(t3, res, m3, s3, g3, b3, True)

        for bn in (bn1, bn2, bn3):
            bn.num_batches_tracked += 1
        ctx.save_for_backward(x, w1, g1, w2, g2, w3, g3, wd, gd,
                              t1, a1, t2, a2, t3, td,
                              mask1, mask2, mask3,
                              m1, s1, m2, s2, m3, s3, md, sd)
        ctx.routes = ((r1d, r1g), (r2d, r2g), (r3d, r3g), (rdd, rdg))
        ctx.stride = stride
        return out

    @staticmethod
    def backward(ctx, dy):
        (x, w1, g1, w2, g2, w3, g3, wd, gd,
         t1, a1, t2, a2, t3, td,
         mask1, mask2, mask3,
         m1, s1, m2, s2, m3, s3, md, sd) = ctx.saved_tensors
        (r1d, r1g), (r2d, r2g), (r3d, r3g), (rdd, rdg) = ctx.routes
        stride = ctx.stride
        dy = _cl(dy)
        if dy.dtype != torch.bfloat16:
            dy = dy.to(torch.bfloat16)

        # wgrads are independent of the dgrad/BN chain: launch them on a
        # side stream so their GEMMs co-run with the chain's small
        # latency-bound kernels (grid-underfilled finalizes)
        use_side = wgrad_stream_enabled() and dy.is_cuda
        main = torch.cuda.current_stream() if use_side else None
        side = _wgrad_stream(dy.device) if use_side else None
        side_evs = []

        def wgrad(dy_t, x_t, w, st, pd, hip_g):
            if not use_side:
                _, dw = conv_gemm.conv_backward(dy_t, x_t, w, st, pd,
                                                False, hip_g, need_dx=False)
                return dw
            ready = torch.cuda.Event()
            ready.record(main)  # dy_t complete on the main stream
            side.wait_event(ready)
            with torch.cuda.stream(side):
                # inputs were allocated on the main stream; tell the caching
                # allocator they are in use on the side stream
                dy_t.record_stream(side)
                x_t.record_stream(side)
                _, dw = conv_gemm.conv_backward(dy_t, x_t, w, st, pd,
                                                False, hip_g, need_dx=False)
                done = torch.cuda.Event()
                done.record(side)
            side_evs.append((done, dw))
            return dw

        # bn3 (+residual, +relu): dres is the shortcut-path gradient
        db3, dg3 = binding.bn_bwd_reduce(dy, mask3, t3, m3, s3, True)
        dt3, dres = binding.bn_bwd_dx(dy, mask3, t3, m3, s3, g3, db3, dg3,
                                      True, True)
        dw3 = wgrad(dt3, a2, w3, 1, 0, r3g)
        da2, db2, dg2 = _dgrad_bn_reduce(dt3, a2, w3, 1, 0, r3d,
                                         t2, mask2, m2, s2)
        dt2, _ = binding.bn_bwd_dx(da2, mask2, t2, m2, s2, g2, db2, dg2,
                                   True, False)
        dw2 = wgrad(dt2, a1, w2, stride, 1, r2g)
        da1, db1, dg1 = _dgrad_bn_reduce(dt2, a1, w2, stride, 1, r2d,
                                         t1, mask1, m1, s1)
        dt1, _ = binding.bn_bwd_dx(da1, mask1, t1, m1, s1, g1, db1, dg1,
                                   True, False)

        dwd = dgd = dbd = None
        if wd is not None:
            # downsample shortcut: dres -> bn_d bwd -> conv_d dgrad -> join
            dbd, dgd = binding.bn_bwd_reduce(dres, None, td, md, sd, False)
            dtd, _ = binding.bn_bwd_dx(dres, None, td, md, sd, gd, dbd, dgd,
                                       False, False)
            dwd = wgrad(dtd, x, wd, stride, 0, rdg)
            dxd, _ = conv_gemm.conv_backward(dtd, x, wd, stride, 0, rdd, False,
                                             need_dw=False)
            join = dxd
        else:
            join = dres
        dw1 = wgrad(dt1, x, w1, 1, 0, r1g)
        # the join-add rides conv1's dgrad epilogue (acc=join)
        dx, _ = conv_gemm.conv_backward(dt1, x, w1, 1, 0, r1d, False,
                                        need_dw=False, acc=join)

        # rejoin: the optimizer (main stream) reads the dw tensors
        for done, dw in side_evs:
            main.wait_event(done)
            dw.record_stream(main)

        return (dx, dw1, dg1, db1, dw2, dg2, db2, dw3, dg3, db3,
                dwd, dgd, dbd, None)


@torch.no_grad()
def bottleneck_eval_forward(block, x: torch.Tensor) -> torch.Tensor:
    """Eval-mode bottleneck: BN folded into the conv epilogues — 4 kernels
    per block (vs conv+bn_apply pairs). Running stats are constants in
    eval, so y = relu(conv*scale + bias [+ res]) is exact BN semantics."""
    x = _cl(x)
    stride = block.stride

    def _conv_ep(inp, w, bn, relu, acc=None, st=1, pad=0):
        return conv_gemm.conv_fwd_kernel(
            inp, w.to(torch.bfloat16), st, pad, acc=acc,
            ep=(*bn.folded_scale_bias(), relu))

    if block.downsample is not None:
        ds = block.downsample
        res = _conv_ep(x, ds.conv.weight, ds.bn, False, st=stride, pad=0)
    else:
        res = x
    a1 = _conv_ep(x, block.conv1.weight, block.bn1, True)
    a2 = _conv_ep(a1, block.conv2.weight, block.bn2, True, st=stride, pad=1)
    return _conv_ep(a2, block.conv3.weight, block.bn3, True, acc=res)


def bottleneck_eval_fusable(block, x: torch.Tensor) -> bool:
    if block.training or torch.is_grad_enabled():
        return False
    if not (x.is_cuda and x.dtype == torch.bfloat16):
        return False
    if os.environ.get("DDLW_DISABLE_HIP_OPS", "0") == "1":
        return False
    if os.environ.get("DDLW_EVAL_FOLD", "1") != "1":
        return False
    if os.environ.get("DDLW_CONV", "auto") == "stock":
        return False
    # every conv must be hip-SUPPORTED (C % 64; the fwd kernel won all 22
    # measured routes, so support — not the batch-keyed table — gates eval)
    for conv in [block.conv1, block.conv2, block.conv3] + (
            [block.downsample.conv] if block.downsample is not None else []):
        if not conv_gemm.fwd_supported(conv.in_channels, conv.out_channels,
                                       *conv.kernel_size):
            return False
    for bn in [block.bn1, block.bn2, block.bn3] + (
            [block.downsample.bn] if block.downsample is not None else []):
        if bn.running_mean.dtype != torch.float32:
            return False
    return True


def bottleneck_fusable(block, x: torch.Tensor) -> bool:
    """Fused path: training, grad-enabled, bf16 CUDA input, fp32 BN
    params/stats, kernel-supported channel counts."""
    if not fused_block_enabled():
        return False
    if not (block.training and torch.is_grad_enabled()):
        return False
    if not (x.is_cuda and x.dtype == torch.bfloat16):
        return False
    if os.environ.get("DDLW_DISABLE_HIP_OPS", "0") == "1":
        return False
    bns = [block.bn1, block.bn2, block.bn3]
    if block.downsample is not None:
        bns.append(block.downsample.bn)
    for bn in bns:
        if bn.weight.dtype != torch.float32 or bn.running_mean.dtype != torch.float32:
            return False
        if not binding.supported_channels(bn.num_features):
            return False
    return True


def bottleneck_forward(block, x: torch.Tensor) -> torch.Tensor:
    if block.downsample is not None:
        wd = block.downsample.conv.weight
        gd = block.downsample.bn.weight
        bd = block.downsample.bn.bias
    else:
        wd = gd = bd = None
    return _BottleneckFn.apply(
        x,
        block.conv1.weight, block.bn1.weight, block.bn1.bias,
        block.conv2.weight, block.bn2.weight, block.bn2.bias,
        block.conv3.weight, block.bn3.weight, block.bn3.bias,
        wd, gd, bd,
        block,
    )
