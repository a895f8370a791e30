"""Autograd bindings over the gfx950 kernels (_dlb_kernels).

Tensors cross into kernels in NHWC ([N, HW, C] contiguous) — the natural
layout for channels_last CV activations: ``permute(0,2,3,1)`` is a free
view on a channels_last tensor.
"""

from __future__ import annotations

import torch

from . import ext


def _to_nhwc3(x: torch.Tensor) -> torch.Tensor:
    n, c, h, w = x.shape
    x3 = x.permute(0, 2, 3, 1).reshape(n, h * w, c)
    return x3 if x3.is_contiguous() else x3.contiguous()


def weight_bf16(weight: torch.Tensor) -> torch.Tensor:
    """bf16 channels_last view of a conv weight.  When the parameter
    lives in the flat arena with a bf16 mirror (FlatSGD), this is a free
    view of the mirror (the SGD kernel keeps it in sync); otherwise a
    per-call cast copy."""
    m = getattr(weight, "_dlb_bf16", None)
    if m is not None:
        return m
    w = weight.detach().to(torch.bfloat16)
    if w.dim() == 4:
        return w.contiguous(memory_format=torch.channels_last)
    return w.contiguous()


class _GroupNormAct(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, num_groups, weight, bias, eps, relu):
        n, c, h, w = x.shape
        x3 = _to_nhwc3(x)
        y3, mean, rstd = ext().gn_fwd([x3], weight, bias, num_groups, eps,
                                      relu)
        ctx.save_for_backward(x3, weight, bias, mean, rstd)
        ctx.gn_dims = (n, c, h, w, num_groups, relu)
        return y3.view(n, h, w, c).permute(0, 3, 1, 2)

    @staticmethod
    def backward(ctx, dz):
        x3, weight, bias, mean, rstd = ctx.saved_tensors
        n, c, h, w, groups, relu = ctx.gn_dims
        dz3 = _to_nhwc3(dz)
        dx3, dgamma, dbeta = ext().gn_bwd([x3], dz3, weight, bias, mean,
                                          rstd, groups, relu)
        dx = dx3.view(n, h, w, c).permute(0, 3, 1, 2)
        return dx, None, dgamma, dbeta, None, None


def group_norm_act(x, num_groups, weight, bias, eps=1e-5, relu=False):
    return _GroupNormAct.apply(x, num_groups, weight, bias, eps, relu)


class _GroupNormActCat(torch.autograd.Function):
    """GroupNorm(+ReLU) over a VIRTUAL channel-concat of segments.

    DenseNet's residual stream is a concat of every previous layer's
    output (reference Net/Densenet.py:20); reading the segments in place
    removes the per-layer cat copy (and the grad-slice copies in
    backward).  Output is packed NHWC; grads come back packed per
    segment."""

    @staticmethod
    def forward(ctx, num_groups, weight, bias, eps, relu, *segs):
        n, _, h, w = segs[0].shape
        segs3 = [_to_nhwc3(s) for s in segs]
        y3, mean, rstd = ext().gn_fwd(segs3, weight, bias, num_groups, eps,
                                      relu)
        ctx.save_for_backward(weight, bias, mean, rstd, *segs3)
        ctx.gn_dims = (n, h, w, num_groups, relu)
        c = y3.shape[-1]
        return y3.view(n, h, w, c).permute(0, 3, 1, 2)

    @staticmethod
    def backward(ctx, dz):
        weight, bias, mean, rstd, *segs3 = ctx.saved_tensors
        n, h, w, groups, relu = ctx.gn_dims
        dz3 = _to_nhwc3(dz)
        outs = ext().gn_bwd(list(segs3), dz3, weight, bias, mean, rstd,
                            groups, relu)
        dxs, dgamma, dbeta = outs[:-2], outs[-2], outs[-1]
        dx4 = [d.view(n, h, w, -1).permute(0, 3, 1, 2) for d in dxs]
        return (None, dgamma, dbeta, None, None, *dx4)


def group_norm_act_cat(segs, num_groups, weight, bias, eps=1e-5, relu=False):
    return _GroupNormActCat.apply(num_groups, weight, bias, eps, relu, *segs)


class _Conv2d(torch.autograd.Function):
    """NHWC bf16 implicit-GEMM conv (gfx950 MFMA kernels).

    Weight/bias parameters stay fp32 masters; the forward obtains the
    bf16 channels_last weight via ``weight_bf16`` — a free view of the
    SGD-maintained mirror when the param lives in the flat arena, or a
    per-call cast otherwise — and the backward returns an fp32 weight
    grad directly, matching the bf16-compute / fp32-master-grad regime
    of the whole framework.
    """

    @staticmethod
    def forward(ctx, x, weight, bias, stride, padding):
        wcl = weight_bf16(weight)
        b32 = bias.detach().float() if bias is not None else None
        y = ext().conv_fwd(x, wcl, b32, stride, padding)
        ctx.save_for_backward(x, wcl)
        ctx.conv_args = (stride, padding, bias is not None)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, wcl = ctx.saved_tensors
        stride, padding, has_bias = ctx.conv_args
        dy = dy.contiguous(memory_format=torch.channels_last)
        co, ci, r, s = wcl.shape

        dx = None
        if ctx.needs_input_grad[0]:
            # 3x3/s1/p1 shapes dispatch to the halo kernel's
            # transpose-read mode inside the binding (flip-free)
            dx = ext().conv_bwd_data(dy, wcl, x.size(2), x.size(3),
                                     stride, padding)

        dweight = None
        if ctx.needs_input_grad[1]:
            dwf = ext().conv_wrw(x, dy, r, s, stride, padding)  # [Co, RSCi]
            # fp32 [Co,R,S,Ci] image -> [Co,Ci,R,S] channels_last-strided
            dweight = dwf.view(co, r, s, ci).permute(0, 3, 1, 2)

        dbias = None
        if has_bias and ctx.needs_input_grad[2]:
            if dy.dtype == torch.bfloat16 and dy.shape[1] % 8 == 0:
                # per-(sample, channel) sums then a deterministic column
                # fold — torch's generic reduce measured 2.4 ms/step on
                # GoogLeNet's biased convs (profiles r2c30)
                s3 = _to_nhwc3(dy)
                dbias = ext().slab_sum(ext().chan_sums(s3)[0])
            else:
                dbias = dy.sum(dim=(0, 2, 3), dtype=torch.float32)

        return dx, dweight, dbias, None, None


def conv2d(x, weight, bias=None, stride=1, padding=0, groups=1):
    assert groups == 1, "grouped conv uses its own kernel path"
    return _Conv2d.apply(x, weight, bias, stride, padding)


def conv_native_ok(x, weight, stride, padding, groups) -> bool:
    """Envelope of the implicit-GEMM kernels: square stride/pad,
    ungrouped, bf16 channels_last activations."""
    if groups != 1 or x.dim() != 4 or x.dtype != torch.bfloat16:
        return False
    if not x.is_contiguous(memory_format=torch.channels_last):
        return False
    s = stride if isinstance(stride, int) else stride[0]
    p = padding if isinstance(padding, int) else padding[0]
    if not isinstance(stride, int) and stride[0] != stride[1]:
        return False
    if not isinstance(padding, int) and padding[0] != padding[1]:
        return False
    return s in (1, 2) and p >= 0


def gn_native_ok(x, num_groups, weight) -> bool:
    """Shape/dtype envelope the fused GN kernel covers (everything the
    zoo produces on the GPU path)."""
    return (x.dim() == 4 and x.dtype == torch.bfloat16
            and weight.dtype == torch.float32
            and x.shape[1] % 8 == 0 and num_groups <= 64
            and x.shape[1] % num_groups == 0)


# ---------------------------------------------------------------- pooling
class _AvgPool2d(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, k):
        ctx.pool_args = (k, x.size(2), x.size(3))
        return ext().avgpool_fwd(x, k)

    @staticmethod
    def backward(ctx, dy):
        k, h, w = ctx.pool_args
        dy = dy.contiguous(memory_format=torch.channels_last)
        return ext().avgpool_bwd(dy, k, h, w), None


class _GlobalAvgPool(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ctx.pool_hw = (x.size(2), x.size(3))
        return ext().gavg_fwd(x)

    @staticmethod
    def backward(ctx, dy):
        h, w = ctx.pool_hw
        return ext().gavg_bwd(dy, h, w)


def avg_pool2d(x, k):
    return _AvgPool2d.apply(x, k)


def global_avg_pool(x):
    return _GlobalAvgPool.apply(x)


# -------------------------------------------------------------- layernorm
class _LayerNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        x = x.contiguous()
        y, mean, rstd = ext().ln_fwd(x, weight, bias, eps)
        ctx.save_for_backward(x, weight, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dz):
        x, weight, mean, rstd = ctx.saved_tensors
        dx, dgamma, dbeta = ext().ln_bwd(x, dz, weight, mean, rstd)
        return dx, dgamma, dbeta, None


def layer_norm(x, weight, bias, eps=1e-5):
    return _LayerNorm.apply(x, weight, bias, eps)


# -------------------------------------------------------------- attention
class _CausalAttention(torch.autograd.Function):
    """Fused causal MHA with philox dropout on attention PROBABILITIES
    (reference regularization: nn.TransformerEncoderLayer drops attn
    weights at p=0.2 — Net/Transformer.py:63-64).  p_save holds the
    PRE-dropout probs; the mask is philox-recomputed in backward from
    the same (seed, element) key, so nothing extra is stored."""

    @staticmethod
    def forward(ctx, q, k, v, nhead, dropout_p):
        # draw the per-call seed from torch's CPU generator so runs are
        # reproducible under torch.manual_seed
        seed = int(torch.randint(0, 2**62, (1,)).item()) if dropout_p > 0 \
            else 0
        o, p_save = ext().attn_fwd(q, k, v, nhead, float(dropout_p), seed)
        ctx.save_for_backward(q, k, v, p_save)
        ctx.nhead = nhead
        ctx.pd = float(dropout_p)
        ctx.seed = seed
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, p_save = ctx.saved_tensors
        dq, dk, dv = ext().attn_bwd(q, k, v, do, p_save, ctx.nhead,
                                    ctx.pd, ctx.seed)
        return dq, dk, dv, None, None


def causal_attention(q, k, v, nhead, dropout_p=0.0):
    return _CausalAttention.apply(q, k, v, nhead, dropout_p)


# ------------------------------------------------------------ log_softmax
class _LogSoftmax(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        y = ext().logsoftmax_fwd(x.contiguous())
        ctx.save_for_backward(y)
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        return ext().logsoftmax_bwd(y, dy)


def log_softmax(x):
    return _LogSoftmax.apply(x)


# ------------------------------------------------------------ grouped conv
class _GroupedConv2d(torch.autograd.Function):
    """RegNet's grouped 3x3 (group_width channels per group, pad 1)."""

    @staticmethod
    def forward(ctx, x, weight, stride):
        wcl = weight_bf16(weight)
        y = ext().gconv_fwd(x, wcl, stride)
        ctx.save_for_backward(x, wcl)
        ctx.gconv_stride = stride
        return y

    @staticmethod
    def backward(ctx, dy):
        x, wcl = ctx.saved_tensors
        stride = ctx.gconv_stride
        dy = dy.contiguous(memory_format=torch.channels_last)
        c, gw = wcl.shape[0], wcl.shape[1]
        dx = None
        if ctx.needs_input_grad[0]:
            dx = ext().gconv_bwd(dy, wcl, x.size(2), x.size(3), stride)
        dweight = None
        if ctx.needs_input_grad[1]:
            dwf = ext().gconv_wrw(x, dy, gw, stride)   # [C, 9*GW]
            dweight = dwf.view(c, 3, 3, gw).permute(0, 3, 1, 2)
        return dx, dweight, None


def grouped_conv2d(x, weight, stride):
    return _GroupedConv2d.apply(x, weight, stride)


def gconv_native_ok(x, weight, stride, padding, groups) -> bool:
    if x.dim() != 4 or x.dtype != torch.bfloat16:
        return False
    if not x.is_contiguous(memory_format=torch.channels_last):
        return False
    co, cig, r, s = weight.shape
    gw = co // groups
    return (r == 3 and s == 3 and padding == 1 and stride in (1, 2)
            and cig == gw and gw in (8, 16) and co == x.shape[1])


# ---------------------------------------------------------------- maxpool
class _MaxPool2d(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, k, stride, pad):
        y, idx = ext().maxpool_fwd(x, k, stride, pad)
        ctx.save_for_backward(idx)
        ctx.mp_args = (k, stride, pad, x.size(2), x.size(3))
        return y

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        k, stride, pad, h, w = ctx.mp_args
        return ext().maxpool_bwd(dy, idx, h, w, k, stride, pad), None, None, None


def max_pool2d(x, k, stride, pad):
    return _MaxPool2d.apply(x, k, stride, pad)


class _LMLoss(torch.autograd.Function):
    """Fused decoder GEMM -> log_softmax -> NLL (mean) — the [T, V]
    logits are never materialized (reference criterion site
    dbs.py:371-374 over Net/Transformer.py:95).
    Backward recomputes logit tiles to produce dh, dW, db."""

    @staticmethod
    def forward(ctx, h2, weight, bias, targets):
        wb = weight_bf16(weight)  # free view of the FlatSGD bf16 mirror
        if not wb.is_contiguous():
            wb = wb.contiguous()
        tgt = targets.to(torch.int32)
        bias32 = bias.detach().float().contiguous()
        loss, lse = ext().lmloss_fwd(h2, wb, bias32, tgt)
        ctx.save_for_backward(h2, wb, bias32, tgt, lse)
        return loss

    @staticmethod
    def backward(ctx, go):
        h2, wb, bias32, tgt, lse = ctx.saved_tensors
        go32 = go.detach().to(torch.float32).reshape(1).contiguous()
        dh, dw, db = ext().lmloss_bwd(h2, wb, bias32, tgt, lse, go32)
        return (dh.to(h2.dtype), dw, db, None)


def lm_loss(h, weight, bias, targets):
    h2 = h.reshape(-1, h.shape[-1])
    if not h2.is_contiguous():
        h2 = h2.contiguous()
    return _LMLoss.apply(h2, weight, bias, targets.reshape(-1))


# ----------------------------------------------------- dropout (K15)
class _Dropout(torch.autograd.Function):
    """philox dropout: the mask is a pure function of (seed, index), so
    backward re-applies the same kernel to the upstream grad."""

    @staticmethod
    def forward(ctx, x, pd):
        seed = int(torch.randint(0, 2**62, (1,)).item())
        ctx.pd, ctx.seed = float(pd), seed
        return ext().dropout(x, float(pd), seed)

    @staticmethod
    def backward(ctx, dy):
        return ext().dropout(dy.contiguous(), ctx.pd, ctx.seed), None


def dropout(x, pd):
    return _Dropout.apply(x, pd)


# ------------------------------------------- embedding x sqrt(d) (K11)
class _EmbeddingScaled(torch.autograd.Function):
    @staticmethod
    def forward(ctx, idx2, weight, scale):
        wb = weight_bf16(weight)
        if not wb.is_contiguous():
            wb = wb.contiguous()
        idx32 = idx2.reshape(-1).to(torch.int32)
        out = ext().embed_fwd(wb, idx32, float(scale))
        ctx.save_for_backward(idx32)
        ctx.meta = (weight.shape[0], float(scale), idx2.shape)
        return out.view(*idx2.shape, -1)

    @staticmethod
    def backward(ctx, dy):
        (idx32,) = ctx.saved_tensors
        V, scale, _ = ctx.meta
        dt = ext().embed_bwd(dy.reshape(idx32.numel(), -1), idx32, V, scale)
        return None, dt, None


def embedding_scaled(idx, weight, scale):
    """table[idx] * scale — reference Net/Transformer.py:91 (K11)."""
    return _EmbeddingScaled.apply(idx, weight, scale)


# ------------------------------------------------- SE sigmoid-mul (K10)
class _SEMul(torch.autograd.Function):
    """y = x * sigmoid(gate) with NHWC x [N,HW,C] and gate [N,C]."""

    @staticmethod
    def forward(ctx, x3, g2):
        y = ext().se_fwd(x3, g2)
        ctx.save_for_backward(x3, g2)
        return y

    @staticmethod
    def backward(ctx, dy):
        x3, g2 = ctx.saved_tensors
        dx, dg = ext().se_bwd(x3, g2, dy)
        return dx, dg.to(g2.dtype)


def se_mul(x, gate):
    """x [N,C,H,W] channels_last * sigmoid(gate [N,C,1,1])."""
    n, c, h, w = x.shape
    x3 = _to_nhwc3(x)
    g2 = gate.reshape(n, c)
    if not g2.is_contiguous():
        g2 = g2.contiguous()
    y3 = _SEMul.apply(x3, g2)
    return y3.view(n, h, w, c).permute(0, 3, 1, 2)


# --------------------------------------- GN + residual-add + ReLU fused
class _GroupNormAddAct(torch.autograd.Function):
    """relu(GN(x) + residual) in one kernel — the ResNet/RegNet block
    junction.  Unfused this costs an add (3 passes) + relu fwd (2) +
    relu/add backward passes in torch elementwise kernels (~13% of the
    ResNet-101 step, profiles r2c22); here the residual rides the GN
    normalize/apply sweeps and the backward emits the masked residual
    grad directly."""

    @staticmethod
    def forward(ctx, x, res, num_groups, weight, bias, eps):
        n, c, h, w = x.shape
        x3 = _to_nhwc3(x)
        r3 = _to_nhwc3(res)
        y3, mean, rstd = ext().gn_fwd([x3], weight, bias, num_groups, eps,
                                      True, res=r3)
        ctx.save_for_backward(x3, r3, weight, bias, mean, rstd)
        ctx.gn_dims = (n, c, h, w, num_groups)
        return y3.view(n, h, w, c).permute(0, 3, 1, 2)

    @staticmethod
    def backward(ctx, dz):
        x3, r3, weight, bias, mean, rstd = ctx.saved_tensors
        n, c, h, w, groups = ctx.gn_dims
        dz3 = _to_nhwc3(dz)
        dx3, dres3, dgamma, dbeta = ext().gn_bwd(
            [x3], dz3, weight, bias, mean, rstd, groups, True, res=r3)
        dx = dx3.view(n, h, w, c).permute(0, 3, 1, 2)
        dres = dres3.view(n, h, w, c).permute(0, 3, 1, 2)
        return dx, dres, None, dgamma, dbeta, None


def group_norm_add_act(x, res, num_groups, weight, bias, eps=1e-5):
    return _GroupNormAddAct.apply(x, res, num_groups, weight, bias, eps)
