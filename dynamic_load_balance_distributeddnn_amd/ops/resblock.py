"""Manual backward over one ResNet bottleneck block (GPU path).

Why this exists: through autograd, each of ResNet-101's 33 bottlenecks
costs one large elementwise add at the residual junction (the skip grad
and conv1's data grad meet on the block input) plus ~10 small
AccumulateGrad adds into the flat-arena views — measured 5.1% of the
step (profiles/steady_r2final_resnet101.txt).  Here backward walks the
block in reverse itself: conv1's data-grad kernel ACCUMULATES into the
skip grad (`conv_bwd_data(..., accum_into=)`), weight grads land
directly in the arena (`conv_wrw(..., out=)`), and the four GroupNorms'
dgamma/dbeta reductions batch into ONE deterministic colsum launch.

Numerics: the same kernels as the autograd path, same association
order except the junction sum (bf16 rounding at one add).

Parity anchor: reference Net/Resnet.py:30-55 (Bottleneck: 1x1 -> GN ->
relu -> 3x3(stride) -> GN -> relu -> 1x1 -> GN, + shortcut, relu).
"""

from __future__ import annotations

import torch

from . import ext
from .native import _to_nhwc3, weight_bf16

_EPS = 1e-5


def _as4(x3, n, h, w):
    return x3.view(n, h, w, -1).permute(0, 3, 1, 2)


class _BottleneckFn(torch.autograd.Function):
    """inputs: meta=(groups, stride, has_proj), x (4D cl bf16), then
    params (w1, g1, b1, w2, g2, b2, w3, go, bo[, wp, gp, bp]).
    output: relu(GN(conv3(...)) + res) (4D cl bf16)."""

    @staticmethod
    def forward(ctx, meta, x, *params):
        groups, stride, has_proj = meta
        n, _, h, w = x.shape
        h2, w2 = h // stride, w // stride
        w1, g1, b1, w2_, g2, b2, w3, go, bo = params[:9]
        w1c, w2c, w3c = weight_bf16(w1), weight_bf16(w2_), weight_bf16(w3)
        x3 = _to_nhwc3(x)
        x4 = _as4(x3, n, h, w)

        h1 = ext().conv_fwd(x4, w1c, None, 1, 0)
        h13 = _to_nhwc3(h1)
        y13, m1, r1 = ext().gn_fwd([h13], g1, b1, groups, _EPS, True)
        h2_ = ext().conv_fwd(_as4(y13, n, h, w), w2c, None, stride, 1)
        h23 = _to_nhwc3(h2_)
        y23, m2, r2 = ext().gn_fwd([h23], g2, b2, groups, _EPS, True)
        h3 = ext().conv_fwd(_as4(y23, n, h2, w2), w3c, None, 1, 0)
        h33 = _to_nhwc3(h3)

        if has_proj:
            wp, gp, bp = params[9:12]
            wpc = weight_bf16(wp)
            hp = ext().conv_fwd(x4, wpc, None, stride, 0)
            hp3 = _to_nhwc3(hp)
            res3, mp, rp = ext().gn_fwd([hp3], gp, bp, groups, _EPS, False)
            proj_saves = (wpc, hp3, mp, rp)
        else:
            res3 = x3
            proj_saves = ()

        z3, mo, ro = ext().gn_fwd([h33], go, bo, groups, _EPS, True,
                                  res=res3)
        ctx.save_for_backward(x3, w1c, h13, m1, r1, y13, w2c, h23, m2, r2,
                              y23, w3c, h33, mo, ro, res3, *proj_saves,
                              *params)
        ctx.blk = (groups, stride, has_proj, n, h, w)
        return _as4(z3, n, h2, w2)

    @staticmethod
    def backward(ctx, dz):
        groups, stride, has_proj, n, h, w = ctx.blk
        h2, w2 = h // stride, w // stride
        np_ = 12 if has_proj else 9
        saved = ctx.saved_tensors
        (x3, w1c, h13, m1, r1, y13, w2c, h23, m2, r2, y23, w3c, h33, mo,
         ro, res3) = saved[:16]
        if has_proj:
            wpc, hp3, mp, rp = saved[16:20]
        params = saved[len(saved) - np_:]
        pgrads = [None] * np_

        sink = getattr(params[0], "_dlb_sink", None)
        direct = sink is not None and params[0].grad is not None
        dgb_batch = []  # (part, gamma_idx, beta_idx)

        def norm_bwd(xs, dz3, gi, bi, mean, rstd, relu, res=None):
            kw = dict(res=res) if res is not None else {}
            if direct:
                outs = ext().gn_bwd(xs, dz3, params[gi], params[bi], mean,
                                    rstd, groups, relu, dgb_defer=True,
                                    **kw)
                dgb_batch.append((outs[-1], gi, bi))
                return outs[:-1]
            outs = ext().gn_bwd(xs, dz3, params[gi], params[bi], mean,
                                rstd, groups, relu, **kw)
            pgrads[gi] = outs[-2]
            pgrads[bi] = outs[-1]
            return outs[:-2]

        def wrw(x4, dy4, R, S, st, pad, wi):
            wp_ = params[wi]
            co, ci = wp_.shape[0], wp_.shape[1]
            if direct:
                g4 = wp_.grad.permute(0, 2, 3, 1)
                # a non-viewable permute would reshape into a COPY and
                # silently drop the grad — the arena stores 4D params
                # channels_last, assert it
                assert g4.is_contiguous(), "arena grad not channels_last"
                ext().conv_wrw(x4, dy4, R, S, st, pad,
                               out=g4.reshape(co, R * S * ci))
                sink.mark_ready(wp_)
            else:
                dw = ext().conv_wrw(x4, dy4, R, S, st, pad)
                pgrads[wi] = dw.view(co, R, S, ci).permute(0, 3, 1, 2)

        dz3 = _to_nhwc3(dz)
        dh33, dres3 = norm_bwd([h33], dz3, 7, 8, mo, ro, True, res=res3)
        dh34 = _as4(dh33, n, h2, w2)
        dy2 = ext().conv_bwd_data(dh34, w3c, h2, w2, 1, 0)
        wrw(_as4(y23, n, h2, w2), dh34, 1, 1, 1, 0, 6)
        (dh23,) = norm_bwd([h23], _to_nhwc3(dy2), 4, 5, m2, r2, True)
        dh24 = _as4(dh23, n, h2, w2)
        dy1 = ext().conv_bwd_data(dh24, w2c, h, w, stride, 1)
        wrw(_as4(y13, n, h, w), dh24, 3, 3, stride, 1, 3)
        (dh13,) = norm_bwd([h13], _to_nhwc3(dy1), 1, 2, m1, r1, True)
        dh14 = _as4(dh13, n, h, w)
        x4 = _as4(x3, n, h, w)
        wrw(x4, dh14, 1, 1, 1, 0, 0)

        if has_proj:
            (dhp3,) = norm_bwd([hp3], dres3, 10, 11, mp, rp, False)
            dhp4 = _as4(dhp3, n, h2, w2)
            wrw(x4, dhp4, 1, 1, stride, 0, 9)
            dx4 = ext().conv_bwd_data(dhp4, wpc, h, w, stride, 0)
        else:
            dx4 = _as4(dres3, n, h, w)  # owned buffer from gn_bwd
        # conv1's data grad rides the skip grad's buffer — the junction
        # add happens in the kernel epilogue, not a separate pass
        ext().conv_bwd_data(dh14, w1c, h, w, 1, 0, accum_into=dx4)

        if direct and dgb_batch:
            ext().gn_dgb_reduce_multi(
                [p for p, _, _ in dgb_batch],
                [params[gi].grad for _, gi, _ in dgb_batch],
                [params[bi].grad for _, _, bi in dgb_batch])
            for _, gi, bi in dgb_batch:
                sink.mark_ready(params[gi])
                sink.mark_ready(params[bi])

        return (None, dx4, *pgrads)


def bottleneck_forward(mod, x):
    """Run a models.resnet._Bottleneck through the manual Function."""
    c1, n1, c2, n2, c3 = mod.a
    params = [c1.weight, n1.weight, n1.bias, c2.weight, n2.weight, n2.bias,
              c3.weight, mod.norm_out.weight, mod.norm_out.bias]
    has_proj = mod.proj is not None
    if has_proj:
        params += [mod.proj[0].weight, mod.proj[1].weight,
                   mod.proj[1].bias]
    meta = (mod.norm_out.num_groups, c2.stride, has_proj)
    return _BottleneckFn.apply(meta, x, *params)


def bottleneck_fn_ok(mod, x) -> bool:
    """Envelope: bf16 channels_last CUDA input, octet channel counts,
    even spatial dims for the stride-2 blocks."""
    if not (x.is_cuda and x.dtype == torch.bfloat16 and x.dim() == 4
            and x.is_contiguous(memory_format=torch.channels_last)):
        return False
    c1, n1, c2, n2, c3 = mod.a
    if c1.bias is not None or c2.bias is not None or c3.bias is not None:
        return False
    g = mod.norm_out.num_groups
    if n1.num_groups != g or n2.num_groups != g:
        return False
    width, cout, cin = c1.weight.shape[0], c3.weight.shape[0], x.shape[1]
    if width % 8 or cout % 8 or cin % 8:
        return False
    if width % g or cout % g:
        return False
    s = c2.stride
    if s not in (1, 2) or (s == 2 and (x.shape[2] % 2 or x.shape[3] % 2)):
        return False
    if mod.proj is not None:
        if mod.proj[0].stride != s or mod.proj[1].num_groups != g:
            return False
    return True
