"""Manual backward over one GoogLeNet Inception module (GPU path).

The block-Function treatment (ops/resblock.py) for the 4-branch
inception: through autograd each module costs THREE large elementwise
adds where the branch data-grads meet on the module input (27 per
GoogLeNet step, 1.7 ms) plus ~28 small AccumulateGrad adds into the
arena views.  Here the pool branch's maxpool backward writes the input
grad fresh and the other three branches' first convs ACCUMULATE into it
(`conv_bwd_data(accum_into=)`); weight grads land directly in the flat
arena, conv bias grads fold through the deterministic chansum+colsum
pair, and all seven GroupNorms' dgamma/dbeta reductions batch into one
launch.

Parity anchor: reference Net/GoogleNet.py:10-55 (_cbr branches + 4-way
cat; the 5x5-reduce branch uses the corrected conv->GN order, see
models/googlenet.py).
"""

from __future__ import annotations

import torch

from . import ext
from .native import _to_nhwc3, weight_bf16

_EPS = 1e-5

# unit = conv(k, bias) -> GN(groups) -> relu.  Fixed order:
#   0: branch1 1x1 | 1: branch2 reduce 1x1 | 2: branch2 3x3
#   3: branch3 reduce 1x1 | 4,5: branch3 3x3s | 6: branch4 1x1 (post-pool)
_KS = (1, 1, 3, 1, 3, 3, 1)


def _as4(x3, n, h, w):
    return x3.view(n, h, w, -1).permute(0, 3, 1, 2)


class _InceptionFn(torch.autograd.Function):
    """inputs: meta=(groups tuple[7], splits tuple[4]), x, then 28
    params as (w, bias, gamma, beta) x 7 units."""

    @staticmethod
    def forward(ctx, meta, x, *params):
        groups, splits = meta
        n, _, h, w = x.shape
        x3 = _to_nhwc3(x)
        x4 = _as4(x3, n, h, w)
        saves = []

        def unit(inp4, ui):
            wp, bp, gp, be = params[4 * ui:4 * ui + 4]
            wc = weight_bf16(wp)
            hc = ext().conv_fwd(inp4, wc, bp.detach().float(), 1,
                                _KS[ui] // 2)
            h3 = _to_nhwc3(hc)
            y3, m, r = ext().gn_fwd([h3], gp, be, groups[ui], _EPS, True)
            saves.extend((wc, h3, m, r, y3))
            return _as4(y3, n, h, w)

        y0 = unit(x4, 0)
        y2 = unit(unit(x4, 1), 2)
        y3c = unit(unit(unit(x4, 3), 4), 5)
        p4, idx = ext().maxpool_fwd(x4, 3, 1, 1)
        y4 = unit(p4, 6)
        out = torch.cat([y0, y2, y3c, y4], dim=1)

        ctx.save_for_backward(x3, _to_nhwc3(p4), idx, *saves, *params)
        ctx.blk = (groups, splits, n, h, w)
        return out

    @staticmethod
    def backward(ctx, dz):
        groups, splits, n, h, w = ctx.blk
        saved = ctx.saved_tensors
        x3, p3, idx = saved[:3]
        saves = saved[3:3 + 5 * 7]
        params = saved[len(saved) - 28:]
        pgrads = [None] * 28
        x4 = _as4(x3, n, h, w)
        p4 = _as4(p3, n, h, w)

        sink = getattr(params[0], "_dlb_sink", None)
        direct = sink is not None and params[0].grad is not None
        dgb_batch = []

        def unit_bwd(dy4, ui, inp4, accum_into=None):
            """reverse of one conv->GN->relu unit; returns d(inp) or
            accumulates it into `accum_into`."""
            wc, h3, m, r, _y3 = saves[5 * ui:5 * ui + 5]
            wi, bi, gi, bei = 4 * ui, 4 * ui + 1, 4 * ui + 2, 4 * ui + 3
            if direct:
                outs = ext().gn_bwd([h3], _to_nhwc3(dy4), params[gi],
                                    params[bei], m, r, groups[ui], True,
                                    dgb_defer=True)
                dgb_batch.append((outs[-1], gi, bei))
                dh3 = outs[0]
            else:
                dh3, dg, db = ext().gn_bwd([h3], _to_nhwc3(dy4), params[gi],
                                           params[bei], m, r, groups[ui],
                                           True)
                pgrads[gi] = dg
                pgrads[bei] = db
            dh4 = _as4(dh3, n, h, w)
            k = _KS[ui]
            wp = params[wi]
            co, ci = wp.shape[0], wp.shape[1]
            if direct:
                g4 = wp.grad.permute(0, 2, 3, 1)
                # reshape of a non-viewable permute would be a silent
                # copy (grad dropped) — the arena stores 4D params
                # channels_last, assert it
                assert g4.is_contiguous(), "arena grad not channels_last"
                ext().conv_wrw(inp4, dh4, k, k, 1, k // 2,
                               out=g4.reshape(co, k * k * ci))
                sink.mark_ready(wp)
            else:
                dw = ext().conv_wrw(inp4, dh4, k, k, 1, k // 2)
                pgrads[wi] = dw.view(co, k, k, ci).permute(0, 3, 1, 2)
            dbias = ext().slab_sum(ext().chan_sums(dh3)[0])
            if direct:
                params[bi].grad.copy_(dbias)
                sink.mark_ready(params[bi])
            else:
                pgrads[bi] = dbias
            return ext().conv_bwd_data(dh4, wc, h, w, 1, k // 2,
                                       accum_into=accum_into)

        c0 = 0
        dzs = []
        for cb in splits:
            dzs.append(dz.narrow(1, c0, cb)
                       .contiguous(memory_format=torch.channels_last))
            c0 += cb

        # pool branch first: its maxpool backward writes the module
        # input grad FRESH; the other branches then accumulate into it
        dp4 = unit_bwd(dzs[3], 6, p4)
        dx4 = ext().maxpool_bwd(dp4, idx, h, w, 3, 1, 1)
        unit_bwd(dzs[0], 0, x4, accum_into=dx4)
        unit_bwd(unit_bwd(dzs[1], 2, _as4(saves[5 * 1 + 4], n, h, w)),
                 1, x4, accum_into=dx4)
        d3b = unit_bwd(dzs[2], 5, _as4(saves[5 * 4 + 4], n, h, w))
        d3a = unit_bwd(d3b, 4, _as4(saves[5 * 3 + 4], n, h, w))
        unit_bwd(d3a, 3, x4, accum_into=dx4)

        if direct and dgb_batch:
            ext().gn_dgb_reduce_multi(
                [pt for pt, _, _ in dgb_batch],
                [params[gi].grad for _, gi, _ in dgb_batch],
                [params[bi].grad for _, _, bi in dgb_batch])
            for _, gi, bi in dgb_batch:
                sink.mark_ready(params[gi])
                sink.mark_ready(params[bi])

        return (None, dx4, *pgrads)


def _units(mod):
    """The 7 (conv, gn) unit pairs of a models.googlenet._Inception."""
    b2r, b2c, b2n = mod.branch2[0], mod.branch2[1], mod.branch2[2]
    return [
        (mod.branch1[0], mod.branch1[1]),
        (b2r[0], b2r[1]), (b2c, b2n),
        (mod.branch3[0][0], mod.branch3[0][1]),
        (mod.branch3[1][0], mod.branch3[1][1]),
        (mod.branch3[2][0], mod.branch3[2][1]),
        (mod.branch4[0][0], mod.branch4[0][1]),
    ]


def inception_forward(mod, x):
    units = _units(mod)
    params = []
    for conv, gn in units:
        params += [conv.weight, conv.bias, gn.weight, gn.bias]
    groups = tuple(gn.num_groups for _, gn in units)
    splits = (units[0][0].weight.shape[0], units[2][0].weight.shape[0],
              units[5][0].weight.shape[0], units[6][0].weight.shape[0])
    return _InceptionFn.apply((groups, splits), x, *params)


def inception_fn_ok(mod, x) -> bool:
    if not (x.is_cuda and x.dtype == torch.bfloat16 and x.dim() == 4
            and x.is_contiguous(memory_format=torch.channels_last)):
        return False
    if x.shape[1] % 8:
        return False
    for conv, gn in _units(mod):
        co = conv.weight.shape[0]
        if conv.bias is None or co % 8 or co % gn.num_groups:
            return False
    return True
