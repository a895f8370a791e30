"""Manual backward over a whole DenseNet block (+ its transition).

Why this exists: in the autograd virtual-concat path every segment of the
dense residual stream is consumed by up to L later GroupNorms, so
autograd materializes a pairwise bf16 `add` chain per segment per step
(~530 extra elementwise kernels on DenseNet-121).  Here backward walks
the block in reverse itself and the fused GroupNorm backward kernel
ACCUMULATES each consumer's contribution directly into one grad buffer
per segment (`gn_bwd(..., dx_accum=...)`), eliminating the adds.  The
transition (or the external grads, for the last block) is processed
first, so the buffers are written fresh — no zero-fill pass either.

Numerics: the same kernels as the autograd path; per-segment grads add
the same bf16 terms in a slightly different association order, so
results match the autograd path to bf16 rounding.

Parity anchor: reference Net/Densenet.py:14-33 (_DenseLayer / _Transition
composition); the math is GN->ReLU->1x1 -> GN->ReLU->3x3 per layer with
cat([out, x], 1), transitions GN->ReLU->1x1 -> avg_pool(2).
"""

from __future__ import annotations

import os

import torch

from . import ext
from .native import _to_nhwc3, weight_bf16

_EPS = 1e-5


def _as4(x3, n, h, w):
    return x3.view(n, h, w, -1).permute(0, 3, 1, 2)


_wcl = weight_bf16


def _conv_bwd_data(dy4, wcl, h, w, stride, pad):
    # 3x3/s1/p1 dispatches to the halo kernel's transpose-read mode
    # inside the binding (flip-free data gradient)
    return ext().conv_bwd_data(dy4, wcl, h, w, stride, pad)


class _DenseBlockFn(torch.autograd.Function):
    """One dense block, optionally ending in its transition.

    inputs:  meta=(nlayers, groups, has_transition), seg0 (4D cl bf16),
             then per layer (g1, b1, w1, g2, b2, w2), then transition
             (gt, bt, wt) when present.
    outputs: the pooled transition tensor, or (when no transition) the
             final segment list newest-first — each consumed exactly once
             by the model's final norm, so no external grad adds appear.
    """

    @staticmethod
    def forward(ctx, meta, seg0, *params):
        nlayers, groups, has_trans = meta
        n, _, h, w = seg0.shape
        # fused GN->1x1 kernels bound the samples a row chunk spans
        hw = h * w
        fused = hw >= 32 or (hw >= 16 and 128 % hw == 0)
        segs3 = [_to_nhwc3(seg0)]
        saves = []
        # Incremental norm1 statistics (OPT-IN, experimental): derive
        # each layer's mean/rstd from accumulated per-channel (sum, ssq)
        # instead of re-reading the stream.  MEASURED SLOWER at the
        # flagship (r2c26: 18.7k vs 19.0k img/s in the same session —
        # the gn_stats stream re-reads are already Infinity-Cache-served,
        # while chansum's thin per-(n,c) grid is latency-bound), so the
        # default stays on gn_stats.
        inc = fused and bool(os.environ.get("DLB_INCSTATS"))
        if inc:
            s0, q0 = ext().chan_sums(segs3[0])
            sums, ssqs = [s0], [q0]
        for li in range(nlayers):
            g1, b1, w1, g2, b2, w2 = params[6 * li:6 * li + 6]
            w1c = _wcl(w1)
            if fused:
                # fused GN->1x1: stats from the accumulated channel
                # sums, normalize at the conv's operand load — the
                # packed norm1 never exists and the stream is not
                # re-read for statistics
                if inc:
                    m1, r1 = ext().gn_stats_from_sums(sums, ssqs, groups,
                                                      hw, _EPS)
                else:
                    m1, r1 = ext().gn_stats(segs3, groups, _EPS)
                h13 = ext().gn_conv1x1_fwd(segs3, m1, r1, g1, b1, True, w1c)
                lay = [m1, r1, w1c, h13]
            else:
                y1, m1, r1 = ext().gn_fwd(segs3, g1, b1, groups, _EPS, True)
                h1 = ext().conv_fwd(_as4(y1, n, h, w), w1c, None, 1, 0)
                h13 = _to_nhwc3(h1)
                lay = [y1, m1, r1, w1c, h13]
            y2, m2, r2 = ext().gn_fwd([h13], g2, b2, groups, _EPS, True)
            w2c = _wcl(w2)
            fresh = ext().conv_fwd(_as4(y2, n, h, w), w2c, None, 1, 1)
            segs3.insert(0, _to_nhwc3(fresh))
            if inc:
                fs, fq = ext().chan_sums(segs3[0])
                sums.insert(0, fs)
                ssqs.insert(0, fq)
            saves += lay + [m2, r2, y2, w2c]
        if has_trans:
            gt, bt, wt = params[6 * nlayers:6 * nlayers + 3]
            wtc = _wcl(wt)
            if fused:
                if inc:
                    mt, rt = ext().gn_stats_from_sums(sums, ssqs, groups,
                                                      hw, _EPS)
                else:
                    mt, rt = ext().gn_stats(segs3, groups, _EPS)
                ht3 = ext().gn_conv1x1_fwd(segs3, mt, rt, gt, bt, True, wtc)
                ht4 = _as4(ht3, n, h, w)
                saves += [mt, rt, wtc]
            else:
                yt, mt, rt = ext().gn_fwd(segs3, gt, bt, groups, _EPS, True)
                ht4 = ext().conv_fwd(_as4(yt, n, h, w), wtc, None, 1, 0)
                saves += [yt, mt, rt, wtc]
            out = ext().avgpool_fwd(ht4, 2)
        ctx.save_for_backward(*segs3, *saves, *params)
        ctx.blk = (nlayers, groups, has_trans, n, h, w, fused)
        if has_trans:
            return out
        return tuple(_as4(s, n, h, w) for s in segs3)

    @staticmethod
    def backward(ctx, *douts):
        nlayers, groups, has_trans, n, h, w, fused = ctx.blk
        nseg = nlayers + 1
        LW = 8 if fused else 9       # saves per layer
        segs3 = list(ctx.saved_tensors[:nseg])
        saves = ctx.saved_tensors[nseg:]
        params = saves[len(saves) - (6 * nlayers + (3 if has_trans else 0)):]
        pgrads = [None] * len(params)

        # When the params live in a GradientSynchronizer arena, write
        # grads straight into their arena views and notify the bucket
        # machinery, returning None to autograd — removes one small add
        # kernel per parameter per step (~364 launches on DenseNet-121).
        sink = getattr(params[0], "_dlb_sink", None)
        direct = sink is not None and params[0].grad is not None

        # Direct mode defers every norm's dgamma/dbeta reduction: gn_bwd
        # returns the raw [N, 2C] partial rows, and ONE batched colsum
        # launch at the end of the block folds them all into the arena
        # grad views (each per-layer launch measured ~4-6 us of
        # latency/boundary; a DenseNet block has up to 49 of them).
        dgb_batch = []  # (part, gamma_idx, beta_idx)

        def norm_bwd(xs, dz3, gi, bi, mean, rstd, dx_accum=None):
            """gn_bwd wrapper: returns the per-segment dx list."""
            kw = dict(dx_accum=dx_accum) if dx_accum is not None else {}
            if direct:
                outs = ext().gn_bwd(xs, dz3, params[gi], params[bi], mean,
                                    rstd, groups, True, dgb_defer=True, **kw)
                dgb_batch.append((outs[-1], gi, bi))
                return outs[:-1]
            outs = ext().gn_bwd(xs, dz3, params[gi], params[bi], mean,
                                rstd, groups, True, **kw)
            pgrads[gi] = outs[-2]
            pgrads[bi] = outs[-1]
            return outs[:-2]

        def put(idx, grad):
            # grad already landed in the arena on the direct path
            if direct:
                sink.mark_ready(params[idx])
            else:
                pgrads[idx] = grad

        if has_trans:
            tbase = LW * nlayers
            if fused:
                mt, rt, wtc = saves[tbase:tbase + 3]
            else:
                yt, mt, rt, wtc = saves[tbase:tbase + 4]
            gt, bt = params[6 * nlayers], params[6 * nlayers + 1]
            dht = ext().avgpool_bwd(
                douts[0].contiguous(memory_format=torch.channels_last),
                2, h, w)
            dyt = _conv_bwd_data(dht, wtc, h, w, 1, 0)
            co, ci = wtc.shape[0], wtc.shape[1]
            wt_p = params[6 * nlayers + 2]
            wt_out = dict(out=wt_p.grad.permute(0, 2, 3, 1).reshape(co, ci)) \
                if direct else {}
            if fused:
                dwt = ext().gn_conv1x1_wrw(segs3, mt, rt, gt, bt, True, dht,
                                           **wt_out)
            else:
                dwt = ext().conv_wrw(_as4(yt, n, h, w), dht, 1, 1, 1, 0,
                                     **wt_out)
            put(6 * nlayers + 2, None if direct else
                dwt.view(co, 1, 1, ci).permute(0, 3, 1, 2))
            # the transition norm is every segment's LAST consumer: its
            # backward writes the per-segment grad buffers fresh
            dsegs = list(norm_bwd(segs3, _to_nhwc3(dyt), 6 * nlayers,
                                  6 * nlayers + 1, mt, rt))
        else:
            # external grads arrive per segment (one consumer each);
            # clone into owned buffers the kernels then accumulate into
            dsegs = [_to_nhwc3(d).clone() for d in douts]

        for li in range(nlayers - 1, -1, -1):
            if fused:
                m1, r1, w1c, h13, m2, r2, y2, w2c = \
                    saves[LW * li:LW * li + LW]
            else:
                y1, m1, r1, w1c, h13, m2, r2, y2, w2c = \
                    saves[LW * li:LW * li + LW]
            g1, b1 = params[6 * li], params[6 * li + 1]

            in_segs = segs3[nlayers - li:]
            dfresh4 = _as4(dsegs[nlayers - 1 - li], n, h, w)
            y24 = _as4(y2, n, h, w)
            dy2 = _conv_bwd_data(dfresh4, w2c, h, w, 1, 1)
            co2, ci2 = w2c.shape[0], w2c.shape[1]
            w2_p = params[6 * li + 5]
            w2_out = dict(out=w2_p.grad.permute(0, 2, 3, 1)
                          .reshape(co2, 9 * ci2)) if direct else {}
            dw2 = ext().conv_wrw(y24, dfresh4, 3, 3, 1, 1, **w2_out)
            put(6 * li + 5, None if direct else
                dw2.view(co2, 3, 3, ci2).permute(0, 3, 1, 2))
            (dh1,) = norm_bwd([h13], _to_nhwc3(dy2), 6 * li + 3,
                              6 * li + 4, m2, r2)
            dh14 = _as4(dh1, n, h, w)
            dy1 = _conv_bwd_data(dh14, w1c, h, w, 1, 0)
            co1, ci1 = w1c.shape[0], w1c.shape[1]
            w1_p = params[6 * li + 2]
            w1_out = dict(out=w1_p.grad.permute(0, 2, 3, 1)
                          .reshape(co1, ci1)) if direct else {}
            if fused:
                # fused weight grad re-normalizes segments at load time
                # from the saved stats (norm1's output was never saved)
                dw1 = ext().gn_conv1x1_wrw(in_segs, m1, r1, g1, b1, True,
                                           dh14, **w1_out)
            else:
                dw1 = ext().conv_wrw(_as4(y1, n, h, w), dh14, 1, 1, 1, 0,
                                     **w1_out)
            put(6 * li + 2, None if direct else
                dw1.view(co1, 1, 1, ci1).permute(0, 3, 1, 2))
            norm_bwd(in_segs, _to_nhwc3(dy1), 6 * li, 6 * li + 1, m1, r1,
                     dx_accum=dsegs[nlayers - li:])

        if direct and dgb_batch:
            # one launch reduces every deferred dgamma/dbeta; only then
            # are the norm params bucket-ready
            for s in range(0, len(dgb_batch), 52):
                chunk = dgb_batch[s:s + 52]
                ext().gn_dgb_reduce_multi(
                    [p for p, _, _ in chunk],
                    [params[gi].grad for _, gi, _ in chunk],
                    [params[bi].grad for _, _, bi in chunk])
            for _, gi, bi in dgb_batch:
                sink.mark_ready(params[gi])
                sink.mark_ready(params[bi])

        dseg0 = _as4(dsegs[-1], n, h, w)
        return (None, dseg0, *pgrads)


def dense_block_forward(block, transition, segs):
    """Run `block` (list of _DenseLayer) + optional `transition` through
    the manual-backward Function.  `segs` is the incoming segment list
    (length 1 in practice: stem output or previous transition)."""
    params = []
    for layer in block:
        params += [layer.norm1.weight, layer.norm1.bias, layer.conv1.weight,
                   layer.norm2.weight, layer.norm2.bias, layer.conv2.weight]
    has_trans = transition is not None
    if has_trans:
        params += [transition.norm.weight, transition.norm.bias,
                   transition.conv.weight]
    meta = (len(block), block[0].norm1.num_groups, has_trans)
    out = _DenseBlockFn.apply(meta, segs[0], *params)
    return [out] if has_trans else list(out)


def block_fn_ok(block, segs) -> bool:
    """Envelope: single incoming bf16 channels_last segment, channel
    counts in the fused-GN octet envelope, segment count within GN_MAXSEG,
    all norms sharing one group count."""
    if len(segs) != 1:
        return False
    x = segs[0]
    if not (x.is_cuda and x.dtype == torch.bfloat16 and x.dim() == 4
            and x.is_contiguous(memory_format=torch.channels_last)):
        return False
    if len(block) + 1 > 56:
        return False
    g = block[0].norm1.num_groups
    growth = block[0].conv2.weight.shape[0]
    mid = block[0].conv1.weight.shape[0]
    cin0 = x.shape[1]
    if growth % 8 or mid % 8 or cin0 % 8:
        return False
    for layer in block:
        if layer.norm1.num_groups != g or layer.norm2.num_groups != g:
            return False
    # every layer's norm1 sees cin0 + k*growth channels; divisibility by g
    # for all k needs both terms divisible (norm2 sees mid)
    return cin0 % g == 0 and growth % g == 0 and mid % g == 0 and g <= 64
