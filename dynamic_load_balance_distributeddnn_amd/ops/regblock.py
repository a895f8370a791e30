"""Manual backward over one RegNet block (GPU path).

The ResNet bottleneck treatment (ops/resblock.py) applied to RegNet's
block shape: 1x1 -> GN/relu -> grouped 3x3(stride) -> GN/relu ->
[SqueezeExcite] -> 1x1 -> GN(+res)/relu.  The residual-junction grad
add rides conv1's data-grad epilogue (`conv_bwd_data(accum_into=)`),
weight grads land directly in the flat arena (the grouped wrw's atomics
accumulate straight into the pre-zeroed arena view), and the block's
GroupNorm dgamma/dbeta reductions batch into one launch.  The SE
middle's backward is chained by hand: the se_bwd kernel gives du and
the gate grad, the two tiny 1x1 convs reduce to [N, C] matrix products.

Parity anchor: reference Net/RegNet.py:26-55 (Block with SE at :21-23).
"""

from __future__ import annotations

import torch

from . import ext
from .native import _to_nhwc3, weight_bf16

_EPS = 1e-5


def _as4(x3, n, h, w):
    return x3.view(n, h, w, -1).permute(0, 3, 1, 2)


def _arena2d(p, rows, cols):
    """p.grad as a contiguous [rows, cols] channels_last image, or None
    when the arena view isn't viewable that way."""
    g = p.grad
    if g is None:
        return None
    g4 = g.permute(0, 2, 3, 1)
    if not g4.is_contiguous():
        return None
    return g4.reshape(rows, cols)


class _RegBlockFn(torch.autograd.Function):
    """inputs: meta=(gb, go, stride, gw, has_se, has_proj), x, params
    (w1, g1, b1, wg, g2, b2, wb, gout, bout
     [, wr, br, we, be][, wp, gp, bp])."""

    @staticmethod
    def forward(ctx, meta, x, *params):
        gb, go, stride, gw, has_se, has_proj = meta
        n, _, h, w = x.shape
        h2, w2 = h // stride, w // stride
        w1, g1, b1, wg, g2, b2, wb = params[:7]
        gout, bout = params[7], params[8]
        w1c, wgc, wbc = weight_bf16(w1), weight_bf16(wg), weight_bf16(wb)
        x3 = _to_nhwc3(x)
        x4 = _as4(x3, n, h, w)

        h1 = ext().conv_fwd(x4, w1c, None, 1, 0)
        h13 = _to_nhwc3(h1)
        y13, m1, r1 = ext().gn_fwd([h13], g1, b1, gb, _EPS, True)
        hg = ext().gconv_fwd(_as4(y13, n, h, w), wgc, stride)
        hg3 = _to_nhwc3(hg)
        u3, m2, r2 = ext().gn_fwd([hg3], g2, b2, gb, _EPS, True)

        se_saves = ()
        if has_se:
            wr, br, we, be = params[9:13]
            s0 = ext().gavg_fwd(_as4(u3, n, h2, w2)).reshape(n, -1)
            # gate math in fp32 regardless of the surrounding autocast:
            # backward reuses these tensors in fp32 matrix products
            with torch.autocast("cuda", enabled=False):
                s0f = s0.float()
                wrf = wr.reshape(wr.shape[0], -1)
                wef = we.reshape(we.shape[0], -1)
                s1 = torch.addmm(br, s0f, wrf.t())
                rr = torch.relu(s1)
                gate = torch.addmm(be, rr, wef.t()).bfloat16().contiguous()
            v3 = ext().se_fwd(u3, gate)
            se_saves = (s0f, s1, rr, gate)
        else:
            v3 = u3

        hb = ext().conv_fwd(_as4(v3, n, h2, w2), wbc, None, 1, 0)
        hb3 = _to_nhwc3(hb)

        if has_proj:
            wp, gp, bp = params[-3:]
            wpc = weight_bf16(wp)
            hp = ext().conv_fwd(x4, wpc, None, stride, 0)
            hp3 = _to_nhwc3(hp)
            res3, mp, rp = ext().gn_fwd([hp3], gp, bp, go, _EPS, False)
            proj_saves = (wpc, hp3, mp, rp)
        else:
            res3 = x3
            proj_saves = ()

        z3, mo, ro = ext().gn_fwd([hb3], gout, bout, go, _EPS, True,
                                  res=res3)
        ctx.save_for_backward(x3, w1c, h13, m1, r1, y13, wgc, hg3, m2, r2,
                              u3, v3, wbc, hb3, mo, ro, res3, *se_saves,
                              *proj_saves, *params)
        ctx.blk = meta + (n, h, w)
        return _as4(z3, n, h2, w2)

    @staticmethod
    def backward(ctx, dz):
        gb, go, stride, gw, has_se, has_proj, n, h, w = ctx.blk
        h2, w2 = h // stride, w // stride
        np_ = 9 + (4 if has_se else 0) + (3 if has_proj else 0)
        saved = ctx.saved_tensors
        (x3, w1c, h13, m1, r1, y13, wgc, hg3, m2, r2, u3, v3, wbc, hb3,
         mo, ro, res3) = saved[:17]
        i = 17
        if has_se:
            s0f, s1, rr, gate = saved[i:i + 4]
            i += 4
        if has_proj:
            wpc, hp3, mp, rp = saved[i:i + 4]
        params = saved[len(saved) - np_:]
        pgrads = [None] * np_

        sink = getattr(params[0], "_dlb_sink", None)
        direct = sink is not None and params[0].grad is not None
        dgb_batch = []

        def norm_bwd(xs, dz3, gi, bi, groups, mean, rstd, relu, res=None):
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
            out = _arena2d(wp_, co, R * S * ci) if direct else None
            if out is not None:
                ext().conv_wrw(x4, dy4, R, S, st, pad, out=out)
                sink.mark_ready(wp_)
            else:
                dw = ext().conv_wrw(x4, dy4, R, S, st, pad)
                dw4 = dw.view(co, R, S, ci).permute(0, 3, 1, 2)
                if direct:
                    wp_.grad.add_(dw4)
                    sink.mark_ready(wp_)
                else:
                    pgrads[wi] = dw4

        def put(wi, grad):
            if direct:
                params[wi].grad.add_(grad.to(params[wi].grad.dtype)
                                     .view_as(params[wi].grad))
                sink.mark_ready(params[wi])
            else:
                pgrads[wi] = grad.view_as(params[wi])

        dz3 = _to_nhwc3(dz)
        dhb3, dres3 = norm_bwd([hb3], dz3, 7, 8, go, mo, ro, True,
                               res=res3)
        dhb4 = _as4(dhb3, n, h2, w2)
        dv = ext().conv_bwd_data(dhb4, wbc, h2, w2, 1, 0)
        wrw(_as4(v3, n, h2, w2), dhb4, 1, 1, 1, 0, 6)

        if has_se:
            wr, we = params[9], params[11]
            du3, dg = ext().se_bwd(u3, gate, _to_nhwc3(dv))
            ds2 = dg.float()
            wrf = wr.reshape(wr.shape[0], -1)
            wef = we.reshape(we.shape[0], -1)
            put(11, ds2.t().mm(rr))             # dWe [wb, se]
            put(12, ds2.sum(0))                 # dbe
            dr = ds2.mm(wef) * (s1 > 0)
            put(9, dr.t().mm(s0f))              # dWr [se, wb]
            put(10, dr.sum(0))                  # dbr
            ds0 = dr.mm(wrf)
            du3 = du3.add_((ds0 * (1.0 / (h2 * w2))).to(du3.dtype)
                           .unsqueeze(1))
            du4 = _as4(du3, n, h2, w2)
        else:
            du4 = dv.contiguous(memory_format=torch.channels_last) \
                if not dv.is_contiguous(memory_format=torch.channels_last) \
                else dv

        (dhg3,) = norm_bwd([hg3], _to_nhwc3(du4), 4, 5, gb, m2, r2, True)
        dhg4 = _as4(dhg3, n, h2, w2)
        dy1 = ext().gconv_bwd(dhg4, wgc, h, w, stride)
        wgp = params[3]
        gout_ = _arena2d(wgp, wgp.shape[0], 9 * gw) if direct else None
        if gout_ is not None:
            ext().gconv_wrw(_as4(y13, n, h, w), dhg4, gw, stride, out=gout_)
            sink.mark_ready(wgp)
        else:
            dwg = ext().gconv_wrw(_as4(y13, n, h, w), dhg4, gw, stride)
            dwg4 = dwg.view(-1, 3, 3, gw).permute(0, 3, 1, 2)
            if direct:
                wgp.grad.add_(dwg4)
                sink.mark_ready(wgp)
            else:
                pgrads[3] = dwg4

        (dh13,) = norm_bwd([h13], _to_nhwc3(dy1), 1, 2, gb, m1, r1, True)
        dh14 = _as4(dh13, n, h, w)
        x4 = _as4(x3, n, h, w)
        wrw(x4, dh14, 1, 1, 1, 0, 0)

        if has_proj:
            (dhp3,) = norm_bwd([hp3], dres3, np_ - 2, np_ - 1, go, mp, rp,
                               False)
            dhp4 = _as4(dhp3, n, h2, w2)
            wrw(x4, dhp4, 1, 1, stride, 0, np_ - 3)
            dx4 = ext().conv_bwd_data(dhp4, wpc, h, w, stride, 0)
        else:
            dx4 = _as4(dres3, n, h, w)
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


def regblock_forward(mod, x):
    """Run a models.regnet._Block through the manual Function."""
    c1, n1, cg, n2 = mod.a
    params = [c1.weight, n1.weight, n1.bias, cg.weight, n2.weight, n2.bias,
              mod.b[0].weight, mod.norm_out.weight, mod.norm_out.bias]
    has_se = mod.se is not None
    if has_se:
        params += [mod.se.reduce.weight, mod.se.reduce.bias,
                   mod.se.expand.weight, mod.se.expand.bias]
    has_proj = mod.proj is not None
    if has_proj:
        params += [mod.proj[0].weight, mod.proj[1].weight,
                   mod.proj[1].bias]
    meta = (n1.num_groups, mod.norm_out.num_groups, cg.stride,
            cg.weight.shape[1], has_se, has_proj)
    return _RegBlockFn.apply(meta, x, *params)


def regblock_fn_ok(mod, x) -> bool:
    if not (x.is_cuda and x.dtype == torch.bfloat16 and x.dim() == 4
            and x.is_contiguous(memory_format=torch.channels_last)):
        return False
    c1, n1, cg, n2 = mod.a
    if c1.bias is not None or mod.b[0].bias is not None:
        return False
    wb_ = c1.weight.shape[0]
    wout = mod.b[0].weight.shape[0]
    if wb_ % 8 or wout % 8 or x.shape[1] % 8:
        return False
    if n1.num_groups != n2.num_groups:
        return False
    gw = cg.weight.shape[1]
    if gw not in (8, 16) or cg.weight.shape[0] != wb_:
        return False
    s = cg.stride
    if s not in (1, 2) or (s == 2 and (x.shape[2] % 2 or x.shape[3] % 2)):
        return False
    if mod.proj is not None:
        if (mod.proj[0].stride != s
                or mod.proj[1].num_groups != mod.norm_out.num_groups):
            return False
    return True
