"""Custom implicit-GEMM convolutions (gather-GEMM on MFMA, gfx950).

Covers every 2D conv in DSIN: direct conv with any
stride/dilation, transposed conv, and the full backward — via ONE device
kernel pair (conv_fwd gather-GEMM + conv_wrw) plus torch-differentiable
geometry transforms:

  * forward flattens the weight to (Co, K=(ci,r,s)); the kernel computes
    out[pixel][cout] = sum_k xv[coords(pixel) + tap(k)] * w[cout][k], where
    xv = pad(stuff(x, sv)) exists only VIRTUALLY: the staging bounds-tests
    packed coordinate tables against the raw tensor and produces zeros for
    the pad ring and stuff holes (no padded buffer, no pad kernel, no
    extra HBM round trip — round-1's pad_stuff path cost ~2 ms/step);
  * backward-data is the transposed virtual geometry on the raw dy
    ((st', sv', pads') = (sv, st, (k-1)*dil - pads)) with spatially-rotated
    ci<->co-swapped weights; stride-1 3x3 shapes take the direct LDS-halo
    kernel (forward and backward) with its own virtual pad;
  * backward-weight is the conv_wrw kernel (per-pixel-chunk fp32 partial
    slices reduced with one sum — no atomics), same virtual staging;
  * the padded/rotated W panel is one fused kernel (wmat_make), cached
    per step per weight (ops.conv.begin_step).

bf16 compute with fp32 accumulate; fp32 master weights. Optional fused
epilogue activation (relu / leaky-0.2) for the no-batchnorm convs (siNet).
The fp8 (e4m3) path keeps explicit pad buffers: its pad kernel doubles as
the float->e4m3 quantizer, which is needed regardless.
CPU path falls back to torch.nn.functional (the numerics oracle).
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F

from . import _require_ext, hip_available

_PLANS = {}

# compute dtype for the conv path: "bf16" (default) or "fp8" (OCP e4m3,
# BASELINE config 5). fp8 tensors are carried as raw-byte uint8 views.
_COMPUTE = "bf16"

# Per-step W-panel cache. One training step evaluates each conv weight up
# to 3x (y_dec eval pass, x forward, backward's rotated panel) but the
# weights only change at the optimizer step, so wmat_make need run once per
# (weight, layout) per step (~236 -> ~100 launches/step). The Trainer bumps
# the epoch at the top of every train_step (and graph capture records the
# in-step recompute, so replays stay correct); eval calls after a step see
# the new epoch and refresh.
_WMAT_CACHE = {}
_WMAT_EPOCH = 0

# gate for the direct 5x5/stride-2 LDS-halo forward kernel
_DIRECT5 = True


def begin_step() -> None:
    """Invalidate the per-step W-panel cache (call when weights change).
    Entries are only ever valid within their own epoch, so clearing is
    free — and it releases the pinned source tensors (see _wmat_cache)."""
    global _WMAT_EPOCH
    _WMAT_EPOCH += 1
    _WMAT_CACHE.clear()


def _wmat_cache(w1: torch.Tensor, tag, builder):
    """Cache a built panel for w1 under (data_ptr, shape, tag) for the
    current epoch. The entry PINS w1 itself: several call sites pass
    TEMPORARIES (probclass weight*mask products, conv-transpose
    flip/permute chains) whose storage would otherwise be freed and
    recycled by the caching allocator — a later conv's lookup could then
    hit this entry's key with a DIFFERENT weight at the same address and
    silently run with another layer's panel. (Exactly that happened under
    no_grad, where autograd keeps no references: every eval-mode forward
    ran the probclass with shuffled panels while training looked fine.)
    Holding the reference makes live keys unique by construction."""
    if not w1.is_cuda:
        return builder()
    key = (w1.data_ptr(), tuple(w1.shape), tag)
    ent = _WMAT_CACHE.get(key)
    if ent is not None and ent[0] == _WMAT_EPOCH:
        # safe: ent pins its source tensor, so a live entry's address
        # cannot have been recycled — a key match means w1 views the very
        # same (unmodified-this-epoch) storage
        return ent[1]
    panel = builder()
    _WMAT_CACHE[key] = (_WMAT_EPOCH, panel, w1)
    return panel


def set_compute_dtype(dtype: str) -> None:
    global _COMPUTE
    assert dtype in ("bf16", "fp8")
    _COMPUTE = dtype


def compute_dtype() -> str:
    return _COMPUTE


def _plan(device, Ci: int, Hp: int, Wp: int, kh: int, kw: int, stride: int,
          dil: int, HO: int, WO: int):
    key = (device.index, Ci, Hp, Wp, kh, kw, stride, dil, HO, WO)
    p = _PLANS.get(key)
    if p is None:
        tables = _require_ext("conv_tables")
        K = Ci * kh * kw
        mbase, koff = tables(HO * WO, K, WO, stride, dil, Wp, Hp * Wp, kh, kw,
                             device)
        p = (mbase, koff)
        _PLANS[key] = p
    return p


def _plan_v2(device, M: int, K: int, WO: int, stride: int, dil: int,
             kh: int, kw: int):
    """Packed-coordinate tables for the virtual-pad gather path."""
    key = ("v2", device.index, M, K, WO, stride, dil, kh, kw)
    p = _PLANS.get(key)
    if p is None:
        p = _require_ext("conv_tables_v2")(M, K, WO, stride, dil, kh, kw,
                                           device)
        _PLANS[key] = p
    return p


def _phase_plans(device, HO: int, WO: int, kh: int, kw: int, Cin: int,
                 pt: int, pl: int):
    """Phase decomposition of a (st=1, sv=2, dil=1) virtually-stuffed gather
    (conv-transpose forward / stride-2 backward-data) into 4 DENSE stride-1
    sub-convs over the raw input: output pixels of parity (a, b) depend only
    on taps of parity ((pt+a)&1, (pl+b)&1), so each phase is an ordinary
    (st=1, sv=1) gather on the halved grids — 1/4 the MFMA work and
    full-vector staging instead of reconstructing stuff holes."""
    key = ("phase", device.index, HO, WO, kh, kw, Cin, pt, pl)
    plans = _PLANS.get(key)
    if plans is not None:
        return plans
    plans = []
    for a in (0, 1):
        for b in (0, 1):
            qa = (pt + a) & 1
            qb = (pl + b) & 1
            nrs = len(range(qa, kh, 2))
            nss = len(range(qb, kw, 2))
            nh = len(range(a, HO, 2))
            nw = len(range(b, WO, 2))
            if nrs == 0 or nss == 0 or nh == 0 or nw == 0:
                continue
            m = torch.arange(nh * nw, device=device, dtype=torch.int32)
            mpack = ((torch.div(m, nw, rounding_mode="floor")) << 16) | (m % nw)
            kk = torch.arange(Cin * nrs * nss, device=device,
                              dtype=torch.int32)
            ci = torch.div(kk, nrs * nss, rounding_mode="floor")
            rem = kk % (nrs * nss)
            dr = torch.div(rem, nss, rounding_mode="floor")
            ds = rem % nss
            kpack = (ci << 20) | (dr << 10) | ds
            ktab = ci * (kh * kw) + (qa + 2 * dr) * kw + (qb + 2 * ds)
            plans.append({
                "mpack": mpack.contiguous(), "kpack": kpack.contiguous(),
                "ktab": ktab.contiguous(),
                "ktab64": ktab.to(torch.int64).contiguous(),
                "nh": nh, "nw": nw, "Kp": int(Cin * nrs * nss),
                "ptp": (pt - a - qa) // 2, "plp": (pl - b - qb) // 2,
                "a": a, "b": b,
            })
    _PLANS[key] = plans
    return plans


def _dummy_tables(device):
    """Placeholder table args for the direct-kernel branch (unused there)."""
    key = ("dummy", device.index)
    p = _PLANS.get(key)
    if p is None:
        t = torch.zeros(1, dtype=torch.int32, device=device)
        p = (t, t)
        _PLANS[key] = p
    return p


def _wmat(w1: torch.Tensor, fp8: bool = False, direct: bool = False,
          khw: int = 9) -> torch.Tensor:
    """(Co, K) any-dtype -> bf16 (or e4m3-as-uint8) zero-padded to
    (Co, KP64+8); the zeros cancel the clamped out-of-range A gathers.
    direct: (chunk, tap, ci) mode-2 layout for the direct LDS-halo kernels
    (khw = 9 for 3x3, 25 for 5x5/s2)."""
    K = w1.shape[1]
    if fp8:
        def build8():
            KP = (K + 63) & ~63
            w8 = w1.to(torch.float8_e4m3fn).view(torch.uint8)
            out = torch.zeros(w1.shape[0], KP + 8, dtype=torch.uint8,
                              device=w1.device)
            out[:, :K] = w8
            return out
        return _wmat_cache(w1, "fp8", build8)
    if w1.is_cuda and hip_available():
        return _wmat_cache(
            w1, ("d", khw) if direct else "p",
            lambda: _require_ext("wmat_make")(w1.contiguous(),
                                              khw if direct else 1,
                                              2 if direct else 0))
    KP = (K + 63) & ~63
    return F.pad(w1.to(torch.bfloat16), (0, KP + 8 - K)).contiguous()


def _wmat_rot(w1: torch.Tensor, khw: int, direct: bool = False) -> torch.Tensor:
    """Padded bf16 W panel for the backward-data gather conv: spatial taps
    reversed and cin<->cout swapped — one kernel instead of the torch
    flip+permute+reshape+pad chain (~4 kernels per conv backward). direct:
    mode-3 (chunk, tap, ci) layout for the 3x3 direct kernel."""
    return _wmat_cache(
        w1, ("rd" if direct else "r", khw),
        lambda: _require_ext("wmat_make")(w1.contiguous(), khw,
                                          3 if direct else 1))


def _act_grad(dy: torch.Tensor, y_act, act: int) -> torch.Tensor:
    """dy * act'(y) fused to one kernel, bf16 out (act=0: plain cast)."""
    dy = dy.contiguous()
    if act:
        return _require_ext("act_bwd")(dy, y_act, act)
    if dy.dtype != torch.bfloat16:
        dy = dy.to(torch.bfloat16)
    return dy


class _GatherConvFn(torch.autograd.Function):
    """Virtual-pad gather conv:
        y[b, co, oh, ow] = sum_{ci,r,s} xv[b, ci, oh*st + r*d, ow*st + s*d]
                           * w1[co, (ci, r, s)]  (+bias, +activation)
    where xv = pad(stuff(x, sv), (pt, pl)) exists only VIRTUALLY — the
    kernels' staging bounds-tests packed coordinates against the raw x and
    produces zeros for pad ring / stuff holes (conv_kernels.h vstage8). No
    padded buffer, no pad kernel, no extra HBM round trip per conv. The
    backward-data pass is the transposed geometry on the raw dy:
    (st', sv', pads') = (sv, st, (k-1)*dil - pads) with rotated weights."""

    @staticmethod
    def forward(ctx, x, w1, bias, st, sv, dil, kh, kw, pt, pl, HO, WO, act,
                direct):
        ext_fwd = _require_ext("conv_fwd")
        assert sv in (1, 2), "virtual stuff supports stride 1/2 only"
        B, Ci, H, W = x.shape
        Co, K = w1.shape
        bias32 = bias.float().contiguous() if bias is not None else None
        if direct == 2:
            dt = _dummy_tables(x.device)
            y = ext_fwd(x, _wmat(w1, direct=True, khw=25), bias32, dt[0],
                        dt[1], Co, K, HO, WO, act, 2, 2, pt, 0, 0, 0, 1,
                        None, 0, 0, 1, WO)
        elif direct:
            dt = _dummy_tables(x.device)
            y = ext_fwd(x, _wmat(w1, direct=True), bias32, dt[0], dt[1], Co,
                        K, HO, WO, act, 1, 1, pt, 0, 0, 0, 1,
                        None, 0, 0, 1, WO)
        elif sv == 2 and dil == 1:
            # conv-transpose: 4 dense stride-1 phase sub-convs (see
            # _phase_plans) writing strided into one full-grid output.
            # A phase with no matching taps (k=1 dims) is mathematically
            # zero but never written -> zero-init unless all 4 phases run.
            pg = _require_ext("panel_gather")
            plans = _phase_plans(x.device, HO, WO, kh, kw, Ci, pt, pl)
            alloc = torch.empty if len(plans) == 4 else torch.zeros
            y = alloc(B, Co, HO, WO, dtype=torch.bfloat16, device=x.device)
            for p_ in plans:
                wp = _wmat_cache(
                    w1, ("ph", p_["a"], p_["b"], kh, kw),
                    lambda p_=p_: pg(w1.contiguous(), p_["ktab"]))
                ext_fwd(x, wp, bias32, p_["mpack"], p_["kpack"], Co,
                        p_["Kp"], p_["nh"], p_["nw"], act, 1, 0, 0, 1,
                        p_["ptp"], p_["plp"], 1, y, p_["a"], p_["b"], 2, WO)
        else:
            mpack, kpack = _plan_v2(x.device, HO * WO, K, WO, st, dil, kh, kw)
            y = ext_fwd(x, _wmat(w1), bias32, mpack, kpack, Co, K, HO, WO,
                        act, st, 0, 0, 1, pt, pl, sv, None, 0, 0, 1, WO)
        ctx.save_for_backward(x, w1, y if act else None)
        ctx.meta = (st, sv, dil, kh, kw, pt, pl, HO, WO, bias is not None, act)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w1, y_act = ctx.saved_tensors
        st, sv, dil, kh, kw, pt, pl, HO, WO, has_bias, act = ctx.meta
        ext_fwd = _require_ext("conv_fwd")
        ext_wrw = _require_ext("conv_wrw")
        B, Ci, H, W = x.shape
        Co, K = w1.shape

        dy = _act_grad(dy, y_act, act)
        dx = None
        if ctx.needs_input_grad[0]:
            with torch.no_grad():
                ptb = (kh - 1) * dil - pt
                plb = (kw - 1) * dil - pl
                dir_b = (sv == 1 and st == 1 and dil == 1 and kh == 3
                         and kw == 3 and Co % 64 == 0 and ptb == plb
                         and ptb >= 1)
                if dir_b:
                    dt = _dummy_tables(x.device)
                    dx = ext_fwd(dy, _wmat_rot(w1, 9, True), None, dt[0],
                                 dt[1], Ci, Co * 9, H, W, 0, 1, 1, ptb,
                                 0, 0, 0, 1, None, 0, 0, 1, W)
                elif st == 2 and dil == 1:
                    # stride-2 backward-data: phase sub-convs over raw dy
                    # with per-phase column-gathers of the rotated panel
                    pg = _require_ext("panel_gather")
                    rot = _wmat_rot(w1, kh * kw)
                    plans_b = _phase_plans(dy.device, H, W, kh, kw, Co,
                                           ptb, plb)
                    alloc = torch.empty if len(plans_b) == 4 else torch.zeros
                    dx = alloc(B, Ci, H, W, dtype=torch.bfloat16,
                               device=x.device)
                    for p_ in plans_b:
                        wp = _wmat_cache(
                            w1, ("phr", p_["a"], p_["b"], kh, kw),
                            lambda p_=p_: pg(rot, p_["ktab"]))
                        ext_fwd(dy, wp, None, p_["mpack"], p_["kpack"], Ci,
                                p_["Kp"], p_["nh"], p_["nw"], 0, 1, 0, 0, 1,
                                p_["ptp"], p_["plp"], 1, dx, p_["a"],
                                p_["b"], 2, W)
                else:
                    mb, kb = _plan_v2(dy.device, H * W, Co * kh * kw, W, sv,
                                      dil, kh, kw)
                    dx = ext_fwd(dy, _wmat_rot(w1, kh * kw), None, mb, kb,
                                 Ci, Co * kh * kw, H, W, 0, sv, 0, 0,
                                 1, ptb, plb, st, None, 0, 0, 1, W)

        dw1 = None
        if ctx.needs_input_grad[1]:
            if sv == 2 and dil == 1:
                # conv-transpose weight grad by phases: each phase's compact
                # dW lands in its own tap columns (disjoint across phases)
                dw32 = torch.zeros(Co, K, dtype=torch.float32,
                                   device=x.device)
                for p_ in _phase_plans(x.device, HO, WO, kh, kw, Ci, pt, pl):
                    dy_p = dy[:, :, p_["a"]::2, p_["b"]::2].contiguous()
                    dwp = ext_wrw(x, dy_p, p_["mpack"], p_["kpack"], Co,
                                  p_["Kp"], p_["nw"], True, 1, 1,
                                  p_["ptp"], p_["plp"], 1)
                    dw32.index_copy_(1, p_["ktab64"], dwp)
                dw1 = dw32.to(w1.dtype)
            else:
                mpack, kpack = _plan_v2(x.device, HO * WO, K, WO, st, dil,
                                        kh, kw)
                dw1 = ext_wrw(x, dy, mpack, kpack, Co, K, WO, st == 1,
                              1, st, pt, pl, sv).to(w1.dtype)

        dbias = (dy.sum(dim=(0, 2, 3), dtype=torch.float32)
                 if has_bias else None)
        return (dx, dw1, dbias, None, None, None, None, None, None, None,
                None, None, None, None)


class _GatherConvFP8Fn(torch.autograd.Function):
    """fp8 (e4m3) twin of _GatherConvFn that owns the pad/stuff step itself:
    fp8 buffers are integer-typed (raw bytes) and cannot carry autograd, so
    the differentiable boundary is the ORIGINAL float x. pads = (pl, pr, pt,
    pb); stuff = zero-stuffing stride of the input (conv-transpose)."""

    @staticmethod
    def forward(ctx, x, w1, bias, stride, dil, kh, kw, HO, WO, act, pads,
                stuff):
        pad_fn = _require_ext("pad_stuff")
        ext_fwd = _require_ext("conv_fwd")
        pl, pr, pt, pb = pads
        xbuf = pad_fn(x.contiguous(), pt, pb, pl, pr, stuff, True)
        B, Ci, Hp, Wp = xbuf.shape
        Co, K = w1.shape
        mbase, koff = _plan(x.device, Ci, Hp, Wp, kh, kw, stride, dil, HO, WO)
        bias32 = bias.float().contiguous() if bias is not None else None
        y = ext_fwd(xbuf, _wmat(w1, fp8=True), bias32, mbase, koff, Co, K,
                    HO, WO, act, stride, 0, 0, 0, 0, 0, 1, None, 0, 0, 1, WO)
        ctx.save_for_backward(xbuf, w1, y if act else None)
        ctx.meta = (stride, dil, kh, kw, HO, WO, bias is not None, act,
                    pads, stuff, x.shape, x.dtype)
        return y

    @staticmethod
    def backward(ctx, dy):
        xbuf, w1, y_act = ctx.saved_tensors
        (stride, dil, kh, kw, HO, WO, has_bias, act, pads, stuff, xshape,
         xdtype) = ctx.meta
        pad_fn = _require_ext("pad_stuff")
        ext_fwd = _require_ext("conv_fwd")
        ext_wrw = _require_ext("conv_wrw")
        B, Ci, Hp, Wp = xbuf.shape
        Co, K = w1.shape
        pl, pr, pt, pb = pads

        dy = dy.contiguous().to(torch.bfloat16)
        if act == 1:
            dy = torch.where(y_act > 0, dy, torch.zeros((), dtype=dy.dtype,
                                                        device=dy.device))
        elif act == 2:
            dy = torch.where(y_act > 0, dy, dy * 0.2)
        dy8 = pad_fn(dy, 0, 0, 0, 0, 1, True).view(B, Co, HO, WO)

        dx = None
        if ctx.needs_input_grad[0]:
            with torch.no_grad():
                pe_h, pe_w = (kh - 1) * dil, (kw - 1) * dil
                dybuf = pad_fn(dy, pe_h, pe_h, pe_w, pe_w, stride, True)
                wrot = (w1.view(Co, Ci, kh, kw).flip(2, 3).permute(1, 0, 2, 3)
                        .reshape(Ci, Co * kh * kw))
                mb2, ko2 = _plan(dy.device, Co, dybuf.shape[2], dybuf.shape[3],
                                 kh, kw, 1, dil, Hp, Wp)
                dxbuf = ext_fwd(dybuf, _wmat(wrot, fp8=True), None, mb2, ko2,
                                Ci, Co * kh * kw, Hp, Wp, 0, 1, 0, 0,
                                0, 0, 0, 1, None, 0, 0, 1, Wp)
                _, _, H, W = xshape
                dx = dxbuf[:, :, pt:pt + (H - 1) * stuff + 1:stuff,
                           pl:pl + (W - 1) * stuff + 1:stuff].to(xdtype)

        dw1 = None
        if ctx.needs_input_grad[1]:
            mbase, koff = _plan(xbuf.device, Ci, Hp, Wp, kh, kw, stride, dil,
                                HO, WO)
            dw1 = ext_wrw(xbuf, dy8, mbase, koff, Co, K, WO, stride == 1,
                          0, 1, 0, 0, 1).to(w1.dtype)

        dbias = (dy.sum(dim=(0, 2, 3), dtype=torch.float32)
                 if has_bias else None)
        return (dx, dw1, dbias, None, None, None, None, None, None, None,
                None, None)


def _out_size(h: int, k: int, s: int, p: int, d: int) -> int:
    return (h + 2 * p - (k - 1) * d - 1) // s + 1


def conv2d(x: torch.Tensor, w: torch.Tensor, bias: Optional[torch.Tensor] = None,
           stride: int = 1, padding: int = 0, dilation: int = 1,
           act: int = 0) -> torch.Tensor:
    """Drop-in conv (NCHW). GPU: gather-GEMM MFMA kernel; CPU: torch oracle."""
    if not x.is_cuda:
        y = F.conv2d(x, w.to(x.dtype), bias.to(x.dtype) if bias is not None else None,
                     stride=stride, padding=padding, dilation=dilation)
        return _act(y, act)
    B, Ci, H, W = x.shape
    Co, _, kh, kw = w.shape
    HO = _out_size(H, kh, stride, padding, dilation)
    WO = _out_size(W, kw, stride, padding, dilation)
    if _COMPUTE == "fp8":
        return _GatherConvFP8Fn.apply(x, w.reshape(Co, Ci * kh * kw), bias,
                                      stride, dilation, kh, kw, HO, WO, act,
                                      (padding, padding, padding, padding), 1)
    x = x.to(torch.bfloat16).contiguous()
    direct = 0
    if stride == 1 and dilation == 1 and kh == 3 and kw == 3 \
            and Ci % 64 == 0 and padding >= 1:
        direct = 1
    elif stride == 2 and dilation == 1 and kh == 5 and kw == 5 \
            and Ci % 64 == 0 and _DIRECT5:
        direct = 2
    return _GatherConvFn.apply(x, w.reshape(Co, Ci * kh * kw), bias,
                               stride, 1, dilation, kh, kw, padding, padding,
                               HO, WO, act, direct)


def conv_transpose2d(x: torch.Tensor, w: torch.Tensor,
                     bias: Optional[torch.Tensor] = None, stride: int = 2,
                     padding: int = 0, output_padding: int = 0,
                     act: int = 0) -> torch.Tensor:
    """Transposed conv (w: (Ci, Co, kh, kw), the torch layout)."""
    if not x.is_cuda:
        y = F.conv_transpose2d(x, w.to(x.dtype),
                               bias.to(x.dtype) if bias is not None else None,
                               stride=stride, padding=padding,
                               output_padding=output_padding)
        return _act(y, act)
    B, Ci, H, W = x.shape
    _, Co, kh, kw = w.shape
    HO = (H - 1) * stride - 2 * padding + kh + output_padding
    WO = (W - 1) * stride - 2 * padding + kw + output_padding
    pl_h, pl_w = kh - 1 - padding, kw - 1 - padding
    w1 = w.flip(2, 3).permute(1, 0, 2, 3).reshape(Co, Ci * kh * kw)
    if _COMPUTE == "fp8":
        return _GatherConvFP8Fn.apply(
            x, w1, bias, 1, 1, kh, kw, HO, WO, act,
            (pl_w, pl_w + output_padding, pl_h, pl_h + output_padding), stride)
    x = x.to(torch.bfloat16).contiguous()
    # sv = stride: the zero-stuffed input exists only virtually
    return _GatherConvFn.apply(x, w1, bias, 1, stride, 1, kh, kw, pl_h, pl_w,
                               HO, WO, act, False)


def _act(y: torch.Tensor, act: int) -> torch.Tensor:
    if act == 1:
        return torch.relu(y)
    if act == 2:
        return F.leaky_relu(y, 0.2)
    return y


# ---------------------------------------------------------------------------
# 3D conv on the same gather-GEMM kernels (the kernels are dimension-blind:
# geometry lives in the offset tables). Used by the masked causal conv3d of
# the entropy model (models/probclass.py) — input is pre-padded by
# pad_for_probclass, the conv itself is VALID, stride 1.
# ---------------------------------------------------------------------------

_PLANS3D = {}


def _plan3d(device, Ci, Dp, Hp, Wp, kd, kh, kw):
    key = (device.index, Ci, Dp, Hp, Wp, kd, kh, kw)
    p = _PLANS3D.get(key)
    if p is None:
        Do, Ho, Wo = Dp - kd + 1, Hp - kh + 1, Wp - kw + 1
        m = torch.arange(Do * Ho * Wo, device=device, dtype=torch.int32)
        od = torch.div(m, Ho * Wo, rounding_mode="floor")
        rem = m % (Ho * Wo)
        oh = torch.div(rem, Wo, rounding_mode="floor")
        ow = rem % Wo
        mbase = (od * (Hp * Wp) + oh * Wp + ow).contiguous()
        k = torch.arange(Ci * kd * kh * kw, device=device, dtype=torch.int32)
        khw = kd * kh * kw
        ci = torch.div(k, khw, rounding_mode="floor")
        rem = k % khw
        r = torch.div(rem, kh * kw, rounding_mode="floor")
        rem2 = rem % (kh * kw)
        a = torch.div(rem2, kw, rounding_mode="floor")
        b = rem2 % kw
        koff = (ci * (Dp * Hp * Wp) + r * (Hp * Wp) + a * Wp + b).contiguous()
        p = (mbase, koff, Do, Ho, Wo)
        _PLANS3D[key] = p
    return p


class _GatherConv3dFn(torch.autograd.Function):
    """VALID stride-1 3D conv: y[b,co,od,oh,ow] = sum xbuf[b,ci,od+r,oh+a,
    ow+b'] * w1[co,(ci,r,a,b')] + bias, optional fused ReLU."""

    @staticmethod
    def forward(ctx, xbuf, w1, bias, kd, kh, kw, act):
        ext_fwd = _require_ext("conv_fwd")
        B, Ci, Dp, Hp, Wp = xbuf.shape
        Co, K = w1.shape
        mbase, koff, Do, Ho, Wo = _plan3d(xbuf.device, Ci, Dp, Hp, Wp, kd, kh, kw)
        bias32 = bias.float().contiguous() if bias is not None else None
        # kernel sees a 2D problem: M = Do*Ho*Wo pixels, "WO" = Wo rows
        y = ext_fwd(xbuf.view(B, Ci, Dp * Hp, Wp), _wmat(w1), bias32, mbase,
                    koff, Co, K, Do * Ho, Wo, act, 1, 0, 0, 0, 0, 0, 1,
                    None, 0, 0, 1, Wo)
        ctx.save_for_backward(xbuf, w1, y if act else None)
        ctx.meta = (kd, kh, kw, Do, Ho, Wo, bias is not None, act)
        return y.view(B, Co, Do, Ho, Wo)

    @staticmethod
    def backward(ctx, dy):
        xbuf, w1, y_act = ctx.saved_tensors
        kd, kh, kw, Do, Ho, Wo, has_bias, act = ctx.meta
        ext_fwd = _require_ext("conv_fwd")
        ext_wrw = _require_ext("conv_wrw")
        B, Ci, Dp, Hp, Wp = xbuf.shape
        Co, K = w1.shape
        dy = _act_grad(dy.contiguous(), y_act, act).view(B, Co, Do, Ho, Wo)

        dxbuf = None
        if ctx.needs_input_grad[0]:
            with torch.no_grad():
                dybuf = _padded_buf3d(dy, kd - 1, kh - 1, kw - 1)
                mb2, ko2, D2, H2, W2 = _plan3d(dy.device, Co, dybuf.shape[2],
                                               dybuf.shape[3], dybuf.shape[4],
                                               kd, kh, kw)
                assert (D2, H2, W2) == (Dp, Hp, Wp)
                dxbuf = ext_fwd(dybuf.view(B, Co, -1, dybuf.shape[4]),
                                _wmat_rot(w1, kd * kh * kw), None,
                                mb2, ko2, Ci, Co * kd * kh * kw, Dp * Hp, Wp,
                                0, 1, 0, 0, 0, 0, 0, 1,
                                None, 0, 0, 1, Wp).view(B, Ci, Dp, Hp, Wp)

        dw1 = None
        if ctx.needs_input_grad[1]:
            mbase, koff, _, _, _ = _plan3d(xbuf.device, Ci, Dp, Hp, Wp,
                                           kd, kh, kw)
            dw1 = ext_wrw(xbuf.view(B, Ci, Dp * Hp, Wp),
                          dy.view(B, Co, Do * Ho, Wo), mbase, koff, Co, K,
                          Wo, True, 0, 1, 0, 0, 1).to(w1.dtype)

        dbias = (dy.sum(dim=(0, 2, 3, 4), dtype=torch.float32)
                 if has_bias else None)
        return dxbuf, dw1, dbias, None, None, None, None


def _padded_buf3d(x: torch.Tensor, pd: int, ph: int, pw: int) -> torch.Tensor:
    """Symmetric-pad NCDHW (all-sides pd/ph/pw) with 16-element tail slack."""
    B, C, D, H, W = x.shape
    Dp, Hp, Wp = D + 2 * pd, H + 2 * ph, W + 2 * pw
    n = C * Dp * Hp * Wp
    store = x.new_zeros(B * n + 16)
    buf = store[:B * n].view(B, C, Dp, Hp, Wp)
    buf[:, :, pd:pd + D, ph:ph + H, pw:pw + W] = x
    return buf


def conv3d_valid(x: torch.Tensor, w: torch.Tensor,
                 bias: Optional[torch.Tensor] = None, act: int = 0
                 ) -> torch.Tensor:
    """VALID stride-1 3D conv; w: (Co, Ci, kD, kH, kW). GPU: gather-GEMM
    kernels; CPU: torch oracle. Input must already carry any padding (the
    probclass pads with centers[0] itself)."""
    if not x.is_cuda:
        y = F.conv3d(x, w.to(x.dtype),
                     bias.to(x.dtype) if bias is not None else None)
        return _act(y, act)
    Co, Ci, kd, kh, kw = w.shape
    # add the vector-staging slack (fresh buffer; conv3d inputs are small)
    B = x.shape[0]
    n = x[0].numel()
    store = x.new_empty(B * n + 16, dtype=torch.bfloat16)
    store[:B * n].copy_(x.reshape(-1))
    xbuf = store[:B * n].view(x.shape)
    return _GatherConv3dFn.apply(xbuf, w.reshape(Co, Ci * kd * kh * kw), bias,
                                 kd, kh, kw, act)
