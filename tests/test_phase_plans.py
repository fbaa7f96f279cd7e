"""CPU math test of the phase decomposition tables (ops/conv._phase_plans):
reconstruct a conv-transpose via the phase-table semantics in pure torch
and compare against F.conv_transpose2d. Covers the geometry the GPU kernel
consumes (mpack/kpack fields, per-phase pads, the ktab column gather and
the strided output placement), independent of the HIP kernels."""

import pytest
import torch
import torch.nn.functional as F

from dsin_amd.ops.conv import _phase_plans


def _emulate_phase_gather(x, w1, plans, HO, WO, kh, kw):
    """Run each phase as the kernel would: out[a + 2i, b + 2j] =
    sum_k xv[(mh + kdh - ptp), (mw + kdw - plp)] * wp[k] over the phase's
    gathered tap columns."""
    B, Ci, H, W = x.shape
    Co = w1.shape[0]
    out = torch.zeros(B, Co, HO, WO, dtype=x.dtype)
    for p in plans:
        nh, nw = p["nh"], p["nw"]
        mp = p["mpack"].to(torch.int64)
        kp = p["kpack"].to(torch.int64)
        wp = w1[:, p["ktab64"]]                       # (Co, Kp) column gather
        mh = (mp >> 16).view(nh, nw, 1)
        mw = (mp & 0xFFFF).view(nh, nw, 1)
        kci = (kp >> 20).view(1, 1, -1)
        kdh = ((kp >> 10) & 1023).view(1, 1, -1)
        kdw = (kp & 1023).view(1, 1, -1)
        hv = mh + kdh - p["ptp"]
        wv = mw + kdw - p["plp"]
        ok = (hv >= 0) & (hv < H) & (wv >= 0) & (wv < W)
        hc = hv.clamp(0, H - 1)
        wc = wv.clamp(0, W - 1)
        for b in range(B):
            vals = x[b, kci, hc, wc] * ok                    # (nh, nw, Kp)
            ph_out = torch.einsum("ijk,ck->cij", vals, wp)   # (Co, nh, nw)
            out[b, :, p["a"]::2, p["b"]::2] = ph_out
    return out


@pytest.mark.parametrize("k,pad,H,W", [(3, 1, 6, 9), (5, 2, 7, 8),
                                       (4, 1, 5, 6)])
def test_phase_tables_match_conv_transpose(k, pad, H, W):
    torch.manual_seed(0)
    Ci, Co, stride = 3, 4, 2
    x = torch.randn(2, Ci, H, W)
    w = torch.randn(Ci, Co, k, k) * 0.3
    ref = F.conv_transpose2d(x, w, stride=stride, padding=pad,
                             output_padding=stride - 1)
    HO, WO = ref.shape[2], ref.shape[3]
    # the Function's convT mapping: rotated ci<->co-swapped flat weight,
    # virtual pads (k-1-pad)
    w1 = w.flip(2, 3).permute(1, 0, 2, 3).reshape(Co, Ci * k * k)
    plans = _phase_plans(torch.device("cpu"), HO, WO, k, k, Ci,
                         k - 1 - pad, k - 1 - pad)
    out = _emulate_phase_gather(x, w1, plans, HO, WO, k, k)
    torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-4)


# property-based widening: random rectangular kernels / pads / sizes
try:
    from hypothesis import given, settings, strategies as st
    HAVE_HYP = True
except ImportError:  # pragma: no cover
    HAVE_HYP = False


if HAVE_HYP:
    @settings(max_examples=30, deadline=None)
    @given(kh=st.integers(2, 5), kw=st.integers(2, 5),
           H=st.integers(3, 9), W=st.integers(3, 9),
           ci=st.integers(1, 4), co=st.integers(1, 5),
           ph_frac=st.floats(0, 1), pw_frac=st.floats(0, 1),
           seed=st.integers(0, 2**31 - 1))
    def test_phase_tables_property(kh, kw, H, W, ci, co, ph_frac, pw_frac,
                                   seed):
        """Any (kh,kw,pad_h,pad_w,H,W,Ci,Co) conv-transpose stride-2 geometry
        must be exactly reproduced by the phase tables (0 <= pad <= k-1, the
        range where the virtual pads are non-negative — every config shape
        is in it)."""
        pad_h = int(ph_frac * (kh - 1))
        pad_w = int(pw_frac * (kw - 1))
        torch.manual_seed(seed)
        x = torch.randn(1, ci, H, W, dtype=torch.float64)
        w = torch.randn(ci, co, kh, kw, dtype=torch.float64) * 0.3
        ref = F.conv_transpose2d(x, w, stride=2, padding=(pad_h, pad_w),
                                 output_padding=1)
        HO, WO = ref.shape[2], ref.shape[3]
        w1 = w.flip(2, 3).permute(1, 0, 2, 3).reshape(co, ci * kh * kw)
        plans = _phase_plans(torch.device("cpu"), HO, WO, kh, kw, ci,
                             kh - 1 - pad_h, kw - 1 - pad_w)
        out = _emulate_phase_gather(x, w1, plans, HO, WO, kh, kw)
        torch.testing.assert_close(out, ref, rtol=1e-9, atol=1e-9)
