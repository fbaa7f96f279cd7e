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
