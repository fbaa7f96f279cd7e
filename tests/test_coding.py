"""Range coder + entropy-model codec tests (CPU)."""

import numpy as np
import torch
import pytest

from dsin_amd.coding import (PredictionNetwork, ProbclassTesting,
                             decode_symbols, decode_with_freqs,
                             encode_symbols, encode_with_freqs)
from dsin_amd.models.probclass import ProbClass


def test_range_coder_roundtrip_uniform():
    rng = np.random.default_rng(0)
    n = 2000
    syms = rng.integers(0, 6, n)
    freqs = np.ones(6, dtype=np.int64) * 100
    data = encode_with_freqs(syms, (freqs for _ in range(n)))
    out = decode_with_freqs(data, n, lambda i, prev: freqs)
    assert list(syms) == out


def test_range_coder_roundtrip_skewed():
    rng = np.random.default_rng(1)
    n = 3000
    freqs = np.array([60000, 3000, 1500, 800, 400, 300], dtype=np.int64)
    p = freqs / freqs.sum()
    syms = rng.choice(6, size=n, p=p)
    data = encode_with_freqs(syms, (freqs for _ in range(n)))
    out = decode_with_freqs(data, n, lambda i, prev: freqs)
    assert list(syms) == out
    # compression close to entropy
    ent_bits = -np.sum(np.log2(p[syms]))
    assert len(data) * 8 < ent_bits * 1.02 + 64


def _make_pc(pc_config):
    torch.manual_seed(0)
    return ProbClass(pc_config, num_centers=6)


def test_probclass_testing_matches_bitcost(pc_config):
    pc = _make_pc(pc_config)
    centers = torch.linspace(-2, 2, 6)
    symbols = torch.randint(0, 6, (4, 6, 8))
    pt = ProbclassTesting(pc, centers)
    total = pt.total_bit_cost(symbols)
    q = centers[symbols.unsqueeze(0)]
    bc = pc.bitcost(q, symbols.unsqueeze(0), centers[0])
    assert abs(total - float(bc.detach().sum())) < 1e-3


def test_prediction_network_shapes(pc_config):
    pc = _make_pc(pc_config)
    centers = torch.linspace(-2, 2, 6)
    pred = PredictionNetwork(pc, centers)
    assert pred.context_shape == (5, 9, 9)
    ctx = torch.randint(0, 6, pred.context_shape)
    f = pred.freqs(ctx)
    assert f.shape == (6,) and (f >= 1).all()


def test_codec_roundtrip(pc_config):
    """Bit-exact wavefront encode/decode through the entropy model."""
    pc = _make_pc(pc_config)
    centers = torch.linspace(-2, 2, 6)
    torch.manual_seed(2)
    symbols = torch.randint(0, 6, (3, 5, 6))
    data = encode_symbols(pc, centers, symbols)
    out = decode_symbols(pc, centers, data, (3, 5, 6))
    assert torch.equal(out.cpu(), symbols)


def test_codec_roundtrip_zero_padding(pc_config):
    """use_centers_for_padding=False: encoder and decoder must agree on the
    0.0 pad *value* (the round-1 bug paired a 0.0-value pad on encode with a
    centers[0] symbol pad on decode)."""
    cfg = pc_config.clone()
    cfg.use_centers_for_padding = False
    pc = _make_pc(cfg)
    centers = torch.linspace(-2, 2, 6)
    torch.manual_seed(5)
    symbols = torch.randint(0, 6, (3, 5, 6))
    data = encode_symbols(pc, centers, symbols)  # exact=True default
    out = decode_symbols(pc, centers, data, (3, 5, 6))
    assert torch.equal(out.cpu(), symbols)


def test_wave_order_causality(pc_config):
    """Every causal dependency of a position must land in an earlier wave."""
    from dsin_amd.coding.entropy import _wave_order
    waves = _wave_order(4, 7, 9, pad=4)
    wave_of = {}
    for i, wv in enumerate(waves):
        for c, h, w in wv:
            wave_of[(c, h, w)] = i
    assert len(wave_of) == 4 * 7 * 9
    for (c, h, w), i in wave_of.items():
        # raster-causal in-plane deps within the +-4 window
        for dw in range(-4, 0):
            if (c, h, w + dw) in wave_of:
                assert wave_of[(c, h, w + dw)] < i
        for dh in range(-4, 0):
            for dw in range(-4, 5):
                if (c, h + dh, w + dw) in wave_of:
                    assert wave_of[(c, h + dh, w + dw)] < i
        # full previous-plane window
        for dh in range(-4, 5):
            for dw in range(-4, 5):
                if (c - 1, h + dh, w + dw) in wave_of:
                    assert wave_of[(c - 1, h + dh, w + dw)] < i


def test_codec_bits_close_to_bitcost(pc_config):
    pc = _make_pc(pc_config)
    centers = torch.linspace(-2, 2, 6)
    torch.manual_seed(3)
    symbols = torch.randint(0, 6, (4, 8, 10))
    data = encode_symbols(pc, centers, symbols)
    est_bits = ProbclassTesting(pc, centers).total_bit_cost(symbols)
    actual_bits = len(data) * 8
    # coder overhead: freq quantization + 4 flush bytes
    assert actual_bits < est_bits * 1.05 + 64, (actual_bits, est_bits)


try:
    from hypothesis import given, settings, strategies as st
    HAVE_HYP = True
except ImportError:  # pragma: no cover
    HAVE_HYP = False


if HAVE_HYP:
    @settings(max_examples=25, deadline=None)
    @given(n=st.integers(1, 400), L=st.integers(2, 12),
           seed=st.integers(0, 2**31 - 1), fmax=st.integers(1, 1 << 14))
    def test_range_coder_property_adaptive(n, L, seed, fmax):
        """Roundtrip with per-position ADAPTIVE frequency tables (the codec's
        real operating mode: freqs depend on already-decoded symbols), random
        alphabet sizes and aggressively skewed counts incl. freq=1 tails."""
        rng = np.random.default_rng(seed)
        tables = rng.integers(1, fmax + 1, size=(n, L)).astype(np.int64)

        def freqs_at(i, prev):
            t = tables[i].copy()
            if i > 0:            # context-dependence: rotate by prev symbol
                t = np.roll(t, int(prev))
            return t

        syms = []
        for i in range(n):
            f = freqs_at(i, syms[-1] if syms else 0)
            p = f / f.sum()
            syms.append(int(rng.choice(L, p=p)))

        def gen():
            for i in range(n):
                yield freqs_at(i, syms[i - 1] if i else 0)

        data = encode_with_freqs(syms, gen())
        out = decode_with_freqs(
            data, n, lambda i, dec: freqs_at(i, dec[-1] if dec else 0))
        assert syms == out
