import torch
import pytest

from dsin_amd.models import Decoder, Encoder
from dsin_amd.ops.reference import heatmap3d_ref, kitti_denormalize, kitti_normalize


def test_encoder_shapes(small_ae_config):
    torch.manual_seed(0)
    enc = Encoder(small_ae_config)
    x = torch.rand(1, 3, 64, 96) * 255
    z = enc(x)
    assert z.qbar.shape == (1, 32, 8, 12)       # 8x subsampling, C=32
    assert z.symbols.shape == (1, 32, 8, 12)
    assert z.heatmap.shape == (1, 32, 8, 12)


def test_decoder_shapes(small_ae_config):
    torch.manual_seed(0)
    dec = Decoder(small_ae_config)
    q = torch.randn(1, 32, 8, 12)
    out = dec(q)
    assert out.shape == (1, 3, 64, 96)
    assert out.min() >= 0 and out.max() <= 255


def test_heatmap3d_values():
    torch.manual_seed(1)
    b = torch.randn(1, 5, 3, 3)  # C = 4
    h = heatmap3d_ref(b)
    assert h.shape == (1, 4, 3, 3)
    assert h.min() >= 0 and h.max() <= 1
    # monotone non-increasing along c
    assert (h[:, :-1] >= h[:, 1:] - 1e-6).all()
    # large positive logit -> all ones
    b2 = torch.full((1, 5, 1, 1), 100.0)
    assert torch.allclose(heatmap3d_ref(b2), torch.ones(1, 4, 1, 1))


def test_normalize_roundtrip():
    torch.manual_seed(2)
    x = torch.rand(2, 3, 8, 8) * 255
    assert torch.allclose(kitti_denormalize(kitti_normalize(x)), x, atol=1e-4)
    # normalized KITTI-mean image is ~0
    mean_img = torch.tensor([93.70454143384742, 98.28243432206516,
                             94.84678088809876]).view(1, 3, 1, 1).expand(1, 3, 4, 4)
    assert kitti_normalize(mean_img).abs().max() < 1e-5


def test_param_count(small_ae_config, pc_config):
    """~10.1 M total params (SURVEY.md section 2b K18 derivation)."""
    from dsin_amd.models import DSIN
    m = DSIN(small_ae_config, pc_config)
    total = sum(p.numel() for p in m.parameters())
    assert 9.5e6 < total < 10.8e6, total
    enc = sum(p.numel() for p in m.encoder.parameters())
    dec = sum(p.numel() for p in m.decoder.parameters())
    pc = sum(p.numel() for p in m.probclass.parameters())
    si = sum(p.numel() for p in m.sinet.parameters())
    assert 4.8e6 < enc < 5.3e6
    assert 4.7e6 < dec < 5.2e6
    assert 20e3 < pc < 28e3
    assert 70e3 < si < 80e3


def test_gradients_reach_encoder_through_rate_mask(small_ae_config, pc_config):
    """PC input is detached, but the heatmap (H_mask) path must carry rate
    gradients into the encoder (reference src/AE.py:74 + loss assembly)."""
    from dsin_amd.models import DSIN
    torch.manual_seed(0)
    m = DSIN(small_ae_config, pc_config)
    x = torch.rand(1, 3, 64, 96) * 255
    z, x_dec = m.autoencode(x)
    bc = m.probclass.bitcost(z.qbar.detach(), z.symbols, m._pad_value())
    H_mask = (bc * z.heatmap).mean()
    H_mask.backward()
    g = m.encoder.to_bn.conv.weight.grad
    assert g is not None and g.abs().sum() > 0
