import torch
import pytest

from dsin_amd.ops.reference import quantize_ref
from dsin_amd.models.quantizer import Quantizer


def test_quantize_shapes_and_symbols():
    torch.manual_seed(0)
    x = torch.randn(2, 4, 5, 7)
    centers = torch.tensor([-1.5, -0.5, 0.0, 0.5, 1.0, 2.0])
    qbar, qsoft, qhard, symbols = quantize_ref(x, centers)
    assert qbar.shape == x.shape and symbols.shape == x.shape
    assert symbols.dtype == torch.int64
    # qhard must be the nearest center
    expected = (x.unsqueeze(-1) - centers).abs().argmin(-1)
    assert torch.equal(symbols, expected)
    assert torch.allclose(qhard, centers[symbols])


def test_quantize_soft_formula():
    torch.manual_seed(1)
    x = torch.randn(1, 2, 3, 3).double()
    centers = torch.linspace(-2, 2, 6).double()
    _, qsoft, _, _ = quantize_ref(x, centers)
    d = (x.unsqueeze(-1) - centers) ** 2
    phi = torch.softmax(-d, dim=-1)
    assert torch.allclose(qsoft, (phi * centers).sum(-1))


def test_straight_through_gradient():
    """qbar's gradient wrt x must equal qsoft's gradient (the hard path is
    detached — reference src/autoencoder_imgcomp.py:131-134)."""
    torch.manual_seed(2)
    x = torch.randn(1, 2, 4, 4, dtype=torch.float64, requires_grad=True)
    centers = torch.linspace(-2, 2, 6, dtype=torch.float64, requires_grad=True)
    qbar, qsoft, _, _ = quantize_ref(x, centers)
    g = torch.randn_like(qbar)
    gx_bar, gc_bar = torch.autograd.grad(qbar, (x, centers), g, retain_graph=True)
    gx_soft, gc_soft = torch.autograd.grad(qsoft, (x, centers), g)
    assert torch.allclose(gx_bar, gx_soft)
    assert torch.allclose(gc_bar, gc_soft)


def test_quantizer_module(small_ae_config):
    q = Quantizer(small_ae_config)
    assert q.centers.shape == (6,)
    assert q.centers.min() >= -2 and q.centers.max() <= 2
    x = torch.randn(1, 32, 8, 12)
    qbar, symbols = q(x)
    assert qbar.shape == x.shape
    assert symbols.max() < 6
    reg = q.regularization_loss()
    assert torch.allclose(reg, 0.1 * 0.5 * (q.centers ** 2).sum())


def test_hard_value_between_extreme_inputs():
    centers = torch.tensor([-1.0, 1.0])
    x = torch.tensor([[[[-5.0, 5.0]]]])
    _, _, qhard, sym = quantize_ref(x, centers)
    assert torch.equal(sym[0, 0, 0], torch.tensor([0, 1]))
    assert torch.allclose(qhard[0, 0, 0], torch.tensor([-1.0, 1.0]))
