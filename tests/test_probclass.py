import torch
import pytest

from dsin_amd.models.probclass import ProbClass, first_mask, other_mask
from dsin_amd.ops.reference import pad_for_probclass_ref


def make_pc(pc_config, L=6):
    torch.manual_seed(0)
    return ProbClass(pc_config, num_centers=L)


def test_mask_structure():
    fm = first_mask(3)[0, 0]
    om = other_mask(3)[0, 0]
    assert fm.shape == (2, 3, 3)
    # earlier depth plane fully visible
    assert fm[0].sum() == 9 and om[0].sum() == 9
    # current plane: rows below center zero
    assert fm[1, 2].sum() == 0 and om[1, 2].sum() == 0
    # center pixel excluded for first mask, included for other
    assert fm[1, 1, 1] == 0 and om[1, 1, 1] == 1
    # right of center excluded in both
    assert fm[1, 1, 2] == 0 and om[1, 1, 2] == 0
    # left of center included in both
    assert fm[1, 1, 0] == 1 and om[1, 1, 0] == 1


def test_context_size(pc_config):
    pc = make_pc(pc_config)
    assert pc.num_layers() == 4
    assert pc.context_size() == 9  # 4*(3-1)+1


def test_bitcost_shape(pc_config):
    pc = make_pc(pc_config)
    q = torch.randn(1, 8, 6, 10)
    symbols = torch.randint(0, 6, (1, 8, 6, 10))
    bc = pc.bitcost(q, symbols, torch.tensor(0.0))
    assert bc.shape == q.shape
    assert (bc >= 0).all()


def test_causality(pc_config):
    """Bit cost at position (c, h, w) must not change when any symbol at a
    causally-later position changes (the defining property of the masked
    conv3d, reference src/probclass_imgcomp.py:150-183)."""
    pc = make_pc(pc_config).double()
    for m in (pc.conv0, pc.res_conv1, pc.res_conv2, pc.conv2):
        m.mask.data = m.mask.double()
    torch.manual_seed(3)
    q = torch.randn(1, 6, 6, 8, dtype=torch.float64)
    symbols = torch.randint(0, 6, (1, 6, 6, 8))
    bc0 = pc.bitcost(q, symbols, torch.tensor(0.0, dtype=torch.float64))

    # flip the value at a "future" location: same (h,w), later channel;
    # same channel later row; same row later column
    pos = (0, 3, 3, 4)  # n, c, h, w
    for later in [(0, 4, 3, 4),    # later channel plane (depth future)
                  (0, 3, 4, 4),    # row below in same plane
                  (0, 3, 3, 5)]:   # right of center in same row/plane
        q2 = q.clone()
        q2[later] += 7.0
        bc1 = pc.bitcost(q2, symbols, torch.tensor(0.0, dtype=torch.float64))
        assert torch.allclose(bc0[pos], bc1[pos]), f"future leak from {later}"

    # and a strictly-past location MUST change it (left neighbor)
    q3 = q.clone()
    q3[0, 3, 3, 3] += 7.0
    bc2 = pc.bitcost(q3, symbols, torch.tensor(0.0, dtype=torch.float64))
    assert not torch.allclose(bc0[pos], bc2[pos])


def test_own_value_not_seen(pc_config):
    """The center symbol itself must not influence its own logits (strict
    causality of the FIRST layer's mask)."""
    pc = make_pc(pc_config).double()
    torch.manual_seed(4)
    q = torch.randn(1, 6, 6, 8, dtype=torch.float64)
    symbols = torch.randint(0, 6, (1, 6, 6, 8))
    pos = (0, 3, 3, 4)
    pad = pc.context_size() // 2
    qp = pad_for_probclass_ref(q, pad, torch.tensor(0.0, dtype=torch.float64))
    logits0 = pc.logits(qp.unsqueeze(1))[(0, slice(None)) + pos[1:]]
    q2 = q.clone()
    q2[pos] += 5.0
    qp2 = pad_for_probclass_ref(q2, pad, torch.tensor(0.0, dtype=torch.float64))
    logits1 = pc.logits(qp2.unsqueeze(1))[(0, slice(None)) + pos[1:]]
    assert torch.allclose(logits0, logits1)


def test_pad_semantics():
    q = torch.arange(24, dtype=torch.float32).view(1, 2, 3, 4)
    out = pad_for_probclass_ref(q, 2, torch.tensor(9.0))
    assert out.shape == (1, 4, 7, 8)
    assert (out[:, :2] == 9).all()          # front channel pad
    assert (out[:, :, :2] == 9).all()       # top pad
    assert torch.equal(out[:, 2:, 2:-2, 2:-2], q)


def test_pc_gradients_flow(pc_config):
    pc = make_pc(pc_config)
    q = torch.randn(1, 6, 6, 8)
    symbols = torch.randint(0, 6, (1, 6, 6, 8))
    bc = pc.bitcost(q, symbols, torch.tensor(0.0))
    bc.mean().backward()
    for name, p in pc.named_parameters():
        assert p.grad is not None, name
        assert torch.isfinite(p.grad).all(), name
