import torch
import pytest

from dsin_amd import ops
from dsin_amd.ops.reference import (assemble_patches, extract_patches,
                                    gaussian_mask_value, ncc_search_ref)


def test_extract_assemble_roundtrip():
    torch.manual_seed(0)
    img = torch.randn(3, 40, 48)
    p = extract_patches(img, 20, 24)
    assert p.shape == (4, 3, 20, 24)
    back = assemble_patches(p, 40, 48)
    assert torch.equal(back, img)


def test_patch_order_row_major():
    img = torch.zeros(1, 4, 6)
    img[0, 0, 3] = 1.0  # patch row 0, col 1 for 2x3 patches
    p = extract_patches(img, 2, 3)
    assert p.shape == (4, 1, 2, 3)
    assert p[1].sum() == 1.0 and p[0].sum() == 0.0


def test_gaussian_mask_properties():
    m = gaussian_mask_value(4, 16, 16, 64, 64, "cpu", torch.float32)
    # P = 16 patches, map (49, 49)
    assert m.shape == (16, 49, 49)
    assert m.max() <= 1.0 + 1e-6 and m.min() > 0
    # each patch's mask peaks near its own center
    p = 5  # row 1, col 1 -> center (24, 24), map coords minus (7, 7)
    idx = m[p].argmax()
    r, c = divmod(int(idx), 49)
    assert abs(r - (24 - 7)) <= 1 and abs(c - (24 - 7)) <= 1


def test_ncc_finds_planted_shift():
    """y is x shifted a few pixels: the search must find each patch at its
    shifted location and reassemble (nearly) x itself."""
    torch.manual_seed(1)
    h, w, ph, pw, shift = 48, 64, 16, 16, 4
    base = torch.rand(3, h, w + shift) * 255
    x = base[:, :, shift:]
    y = base[:, :, :w]
    y_syn, rows, cols = ncc_search_ref(x, y, y, ph, pw, use_mask=True)
    # every patch (gr, gc) should be found at (gr*ph, gc*pw + shift) in y
    gh, gw = h // ph, w // pw
    for p in range(gh * gw):
        gr, gc = divmod(p, gw)
        expect_c = gc * pw + shift
        if expect_c + pw <= w:  # findable only if inside y
            assert int(rows[p]) == gr * ph, (p, int(rows[p]))
            assert int(cols[p]) == expect_c, (p, int(cols[p]))
    # reassembled y_syn equals x on all fully-findable patches
    findable_w = (w - shift) // pw * pw
    assert torch.allclose(y_syn[:, :, :findable_w], x[:, :, :findable_w], atol=1e-3)


def test_ncc_identical_images_identity_match():
    torch.manual_seed(2)
    x = torch.rand(3, 32, 48) * 255
    y_syn, rows, cols = ncc_search_ref(x, x, x, 16, 16, use_mask=True)
    assert torch.allclose(y_syn, x, atol=1e-4)


def test_sifinder_module_batch(small_ae_config):
    from dsin_amd.models.sifinder import SiFinder
    sf = SiFinder(small_ae_config)
    torch.manual_seed(3)
    x = torch.rand(2, 3, 64, 96) * 255
    y = torch.rand(2, 3, 64, 96) * 255
    out = sf(x, y, y)
    assert out.shape == x.shape


def test_l2lab_mode_planted_patch():
    """use_L2andLAB search (reference src/siFinder.py:13-31, 102-103,
    145-195): LAB transform, squared-L2 score, argmin. A patch of y copied
    into x must match itself at its own location (no mask)."""
    torch.manual_seed(5)
    H, W, ph, pw = 40, 48, 8, 8
    y = torch.rand(3, H, W) * 255
    x = torch.rand(3, H, W) * 255
    # plant y's (16, 24) patch at x patch-slot (1, 2)
    x[:, 8:16, 16:24] = y[:, 16:24, 24:32]
    y_syn, rows, cols = ops.ncc_search(x, y, y, ph, pw, use_mask=False,
                                       l2lab=True)
    p = (8 // ph) * (W // pw) + 16 // pw
    assert int(rows[p]) == 16 and int(cols[p]) == 24
    assert torch.allclose(y_syn[:, 8:16, 16:24], y[:, 16:24, 24:32])


def test_l2lab_differs_from_pearson():
    torch.manual_seed(6)
    x = torch.rand(3, 32, 32) * 255
    y = torch.rand(3, 32, 32) * 255
    a, _, _ = ops.ncc_search(x, y, y, 8, 8, use_mask=True, l2lab=False)
    b, _, _ = ops.ncc_search(x, y, y, 8, 8, use_mask=True, l2lab=True)
    assert a.shape == b.shape
