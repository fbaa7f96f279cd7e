import os

import torch
import pytest

from dsin_amd.models import DSIN
from dsin_amd.training import Trainer, checkpoint


def make_model(small_ae_config, pc_config, seed):
    torch.manual_seed(seed)
    return DSIN(small_ae_config, pc_config)


def test_model_name_grammar(small_ae_config):
    name = checkpoint.model_name_for(small_ae_config, now="01012026-1200")
    assert name == "target_bpp0.02_sinet_01012026-1200"
    cfg = small_ae_config.clone(AE_only=True)
    assert "_AE_only_" in checkpoint.model_name_for(cfg, now="x")


def test_save_load_roundtrip(small_ae_config, pc_config, tmp_path):
    m1 = make_model(small_ae_config, pc_config, 0)
    tr1 = Trainer(m1, small_ae_config, pc_config, num_training_imgs=10)
    root = str(tmp_path) + os.sep
    checkpoint.save(m1, tr1.optimizers, 42, root, "testmodel", 5, 10, 1.5,
                    small_ae_config, pc_config)
    assert os.path.exists(os.path.join(root, "testmodel", "model.pt"))
    assert os.path.exists(os.path.join(root, "last_saved_testmodel.txt"))
    assert os.path.exists(os.path.join(root, "configs_testmodel.txt"))

    m2 = make_model(small_ae_config, pc_config, 1)
    tr2 = Trainer(m2, small_ae_config, pc_config, num_training_imgs=10)
    # default staged load: encoder/decoder/imgcomp only, no training-step
    cfg = small_ae_config.clone(load_train_step=False, test_model=False,
                                train_model=True)
    step = checkpoint.load(m2, tr2.optimizers, os.path.join(root, "testmodel"), cfg)
    assert step == 0
    assert torch.equal(m2.encoder.h1.conv.weight, m1.encoder.h1.conv.weight)
    assert torch.equal(m2.encoder.quantizer.centers, m1.encoder.quantizer.centers)
    assert torch.equal(m2.probclass.conv0.weight, m1.probclass.conv0.weight)
    # siNetwork NOT restored in this mode (fresh SI training)
    assert not torch.equal(m2.sinet.last.weight, m1.sinet.last.weight)


def test_load_with_train_step(small_ae_config, pc_config, tmp_path):
    m1 = make_model(small_ae_config, pc_config, 0)
    tr1 = Trainer(m1, small_ae_config, pc_config, num_training_imgs=10)
    root = str(tmp_path) + os.sep
    checkpoint.save(m1, tr1.optimizers, 42, root, "m", 5, 10, 1.5)
    m2 = make_model(small_ae_config, pc_config, 1)
    tr2 = Trainer(m2, small_ae_config, pc_config, num_training_imgs=10)
    cfg = small_ae_config.clone(load_train_step=True)
    step = checkpoint.load(m2, tr2.optimizers, os.path.join(root, "m"), cfg)
    assert step == 42
    assert torch.equal(m2.sinet.last.weight, m1.sinet.last.weight)  # si restored


def test_load_for_si_inference(small_ae_config, pc_config, tmp_path):
    m1 = make_model(small_ae_config, pc_config, 0)
    tr1 = Trainer(m1, small_ae_config, pc_config, num_training_imgs=10)
    root = str(tmp_path) + os.sep
    checkpoint.save(m1, tr1.optimizers, 7, root, "m", 5, 10, 1.5)
    m2 = make_model(small_ae_config, pc_config, 1)
    tr2 = Trainer(m2, small_ae_config, pc_config, num_training_imgs=10)
    cfg = small_ae_config.clone(load_train_step=False, test_model=True,
                                train_model=False)
    step = checkpoint.load(m2, tr2.optimizers, os.path.join(root, "m"), cfg)
    assert step == 0
    assert torch.equal(m2.sinet.last.weight, m1.sinet.last.weight)


def test_resume_equivalence(small_ae_config, pc_config, tmp_path):
    """Crash-recovery semantics: train 2 steps, checkpoint with training
    state, resume in a fresh process-equivalent Trainer, train 2 more —
    must equal 4 uninterrupted steps (weights AND optimizer state). Uses
    the fused flat-buffer Adam (the GPU path) via its CPU fallback."""
    from dsin_amd.data import SyntheticStereo
    root = str(tmp_path) + os.sep
    cfg = small_ae_config
    gen = lambda: SyntheticStereo(cfg.crop_size[0], cfg.crop_size[1], seed=11)

    def steps(tr, g, n):
        out = None
        for _ in range(n):
            x, y = g.next_batch()
            out = tr.train_step(x, y)
        return out

    # uninterrupted run
    m1 = make_model(cfg, pc_config, 7)
    tr1 = Trainer(m1, cfg, pc_config, num_training_imgs=10, fused_adam=True)
    g1 = gen()
    steps(tr1, g1, 4)

    # interrupted + resumed run
    m2 = make_model(cfg, pc_config, 7)
    tr2 = Trainer(m2, cfg, pc_config, num_training_imgs=10, fused_adam=True)
    g2 = gen()
    steps(tr2, g2, 2)
    checkpoint.save(m2, tr2.optimizers, tr2.global_step, root, "resume",
                    2, 4, 9.9, cfg, pc_config)
    lcfg = cfg.clone(load_train_step=True)
    m4 = make_model(cfg, pc_config, 8)   # different init; load must fix
    tr4 = Trainer(m4, cfg, pc_config, num_training_imgs=10, fused_adam=True)
    step = checkpoint.load(m4, tr4.optimizers, os.path.join(root, "resume"),
                           lcfg)
    tr4.global_step = step
    g4 = gen()
    for _ in range(2):      # skip the two already-trained batches
        g4.next_batch()
    steps(tr4, g4, 2)

    torch.testing.assert_close(tr4.opt_ae.flat_p, tr1.opt_ae.flat_p,
                               rtol=0, atol=0)
    torch.testing.assert_close(tr4.opt_pc.flat_p, tr1.opt_pc.flat_p,
                               rtol=0, atol=0)
    torch.testing.assert_close(tr4.opt_ae.exp_avg, tr1.opt_ae.exp_avg,
                               rtol=0, atol=0)
