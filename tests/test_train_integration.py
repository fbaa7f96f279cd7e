"""End-to-end training integration tests on CPU (BASELINE config 1:
AE encoder+quantizer+decoder forward on 64x64 random crops, CPU eager)."""

import torch
import pytest

from dsin_amd.data import SyntheticStereo
from dsin_amd.models import DSIN
from dsin_amd.training import Trainer
from dsin_amd.training.helpers import lr_at_step, num_itr_per_epoch


def test_baseline_config1_forward(small_ae_config, pc_config):
    cfg = small_ae_config.clone(crop_size=(64, 64), y_patch_size=(16, 16))
    torch.manual_seed(0)
    m = DSIN(cfg, pc_config)
    x = torch.rand(1, 3, 64, 64) * 255
    z, x_dec = m.autoencode(x)
    assert x_dec.shape == x.shape
    assert z.qbar.shape == (1, 32, 8, 8)


def test_two_train_steps_finite(small_ae_config, pc_config):
    torch.manual_seed(0)
    m = DSIN(small_ae_config, pc_config)
    tr = Trainer(m, small_ae_config, pc_config, num_training_imgs=1576)
    gen = SyntheticStereo(64, 96)
    losses = []
    for _ in range(2):
        x, y = gen.next_batch()
        loss, bpp = tr.train_step(x, y)
        assert torch.isfinite(loss) and torch.isfinite(bpp)
        losses.append(float(loss))
    assert tr.global_step == 2


def test_ae_only_mode(small_ae_config, pc_config):
    cfg = small_ae_config.clone(AE_only=True, batch_size=2)
    torch.manual_seed(0)
    m = DSIN(cfg, pc_config)
    assert m.sinet is None and m.si_weight == 0.0
    tr = Trainer(m, cfg, pc_config, num_training_imgs=1576)
    x = torch.rand(2, 3, 64, 96) * 255
    loss, bpp = tr.train_step(x, None)
    assert torch.isfinite(loss)


def test_validate_and_reconstruct(small_ae_config, pc_config):
    torch.manual_seed(0)
    m = DSIN(small_ae_config, pc_config)
    gen = SyntheticStereo(64, 96)
    x, y = gen.next_batch()
    v = m.validate_loss(x, y)
    assert torch.isfinite(v)
    y_dec, y_syn, x_dec, x_with_si, bpp = m.reconstruct(x, y)
    assert x_with_si.shape == x.shape and torch.isfinite(bpp)


def test_lr_schedule_staircase(small_ae_config):
    itr_ep = num_itr_per_epoch(1, 1, 1576, False)
    assert itr_ep == 1576
    decay_steps = itr_ep * 20
    assert lr_at_step(small_ae_config, 0, itr_ep) == pytest.approx(1e-4)
    assert lr_at_step(small_ae_config, decay_steps - 1, itr_ep) == pytest.approx(1e-4)
    assert lr_at_step(small_ae_config, decay_steps, itr_ep) == pytest.approx(1e-5)
    assert lr_at_step(small_ae_config, 2 * decay_steps, itr_ep) == pytest.approx(1e-6)


def test_imagenet_epoch_for_ae_only():
    assert num_itr_per_epoch(1, 1, 1576, True) == 1_281_000


def test_two_optimizer_split(small_ae_config, pc_config):
    m = DSIN(small_ae_config, pc_config)
    ae_params, pc_params = m.param_groups()
    n_pc = sum(p.numel() for p in pc_params)
    assert 20e3 < n_pc < 28e3  # probclass only
    ids = {id(p) for p in ae_params} | {id(p) for p in pc_params}
    assert len(ids) == len(ae_params) + len(pc_params)  # disjoint
    assert ids == {id(p) for p in m.parameters()}       # complete


def test_gradient_isolation_sinet_vs_search(small_ae_config, pc_config):
    """siNet loss must backprop into the DECODER through x_dec, but not into
    anything through y_syn (stop_gradient, reference src/AE.py:67)."""
    torch.manual_seed(0)
    m = DSIN(small_ae_config, pc_config)
    gen = SyntheticStereo(64, 96)
    x, y = gen.next_batch()
    out = m.train_losses(x, y)
    loss_sinet = out["loss_sinet"]
    loss_sinet.backward()
    dec_grad = m.decoder.from_bn.conv.weight.grad
    assert dec_grad is not None and dec_grad.abs().sum() > 0


def test_nan_guard_blackbox(small_ae_config, pc_config, tmp_path):
    """Failure detection: a poisoned weight must trip
    the per-step non-finite guard, dump a replayable black box, and point at
    the corrupt buffer in the corruption map."""
    torch.manual_seed(0)
    m = DSIN(small_ae_config, pc_config)
    tr = Trainer(m, small_ae_config, pc_config, num_training_imgs=10,
                 fused_adam=True, nan_guard=True,
                 blackbox_dir=str(tmp_path / "bb"))
    gen = SyntheticStereo(64, 96, seed=3)
    x, y = gen.next_batch()
    tr.train_step(x, y)  # clean step: guard must not trip

    with torch.no_grad():
        m.encoder.h1.conv.weight[0, 0, 0, 0] = float("nan")
    x, y = gen.next_batch()
    with pytest.warns(UserWarning, match="nan_guard tripped"):
        with pytest.raises(RuntimeError, match="non-finite training loss"):
            tr.train_step(x, y)

    import os
    files = os.listdir(tmp_path / "bb")
    assert len(files) == 1 and files[0].startswith("step_")
    box = torch.load(tmp_path / "bb" / files[0], weights_only=False)
    assert box["x"].shape == x.shape
    # the corruption map localizes the poisoned parameter
    stats = box["param_stats"]["encoder.h1.conv.weight"]
    assert stats["nonfinite"] >= 1
    # model + optimizer state present for offline replay
    assert "encoder.h1.conv.weight" in box["model_state"]
    assert len(box["optim_state"]) == 2


def test_replay_blackbox_tool(small_ae_config, pc_config, tmp_path, capsys):
    """tools/replay_blackbox.py localizes the failure from a black box: the
    poisoned encoder weight must show up as the first non-finite forward."""
    torch.manual_seed(0)
    m = DSIN(small_ae_config, pc_config)
    tr = Trainer(m, small_ae_config, pc_config, num_training_imgs=10,
                 fused_adam=True, nan_guard=True,
                 blackbox_dir=str(tmp_path / "bb"))
    gen = SyntheticStereo(64, 96, seed=3)
    with torch.no_grad():
        m.encoder.h1.conv.weight[0, 0, 0, 0] = float("nan")
    x, y = gen.next_batch()
    with pytest.warns(UserWarning), pytest.raises(RuntimeError):
        tr.train_step(x, y)
    import os
    box = os.path.join(tmp_path / "bb", os.listdir(tmp_path / "bb")[0])

    # configs via the sidecar dump round-trip (the tool's real input format)
    aef, pcf = str(tmp_path / "ae"), str(tmp_path / "pc")
    with open(aef, "w") as f:
        f.write(str(small_ae_config))
    with open(pcf, "w") as f:
        f.write(str(pc_config))

    import importlib.util
    spec = importlib.util.spec_from_file_location(
        "replay_blackbox",
        os.path.join(os.path.dirname(__file__), "..", "tools",
                     "replay_blackbox.py"))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    mod.main([box, "-ae_config", aef, "-pc_config", pcf, "--device", "cpu"])
    out = capsys.readouterr().out
    assert "corrupt buffers at dump time" in out
    assert "encoder.h1.conv.weight" in out       # corruption map names it
    assert "first non-finite FORWARD output" in out
    assert "encoder.h1" in out.split("FORWARD output:")[1]
