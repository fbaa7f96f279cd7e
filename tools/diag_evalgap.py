"""Diagnose the eval-mode rate divergence: track val loss/bpp with
eval-mode BN (running stats) vs train-mode BN (batch stats) during a
longer training run. If only the eval-mode curve diverges, the gap is the
batch-1 BN train/eval statistics mismatch (shared with the reference's
fused-BN-at-batch-1 design), not a kernel bug."""
import os, sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from dsin_amd import config as cm
from dsin_amd.models import DSIN
from dsin_amd.training import Trainer
from dsin_amd.data import SyntheticStereo

here = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
ae, _ = cm.parse(os.path.join(here, "run_configs", "ae_run_configs"))
pc, _ = cm.parse(os.path.join(here, "run_configs", "pc_run_configs"))
dev = torch.device("cuda:0")
steps = int(sys.argv[1]) if len(sys.argv) > 1 else 8000

torch.manual_seed(0)
model = DSIN(ae, pc).to(dev)
tr = Trainer(model, ae, pc, 1576, device=dev, autocast_bf16=True,
             use_cuda_graph=True)
gen = SyntheticStereo(320, 960, seed=5, device="cuda:0")
vgen = SyntheticStereo(320, 960, seed=6, device="cuda:0")
vx, vy = vgen.next_batch()

for i in range(steps):
    x, y = gen.next_batch()
    loss, bpp = tr.train_step(x, y)
    if (i + 1) % 500 == 0:
        with torch.no_grad(), torch.autocast(device_type="cuda",
                                             dtype=torch.bfloat16):
            model.eval()
            ev = model.compute_losses(vx, vy, model.create_y_dec(vy))
            model.train()
            tv = model.compute_losses(vx, vy, model.create_y_dec(vy))
        print(f"step {i+1:6d} train_loss {float(loss):9.2f} "
              f"train_bpp {float(bpp):.4f} | eval-BN val "
              f"{float(ev['loss']):9.2f} bpp {float(ev['bpp']):.4f} | "
              f"batch-BN val {float(tv['loss']):9.2f} bpp "
              f"{float(tv['bpp']):.4f}", flush=True)
