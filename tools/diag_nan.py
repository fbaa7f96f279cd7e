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
torch.manual_seed(1234)   # main.py seeds differently; NaN was at ~4.3k
model = DSIN(ae, pc).to(dev)
graphs = os.environ.get("G", "0") == "1"
tr = Trainer(model, ae, pc, 1576, device=dev, autocast_bf16=True,
             use_cuda_graph=graphs)
gen = SyntheticStereo(320, 960, seed=1, device="cuda:0")
for i in range(int(sys.argv[1]) if len(sys.argv) > 1 else 6000):
    x, y = gen.next_batch()
    loss, bpp = tr.train_step(x, y)
    if (i + 1) % 250 == 0 or not torch.isfinite(loss):
        with torch.no_grad(), tr._autocast():
            o = model.train_losses(x, y)
        fp = tr.opt_ae.flat_p
        print(f"{i+1:6d} loss {float(o['loss']):10.2f} d {float(o['d_loss']):8.2f} "
              f"pc {float(o['pc_loss']):9.2f} si {float(o['loss_sinet']):8.2f} "
              f"reg {float(o['reg']):7.2f} bpp {float(o['bpp']):.4f} "
              f"|p|max {float(fp.abs().max()):.3e} "
              f"centers {model.encoder.quantizer.centers.detach().cpu().numpy().round(2).tolist()}",
              flush=True)
        if not torch.isfinite(loss) or not torch.isfinite(o['loss']):
            print("NON-FINITE at", i + 1, flush=True)
            break
