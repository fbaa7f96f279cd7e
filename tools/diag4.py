"""Stage-wise GPU vs CPU-oracle comparison of one full-size forward."""
import os, sys, copy, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from dsin_amd import config as cm
from dsin_amd.models import DSIN
from dsin_amd.training import Trainer
from dsin_amd.data import SyntheticStereo

here = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
ae, _ = cm.parse(os.path.join(here, "run_configs", "ae_run_configs"))
pc, _ = cm.parse(os.path.join(here, "run_configs", "pc_run_configs"))
dev = torch.device("cuda:0")
torch.manual_seed(0)
model = DSIN(ae, pc).to(dev)
tr = Trainer(model, ae, pc, 1576, device=dev, autocast_bf16=True,
             use_cuda_graph=False)
gen = SyntheticStereo(320, 960, seed=5, device="cuda:0")
for i in range(30):
    x, y = gen.next_batch()
    loss, bpp = tr.train_step(x, y)
print(f"step30 train: loss {float(loss):.2f} bpp {float(bpp):.4f}", flush=True)

mcpu = DSIN(ae, pc)
mcpu.load_state_dict(copy.deepcopy(model.state_dict()))
mcpu = mcpu.float()

xc, yc = x.cpu().float(), y.cpu().float()
with torch.no_grad():
    mcpu.train()
    ydc = mcpu.create_y_dec(yc)
    zc, xdc = mcpu.autoencode(xc)
    bcc = mcpu.probclass.bitcost(zc.qbar.float(), zc.symbols, mcpu._pad_value())
    print(f"CPU : bpp {float(bcc.sum())/ (320*960):.4f} x_dec mean {float(xdc.mean()):.4f} "
          f"sym hist {torch.bincount(zc.symbols.flatten(), minlength=6).tolist()}", flush=True)

def gpu_fwd(tag, grad):
    import contextlib
    ctx = contextlib.nullcontext() if grad else torch.no_grad()
    with ctx, tr._autocast():
        model.train()
        zg, xdg = model.autoencode(x)
        bcg = model.probclass.bitcost(zg.qbar.detach().float(), zg.symbols,
                                      model._pad_value())
    print(f"GPU {tag}: bpp {float(bcg.sum())/(320*960):.4f} x_dec mean "
          f"{float(xdg.float().mean()):.4f} sym hist "
          f"{torch.bincount(zg.symbols.flatten(), minlength=6).tolist()} "
          f"qbar-diff-vs-cpu {float((zg.qbar.float().cpu()-zc.qbar.float()).abs().mean()):.5f} "
          f"xdec-diff {float((xdg.float().cpu()-xdc).abs().mean()):.4f} "
          f"symdiff {int((zg.symbols.cpu()!=zc.symbols).sum())}", flush=True)

gpu_fwd("grad  ", True)
gpu_fwd("nograd", False)
gpu_fwd("grad2 ", True)
