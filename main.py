"""CLI driver: train / validate / test, mirroring the reference entry point
(/root/reference/src/main.py): ``python main.py [-ae_config PATH]
[-pc_config PATH]``; behavior (train loop with shrinking validation
intervals, best-val checkpointing, test loop writing `<i>_<bpp>bpp.png`
images and metric list files) follows src/main.py:21-126.
"""

from __future__ import annotations

import argparse
import datetime
import os

import numpy as np
import torch

from dsin_amd import config as config_mod
from dsin_amd.data import make_dataset
from dsin_amd.models import DSIN
from dsin_amd.parallel import init_distributed, rank
from dsin_amd.training import Trainer, checkpoint
from dsin_amd.utils import (MetricsLogger, eval_msssim_bpp, loss_list_saver,
                            save_test_img)


def get_validate_every(iteration, total_iterations, validate_every, p1, p2):
    """Validation-interval decay: /10 after 50%, /2 more after 75%
    (reference src/main.py:129-138)."""
    if iteration > total_iterations // 2 and not p1:
        validate_every //= 10
        p1 = True
    if iteration > 3 * (total_iterations // 4) and not p2:
        validate_every //= 2
        p2 = True
    return max(validate_every, 1), p1, p2


def main(argv=None):
    cur = os.getcwd()
    ap = argparse.ArgumentParser()
    ap.add_argument("-ae_config", "--ae_config_path", type=str,
                    default=os.path.join(cur, "run_configs", "ae_run_configs"))
    ap.add_argument("-pc_config", "--pc_config_path", type=str,
                    default=os.path.join(cur, "run_configs", "pc_run_configs"))
    ap.add_argument("--device", type=str, default=None)
    ap.add_argument("--bf16", action="store_true", help="bf16 autocast compute")
    ap.add_argument("--graphs", action="store_true",
                    help="hipGraph whole-step capture (same engine path as "
                         "bench.py; validation runs eager between replays)")
    ap.add_argument("--metrics", type=str, default=None, help="JSONL metrics path")
    ap.add_argument("--checkpoint-every", type=int, default=0,
                    help="periodic crash-recovery checkpoint interval "
                         "(iterations; 0 = best-val only, the reference's "
                         "behavior)")
    ap.add_argument("--nan-guard", action="store_true",
                    help="per-step non-finite loss check; on failure dumps "
                         "batch + model/optimizer state + corruption map to "
                         "nan_blackbox/ and aborts (also: DSIN_NANCHECK=1)")
    ap.add_argument("--plots", action="store_true",
                    help="save loss / inference figures (headless "
                         "equivalents of the reference's plt.show windows, "
                         "src/utils.py:12-79) under the images root")
    args = ap.parse_args(argv)

    ae_config, _ = config_mod.parse(args.ae_config_path)
    pc_config, _ = config_mod.parse(args.pc_config_path)

    local_rank = init_distributed()
    device = torch.device(args.device if args.device else
                          (f"cuda:{local_rank}" if torch.cuda.is_available() else "cpu"))

    data_dir = os.path.join(cur, "data_paths") + os.sep
    root_weights = os.path.join(cur, "weights") + os.sep
    root_save_img = os.path.join(cur, "images") + os.sep
    os.makedirs(root_weights, exist_ok=True)

    model = DSIN(ae_config, pc_config).to(device)
    data = make_dataset(ae_config, data_dir, seed=rank())
    num_train = len(data.train_pairs)
    trainer = Trainer(model, ae_config, pc_config, num_train, device=device,
                     autocast_bf16=args.bf16, use_cuda_graph=args.graphs,
                     nan_guard=args.nan_guard or None)
    metrics = MetricsLogger(args.metrics if rank() == 0 else None)

    model_name = "NA"
    best_val, val_loss = np.inf, np.inf
    now = datetime.datetime.today().strftime("%d%m%Y-%H%M")
    total_iterations = int(ae_config.iterations)
    validate_every = int(ae_config.validate_every)
    p1 = p2 = False

    if ae_config.load_model:
        model_name = ae_config.load_model_name
        step = checkpoint.load(model, trainer.optimizers,
                               root_weights + model_name, ae_config)
        trainer.global_step = step
        print(f"Loaded {root_weights + model_name} (global step {step})")

    if ae_config.train_model:
        val_names, _ = data.get_data_size()
        val_iterations = max(len(val_names) // data.batch_size, 1)
        train_sum = bpp_sum = 0.0
        show_every = int(ae_config.show_every)
        train_hist, train_hist_iters = [], []
        val_hist, val_hist_iters = [], []
        best_iter = 0
        for iteration in range(1, total_iterations + 1):
            x, y = data.get_data_for_train()
            x, y = x.to(device), y.to(device)
            loss, bpp = trainer.train_step(x, y)
            train_sum += float(loss)
            bpp_sum += float(bpp)

            if ae_config.decrease_val_steps:
                validate_every, p1, p2 = get_validate_every(
                    iteration, total_iterations, validate_every, p1, p2)

            if iteration % validate_every == 0:
                val_sum = 0.0
                for _ in range(val_iterations):
                    xv, yv = data.get_data_for_val()
                    xv, yv = xv.to(device), yv.to(device)
                    val_sum += float(trainer.validate(xv, yv))
                val_loss = val_sum / val_iterations
                # RD operating point: MS-SSIM (numpy oracle) + bpp on the
                # last val batch — the quality half of the headline metric
                # (BASELINE: "imgs/sec ...; MS-SSIM @ 0.02 bpp")
                val_ms, val_bpp = eval_msssim_bpp(model, xv, yv)
                val_hist.append(val_loss)
                val_hist_iters.append(iteration)
                metrics.log("val", iteration=iteration, val_loss=val_loss,
                            val_msssim=val_ms, val_bpp=val_bpp)
                if val_loss < best_val and rank() == 0:
                    best_val = val_loss
                    best_iter = iteration
                    if ae_config.save_model:
                        model_name = checkpoint.model_name_for(ae_config, now)
                        checkpoint.save(model, trainer.optimizers, trainer.global_step,
                                        root_weights, model_name, iteration,
                                        total_iterations, best_val, ae_config, pc_config)
                        print(f"Saved {root_weights + model_name}")

            if (args.checkpoint_every and rank() == 0
                    and iteration % args.checkpoint_every == 0):
                checkpoint.save(model, trainer.optimizers, trainer.global_step,
                                root_weights, "periodic", iteration,
                                total_iterations, float(val_loss),
                                ae_config, pc_config)

            if iteration % show_every == 0:
                print(f"[{iteration}/{total_iterations}] loss {train_sum/show_every:.4f} "
                      f"bpp {bpp_sum/show_every:.4f} val {val_loss:.4f}")
                metrics.log("train", iteration=iteration,
                            loss=train_sum / show_every, bpp=bpp_sum / show_every)
                train_hist.append(train_sum / show_every)
                train_hist_iters.append(iteration)
                train_sum = bpp_sum = 0.0

        if args.plots and rank() == 0 and train_hist:
            from dsin_amd.utils.plots import plot_loss
            path = plot_loss(train_hist, val_hist, val_hist_iters,
                             train_hist_iters, total_iterations, best_val,
                             best_iter, str(model_name),
                             out_path=os.path.join(root_save_img,
                                                   f"loss_{model_name}.png"))
            print(f"Saved loss figure {path}")

    if ae_config.test_model and rank() == 0:
        _, test_names = data.get_data_size()
        for i in range(len(test_names)):
            x, y = data.get_data_for_test()
            x, y = x.to(device), y.to(device)
            y_dec, y_syn, x_dec, x_with_si, bpp = model.reconstruct(x, y)
            x_rec = (x_with_si if x_with_si is not None and
                     float(x_with_si.abs().mean()) > 0 else x_dec)
            x_rec_np = x_rec.clamp(0, 255).cpu().numpy()
            save_test_img(root_save_img, str(model_name), x_rec_np[0], i, float(bpp))
            loss_list_saver(x.cpu().numpy(), y.cpu().numpy(), x_rec_np,
                            y_syn.cpu().numpy() if y_syn is not None else None,
                            str(model_name), float(bpp), root_save_img,
                            *(int(v) for v in ae_config.y_patch_size))
            if args.plots and i == 0 and y_syn is not None:
                from dsin_amd.utils.plots import plot_inference
                path = plot_inference(
                    x[0].cpu().numpy(), x_dec[0].float().cpu().numpy(),
                    y[0].cpu().numpy(), y_syn[0].float().cpu().numpy(),
                    x_with_si[0].float().cpu().numpy(), str(model_name),
                    total_iterations, cnt=trainer.global_step,
                    bpp=f"{float(bpp):.4f}",
                    out_path=os.path.join(root_save_img,
                                          f"inference_{model_name}.png"))
                print(f"Saved inference figure {path}")
            print(f"test image {i}: bpp {float(bpp):.4f}")

    metrics.close()


if __name__ == "__main__":
    main()
