"""Render the rate-distortion training evidence from a metrics JSONL
(main.py --metrics): training loss/bpp plus validation bpp & MS-SSIM.

    python tools/plot_rd.py gpurun_out/rd2_metrics.jsonl profiles/rd_curve.png
"""
import json
import sys

import matplotlib
matplotlib.use("Agg")
import matplotlib.pyplot as plt


def main(src, out):
    tr_it, tr_loss, tr_bpp = [], [], []
    v_it, v_bpp, v_ms = [], [], []
    for line in open(src):
        d = json.loads(line)
        if d["event"] == "train":
            tr_it.append(d["iteration"])
            tr_loss.append(d["loss"])
            tr_bpp.append(d["bpp"])
        elif d["event"] == "val":
            v_it.append(d["iteration"])
            v_bpp.append(d["val_bpp"])
            v_ms.append(d["val_msssim"])
    fig, axes = plt.subplots(1, 3, figsize=(16, 4.5))
    axes[0].plot(tr_it, tr_loss, ".-")
    axes[0].set_title("train loss")
    axes[0].set_xlabel("iteration")
    axes[1].plot(tr_it, tr_bpp, ".-", label="train")
    axes[1].plot(v_it, v_bpp, "o-", label="val (eval-mode)")
    axes[1].axhline(0.02, color="r", ls="--", label="0.02 bpp target")
    axes[1].set_title("bpp")
    axes[1].set_xlabel("iteration")
    axes[1].legend()
    axes[2].plot(v_it, v_ms, "o-")
    axes[2].set_title("val MS-SSIM (numpy oracle)")
    axes[2].set_xlabel("iteration")
    fig.tight_layout()
    fig.savefig(out, dpi=110)
    print("wrote", out)


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2] if len(sys.argv) > 2 else "rd_curve.png")
