from .metrics import (MetricsLogger, eval_msssim_bpp, l1_np, loss_list_saver,
                      pearson_per_patch, psnr_np, save_test_img)
from .profiling import trace_step

__all__ = ["loss_list_saver", "pearson_per_patch", "l1_np", "psnr_np",
           "save_test_img", "MetricsLogger", "eval_msssim_bpp", "trace_step"]
