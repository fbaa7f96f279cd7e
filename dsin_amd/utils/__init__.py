from .metrics import (loss_list_saver, pearson_per_patch, l1_np, psnr_np,
                      save_test_img, MetricsLogger)
from .profiling import trace_step

__all__ = ["loss_list_saver", "pearson_per_patch", "l1_np", "psnr_np",
           "save_test_img", "MetricsLogger", "trace_step"]
