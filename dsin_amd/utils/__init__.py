from .metrics import (loss_list_saver, pearson_per_patch, l1_np, psnr_np,
                      save_test_img, MetricsLogger)

__all__ = ["loss_list_saver", "pearson_per_patch", "l1_np", "psnr_np",
           "save_test_img", "MetricsLogger"]
