from .distortions import Distortions, get_loss, bitcost_to_bpp
from .msssim import multiscale_ssim
from .msssim_np import multiscale_ssim_np

__all__ = ["Distortions", "get_loss", "bitcost_to_bpp", "multiscale_ssim",
           "multiscale_ssim_np"]
