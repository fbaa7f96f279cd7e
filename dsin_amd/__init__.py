"""dsin_amd — MI355X-native learned image compression with decoder side
information (DSIN capabilities, rebuilt for CDNA4/gfx950).

Stack: PyTorch-ROCm autograd/optimizers + hand-written HIP kernels for the
custom ops (soft quantizer, streaming NCC side-information search, masked
causal conv3d entropy model, fused conv epilogues) + RCCL over xGMI for data
parallelism. See SURVEY.md for the blueprint and /root/reference for the
behavioral reference (TF 1.11; no code is ported from it).
"""

__version__ = "0.1.0"

from .config import Config, parse, parse_string  # noqa: F401
