"""Fixed dataset statistics and shared constants.

KITTI per-channel statistics used by the FIXED normalization mode
(reference ``src/AE.py:240-248`` / ``src/autoencoder_imgcomp.py:160-170``)
and by the SI patch-search normalization (reference ``src/siFinder.py:56-73``,
whose ``variances`` are in fact standard deviations ~= sqrt of the AE vars).
"""

import math

# (R, G, B) channel means / variances over KITTI, pixel range 0..255.
KITTI_MEAN = (93.70454143384742, 98.28243432206516, 94.84678088809876)
KITTI_VAR = (5411.79935676, 5758.60456747, 5890.31451232)
# stds used by siFinder's normalization (named `variances` there).
KITTI_STD_SIFINDER = (73.56493292844912, 75.88547006820752, 76.74838442810665)

NORM_EPS = 1e-10  # added to var before sqrt (reference src/AE.py:228)

KITTI_STD = tuple(math.sqrt(v + NORM_EPS) for v in KITTI_VAR)

LOG2_E = math.log2(math.e)
