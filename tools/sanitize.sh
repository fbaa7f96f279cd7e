#!/usr/bin/env bash
# Kernel sanity harness (the reference has no
# sanitizers; this is the rebuild's hook). Run ON A GPU BOX.
#
# Layer 1 — serialized-kernel run: every kernel launch is synchronized and
# checked, so the first faulting kernel is identified precisely (the async
# default can attribute a fault to a later launch):
#   AMD_SERIALIZE_KERNEL=3 AMD_SERIALIZE_COPY=3 python -m pytest tests -m gpu -x -q
#
# Layer 2 — BITWISE determinism: every custom kernel reduces through plain
# partial stores + ordered sums (no cross-block/cross-wave fp32 atomic
# accumulation; the NCC argmax is a packed-u64 max, order-independent with
# index tie-break), so two identical training runs must produce bit-equal
# weights. tests/test_gpu_kernels.py::test_step_bitwise_determinism
# asserts exactly that.
#
# Layer 3 — host ASAN (HIP host code): rebuild the extension with
#   CFLAGS='-fsanitize=address -shared-libasan' python setup.py build_ext --inplace
# and LD_PRELOAD the matching libclang_rt.asan before python. Heavyweight;
# only needed when chasing host-side memory bugs.
set -euo pipefail
cd "$(dirname "$0")/.."
echo "== serialized-kernel GPU test pass =="
AMD_SERIALIZE_KERNEL=3 AMD_SERIALIZE_COPY=3 python -m pytest tests -m gpu -x -q
echo "== bitwise determinism pass =="
python -m pytest tests/test_gpu_kernels.py -q -x -k determinism
echo "sanitize: all passes completed"
