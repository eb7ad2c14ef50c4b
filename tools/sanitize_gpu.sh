#!/bin/bash
# Hazard-check pass over the HIP kernel suite (the reference
# leaned on dist.barrier() and shipped no sanitizer; ROCm has no
# compute-sanitizer equivalent, so this uses the debugging levers it does
# have):
#
#  AMD_SERIALIZE_KERNEL=3        every kernel launch is synchronous (wait
#                                before AND after) — surfaces missing-sync /
#                                stream-ordering bugs as immediate failures
#                                at the faulting launch
#  PYTORCH_NO_HIP_MEMORY_CACHING=1  no caching-allocator reuse — reads of
#                                freed-and-reallocated memory (a masked
#                                use-after-free) change answers and fail
#                                the parity assertions
#  AMD_LOG_LEVEL=1               abort-level runtime errors to stderr
#
# Numerics tests re-run under these conditions: any kernel relying on
# accidental stream ordering or stale allocator contents diverges from the
# fp32 references. Run on an MI355X box:
#   bash tools/sanitize_gpu.sh [pytest-args]
set -u
cd "$(dirname "$0")/.."
export AMD_SERIALIZE_KERNEL=3
export PYTORCH_NO_HIP_MEMORY_CACHING=1
export AMD_LOG_LEVEL=1
exec python -m pytest tests/test_ops_gpu.py tests/test_conv_gpu.py -q "$@"
