#!/usr/bin/env bash
# Sanitizer-tier GPU test pass (SURVEY §5 race/sanitizer hygiene; the
# reference has none — Go's runtime covers it there, HIP needs this).
#
# ROCm on this image has no compute-sanitizer binary, so the tier is built
# from the runtime's own hardening knobs:
#   AMD_SERIALIZE_KERNEL=3  serialize launches + sync after each kernel —
#                           async faults surface AT the faulting kernel
#   AMD_SERIALIZE_COPY=3    same for copies
#   HIP_LAUNCH_BLOCKING=1   host-synchronous launches
#   PYTORCH_NO_HIP_MEMORY_CACHING=1  every alloc/free hits the HIP
#                           allocator — use-after-free / OOB writes into
#                           cached blocks become hard faults
# Run on a GPU box:  bash scripts/sanitize_gpu.sh [pytest-args]
set -euo pipefail
cd "$(dirname "$0")/.."
export AMD_SERIALIZE_KERNEL=3
export AMD_SERIALIZE_COPY=3
export HIP_LAUNCH_BLOCKING=1
export PYTORCH_NO_HIP_MEMORY_CACHING=1
# serialized launches invalidate hipGraph stream capture -> eager decode
export SUB_EAGER_DECODE=1
exec python -m pytest tests -m gpu -q "$@"
