#!/usr/bin/env bash
# Sanitizer-class pass for the gfx950 kernels (SURVEY §5 requirement).
# No ASAN runtime ships for amdgpu in this image, so the pass is:
#  1. canary-guard tests: every kernel's outputs/scratch carved from
#     guarded buffers, guards checked after each launch
#     (tests/test_ops_guard_gpu.py);
#  2. the full GPU suite under AMD_SERIALIZE_KERNEL=3 AMD_SERIALIZE_COPY=3
#     (every launch synchronized: an async fault aborts AT the offending
#     kernel instead of a later sync point).
# Run on the GPU box:  bash tools/sanitize_gpu.sh [logfile]
set -uo pipefail
LOG="${1:-gpurun_out/sanitize_gpu.log}"
mkdir -p "$(dirname "$LOG")"
{
  echo "== canary-guard kernel tests =="
  python -m pytest tests/test_ops_guard_gpu.py -v -q
  G1=$?
  echo "== serialized-kernel full GPU suite =="
  AMD_SERIALIZE_KERNEL=3 AMD_SERIALIZE_COPY=3 \
    python -m pytest tests -m gpu -q -x
  G2=$?
  echo "guard_rc=$G1 serialized_rc=$G2"
  if [ "$G1" -eq 0 ] && [ "$G2" -eq 0 ]; then
    echo "SANITIZE_PASS"
  else
    echo "SANITIZE_FAIL"
  fi
} 2>&1 | tee "$LOG"
