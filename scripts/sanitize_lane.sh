#!/bin/bash
# Kernel sanitizer lane (SURVEY §5 — net-new over the reference, TODO r1 #16).
#
# Two passes over the GPU kernel/engine tests:
#   1) serialized: AMD_SERIALIZE_KERNEL=3 AMD_SERIALIZE_COPY=3 makes every
#      kernel launch + copy synchronous, so async faults surface at the
#      guilty call site and stream-ordering races change behavior visibly.
#   2) normal: the usual concurrent schedule.
# Both must pass; the determinism tests inside the suite
# (test_update_is_run_to_run_deterministic, alias bit-exactness) then pin
# bitwise reproducibility under the concurrent schedule.
#
#   gpurun -- 'bash scripts/sanitize_lane.sh > gpurun_out/sanitize.log 2>&1'
set -uo pipefail
cd "$(dirname "$0")/.."
echo "== pass 1: serialized kernels/copies =="
AMD_SERIALIZE_KERNEL=3 AMD_SERIALIZE_COPY=3 HSA_ENABLE_SDMA=0 \
  timeout 1200 python -m pytest tests/test_kernels_gpu.py tests/test_engine_gpu.py -q -x || exit 1
echo "== pass 2: concurrent schedule =="
timeout 1200 python -m pytest tests/test_kernels_gpu.py tests/test_engine_gpu.py -q -x || exit 1
echo "SANITIZE_LANE_OK"
