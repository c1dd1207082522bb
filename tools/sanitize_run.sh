#!/bin/bash
# Race/sanitizer lane: run a short GPU training with
# serialized kernel launches + synchronous HIP error checking, so any
# stream-ordering bug or async kernel fault surfaces at its call site.
#   gpurun -- 'bash tools/sanitize_run.sh'
set -e
export AMD_SERIALIZE_KERNEL=3 AMD_SERIALIZE_COPY=3 HIP_LAUNCH_BLOCKING=1
cd "$(dirname "$0")/.."
timeout 600 python -m pytest tests/test_kernels_gpu.py -q -x
timeout 300 python bench.py --steps 3 --warmup 1 --batch 64
echo "sanitize lane OK"
