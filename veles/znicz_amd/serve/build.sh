#!/bin/bash
# Build the standalone mzserve inference binary (gfx950).  Links the
# same kernel translation units as the _mzk extension; hipcc
# cross-compiles on CPU-only boxes.  Skips the build when the binary
# is newer than every source.
set -e
cd "$(dirname "$0")"
SRCS="mzserve.cpp ../ops/hip/gemm.hip ../ops/hip/gemm8.hip \
      ../ops/hip/elementwise.hip ../ops/hip/softmax.hip \
      ../ops/hip/pool_lrn.hip"
DEPS="$SRCS ../ops/hip/api.h ../ops/hip/common.h ../ops/hip/loaders.h \
      ../ops/hip/sgd_multi.h"
if [ -x mzserve ]; then
  stale=0
  for f in $DEPS; do
    [ "$f" -nt mzserve ] && stale=1
  done
  [ "$stale" = 0 ] && { echo "mzserve up to date"; exit 0; }
fi
hipcc -O3 -std=c++17 --offload-arch=gfx950 -x hip $SRCS -o mzserve
echo "built mzserve"
