// Multi-tensor SGD descriptor — shared between the HIP kernel
// (elementwise.hip) and the host-side blob builder (bindings.cpp).
// One kernel launch applies the fused SGD update (master fp32 +
// optional bf16 working-copy refresh) to every tensor of a gradient
// bucket: the per-layer sgd_k launches were ~5 µs each at CIFAR size,
// a pure launch-latency tail.
#pragma once

struct SgdDesc {
  float* w;                 // fp32 master
  const float* g;           // raw batch-summed fp32 gradient
  float* vel;               // momentum buffer
  unsigned short* wb;       // bf16 working copy (null: none)
  long long n;              // elements
  long long blk0;           // first grid block assigned to this tensor
  float lr, decay, l1, moment, scale;
};

#define SGD_MULTI_SPAN 2048   // elements per block (256 thr x 8)
