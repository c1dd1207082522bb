// Host-safe shared types for the _mzk kernel set (no device code) —
// included by both bindings.cpp (g++) and the .hip kernels (hipcc).
#pragma once

enum Act { ACT_LINEAR = 0, ACT_TANH = 1, ACT_RELU = 2,
           ACT_STRICT_RELU = 3, ACT_SIGMOID = 4 };

struct ConvDesc {
  int N, H, W, C;      // input (NHWC)
  int K, KH, KW;       // kernels
  int OH, OW;          // output spatial
  int SY, SX;          // stride
  int PT, PL;          // top/left padding
};

enum PoolMode { POOL_MAX = 0, POOL_AVG = 1, POOL_MAXABS = 2,
                POOL_STOCHASTIC = 3 };

struct PoolDesc {
  int N, H, W, C;
  int KY, KX, SY, SX;
  int OH, OW;
  unsigned long long seed;   // for stochastic pooling
};
