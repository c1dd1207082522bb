"""Activation codes shared by numpy paths and the HIP extension.

The codes must match ``ops/hip/common.h`` (enum Act).

Znicz semantics preserved (reference znicz/all2all.py, znicz/activation.py):
* TANH is the scaled LeCun tanh  y = 1.7159 * tanh(0.6666 * x)
  with derivative  dy/dx = 1.143819 - 0.388484 * y^2  (from y);
* RELU is softplus  y = log(1 + exp(x))  (znicz's historic "RELU"),
  derivative  1 - exp(-y);
* STRICT_RELU is max(0, x), derivative (y > 0);
* SIGMOID is 1/(1+exp(-x)), derivative y*(1-y).
"""

import numpy

ACT_LINEAR = 0
ACT_TANH = 1
ACT_RELU = 2          # softplus (znicz "RELU")
ACT_STRICT_RELU = 3   # max(0, x)
ACT_SIGMOID = 4

TANH_A = 1.7159
TANH_B = 0.6666
TANH_DA = 1.143819
TANH_DB = 0.388484


def apply_forward(code, x):
    """numpy forward activation (in-place-safe: returns new array)."""
    if code == ACT_LINEAR:
        return x
    if code == ACT_TANH:
        return TANH_A * numpy.tanh(TANH_B * x)
    if code == ACT_RELU:
        return numpy.where(x > 15, x, numpy.log1p(numpy.exp(
            numpy.minimum(x, 15))))
    if code == ACT_STRICT_RELU:
        return numpy.maximum(x, 0)
    if code == ACT_SIGMOID:
        return 1.0 / (1.0 + numpy.exp(-x))
    raise ValueError(code)


def derivative_from_output(code, y):
    """numpy d(act)/dx expressed through the *output* y."""
    if code == ACT_LINEAR:
        return numpy.ones_like(y)
    if code == ACT_TANH:
        return TANH_DA - TANH_DB * y * y
    if code == ACT_RELU:
        return 1.0 - numpy.exp(-y)
    if code == ACT_STRICT_RELU:
        return (y > 0).astype(y.dtype)
    if code == ACT_SIGMOID:
        return y * (1.0 - y)
    raise ValueError(code)
