"""Learning-rate schedules.

Parity target: reference ``znicz/lr_adjust.py`` (``LearningRateAdjust``
with Caffe-like step / exp / inv policies mutating the bound GD units'
learning rates each iteration).
"""

from veles.units import Unit


def step_policy(base_lr, gamma, step):
    def f(it):
        return base_lr * (gamma ** (it // step))
    return f


def exp_policy(base_lr, gamma):
    def f(it):
        return base_lr * (gamma ** it)
    return f


def inv_policy(base_lr, gamma, power):
    def f(it):
        return base_lr * ((1.0 + gamma * it) ** -power)
    return f


def fixed_policy(base_lr):
    def f(it):
        return base_lr
    return f


def arbitrary_step_policy(schedule):
    """schedule = [(lr, n_iters), ...] — lr for the next n_iters."""
    bounds = []
    acc = 0
    for lr, n in schedule:
        acc += n
        bounds.append((acc, lr))

    def f(it):
        for b, lr in bounds:
            if it < b:
                return lr
        return bounds[-1][1]
    return f


class LearningRateAdjust(Unit):
    """Applies lr_policy/bias_lr_policy to all bound GD units each run."""

    def __init__(self, workflow, **kwargs):
        super(LearningRateAdjust, self).__init__(workflow, **kwargs)
        self.lr_policy = kwargs.get("lr_policy", None)
        self.bias_lr_policy = kwargs.get("bias_lr_policy",
                                         self.lr_policy)
        self.gds = []
        self.iteration = 0

    def initialize(self, **kwargs):
        super(LearningRateAdjust, self).initialize(**kwargs)

    def run(self):
        if self.lr_policy is not None:
            lr = self.lr_policy(self.iteration)
            for gd in self.gds:
                gd.learning_rate = lr
        if self.bias_lr_policy is not None:
            blr = self.bias_lr_policy(self.iteration)
            for gd in self.gds:
                gd.learning_rate_bias = blr
        self.iteration += 1
