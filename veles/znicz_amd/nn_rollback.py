"""NNRollback — restore best weights / cut lr on divergence.

Parity target: reference ``znicz/nn_rollback.py`` (``NNRollback``): when
``improved`` stays False for ``rollback_to_best`` consecutive epochs,
restore the best snapshot of every registered weights/bias Array and
multiply the GD units' learning rates by ``lr_cut_ratio``.
"""

from veles.units import Unit


class NNRollback(Unit):
    def __init__(self, workflow, **kwargs):
        super(NNRollback, self).__init__(workflow, **kwargs)
        self.improved = None            # linked: decision.improved (Bool)
        self.lr_cut_ratio = kwargs.get("lr_cut_ratio", 0.5)
        self.rollback_to_best = kwargs.get("rollback_to_best", 3)
        self.gds = []
        self._fail_count = 0
        self._best = {}                 # Array -> numpy copy

    def _arrays(self):
        for gd in self.gds:
            for name in ("weights", "bias"):
                arr = getattr(gd, name, None)
                if arr is not None and arr:
                    yield arr

    def run(self):
        if self.improved:
            self._fail_count = 0
            for arr in self._arrays():
                arr.map_read()
                self._best[id(arr)] = (arr, arr.mem.copy())
            return
        self._fail_count += 1
        if self._fail_count < self.rollback_to_best or not self._best:
            return
        self._fail_count = 0
        for arr, saved in self._best.values():
            arr.map_write()
            arr.mem[...] = saved
            # push the restored host copy to the fp32 master on device,
            # THEN rebuild whichever bf16 working copies this array
            # carries (fc: bf16+bf16_t, conv: bf16+bf16_dgrad) — without
            # the unmap the kernels keep computing on the old weights
            arr.unmap()
            if getattr(arr, "bf16", None) is not None:
                from veles.znicz_amd import ops
                ops.refresh_all_copies(arr)
        for gd in self.gds:
            gd.learning_rate *= self.lr_cut_ratio
            gd.learning_rate_bias *= self.lr_cut_ratio
        self.info("rolled back to best weights, lr cut by %.2f",
                  self.lr_cut_ratio)
