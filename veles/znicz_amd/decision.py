"""Decision units — training control.

Parity target: reference ``znicz/decision.py`` (``DecisionBase``,
``DecisionGD``, ``DecisionMSE``): per-set epoch error accounting,
``max_epochs`` / ``fail_iterations`` early stop, ``complete`` gate,
``improved`` flag gating the snapshotter.

MI355X redesign: the reference's host ``map_read`` of evaluator counters
every minibatch is the workflow's only hot-loop host sync; here counters
stay device-resident and are read once per epoch-set boundary (the
``last_minibatch`` flag), and under data parallelism the read value is
all-reduced across ranks with one tiny collective (the replacement for
the master-side merge in reference §3.3).
"""

from veles.loader.base import TRAIN, VALID, TEST, CLASS_NAME
from veles.mutable import Bool
from veles.units import Unit


class DecisionBase(Unit):
    def __init__(self, workflow, **kwargs):
        super(DecisionBase, self).__init__(workflow, **kwargs)
        self.max_epochs = kwargs.get("max_epochs", None)
        self.fail_iterations = kwargs.get("fail_iterations", 100)
        self.complete = Bool(False)
        self.improved = Bool(False)
        self.train_improved = Bool(False)
        #: True when the current minibatch must not train (valid/test set)
        self.gd_skip = Bool(False)
        self.snapshot_suffix = ""
        # linked from loader
        self.minibatch_class = None
        self.minibatch_size = None
        self.last_minibatch = None
        self.class_lengths = None
        self.epoch_ended = None
        self.epoch_number = None
        self.fails = 0
        self.best_epoch = 0

    def initialize(self, **kwargs):
        super(DecisionBase, self).initialize(**kwargs)

    def run(self):
        self.gd_skip <<= int(self.minibatch_class) != TRAIN
        if not self.last_minibatch:
            return
        clazz = int(self.minibatch_class)
        self.on_last_minibatch(clazz)
        if clazz == TRAIN:
            self.on_epoch_ended()
            if self.max_epochs is not None and \
                    int(self.epoch_number) >= self.max_epochs:
                self.complete <<= True
            if self.fails >= self.fail_iterations:
                self.complete <<= True

    # -- overridables -------------------------------------------------------
    def on_last_minibatch(self, clazz):
        pass

    def on_epoch_ended(self):
        pass

    def _dist_sum(self, value):
        """All-reduce a scalar across DP ranks (identity when world==1)."""
        ctx = getattr(self.workflow, "dist", None)
        if ctx is None:
            return value
        return ctx.all_reduce_scalar(value)

    def _world(self):
        ctx = getattr(self.workflow, "dist", None)
        return ctx.world if ctx is not None else 1


class DecisionGD(DecisionBase):
    """Classification decision driven by evaluator n_err counts."""

    def __init__(self, workflow, **kwargs):
        super(DecisionGD, self).__init__(workflow, **kwargs)
        self.minibatch_n_err = None     # linked: evaluator.n_err (Array)
        self.epoch_n_err = [1.0e30, 1.0e30, 1.0e30]
        self.epoch_n_err_pt = [100.0, 100.0, 100.0]
        self.min_validation_n_err = None
        self.min_validation_n_err_epoch = 0
        self.min_train_n_err = None
        self.confusion_matrixes = None  # linked optionally
        self.evaluator = None           # set by StandardWorkflow

    def on_last_minibatch(self, clazz):
        arr = self.minibatch_n_err
        arr.map_read()
        n_err = self._dist_sum(int(arr.mem[0]))
        # class_lengths are PER-RANK after sharding; the summed error
        # count is global, so the denominator scales by world
        length = max(1, int(self.class_lengths[clazz]) * self._world())
        self.epoch_n_err[clazz] = n_err
        self.epoch_n_err_pt[clazz] = 100.0 * n_err / length
        # reset the device counter for the next set
        arr.map_write()
        arr.mem[...] = 0
        arr.unmap()
        if clazz == VALID:
            self._check_improvement(n_err)
        elif clazz == TRAIN:
            if self.class_lengths[VALID] == 0:
                self._check_improvement(n_err)
            if self.min_train_n_err is None or \
                    n_err < self.min_train_n_err:
                self.min_train_n_err = n_err
                self.train_improved <<= True
            else:
                self.train_improved <<= False

    def _check_improvement(self, n_err):
        if self.min_validation_n_err is None or \
                n_err < self.min_validation_n_err:
            self.min_validation_n_err = n_err
            self.min_validation_n_err_epoch = int(self.epoch_number)
            self.improved <<= True
            self.fails = 0
        else:
            self.improved <<= False
            self.fails += 1

    def on_epoch_ended(self):
        ep = int(self.epoch_number)
        msgs = []
        for c in (TEST, VALID, TRAIN):
            if self.class_lengths[c]:
                msgs.append("%s: %d (%.2f%%)" % (
                    CLASS_NAME[c], int(self.epoch_n_err[c]),
                    self.epoch_n_err_pt[c]))
        self.info("epoch %d: %s", ep, "; ".join(msgs))


class DecisionMSE(DecisionBase):
    """Regression decision driven by evaluator MSE metrics."""

    def __init__(self, workflow, **kwargs):
        super(DecisionMSE, self).__init__(workflow, **kwargs)
        self.minibatch_metrics = None   # linked: evaluator.metrics
        self.epoch_metrics = [None, None, None]
        self.min_validation_mse = None
        self.min_validation_mse_epoch = 0
        self.min_train_mse = None
        self.mse_history = []           # train mse per epoch

    def on_last_minibatch(self, clazz):
        arr = self.minibatch_metrics
        arr.map_read()
        total = self._dist_sum(float(arr.mem[0]))
        length = max(1, int(self.class_lengths[clazz]) * self._world())
        mse = total / length
        self.epoch_metrics[clazz] = (mse, float(arr.mem[1]),
                                     float(arr.mem[2]))
        arr.map_write()
        arr.mem[0] = 0
        arr.mem[1] = 0
        arr.mem[2] = 1.0e30
        arr.unmap()
        if clazz == VALID or (clazz == TRAIN and
                              self.class_lengths[VALID] == 0):
            if self.min_validation_mse is None or \
                    mse < self.min_validation_mse:
                self.min_validation_mse = mse
                self.min_validation_mse_epoch = int(self.epoch_number)
                self.improved <<= True
                self.fails = 0
            else:
                self.improved <<= False
                self.fails += 1
        if clazz == TRAIN:
            self.mse_history.append(mse)
            if self.min_train_mse is None or mse < self.min_train_mse:
                self.min_train_mse = mse

    def on_epoch_ended(self):
        ep = int(self.epoch_number)
        msgs = []
        for c in (TEST, VALID, TRAIN):
            if self.class_lengths[c] and self.epoch_metrics[c]:
                msgs.append("%s rmse: %.6f" % (CLASS_NAME[c],
                                               self.epoch_metrics[c][0]))
        self.info("epoch %d: %s", ep, "; ".join(msgs))


# reference name kept
TrivialDecision = DecisionBase
