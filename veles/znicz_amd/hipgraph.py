"""hipGraph-captured training step.

MI355X-native feature with no reference counterpart (the reference
re-JITed OpenCL per unit; CDNA4's answer to launch-bound inner loops is
graph replay): the device portion of one training step — forwards,
evaluator, GD units, the bucketed reducer's comm-stream work — is
captured once into a hipGraph (``torch.cuda.CUDAGraph`` = hipGraph on
ROCm) and replayed on subsequent steps, collapsing ~40+ kernel-launch
round-trips into one ``hipGraphLaunch``.  Small models (CIFAR-size) are
launch-bound, so this is a straight step-time win; big models lose
nothing.

What stays eager (host work, outside the graph):

* the Loader: index bookkeeping, shuffle, the pinned-staging index
  upload and the device gather (its output buffer feeds the graph at a
  fixed address);
* the Decision: flag logic every step, counter ``map_read`` only at
  epoch-set boundaries (those syncs are legal outside the graph).

Replayability invariants (the units were built for these):

* every Array is allocated once at ``initialize`` — fixed addresses;
* no unit syncs the host inside the step (Decision maps only on
  ``last_minibatch``, which runs after the replay);
* dropout's PRNG step counter lives in device memory and is advanced by
  a leading bump kernel (dropout.py), so each replay draws fresh masks;
* hyper-parameters passed by value into kernels (learning rate, decay)
  are part of :meth:`_signature`; a change drops the graph and the next
  train step re-captures.

The stepper only replays full-size TRAIN minibatches; short or
valid/test minibatches take the eager path (which also honours
``decision.gd_skip``).  Capture failures fall back to eager permanently
— the feature degrades, never breaks.
"""

from veles.loader.base import TRAIN
from veles.memory import Array


class GraphStepper(object):
    """Drives one training iteration with hipGraph replay of the device
    work.  ``workflow`` must expose ``loader``, ``forwards``,
    ``evaluator``, ``decision``, ``gds`` and ``comm_sync`` (the
    StandardWorkflow surface bench.py uses)."""

    def __init__(self, workflow, warmup=2, enabled=True):
        self.wf = workflow
        self.loader = workflow.loader
        self.forwards = list(workflow.forwards)
        self.evaluator = workflow.evaluator
        self.decision = workflow.decision
        self.gds_rev = [g for g in reversed(workflow.gds)]
        self.comm_sync = workflow.comm_sync
        self.warmup = warmup
        self._train_steps = 0
        self._graph = None
        self._sig = None
        self._touched = ()
        self._disabled = not enabled or not self._graphs_supported()
        # pull the loader's device gather INTO the captured region: its
        # index buffer has a fixed address and the pinned upload stays
        # eager, so each replay gathers that step's fresh minibatch
        if hasattr(self.loader, "device_gather"):
            self.loader.defer_device_gather = True

    @staticmethod
    def _graphs_supported():
        try:
            import torch
            return torch.cuda.is_available() and \
                hasattr(torch.cuda, "CUDAGraph")
        except Exception:
            return False

    # -- step ----------------------------------------------------------------
    def step(self):
        ld = self.loader
        ld.run()
        full_train = int(ld.minibatch_class) == TRAIN and \
            int(ld.minibatch_size) == int(ld.max_minibatch_size)
        if self._disabled or not full_train:
            self._eager(check_skip=not full_train)
            return
        self._train_steps += 1
        if self._graph is not None and self._sig != self._signature():
            # hyper-params changed: drop the graph and run ONE eager
            # step first so downstream caches (e.g. the reducer's SGD
            # descriptor blobs) rebuild outside of stream capture
            self._graph = None
            self._eager()
        elif self._graph is not None:
            self._graph.replay()
            self._post_replay()
        elif self._train_steps > self.warmup:
            self._capture()
        else:
            self._eager()
        self.decision.run()

    def _middle(self):
        if hasattr(self.loader, "device_gather_if_pending"):
            self.loader.device_gather_if_pending()
        for f in self.forwards:
            f.run()
        self.evaluator.run()
        for g in self.gds_rev:
            g.run()
        self.comm_sync.run()

    def _eager(self, check_skip=False):
        if hasattr(self.loader, "device_gather_if_pending"):
            self.loader.device_gather_if_pending()
        for f in self.forwards:
            f.run()
        self.evaluator.run()
        if check_skip:
            self.decision.run()
            if bool(self.decision.gd_skip):
                return
        for g in self.gds_rev:
            g.run()
        self.comm_sync.run()
        if check_skip:
            return
        # decision runs in step() for the train path

    # -- capture/replay ------------------------------------------------------
    def _signature(self):
        sig = []
        for g in self.gds_rev:
            sig.append((float(getattr(g, "learning_rate", 0.0)),
                        float(getattr(g, "learning_rate_bias", 0.0)),
                        float(getattr(g, "weights_decay", 0.0)),
                        float(getattr(g, "gradient_moment", 0.0))))
        return tuple(sig)

    def _capture(self):
        import torch
        touched = set()
        Array._capture_log = touched
        try:
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self._middle()
            self._graph = g
            self._sig = self._signature()
            self._touched = tuple(touched)
            # stream capture RECORDS the kernels without executing
            # them — replay once immediately so the capture iteration's
            # minibatch still trains (otherwise its weight update and
            # n_err contribution are silently dropped, repeated on
            # every re-capture after an lr change)
            g.replay()
            self._post_replay()
        except Exception as e:
            # capture is an optimization; any failure (a unit that syncs,
            # an unsupported collective) falls back to eager for good
            import traceback
            self._disabled = True
            self._graph = None
            self.capture_error = traceback.format_exc()
            import logging
            logging.getLogger("GraphStepper").warning(
                "hipGraph capture failed, falling back to eager: %s", e)
            self._middle()
        finally:
            Array._capture_log = None

    def _post_replay(self):
        # replays bypass python run(): restore the side effects the
        # eager path would have had
        for a in self._touched:
            a.touch_device()
        for f in self.forwards:
            if hasattr(f, "counter") and hasattr(f, "_ctr_dev"):
                f.counter += 1            # mirror the device bump kernel
