"""Functional: MNIST-shape MLP end-to-end on numpy (BASELINE config 1).

Parity: reference znicz/tests/functional/test_mnist.py — run the sample
workflow a few epochs with fixed seeds and assert the error trajectory,
then snapshot/resume and assert training continues.
"""

import os
import pickle

import numpy
import pytest

from veles import prng
from veles.dummy import DummyLauncher
from veles.loader.base import VALID, TRAIN


def build_workflow(max_epochs=4):
    prng.get().seed(1234)
    prng.get("loader").seed(5678)
    from veles.config import root
    root.mnist.decision.max_epochs = max_epochs
    from veles.znicz_amd.models.mnist import MnistWorkflow
    wf = MnistWorkflow(DummyLauncher())
    return wf


def test_mnist_mlp_trains():
    wf = build_workflow(max_epochs=4)
    wf.initialize(device=None)
    wf.run()
    assert wf.is_finished
    d = wf.decision
    assert int(wf.loader.epoch_number) >= 4
    # synthetic prototype data: a correct stack reaches near-zero error
    assert d.epoch_n_err_pt[TRAIN] < 10.0, \
        "train error did not drop: %s" % d.epoch_n_err_pt
    assert d.min_validation_n_err is not None
    assert d.min_validation_n_err <= 10


def test_mnist_snapshot_resume(tmp_path):
    wf = build_workflow(max_epochs=2)
    wf.initialize(device=None)
    wf.run()
    err_before = wf.decision.epoch_n_err_pt[TRAIN]
    # snapshot (the same path NNSnapshotter uses)
    path = os.path.join(str(tmp_path), "wf.pickle")
    with open(path, "wb") as f:
        pickle.dump(wf, f, protocol=4)
    with open(path, "rb") as f:
        wf2 = pickle.load(f)
    # resume: re-initialize and continue for 2 more epochs
    wf2.decision.max_epochs = 4
    wf2.decision.complete <<= False
    wf2.initialize(device=None)
    wf2.run()
    assert wf2.is_finished
    assert int(wf2.loader.epoch_number) >= 4
    assert wf2.decision.epoch_n_err_pt[TRAIN] <= err_before + 5.0


def test_weights_change_only_on_train():
    wf = build_workflow(max_epochs=1)
    wf.initialize(device=None)
    w0 = wf.forwards[0].weights.mem.copy()
    # run exactly one VALID minibatch through the graph
    ld = wf.loader
    ld.run()
    assert ld.minibatch_class == VALID
    wf.forwards[0].run()
    wf.forwards[1].run()
    wf.evaluator.run()
    wf.decision.run()
    assert bool(wf.decision.gd_skip)
    numpy.testing.assert_array_equal(w0, wf.forwards[0].weights.mem)
