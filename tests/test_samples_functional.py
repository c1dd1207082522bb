"""Functional (integration) tests for the sample model zoo on the numpy
backend — the reference's znicz/tests/functional pattern: run a real
sample workflow a few epochs with fixed seeds, assert the loss/error
trajectory improves, then resume-from-snapshot style checks are in
test_mnist_functional.py.
"""

import numpy
import pytest

from veles import prng
from veles.backends import NumpyDevice
from veles.config import root
from veles.dummy import DummyLauncher
from veles.loader.base import TRAIN, VALID


@pytest.mark.timeout(600)
def test_mnist_ae_reconstructs():
    prng.seed_all(2)
    from veles.znicz_amd.models.mnist_ae import MnistAEWorkflow
    root.mnist_ae.loader.update({"minibatch_size": 50, "n_train": 100,
                                 "n_valid": 50})
    root.mnist_ae.decision.update({"max_epochs": 4})
    wf = MnistAEWorkflow(DummyLauncher())
    wf.initialize(device=NumpyDevice())
    wf.run()
    assert wf.is_finished
    first, last = wf.decision.mse_history[0], wf.decision.mse_history[-1]
    assert last < 0.8 * first, (first, last)


@pytest.mark.timeout(600)
def test_cifar_short_numpy():
    prng.seed_all(3)
    from veles.znicz_amd.models.cifar import CifarWorkflow
    root.cifar.loader.update({"minibatch_size": 64, "n_train": 128,
                              "n_valid": 64, "on_device": False,
                              "device_dtype": "fp32"})
    root.cifar.decision.update({"max_epochs": 2})
    wf = CifarWorkflow(DummyLauncher())
    wf.initialize(device=NumpyDevice())
    wf.run()
    assert wf.is_finished
    assert wf.loader.epoch_number == 2


@pytest.mark.timeout(600)
def test_wine_mlp():
    """Reference znicz/samples/Wine — tiny 3-class MLP, must hit ~0
    train error within a few epochs."""
    prng.seed_all(4)
    from veles.znicz_amd.models.wine import WineWorkflow
    root.wine.decision.update({"max_epochs": 8})
    wf = WineWorkflow(DummyLauncher())
    wf.initialize(device=NumpyDevice())
    wf.run()
    assert wf.is_finished
    assert wf.decision.epoch_n_err_pt[TRAIN] < 15.0, \
        wf.decision.epoch_n_err_pt


@pytest.mark.timeout(600)
def test_rbm_cd1_improves_reconstruction():
    prng.seed_all(5)
    from veles.znicz_amd.models.rbm import RBMWorkflow
    root.rbm.decision.update({"max_epochs": 3})
    wf = RBMWorkflow(DummyLauncher())
    wf.initialize(device=NumpyDevice())
    wf.run()
    assert wf.is_finished
    hist = wf.decision.mse_history
    assert hist[-1] < hist[0], hist
    assert numpy.isfinite(wf.energy.energy)


@pytest.mark.timeout(600)
def test_kohonen_map_spreads():
    prng.seed_all(6)
    from veles.znicz_amd.models.kohonen_map import KohonenWorkflow
    root.kohonen.decision.update({"max_epochs": 5})
    wf = KohonenWorkflow(DummyLauncher())
    wf.initialize(device=NumpyDevice())
    wf.run()
    assert wf.is_finished
    wf.forward.output.map_read()
    # the 4 clusters must spread over several SOM cells, not collapse
    assert len(numpy.unique(wf.forward.output.mem)) >= 4


@pytest.mark.timeout(600)
def test_approximator_regression():
    prng.seed_all(7)
    from veles.znicz_amd.models.approximator import ApproximatorWorkflow
    root.approximator.decision.update({"max_epochs": 6})
    wf = ApproximatorWorkflow(DummyLauncher())
    wf.initialize(device=NumpyDevice())
    wf.run()
    assert wf.is_finished
    hist = wf.decision.mse_history
    assert hist[-1] < 0.6 * hist[0], hist


@pytest.mark.timeout(600)
def test_lines_learns_orientation():
    prng.seed_all(8)
    from veles.znicz_amd.models.lines import LinesWorkflow
    root.lines.decision.update({"max_epochs": 9})
    wf = LinesWorkflow(DummyLauncher())
    wf.initialize(device=NumpyDevice())
    wf.run()
    assert wf.is_finished
    assert wf.decision.epoch_n_err_pt[TRAIN] < 20.0, \
        wf.decision.epoch_n_err_pt


@pytest.mark.timeout(600)
def test_channels_classifier():
    prng.seed_all(9)
    from veles.znicz_amd.models.channels import ChannelsWorkflow
    root.channels.decision.update({"max_epochs": 6})
    root.channels.loader.update({"n_train": 200, "n_valid": 40})
    wf = ChannelsWorkflow(DummyLauncher())
    wf.initialize(device=NumpyDevice())
    wf.run()
    assert wf.is_finished
    # position-jittered logos are genuinely hard for 6 epochs — assert
    # steady learning, not mastery
    assert wf.decision.epoch_n_err_pt[TRAIN] < 80.0, \
        wf.decision.epoch_n_err_pt


@pytest.mark.timeout(600)
def test_video_ae():
    prng.seed_all(10)
    from veles.znicz_amd.models.video_ae import VideoAEWorkflow
    root.video_ae.decision.update({"max_epochs": 6})
    wf = VideoAEWorkflow(DummyLauncher())
    wf.initialize(device=NumpyDevice())
    wf.run()
    assert wf.is_finished
    hist = wf.decision.mse_history
    assert hist[-1] < 0.85 * hist[0], hist


@pytest.mark.timeout(600)
def test_mnist_simple():
    """Reference znicz/samples/MnistSimple — minimal 1-hidden MLP."""
    prng.seed_all(11)
    from veles.znicz_amd.models.mnist_simple import MnistSimpleWorkflow
    wf = MnistSimpleWorkflow(DummyLauncher())
    wf.initialize(device=NumpyDevice())
    wf.run()
    assert wf.is_finished
    assert wf.decision.epoch_n_err_pt[TRAIN] < 40.0, \
        wf.decision.epoch_n_err_pt


@pytest.mark.timeout(600)
def test_yale_faces():
    """Reference znicz/samples/YaleFaces — illumination-invariant face
    id; the conv net must beat chance (90%) clearly despite the
    per-sample lighting scale."""
    prng.seed_all(12)
    from veles.znicz_amd.models.yale_faces import YaleFacesWorkflow
    from veles.config import root
    root.yale_faces.decision.update({"max_epochs": 8})
    root.yale_faces.loader.update({"n_train": 200, "n_valid": 40})
    wf = YaleFacesWorkflow(DummyLauncher())
    wf.initialize(device=NumpyDevice())
    wf.run()
    assert wf.is_finished
    assert wf.decision.epoch_n_err_pt[TRAIN] < 50.0, \
        wf.decision.epoch_n_err_pt


@pytest.mark.timeout(600)
def test_spam_kohonen():
    """Reference znicz/samples/SpamKohonen — the SOM must separate the
    two word-distribution topics: most neurons end up class-pure."""
    import numpy
    prng.seed_all(13)
    from veles.znicz_amd.models.spam_kohonen import SpamKohonenWorkflow
    wf = SpamKohonenWorkflow(DummyLauncher())
    wf.initialize(device=NumpyDevice())
    wf.run()
    assert wf.is_finished
    # winner purity: map each sample to its BMU, check dominant-class
    # fraction per neuron
    ld, fwd = wf.loader, wf.forward
    data = ld.original_data.mem
    labels = ld.original_labels.mem
    fwd.weights.map_read()
    w = fwd.weights.mem                      # [n_neurons, features]
    d2 = ((data[:, None, :] - w[None, :, :]) ** 2).sum(axis=2)
    bmu = d2.argmin(axis=1)
    pure = total = 0
    for n in set(bmu):
        cls = labels[bmu == n]
        pure += max((cls == 0).sum(), (cls == 1).sum())
        total += len(cls)
    assert pure / total > 0.85, pure / total


@pytest.mark.timeout(600)
def test_imagenet_ae():
    """Reference znicz/samples/ImagenetAE — conv AE on natural-image
    statistics; reconstruction must improve."""
    prng.seed_all(14)
    from veles.znicz_amd.models.imagenet_ae import ImagenetAEWorkflow
    from veles.config import root
    root.imagenet_ae.decision.update({"max_epochs": 4})
    wf = ImagenetAEWorkflow(DummyLauncher())
    wf.initialize(device=NumpyDevice())
    wf.run()
    assert wf.is_finished
    h = wf.decision.mse_history
    assert h[-1] < 0.9 * h[0], h
