"""MNIST MLP sample (parity: reference ``znicz/samples/MNIST/mnist.py``).

The BASELINE.json config-1 model: 784-100-10 MLP (all2all_tanh + softmax)
trained end-to-end on the numpy backend — the plumbing benchmark — and on
the HIP backend on a GPU box.  Data is synthetic (no network); shape and
class count match MNIST.
"""

from veles.config import root
from veles.znicz_amd.models.synthetic import SyntheticClassLoader
from veles.znicz_amd.standard_workflow import StandardWorkflow

root.mnist.update({
    "loader": {"minibatch_size": 60, "n_train": 600, "n_valid": 100,
               "sample_shape": (784,), "n_classes": 10},
    "decision": {"max_epochs": 10, "fail_iterations": 50},
    "snapshotter": {"enabled": False},
    "learning_rate": 0.03,
    "weights_decay": 0.0,
    "layers": [100, 10],
})


class MnistWorkflow(StandardWorkflow):
    def __init__(self, workflow=None, **kwargs):
        cfg = root.mnist
        layers = kwargs.pop("layers", None) or [
            {"type": "all2all_tanh",
             "->": {"output_sample_shape": cfg.layers[0]},
             "<-": {"learning_rate": cfg.learning_rate,
                    "weights_decay": cfg.weights_decay}},
            {"type": "softmax",
             "->": {"output_sample_shape": cfg.layers[1]},
             "<-": {"learning_rate": cfg.learning_rate,
                    "weights_decay": cfg.weights_decay}},
        ]
        kwargs.setdefault("layers", layers)
        # real-file ingestion: point root.mnist.loader.idx_files at the
        # idx-ubyte files (train_images/train_labels[/validation_*],
        # .gz accepted) and the sample trains on them; otherwise it
        # runs on the synthetic stand-in (BASELINE: no network).
        loader_cfg = cfg.loader.to_dict()
        idx_files = loader_cfg.pop("idx_files", None)
        if idx_files:
            from veles.loader.binary import MnistIdxLoader
            for k in ("sample_shape", "n_train", "n_valid", "n_classes"):
                loader_cfg.pop(k, None)
            loader_cfg.update(idx_files)
            kwargs.setdefault(
                "loader_factory",
                lambda wf, **kw: MnistIdxLoader(wf, **kw))
        else:
            kwargs.setdefault(
                "loader_factory",
                lambda wf, **kw: SyntheticClassLoader(wf, **kw))
        kwargs.setdefault("loader_config", loader_cfg)
        kwargs.setdefault("decision_config", cfg.decision.to_dict())
        kwargs.setdefault("snapshotter_config", cfg.snapshotter.to_dict())
        kwargs.setdefault("loss_function", "softmax")
        super(MnistWorkflow, self).__init__(workflow, **kwargs)


def run(load, main):
    load(MnistWorkflow)
    main()
