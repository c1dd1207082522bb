"""MnistSimple sample (parity: reference ``znicz/samples/MnistSimple``).

The reference's minimal single-hidden-layer softmax workflow — the
"hello world" config distinct from the tuned MNIST sample: one small
all2all_sigmoid layer + softmax, plain SGD, few epochs.  Synthetic
MNIST-shaped data (no network).
"""

from veles.config import root
from veles.znicz_amd.models.synthetic import SyntheticClassLoader
from veles.znicz_amd.standard_workflow import StandardWorkflow

root.mnist_simple.update({
    "loader": {"minibatch_size": 50, "n_train": 500, "n_valid": 100,
               "sample_shape": (784,), "n_classes": 10},
    "decision": {"max_epochs": 6, "fail_iterations": 30},
    "snapshotter": {"enabled": False},
    "learning_rate": 0.05,
    "hidden": 64,
})


class MnistSimpleWorkflow(StandardWorkflow):
    def __init__(self, workflow=None, **kwargs):
        cfg = root.mnist_simple
        kwargs.setdefault("layers", [
            {"type": "all2all_sigmoid",
             "->": {"output_sample_shape": cfg.hidden},
             "<-": {"learning_rate": cfg.learning_rate}},
            {"type": "softmax", "->": {"output_sample_shape": 10},
             "<-": {"learning_rate": cfg.learning_rate}},
        ])
        kwargs.setdefault("loader_factory",
                          lambda wf, **kw: SyntheticClassLoader(wf, **kw))
        kwargs.setdefault("loader_config", cfg.loader.to_dict())
        kwargs.setdefault("decision_config", cfg.decision.to_dict())
        kwargs.setdefault("snapshotter_config", cfg.snapshotter.to_dict())
        super(MnistSimpleWorkflow, self).__init__(workflow, **kwargs)


def run(load, main):
    load(MnistSimpleWorkflow)
    main()
