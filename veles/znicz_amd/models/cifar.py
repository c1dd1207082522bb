"""CIFAR-10 3-conv CNN (BASELINE config 2).

Parity target: reference ``znicz/samples/CIFAR10/cifar.py`` — conv/pool
stack + softmax on 32×32 images.  Synthetic data (RGB padded to 8
channels for the HIP conv path).
"""

from veles.config import root
from veles.znicz_amd.models.synthetic import SyntheticClassLoader
from veles.znicz_amd.standard_workflow import StandardWorkflow

root.cifar.update({
    "loader": {"minibatch_size": 256, "n_train": 2048, "n_valid": 512,
               "sample_shape": (32, 32, 8), "n_classes": 10,
               "noise": 0.5},
    "decision": {"max_epochs": 5, "fail_iterations": 100},
    "snapshotter": {"enabled": False},
    "lr": 0.02,
    "wd": 0.0001,
    "moment": 0.9,
})


def cifar_layers(lr, wd, moment):
    gd_kw = {"learning_rate": lr, "weights_decay": wd,
             "gradient_moment": moment}
    return [
        {"type": "conv_str", "->": {"n_kernels": 32, "kx": 5, "ky": 5,
                                    "padding": (2, 2, 2, 2)},
         "<-": dict(gd_kw)},
        {"type": "max_pooling", "->": {"kx": 2, "ky": 2}},
        {"type": "conv_str", "->": {"n_kernels": 32, "kx": 5, "ky": 5,
                                    "padding": (2, 2, 2, 2)},
         "<-": dict(gd_kw)},
        {"type": "avg_pooling", "->": {"kx": 2, "ky": 2}},
        {"type": "conv_str", "->": {"n_kernels": 64, "kx": 5, "ky": 5,
                                    "padding": (2, 2, 2, 2)},
         "<-": dict(gd_kw)},
        {"type": "avg_pooling", "->": {"kx": 2, "ky": 2}},
        {"type": "softmax", "->": {"output_sample_shape": 10},
         "<-": dict(gd_kw)},
    ]


class CifarWorkflow(StandardWorkflow):
    def __init__(self, workflow=None, **kwargs):
        cfg = root.cifar
        kwargs.setdefault("layers",
                          cifar_layers(cfg.lr, cfg.wd, cfg.moment))
        kwargs.setdefault("loader_factory",
                          lambda wf, **kw: SyntheticClassLoader(wf, **kw))
        kwargs.setdefault("loader_config", cfg.loader.to_dict())
        kwargs.setdefault("decision_config", cfg.decision.to_dict())
        kwargs.setdefault("snapshotter_config", cfg.snapshotter.to_dict())
        super(CifarWorkflow, self).__init__(workflow, **kwargs)


def run(load, main):
    load(CifarWorkflow)
    main()
