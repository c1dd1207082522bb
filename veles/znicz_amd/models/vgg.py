"""VGG-16-style deep conv stack (BASELINE config 4).

Parity target: the reference has no VGG sample; this follows the same
``StandardWorkflow(layers=[...])`` convention as its AlexNet-class
imagenet workflow (znicz/samples/ imagenet) at VGG-16 depth: 13 conv3×3
(64-64, 128-128, 256×3, 512×3, 512×3) with 5 max-pools, then
FC4096-FC4096-FC1000.  Synthetic data, RGB padded to 8 channels for the
HIP conv path (C%8 == 0 DMA constraint).
"""

from veles.config import root
from veles.znicz_amd.models.synthetic import SyntheticClassLoader
from veles.znicz_amd.standard_workflow import StandardWorkflow

root.vgg16.update({
    "loader": {"minibatch_size": 64, "n_train": 256, "n_valid": 64,
               "sample_shape": (224, 224, 8), "n_classes": 1000,
               "noise": 0.5, "on_device": True, "device_dtype": "bf16"},
    "decision": {"max_epochs": 3, "fail_iterations": 100},
    "snapshotter": {"enabled": False},
    "lr": 0.01,
    "wd": 0.0005,
    "moment": 0.9,
})


def vgg16_layers(lr, wd, moment, n_classes=1000):
    gd = {"learning_rate": lr, "weights_decay": wd,
          "gradient_moment": moment}

    def conv(k):
        return {"type": "conv_str",
                "->": {"n_kernels": k, "kx": 3, "ky": 3,
                       "padding": (1, 1, 1, 1), "sliding": (1, 1),
                       "weights_stddev": 0.01},
                "<-": dict(gd)}

    def pool():
        return {"type": "max_pooling", "->": {"kx": 2, "ky": 2,
                                              "sliding": (2, 2)}}

    layers = []
    for block in ([64, 64], [128, 128], [256, 256, 256],
                  [512, 512, 512], [512, 512, 512]):
        for k in block:
            layers.append(conv(k))
        layers.append(pool())
    layers += [
        {"type": "all2all_relu", "->": {"output_sample_shape": 4096,
                                        "weights_stddev": 0.005},
         "<-": dict(gd)},
        {"type": "dropout", "->": {"dropout_ratio": 0.5}},
        {"type": "all2all_relu", "->": {"output_sample_shape": 4096,
                                        "weights_stddev": 0.005},
         "<-": dict(gd)},
        {"type": "dropout", "->": {"dropout_ratio": 0.5}},
        {"type": "softmax", "->": {"output_sample_shape": n_classes,
                                   "weights_stddev": 0.01},
         "<-": dict(gd)},
    ]
    return layers


class VGG16Workflow(StandardWorkflow):
    def __init__(self, workflow=None, **kwargs):
        cfg = root.vgg16
        kwargs.setdefault("layers", vgg16_layers(
            cfg.lr, cfg.wd, cfg.moment,
            cfg.loader.n_classes if cfg.loader.n_classes else 1000))
        kwargs.setdefault("loader_factory",
                          lambda wf, **kw: SyntheticClassLoader(wf, **kw))
        kwargs.setdefault("loader_config", cfg.loader.to_dict())
        kwargs.setdefault("decision_config", cfg.decision.to_dict())
        kwargs.setdefault("snapshotter_config", cfg.snapshotter.to_dict())
        super(VGG16Workflow, self).__init__(workflow, **kwargs)


def run(load, main):
    load(VGG16Workflow)
    main()
