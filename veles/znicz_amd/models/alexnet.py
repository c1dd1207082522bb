"""AlexNet-class ImageNet workflow — the BASELINE.json headline model.

Parity target: the reference's AlexNet-class imagenet sample
(znicz/samples/ imagenet workflow): conv/LRN/pool stack + 2×(FC 4096 +
dropout) + softmax 1000, trained with the StandardWorkflow graph.

MI355X specifics: NHWC activations with the RGB input padded 3→8
channels (C=4 packed staging exists and is tested, but measured ~3×
lower GEMM efficiency on conv1 — profiles/perf_r1_session2.md — so the
2.67× FLOP saving nets a loss; C%8 keeps every 16-B DMA chunk
pixel-aligned), strict-ReLU activations, bf16
compute / fp32 accumulate.  Synthetic 227×227 data (BASELINE: no
network, random-init weights).
"""

from veles.config import root
from veles.znicz_amd.models.synthetic import SyntheticClassLoader
from veles.znicz_amd.standard_workflow import StandardWorkflow

root.alexnet.update({
    "loader": {"minibatch_size": 256, "n_train": 1024, "n_valid": 0,
               "sample_shape": (227, 227, 8), "n_classes": 1000,
               "on_device": True, "device_dtype": "bf16"},
    "decision": {"max_epochs": 1, "fail_iterations": 1000000},
    "snapshotter": {"enabled": False},
    "lr": 0.01,
    "wd": 0.0005,
    "moment": 0.9,
})


def alexnet_layers(lr=0.01, wd=0.0005, moment=0.9, n_classes=1000):
    gd_kw = {"learning_rate": lr, "weights_decay": wd,
             "gradient_moment": moment}
    lrn = {"alpha": 0.00002, "beta": 0.75, "n": 5, "k": 1.0}
    return [
        {"type": "conv_str", "->": {"n_kernels": 96, "kx": 11, "ky": 11,
                                    "sliding": (4, 4),
                                    "weights_filling": "gaussian",
                                    "weights_stddev": 0.01},
         "<-": dict(gd_kw)},
        {"type": "norm", "->": dict(lrn), "<-": dict(lrn)},
        {"type": "max_pooling", "->": {"kx": 3, "ky": 3,
                                       "sliding": (2, 2)}},
        {"type": "conv_str", "->": {"n_kernels": 256, "kx": 5, "ky": 5,
                                    "padding": (2, 2, 2, 2),
                                    "weights_filling": "gaussian",
                                    "weights_stddev": 0.01},
         "<-": dict(gd_kw)},
        {"type": "norm", "->": dict(lrn), "<-": dict(lrn)},
        {"type": "max_pooling", "->": {"kx": 3, "ky": 3,
                                       "sliding": (2, 2)}},
        {"type": "conv_str", "->": {"n_kernels": 384, "kx": 3, "ky": 3,
                                    "padding": (1, 1, 1, 1),
                                    "weights_filling": "gaussian",
                                    "weights_stddev": 0.01},
         "<-": dict(gd_kw)},
        {"type": "conv_str", "->": {"n_kernels": 384, "kx": 3, "ky": 3,
                                    "padding": (1, 1, 1, 1),
                                    "weights_filling": "gaussian",
                                    "weights_stddev": 0.01},
         "<-": dict(gd_kw)},
        {"type": "conv_str", "->": {"n_kernels": 256, "kx": 3, "ky": 3,
                                    "padding": (1, 1, 1, 1),
                                    "weights_filling": "gaussian",
                                    "weights_stddev": 0.01},
         "<-": dict(gd_kw)},
        {"type": "max_pooling", "->": {"kx": 3, "ky": 3,
                                       "sliding": (2, 2)}},
        {"type": "all2all_str", "->": {"output_sample_shape": 4096,
                                       "weights_filling": "gaussian",
                                       "weights_stddev": 0.005},
         "<-": dict(gd_kw)},
        {"type": "dropout", "->": {"dropout_ratio": 0.5}},
        {"type": "all2all_str", "->": {"output_sample_shape": 4096,
                                       "weights_filling": "gaussian",
                                       "weights_stddev": 0.005},
         "<-": dict(gd_kw)},
        {"type": "dropout", "->": {"dropout_ratio": 0.5}},
        {"type": "softmax", "->": {"output_sample_shape": n_classes,
                                   "weights_filling": "gaussian",
                                   "weights_stddev": 0.01},
         "<-": dict(gd_kw)},
    ]


class AlexNetWorkflow(StandardWorkflow):
    def __init__(self, workflow=None, **kwargs):
        cfg = root.alexnet
        kwargs.setdefault("layers", alexnet_layers(
            cfg.lr, cfg.wd, cfg.moment,
            cfg.loader.get("n_classes", 1000)))
        kwargs.setdefault("loader_factory",
                          lambda wf, **kw: SyntheticClassLoader(wf, **kw))
        kwargs.setdefault("loader_config", cfg.loader.to_dict())
        kwargs.setdefault("decision_config", cfg.decision.to_dict())
        kwargs.setdefault("snapshotter_config", cfg.snapshotter.to_dict())
        super(AlexNetWorkflow, self).__init__(workflow, **kwargs)


def run(load, main):
    load(AlexNetWorkflow)
    main()
