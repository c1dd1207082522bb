#!/usr/bin/env python3
"""Diagnose GraphStepper capture/replay on CIFAR (bench config)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import torch  # noqa: E402

from veles import prng  # noqa: E402
from veles.backends import ROCmDevice  # noqa: E402
from veles.config import root  # noqa: E402
from veles.dummy import DummyLauncher  # noqa: E402


def main():
    batch = int(sys.argv[1]) if len(sys.argv) > 1 else 256
    prng.seed_all(41)
    from veles.znicz_amd.models.cifar import CifarWorkflow
    root.cifar.loader.update({"minibatch_size": batch,
                              "n_train": batch * 8, "n_valid": 0,
                              "on_device": True, "device_dtype": "bf16"})
    root.cifar.decision.update({"max_epochs": 10 ** 9,
                                "fail_iterations": 10 ** 9})
    wf = CifarWorkflow(DummyLauncher())
    wf.initialize(device=ROCmDevice(0))
    from veles.znicz_amd.parallel import GradientReducer
    from bench import _NullCtx
    wf.grad_reducer = GradientReducer(_NullCtx(), ROCmDevice(0))

    from veles.znicz_amd.hipgraph import GraphStepper
    st = GraphStepper(wf, warmup=2)
    for i in range(10):
        st.step()
    torch.cuda.synchronize()
    print("captured:", st._graph is not None, "disabled:", st._disabled)
    if getattr(st, "capture_error", None):
        print("CAPTURE ERROR:\n", st.capture_error)
        return

    # timings: full stepper step vs raw replay vs eager middle
    t0 = time.perf_counter()
    for _ in range(40):
        st.step()
    torch.cuda.synchronize()
    print("stepper step: %.3f ms" % ((time.perf_counter() - t0) / 40 * 1e3))

    g = st._graph
    t0 = time.perf_counter()
    for _ in range(40):
        g.replay()
    torch.cuda.synchronize()
    print("raw replay:   %.3f ms" % ((time.perf_counter() - t0) / 40 * 1e3))

    t0 = time.perf_counter()
    for _ in range(40):
        wf.loader.run()
    torch.cuda.synchronize()
    print("loader only:  %.3f ms" % ((time.perf_counter() - t0) / 40 * 1e3))

    t0 = time.perf_counter()
    for _ in range(40):
        st._middle()
    torch.cuda.synchronize()
    print("eager middle: %.3f ms" % ((time.perf_counter() - t0) / 40 * 1e3))


if __name__ == "__main__":
    main()
