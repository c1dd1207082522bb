"""Benchmark the standalone C++ inference engine against the Python
InferenceEngine on the same snapshot (CIFAR CNN by default).

Usage (on a GPU box):
    python tools/bench_mzserve.py [--batch 256] [--batches 200]

Trains the model a few steps (synthetic data), snapshots, exports the
flat manifest, then times (a) the Python engine's graph-replay predict
loop and (b) mzserve.  Both run the identical kernels; the delta is
the host path (Python pinned-buffer upload vs C++'s resident input).
Results recorded in profiles/perf_r2.md §serving.
"""

import argparse
import json
import os
import subprocess
import sys
import tempfile
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--batch", type=int, default=256)
    p.add_argument("--batches", type=int, default=200)
    p.add_argument("--steps", type=int, default=6)
    args = p.parse_args()

    import numpy
    import torch
    from veles import prng
    prng.seed_all(5)
    from veles.backends import ROCmDevice
    from veles.config import root
    from veles.dummy import DummyLauncher
    import veles.znicz_amd.models.cifar  # noqa: F401 (sets defaults)
    root.cifar.loader.update({"minibatch_size": args.batch,
                              "n_train": 4 * args.batch, "n_valid": 0,
                              "on_device": True, "device_dtype": "bf16"})
    root.cifar.decision.update({"max_epochs": 1,
                                "fail_iterations": 10 ** 9})
    from veles.znicz_amd.models.cifar import CifarWorkflow
    wf = CifarWorkflow(DummyLauncher())
    wf.initialize(device=ROCmDevice(0))
    for _ in range(args.steps):
        wf.loader.run()
        for f in wf.forwards:
            f.run()
        wf.evaluator.run()
        wf.decision.run()
        for g in reversed(wf.gds):
            g.run()
        wf.comm_sync.run()
    torch.cuda.synchronize()

    tmp = tempfile.mkdtemp()
    from veles.snapshotter import SnapshotterToFile
    snap = SnapshotterToFile(wf, prefix="bz", directory=tmp,
                             compression="")
    wf.del_ref(snap)
    path = snap.export()

    from veles.serving import InferenceEngine, export_flat
    eng = InferenceEngine(path, device=ROCmDevice(0))
    flat = os.path.join(tmp, "flat")
    export_flat(eng, flat)

    rng = numpy.random.RandomState(1)
    x = rng.normal(0, 1, (args.batch,) + tuple(
        eng._in.shape[1:])).astype(numpy.float32)
    eng.predict(x)                     # warm + capture
    t0 = time.perf_counter()
    for _ in range(args.batches):
        eng.predict(x)
    t1 = time.perf_counter()
    py_ms = (t1 - t0) / args.batches * 1000
    print(json.dumps({"engine": "python", "batch": args.batch,
                      "batches": args.batches, "ms_per_batch":
                      round(py_ms, 4),
                      "img_s": round(args.batch * 1000 / py_ms, 1)}))

    binpath = os.path.join(REPO, "veles", "znicz_amd", "serve",
                           "mzserve")
    for extra in ([], ["--h2d"]):
        r = subprocess.run(
            [binpath, flat, "--batches", str(args.batches),
             "--warmup", "5"] + extra,
            capture_output=True, text=True, timeout=600)
        sys.stderr.write(r.stderr)
        print(r.stdout.strip())
        assert r.returncode == 0


if __name__ == "__main__":
    main()
