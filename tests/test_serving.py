"""Serving path: snapshot -> InferenceEngine -> predictions (and the
HTTP JSON front end) on the numpy backend; the GPU graph-replay lane
lives in test_serving_gpu.py."""

import json
import threading
import urllib.request

import numpy
import pytest

from veles import prng
from veles.backends import NumpyDevice
from veles.config import root
from veles.dummy import DummyLauncher


def _train_and_snapshot(tmp_path):
    prng.seed_all(99)
    import veles.znicz_amd.models.wine  # noqa: F401
    root.wine.decision.update({"max_epochs": 6, "fail_iterations": 50})
    root.wine.loader.update({"n_train": 90, "n_valid": 30,
                             "minibatch_size": 30})
    from veles.znicz_amd.models.wine import WineWorkflow
    wf = WineWorkflow(DummyLauncher())
    wf.initialize(device=NumpyDevice())
    wf.run()
    from veles.snapshotter import SnapshotterToFile
    snap = SnapshotterToFile(wf, prefix="srv", directory=str(tmp_path),
                             compression="gz")
    wf.del_ref(snap)
    return snap.export(), wf


def test_inference_engine_matches_workflow(tmp_path):
    path, wf = _train_and_snapshot(tmp_path)
    from veles.serving import InferenceEngine
    eng = InferenceEngine(path, device=NumpyDevice())
    rng = numpy.random.RandomState(3)
    x = rng.normal(0, 1, (7,) + tuple(
        eng._in.shape[1:])).astype(numpy.float32)
    probs = eng.predict(x)
    assert probs.shape[0] == 7
    # cross-check against the trained workflow's own forward chain
    wf.loader.minibatch_data.map_invalidate()
    wf.loader.minibatch_data.mem[:7] = x
    wf.loader.minibatch_data.mem[7:] = 0
    for f in wf.forwards:
        f.run()
    wf.forwards[-1].output.map_read()
    want = wf.forwards[-1].output.mem[:7]
    numpy.testing.assert_allclose(probs.reshape(7, -1),
                                  want.reshape(7, -1), rtol=1e-5,
                                  atol=1e-6)
    # chunking: n > batch_size goes through multiple passes
    xn = rng.normal(0, 1, (75,) + tuple(
        eng._in.shape[1:])).astype(numpy.float32)
    out = eng.predict(xn)
    assert out.shape[0] == 75
    labels = eng.predict_labels(xn)
    assert labels.shape == (75,) and labels.max() < out.shape[-1]


def test_export_flat_manifest(tmp_path):
    # flat export for the standalone C++ engine: manifest + raw blob
    import os
    path, _ = _train_and_snapshot(tmp_path)
    from veles.serving import InferenceEngine, export_flat
    eng = InferenceEngine(path, device=NumpyDevice())
    outdir = str(tmp_path / "flat")
    export_flat(eng, outdir)
    with open(os.path.join(outdir, "manifest.txt")) as f:
        lines = f.read().splitlines()
    assert lines[0] == "mzserve 1" and lines[-1] == "end"
    ops = [l.split() for l in lines if l.startswith("op ")]
    # wine MLP: fc(tanh) + fc(f32) + softmax
    assert [o[1] for o in ops] == ["fc", "fc", "softmax"]
    tanh_fc = ops[0]
    assert tanh_fc[4] == "1" and tanh_fc[-1] == "bf16"  # ACT_TANH
    soft_fc = ops[1]
    assert soft_fc[4] == "0" and soft_fc[-1] == "f32"
    # tensor table: offsets 16-B aligned, blob covers every tensor
    tens = [l.split() for l in lines if l.startswith("tensor ")]
    assert len(tens) == 4          # 2 weights + 2 biases
    blob = open(os.path.join(outdir, "weights.bin"), "rb").read()
    for _, name, dtype, numel, off in tens:
        assert int(off) % 16 == 0
        size = int(numel) * (2 if dtype == "bf16" else 4)
        assert int(off) + size <= len(blob)
    # fc weight [N,K] bf16 must round-trip the fp32 master (RNE)
    eng.forwards[0].weights.map_read()
    w = eng.forwards[0].weights.mem
    _, name, dtype, numel, off = tens[0]
    assert int(numel) == w.size
    got = numpy.frombuffer(
        blob[int(off):int(off) + 2 * w.size], numpy.uint16)
    import torch
    want = torch.from_numpy(numpy.ascontiguousarray(w)).to(
        torch.bfloat16).view(torch.int16).numpy().astype(
        numpy.uint16).ravel()
    numpy.testing.assert_array_equal(got, want)
    # the standalone binary (if built) parses the manifest
    binpath = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "veles", "znicz_amd", "serve",
        "mzserve")
    if os.path.exists(binpath):
        import json as js
        import subprocess
        out = subprocess.run([binpath, outdir, "--parse-only"],
                             capture_output=True, text=True, timeout=60)
        assert out.returncode == 0, out.stderr
        info = js.loads(out.stdout)
        assert info["ops"] == 3 and info["tensors"] == 4


def test_export_flat_edge_cases(tmp_path):
    import os
    import types
    from veles.dummy import DummyWorkflow
    from veles.memory import Array
    from veles.serving import export_flat
    from veles.znicz_amd.all2all import All2AllTanh

    # weights_transposed FC: export must emit the [N,K] forward image
    wf = DummyWorkflow()
    u = All2AllTanh(wf, output_sample_shape=(5,),
                    weights_transposed=True)
    u.input = Array(numpy.zeros((4, 7), numpy.float32))
    u.initialize(device=NumpyDevice())
    eng = types.SimpleNamespace(_in=u.input, batch_size=4,
                                forwards=[u])
    d1 = str(tmp_path / "t")
    export_flat(eng, d1)
    blob = open(os.path.join(d1, "weights.bin"), "rb").read()
    lines = open(os.path.join(d1, "manifest.txt")).read().splitlines()
    trow = next(l.split() for l in lines if l.startswith("tensor w"))
    import torch
    got = numpy.frombuffer(
        blob[int(trow[4]):int(trow[4]) + 2 * int(trow[3])],
        numpy.uint16)
    u.weights.map_read()
    want = torch.from_numpy(numpy.ascontiguousarray(
        u.weights.mem.T)).to(torch.bfloat16).view(
        torch.int16).numpy().astype(numpy.uint16).ravel()
    numpy.testing.assert_array_equal(got, want)
    oprow = next(l.split() for l in lines if l.startswith("op fc"))
    assert oprow[5:8] == ["4", "5", "7"]     # M N K

    # unsupported unit type -> clear NotImplementedError
    class Alien(object):
        output = None
    eng2 = types.SimpleNamespace(_in=u.input, batch_size=4,
                                 forwards=[Alien()])
    with pytest.raises(NotImplementedError, match="Alien"):
        export_flat(eng2, str(tmp_path / "x"))

    # stochastic pooling -> refused
    from veles.znicz_amd.pooling import StochasticPooling
    sp = StochasticPooling(wf, kx=2, ky=2)
    sp.input = Array(numpy.zeros((2, 4, 4, 4), numpy.float32))
    sp.initialize(device=NumpyDevice())
    eng3 = types.SimpleNamespace(_in=sp.input, batch_size=2,
                                 forwards=[sp])
    with pytest.raises(NotImplementedError, match="stochastic"):
        export_flat(eng3, str(tmp_path / "y"))

    # conv with C % 4 != 0 -> refused with guidance
    from veles.znicz_amd.conv import ConvStrictRELU
    cv = ConvStrictRELU(wf, n_kernels=8, kx=3, ky=3,
                        padding=(1, 1, 1, 1))
    cv.input = Array(numpy.zeros((2, 6, 6, 3), numpy.float32))
    cv.initialize(device=NumpyDevice())
    eng4 = types.SimpleNamespace(_in=cv.input, batch_size=2,
                                 forwards=[cv])
    with pytest.raises(NotImplementedError, match="C%4"):
        export_flat(eng4, str(tmp_path / "z"))


def test_mzserve_rejects_malformed_manifests(tmp_path):
    import os
    import subprocess
    binpath = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "veles", "znicz_amd", "serve",
        "mzserve")
    if not os.path.exists(binpath):
        pytest.skip("mzserve binary not built")

    def run(d):
        return subprocess.run([binpath, str(d), "--parse-only"],
                              capture_output=True, text=True,
                              timeout=60)
    # missing manifest
    r = run(tmp_path / "nope")
    assert r.returncode != 0 and "manifest" in r.stderr
    # bad header
    d1 = tmp_path / "bad1"
    d1.mkdir()
    (d1 / "manifest.txt").write_text("something else\n")
    r = run(d1)
    assert r.returncode != 0 and "header" in r.stderr
    # unknown op
    d2 = tmp_path / "bad2"
    d2.mkdir()
    (d2 / "manifest.txt").write_text(
        "mzserve 1\nbatch 4\nin_shape 4 8\nop warp\nend\n")
    r = run(d2)
    assert r.returncode != 0 and "unknown op" in r.stderr
    # structurally empty (no ops)
    d3 = tmp_path / "bad3"
    d3.mkdir()
    (d3 / "manifest.txt").write_text("mzserve 1\nbatch 4\nend\n")
    r = run(d3)
    assert r.returncode != 0 and "incomplete" in r.stderr
    # chain starting with an in-place op (would mutate the resident
    # input across replays)
    d4 = tmp_path / "bad4"
    d4.mkdir()
    (d4 / "manifest.txt").write_text(
        "mzserve 1\nbatch 4\nin_shape 4 8\nop act 1\nend\n")
    r = run(d4)
    assert r.returncode != 0 and "in-place" in r.stderr
    # tensor extending past weights.bin (caught before any hip call,
    # so this lane runs on CPU boxes too)
    d5 = tmp_path / "bad5"
    d5.mkdir()
    (d5 / "manifest.txt").write_text(
        "mzserve 1\nbatch 4\nin_shape 4 8\n"
        "tensor w0 bf16 100 0\n"
        "op fc w0 - 0 4 10 8 bf16\nend\n")
    (d5 / "weights.bin").write_bytes(b"\0" * 64)   # 100 bf16 > 64 B
    r = subprocess.run([binpath, str(d5)], capture_output=True,
                       text=True, timeout=60)
    assert r.returncode != 0 and "exceeds" in r.stderr


def test_serving_http(tmp_path):
    path, _ = _train_and_snapshot(tmp_path)
    from veles.serving import InferenceEngine, serve
    eng = InferenceEngine(path, device=NumpyDevice())
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    t = threading.Thread(target=serve, args=(eng, port), daemon=True)
    t.start()
    import time
    x = numpy.zeros((2,) + tuple(eng._in.shape[1:]),
                    numpy.float32).tolist()
    req = urllib.request.Request(
        "http://127.0.0.1:%d/" % port,
        data=json.dumps({"inputs": x, "labels_only": True}).encode(),
        headers={"Content-Type": "application/json"})
    for attempt in range(20):
        try:
            with urllib.request.urlopen(req, timeout=10) as r:
                resp = json.loads(r.read())
            break
        except (ConnectionError, OSError):
            time.sleep(0.2)
    assert "labels" in resp and len(resp["labels"]) == 2
