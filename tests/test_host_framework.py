"""Host-framework surface tests: CLI, import_file, graph export,
genetics GA, ensemble, image/pickle loaders, interaction shell,
veles.znicz pickle-compat aliases.
"""

import os
import pickle
import subprocess
import sys

import numpy
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_cli_dry_run():
    out = subprocess.run(
        [sys.executable, "-m", "veles",
         "veles/znicz_amd/models/wine.py", "--dry-run", "-b", "numpy"],
        cwd=REPO, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    assert "dry run:" in out.stdout


def test_cli_overrides_and_run(tmp_path):
    dot = tmp_path / "g.dot"
    out = subprocess.run(
        [sys.executable, "-m", "veles",
         "veles/znicz_amd/models/wine.py",
         "root.wine.decision.max_epochs=1",
         "root.wine.loader.n_train=30", "root.wine.loader.n_valid=0",
         "-b", "numpy", "--random-seed", "4",
         "--dump-graph", str(dot)],
        cwd=REPO, capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-2000:]
    assert dot.exists() and "digraph" in dot.read_text()


def test_cli_config_file(tmp_path):
    cfg = tmp_path / "cfg.py"
    cfg.write_text("root.wine.decision.max_epochs = 1\n"
                   "root.wine.loader.n_train = 30\n"
                   "root.wine.loader.n_valid = 0\n")
    out = subprocess.run(
        [sys.executable, "-m", "veles",
         "veles/znicz_amd/models/wine.py", str(cfg), "-b", "numpy"],
        cwd=REPO, capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-2000:]
    # the CLI must not run silently: epoch accounting reaches the user
    # (a missing setup_logging() once swallowed every self.info line)
    assert "epoch 1" in out.stderr or "epoch 1" in out.stdout, \
        (out.stdout[-500:], out.stderr[-500:])


def test_cli_genetics_optimize(tmp_path):
    # SURVEY §3.5 multi-run tuning call stack: --optimize runs the GA
    # over Tune() config leaves, one full workflow run per individual
    cfg = tmp_path / "cfg.py"
    cfg.write_text("from veles.genetics import Tune\n"
                   "root.wine.decision.max_epochs = 2\n"
                   "root.wine.loader.n_train = 60\n"
                   "root.wine.lr = Tune(0.05, 0.01, 0.2)\n")
    out = subprocess.run(
        [sys.executable, "-m", "veles",
         "veles/znicz_amd/models/wine.py", str(cfg), "-b", "numpy",
         "--optimize", "2", "--population", "3"],
        cwd=REPO, capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-2000:]
    text = out.stdout + out.stderr
    assert "best root.wine.lr" in text, text[-1000:]
    assert "best fitness" in text, text[-1000:]


def test_genetics_ga():
    from veles.config import root
    from veles.genetics import Tune, find_tunes, GeneticsOptimizer
    root.tga.lr = Tune(0.01, 0.001, 0.1)
    tunes = find_tunes(root.tga, "root.tga")
    assert len(tunes) == 1

    def ev(cfg):
        return -(cfg["root.tga.lr"] - 0.05) ** 2
    ga = GeneticsOptimizer(tunes, ev, population=8, generations=6)
    best = ga.run()
    assert abs(best.genes[0] - 0.05) < 0.02
    assert len(ga.history) == 6


def _ga_pmap_fitness(cfg):
    return -(cfg["root.tga2.lr"] - 0.05) ** 2


def test_genetics_ga_parallel_map():
    # pmap evaluates a whole generation concurrently; leaves are NOT
    # applied to the shared root (each worker applies its own cfg)
    import multiprocessing
    from veles.config import root
    from veles.genetics import Tune, find_tunes, GeneticsOptimizer
    root.tga2.lr = Tune(0.01, 0.001, 0.1)
    tunes = find_tunes(root.tga2, "root.tga2")
    with multiprocessing.Pool(2) as pool:
        ga = GeneticsOptimizer(tunes, _ga_pmap_fitness, population=8,
                               generations=4, pmap=pool.map)
        best = ga.run()
    assert abs(best.genes[0] - 0.05) < 0.03
    # pmap mode must leave the Tune marker in place (no root mutation)
    assert isinstance(root.tga2.lr, Tune)


def test_ensemble():
    from veles.ensemble import EnsembleTrainer, majority_vote
    models = EnsembleTrainer(lambda i, seed: ("m%d" % i, seed),
                             n_models=3).run()
    assert len(models) == 3
    votes = majority_vote([[0, 1, 1], [0, 1, 2], [1, 1, 2]])
    numpy.testing.assert_array_equal(votes, [0, 1, 2])


def test_image_loader(tmp_path):
    from PIL import Image
    for cls in ("cats", "dogs"):
        d = tmp_path / "train" / cls
        d.mkdir(parents=True)
        for i in range(3):
            arr = (numpy.random.RandomState(i).uniform(
                0, 255, (12, 10, 3))).astype(numpy.uint8)
            Image.fromarray(arr).save(d / ("%d.png" % i))
    from veles.dummy import DummyWorkflow
    from veles.backends import NumpyDevice
    from veles.loader.image import FileImageLoader
    ldr = FileImageLoader(
        DummyWorkflow(), minibatch_size=6,
        train_paths=[str(tmp_path / "train" / "*" / "*.png")],
        scale=(8, 8))
    ldr.initialize(device=NumpyDevice())
    assert ldr.original_data.shape == (6, 8, 8, 8)  # padded to 8 chans
    assert sorted(numpy.unique(ldr.original_labels.mem)) == [0, 1]
    ldr.run()
    assert ldr.minibatch_data.mem.shape[0] == 6


def test_pickles_loader_and_saver(tmp_path):
    from veles.dummy import DummyWorkflow
    from veles.backends import NumpyDevice
    from veles.loader.pickles import PicklesLoader, MinibatchesSaver
    data = numpy.random.RandomState(0).uniform(
        -1, 1, (20, 4)).astype(numpy.float32)
    labels = (numpy.arange(20) % 3).astype(numpy.int32)
    p = tmp_path / "train.pickle"
    with open(p, "wb") as f:
        pickle.dump((data, labels), f)
    ldr = PicklesLoader(DummyWorkflow(), minibatch_size=10,
                        train_path=str(p))
    ldr.initialize(device=NumpyDevice())
    assert ldr.class_lengths == [0, 0, 20]
    ldr.run()

    sv = MinibatchesSaver(DummyWorkflow(),
                          file_name=str(tmp_path / "mb.pickle.gz"))
    sv.input = ldr.minibatch_data
    sv.labels = ldr.minibatch_labels
    sv.minibatch_size = 10
    sv.initialize()
    sv.run()
    sv.close()
    import gzip
    with gzip.open(tmp_path / "mb.pickle.gz") as f:
        d, l = pickle.load(f)
    assert d.shape == (10, 4) and l.shape == (10,)


def test_shell_noop():
    from veles.dummy import DummyWorkflow
    from veles.interaction import Shell
    Shell(DummyWorkflow()).run()        # enabled=False -> returns


def test_znicz_alias_unpickles_reference_paths():
    import veles.znicz  # registers aliases  # noqa: F401
    import importlib
    mod = importlib.import_module("veles.znicz.all2all")
    assert mod.All2All.__module__ == "veles.znicz_amd.all2all"
    s = pickle.dumps(mod.All2All)
    assert pickle.loads(s) is mod.All2All


def test_uniform_unit_numpy():
    from veles import prng
    from veles.backends import NumpyDevice
    from veles.dummy import DummyWorkflow
    from veles.prng.uniform import Uniform
    prng.get().seed(2)
    u = Uniform(DummyWorkflow(), shape=(1000,), low=-1.0, high=1.0)
    assert u.initialize(device=NumpyDevice()) is not False
    u.run()
    v = u.output.mem
    assert -1.0 <= v.min() and v.max() <= 1.0
    assert abs(v.mean()) < 0.1


def test_web_status():
    import json
    import urllib.request
    from veles.web_status import StatusServer
    srv = StatusServer(port=0).start()
    try:
        srv.publish(epoch=3, err_pt=[1.0, 2.0, 3.0])
        with urllib.request.urlopen(
                "http://127.0.0.1:%d/" % srv.port, timeout=10) as r:
            data = json.loads(r.read())
        assert data["epoch"] == 3 and data["err_pt"] == [1.0, 2.0, 3.0]
    finally:
        srv.stop()


def _boom():
    raise RuntimeError("injected rank failure")


def test_emergency_snapshot(tmp_path):
    """a failing run writes an emergency snapshot that can
    be re-imported."""
    from veles import prng
    from veles.config import root
    from veles.launcher import Launcher
    from veles.snapshotter import SnapshotterToFile
    root.common.dirs.snapshots = str(tmp_path)
    prng.seed_all(4)
    from veles.znicz_amd.models.wine import WineWorkflow
    root.wine.decision.update({"max_epochs": 1})
    root.wine.loader.update({"n_train": 30, "n_valid": 0})
    launcher = Launcher(backend="numpy")
    wf = WineWorkflow(launcher)

    # inject a picklable failure into the decision unit
    orig_run = wf.decision.run
    wf.decision.run = _boom
    with pytest.raises(RuntimeError, match="injected"):
        launcher.boot()
    del wf.decision.run                  # drop the instance override
    snaps = list(tmp_path.glob("emergency_r0*.pickle.gz"))
    assert len(snaps) == 1
    wf2 = SnapshotterToFile.import_(str(snaps[0]))
    assert type(wf2).__name__ == "WineWorkflow"
    # resumed workflow re-initializes and completes with the fault gone
    if "run" in wf2.decision.__dict__:
        del wf2.decision.run
    wf2.workflow = Launcher(backend="numpy")
    wf2.workflow.workflow = wf2
    wf2.initialize(device=__import__(
        "veles.backends", fromlist=["NumpyDevice"]).NumpyDevice())
    wf2.run()
    assert wf2.is_finished


def test_graph_stepper_cpu_fallback():
    """GraphStepper on a CPU device must disable capture and keep the
    exact eager semantics (incl. gd_skip on valid minibatches)."""
    import numpy
    from veles import prng
    from veles.backends import NumpyDevice
    from veles.config import root
    from veles.dummy import DummyLauncher
    from veles.znicz_amd.hipgraph import GraphStepper

    prng.seed_all(7)
    from veles.znicz_amd.models.wine import WineWorkflow
    root.wine.decision.update({"max_epochs": 100})
    wf = WineWorkflow(DummyLauncher())
    wf.initialize(device=NumpyDevice())
    st = GraphStepper(wf)
    assert st._disabled
    w0 = wf.forwards[0].weights.mem.copy()
    for _ in range(6):
        st.step()
    w1 = wf.forwards[0].weights.mem
    assert not numpy.allclose(w0, w1)   # training progressed


def test_launcher_rejects_cpu_multirank():
    """Multi-rank without the GPU reducer would silently train divergent
    replicas — boot must refuse instead."""
    from veles.config import root
    from veles.launcher import Launcher
    from veles import prng
    prng.seed_all(9)
    from veles.znicz_amd.models.wine import WineWorkflow
    root.wine.decision.update({"max_epochs": 1})
    lch = Launcher(backend="numpy")
    wf = WineWorkflow(lch)
    assert lch.workflow is wf

    class _Stub(object):
        rank, world = 0, 2
    lch.dist = _Stub()
    with pytest.raises(RuntimeError, match="single-process"):
        lch.boot()


def test_cli_snapshot_resume(tmp_path):
    """ADVICE r1 regression: `-s snapshot` resume through the CLI (the
    launcher must be handed the unpickled workflow — boot() crashed on
    launcher.workflow None before the fix)."""
    snapdir = tmp_path / "snaps"
    cfg = tmp_path / "cfg.py"
    cfg.write_text(
        "root.common.dirs.snapshots = %r\n"
        "root.wine.decision.max_epochs = 2\n"
        "root.wine.loader.n_train = 30\n"
        "root.wine.loader.n_valid = 10\n"
        "root.wine.snapshotter.enabled = True\n"
        "root.wine.snapshotter.interval = 1\n"
        "root.wine.snapshotter.time_interval = 0\n" % str(snapdir))
    out = subprocess.run(
        [sys.executable, "-m", "veles",
         "veles/znicz_amd/models/wine.py", str(cfg), "-b", "numpy",
         "--random-seed", "7"],
        cwd=REPO, capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-2000:]
    snaps = sorted(snapdir.glob("*.pickle*"))
    assert snaps, "training wrote no snapshot"
    # resume from the snapshot and run 2 more epochs
    cfg2 = tmp_path / "cfg2.py"
    cfg2.write_text(cfg.read_text() +
                    "root.wine.decision.max_epochs = 4\n")
    out2 = subprocess.run(
        [sys.executable, "-m", "veles",
         "veles/znicz_amd/models/wine.py", str(cfg2), "-b", "numpy",
         "-s", str(snaps[-1])],
        cwd=REPO, capture_output=True, text=True, timeout=600)
    assert out2.returncode == 0, out2.stderr[-2000:]


def test_snapshot_codecs_roundtrip(tmp_path):
    # all four reference codecs ("", gz, bz2, xz) must export+import
    from veles import prng
    prng.seed_all(7)
    import veles.znicz_amd.models.wine  # noqa: F401
    from veles.config import root
    root.wine.decision.max_epochs = 1
    root.wine.loader.update({"n_train": 30, "n_valid": 0})
    from veles.dummy import DummyLauncher
    from veles.backends import NumpyDevice
    from veles.snapshotter import SnapshotterBase, SnapshotterToFile
    from veles.znicz_amd.models.wine import WineWorkflow
    wf = WineWorkflow(DummyLauncher())
    wf.initialize(device=NumpyDevice())
    wf.run()
    for codec in ("", "gz", "bz2", "xz"):
        snap = SnapshotterToFile(wf, prefix="c_%s" % (codec or "raw"),
                                 directory=str(tmp_path),
                                 compression=codec)
        wf.del_ref(snap)
        path = snap.export()
        assert os.path.exists(path), codec
        wf2 = SnapshotterBase.import_(path)
        numpy.testing.assert_array_equal(
            wf2.forwards[0].weights.mem, wf.forwards[0].weights.mem,
            err_msg=codec)


def test_routing_table_loader_robust(tmp_path, monkeypatch):
    """A corrupt gfx950_routing.json must never block the ops import."""
    import veles.znicz_amd.ops as ops_mod

    class FakeMod:
        calls = []

        @staticmethod
        def set_routing(a, b, c, d):
            FakeMod.calls.append((a, b, c, d))

    bad = tmp_path / "gfx950_routing.json"
    bad.write_text("{not json")
    real_join = os.path.join

    def fake_join(*a):
        if a and a[-1] == "gfx950_routing.json":
            return str(bad)
        return real_join(*a)

    monkeypatch.setattr(os.path, "join", fake_join)
    ops_mod._apply_routing(FakeMod)          # must not raise
    assert FakeMod.calls == []
    bad.write_text('{"nt_split_budget": 640}')
    ops_mod._apply_routing(FakeMod)
    assert FakeMod.calls == [(640, -1, -1, -1)]

    # per-shape FC split rows reach set_fc_split; malformed rows only
    # drop the table, never the import
    class FakeMod2:
        calls = []
        fc = []

        @staticmethod
        def set_routing(a, b, c, d):
            FakeMod2.calls.append((a, b, c, d))

        @staticmethod
        def set_fc_split(is_tn, m, n, k, s):
            FakeMod2.fc.append((is_tn, m, n, k, s))

    bad.write_text('{"fc_nt_splits": [[256, 1000, 4096, 16]], '
                   '"fc_tn_splits": [[1000, 4096, 256, 1]]}')
    ops_mod._apply_routing(FakeMod2)
    assert FakeMod2.fc == [(0, 256, 1000, 4096, 16),
                           (1, 1000, 4096, 256, 1)]
    bad.write_text('{"fc_nt_splits": [[256, "x"]]}')
    ops_mod._apply_routing(FakeMod2)         # must not raise

    # the COMMITTED table must parse and contain the swept row kinds
    import json as js
    real = js.load(open(real_join(
        os.path.dirname(os.path.abspath(ops_mod.__file__)),
        "gfx950_routing.json")))
    for field in ("fc_nt_splits", "fc_tn_splits"):
        for row in real.get(field, []):
            assert len(row) == 4 and all(
                isinstance(v, int) and v > 0 for v in row), row


def test_golden_snapshot_format_stable():
    """Format-stability guard: the committed round-2 wine snapshot
    (tests/data/golden_wine.0.pickle.gz) must keep loading, resuming
    and training in every future revision — the pickle object graph
    with numpy-only Arrays IS the on-disk contract."""
    from veles.backends import NumpyDevice
    from veles.loader.base import TRAIN
    from veles.snapshotter import SnapshotterToFile
    path = os.path.join(REPO, "tests", "data",
                        "golden_wine.0.pickle.gz")
    wf = SnapshotterToFile.import_(path)
    # the pickled state carries the trained weights and epoch counters
    assert int(wf.loader.epoch_number) == 2
    assert wf.decision.epoch_n_err_pt[TRAIN] == pytest.approx(
        3.3333333, abs=1e-3)
    for f in wf.forwards:
        w = getattr(f, "weights", None)
        if w is not None and w:
            assert numpy.abs(w.mem).max() > 0
    # resume: 2 more epochs on the numpy reference device
    wf.decision.max_epochs = 4
    wf.decision.complete <<= False
    wf.initialize(device=NumpyDevice())
    wf.run()
    assert int(wf.loader.epoch_number) == 4


def test_velescli_shim(tmp_path):
    # the reference's scripts/velescli.py entry point works like -m veles
    cfg = tmp_path / "cfg.py"
    cfg.write_text("root.wine.decision.max_epochs = 1\n"
                   "root.wine.loader.n_train = 30\n"
                   "root.wine.loader.n_valid = 0\n")
    out = subprocess.run(
        [sys.executable, "veles/scripts/velescli.py",
         "veles/znicz_amd/models/wine.py", str(cfg), "-b", "numpy"],
        cwd=REPO, capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-2000:]
    assert "epoch 1" in out.stderr or "epoch 1" in out.stdout
