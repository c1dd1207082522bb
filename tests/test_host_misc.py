"""Direct tests for the small host-framework modules that are
otherwise only exercised indirectly (thread pool failure protection,
distribution hooks, DOT export, path/import/timing utils)."""

import os
import threading
import time

import numpy
import pytest  # noqa: F401


def test_thread_pool_runs_and_joins():
    from veles.thread_pool import ThreadPool
    pool = ThreadPool(maxthreads=3)
    hits = []
    lock = threading.Lock()

    def work(i):
        with lock:
            hits.append(i)
    for i in range(20):
        pool.callInThread(work, i)
    pool.join()
    assert sorted(hits) == list(range(20))
    pool.shutdown()
    # restartable after shutdown
    pool.callInThread(work, 99)
    pool.join()
    assert 99 in hits
    pool.shutdown()


def test_thread_pool_failure_callback():
    """Silent-failure protection: an exploding task reaches the
    registered failure callback and the pool keeps serving."""
    from veles.thread_pool import ThreadPool
    pool = ThreadPool(maxthreads=1)
    caught = []
    pool.register_on_failure(caught.append)

    def boom():
        raise RuntimeError("task failed")

    ok = []
    pool.callInThread(boom)
    pool.callInThread(ok.append, 1)
    pool.join()
    pool.shutdown()
    assert len(caught) == 1 and "task failed" in str(caught[0])
    assert ok == [1]


def test_thread_pool_pause_resume():
    from veles.thread_pool import ThreadPool
    pool = ThreadPool(maxthreads=1)
    ran = []
    pool.pause()
    pool.callInThread(ran.append, 1)
    time.sleep(0.1)
    assert ran == []          # paused: queued but not executed
    pool.resume()
    pool.join()
    assert ran == [1]
    pool.shutdown()


def test_distributable_hooks_are_noops():
    from veles.distributable import (Distributable,
                                     TriviallyDistributable)
    d = TriviallyDistributable()
    assert d.generate_data_for_master() is None
    assert d.generate_data_for_slave() is None
    d.apply_data_from_master({"x": 1})
    d.apply_data_from_slave({"x": 1})
    d.drop_slave_state()
    assert isinstance(d, Distributable)


def test_workflow_to_dot(tmp_path):
    from veles.dummy import DummyLauncher
    from veles.config import root
    import veles.znicz_amd.models.wine  # noqa: F401
    root.wine.decision.max_epochs = 1
    from veles.znicz_amd.models.wine import WineWorkflow
    from veles.graph_plotter import workflow_to_dot, save_dot
    wf = WineWorkflow(DummyLauncher())
    dot = workflow_to_dot(wf)
    assert dot.startswith("digraph") and dot.rstrip().endswith("}")
    # every unit appears as a node; control links appear as edges
    assert dot.count("->") >= len(list(wf)) - 2
    for u in wf:
        assert type(u).__name__ in dot
    p = save_dot(wf, str(tmp_path / "wf.dot"))
    assert os.path.exists(p) and open(p).read() == dot


def test_import_file_as_module(tmp_path):
    from veles.import_file import import_file_as_module, \
        try_to_import_file
    mod_path = tmp_path / "my_wf.py"
    mod_path.write_text("VALUE = 41\n\ndef get():\n    return VALUE + 1\n")
    mod = import_file_as_module(str(mod_path))
    assert mod.get() == 42
    bad = tmp_path / "broken.py"
    bad.write_text("raise ValueError('nope')\n")
    err = try_to_import_file(str(bad))
    assert isinstance(err, ValueError)


def test_timeit_helpers(capsys):
    from veles.timeit2 import timeit, timed
    res, dt = timeit(lambda a, b: a + b, 2, 3)
    assert res == 5 and dt >= 0

    @timed
    def f(x):
        return x * 2
    assert f(4) == 8
    assert "took" in capsys.readouterr().out


def test_error_types():
    from veles import error
    for name in ("BadFormatError", "NotExistsError",
                 "AlreadyExistsError", "DeviceError",
                 "MasterSlaveCommunicationError"):
        cls = getattr(error, name)
        assert issubclass(cls, error.VelesException)
        assert issubclass(cls, Exception)


def test_paths_expand(monkeypatch):
    from veles.paths import expand, __root__, __project_root__
    assert os.path.isdir(__root__) and os.path.isdir(__project_root__)
    monkeypatch.setenv("VELES_T_DIR", "/tmp/somewhere")
    assert expand("$VELES_T_DIR/x") == "/tmp/somewhere/x"
    assert expand("~").startswith("/")


def test_numpy_available():
    # keeps the numpy import (used by other tests' fixtures) honest
    assert numpy.zeros(3).sum() == 0


def test_accelerated_unit_dispatch():
    """run() dispatches numpy_run on host devices, and missing
    implementations fail loudly instead of silently no-opping."""
    from veles.accelerated_units import AcceleratedUnit
    from veles.backends import NumpyDevice
    from veles.dummy import DummyWorkflow

    calls = []

    class U(AcceleratedUnit):
        def numpy_run(self):
            calls.append("numpy")

        def hip_run(self):
            calls.append("hip")

    wf = DummyWorkflow()
    u = U(wf)
    u.initialize(device=NumpyDevice())
    assert not u.on_gpu
    u.run()
    assert calls == ["numpy"]

    class Bare(AcceleratedUnit):
        pass
    b = Bare(wf)
    b.initialize(device=NumpyDevice())
    with pytest.raises(NotImplementedError, match="numpy_run"):
        b.run()
    # legacy kernel-compilation shims: parity surface behaves sanely
    assert u.build_program(defines={"X": 1}) is None
    with pytest.raises(RuntimeError, match="precompiled"):
        u.execute_kernel(global_size=(1,))


def test_logger_setup_idempotent(capsys):
    import logging
    from veles.logger import setup_logging
    setup_logging(logging.INFO)
    n = len(logging.getLogger().handlers)
    setup_logging(logging.INFO)     # second call must not stack handlers
    assert len(logging.getLogger().handlers) == n


def test_per_unit_runtime_accounting():
    """per-unit wall-time accounting inside Unit.run()
    wrappers + the workflow's print_stats table."""
    from veles import prng
    prng.seed_all(4)
    from veles.config import root
    import veles.znicz_amd.models.wine  # noqa: F401
    root.wine.decision.max_epochs = 1
    from veles.backends import NumpyDevice
    from veles.dummy import DummyLauncher
    from veles.znicz_amd.models.wine import WineWorkflow
    wf = WineWorkflow(DummyLauncher())
    wf.initialize(device=NumpyDevice())
    wf.run()
    ran = [u for u in wf if u.run_executed_counter > 0]
    assert len(ran) >= 5
    assert all(u.run_time >= 0 for u in ran)
    assert any(u.run_time > 0 for u in ran)
    rows = wf.print_stats(top=5)
    assert len(rows) == 5
    assert rows == sorted(rows, reverse=True)   # slowest first


def test_compat_shims():
    from veles.compat import from_none, which, BytesIO, StringIO, PYPY
    e = from_none(ValueError("x"))
    assert e.__suppress_context__ and e.__cause__ is None
    assert which("python3")
    assert BytesIO and StringIO and PYPY is False
