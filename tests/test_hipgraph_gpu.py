"""hipGraph step-capture tests (veles/znicz_amd/hipgraph.py).

The captured step must (a) actually capture — a silent eager fallback
would hide a broken graph path — and (b) produce the exact same weight
trajectory as the eager step, since both run the same kernels on the
same buffers (dropout reads its device-side counter either way).
"""

import numpy
import pytest

pytestmark = pytest.mark.gpu

torch = pytest.importorskip("torch")
if not torch.cuda.is_available():
    pytest.skip("no GPU", allow_module_level=True)

from veles import prng  # noqa: E402
from veles.backends import ROCmDevice  # noqa: E402
from veles.config import root  # noqa: E402
from veles.dummy import DummyLauncher  # noqa: E402


def build_cifar_small():
    from veles.znicz_amd.models.cifar import CifarWorkflow
    root.cifar.loader.update({"minibatch_size": 64, "n_train": 256,
                              "n_valid": 0, "on_device": True,
                              "device_dtype": "bf16"})
    root.cifar.decision.update({"max_epochs": 10 ** 9,
                                "fail_iterations": 10 ** 9})
    wf = CifarWorkflow(DummyLauncher())
    dev = ROCmDevice(0)
    wf.initialize(device=dev)
    # the comm-stream reducer path is what bench runs (world=1 included);
    # without it the GD units' inline update path is not capture-safe
    from bench import _NullCtx
    from veles.znicz_amd.parallel import GradientReducer
    wf.grad_reducer = GradientReducer(_NullCtx(), dev)
    return wf


def first_weights(wf):
    out = []
    for f in wf.forwards:
        w = getattr(f, "weights", None)
        if w is not None and w:
            w.map_read()
            out.append(w.mem.copy())
    return out


def run_steps(wf, n, graph):
    from veles.znicz_amd.hipgraph import GraphStepper
    if graph:
        st = GraphStepper(wf, warmup=2)
        for _ in range(n):
            st.step()
        return st
    loader, ev, dec = wf.loader, wf.evaluator, wf.decision
    gds_rev = [g for g in reversed(wf.gds)]
    for _ in range(n):
        loader.run()
        for f in wf.forwards:
            f.run()
        ev.run()
        dec.run()
        for g in gds_rev:
            g.run()
        wf.comm_sync.run()
    return None


def test_graph_stepper_matches_eager():
    """Graph-vs-eager weight trajectories must agree to within the
    run-to-run drift of the eager path itself (the split-K kernels
    accumulate through fp32 atomics, so identical seeds are not bitwise
    reproducible — twin eager runs calibrate the tolerance)."""
    prng.seed_all(41)
    wf_e1 = build_cifar_small()
    prng.seed_all(41)
    wf_e2 = build_cifar_small()
    prng.seed_all(41)
    wf_g = build_cifar_small()
    run_steps(wf_e1, 8, graph=False)
    run_steps(wf_e2, 8, graph=False)
    st = run_steps(wf_g, 8, graph=True)
    assert st._graph is not None, \
        getattr(st, "capture_error", "capture fell back to eager")
    torch.cuda.synchronize()
    we1, we2, wg = (first_weights(w) for w in (wf_e1, wf_e2, wf_g))
    assert len(we1) == len(wg) > 0
    for a, b, g in zip(we1, we2, wg):
        drift = numpy.abs(a - b).max()          # eager nondeterminism
        budget = max(50.0 * drift, 2e-3)
        err = numpy.abs(a - g).max()
        assert err <= budget, (err, drift)


def test_graph_stepper_learns():
    """Training must actually progress under replay: a frozen minibatch
    or stale-counter replay would stall the synthetic-data error."""
    from veles.loader.base import TRAIN
    prng.seed_all(43)
    wf = build_cifar_small()
    from veles.znicz_amd.hipgraph import GraphStepper
    st = GraphStepper(wf, warmup=2)
    steps_per_epoch = 4                       # 256 train / 64 batch
    for _ in range(steps_per_epoch):
        st.step()
    first = wf.decision.epoch_n_err_pt[TRAIN]
    for _ in range(steps_per_epoch * 14):
        st.step()
    last = wf.decision.epoch_n_err_pt[TRAIN]
    assert st._graph is not None
    assert last < 0.7 * first, (first, last)


def test_graph_stepper_recaptures_on_lr_change():
    prng.seed_all(42)
    wf = build_cifar_small()
    from veles.znicz_amd.hipgraph import GraphStepper
    st = GraphStepper(wf, warmup=1)
    for _ in range(4):
        st.step()
    g1 = st._graph
    assert g1 is not None
    for gd in wf.gds:
        gd.learning_rate *= 0.5
    st.step()                      # signature mismatch -> eager rebuild
    assert st._graph is None
    st.step()                      # -> re-capture with new params
    assert st._graph is not None and st._graph is not g1


def test_dropout_fresh_masks_across_replays():
    """The device-side counter must advance inside the graph, so two
    replays of the same graph draw different masks."""
    from veles.znicz_amd import ops
    n = 4096
    x = torch.ones(n, device="cuda", dtype=torch.bfloat16)
    y = torch.empty_like(x)
    mask = torch.empty(n, device="cuda", dtype=torch.uint8)
    ctr = torch.zeros(1, device="cuda", dtype=torch.int64)
    # warmup then capture
    ops.dropout_fwd(x, y, mask, 0.5, 1234, 0, ctr)
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        ops.dropout_fwd(x, y, mask, 0.5, 1234, 0, ctr)
    g.replay()
    torch.cuda.synchronize()
    m1 = mask.cpu().numpy().copy()
    c1 = int(ctr.item())
    g.replay()
    torch.cuda.synchronize()
    m2 = mask.cpu().numpy().copy()
    assert int(ctr.item()) == c1 + 1
    assert (m1 != m2).any(), "replayed dropout mask is frozen"
    # keep-rate sane both times
    for m in (m1, m2):
        assert 0.4 < m.mean() < 0.6
