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
    wf.initialize(device=ROCmDevice(0))
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
    prng.seed_all(41)
    wf_e = build_cifar_small()
    prng.seed_all(41)
    wf_g = build_cifar_small()
    run_steps(wf_e, 8, graph=False)
    st = run_steps(wf_g, 8, graph=True)
    assert st._graph is not None, "capture fell back to eager"
    torch.cuda.synchronize()
    we, wg = first_weights(wf_e), first_weights(wf_g)
    assert len(we) == len(wg) > 0
    for a, b in zip(we, wg):
        numpy.testing.assert_allclose(a, b, rtol=1e-4, atol=1e-5)
    # training actually progressed (n_err counters accumulated)
    wf_g.evaluator.n_err.map_read()


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
    st.step()                      # signature mismatch -> re-capture
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
