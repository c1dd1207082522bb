"""Unit link/gate semantics (parity: veles/tests/test_units.py)."""

import numpy
import pytest

from veles.dummy import DummyWorkflow
from veles.units import Unit, TrivialUnit


class Recorder(TrivialUnit):
    log_ = []

    def run(self):
        Recorder.log_.append(self.name)


def build(wf, names):
    return {n: Recorder(wf, name=n) for n in names}


def test_linear_chain_order():
    Recorder.log_ = []
    wf = DummyWorkflow()
    u = build(wf, "abc")
    u["a"].link_from(wf.start_point)
    u["b"].link_from(u["a"])
    u["c"].link_from(u["b"])
    wf.end_point.link_from(u["c"])
    wf.initialize()
    wf.run()
    assert Recorder.log_ == ["a", "b", "c"]
    assert wf.is_finished


def test_fanin_waits_for_all_inputs():
    Recorder.log_ = []
    wf = DummyWorkflow()
    u = build(wf, "abcd")
    u["a"].link_from(wf.start_point)
    u["b"].link_from(wf.start_point)
    u["c"].link_from(u["a"], u["b"])   # c fires after BOTH
    wf.end_point.link_from(u["c"])
    wf.initialize()
    wf.run()
    assert Recorder.log_.count("c") == 1
    assert set(Recorder.log_) == {"a", "b", "c"}


def test_gate_block_swallows():
    Recorder.log_ = []
    wf = DummyWorkflow()
    u = build(wf, "ab")
    u["a"].link_from(wf.start_point)
    u["b"].link_from(u["a"])
    u["b"].gate_block <<= True
    wf.end_point.link_from(u["a"])
    wf.initialize()
    wf.run()
    assert Recorder.log_ == ["a"]


def test_gate_skip_propagates():
    Recorder.log_ = []
    wf = DummyWorkflow()
    u = build(wf, "abc")
    u["a"].link_from(wf.start_point)
    u["b"].link_from(u["a"])
    u["b"].gate_skip <<= True
    u["c"].link_from(u["b"])
    wf.end_point.link_from(u["c"])
    wf.initialize()
    wf.run()
    assert Recorder.log_ == ["a", "c"]


def test_link_attrs_by_reference():
    wf = DummyWorkflow()
    a = TrivialUnit(wf, name="a")
    b = TrivialUnit(wf, name="b")
    a.output = numpy.zeros(3)
    b.link_attrs(a, ("input", "output"))
    assert b.input is a.output
    a.output = numpy.ones(5)       # rebind on source
    assert b.input is a.output     # still aliased
    b.input = numpy.zeros(2)       # two-way: writes through
    assert a.output.shape == (2,)


def test_unlink():
    wf = DummyWorkflow()
    a = TrivialUnit(wf, name="a")
    b = TrivialUnit(wf, name="b")
    b.link_from(a)
    assert a in b.links_from and b in a.links_to
    b.unlink_from(a)
    assert a not in b.links_from and b not in a.links_to


def test_repeater_loop_with_decision_gate():
    """The canonical Veles loop: repeater -> body -> repeater, stopped by
    a complete flag flipping the repeater's gate_block."""
    from veles.workflow import Repeater
    from veles.mutable import Bool
    wf = DummyWorkflow()
    complete = Bool(False)
    counter = {"n": 0}

    class Body(Unit):
        def initialize(self, **kwargs):
            super(Body, self).initialize(**kwargs)

        def run(self):
            counter["n"] += 1
            if counter["n"] >= 5:
                complete.set(True)

    rep = Repeater(wf)
    body = Body(wf, name="body")
    rep.link_from(wf.start_point)
    body.link_from(rep)
    rep.link_from(body)
    rep.gate_block = complete
    wf.end_point.link_from(body)
    wf.end_point.gate_block = ~complete
    wf.initialize()
    wf.run()
    assert counter["n"] == 5
    assert wf.is_finished


def test_initialize_surfaces_genuine_typeerror():
    """a unit whose initialize RAISES TypeError (e.g. a bug
    mentioning 'unexpected keyword argument') must not be silently
    retried without kwargs — the signature-based dispatch never
    swallows unit exceptions."""
    from veles.workflow import Workflow
    from veles.units import Unit

    class Buggy(Unit):
        def initialize(self, device=None, **kwargs):
            raise TypeError(
                "got an unexpected keyword argument 'frobnicate'")

    wf = Workflow()
    Buggy(wf)
    with pytest.raises(TypeError, match="frobnicate"):
        wf.initialize(device=None)


def test_initialize_filters_unknown_kwargs():
    """A unit whose initialize() takes no 'snapshot' kwarg still
    initializes when the workflow-level kwargs carry one."""
    from veles.workflow import Workflow
    from veles.units import Unit

    seen = {}

    class Narrow(Unit):
        def initialize(self, device=None):
            seen["device"] = device

    wf = Workflow()
    Narrow(wf)
    wf.initialize(device="d", snapshot="ignored.pickle")
    assert seen["device"] == "d"


def test_link_attrs_one_way():
    # two_way=False: reads alias, rebinding through the alias refuses
    wf = DummyWorkflow()
    a = TrivialUnit(wf, name="a")
    b = TrivialUnit(wf, name="b")
    a.output = numpy.zeros(3)
    b.link_attrs(a, ("input", "output"), two_way=False)
    assert b.input is a.output
    try:
        b.input = numpy.ones(2)
    except AttributeError:
        pass
    else:
        raise AssertionError("one-way link accepted a write")
    assert a.output.shape == (3,)
