"""Workflow — the unit-graph container and execution driver.

Parity target: reference ``veles/workflow.py`` (``Workflow``,
``StartPoint``, ``EndPoint``, ``Repeater``; ``initialize``/``run``/
``stop``; ``generate_graph``; unit enumeration and references).

MI355X redesign: the reference schedules unit ``run()`` calls on a thread
pool; here the default driver is a deterministic sequential event loop
(GPU work is HIP-stream-async so host threads add only nondeterminism).
Gate semantics (``gate_block`` swallows, ``gate_skip`` forwards) and the
Repeater-loop construct are preserved exactly.
"""

import inspect
from collections import deque

from veles.units import Unit, TrivialUnit, Container
from veles.mutable import Bool


def _init_signature(unit):
    """Signature of unit.initialize, or None if it takes **kwargs (or
    cannot be inspected) — None means 'pass everything through'."""
    try:
        sig = inspect.signature(unit.initialize)
    except (TypeError, ValueError):
        return None
    for p in sig.parameters.values():
        if p.kind is inspect.Parameter.VAR_KEYWORD:
            return None
    return sig


class StartPoint(TrivialUnit):
    """The unit the workflow fires first."""

    def __init__(self, workflow, **kwargs):
        kwargs.setdefault("name", "START")
        super(StartPoint, self).__init__(workflow, **kwargs)


class EndPoint(TrivialUnit):
    """Reaching this unit (gate open) completes the workflow run."""

    def __init__(self, workflow, **kwargs):
        kwargs.setdefault("name", "END")
        super(EndPoint, self).__init__(workflow, **kwargs)

    def run(self):
        self.workflow.on_workflow_finished()


class Repeater(TrivialUnit):
    """Loop hub: units link back to it; Decision flips its gate to stop."""

    def __init__(self, workflow, **kwargs):
        kwargs.setdefault("name", "REPEATER")
        super(Repeater, self).__init__(workflow, **kwargs)

    def open_gate(self, src):
        # A repeater fires whenever ANY input fires (loop-back
        # semantics), matching the reference behavior for loop hubs.
        # Note: this resets EVERY input flag on each fire, which
        # diverges from per-edge gate accounting when a Repeater has
        # more than one non-loop input — by design: a Repeater is a
        # loop hub, not a join point (use a plain Unit to join).
        if src in self.links_from:
            for u in self.links_from:
                self.links_from[u] = False
        return True


class Workflow(Container):
    """Owns units; drives the event loop from start_point to end_point."""

    def __init__(self, workflow=None, **kwargs):
        self._units = []
        super(Workflow, self).__init__(workflow, **kwargs)
        self.start_point = StartPoint(self)
        self.end_point = EndPoint(self)
        self.negotiates_on_connect = False
        self._is_running = False
        self._finished = Bool(False)
        self._stop_requested = False
        self.device = None
        self._run_callbacks = []

    # -- unit registry ------------------------------------------------------
    def add_ref(self, unit):
        if unit is self:
            return
        if unit not in self._units:
            self._units.append(unit)

    def del_ref(self, unit):
        if unit in self._units:
            self._units.remove(unit)

    @property
    def units(self):
        return list(self._units)

    def __iter__(self):
        return iter(self._units)

    def __len__(self):
        return len(self._units)

    def get_unit(self, name):
        for u in self._units:
            if u.name == name:
                return u
        raise KeyError(name)

    #: reference API parity: this engine's run() is always blocking (the
    #: reference could also run detached under its Twisted reactor)
    run_is_blocking = True

    @property
    def is_running(self):
        return self._is_running

    @property
    def is_finished(self):
        return bool(self._finished)

    # -- lifecycle ----------------------------------------------------------
    def initialize(self, device=None, **kwargs):
        """Multi-pass unit initialization.

        A unit's ``initialize`` may return False to defer until its data
        dependencies (linked attrs) are materialized by upstream units —
        the reference's re-initialization ordering, preserved here.
        """
        super(Workflow, self).initialize(**kwargs)
        self.device = device
        pending = [u for u in self._units if u is not self]
        max_passes = len(pending) + 2
        for _ in range(max_passes):
            deferred = []
            for u in pending:
                kw = dict(kwargs)
                kw["device"] = device
                # Pass only the kwargs the unit's initialize() accepts
                # (signature-inspected, so a unit's own genuine
                # TypeError is never swallowed).
                sig = _init_signature(u)
                if sig is not None:
                    kw = {k: v for k, v in kw.items()
                          if k in sig.parameters}
                result = u.initialize(**kw)
                if result is False:
                    deferred.append(u)
            if not deferred:
                break
            pending = deferred
        else:
            raise RuntimeError(
                "units failed to initialize after %d passes: %s" %
                (max_passes, pending))
        self._finished <<= False
        self._stop_requested = False
        return self

    def run(self):
        """Drive the event loop: fire start_point, propagate until the
        end_point runs or stop() is requested."""
        self._is_running = True
        self._finished <<= False
        try:
            queue = deque()
            queue.append((self.start_point, False))
            while queue and not self._finished and not self._stop_requested:
                unit, skip = queue.popleft()
                if not skip:
                    unit._execute()
                if self._finished or self._stop_requested:
                    break
                for nxt in unit.links_to:
                    if nxt.open_gate(unit):
                        if nxt.gate_block:
                            continue
                        queue.append((nxt, bool(nxt.gate_skip)))
        finally:
            self._is_running = False
        return self

    def stop(self):
        self._stop_requested = True
        for u in self._units:
            u.stop()

    def on_workflow_finished(self):
        self._finished <<= True
        for cb in self._run_callbacks:
            cb()

    def add_run_callback(self, cb):
        self._run_callbacks.append(cb)

    # -- introspection -------------------------------------------------------
    def generate_graph(self, filename=None, write_on_disk=True):
        """Emit the unit graph in Graphviz dot text (reference used pydot)."""
        lines = ["digraph %s {" % type(self).__name__.replace(".", "_")]
        ids = {u: "u%d" % i for i, u in enumerate(self._units)}
        for u in self._units:
            lines.append('  %s [label="%s"];' % (ids[u], u.name))
        for u in self._units:
            for v in u.links_to:
                if v in ids:
                    lines.append("  %s -> %s;" % (ids[u], ids[v]))
        lines.append("}")
        text = "\n".join(lines)
        if filename and write_on_disk:
            with open(filename, "w") as f:
                f.write(text)
        return text

    def print_stats(self, top=20):
        """Per-unit wall-time accounting table (reference parity)."""
        rows = sorted(((u.run_time, u.run_executed_counter, u.name)
                       for u in self._units), reverse=True)[:top]
        self.info("%-32s %12s %10s", "unit", "total_s", "runs")
        for t, n, name in rows:
            self.info("%-32s %12.4f %10d", name, t, n)
        return rows

    # -- pickling ------------------------------------------------------------
    def __getstate__(self):
        state = super(Workflow, self).__getstate__()
        state["_is_running"] = False
        state["_run_callbacks"] = []
        state["device"] = None
        return state


# Backwards-compatible alias used by some reference samples.
class NNWorkflowBase(Workflow):
    pass
