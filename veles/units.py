"""Unit — the node of the workflow graph.

Parity target: reference ``veles/units.py`` (``Unit``, ``TrivialUnit``,
``IUnit``): control links (``link_from``), data links (``link_attrs`` —
attribute aliasing by reference), gate logic (``gate_block``,
``gate_skip``), deferred ``initialize`` ordering, per-unit run-time
accounting.

MI355X redesign notes: execution is event-driven exactly like the
reference, but the default driver is a deterministic sequential loop in
:class:`veles.workflow.Workflow` — on this hardware all kernel work is
HIP-stream-async, so host-thread concurrency buys nothing in the hot loop
and costs determinism.  The gate semantics are preserved bit-for-bit:
a unit fires when *all* its control inputs have fired since its last run;
``gate_block`` swallows the trigger; ``gate_skip`` propagates without
running.
"""

import time

from veles.config import root
from veles.distributable import Distributable
from veles.logger import Logger
from veles.mutable import Bool, LinkableAttribute


class IUnit(object):
    """Marker kept for zope.interface-free API parity."""


class Unit(Logger, Distributable):
    """Graph node.  See module docstring for semantics."""

    hide_from_registry = False

    def __init__(self, workflow, **kwargs):
        super(Unit, self).__init__()
        self._name = kwargs.get("name", type(self).__name__)
        self.view_group = kwargs.get("view_group", "PLUMBING")
        # control-flow links
        self.links_from = {}    # {unit: fired_flag(bool)}
        self.links_to = {}      # {unit: True}
        # gates
        self.gate_block = Bool(False)
        self.gate_skip = Bool(False)
        # state
        self._initialized = False
        self._workflow = None
        self.run_executed_counter = 0
        self.run_time = 0.0
        self._stopped = False
        self.workflow = workflow
        if workflow is not None:
            workflow.add_ref(self)

    # -- identity ----------------------------------------------------------
    @property
    def name(self):
        return self._name

    @name.setter
    def name(self, value):
        self._name = value
        self._reset_logger()

    @property
    def workflow(self):
        return self._workflow

    @workflow.setter
    def workflow(self, wf):
        self._workflow = wf

    @property
    def is_initialized(self):
        return self._initialized

    def __repr__(self):
        return '<%s "%s">' % (type(self).__name__, self._name)

    # -- linking -----------------------------------------------------------
    def link_from(self, *units):
        """Run self after each of *units* finishes (control dependency)."""
        for u in units:
            if u is self:
                raise ValueError("cannot link a unit to itself")
            self.links_from[u] = False
            u.links_to[self] = True
        return self

    def unlink_from(self, *units):
        for u in units:
            self.links_from.pop(u, None)
            u.links_to.pop(self, None)
        return self

    def unlink_all(self):
        for u in list(self.links_from):
            self.unlink_from(u)
        for u in list(self.links_to):
            u.unlink_from(self)
        return self

    def unlink_before(self):
        for u in list(self.links_from):
            self.unlink_from(u)
        return self

    def unlink_after(self):
        for u in list(self.links_to):
            u.unlink_from(self)
        return self

    def link_attrs(self, other, *args, two_way=True):
        """Alias attributes of *other* into self **by reference**.

        Each arg is either ``"attr"`` (same name both sides) or a tuple
        ``("my_attr", "other_attr")``.  Two-way by default (rebinding
        the alias writes through to *other*, the reference semantics);
        ``two_way=False`` makes rebinding through the alias an error.
        """
        table = self.__dict__.setdefault("_linked_attrs_", {})
        for arg in args:
            if isinstance(arg, tuple):
                mine, theirs = arg
            else:
                mine = theirs = arg
            # an existing instance attribute would shadow the link
            self.__dict__.pop(mine, None)
            table[mine] = LinkableAttribute(other, theirs,
                                            two_way=two_way)
        return self

    def unlink_attrs(self, *names):
        table = self.__dict__.get("_linked_attrs_", {})
        for n in names:
            table.pop(n, None)

    # attribute forwarding for linked attrs
    def __getattr__(self, name):
        # only called when normal lookup fails OR never for instance attrs;
        # linked attrs are not stored in __dict__ directly.
        if name.startswith("_"):
            raise AttributeError(name)
        table = self.__dict__.get("_linked_attrs_")
        if table is not None and name in table:
            return table[name].get()
        raise AttributeError("%r has no attribute %r" % (self, name))

    def __setattr__(self, name, value):
        table = self.__dict__.get("_linked_attrs_")
        if table is not None and name in table and not name.startswith("_"):
            table[name].set(value)
            return
        object.__setattr__(self, name, value)

    # -- lifecycle ---------------------------------------------------------
    def initialize(self, **kwargs):
        """Prepare for run.  May return False to ask the workflow to retry
        after other units have initialized (re-initialization ordering)."""
        self._initialized = True

    def run(self):
        """The unit's work.  Subclasses override."""

    def stop(self):
        self._stopped = True

    @property
    def stopped(self):
        return self._stopped

    # -- gate/fire machinery (driven by Workflow) ---------------------------
    def open_gate(self, src):
        """Record that *src* fired; return True when all inputs have."""
        if src in self.links_from:
            self.links_from[src] = True
        if all(self.links_from.values()):
            for u in self.links_from:
                self.links_from[u] = False
            return True
        return False

    def _execute(self):
        """Run with accounting; returns False if gate_skip suppressed it."""
        if self.gate_skip:
            return False
        t0 = time.time()
        try:
            self.run()
        finally:
            dt = time.time() - t0
            self.run_time += dt
            self.run_executed_counter += 1
            if root.common.trace.get("run_time", False):
                self.debug("run() took %.3f ms", dt * 1e3)
        return True

    # -- pickling ----------------------------------------------------------
    def __getstate__(self):
        state = super(Unit, self).__getstate__()
        state["_initialized"] = False
        state.pop("_logger_", None)
        return state

    def __setstate__(self, state):
        self.__dict__.update(state)
        self._logger_ = None


class TrivialUnit(Unit):
    """A unit that does nothing when run (plumbing node)."""

    def initialize(self, **kwargs):
        super(TrivialUnit, self).initialize(**kwargs)

    def run(self):
        pass


class Container(Unit):
    """Marker base for units that contain other units (Workflow)."""
