"""Linkable mutable values — the gate mechanism.

Parity target: reference ``veles/mutable.py`` (``Bool`` with live logical
expressions and on-change callbacks, ``LinkableAttribute``).  Units share
``Bool`` objects: e.g. ``decision.complete`` is the *same object* (or a
live derivation) as ``repeater.gate_block``, so flipping one flips the
gate.
"""


class Bool(object):
    """A mutable boolean with live derivations.

    ``~b``, ``a & b``, ``a | b`` return *derived* Bools that recompute
    whenever a source changes.  ``b <<= value`` assigns (reference's
    ``__ilshift__`` idiom) and fires on-change callbacks.
    """

    __slots__ = ("_value", "_derived", "_compute", "_sources", "_op",
                 "on_change")

    _OPS = {
        "not": lambda a: not bool(a),
        "and": lambda a, b: bool(a) and bool(b),
        "or": lambda a, b: bool(a) or bool(b),
        "xor": lambda a, b: bool(a) != bool(b),
    }

    def __init__(self, value=False):
        self._value = bool(value)
        self._derived = []      # Bools depending on self
        self._compute = None    # callable for derived bools
        self._sources = ()
        self._op = None         # op tag for pickling derived bools
        self.on_change = []     # callbacks f(bool_obj)

    # -- value access ------------------------------------------------------
    def __bool__(self):
        # derived bools also PULL on read: makes them correct even if the
        # push registration was lost (e.g. unpickle ordering).
        if self._compute is not None:
            self._set(self._compute())
        return self._value

    __nonzero__ = __bool__

    @property
    def value(self):
        return self._value

    def _set(self, v):
        v = bool(v)
        if v == self._value:
            return
        self._value = v
        for cb in list(self.on_change):
            cb(self)
        for d in list(self._derived):
            d._recompute()

    def __ilshift__(self, other):
        """``b <<= x`` — assign value (x may be a Bool or a bool)."""
        self._set(bool(other))
        return self

    def set(self, v=True):
        self._set(v)

    def unset(self):
        self._set(False)

    def toggle(self):
        self._set(not self._value)

    # -- derivations -------------------------------------------------------
    @staticmethod
    def _derive(op, *sources):
        fn = Bool._OPS[op]
        b = Bool(fn(*sources))
        b._compute = lambda: fn(*b._sources)
        b._sources = sources
        b._op = op
        for s in sources:
            if isinstance(s, Bool):
                s._derived.append(b)
        return b

    def _recompute(self):
        if self._compute is not None:
            self._set(self._compute())

    def __invert__(self):
        return Bool._derive("not", self)

    def __and__(self, other):
        return Bool._derive("and", self, other)

    def __or__(self, other):
        return Bool._derive("or", self, other)

    def __xor__(self, other):
        return Bool._derive("xor", self, other)

    def __repr__(self):
        return "Bool(%s)" % self._value

    # pickling: drop callbacks (units re-link on initialize), keep the
    # derivation topology so gates stay LIVE after snapshot/resume
    def __getstate__(self):
        return (self._value, self._op, self._sources)

    def __setstate__(self, state):
        self._value = state[0]
        self._derived = []
        self._compute = None
        self._op = state[1] if len(state) > 1 else None
        self._sources = state[2] if len(state) > 2 else ()
        self.on_change = []
        if self._op is not None:
            fn = Bool._OPS[self._op]
            self._compute = lambda: fn(*self._sources)
            for s in self._sources:
                if isinstance(s, Bool):
                    s._derived.append(self)


class LinkableAttribute(object):
    """Descriptor-free attribute alias record.

    The reference installs class descriptors; here :class:`veles.units.Unit`
    consults a per-instance table in ``__getattr__``/``__setattr__``, which
    gives the same by-reference semantics (B.input *is* A.output) without
    mutating classes.
    """

    __slots__ = ("src", "src_attr", "two_way")

    def __init__(self, src, src_attr, two_way=True):
        self.src = src
        self.src_attr = src_attr
        self.two_way = two_way

    def get(self):
        return getattr(self.src, self.src_attr)

    def set(self, value):
        if not self.two_way:
            raise AttributeError(
                "attribute is linked one-way from %s.%s" %
                (self.src, self.src_attr))
        setattr(self.src, self.src_attr, value)
