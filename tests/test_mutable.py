"""Bool/LinkableAttribute semantics (parity: veles/tests test_mutable)."""

from veles.mutable import Bool


def test_bool_basic():
    b = Bool()
    assert not b
    b <<= True
    assert b
    b.unset()
    assert not b


def test_bool_derived_live():
    a, b = Bool(False), Bool(True)
    c = a & b
    d = a | b
    n = ~a
    assert not c and d and n
    a <<= True
    assert c and d and not n
    b <<= False
    assert not c and d


def test_bool_callbacks():
    a = Bool(False)
    fired = []
    a.on_change.append(lambda x: fired.append(bool(x)))
    a <<= True
    a <<= True  # no change, no fire
    a <<= False
    assert fired == [True, False]


def test_bool_shared_object_gate_idiom():
    complete = Bool(False)
    gate_block = complete          # shared object, the veles idiom
    assert not gate_block
    complete <<= True
    assert gate_block


def test_bool_pickle():
    import pickle
    a = Bool(True)
    b = pickle.loads(pickle.dumps(a))
    assert bool(b) is True
