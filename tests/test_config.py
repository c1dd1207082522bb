"""Config tree semantics (parity: veles/tests/test_config.py)."""

import pickle

from veles.config import Config


def test_autovivify_and_leaf():
    c = Config("r")
    c.a.b.c = 42
    assert c.a.b.c == 42
    assert c.defined("a")
    assert not c.defined("zzz")


def test_update_deep_merge():
    c = Config("r")
    c.update({"x": {"y": 1, "z": 2}})
    c.update({"x": {"z": 3}, "w": 4})
    assert c.x.y == 1 and c.x.z == 3 and c.w == 4


def test_to_dict_roundtrip():
    c = Config("r")
    c.update({"a": {"b": [1, 2, 3]}, "c": "s"})
    assert c.to_dict() == {"a": {"b": [1, 2, 3]}, "c": "s"}


def test_pickle():
    c = Config("r")
    c.update({"a": {"b": 7}})
    c2 = pickle.loads(pickle.dumps(c))
    assert c2.a.b == 7


def test_get_default():
    c = Config("r")
    assert c.get("missing", 5) == 5
    c.v = 9
    assert c.get("v") == 9
