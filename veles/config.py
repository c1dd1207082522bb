"""Global configuration tree.

Parity target: reference ``veles/config.py`` (class ``Config``, global
``root``): attribute auto-vivification, ``update()`` deep-merge,
pretty-printing.  Sample ``*_config.py`` files mutate ``root.<sample>.*``
and units read ``root.common.*`` for directories / precision / engine
settings.
"""

import os
import pprint


class Config(object):
    """A tree node with attribute auto-vivification.

    ``root.foo.bar.baz = 1`` creates intermediate Config nodes on the fly.
    Reading a non-existent leaf also auto-vivifies (matching the reference
    behavior where configs are built incrementally); use ``in`` or
    :meth:`defined` to test for presence without creating nodes.
    """

    def __init__(self, path="", **kwargs):
        # NOTE: use object.__setattr__ to dodge our own __setattr__.
        object.__setattr__(self, "_path", path)
        object.__setattr__(self, "_children", {})
        for k, v in kwargs.items():
            setattr(self, k, v)

    # -- attribute protocol ------------------------------------------------
    def __getattr__(self, name):
        if name.startswith("_"):
            raise AttributeError(name)
        children = object.__getattribute__(self, "_children")
        if name not in children:
            children[name] = Config(self._subpath(name))
        return children[name]

    def __setattr__(self, name, value):
        if name.startswith("_"):
            object.__setattr__(self, name, value)
            return
        if isinstance(value, dict):
            node = Config(self._subpath(name))
            node.update(value)
            value = node
        self._children[name] = value

    def __delattr__(self, name):
        self._children.pop(name, None)

    def _subpath(self, name):
        return "%s.%s" % (self._path, name) if self._path else name

    # -- mapping-ish helpers ----------------------------------------------
    def __contains__(self, name):
        return name in self._children

    def __iter__(self):
        return iter(self._children.items())

    def __len__(self):
        return len(self._children)

    def defined(self, name):
        """True if *name* exists and is not an empty auto-vivified node."""
        v = self._children.get(name)
        if v is None:
            return False
        if isinstance(v, Config) and len(v._children) == 0:
            return False
        return True

    def get(self, name, default=None):
        """Return leaf value or *default* without auto-vivifying."""
        if not self.defined(name):
            return default
        return self._children[name]

    def update(self, tree):
        """Deep-merge a dict (or another Config) into this node."""
        if isinstance(tree, Config):
            tree = tree.to_dict()
        if not isinstance(tree, dict):
            raise TypeError("Config.update expects a dict, got %r" % (tree,))
        for k, v in tree.items():
            if isinstance(v, dict):
                node = self._children.get(k)
                if not isinstance(node, Config):
                    node = Config(self._subpath(k))
                    self._children[k] = node
                node.update(v)
            else:
                setattr(self, k, v)
        return self

    def to_dict(self):
        out = {}
        for k, v in self._children.items():
            out[k] = v.to_dict() if isinstance(v, Config) else v
        return out

    def print_(self):
        pprint.pprint(self.to_dict())

    def __repr__(self):
        return "Config(%s: %s)" % (self._path or "root",
                                   pprint.pformat(self.to_dict(), width=100))

    # pickling: keep the plain tree
    def __getstate__(self):
        return {"_path": self._path, "_tree": self.to_dict()}

    def __setstate__(self, state):
        object.__setattr__(self, "_path", state["_path"])
        object.__setattr__(self, "_children", {})
        self.update(state["_tree"])


#: The global configuration tree (reference: ``veles.config.root``).
root = Config()

# Defaults mirroring the reference's root.common layout, adapted to this
# environment (no $HOME datasets; everything under the repo or /tmp).
_base = os.environ.get("VELES_DIR", os.path.join(os.path.expanduser("~"),
                                                 ".veles"))
root.common.update({
    "dirs": {
        "veles": os.path.dirname(os.path.abspath(__file__)),
        "base": _base,
        "cache": os.path.join(_base, "cache"),
        "snapshots": os.path.join(_base, "snapshots"),
        "datasets": os.path.join(_base, "datasets"),
    },
    # compute precision of the *reference/numpy* path; the HIP path is
    # bf16-compute/fp32-accumulate with fp32 master weights.
    "precision_type": "float",      # "float" | "double"
    "precision_level": 0,
    "engine": {
        "backend": "auto",          # "auto" | "rocm" | "numpy"
    },
    "trace": {
        "run_time": False,
    },
})


def get(cfg_leaf, default=None):
    """Reference-style helper: return leaf value if set, else default."""
    if isinstance(cfg_leaf, Config):
        if len(cfg_leaf._children) == 0:
            return default
        return cfg_leaf
    return cfg_leaf if cfg_leaf is not None else default


def dtype_map():
    import numpy
    return {"float": numpy.float32, "double": numpy.float64}


def precision_dtype():
    return dtype_map()[root.common.get("precision_type", "float")]
