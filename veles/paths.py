"""Path roots (parity: reference ``veles/paths.py``)."""

import os

#: the veles package directory
__root__ = os.path.dirname(os.path.abspath(__file__))
#: the repository root (one level up)
__project_root__ = os.path.dirname(__root__)


def expand(path):
    return os.path.abspath(os.path.expanduser(os.path.expandvars(path)))
