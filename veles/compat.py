"""Py2/Py3 compatibility shims (parity: reference ``veles/compat.py``).

The reference supported Python 2; on Python 3 these collapse to
trivial passthroughs, kept so reference-era imports keep working.
"""

from io import BytesIO, StringIO  # noqa: F401


def from_none(exc):
    """raise ... from None helper (reference API)."""
    exc.__cause__ = None
    exc.__suppress_context__ = True
    return exc


try:
    from shutil import which  # noqa: F401
except ImportError:  # pragma: no cover
    which = None

PYPY = False
IS_INTERACTIVE = hasattr(__import__("sys"), "ps1")
