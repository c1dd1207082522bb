#!/usr/bin/env python3
"""Reference-compatible CLI entry (parity: ``scripts/velescli.py``).

The reference installed a ``velescli.py`` console script alongside
``python -m veles``; both shared one ``Main``.  Same here — this shim
delegates to :mod:`veles.__main__` so either invocation works:

    python -m veles             workflow.py config.py ...
    python veles/scripts/velescli.py workflow.py config.py ...
"""

import os
import sys

# direct execution puts scripts/ on sys.path; the repo root (two
# levels up) is what holds the veles package
sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.dirname(os.path.abspath(__file__)))))


def main(argv=None):
    from veles.__main__ import main as _main
    return _main(argv)


if __name__ == "__main__":
    sys.exit(main())
