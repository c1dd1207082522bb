"""Import a Python module by file path.

Parity target: reference ``veles/import_file.py`` — the CLI imports the
workflow module given on the command line by path, not by package name.
"""

import importlib.util
import os
import sys


def import_file_as_module(path, name=None):
    path = os.path.abspath(path)
    if name is None:
        name = os.path.splitext(os.path.basename(path))[0]
    spec = importlib.util.spec_from_file_location(name, path)
    if spec is None:
        raise ImportError("cannot import %s" % path)
    mod = importlib.util.module_from_spec(spec)
    sys.modules[name] = mod
    spec.loader.exec_module(mod)
    return mod


def try_to_import_file(path):
    try:
        return import_file_as_module(path)
    except Exception as e:  # reference returns the error object
        return e
