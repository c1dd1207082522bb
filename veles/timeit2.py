"""Timing helpers (parity: reference ``veles/timeit2.py``)."""

import time
from functools import wraps


def timeit(fn, *args, **kwargs):
    """Run fn(*args, **kwargs); return (result, seconds)."""
    t0 = time.perf_counter()
    res = fn(*args, **kwargs)
    return res, time.perf_counter() - t0


def timed(fn):
    """Decorator printing the call duration via the object's logger
    (or stdout)."""
    @wraps(fn)
    def wrapper(*args, **kwargs):
        res, dt = timeit(fn, *args, **kwargs)
        obj = args[0] if args else None
        msg = "%s took %.3f s" % (fn.__name__, dt)
        if hasattr(obj, "debug"):
            obj.debug(msg)
        else:
            print(msg)
        return res
    return wrapper
