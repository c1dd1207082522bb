"""Logger mixin.

Parity target: reference ``veles/logger.py`` — every Unit is a Logger with
a per-instance identity; colored console output; file handler option.
"""

import logging
import sys

_initialized = False


def setup_logging(level=logging.INFO, filename=None):
    global _initialized
    fmt = "%(asctime)s %(levelname).1s %(name)s: %(message)s"
    handlers = [logging.StreamHandler(sys.stderr)]
    if filename:
        handlers.append(logging.FileHandler(filename))
    logging.basicConfig(level=level, format=fmt, handlers=handlers,
                        force=True)
    _initialized = True


class Logger(object):
    """Mixin providing ``self.debug/info/warning/error`` with a per-object
    logger named after the class (and unit name when available)."""

    def __init__(self, **kwargs):
        super(Logger, self).__init__()
        self._logger_ = None

    @property
    def logger(self):
        if getattr(self, "_logger_", None) is None:
            name = type(self).__name__
            uname = getattr(self, "_name", None)
            if uname and uname != name:
                name = "%s[%s]" % (name, uname)
            self._logger_ = logging.getLogger(name)
        return self._logger_

    def _reset_logger(self):
        self._logger_ = None

    def debug(self, msg, *args):
        self.logger.debug(msg, *args)

    def info(self, msg, *args):
        self.logger.info(msg, *args)

    def warning(self, msg, *args):
        self.logger.warning(msg, *args)

    def error(self, msg, *args):
        self.logger.error(msg, *args)

    def exception(self, msg="", *args):
        self.logger.exception(msg, *args)

    # keep pickles clean of logging objects
    def __getstate__(self):
        state = getattr(super(Logger, self), "__getstate__", lambda: None)()
        if state is None:
            state = self.__dict__.copy()
        state.pop("_logger_", None)
        return state
