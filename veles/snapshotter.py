"""Snapshotter — whole-workflow pickle checkpointing.

Parity target: reference ``veles/snapshotter.py`` (``SnapshotterBase``,
``SnapshotterToFile``): gated on ``Decision.improved``; ``interval`` /
``time_interval`` periodic saves; compression codecs "", "gz", "bz2",
"xz"; filenames stamped with the current metric suffix; classmethod
``import_`` to load.

Snapshot format contract (reference §3.4): a pickled object graph of
Workflow/Unit/Array where ``Array.__getstate__`` yields host numpy only
(device state stripped) — resume = unpickle + re-``initialize(device)``.
"""

import bz2
import gzip
import lzma
import os
import pickle
import time

from veles.config import root
from veles.mutable import Bool
from veles.units import Unit

_CODECS = {
    "": (open, ""),
    None: (open, ""),
    "gz": (gzip.open, ".gz"),
    "bz2": (bz2.open, ".bz2"),
    "xz": (lzma.open, ".xz"),
}


class SnapshotterBase(Unit):
    def __init__(self, workflow, **kwargs):
        super(SnapshotterBase, self).__init__(workflow, **kwargs)
        self.prefix = kwargs.get("prefix", "snapshot")
        self.directory = kwargs.get(
            "directory", root.common.dirs.get(
                "snapshots", "/tmp/veles_snapshots"))
        self.compression = kwargs.get("compression", "gz")
        self.interval = kwargs.get("interval", 1)      # epochs between saves
        self.time_interval = kwargs.get("time_interval", 15)  # seconds min
        self.skip = Bool(False)
        self.suffix = None
        self.destination = None     # last written file
        self._last_time = 0.0
        self._counter = 0

    def initialize(self, **kwargs):
        super(SnapshotterBase, self).initialize(**kwargs)
        os.makedirs(self.directory, exist_ok=True)
        self._last_time = time.time()

    def run(self):
        if self.skip:
            return
        self._counter += 1
        if self._counter % max(1, self.interval) != 0:
            return
        if time.time() - self._last_time < self.time_interval:
            return
        self.export()
        self._last_time = time.time()

    def export(self):
        raise NotImplementedError

    @staticmethod
    def import_(file_name):
        """Load a snapshot -> the pickled object (usually a Workflow)."""
        openers = {".gz": gzip.open, ".bz2": bz2.open, ".xz": lzma.open}
        opener = openers.get(os.path.splitext(file_name)[1], open)
        with opener(file_name, "rb") as f:
            return pickle.load(f)


class SnapshotterToFile(SnapshotterBase):
    """Pickle the whole workflow object graph to a compressed file."""

    WRITE_PROTOCOL = 4

    def export(self):
        wf = self.workflow
        opener, ext = _CODECS[self.compression]
        suffix = ("_" + str(self.suffix)) if self.suffix is not None else ""
        fname = "%s%s.%d.pickle%s" % (self.prefix, suffix, self._counter,
                                      ext)
        path = os.path.join(self.directory, fname)
        tmp = path + ".tmp"
        try:
            with opener(tmp, "wb") as f:
                pickle.dump(wf, f, protocol=self.WRITE_PROTOCOL)
        except BaseException:
            try:
                os.remove(tmp)           # never leave a truncated file
            except OSError:
                pass
            raise
        os.replace(tmp, path)
        self.destination = path
        self.info("snapshotted to %s", path)
        return path


# Reference-compatible default name
Snapshotter = SnapshotterToFile
