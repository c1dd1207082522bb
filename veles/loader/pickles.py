"""Pickled-dataset loader + minibatch saver.

Parity target: reference ``veles/loader/pickles.py`` (``PicklesLoader``)
and ``veles/loader/saver.py`` (``MinibatchesSaver``).
"""

import gzip
import pickle

import numpy

from veles.loader.fullbatch import FullBatchLoader
from veles.units import Unit


def _open(path, mode="rb"):
    return gzip.open(path, mode) if path.endswith(".gz") else \
        open(path, mode)


class PicklesLoader(FullBatchLoader):
    """Loads (data, labels) arrays from one pickle per class set:
    ``test_path`` / ``validation_path`` / ``train_path`` each point to a
    pickle of (ndarray[N, ...], ndarray[N]) or a dict with "data"/
    "labels" keys."""

    def __init__(self, workflow, **kwargs):
        super(PicklesLoader, self).__init__(workflow, **kwargs)
        self.test_path = kwargs.get("test_path", None)
        self.validation_path = kwargs.get("validation_path", None)
        self.train_path = kwargs.get("train_path", None)

    @staticmethod
    def _read(path):
        if path is None:
            return None, None
        with _open(path) as f:
            obj = pickle.load(f)
        if isinstance(obj, dict):
            return (numpy.asarray(obj["data"], numpy.float32),
                    numpy.asarray(obj["labels"], numpy.int32))
        data, labels = obj
        return (numpy.asarray(data, numpy.float32),
                numpy.asarray(labels, numpy.int32))

    def load_data(self):
        parts = [self._read(p) for p in (self.test_path,
                                         self.validation_path,
                                         self.train_path)]
        datas = [d for d, _ in parts if d is not None]
        labels = [l for _, l in parts if l is not None]
        if not datas:
            raise ValueError("PicklesLoader: no dataset paths given")
        self.original_data.mem = numpy.concatenate(datas)
        self.original_labels.mem = numpy.concatenate(labels)
        self.class_lengths = [0 if d is None else len(d)
                              for d, _ in parts]


class MinibatchesSaver(Unit):
    """Dumps every served minibatch to one growing pickle stream
    (reference veles/loader/saver.py) — for dataset capture/replay."""

    def __init__(self, workflow, **kwargs):
        super(MinibatchesSaver, self).__init__(workflow, **kwargs)
        self.file_name = kwargs.get("file_name", "minibatches.pickle.gz")
        self.input = None               # minibatch_data
        self.labels = None              # minibatch_labels
        self.minibatch_size = 0
        self._file = None

    def initialize(self, **kwargs):
        super(MinibatchesSaver, self).initialize(**kwargs)
        self._file = gzip.open(self.file_name, "wb") if \
            self.file_name.endswith(".gz") else open(self.file_name, "wb")

    def run(self):
        self.input.map_read()
        self.labels.map_read()
        n = int(self.minibatch_size)
        pickle.dump((self.input.mem[:n].copy(),
                     self.labels.mem[:n].copy()), self._file,
                    protocol=4)

    def close(self):
        if self._file is not None:
            self._file.close()
            self._file = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass
