"""Loader — the minibatch engine.

Parity target: reference ``veles/loader/base.py`` (``Loader``, ``ILoader``,
TEST/VALID/TRAIN sets, minibatch state, shuffling, epoch accounting).

MI355X redesign: the master-side index distribution of the reference's
distributed mode becomes *sharding* — with one process per GPU each rank
serves a disjoint stripe of every global minibatch (``shard(rank,
world)``), keeping all ranks' epoch/shuffle state identical (same seed)
so the loop control flow is replicated and only gradients need RCCL.
"""

import numpy

from veles import prng
from veles.memory import Array
from veles.mutable import Bool
from veles.units import Unit

TEST = 0
VALID = 1
TRAIN = 2
CLASS_NAME = {TEST: "test", VALID: "validation", TRAIN: "train"}


class ILoader(object):
    """Marker (parity)."""


class Loader(Unit):
    """Serves minibatches from three sample sets."""

    def __init__(self, workflow, **kwargs):
        super(Loader, self).__init__(workflow, **kwargs)
        self.max_minibatch_size = kwargs.get("minibatch_size", 100)
        self.shuffle_limit = kwargs.get("shuffle_limit", numpy.iinfo(
            numpy.int64).max)
        self.prng = kwargs.get("prng", prng.get("loader"))
        #: [n_test, n_valid, n_train]
        self.class_lengths = [0, 0, 0]
        # minibatch state (read by downstream units via link_attrs)
        self.minibatch_data = Array()
        self.minibatch_labels = Array()
        self.minibatch_indices = Array()
        self.minibatch_class = TRAIN
        self.minibatch_size = 0
        self.minibatch_offset = 0
        self.epoch_number = 0
        self.epoch_ended = Bool(False)
        self.last_minibatch = Bool(False)
        self.train_ended = Bool(False)
        self.samples_served = 0
        # DP sharding (identity by default)
        self.rank = 0
        self.world = 1
        self._shuffled_indices = None
        self._class_order = ()
        self._pos_in_class = 0
        self._cur_class_i = 0
        self.device = None

    # -- overridables -------------------------------------------------------
    def load_data(self):
        """Set class_lengths and materialize the dataset."""
        raise NotImplementedError

    def create_minibatch_data(self):
        """Allocate minibatch_data (shape [max_minibatch_size, ...])."""
        raise NotImplementedError

    def fill_minibatch(self):
        """Fill minibatch_data/labels for indices in minibatch_indices."""
        raise NotImplementedError

    # -- helpers ------------------------------------------------------------
    @property
    def total_samples(self):
        return int(sum(self.class_lengths))

    @property
    def class_ends(self):
        ends, acc = [], 0
        for ln in self.class_lengths:
            acc += ln
            ends.append(acc)
        return ends

    def class_offset(self, clazz):
        return int(sum(self.class_lengths[:clazz]))

    def shard(self, rank, world):
        """Serve only stripe ``rank`` of every global minibatch."""
        self.rank, self.world = int(rank), int(world)
        return self

    @property
    def effective_minibatch_size(self):
        """Per-rank samples in the current minibatch."""
        return self.minibatch_size

    # -- lifecycle ----------------------------------------------------------
    def initialize(self, device=None, **kwargs):
        super(Loader, self).initialize(**kwargs)
        self.device = device
        if self.total_samples == 0:
            self.load_data()
        if self.total_samples == 0:
            raise ValueError("load_data() produced an empty dataset")
        self.create_minibatch_data()
        if not self.minibatch_labels:
            self.minibatch_labels.mem = numpy.zeros(
                self.max_minibatch_size, dtype=numpy.int32)
        self.minibatch_indices.mem = numpy.zeros(
            self.max_minibatch_size, dtype=numpy.int32)
        self._shuffled_indices = numpy.arange(
            self.total_samples, dtype=numpy.int32)
        if self.world > 1:
            self._apply_shard()
        self._class_order = tuple(
            c for c in (TEST, VALID, TRAIN) if self.class_lengths[c] > 0)
        self._reset_epoch(first=True)
        for arr in (self.minibatch_data, self.minibatch_labels,
                    self.minibatch_indices):
            arr.initialize(device)

    def _apply_shard(self):
        """Each rank keeps a disjoint 1/world stripe of every class
        (equal lengths across ranks -> aligned epoch boundaries)."""
        segs, new_lengths = [], []
        off = 0
        for ln in self.class_lengths:
            seg = self._shuffled_indices[off:off + ln]
            off += ln
            keep = ln // self.world
            if ln and keep == 0:
                raise ValueError(
                    "class with %d samples cannot be sharded %d ways" %
                    (ln, self.world))
            segs.append(seg[self.rank::self.world][:keep])
            new_lengths.append(keep)
        self._shuffled_indices = numpy.concatenate(segs) if segs else \
            self._shuffled_indices
        self.class_lengths = new_lengths

    def _reset_epoch(self, first=False):
        self._cur_class_i = 0
        self._pos_in_class = 0
        if not first:
            self.epoch_number += 1
        self.shuffle()

    def shuffle(self):
        """Shuffle the TRAIN portion only (test/valid order is fixed)."""
        if self.epoch_number >= self.shuffle_limit:
            return
        off = self.class_offset(TRAIN)
        n = self.class_lengths[TRAIN]
        if n > 1:
            seg = self._shuffled_indices[off:off + n]
            self.prng.shuffle(seg)

    # -- serving ------------------------------------------------------------
    def run(self):
        self.epoch_ended <<= False
        self.last_minibatch <<= False
        clazz = self._class_order[self._cur_class_i]
        length = self.class_lengths[clazz]
        offset = self.class_offset(clazz) + self._pos_in_class
        size = min(self.max_minibatch_size, length - self._pos_in_class)
        self.minibatch_class = clazz
        self.minibatch_offset = offset
        self.minibatch_size = size
        idx = self._shuffled_indices[offset:offset + size]
        self.minibatch_indices.map_invalidate()
        self.minibatch_indices.mem[:size] = idx
        if size < self.max_minibatch_size:
            self.minibatch_indices.mem[size:] = -1
        self.fill_minibatch()
        if size < self.max_minibatch_size:
            self._pad_minibatch(size)
        for arr in (self.minibatch_data, self.minibatch_labels,
                    self.minibatch_indices):
            arr.unmap()
        self.samples_served += size
        # advance
        self._pos_in_class += size
        if self._pos_in_class >= length:
            self.last_minibatch <<= True
            self._cur_class_i += 1
            self._pos_in_class = 0
            if clazz == TRAIN or self._cur_class_i >= len(self._class_order):
                self.epoch_ended <<= True
                self.train_ended <<= clazz == TRAIN
                self._reset_epoch()

    def _pad_minibatch(self, size):
        """Zero the tail of a short minibatch (keeps shapes static)."""
        if self.minibatch_data.mem is not None:
            self.minibatch_data.mem[size:] = 0
        if self.minibatch_labels.mem is not None:
            self.minibatch_labels.mem[size:] = -1

    # reference distributed hooks -> sharding note in module docstring
    def generate_data_for_slave(self, slave=None):
        return None
