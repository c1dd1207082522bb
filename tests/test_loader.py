"""Loader minibatch/epoch accounting and sharding."""

import numpy

from veles.dummy import DummyWorkflow
from veles.loader.base import TRAIN, VALID, TEST
from veles.znicz_amd.models.synthetic import SyntheticClassLoader


def make_loader(mb=10, n_train=40, n_valid=20, world=1, rank=0):
    wf = DummyWorkflow()
    ld = SyntheticClassLoader(wf, sample_shape=(8,), n_classes=4,
                              n_train=n_train, n_valid=n_valid,
                              minibatch_size=mb)
    if world > 1:
        ld.shard(rank, world)
    ld.initialize(device=None)
    return ld


def test_epoch_walkthrough():
    ld = make_loader()
    seen = {TEST: 0, VALID: 0, TRAIN: 0}
    ended = 0
    for _ in range(6):   # 2 valid + 4 train minibatches
        ld.run()
        seen[ld.minibatch_class] += ld.minibatch_size
    assert seen[VALID] == 20 and seen[TRAIN] == 40
    assert bool(ld.epoch_ended)
    assert ld.epoch_number == 1


def test_last_minibatch_flags():
    ld = make_loader(mb=15, n_train=30, n_valid=15)
    flags = []
    for _ in range(3):
        ld.run()
        flags.append((ld.minibatch_class, bool(ld.last_minibatch),
                      bool(ld.epoch_ended)))
    assert flags == [(VALID, True, False), (TRAIN, False, False),
                     (TRAIN, True, True)]


def test_train_shuffles_between_epochs():
    ld = make_loader(mb=40, n_train=40, n_valid=0)
    ld.run()
    first = ld.minibatch_indices.mem.copy()
    ld.run()
    second = ld.minibatch_indices.mem.copy()
    assert not numpy.array_equal(first, second)
    assert set(first) == set(second)


def test_sharding_disjoint_and_aligned():
    lds = [make_loader(mb=10, n_train=40, n_valid=20, world=2, rank=r)
           for r in (0, 1)]
    assert lds[0].class_lengths == [0, 10, 20]
    served = [[], []]
    for r, ld in enumerate(lds):
        for _ in range(3):
            ld.run()
            served[r].extend(ld.minibatch_indices.mem[
                :ld.minibatch_size].tolist())
        assert bool(ld.epoch_ended)
    assert not (set(served[0]) & set(served[1]))


def test_short_minibatch_padding():
    ld = make_loader(mb=16, n_train=40, n_valid=20)
    ld.run()  # valid: 16
    ld.run()  # valid: 4 (short)
    assert ld.minibatch_size == 4
    assert (ld.minibatch_labels.mem[4:] == -1).all()
    assert (ld.minibatch_data.mem[4:] == 0).all()


def test_sharding_non_divisible_world():
    """world=3 over lengths not divisible by 3: the contract is EQUAL
    per-rank stripes (ln // world per class — the remainder samples are
    dropped that epoch) so epoch boundaries stay aligned across ranks;
    what IS served must be disjoint and duplicate-free."""
    lds = [make_loader(mb=9, n_train=40, n_valid=20, world=3, rank=r)
           for r in (0, 1, 2)]
    served = [[] for _ in range(3)]
    for r, ld in enumerate(lds):
        # equal truncated stripes on every rank
        assert ld.class_lengths == [0, 20 // 3, 40 // 3]
        guard = 0
        while True:
            ld.run()
            served[r].extend(ld.minibatch_indices.mem[
                :ld.minibatch_size].tolist())
            guard += 1
            assert guard < 40
            if bool(ld.epoch_ended):
                break
        # all ranks finish their epoch after the same number of
        # steps: ceil(6/9) VALID + ceil(13/9) TRAIN minibatches
        assert guard == 3
    flat = sum(served, [])
    for a in range(3):
        for b in range(a + 1, 3):
            assert not (set(served[a]) & set(served[b])), (a, b)
    # each rank served exactly its stripe, no duplicates; the
    # remainder (60 - 3*(6+13) = 3 samples) sits out this epoch
    assert len(flat) == len(set(flat)) == 3 * (6 + 13)
