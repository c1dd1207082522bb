"""Property-based tests (hypothesis) for the pure-math contracts the
kernels and host framework rely on.

The DivM test transliterates ops/hip/api.h ``make_divm``/``divq``
exactly: every im2col/dgrad address decomposition on the GPU uses this
magic-number division, so its exactness for all n < 2^31 is a
load-bearing invariant — violated once, the loaders read the wrong
rows silently.
"""

import os

import numpy
from hypothesis import given, settings, strategies as st


# -- DivM: exact transliteration of ops/hip/api.h ---------------------------

def make_divm(d):
    assert d >= 1
    if (d & (d - 1)) == 0:                 # power of two (incl. 1)
        z = 0
        while (1 << z) < d:
            z += 1
        return (0x80000000, 31 + z)
    lg = d.bit_length() - 1
    shift = 32 + lg
    mul = ((1 << shift) + d - 1) // d      # ceil
    return (mul, shift)


def divq(n, m):
    mul, shift = m
    return ((n * mul) >> shift) & 0xFFFFFFFF


@settings(max_examples=2000, deadline=None)
@given(st.integers(min_value=0, max_value=2**31 - 1),
       st.integers(min_value=1, max_value=2**21))
def test_divm_exact(n, d):
    assert divq(n, make_divm(d)) == n // d


def test_divm_kernel_shapes_exhaustive_tail():
    # the divisors the conv descs actually use (C, KW, OW, OH, K, SY,
    # SX, W, H of the model zoo) at every n near the 2^31 boundary and
    # near multiples of d — the spots a wrong ceil/shift breaks first
    divisors = [1, 2, 3, 4, 5, 7, 8, 11, 13, 27, 32, 55, 96, 224, 227,
                256, 384, 1000, 4096, 9216, 25088]
    for d in divisors:
        m = make_divm(d)
        for base in (0, 1, d - 1, d, d + 1, 123456789, 2**31 - 1):
            for n in range(max(0, base - 2), min(2**31, base + 3)):
                assert divq(n, m) == n // d, (n, d)
        for k in range(1, 50):
            n = (2**31 - 1) // d * d - k
            if n >= 0:
                assert divq(n, m) == n // d, (n, d)


# -- genetics: genes stay inside their Tune ranges --------------------------

@settings(max_examples=100, deadline=None)
@given(st.integers(min_value=0, max_value=2**32 - 1),
       st.floats(min_value=-100, max_value=100),
       st.floats(min_value=0.01, max_value=100))
def test_chromosome_bounds(seed, vmin, width):
    from veles.genetics import Tune, Chromosome
    vmax = vmin + width
    tunes = [("root.x.a", Tune(vmin, vmin, vmax)),
             ("root.x.b", Tune(vmax, vmin, vmax))]
    rng = numpy.random.RandomState(seed)
    c = Chromosome([float(rng.uniform(vmin, vmax)) for _ in tunes])
    for _ in range(5):
        c.mutate(tunes, rng, rate=1.0)
        for g, (_, t) in zip(c.genes, tunes):
            assert t.vmin <= g <= t.vmax
    d = Chromosome.crossover(c, Chromosome([vmin, vmax]), rng)
    for g, (_, t) in zip(d.genes, tunes):
        assert t.vmin <= g <= t.vmax


# -- config: update() deep-merge and to_dict round-trip ---------------------

_leaf = st.one_of(st.integers(-10**6, 10**6),
                  st.floats(allow_nan=False, allow_infinity=False),
                  st.text(max_size=8), st.booleans())
_tree = st.recursive(
    _leaf, lambda inner: st.dictionaries(
        st.text(st.characters(whitelist_categories=("Ll",)),
                min_size=1, max_size=6),
        inner, max_size=4), max_leaves=12)


@settings(max_examples=100, deadline=None)
@given(st.dictionaries(st.text(st.characters(
    whitelist_categories=("Ll",)), min_size=1, max_size=6), _tree,
    max_size=4))
def test_config_update_roundtrip(tree):
    from veles.config import Config
    cfg = Config("t")
    cfg.update(tree)
    out = cfg.to_dict()
    assert out == tree
    cfg.update(tree)               # idempotent re-merge
    assert cfg.to_dict() == tree


# -- binary loader: idx round-trip ------------------------------------------

@settings(max_examples=50, deadline=None)
@given(st.sampled_from(["uint8", "int8", "int16", "int32", "float32",
                        "float64"]),
       st.lists(st.integers(1, 5), min_size=1, max_size=3),
       st.booleans())
def test_idx_roundtrip(dtype, shape, gz):
    import tempfile
    from veles.loader.binary import read_idx, write_idx
    rng = numpy.random.RandomState(3)
    arr = (rng.uniform(0, 100, shape)).astype(numpy.dtype(dtype))
    with tempfile.TemporaryDirectory() as d:
        path = os.path.join(d, "a.idx.gz" if gz else "a.idx")
        write_idx(path, arr)
        back = read_idx(path)
    assert back.dtype == arr.dtype and back.shape == arr.shape
    numpy.testing.assert_array_equal(back, arr)


# -- memory: roundup ---------------------------------------------------------

@settings(max_examples=200, deadline=None)
@given(st.integers(0, 10**9), st.integers(1, 4096))
def test_roundup_properties(n, align):
    from veles.memory import roundup
    r = roundup(n, align)
    assert r >= n and r % align == 0 and r - n < align


# -- mzserve f2bf: RNE float->bf16 must match torch's host convert ----------

def _f2bf(f):
    """Transliteration of serve/mzserve.cpp f2bf (round-to-nearest-even)."""
    import struct
    u = struct.unpack("<I", struct.pack("<f", f))[0]
    if (u & 0x7F800000) == 0x7F800000:     # inf/nan: truncate
        return (u >> 16) & 0xFFFF
    u = (u + 0x7FFF + ((u >> 16) & 1)) & 0xFFFFFFFF
    return (u >> 16) & 0xFFFF


@settings(max_examples=500, deadline=None)
@given(st.floats(width=32, allow_nan=False))
def test_f2bf_matches_torch(f):
    import torch
    want = torch.tensor([f], dtype=torch.float32).to(
        torch.bfloat16).view(torch.int16).item() & 0xFFFF
    assert _f2bf(f) == want, f


def test_f2bf_specials():
    import torch
    for f in (0.0, -0.0, 1.0, -1.0, 3.0e38, -3.0e38, 1e-38, 6e-39):
        want = torch.tensor([f], dtype=torch.float32).to(
            torch.bfloat16).view(torch.int16).item() & 0xFFFF
        assert _f2bf(f) == want, f
    assert _f2bf(float("inf")) == 0x7F80
    assert _f2bf(float("-inf")) == 0xFF80
