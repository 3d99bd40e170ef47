"""Property-based oracle tests (hypothesis): randomized volumes must
uphold the canonical contract's invariants — the test strategy SURVEY §4
recommends beyond the reference's existence-only assertions."""
import collections

import numpy as np
import pytest
from hypothesis import given, settings, strategies as st

import oracle


def volumes(max_dim=14, max_labels=6):
    @st.composite
    def _vol(draw):
        sx = draw(st.integers(2, max_dim))
        sy = draw(st.integers(2, max_dim))
        sz = draw(st.integers(2, max_dim))
        seed = draw(st.integers(0, 2**31 - 1))
        nlab = draw(st.integers(1, max_labels))
        dtype = draw(st.sampled_from([np.uint32, np.uint64]))
        rng = np.random.default_rng(seed)
        data = rng.integers(0, nlab + 1, size=(sx, sy, sz)).astype(dtype)
        return np.asfortranarray(data)
    return _vol()


@settings(max_examples=40, deadline=None)
@given(volumes())
def test_surfaces_closed_for_interior_labels(data):
    """Every label fully inside the volume (zero-padded) yields a CLOSED
    surface: each directed edge balanced by its reverse."""
    padded = np.zeros(tuple(s + 2 for s in data.shape), data.dtype, order="F")
    padded[1:-1, 1:-1, 1:-1] = data
    r = oracle.mesh_chunk(padded, resolution=(2, 3, 5))
    for label, (v, f) in r.items():
        assert f.shape[0] == 0 or f.max() < v.shape[0]
        directed = collections.Counter()
        for tri in f:
            a, b, c = (int(x) for x in tri)
            assert a != b and b != c and a != c
            for u, w in ((a, b), (b, c), (c, a)):
                directed[(u, w)] += 1
        for (u, w), ct in directed.items():
            assert directed[(w, u)] == ct, f"label {label} open"


@settings(max_examples=30, deadline=None)
@given(volumes())
def test_u32_u64_and_determinism(data):
    a = oracle.mesh_chunk(data.astype(np.uint32), resolution=(1, 1, 1))
    b = oracle.mesh_chunk(data.astype(np.uint64), resolution=(1, 1, 1))
    c = oracle.mesh_chunk(data.astype(np.uint32), resolution=(1, 1, 1))
    assert a.keys() == b.keys() == c.keys()
    for k in a:
        assert np.array_equal(a[k][0], b[k][0])
        assert np.array_equal(a[k][1], b[k][1])
        assert np.array_equal(a[k][0], c[k][0])
        assert np.array_equal(a[k][1], c[k][1])


@settings(max_examples=30, deadline=None)
@given(volumes(), st.integers(2, 50))
def test_simplify_valid_and_bounded(data, factor):
    """Simplification keeps meshes valid, never grows them, and respects
    a zero error bound on bounding boxes."""
    full = oracle.mesh_chunk(data, resolution=(4, 4, 40))
    simp = oracle.mesh_chunk(data, resolution=(4, 4, 40),
                             reduction_factor=factor, max_error=1e9)
    assert full.keys() == simp.keys()
    for k in full:
        fv, ff = full[k]
        sv, sf = simp[k]
        assert sf.shape[0] <= ff.shape[0]
        assert sv.shape[0] <= fv.shape[0]
        if sf.shape[0]:
            assert sf.max() < sv.shape[0]


@settings(max_examples=25, deadline=None)
@given(volumes())
def test_vertices_on_half_grid(data):
    res = (2.0, 4.0, 8.0)
    r = oracle.mesh_chunk(data, resolution=res)
    for _, (v, f) in r.items():
        k = v / (0.5 * np.array(res, dtype=np.float32))
        assert np.array_equal(k, np.round(k))
