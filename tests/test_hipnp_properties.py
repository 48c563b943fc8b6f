"""Property-based parity: random programs of routed numpy operations on
a DeviceArray (fake numpy backend) must produce numpy's own values.
Complements the example-based dispatch tests by searching the edge
space (shapes, axes, NaNs, duplicates) automatically."""

import sys
from pathlib import Path

import numpy as np
import pytest
from hypothesis import given, settings, strategies as st

OPS_DIR = Path(__file__).resolve().parent.parent / "code_interpreter_amd" / "ops"
sys.path.insert(0, str(OPS_DIR))

import hipnp  # noqa: E402

from test_hipnp_dispatch import FakeBackend  # noqa: E402


@pytest.fixture
def fake(monkeypatch):
    backend = FakeBackend()
    monkeypatch.setitem(hipnp._state, "backend", backend)
    monkeypatch.setitem(hipnp._state, "failed", None)
    return backend


def _dev(fake, arr):
    arr = np.ascontiguousarray(arr)
    h = fake._new(arr.copy())
    return hipnp.DeviceArray(h, arr.shape, arr.dtype)


arrays_1d = st.builds(
    lambda seed, n, with_nan: _mk(seed, (n,), with_nan),
    st.integers(0, 2**31 - 1), st.integers(1, 400), st.booleans(),
)
arrays_2d = st.builds(
    lambda seed, r, c, with_nan: _mk(seed, (r, c), with_nan),
    st.integers(0, 2**31 - 1), st.integers(1, 40), st.integers(1, 40),
    st.booleans(),
)


def _mk(seed, shape, with_nan):
    rng = np.random.default_rng(seed)
    # duplicate-heavy integers exercise sort stability and ties
    a = rng.integers(-5, 6, shape).astype(np.float64)
    if with_nan:
        m = rng.random(shape) < 0.15
        a[m] = np.nan
    return a


@settings(max_examples=60, deadline=None)
@given(a=arrays_1d)
def test_sort_argsort_parity(a):
    fake = FakeBackend()
    hipnp._state["backend"] = fake
    try:
        np.testing.assert_array_equal(
            np.asarray(np.sort(_dev(fake, a))), np.sort(a))
        np.testing.assert_array_equal(
            np.asarray(np.argsort(_dev(fake, a))),
            np.argsort(a, kind="stable"))
        u = np.unique(_dev(fake, a))
        np.testing.assert_array_equal(np.asarray(u), np.unique(a))
    finally:
        hipnp._state["backend"] = None


@settings(max_examples=60, deadline=None)
@given(a=arrays_2d, axis=st.sampled_from([0, 1, -1]))
def test_axis_sort_cumsum_parity(a, axis):
    fake = FakeBackend()
    hipnp._state["backend"] = fake
    try:
        np.testing.assert_array_equal(
            np.asarray(np.sort(_dev(fake, a), axis=axis)),
            np.sort(a, axis=axis))
        np.testing.assert_allclose(
            np.asarray(np.cumsum(_dev(fake, a), axis=axis)),
            np.cumsum(a, axis=axis), rtol=1e-12)
    finally:
        hipnp._state["backend"] = None


@settings(max_examples=60, deadline=None)
@given(a=arrays_1d, q=st.floats(0.0, 1.0))
def test_quantile_parity(a, q):
    fake = FakeBackend()
    hipnp._state["backend"] = fake
    try:
        got = np.quantile(_dev(fake, a), q)
        ref = np.quantile(a, q)
        if np.isnan(ref):
            assert np.isnan(got)
        else:
            assert got == pytest.approx(ref, abs=1e-12)
    finally:
        hipnp._state["backend"] = None


@settings(max_examples=40, deadline=None)
@given(a=arrays_2d)
def test_histogram_and_mask_parity(a):
    fake = FakeBackend()
    hipnp._state["backend"] = fake
    try:
        if np.isnan(a).any():
            # numpy raises on NaN autorange; the device route must too
            with pytest.raises(ValueError):
                np.histogram(_dev(fake, a), bins=7)
        else:
            hist, edges = np.histogram(_dev(fake, a), bins=7)
            rh, re = np.histogram(a, bins=7)
            np.testing.assert_array_equal(hist, rh)
            np.testing.assert_allclose(edges, re, rtol=0)
        with np.errstate(invalid="ignore"):
            m = _dev(fake, a) > 0
            ref = a > 0
            np.testing.assert_array_equal(np.asarray(m.sum(axis=1)),
                                          ref.sum(axis=1))
            assert int(m.sum()) == int(ref.sum())
    finally:
        hipnp._state["backend"] = None


@settings(max_examples=50, deadline=None)
@given(a=arrays_1d, b=arrays_1d, side=st.sampled_from(["left", "right"]))
def test_searchsorted_digitize_parity(a, b, side):
    fake = FakeBackend()
    hipnp._state["backend"] = fake
    try:
        srt = np.sort(a)
        got = np.searchsorted(_dev(fake, srt), _dev(fake, b), side=side)
        np.testing.assert_array_equal(
            np.asarray(got), np.searchsorted(srt, b, side=side))
        bins = np.unique(a[~np.isnan(a)])
        if bins.size:
            got2 = np.digitize(_dev(fake, b), bins)
            np.testing.assert_array_equal(
                np.asarray(got2), np.digitize(b, bins))
    finally:
        hipnp._state["backend"] = None


@settings(max_examples=50, deadline=None)
@given(a=arrays_2d, b=arrays_2d)
def test_concat_reshape_parity(a, b):
    fake = FakeBackend()
    hipnp._state["backend"] = fake
    try:
        if a.shape[1] == b.shape[1]:
            got = np.vstack([_dev(fake, a), _dev(fake, b)])
            np.testing.assert_array_equal(
                np.asarray(got), np.vstack([a, b]))
        flat = _dev(fake, a).ravel()
        np.testing.assert_array_equal(np.asarray(flat), a.ravel())
        r = _dev(fake, a).reshape(-1, a.shape[0])
        np.testing.assert_array_equal(
            np.asarray(r), a.reshape(-1, a.shape[0]))
    finally:
        hipnp._state["backend"] = None


@settings(max_examples=50, deadline=None)
@given(a=arrays_1d)
def test_nan_reduction_parity(a):
    fake = FakeBackend()
    hipnp._state["backend"] = fake
    try:
        import warnings
        with warnings.catch_warnings(), np.errstate(all="ignore"):
            warnings.simplefilter("ignore")
            for dev_f, ref_f in (
                (np.nansum, np.nansum), (np.nanmean, np.nanmean),
                (np.nanmedian, np.nanmedian),
            ):
                got = dev_f(_dev(fake, a))
                ref = ref_f(a)
                if np.isnan(ref):
                    assert np.isnan(got)
                else:
                    assert float(got) == pytest.approx(ref, abs=1e-10)
    finally:
        hipnp._state["backend"] = None


@settings(max_examples=50, deadline=None)
@given(a=arrays_1d, rtol=st.sampled_from([1e-5, 1e-9, 0.5]))
def test_isclose_allclose_parity(a, rtol):
    fake = FakeBackend()
    hipnp._state["backend"] = fake
    try:
        b = a + np.where(np.isnan(a), 0.0, 1e-7)
        with np.errstate(invalid="ignore"):
            got = np.isclose(_dev(fake, a), _dev(fake, b), rtol=rtol)
            ref = np.isclose(a, b, rtol=rtol)
            np.testing.assert_array_equal(np.asarray(got), ref)
            assert bool(np.allclose(_dev(fake, a), _dev(fake, b),
                                    rtol=rtol)) == bool(
                np.allclose(a, b, rtol=rtol))
    finally:
        hipnp._state["backend"] = None


@settings(max_examples=40, deadline=None)
@given(a=arrays_2d)
def test_median_quantile_axis_parity(a):
    fake = FakeBackend()
    hipnp._state["backend"] = fake
    try:
        import warnings
        with warnings.catch_warnings(), np.errstate(all="ignore"):
            warnings.simplefilter("ignore")
            for axis in (0, 1):
                got = np.median(_dev(fake, a), axis=axis)
                ref = np.median(a, axis=axis)
                nans = np.isnan(ref)
                assert np.array_equal(np.isnan(np.asarray(got)), nans)
                np.testing.assert_allclose(
                    np.asarray(got)[~nans], ref[~nans], rtol=1e-12)
    finally:
        hipnp._state["backend"] = None


@settings(max_examples=40, deadline=None)
@given(
    seed=st.integers(0, 2**31 - 1),
    m=st.integers(1, 24), k=st.integers(1, 24), n=st.integers(1, 24),
)
def test_matmul_shapes_parity(seed, m, k, n):
    fake = FakeBackend()
    hipnp._state["backend"] = fake
    try:
        rng = np.random.default_rng(seed)
        A = rng.standard_normal((m, k))
        B = rng.standard_normal((k, n))
        x = rng.standard_normal(k)
        got = np.asarray(_dev(fake, A) @ _dev(fake, B))
        np.testing.assert_allclose(got, A @ B, rtol=1e-10, atol=1e-12)
        got_mv = np.asarray(_dev(fake, A) @ _dev(fake, x))
        np.testing.assert_allclose(got_mv, A @ x, rtol=1e-10, atol=1e-12)
        got_vm = np.asarray(_dev(fake, x) @ _dev(fake, B))
        np.testing.assert_allclose(got_vm, x @ B, rtol=1e-10, atol=1e-12)
        got_e = np.asarray(np.einsum("ij,jk->ik", _dev(fake, A),
                                     _dev(fake, B)))
        np.testing.assert_allclose(got_e, A @ B, rtol=1e-10, atol=1e-12)
    finally:
        hipnp._state["backend"] = None
