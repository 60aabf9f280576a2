"""C++ group-gather engine (groupby/_gather) vs the pandas oracle.

The engine replaces the Spark JVM hash-shuffle of the reference
(group_apply/02_Fine_Grained_Demand_Forecasting.py:525-528) with a
multithreaded factorize + scatter; pd.factorize(sort=True) is the
semantic contract.
"""
import numpy as np
import pandas as pd
import pytest

from mi355x_scale.groupby.gather import (HAVE_GATHER_EXT, fast_factorize,
                                         panel_from_long)

pytestmark = pytest.mark.skipif(not HAVE_GATHER_EXT,
                                reason="_gather ext not built")


def _check_against_pandas(col):
    codes, uniq = fast_factorize(col)
    ref_codes, ref_uniq = pd.factorize(col, sort=True)
    assert (codes == ref_codes).all()
    assert (np.asarray(uniq) == np.asarray(ref_uniq)).all()


def test_strings_with_duplicates():
    rng = np.random.default_rng(0)
    _check_against_pandas(
        pd.Series(rng.choice([f"K{i:04d}" for i in range(500)], 50_000)))


def test_strings_unsorted_unicode():
    s = pd.Series(["b", "a", "ü", "A", "z", "a", "", "b"])
    _check_against_pandas(s)


def test_arrow_backed_strings():
    rng = np.random.default_rng(1)
    s = pd.Series(rng.choice([f"G{i}" for i in range(100)], 20_000))
    codes_obj, uniq_obj = fast_factorize(s)
    codes_arr, uniq_arr = fast_factorize(s.astype("string[pyarrow]"))
    assert (codes_obj == codes_arr).all()
    assert list(uniq_obj) == list(uniq_arr)


def test_nulls_get_minus_one():
    s = pd.Series(["x", None, "y", "x", None])
    codes, uniq = fast_factorize(s.astype("string[pyarrow]"))
    assert list(codes) == [0, -1, 1, 0, -1]
    assert list(uniq) == ["x", "y"]


def test_int64_and_datetime():
    rng = np.random.default_rng(2)
    _check_against_pandas(pd.Series(rng.integers(-10**15, 10**15, 30_000)))
    _check_against_pandas(pd.Series(
        pd.to_datetime(rng.integers(10**18, 2 * 10**18, 10_000))))


def test_single_and_empty():
    _check_against_pandas(pd.Series(["only"]))
    codes, uniq = fast_factorize(pd.Series([], dtype=object))
    assert len(codes) == 0 and len(uniq) == 0


def test_factorize_nul_bytes_distinct():
    """Strings differing only by embedded/trailing NUL bytes are distinct
    groups here (length-delimited arrow buffers); pandas' C-string hash
    table conflates them — a documented, deliberate divergence."""
    codes, uniq = fast_factorize(pd.Series(["", "\x00", "a", "a\x00"]))
    assert len(set(codes.tolist())) == 4
    assert list(uniq) == ["", "\x00", "a", "a\x00"]


def test_all_unique_growth_path():
    # forces local-table rehash growth (cardinality == rows)
    _check_against_pandas(pd.Series([f"u{i:07d}" for i in range(300_000)]))


def test_panel_multikey_matches_pandas_path():
    rng = np.random.default_rng(3)
    df = pd.DataFrame({
        "P": rng.choice(["prodA", "prodB", "prodC"], 5000),
        "S": rng.choice([f"s{i}" for i in range(40)], 5000),
        "t": rng.integers(0, 25, 5000),
        "v": rng.normal(size=5000).astype(np.float32),
    }).drop_duplicates(["P", "S", "t"])
    panel, gindex, tvals = panel_from_long(df, ["P", "S"], "t", "v")
    gc, gi_ref = pd.factorize(pd.MultiIndex.from_frame(df[["P", "S"]]),
                              sort=True)
    tc, tv_ref = pd.factorize(df["t"], sort=True)
    ref = np.full((len(gi_ref), len(tv_ref)), np.nan, dtype=np.float32)
    ref[gc, tc] = df["v"].to_numpy(np.float32)
    assert np.array_equal(panel, ref, equal_nan=True)
    assert list(gindex) == list(gi_ref)
    assert (tvals == np.asarray(tv_ref)).all()


def test_scatter_oob_raises():
    from mi355x_scale.groupby import _gather
    panel = np.zeros((2, 2), dtype=np.float32)
    with pytest.raises(Exception):
        _gather.scatter_f32(panel, np.array([5], dtype=np.int32),
                            np.array([0], dtype=np.int32),
                            np.array([1.0], dtype=np.float32))


try:
    from hypothesis import given, settings
    from hypothesis import strategies as st

    # NUL excluded: pandas' StringHashTable hashes NUL-terminated C
    # strings, so pd.factorize conflates '\x00' with '' — our factorize
    # hashes length-delimited arrow buffers and correctly distinguishes
    # them (see test_factorize_nul_bytes_distinct below).
    @settings(max_examples=40, deadline=None)
    @given(st.lists(st.one_of(
        st.text(st.characters(exclude_characters="\x00"),
                min_size=0, max_size=12),
        st.sampled_from(["dup1", "dup2", ""])), min_size=0, max_size=300))
    def test_property_strings_match_pandas(vals):
        col = pd.Series(vals, dtype=object)
        codes, uniq = fast_factorize(col)
        ref_codes, ref_uniq = pd.factorize(col, sort=True)
        assert (codes == ref_codes).all()
        assert list(uniq) == list(ref_uniq)

    @settings(max_examples=40, deadline=None)
    @given(st.lists(st.integers(min_value=-2**62, max_value=2**62),
                    min_size=0, max_size=300))
    def test_property_ints_match_pandas(vals):
        col = pd.Series(vals, dtype=np.int64)
        codes, uniq = fast_factorize(col)
        ref_codes, ref_uniq = pd.factorize(col, sort=True)
        assert (codes == ref_codes).all()
        assert (np.asarray(uniq) == np.asarray(ref_uniq)).all()
except ImportError:  # pragma: no cover
    pass


def test_panel_three_keys_matches_pandas_path():
    """Guards the in-place combined-key build beyond two keys."""
    rng = np.random.default_rng(3)
    n = 4000
    df = pd.DataFrame({
        "A": rng.choice(["x", "y"], n),
        "B": rng.choice([f"b{i}" for i in range(7)], n),
        "C": rng.choice([f"c{i}" for i in range(5)], n),
        "t": rng.integers(0, 10, n).astype(np.int64),
        "v": rng.standard_normal(n).astype(np.float32),
    }).drop_duplicates(["A", "B", "C", "t"])
    from mi355x_scale.groupby.gather import panel_from_long
    panel, gindex, tvals = panel_from_long(df, ["A", "B", "C"], "t", "v")
    # oracle: pandas pivot on the MultiIndex
    ref = df.set_index(["A", "B", "C", "t"])["v"]
    for gi, key in enumerate(gindex):
        for ti, t in enumerate(tvals):
            want = ref.get(key + (t,))
            got = panel[gi, ti]
            if want is None:
                assert np.isnan(got)
            else:
                assert got == np.float32(want)
    # group index sorted like pandas MultiIndex factorize
    codes_ref, idx_ref = pd.factorize(
        pd.MultiIndex.from_frame(df[["A", "B", "C"]]), sort=True)
    assert list(gindex) == list(idx_ref)
