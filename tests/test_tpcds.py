"""TPC-DS query correctness vs pandas oracle at small SF (oracle (b) of
SURVEY.md §4: vanilla-engine results are ground truth)."""
import math
import os

import numpy as np
import pytest

from auron_amd import AuronSession
from auron_amd.tpcds import datagen
from auron_amd.tpcds.oracle import ORACLES
from auron_amd.tpcds.queries import QUERIES, Catalog

SF = float(os.environ.get("AURON_TEST_SF", "0.01"))
ROOT = os.path.join(os.path.dirname(__file__), "..", ".tpcds_cache")


@pytest.fixture(scope="module")
def dataset():
    datagen.write_dataset(ROOT, SF)
    return ROOT


def rows_of(df):
    import pandas as pd

    out = []
    for _, r in df.iterrows():
        row = []
        for v in r:
            if v is pd.NA or (isinstance(v, float) and math.isnan(v)):
                row.append(None)
            elif isinstance(v, (np.integer,)):
                row.append(int(v))
            elif isinstance(v, (np.floating,)):
                row.append(float(v))
            else:
                row.append(v)
        out.append(tuple(row))
    return out


def _skey(v):
    if v is None:
        return (True, "")
    if isinstance(v, float):
        return (False, f"{v:.6e}")  # round for pairing; exactness checked later
    return (False, str(v))


# fp summation-order differences make exact tie-ranks on float keys
# ambiguous across engines; allow a small delta on these columns
# (GPU sums associate differently from pandas, so q67's dense float
# ranking drifts a little further than the CPU run does)
RANK_TOLERANT = {"q36": {"rank_within_parent": 1}, "q67": {"rk": 3},
                 "q49": {"return_rank": 2, "currency_rank": 2}}
# queries whose ORDER BY keys tie across rows: the LIMIT keeps an
# engine-dependent subset, so the oracle returns the FULL result and the
# engine rows must be a subset of it (official TPC-DS answer sets have the
# same ambiguity)
SUBSET_OF_FULL = {"q59", "q18"}
# boundary-epsilon queries: oracle returns an epsilon-relaxed FULL superset;
# engine rows must be members (threshold rows may legitimately differ by
# one fp ulp between engines)
SUBSET_LOOSE = {"q65"}


def assert_result_matches(batch, df, qname=None):
    tolerant = RANK_TOLERANT.get(qname, {})
    got_d = batch.to_pydict()
    got_cols = list(got_d.keys())
    want_cols = list(df.columns)
    assert got_cols == want_cols, f"{got_cols} != {want_cols}"
    if qname == "q67":
        # q67 ranks dense fp64 sums inside a rollup and keeps rk<=100:
        # atomic fp64 summation order legitimately flips near-tie ranks
        # between runs, which both perturbs rk and swaps rows across the
        # rk<=100 boundary. Compare the non-rank columns as a rounded
        # multiset with a small boundary slack instead.
        import collections

        ri = got_cols.index("rk")

        def soft(v):
            # 4 significant digits: engine and pandas associate fp64
            # sums differently; pairwise isclose is unavailable in a
            # multiset compare. None and NaN both mean SQL NULL here.
            if v is None:
                return "~"
            if isinstance(v, float):
                if v != v:
                    return "~"
                if abs(v) < 1e12 and v == int(v):
                    return int(v)  # pandas floats int-typed engine cols
                return f"{v:.3e}"
            return v

        strip = lambda rows: collections.Counter(
            tuple(soft(v) for v in (r[:ri] + r[ri + 1:])) for r in rows)
        g = strip(list(zip(*got_d.values())))
        w = strip(rows_of(df))
        extra = sum((g - w).values()) + sum((w - g).values())
        assert extra <= 10, f"q67 row multiset diverges by {extra}"
        assert all(1 <= v <= 100 for v in got_d["rk"])
        return
    got_rows = sorted(zip(*got_d.values()), key=lambda r: tuple(_skey(v) for v in r))
    want_rows = sorted(rows_of(df), key=lambda r: tuple(_skey(v) for v in r))
    assert len(got_rows) == len(want_rows), \
        f"{len(got_rows)} rows != {len(want_rows)}\n{got_rows[:5]}\n{want_rows[:5]}"
    for a, b in zip(got_rows, want_rows):
        for x, y, cname in zip(a, b, got_cols):
            if cname in tolerant and isinstance(x, int) and isinstance(y, int):
                assert abs(x - y) <= tolerant[cname], (a, b)
            elif isinstance(x, float) and isinstance(y, float):
                assert math.isclose(x, y, rel_tol=1e-6, abs_tol=1e-6), (a, b)
            elif isinstance(x, float) or isinstance(y, float):
                assert x is not None and y is not None and math.isclose(float(x), float(y), rel_tol=1e-6), (a, b)
            else:
                assert x == y, (a, b)


def _round_row(r):
    return tuple("~" if v is None else (f"{v:.6e}" if isinstance(v, float) else v)
                 for v in r)


@pytest.mark.parametrize("qname", sorted(QUERIES.keys()))
def test_query_vs_oracle(dataset, qname):
    s = AuronSession()
    cat = Catalog(dataset, SF)
    plan = QUERIES[qname](cat, s)
    got = s.collect(plan)
    want = ORACLES[qname](dataset, SF)
    if qname in SUBSET_OF_FULL or qname in SUBSET_LOOSE:
        got_d = got.to_pydict()
        assert list(got_d.keys()) == list(want.columns)
        full = set(_round_row(r) for r in rows_of(want))
        got_rows = list(zip(*got_d.values()))
        if qname in SUBSET_OF_FULL:
            assert len(got_rows) == min(100, len(full))
        else:
            assert len(got_rows) <= 100 and len(got_rows) > 0
        for r in got_rows:
            assert _round_row(r) in full, r
        return
    assert_result_matches(got, want, qname)
