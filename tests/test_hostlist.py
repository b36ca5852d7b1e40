import pytest

from acco_amd.utils.hostlist import (BadHostlist, expand_hostlist,
                                     parse_slurm_tasks_per_node)


def test_simple():
    assert expand_hostlist("n1") == ["n1"]
    assert expand_hostlist("n1,n2") == ["n1", "n2"]


def test_ranges():
    assert expand_hostlist("n[9-11]") == ["n9", "n10", "n11"]
    assert expand_hostlist("gpu[01-03]") == ["gpu01", "gpu02", "gpu03"]
    assert expand_hostlist("n[1,3,5-6]") == ["n1", "n3", "n5", "n6"]


def test_mixed_and_suffix():
    assert expand_hostlist("a[1-2]b,c") == ["a1b", "a2b", "c"]


def test_bad():
    with pytest.raises(BadHostlist):
        expand_hostlist("n[1")


def test_tasks_per_node():
    assert parse_slurm_tasks_per_node("2(x3),1") == [2, 2, 2, 1]
    assert parse_slurm_tasks_per_node("8") == [8]


# ------------------------- property: generated ranges expand correctly

from hypothesis import given, settings
from hypothesis import strategies as st


@st.composite
def _ranged_hostlists(draw):
    """Build (expr, expected) pairs from random prefixes + numeric ranges."""
    n_parts = draw(st.integers(1, 3))
    exprs, expected = [], []
    for _ in range(n_parts):
        prefix = draw(st.text(alphabet="abcz-", min_size=1, max_size=4)
                      .filter(lambda s: not s.startswith("-")))
        kind = draw(st.sampled_from(["plain", "range", "padded"]))
        if kind == "plain":
            exprs.append(prefix)
            expected.append(prefix)
        else:
            lo = draw(st.integers(0, 30))
            hi = lo + draw(st.integers(0, 8))
            if kind == "padded":
                width = draw(st.integers(2, 4))
                exprs.append(f"{prefix}[{lo:0{width}d}-{hi:0{width}d}]")
                expected.extend(f"{prefix}{i:0{width}d}"
                                for i in range(lo, hi + 1))
            else:
                exprs.append(f"{prefix}[{lo}-{hi}]")
                expected.extend(f"{prefix}{i}" for i in range(lo, hi + 1))
    return ",".join(exprs), expected


@settings(max_examples=200, deadline=None)
@given(_ranged_hostlists())
def test_expand_property(pair):
    expr, expected = pair
    assert expand_hostlist(expr) == expected
