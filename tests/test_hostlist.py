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
