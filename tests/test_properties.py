"""Property-based invariants (hypothesis, derandomized so driver runs are
deterministic): ShardSpec bucket-major geometry and const-length packing."""

from hypothesis import given, settings, strategies as st

from acco_amd.data.packing import pack_const_len
from acco_amd.parallel.comm import ShardSpec

FAST = settings(max_examples=60, deadline=None, derandomize=True)


@FAST
@given(n=st.integers(1, 10_000_000),
       world=st.sampled_from([1, 2, 4, 8]),
       buckets=st.integers(1, 16))
def test_shard_spec_invariants(n, world, buckets):
    spec = ShardSpec.build(n, world, buckets=buckets)
    assert spec.total >= n                       # live data always fits
    assert spec.seg % 256 == 0                   # aligned segments
    assert spec.total == spec.nb * spec.world * spec.seg
    assert spec.owned * world == spec.total      # exact ZeRO-1 partition
    # (bucket, rank) tiling covers [0, total) exactly once, in order
    off = 0
    for j in range(spec.nb):
        for r in range(world):
            assert j * spec.bucket_elems + r * spec.seg == off
            off += spec.seg
    assert off == spec.total


@FAST
@given(docs=st.lists(st.lists(st.integers(0, 100), min_size=0, max_size=20),
                     min_size=0, max_size=20),
       eos=st.integers(101, 105),
       L=st.integers(1, 16))
def test_pack_const_len_invariants(docs, eos, L):
    out = pack_const_len(docs, eos_token_id=eos, max_length=L)
    concat = []
    for d in docs:
        concat.extend(d)
        concat.append(eos)
    # chop semantics (reference tokenize_data_const_len trainer_base.py:84-97):
    # full blocks of the concatenation, remainder dropped
    assert out.shape == (len(concat) // L, L)
    assert out.flatten().tolist() == concat[:(len(concat) // L) * L]
