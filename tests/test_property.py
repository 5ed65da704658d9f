"""Property-based tests (hypothesis) for core invariants."""
import json

from hypothesis import given, settings
from hypothesis import strategies as st

from maggy_amd import Searchspace, Trial
from maggy_amd.core.shm import MetricRing

names = st.text(alphabet="abcdefghij_", min_size=1, max_size=8)
finite = st.floats(allow_nan=False, allow_infinity=False, width=32)


@settings(max_examples=50, deadline=None)
@given(st.dictionaries(
    names,
    st.one_of(st.integers(-10**6, 10**6), finite,
              st.text(max_size=12), st.booleans()),
    min_size=1, max_size=6))
def test_trial_id_stable_and_json_roundtrip(params):
    t1 = Trial(dict(params))
    t2 = Trial(dict(sorted(params.items(), reverse=True)))
    # content-addressed id is insertion-order independent
    assert t1.trial_id == t2.trial_id
    restored = Trial.from_json(t1.to_json())
    assert restored.trial_id == t1.trial_id
    assert restored.params == t1.params
    # the id is exactly the md5[:16] of the sorted-params json
    import hashlib

    expect = hashlib.md5(
        json.dumps(params, sort_keys=True).encode()).hexdigest()[:16]
    assert t1.trial_id == expect


@settings(max_examples=40, deadline=None)
@given(st.floats(-1e6, 1e6), st.floats(-1e6, 1e6), st.floats(0, 1))
def test_searchspace_double_roundtrip(lo, hi, frac):
    if not hi > lo + 1e-6:
        return
    sp = Searchspace(x=("DOUBLE", [lo, hi]))
    v = lo + (hi - lo) * frac
    t = sp.transform([v])
    assert -1e-9 <= t[0] <= 1 + 1e-9
    back = sp.inverse_transform(t)[0]
    assert abs(back - v) <= 1e-6 * max(1.0, abs(hi), abs(lo))


@settings(max_examples=40, deadline=None)
@given(st.integers(-1000, 1000), st.integers(1, 2000), st.integers(0, 100))
def test_searchspace_integer_roundtrip(lo, span, off):
    hi = lo + span
    v = lo + min(off, span)
    sp = Searchspace(n=("INTEGER", [lo, hi]))
    assert sp.inverse_transform(sp.transform([v]))[0] == v


@settings(max_examples=30, deadline=None)
@given(st.lists(st.tuples(st.integers(1, 2**60), st.integers(0, 10**6),
                          finite),
                min_size=0, max_size=200),
       st.integers(1, 4))
def test_metric_ring_preserves_order(records, chunks):
    ring = MetricRing(slots=64, create=True)
    try:
        got = []
        # interleave pushes with partial drains
        step = max(1, len(records) // chunks)
        for i in range(0, len(records), step):
            for tag, s, v in records[i:i + step]:
                ring.push(tag, s, v)
            got.extend(ring.drain())
        got.extend(ring.drain())
        # with < 64 pushes between drains nothing is dropped (a full-ring
        # burst sacrifices its oldest slot to the torn-read guard)
        if step < 64:
            assert [(t, s) for t, s, _ in records] == \
                   [(t, s) for t, s, _ in got]
            for (_, _, v0), (_, _, v1) in zip(records, got):
                assert v0 == v1 or (v0 != v0 and v1 != v1)
        else:
            assert len(got) <= len(records)
    finally:
        ring.close()
        ring.unlink()
