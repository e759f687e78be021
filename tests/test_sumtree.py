import numpy as np
import pytest
from hypothesis import given, settings
from hypothesis import strategies as st

from rainbowiqn_amd.replay.sumtree import SumTree


def test_update_and_total():
    t = SumTree(10)
    t.update(np.arange(10), np.arange(10, dtype=float))
    assert t.total() == pytest.approx(45.0)
    assert t.get(np.array([3]))[0] == pytest.approx(3.0)


def test_find_deterministic():
    t = SumTree(8)
    t.update(np.arange(8), np.array([1.0, 0.0, 2.0, 0.0, 3.0, 0.0, 0.0, 4.0]))
    # prefix sums: [0,1) -> 0, [1,3) -> 2, [3,6) -> 4, [6,10) -> 7
    vals = np.array([0.0, 0.5, 1.0, 2.9, 3.0, 5.9, 6.0, 9.99])
    expect = np.array([0, 0, 2, 2, 4, 4, 7, 7])
    assert (t.find(vals) == expect).all()


def test_overwrite_updates_total():
    t = SumTree(4)
    t.update(np.array([0, 1]), np.array([5.0, 5.0]))
    t.update(np.array([0]), np.array([1.0]))
    assert t.total() == pytest.approx(6.0)


@settings(max_examples=50, deadline=None)
@given(
    caps=st.integers(min_value=1, max_value=65),
    data=st.data(),
)
def test_property_total_and_find(caps, data):
    t = SumTree(caps)
    n_updates = data.draw(st.integers(1, 20))
    ref = np.zeros(caps)
    rng = np.random.default_rng(0)
    for _ in range(n_updates):
        k = data.draw(st.integers(1, caps))
        idx = rng.choice(caps, size=k, replace=False)
        pri = rng.random(k) * 10
        t.update(idx, pri)
        ref[idx] = pri
    assert t.total() == pytest.approx(ref.sum(), rel=1e-9)
    if ref.sum() > 0:
        v = rng.random(16) * ref.sum() * (1 - 1e-12)
        leaves = t.find(v)
        cum = np.cumsum(ref)
        expect = np.searchsorted(cum, v, side="right")
        assert (leaves == expect).all()


def test_max_priority_tracking():
    t = SumTree(4)
    assert t.max_priority == 1.0
    t.update(np.array([2]), np.array([7.5]))
    assert t.max_priority == 7.5
