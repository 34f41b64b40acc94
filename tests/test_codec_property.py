"""Property-based tests for the tensor codec's flatten/rebuild cycle
(parallel/codec.py) — the wire format under every federation method."""

import hypothesis.strategies as st
import torch
from hypothesis import given, settings

from flreid_amd.parallel.codec import (_flatten, _to_cpu, _unflatten_state)


def _tensors(draw):
    shape = draw(st.lists(st.integers(1, 4), min_size=0, max_size=3))
    return torch.randn(*shape) if shape else torch.tensor(draw(
        st.floats(-10, 10, allow_nan=False, width=32)))


scalars = st.one_of(st.integers(-100, 100),
                    st.floats(-100, 100, allow_nan=False, width=32),
                    st.text(max_size=8), st.none(), st.booleans())


@st.composite
def states(draw, depth=2):
    if depth == 0:
        return draw(st.one_of(scalars, st.just("T")))
    return draw(st.one_of(
        scalars,
        st.just("T"),
        st.dictionaries(st.text(min_size=1, max_size=6),
                        states(depth=depth - 1), max_size=3),
        st.lists(states(depth=depth - 1), max_size=3),
    ))


def _materialize(node, rng):
    """Replace 'T' markers with random tensors (deterministic per test)."""
    if node == "T":
        shape = [int(rng.integers(1, 4)) for _ in range(int(rng.integers(0, 3)))]
        return torch.randn(shape) if shape else torch.randn(())
    if isinstance(node, dict):
        return {k: _materialize(v, rng) for k, v in node.items()}
    if isinstance(node, list):
        return [_materialize(v, rng) for v in node]
    return node


@settings(max_examples=60, deadline=None, derandomize=True)
@given(skeleton=states(), seed=st.integers(0, 2 ** 16))
def test_flatten_unflatten_roundtrip(skeleton, seed):
    import numpy as np
    rng = np.random.default_rng(seed)
    state = _materialize(skeleton, rng)

    tensors = []
    meta = _flatten(state, "root", tensors)
    if tensors:
        flat = torch.cat([t.reshape(-1).float() for _n, t in tensors])
    else:
        flat = torch.zeros(0)
    rebuilt = _unflatten_state(meta, flat)

    def check(a, b):
        if torch.is_tensor(a):
            assert torch.is_tensor(b)
            assert a.shape == b.shape
            assert torch.allclose(a.float(), b.float())
            return
        if isinstance(a, dict):
            assert set(a) == set(b)
            for k in a:
                check(a[k], b[k])
            return
        if isinstance(a, list):
            # codec may rebuild lists as lists (structure preserved)
            assert isinstance(b, list) and len(a) == len(b)
            for x, y in zip(a, b):
                check(x, y)
            return
        assert a == b

    check(state, rebuilt)

    # _to_cpu is structure-preserving too
    check(state, _to_cpu(state))
