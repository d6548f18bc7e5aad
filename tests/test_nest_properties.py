"""Property tests for the nest module (hypothesis): the structural
identities the runtime relies on hold for arbitrary nested shapes.

Analogue of the reference's nest unit tests (nest/nest_test.py), but
property-based: random nest structures of tuples/lists/dicts/leaves.
"""

import hypothesis.strategies as st
import numpy as np
import torch
from hypothesis import given, settings

import torchbeast_amd.nest as nest


def leaves():
    return st.builds(
        lambda seed, shape: torch.from_numpy(
            np.random.RandomState(seed).standard_normal(shape)
            .astype(np.float32)),
        st.integers(0, 2**31 - 1),
        st.lists(st.integers(0, 3), min_size=0, max_size=3),
    )


def nests():
    return st.recursive(
        leaves(),
        lambda children: st.one_of(
            st.lists(children, min_size=1, max_size=3).map(tuple),
            st.lists(children, min_size=1, max_size=3),
            st.dictionaries(st.text(min_size=1, max_size=8), children,
                            min_size=1, max_size=3),
        ),
        max_leaves=8,
    )


def assert_same_structure_and_values(a, b):
    fa, fb = nest.flatten(a), nest.flatten(b)
    assert len(fa) == len(fb)
    for x, y in zip(fa, fb):
        torch.testing.assert_close(x, y)


@settings(max_examples=60, deadline=None)
@given(nests())
def test_pack_as_flatten_roundtrip(n):
    flat = nest.flatten(n)
    packed = nest.pack_as(n, flat)
    assert_same_structure_and_values(n, packed)


@settings(max_examples=60, deadline=None)
@given(nests())
def test_map_preserves_structure(n):
    doubled = nest.map(lambda t: t * 2, n)
    flat, dflat = nest.flatten(n), nest.flatten(doubled)
    assert len(flat) == len(dflat)
    for x, y in zip(flat, dflat):
        torch.testing.assert_close(y, x * 2)
    # pack_as accepts the mapped leaves against the original structure.
    assert_same_structure_and_values(doubled, nest.pack_as(n, dflat))


@settings(max_examples=40, deadline=None)
@given(nests())
def test_map_many2_matches_elementwise(n):
    other = nest.map(lambda t: t + 1, n)
    summed = nest.map_many2(lambda a, b: a + b, n, other)
    for x, y in zip(nest.flatten(n), nest.flatten(summed)):
        torch.testing.assert_close(y, 2 * x + 1)


@settings(max_examples=40, deadline=None)
@given(nests())
def test_front_is_first_flat_leaf(n):
    torch.testing.assert_close(nest.front(n), nest.flatten(n)[0])
