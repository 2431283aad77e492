"""Property-based invariants of the histogram tree builder (hypothesis).

Reference test style analog: exact small-data asserts
(skdist tests) — extended here with randomized structural
invariants: full-binary topology, reachable leaves, simplex leaf
payloads, bounded depth.
"""

import numpy as np
from hypothesis import given, settings, strategies as st

from skdist_amd.models.forest import BinnedDataset, ForestBuilder


@settings(max_examples=12, deadline=None)
@given(
    n=st.integers(60, 400),
    f=st.integers(2, 8),
    k=st.integers(2, 4),
    seed=st.integers(0, 2 ** 20),
    depth=st.integers(1, 6),
    bootstrap=st.booleans(),
    extra=st.booleans(),
)
def test_tree_structure_invariants(n, f, k, seed, depth, bootstrap, extra):
    rng = np.random.default_rng(seed)
    X = rng.standard_normal((n, f)).astype(np.float32)
    y = rng.integers(0, k, size=n)
    ds = BinnedDataset(X, y, "cpu", is_cls=True, nbins=32)
    tree = ForestBuilder(
        ds, "gini", max_depth=depth, max_features="sqrt",
        bootstrap=bootstrap, extra_mode=extra,
    ).build([seed])[0]

    nn = tree.node_count
    internal = tree.feature >= 0
    leaves = ~internal
    # full binary tree: one more leaf than internal nodes
    assert leaves.sum() == internal.sum() + 1
    assert nn % 2 == 1
    # children point inside the tree; leaf slots index the value table
    assert (tree.left[internal] >= 0).all()
    assert (tree.right[internal] < nn).all()
    assert (tree.left[leaves] < len(tree.value)).all()
    # every row lands on a leaf with a simplex payload
    proba = tree.predict_proba(X)
    assert proba.shape == (n, ds.S)
    np.testing.assert_allclose(proba.sum(axis=1), 1.0, atol=1e-5)
    assert (proba >= 0).all()
    # depth bound: walk down from the root counting levels
    level = {0: 0}
    maxd = 0
    for i in range(nn):
        if internal[i]:
            level[tree.left[i]] = level[i] + 1
            level[tree.right[i]] = level[i] + 1
            maxd = max(maxd, level[i] + 1)
    assert maxd <= depth
