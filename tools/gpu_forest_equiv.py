"""Old-vs-new builder equivalence on the HIP engine (debug aid)."""
import importlib.util
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np


def load(path, name):
    import skdist_amd.models  # noqa
    spec = importlib.util.spec_from_file_location(
        f"skdist_amd.models.{name}", path)
    mod = importlib.util.module_from_spec(spec)
    sys.modules[f"skdist_amd.models.{name}"] = mod
    spec.loader.exec_module(mod)
    return mod


import skdist_amd.models.forest as newf

old = load(os.path.join(os.path.dirname(__file__), "_forest_old_check.py"),
           "forest_old")
rng = np.random.default_rng(0)
n, f = 1_000_000, 64
X = rng.standard_normal((n, f)).astype(np.float32)
y = ((X @ rng.standard_normal(f)) > 0).astype(np.int64)
kw = dict(max_depth=12, max_features="sqrt", bootstrap=True, tree_batch=8)
ds_n = newf.BinnedDataset(X, y, "cuda", is_cls=True)
ds_o = old.BinnedDataset(X, y, "cuda", is_cls=True)
tn = newf.ForestBuilder(ds_n, "gini", subtract=False, **kw).build(
    list(range(8)))
to = old.ForestBuilder(ds_o, "gini", subtract=False, **kw).build(
    list(range(8)))
diff = 0
for i, (a, b) in enumerate(zip(tn, to)):
    same = (a.node_count == b.node_count
            and np.array_equal(a.feature, b.feature)
            and np.array_equal(a.threshold, b.threshold))
    if not same:
        diff += 1
        print(f"tree {i}: nodes {a.node_count} vs {b.node_count}")
        if a.node_count == b.node_count:
            j = np.flatnonzero(a.feature != b.feature)[:5]
            print("  first feat diffs at", j, a.feature[j], b.feature[j])
print("identical" if not diff else f"{diff}/8 trees differ")
