"""
Encoderizer (reference analog: examples/encoder/basic_usage.py —
size-tiered automatic feature encoding over a mixed DataFrame).

Column types are inferred (numeric / categorical / text / dict / list);
text columns hash through the HIP murmur3 vectorizer when a GPU is
present (bit-identical to sklearn's HashingVectorizer on ASCII input).
"""

import numpy as np
import pandas as pd

from skdist_amd.distribute.encoder import Encoderizer

df = pd.DataFrame({
    "text": [f"document number {i} about topic {i % 5}" for i in range(200)],
    "cat": [f"c{i % 4}" for i in range(200)],
    "num": np.random.default_rng(0).standard_normal(200),
})
for size in ("small", "medium"):
    enc = Encoderizer(size=size)
    T = enc.fit_transform(df)
    print(size, "encoded width:", T.shape[1])
