"""GPU tests: HIP hashing vectorizer vs sklearn, bit-exact on ASCII.

Reference behavior: sklearn HashingVectorizer as used by the reference's
Encoderizer text pipelines (skdist/preprocessing.py:264-310,
skdist/distribute/_defaults.py:91-198).
"""

import numpy as np
import pytest
import torch
from sklearn.feature_extraction.text import HashingVectorizer

pytestmark = pytest.mark.gpu

DOCS = [
    "The quick brown fox jumps over the lazy dog",
    "pack my box with five dozen liquor jugs",
    "a ab abc abcd _under_score_ 1234 mixedCASE Token99",
    "",
    "   leading and trailing   spaces   ",
    "punctuation, should; split: tokens! right? (yes) [ok] {fine}",
    "repeat repeat repeat repeat repeat",
    "x " * 200 + "tail",
]


def _cases():
    return [
        dict(analyzer="word", ngram_range=(1, 1)),
        dict(analyzer="word", ngram_range=(1, 2)),
        dict(analyzer="word", ngram_range=(1, 3), norm=None),
        dict(analyzer="word", ngram_range=(2, 3), alternate_sign=False),
        dict(analyzer="word", ngram_range=(1, 2), binary=True),
        dict(analyzer="word", ngram_range=(1, 1), norm="l1"),
        dict(analyzer="char_wb", ngram_range=(3, 4)),
        dict(analyzer="char_wb", ngram_range=(2, 5)),
        dict(analyzer="char_wb", ngram_range=(1, 3), norm=None),
        dict(analyzer="word", ngram_range=(1, 2), n_features=4096),
    ]


@pytest.mark.parametrize("kw", _cases())
def test_device_hash_matches_sklearn(kw):
    from skdist_amd.ops import hash_vectorize

    ref = HashingVectorizer(**kw).transform(DOCS)
    out = hash_vectorize(
        DOCS,
        n_features=kw.get("n_features", 2 ** 20),
        analyzer=kw.get("analyzer", "word"),
        ngram_range=kw.get("ngram_range", (1, 1)),
        alternate_sign=kw.get("alternate_sign", True),
        binary=kw.get("binary", False),
        norm=kw.get("norm", "l2"),
    )
    assert out.shape == ref.shape
    d = (out - ref)
    assert abs(d).max() < 1e-12 if d.nnz else True


def test_chunked_vectorizer_takes_device_path():
    from skdist_amd.preprocessing import HashingVectorizerChunked

    v = HashingVectorizerChunked(ngram_range=(1, 2))
    assert v._try_device_transform(DOCS) is not None
    out = v.transform(np.asarray(DOCS, dtype=object))
    ref = HashingVectorizer(ngram_range=(1, 2)).transform(DOCS)
    assert abs(out - ref).max() < 1e-12

    # non-ASCII falls back to sklearn, same result
    docs2 = DOCS + ["café ümlaut — em-dash"]
    out2 = v.transform(np.asarray(docs2, dtype=object))
    ref2 = HashingVectorizer(ngram_range=(1, 2)).transform(docs2)
    assert abs(out2 - ref2).max() < 1e-12


def test_encoderizer_text_on_gpu():
    import pandas as pd

    from skdist_amd.distribute.encoder import Encoderizer

    df = pd.DataFrame({
        "txt": [f"some document number {i} with words {i % 7}"
                for i in range(60)],
        "num": np.arange(60, dtype=float),
    })
    enc = Encoderizer(size="small")
    T = enc.fit_transform(df)
    assert T.shape[0] == 60
