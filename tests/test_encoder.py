"""
Encoderizer tests (reference mirror:
skdist/distribute/tests/test_encoder.py).
"""

import numpy as np
import pandas as pd
import pytest

from skdist_amd import Cluster
from skdist_amd.distribute.encoder import Encoderizer, EncoderizerExtractor


@pytest.fixture
def mixed_df():
    return pd.DataFrame({
        "text": [
            "the quick brown fox", "jumps over the lazy dog",
            "pack my box", "with five dozen", "liquor jugs",
            "how vexingly quick", "daft zebras jump", "bright vixens",
            "quick wafting zephyrs", "vex bold jim", "sphinx of black",
            "quartz judge my vow",
        ],
        "num": [1.0, 2.5, 3.3, 0.1, 5.5, 2.2, 8.8, 1.1, 0.4, 3.3, 2.8, 9.9],
        "cat": ["a", "b"] * 6,
    })


@pytest.fixture
def mixed_df24(mixed_df):
    # 24 rows so "cat" is categorical (2/24 < 0.10 unique ratio)
    return pd.concat([mixed_df, mixed_df], ignore_index=True)


def test_infer_and_fit_transform(mixed_df24):
    mixed_df = mixed_df24
    enc = Encoderizer(size="small")
    out = enc.fit_transform(mixed_df)
    assert out.shape[0] == 24
    assert out.shape[1] == sum(enc.transformer_lengths)
    names = enc.step_names
    assert any("word_vec" in n for n in names)
    assert any("scaler" in n for n in names)
    assert any("onehot" in n for n in names)


def test_medium_has_char_vec(mixed_df24):
    enc = Encoderizer(size="medium")
    enc.fit(mixed_df24)
    assert any("char_vec" in n for n in enc.step_names)


def test_dict_and_list_columns():
    df = pd.DataFrame({
        "d": [{"a": 1}, {"b": 2}, {"a": 3}] * 4,
        "l": [["x"], ["y"], ["x", "y"]] * 4,
    })
    enc = Encoderizer(size="small")
    out = enc.fit_transform(df)
    assert out.shape[0] == 12
    assert any("dict_encoder" in n for n in enc.step_names)
    assert any("multihot" in n for n in enc.step_names)


def test_stringified_container_raises():
    df = pd.DataFrame({"d": ["{'a': 1}"] * 6})
    with pytest.raises(ValueError, match="Convert this column"):
        Encoderizer(size="small").fit(df)


def test_numpy_input_requires_col_names(mixed_df24):
    mixed_df = mixed_df24
    X = mixed_df.values
    with pytest.raises(ValueError):
        Encoderizer(size="small").fit(X)
    enc = Encoderizer(size="small", col_names=list(mixed_df.columns))
    out = enc.fit_transform(X)
    assert out.shape[0] == 24


def test_dict_input(mixed_df24):
    enc = Encoderizer(size="small")
    out = enc.fit_transform(mixed_df24.to_dict(orient="list"))
    assert out.shape[0] == 24


def test_config_override(mixed_df):
    enc = Encoderizer(size="small", config={"text": "string_vectorizer",
                                            "num": "numeric"})
    out = enc.fit_transform(mixed_df)
    assert out.shape[0] == 12
    assert len(enc.step_names) == 2


def test_feature_origin(mixed_df24):
    mixed_df = mixed_df24
    enc = Encoderizer(size="small")
    enc.fit(mixed_df)
    name = enc.feature_origin(0)
    assert name == enc.step_names[0]
    last = enc.feature_origin(sum(enc.transformer_lengths) - 1)
    assert last == enc.step_names[-1]


def test_extract_and_extractor(mixed_df24):
    mixed_df = mixed_df24
    enc = Encoderizer(size="small")
    enc.fit(mixed_df)
    keep = [n for n in enc.step_names if "scaler" in n]
    sub = enc.extract(keep)
    out = sub.transform(mixed_df)
    assert out.shape == (24, 1)
    ext = EncoderizerExtractor(enc, keep)
    out2 = ext.fit(mixed_df).transform(mixed_df)
    assert np.allclose(np.asarray(out), np.asarray(out2))


def test_encoder_cluster(mixed_df24):
    enc = Encoderizer(size="small", sc=Cluster())
    out = enc.fit_transform(mixed_df24)
    assert out.shape[0] == 24
    assert enc.sc is None


def test_transformer_weights(mixed_df24):
    mixed_df = mixed_df24
    enc = Encoderizer(size="small")
    enc.fit(mixed_df)
    scaler = [n for n in enc.step_names if "scaler" in n][0]
    enc_w = Encoderizer(size="small",
                        transformer_weights={scaler: 2.0})
    enc_w.fit(mixed_df)
    a = enc.transform(mixed_df)
    b = enc_w.transform(mixed_df)
    assert abs(b.sum() - a.sum()) > 0  # weighting changed something


def test_list_of_record_dicts():
    """Record-style input coerces to a frame (reference
    encoder.py:237-266 accepts pandas/dict/numpy/list)."""
    recs = [{"txt": f"doc {i % 3}", "num": float(i)} for i in range(40)]
    enc = Encoderizer(size="small")
    T = enc.fit_transform(recs)
    assert T.shape[0] == 40
    assert enc.transform(recs[:5]).shape == (5, T.shape[1])


def test_config_validation_errors():
    df = pd.DataFrame({"a": [1.0, 2.0], "b": ["x", "y"]})
    with pytest.raises(ValueError, match="not in the input"):
        Encoderizer(size="small", config={"zzz": "numeric"}).fit(df)
    with pytest.raises(ValueError, match="unknown encoder kind"):
        Encoderizer(size="small", config={"a": "bogus"}).fit(df)


def test_defaults_registry_factories():
    """Every tier's factory returns [(name, pipeline)] steps (reference
    test_defaults.py:18-57) and the identity tokenizer is identity."""
    from skdist_amd.distribute import _defaults

    assert _defaults.tokenizer(5) == 5
    for size in ("small", "medium", "large"):
        for kind, factory in _defaults._default_encoders[size].items():
            steps = factory("c")
            assert isinstance(steps, list) and steps
            for name, pipe in steps:
                assert "c" in name
                assert hasattr(pipe, "fit") and hasattr(pipe, "transform")
