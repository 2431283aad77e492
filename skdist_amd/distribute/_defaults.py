"""
Default per-column encoder pipeline factories, size-tiered
(reference: skdist/distribute/_defaults.py).

Registry shape: ``_default_encoders[size][kind](col) -> [(name, pipeline)]``.
The string vectorizers differ per tier (word/char n-gram ranges,
reference _defaults.py:91-198); the other encoder kinds are tier-invariant.
"""

from sklearn.feature_extraction import DictVectorizer
from sklearn.feature_extraction.text import CountVectorizer
from sklearn.feature_selection import VarianceThreshold
from sklearn.impute import SimpleImputer
from sklearn.pipeline import Pipeline
from sklearn.preprocessing import StandardScaler

from ..preprocessing import (
    FeatureCast,
    HashingVectorizerChunked,
    ImputeNull,
    MultihotEncoder,
    SelectField,
    UniqueFactorizedVectorizer,
)


def tokenizer(x):
    """Identity tokenizer for pre-tokenized input
    (reference _defaults.py:23-25)."""
    return x


def dict_encoder(col):
    return [(
        f"{col}_dict_encoder",
        Pipeline([
            ("var", SelectField(cols=[col], single_dimension=True)),
            ("fillna", ImputeNull({})),
            ("vec", DictVectorizer()),
        ]),
    )]


def onehot_encoder(col):
    return [(
        f"{col}_onehot",
        Pipeline([
            ("var", SelectField(cols=[col], single_dimension=True)),
            ("cast", FeatureCast(cast_type=str)),
            ("fillna", ImputeNull("")),
            # CountVectorizer on the column's UNIQUE values only +
            # row gather: sklearn-identical output, O(n_unique) python
            # instead of O(n) (UniqueFactorizedVectorizer docstring)
            ("vec", UniqueFactorizedVectorizer(CountVectorizer(
                token_pattern=None, tokenizer=tokenizer, binary=True,
                decode_error="ignore",
            ))),
        ]),
    )]


def multihot_encoder(col):
    return [(
        f"{col}_multihot",
        Pipeline([
            ("var", SelectField(cols=[col], single_dimension=True)),
            ("fillna", ImputeNull([])),
            ("vec", UniqueFactorizedVectorizer(MultihotEncoder())),
        ]),
    )]


def numeric_encoder(col):
    return [(
        f"{col}_scaler",
        Pipeline([
            ("var", SelectField(cols=[col])),
            ("imputer", SimpleImputer(strategy="median")),
            ("scaler", StandardScaler(copy=False)),
        ]),
    )]


def _text_vec(col, suffix, analyzer, ngram_range):
    return (
        f"{col}_{suffix}",
        Pipeline([
            ("var", SelectField(cols=[col], single_dimension=True)),
            ("fillna", ImputeNull(" ")),
            ("vec", HashingVectorizerChunked(
                ngram_range=ngram_range, analyzer=analyzer,
                decode_error="ignore",
            )),
            ("var_thresh", VarianceThreshold()),
        ]),
    )


# per-tier text vectorizer specs: (suffix, analyzer, ngram_range)
_TEXT_TIERS = {
    "small": [("word_vec", "word", (1, 2))],
    "medium": [("word_vec", "word", (1, 3)),
               ("char_vec", "char_wb", (3, 4))],
    "large": [("word_vec", "word", (1, 3)),
              ("char_vec", "char_wb", (2, 5))],
}


def _make_tier(size):
    return {
        "string_vectorizer": lambda c, s=size: [
            _text_vec(c, *spec) for spec in _TEXT_TIERS[s]
        ],
        "onehotencoder": onehot_encoder,
        "multihotencoder": multihot_encoder,
        "numeric": numeric_encoder,
        "dict": dict_encoder,
    }


_default_encoders = {size: _make_tier(size) for size in _TEXT_TIERS}
