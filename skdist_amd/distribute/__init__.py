"""
Distributed meta-estimators — the product layer (reference SURVEY.md §1 L2).
"""

from .eliminate import DistFeatureEliminator
from .encoder import Encoderizer, EncoderizerExtractor
from .ensemble import (
    DistExtraTreesClassifier,
    DistExtraTreesRegressor,
    DistRandomForestClassifier,
    DistRandomForestRegressor,
    DistRandomTreesEmbedding,
)
from .multiclass import DistOneVsOneClassifier, DistOneVsRestClassifier
from .predict import DistPredictor, get_prediction_fn
from .search import (
    DistGridSearchCV,
    DistMultiModelSearch,
    DistRandomizedSearchCV,
)

__all__ = [
    "DistGridSearchCV", "DistRandomizedSearchCV", "DistMultiModelSearch",
    "DistOneVsRestClassifier", "DistOneVsOneClassifier",
    "DistRandomForestClassifier", "DistExtraTreesClassifier",
    "DistRandomForestRegressor", "DistExtraTreesRegressor",
    "DistRandomTreesEmbedding", "DistFeatureEliminator",
    "Encoderizer", "EncoderizerExtractor",
    "DistPredictor", "get_prediction_fn",
]
