"""Import-guard tests (reference pattern: every module asserts
`_import_error == None`, e.g. skdist/distribute/tests/test_search.py:20-34
— here every public module simply imports and exposes its API)."""

import importlib

import pytest

MODULES = [
    "skdist_amd",
    "skdist_amd.parallel.cluster",
    "skdist_amd.parallel.local",
    "skdist_amd.distribute.base",
    "skdist_amd.distribute.utils",
    "skdist_amd.distribute.validation",
    "skdist_amd.distribute.search",
    "skdist_amd.distribute.multiclass",
    "skdist_amd.distribute.ensemble",
    "skdist_amd.distribute.eliminate",
    "skdist_amd.distribute.encoder",
    "skdist_amd.distribute._defaults",
    "skdist_amd.distribute.predict",
    "skdist_amd.preprocessing",
    "skdist_amd.postprocessing",
    "skdist_amd.models",
    "skdist_amd.models.linear",
    "skdist_amd.models._sgd",
    "skdist_amd.models.forest",
    "skdist_amd.models.boosting",
    "skdist_amd.ops",
    "skdist_amd.ops.build",
]


@pytest.mark.parametrize("mod", MODULES)
def test_imports(mod):
    importlib.import_module(mod)


def test_public_api_surface():
    from skdist_amd import Cluster, __version__  # noqa: F401
    from skdist_amd.distribute.eliminate import DistFeatureEliminator  # noqa: F401
    from skdist_amd.distribute.encoder import (  # noqa: F401
        Encoderizer,
        EncoderizerExtractor,
    )
    from skdist_amd.distribute.ensemble import (  # noqa: F401
        DistExtraTreesClassifier,
        DistExtraTreesRegressor,
        DistRandomForestClassifier,
        DistRandomForestRegressor,
        DistRandomTreesEmbedding,
    )
    from skdist_amd.distribute.multiclass import (  # noqa: F401
        DistOneVsOneClassifier,
        DistOneVsRestClassifier,
    )
    from skdist_amd.distribute.predict import (  # noqa: F401
        DistPredictor,
        get_prediction_fn,
    )
    from skdist_amd.distribute.search import (  # noqa: F401
        DistGridSearchCV,
        DistMultiModelSearch,
        DistRandomizedSearchCV,
    )
    from skdist_amd.models import (  # noqa: F401
        HistGradientBoostingClassifier,
        HistGradientBoostingRegressor,
        LinearSVC,
        LogisticRegression,
        Ridge,
    )
    from skdist_amd.distribute.ensemble import (  # noqa: F401
        get_oof,
        get_single_oof,
    )
    from skdist_amd.parallel.cluster import TaskFailedError  # noqa: F401
    from skdist_amd.postprocessing import SimpleVoter  # noqa: F401
    from skdist_amd.preprocessing import (  # noqa: F401
        DenseTransformer,
        FeatureCast,
        HashingVectorizerChunked,
        ImputeNull,
        LabelEncoderPipe,
        MultihotEncoder,
        SelectField,
        SelectorMem,
        SparseTransformer,
    )
