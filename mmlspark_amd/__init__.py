"""mmlspark_amd — MI355X-native distributed ML toolkit.

Brand-new framework with the capabilities of Azure/mmlspark (MMLSpark):
SparkML-shaped Estimator/Transformer API over pandas DataFrames, typed
Params with pipeline save/load, distributed gradient-boosted trees
(LightGBM-equivalent) with CDNA4 HIP histogram kernels and RCCL histogram
sync over xGMI, VW-equivalent hashed sparse online learning, Isolation
Forest, Conditional KNN, ResNet ImageFeaturizer on PyTorch-ROCm,
LIME/KernelSHAP explainers, and a low-latency HTTP serving path.
"""

__version__ = "0.1.0"

from .core.param import Param, Params  # noqa: F401
from .core.pipeline import (  # noqa: F401
    Estimator,
    Model,
    Pipeline,
    PipelineModel,
    PipelineStage,
    Transformer,
)
from .core.serialize import load_stage, save_stage  # noqa: F401


def _register_all():
    """Import every module that registers public stages (load()/fuzzing need this)."""
    from importlib import import_module
    for mod in (
        "mmlspark_amd.models.gbdt.estimators",
        "mmlspark_amd.models.vw.estimators",
        "mmlspark_amd.models.vw.featurizer",
        "mmlspark_amd.models.iforest",
        "mmlspark_amd.models.knn",
        "mmlspark_amd.models.sar",
        "mmlspark_amd.models.images",
        "mmlspark_amd.models.image_featurizer",
        "mmlspark_amd.explainers.lime",
        "mmlspark_amd.explainers.shap",
        "mmlspark_amd.stages.basic",
        "mmlspark_amd.stages.batching",
        "mmlspark_amd.stages.featurize",
        "mmlspark_amd.stages.text",
        "mmlspark_amd.stages.train",
        "mmlspark_amd.stages.automl",
        "mmlspark_amd.io_http.client",
        "mmlspark_amd.io_http.cognitive",
        "mmlspark_amd.io_http.files",
        "mmlspark_amd.models.cyber",
        "mmlspark_amd.stages.udfs",
    ):
        try:
            import_module(mod)
        except ImportError:
            pass  # module not built yet during incremental development
