"""distributedkernelshap_amd — MI355X-native distributed KernelSHAP engine.

Brand-new framework with the capabilities of alexcoca/DistributedKernelShap:
the KernelSHAP inner loop runs as hand-written CDNA4 HIP kernels (MFMA-tiled
fused masked-background predict + batched constrained WLS), instances are
sharded data-parallel across MI355X GPUs with RCCL over xGMI, and the
``explainers.kernel_shap`` / ``explainers.distributed`` Python API and
``shap_values`` output layout stay compatible with the reference.
"""
__version__ = "0.1.0"

from .interface import Explanation, Explainer, FitMixin  # noqa: F401
from .explainers.kernel_shap import (  # noqa: F401
    KernelShap,
    KernelExplainerWrapper,
    rank_by_importance,
    sum_categories,
)
from .explainers.distributed import DistributedExplainer  # noqa: F401
from .core.engine import KernelShapEngine  # noqa: F401
