from .kernel_shap import KernelShap, KernelExplainerWrapper  # noqa: F401
from .distributed import DistributedExplainer  # noqa: F401
