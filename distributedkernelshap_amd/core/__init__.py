from .engine import KernelShapEngine  # noqa: F401
from .sampler import plan_coalitions, sample_masks, default_nsamples  # noqa: F401
from .solver import solve_wls  # noqa: F401
