from .app import KernelShapModel, BatchKernelShapModel, create_app  # noqa: F401
