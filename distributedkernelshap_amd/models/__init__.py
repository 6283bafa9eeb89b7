from .predictors import LinearPredictor, TorchPredictor, make_predictor  # noqa: F401
from .synthetic import SyntheticData, make_adult_like, make_tabular  # noqa: F401
