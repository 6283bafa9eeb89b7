import numpy as np
import pytest


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that require an MI355X GPU (run via gpurun)"
    )


@pytest.fixture(scope="session")
def adult_like():
    from distributedkernelshap_amd.models import make_adult_like

    return make_adult_like(n_instances=8, n_background=25, seed=0)


@pytest.fixture(scope="session")
def linear_predictor(adult_like):
    from distributedkernelshap_amd.models import LinearPredictor

    return LinearPredictor.random(adult_like.X.shape[1], 2, seed=0)


@pytest.fixture
def rng():
    return np.random.Generator(np.random.Philox(key=[7, 7]))
