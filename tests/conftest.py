import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU")
    config.addinivalue_line("markers", "distributed: multi-process test")


@pytest.fixture(autouse=True)
def _clear_state():
    yield
    from realhf_amd.base import constants

    constants.clear_grids()
