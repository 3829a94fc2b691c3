import pytest


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run on the GPU box)")


@pytest.fixture
def small_batch():
    from distegnn_amd.data.graph import collate
    from distegnn_amd.data.synthetic import make_cutoff_dataset

    return collate(make_cutoff_dataset("nbody_100", 3, seed=0))
