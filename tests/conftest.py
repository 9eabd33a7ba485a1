import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs an MI355X GPU (run via gpurun)")


@pytest.fixture
def vocab():
    from spacy_ray_amd.vocab.doc import Vocab

    return Vocab("en")
