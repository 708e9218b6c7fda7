import os
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "oracle"))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an AMD GPU (MI355X)")


@pytest.fixture(scope="session")
def oracle():
    import pywrap
    pywrap.build()
    return pywrap


@pytest.fixture(scope="session")
def golden():
    import json

    def load(name):
        with open(os.path.join(REPO, "tests", "golden", name)) as f:
            return json.load(f)

    return load


@pytest.fixture(scope="session")
def gpu():
    from spectre_amd import SpectreGpu
    g = SpectreGpu([0])
    yield g
    g.close()
