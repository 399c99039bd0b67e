import os
import sys

import pytest

# repo root on sys.path so `opendiloco_amd` and `oracle` import without install
REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO_ROOT not in sys.path:
    sys.path.insert(0, REPO_ROOT)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an AMD GPU (MI355X) and the built HIP extension")


@pytest.fixture(scope="session")
def repo_root() -> str:
    return REPO_ROOT


@pytest.fixture(scope="session")
def fixture_2m(repo_root) -> str:
    path = os.path.join(repo_root, "tests", "models", "llama-2m")
    assert os.path.exists(os.path.join(path, "config.json"))
    return path


@pytest.fixture(scope="session")
def golden_dir(repo_root) -> str:
    return os.path.join(repo_root, "tests", "golden")
