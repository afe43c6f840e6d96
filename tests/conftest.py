import os
import sys

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO_ROOT not in sys.path:
    sys.path.insert(0, REPO_ROOT)

REFERENCE_DIR = "/root/reference"


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU (run via gpurun)")
    config.addinivalue_line(
        "markers",
        "reference: requires the read-only reference checkout at /root/reference "
        "(dev-container only; skipped elsewhere)",
    )


def pytest_collection_modifyitems(config, items):
    have_ref = os.path.isdir(REFERENCE_DIR)
    skip_ref = pytest.mark.skip(reason="reference checkout not available")
    for item in items:
        if "reference" in item.keywords and not have_ref:
            item.add_marker(skip_ref)


@pytest.fixture(scope="session")
def data_dir():
    return os.path.join(REPO_ROOT, "tests", "data")


@pytest.fixture(scope="session")
def sample_profile_dir():
    """Synthetic MI355X-shaped sample profiles generated for tests."""
    return os.path.join(REPO_ROOT, "tests", "data", "profiles_synth")
