import os
import sys

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO_ROOT not in sys.path:
    sys.path.insert(0, REPO_ROOT)


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test needs a real MI355X GPU (run via gpurun)"
    )


@pytest.fixture(scope="session")
def golden():
    """Committed golden fixtures: {'fixtures': {...}, 'answers': {...}},
    with file paths resolved to absolute."""
    import json

    gpath = os.path.join(REPO_ROOT, "tests", "golden", "answers.json")
    with open(gpath) as fh:
        g = json.load(fh)
    gdir = os.path.join(REPO_ROOT, "tests", "golden", "data")
    for fx in g["fixtures"].values():
        fx["files"] = [os.path.join(gdir, f) for f in fx["files"]]
    return g
