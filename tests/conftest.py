import os
import sys

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO_ROOT not in sys.path:
    sys.path.insert(0, REPO_ROOT)


def pytest_configure(config):
    config.addinivalue_line(
        'markers', 'gpu: needs a real MI355X GPU (run with -m gpu)')
    config.addinivalue_line(
        'markers', 'slow: takes more than a few seconds on CPU')


def pytest_collection_modifyitems(config, items):
    # skip gpu tests automatically when no GPU is visible, so a plain
    # `pytest tests` run works both here (CPU) and on the GPU box
    try:
        import torch
        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip = pytest.mark.skip(reason='no GPU visible')
    for item in items:
        if 'gpu' in item.keywords:
            item.add_marker(skip)


@pytest.fixture(scope='session')
def golden():
    import json
    import numpy as np
    gdir = os.path.join(REPO_ROOT, 'tests', 'golden')
    with open(os.path.join(gdir, 'golden.json')) as f:
        meta = json.load(f)
    arrays = np.load(os.path.join(gdir, 'golden.npz'))
    return meta, arrays


@pytest.fixture(scope='session')
def golden_dir():
    return os.path.join(REPO_ROOT, 'tests', 'golden')
