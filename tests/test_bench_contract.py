"""bench.py driver-contract guards that run WITHOUT a GPU: the module
imports cleanly, the argument surface matches the driver's launch line,
and a GPU-less invocation fails loudly (no silent CPU benchmark)."""
import os
import subprocess
import sys

import pytest
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_parse_args_defaults():
    sys.path.insert(0, REPO)
    import bench
    old = sys.argv
    try:
        sys.argv = ['bench.py']
        a = bench.parse_args()
    finally:
        sys.argv = old
    # the driver runs `python bench.py --gpus N --steps K --warmup W`
    assert a.gpus == 1 and a.steps >= 1 and a.warmup >= 1
    assert tuple(a.chunk_size) == (512, 512, 512)       # config-2 workload
    assert tuple(a.patch_size) == (20, 256, 256)
    assert tuple(a.overlap) == (4, 64, 64)
    assert a.batch_size == 12
    assert a.dtype == 'float32'


def test_driver_flags_accepted():
    sys.path.insert(0, REPO)
    import bench
    old = sys.argv
    try:
        sys.argv = ['bench.py', '--gpus', '8', '--steps', '3',
                    '--warmup', '1']
        a = bench.parse_args()
    finally:
        sys.argv = old
    assert (a.gpus, a.steps, a.warmup) == (8, 3, 1)


@pytest.mark.skipif(torch.cuda.is_available(), reason='GPU present')
def test_no_silent_cpu_fallback():
    r = subprocess.run([sys.executable, 'bench.py', '--steps', '1'],
                       capture_output=True, text=True, cwd=REPO,
                       env=dict(os.environ, PYTHONPATH=REPO), timeout=300)
    assert r.returncode != 0
    assert 'requires a GPU' in (r.stderr + r.stdout)
