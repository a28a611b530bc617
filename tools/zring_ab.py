"""A/B the plain vs software-pipelined persistent-z ring conv.

Run twice: CFX_ZRING_PL=0 (plain) and CFX_ZRING_PL=1 (pipelined) — the
variant choice is cached per process.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tools.conv_probe import probe  # noqa: E402

print('CFX_ZRING_PL =', os.environ.get('CFX_ZRING_PL', '(default 1)'),
      flush=True)
probe(28, 20, 256, 256, zring=True)
probe(36, 20, 128, 128, zring=True)
if '--c48' in sys.argv:
    probe(48, 20, 64, 64, zring=True)
    probe(48, 20, 64, 64)
