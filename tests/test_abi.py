"""The C-ABI shared library builds, loads, and exports every symbol the
public header declares (no GPU needed — hipcc cross-compiles and host-side
entry points run anywhere)."""
import ctypes
import os
import re

from chunkflow_amd.build import build, SO_PATH


def header_symbols():
    header = os.path.join(os.path.dirname(SO_PATH), '..', 'include',
                          'chunkflow_amd.h')
    with open(header) as f:
        text = f.read()
    return sorted(set(re.findall(r'\b(cfx_\w+)\s*\(', text)))


def test_build_and_symbols():
    path = build()
    lib = ctypes.CDLL(path)
    syms = header_symbols()
    assert len(syms) >= 18
    missing = [s for s in syms if not hasattr(lib, s)]
    assert not missing, f'missing C-ABI symbols: {missing}'


def test_python_binding_symbol_list_matches_header():
    from chunkflow_amd.hip import SYMBOLS
    assert sorted(SYMBOLS) == header_symbols()


def test_version_and_error_string():
    from chunkflow_amd.hip import load_library
    lib = load_library()
    assert lib.cfx_version() >= 1
    assert isinstance(lib.cfx_last_error(), bytes)
