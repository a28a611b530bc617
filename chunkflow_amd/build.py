"""Build the in-tree HIP extension (gfx950)."""
import os
import subprocess

PKG_DIR = os.path.dirname(os.path.abspath(__file__))
SO_PATH = os.path.join(PKG_DIR, 'libchunkflow_amd.so')
SRC = os.path.join(PKG_DIR, 'csrc', 'cfx.hip')
SRC_CC = os.path.join(PKG_DIR, 'csrc', 'cc.hip')
SRC_IMG = os.path.join(PKG_DIR, 'csrc', 'image.hip')
SRC_CONV = os.path.join(PKG_DIR, 'csrc', 'conv.hip')
SRC_UPDOWN = os.path.join(PKG_DIR, 'csrc', 'updown.hip')
SRC_INT = os.path.join(PKG_DIR, 'csrc', 'cfx_internal.h')
HEADER = os.path.join(PKG_DIR, '..', 'include', 'chunkflow_amd.h')


def so_is_fresh() -> bool:
    if not os.path.exists(SO_PATH):
        return False
    so_mtime = os.path.getmtime(SO_PATH)
    return all(os.path.getmtime(p) <= so_mtime
               for p in (SRC, SRC_CC, SRC_IMG, SRC_CONV, SRC_UPDOWN, SRC_INT,
                         HEADER))


def build(force: bool = False) -> str:
    """Compile chunkflow_amd/csrc/cfx.hip -> libchunkflow_amd.so in-tree.

    hipcc cross-compiles for gfx950 without a GPU; the .so ships to the GPU
    box inside the repo snapshot.
    """
    if not force and so_is_fresh():
        return SO_PATH
    cmd = [
        'hipcc', '--offload-arch=gfx950', '-O3', '-std=c++17',
        '-ffp-contract=off', '-fPIC', '-shared', SRC, SRC_CC, SRC_IMG,
        SRC_CONV, SRC_UPDOWN, '-o', SO_PATH,
    ]
    subprocess.run(cmd, check=True)
    return SO_PATH


if __name__ == '__main__':
    print(build(force=True))
