"""Stub-import harness for running the reference in THIS build container.

TEST INFRASTRUCTURE ONLY (see oracle/__init__.py). The reference
(seung-lab/chunkflow v1.1.7 at /root/reference) is pure Python, but imports
several packages that are absent here (h5py, tifffile, cc3d, cloudvolume,
skimage). The hot path never calls them for in-memory uint8/f32 chunks
(SURVEY.md §4 'Verified in this container'), so tiny stub modules are enough
to import and run the real Inferencer as a golden-vector generator.

Nothing in here runs on the GPU box: /root/reference does not exist there.
Golden vectors are generated here (oracle/gen_golden.py) and committed as
fixtures under tests/golden/.
"""
import sys
import types

REFERENCE_PATH = '/root/reference'


def reference_available() -> bool:
    import os
    return os.path.isdir(REFERENCE_PATH)


def install_stubs():
    def stub(name, **attrs):
        if name in sys.modules:
            return sys.modules[name]
        mod = types.ModuleType(name)
        for k, v in attrs.items():
            setattr(mod, k, v)
        sys.modules[name] = mod
        return mod

    class _Bbox:  # only used for isinstance checks in Chunk.cutout
        pass

    class _Vec(tuple):
        def __new__(cls, *args):
            return super().__new__(cls, args)

    stub('h5py')
    stub('tifffile')
    stub('cc3d')
    stub('cv2')  # chunk/image/adjust_grey.py imports it at module scope
    cv = stub('cloudvolume', CloudVolume=None)
    cvlib = stub('cloudvolume.lib', Bbox=_Bbox, Vec=_Vec,
                 yellow=lambda s: s)
    cv.lib = cvlib
    skimage = stub('skimage')
    feat = stub('skimage.feature', match_template=None)
    skimage.feature = feat


def import_reference():
    """Import the reference chunkflow from /root/reference with stubs.

    Returns the (Inferencer class, Chunk class, make_patch_mask fn) triple.
    """
    install_stubs()
    if REFERENCE_PATH not in sys.path:
        sys.path.insert(0, REFERENCE_PATH)
    from chunkflow.flow.divid_conquer.inferencer import Inferencer
    from chunkflow.chunk import Chunk
    from chunkflow.flow.divid_conquer.patch.patch_mask import make_patch_mask
    return Inferencer, Chunk, make_patch_mask
