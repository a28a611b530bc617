"""Plugin operator: run a user python file's `execute(*inputs, **kwargs)`
over named task chunks.

Keeps the reference contract (chunkflow/flow/plugin.py): the file is searched
in ./, the package plugins dir and $CHUNKFLOW_PLUGIN_DIR (:45-53); ndarray
outputs are re-wrapped as Chunks with the symmetric-crop offset correction
(:19-26); string args of the form "k=v;k2=v2" become kwargs (:74-76).
"""
import os
import os.path as path
from typing import Union

import numpy as np

from .chunk import Chunk
from .cartesian import Cartesian
from .model_loader import load_source


def str_to_dict(s: str) -> dict:
    d = {}
    for item in s.split(';'):
        if not item:
            continue
        k, v = item.split('=', 1)
        try:
            v = eval(v, {'__builtins__': {}})
        except Exception:
            pass
        d[k] = v
    return d


def array_to_chunk(arr: Union[np.ndarray, Chunk], voxel_offset: Cartesian,
                   voxel_size: Cartesian, shape: tuple):
    if isinstance(arr, np.ndarray):
        # if the plugin symmetric-cropped, shift the offset accordingly
        offset = tuple(vo + (ins - outs) // 2 for vo, ins, outs in
                       zip(voxel_offset, shape[-3:], arr.shape[-3:]))
        return Chunk(arr, voxel_offset=offset, voxel_size=voxel_size)
    return arr


class Plugin:
    def __init__(self, plugin_file_name: str, name: str = 'plugin-1'):
        self.name = name
        if not plugin_file_name.endswith('.py'):
            plugin_file_name += '.py'
        pkg_plugin_dir = path.join(
            path.dirname(path.realpath(__file__)), 'plugins')
        plugin_dirs = ['./', pkg_plugin_dir]
        if 'CHUNKFLOW_PLUGIN_DIR' in os.environ:
            plugin_dirs.append(os.environ['CHUNKFLOW_PLUGIN_DIR'])
        fname = plugin_file_name
        for d in plugin_dirs:
            cand = path.join(d, plugin_file_name)
            if path.exists(cand):
                fname = cand
                break
        assert path.exists(fname), f'did not find plugin: {plugin_file_name}'
        program = load_source(fname)
        assert hasattr(program, 'execute'), \
            f'plugin {fname} must define execute(*inputs, **kwargs)'
        self.execute = program.execute

    def __call__(self, inputs: list, args: str = None):
        voxel_offset = voxel_size = shape = None
        for inp in inputs:
            if isinstance(inp, Chunk):
                voxel_offset = inp.voxel_offset
                voxel_size = inp.voxel_size
                shape = inp.shape
                break
        if args is not None and '=' in args:
            args = str_to_dict(args)

        if len(inputs) == 0 and args is None:
            outputs = self.execute()
        elif len(inputs) == 0:
            outputs = (self.execute(args=args) if isinstance(args, str)
                       else self.execute(**args))
        elif args is None:
            outputs = self.execute(*inputs)
        elif isinstance(args, str):
            outputs = self.execute(*inputs, args=args)
        else:
            outputs = self.execute(*inputs, **args)

        if isinstance(outputs, tuple):
            outputs = [*outputs]
        if voxel_offset is not None and outputs is not None:
            if isinstance(outputs, list):
                outputs = [array_to_chunk(o, voxel_offset, voxel_size, shape)
                           for o in outputs]
            elif isinstance(outputs, np.ndarray):
                outputs = array_to_chunk(outputs, voxel_offset, voxel_size,
                                         shape)
        return outputs
