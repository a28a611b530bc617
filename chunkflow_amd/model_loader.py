"""Load a user model/plugin python file as a module (reference
chunkflow/lib/__init__.py:5-16 contract: plain source-file import)."""
import types
from importlib.machinery import SourceFileLoader


def load_source(fname: str):
    loader = SourceFileLoader('Model', fname)
    mod = types.ModuleType(loader.name)
    loader.exec_module(mod)
    return mod
