"""greengage_amd — MI355X-native columnar query-operator engine.

Python face of the C-ABI engine library (libgreengage_engine.so, built
in-tree from csrc/ by `make -C greengage_amd/csrc` or
__graft_entry__.build()).  The compute path is the HIP/CDNA4 engine —
there is NO CPU fallback: on a machine with a GPU, a missing or
unloadable engine library raises immediately.
"""
from .engine import (  # noqa: F401
    Engine,
    EngineError,
    PGDate,
    pgdate,
)

__version__ = "0.1.0"
