"""dgraph_amd — MI355X-native posting-list set-algebra engine.

From-scratch GPU replacement for dgraph's algo/uidlist.go + codec/ hot path
(see DESIGN.md).  The compute path is hand-written HIP for gfx950 behind the
C-ABI in include/uidalgo.h; this package is the host-side mirror of the
reference's `algo` interface plus batching/sharding plumbing.

The GPU extension is mandatory on a GPU box: there is NO CPU fallback here —
ops raise if libuidalgo.so is missing or no device is visible.
"""
from dgraph_amd._lib import lib, UA_OK  # noqa: F401
from dgraph_amd import algo  # noqa: F401

__version__ = "0.1.0"
