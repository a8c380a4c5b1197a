"""incubator_pegasus_amd — MI355X-native (gfx950) re-implementation of the
apache/incubator-pegasus per-replica storage hot path: get / multi_get / scan
read handlers, k-way sorted-run merge, and TTL / user-rule compaction-filter
evaluation, as hand-written HIP kernels behind the reference's storage-engine
C-ABI boundary (include/rrdb_engine.h).

The product backend is incubator_pegasus_amd/csrc/librrdb_hip.so; it requires
a gfx950 GPU at runtime and fails loudly without one.  The CPU oracle
(oracle/liboracle.so) implements the same ABI but is test infrastructure only.
"""
import os

from . import capi  # noqa: F401
from .capi import (  # noqa: F401
    FT_MATCH_ANYWHERE,
    FT_MATCH_POSTFIX,
    FT_MATCH_PREFIX,
    FT_NO_FILTER,
    INCOMPLETE,
    INVALID_ARGUMENT,
    KIND_DELETE,
    KIND_PUT,
    NOT_FOUND,
    OK,
    SCAN_COMPLETED,
    RrdbLib,
    RrdbPartition,
)

_PKG_DIR = os.path.dirname(os.path.abspath(__file__))
HIP_LIB_PATH = os.path.join(_PKG_DIR, "csrc", "librrdb_hip.so")

_hip_lib = None


def hip_lib() -> RrdbLib:
    """The product engine library.  Raises if the extension is not built —
    there is no CPU fallback on the product path."""
    global _hip_lib
    if _hip_lib is None:
        if not os.path.exists(HIP_LIB_PATH):
            raise RuntimeError(
                f"HIP engine not built: {HIP_LIB_PATH} missing. "
                "Run __graft_entry__.build() (hipcc --offload-arch=gfx950).")
        _hip_lib = RrdbLib(HIP_LIB_PATH)
    return _hip_lib
