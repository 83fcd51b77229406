"""opengemini_amd — MI355X-native TSSP scan-and-aggregate engine.

The product package: a from-scratch gfx950/CDNA4 engine for openGemini's
immutable-columnar decode + GROUP BY time aggregation hot path, behind the
reference's cursor surface (engine/comm/cursor.go:46 comm.KeyCursor).
Requires a GPU — there is no CPU fallback; use oracle/ (test infrastructure)
for CPU checking only.
"""

from .engine import (  # noqa: F401
    AGG_ROW_DTYPE,
    SEG_DESC_DTYPE,
    GemxError,
    Shard,
    AggCursor,
    device_count,
    abi_version,
    build_extension,
    encode_shard,
)
