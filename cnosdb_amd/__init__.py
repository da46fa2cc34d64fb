"""cnosdb_amd — MI355X-native TSM columnar scan/decode engine.

The GPU replacement for CnosDB's tskv hot path: TSM
DataBlock codec decode (DeltaTs/Delta/Gorilla/BitPack), tombstone masking,
closed-interval time-range filtering and downsampling aggregates, executed
by hand-written HIP kernels for CDNA4/gfx950 behind the C ABI declared in
include/cnosdb_gs.h.

The product path REQUIRES the HIP extension and a GPU: there is no CPU
fallback (the CPU restatement under oracle/ is test infrastructure only).
"""
from .host import (
    Engine,
    GroupSet,
    PageLib,
    lib_path,
    encode_ts,
    encode_i64,
    encode_f64,
    encode_bool,
    encode_str,
    str_page_of,
    build_page,
    page_of,
    prune_column_groups,
    groupby_tag,
    scan_fields,
    scan_fields_async,
    CT_TIME,
    CT_I64,
    CT_F64,
    CT_BOOL,
    CT_U64,
    CT_STR,
)

__all__ = [
    "Engine", "GroupSet", "PageLib", "lib_path",
    "encode_ts", "encode_i64", "encode_f64", "encode_bool", "encode_str",
    "build_page", "page_of", "str_page_of", "prune_column_groups", "groupby_tag", "scan_fields", "scan_fields_async",
    "CT_TIME", "CT_I64", "CT_F64", "CT_BOOL", "CT_U64", "CT_STR",
]
__version__ = "0.1"
