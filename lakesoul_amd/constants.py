"""Constants shared across the engine.

Compatibility sentinels mirror the reference's
``rust/lakesoul-io/src/constant.rs:17-23`` and
``rust/lakesoul-metadata/src/transfusion.rs:44-50``.
"""

HASH_SEED = 42  # Spark murmur3 seed (reference: utils/hash/mod.rs:22)

# Partition-desc encoding (reference: transfusion.rs:44-50, helpers/mod.rs:453-501)
NON_PARTITION_TABLE_PART_DESC = "-5"
RANGE_PARTITION_SPLITTER = ","
HASH_PARTITION_SPLITTER = ";"
PARTITION_SPLITTER_OF_RANGE_AND_HASH = ";"

# Null / special-value sentinels (reference: constant.rs:18-21)
LAKESOUL_NULL_STRING = "__L@KE$OUL_NULL__"
LAKESOUL_EMPTY_STRING = "__L@KE$OUL_EMPTY_STRING__"
LAKESOUL_EQ = "__L@KE$OUL_EQ__"
LAKESOUL_COMMA = "__L@KE$OUL_COMMA__"


def encode_partition_value(v) -> str:
    """Encode one range-partition value for a partition_desc.

    Mirrors the reference's format_scalar_value (helpers/mod.rs:206-221):
    NULL -> LAKESOUL_NULL_STRING, "" -> LAKESOUL_EMPTY_STRING, and '='/','
    escaped so the "col=val,col2=val2" desc stays parseable. (The
    reference's encoder at helpers/mod.rs:219 escapes ',' twice and '='
    never — its decoder at :325 expects EQ/COMMA sentinels, so we encode
    what the decoder expects.)"""
    if v is None:
        return LAKESOUL_NULL_STRING
    s = str(v)
    if s == "":
        return LAKESOUL_EMPTY_STRING
    return s.replace("=", LAKESOUL_EQ).replace(",", LAKESOUL_COMMA)


def decode_partition_value(s: str):
    """Inverse of encode_partition_value (reference helpers/mod.rs:325).
    Returns None for the NULL sentinel."""
    if s == LAKESOUL_NULL_STRING:
        return None
    if s == LAKESOUL_EMPTY_STRING:
        return ""
    return s.replace(LAKESOUL_EQ, "=").replace(LAKESOUL_COMMA, ",")

# CDC row-kind column values (reference: flink LakeSoulRecordConvert RowKind -> cdc column)
CDC_INSERT = "insert"
CDC_UPDATE = "update"
CDC_DELETE = "delete"

# Compaction directory convention (reference: merge/mod.rs:358-363)
COMPACT_DIR = "compactdir"

DEFAULT_NAMESPACE = "default"
DEFAULT_DOMAIN = "public"

# Writer defaults (reference: config/mod.rs:67-115, writer/mod.rs:224-245)
DEFAULT_BATCH_SIZE = 8192
DEFAULT_MAX_ROW_GROUP_SIZE = 250_000
DEFAULT_MAX_ROW_GROUP_NUM_VALUES = 2_147_483_647
DEFAULT_COMPRESSION = "zstd"
DEFAULT_COMPRESSION_LEVEL = 1
DEFAULT_HASH_BUCKET_NUM = 1
DEFAULT_PREFETCH_SIZE = 1
