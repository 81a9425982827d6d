/* lakesoul_amd C ABI — engine-connector surface (torch-free).
 *
 * MI355X-native analog of the reference's rust/lakesoul-io-c
 * (lib.rs:113-1324): config-builder setters, reader create/next via the
 * Arrow C Data Interface, writer create/write/flush, spark-murmur3
 * helpers. JVM/other engines dlopen liblakesoul_amd_c.so and move
 * batches zero-copy through ArrowArray/ArrowSchema, exactly as the
 * reference's JNR-FFI binding does (native-io/lakesoul-io-java).
 *
 * Surface parity with lakesoul-io-c (lib.rs:113-1324): config-builder
 * option map, multi-column int/string PKs, the full merge-operator set
 * (UseLast, UseLastNotNull, SumAll, SumLast, JoinedAll/LastBy...),
 * filter pushdown (string DSL and Substrait protobuf bytes), CDC
 * delete-row handling, blocking + callback-async reads via the Arrow C
 * Data Interface, and writer flush results {path,size,rows,exist_cols}.
 */
#ifndef LAKESOUL_AMD_C_H
#define LAKESOUL_AMD_C_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- Arrow C Data Interface (stable ABI, arrow.apache.org) ---- */
struct ArrowSchema {
  const char* format;
  const char* name;
  const char* metadata;
  int64_t flags;
  int64_t n_children;
  struct ArrowSchema** children;
  struct ArrowSchema* dictionary;
  void (*release)(struct ArrowSchema*);
  void* private_data;
};

struct ArrowArray {
  int64_t length;
  int64_t null_count;
  int64_t offset;
  int64_t n_buffers;
  int64_t n_children;
  const void** buffers;
  struct ArrowArray** children;
  struct ArrowArray* dictionary;
  void (*release)(struct ArrowArray*);
  void* private_data;
};

/* ---- error model (reference: CResult/CStatus) ---- */
/* functions return 0 on success; lakesoul_c_last_error() describes failures */
const char* lakesoul_c_last_error(void);

/* ---- config builder (reference: lakesoul-io-c lib.rs:113-614) ---- */
/* Files in snapshot order (oldest..newest), requested columns, PKs,
 * per-column merge operators, filters, and an untyped option map with
 * LAKESOUL_<KEY> env fallback (reference config/mod.rs:160-165).
 * Recognized options: batch_size, max_row_group_size, compression,
 * compression_level, hash_bucket_num, hash_bucket_id, cdc_column,
 * skip_merge_on_read ("1"/"true"). Unknown keys are stored (queryable
 * via get_option) and ignored by the engine. */
typedef struct LakesoulCConfig LakesoulCConfig;

LakesoulCConfig* lakesoul_c_config_create(void);
int lakesoul_c_config_add_file(LakesoulCConfig*, const char* path);
int lakesoul_c_config_add_column(LakesoulCConfig*, const char* name);
int lakesoul_c_config_add_primary_key(LakesoulCConfig*, const char* name);
/* op: UseLast | UseLastNotNull | SumAll | SumLast | JoinedAllByComma |
 * JoinedAllBySemicolon | JoinedLastByComma | JoinedLastBySemicolon */
int lakesoul_c_config_add_merge_op(LakesoulCConfig*, const char* col, const char* op);
/* filter in the reference's string DSL, e.g. "and(gt(id, 10), eq(s, 'x'))";
 * multiple calls AND together (reference parser.rs:52-120) */
int lakesoul_c_config_add_filter(LakesoulCConfig*, const char* dsl);
/* filter as Substrait protobuf bytes (Plan or ExtendedExpression, the
 * encodings Spark/Flink push; reference parser.rs:44-49 FilterContainer) */
int lakesoul_c_config_set_filter_substrait(LakesoulCConfig*, const uint8_t* buf, int64_t len);
int lakesoul_c_config_set_option(LakesoulCConfig*, const char* key, const char* value);
/* returns NULL if the key is unset and no LAKESOUL_<KEY> env exists */
const char* lakesoul_c_config_get_option(LakesoulCConfig*, const char* key);
void lakesoul_c_config_free(LakesoulCConfig*);

/* ---- reader ---- */
/* config: file paths in snapshot order (oldest..newest), requested
 * column names, primary-key names (empty => pass-through concat). */
typedef struct LakesoulCReader LakesoulCReader;

/* build a reader from a config (files/columns/pks/merge-ops/filters/
 * options); call lakesoul_c_reader_start next */
LakesoulCReader* lakesoul_c_reader_create_from_config(const LakesoulCConfig*);

LakesoulCReader* lakesoul_c_reader_create(void);
int lakesoul_c_reader_add_file(LakesoulCReader*, const char* path);
int lakesoul_c_reader_add_column(LakesoulCReader*, const char* name);
int lakesoul_c_reader_add_primary_key(LakesoulCReader*, const char* name);
int lakesoul_c_reader_set_batch_size(LakesoulCReader*, int64_t rows);
/* resolves schemas + runs the merge; call once before next_batch */
int lakesoul_c_reader_start(LakesoulCReader*);
/* exports the result schema (caller owns: call schema->release) */
int lakesoul_c_reader_schema(LakesoulCReader*, struct ArrowSchema* out);
/* next batch as an Arrow struct array; returns 1 = batch written,
 * 0 = end of stream, -1 = error */
int lakesoul_c_reader_next(LakesoulCReader*, struct ArrowArray* out);
/* callback-async next (reference lib.rs next_record_batch callback form):
 * fills *out on a background thread, then invokes cb(user, rc, err) with
 * rc 1 = batch written, 0 = end of stream, -1 = error (err non-NULL).
 * One in-flight call per reader; returns 0 if the job was queued. */
int lakesoul_c_reader_next_async(LakesoulCReader*, struct ArrowArray* out,
                                 void (*cb)(void* user, int rc, const char* err),
                                 void* user);
void lakesoul_c_reader_close(LakesoulCReader*);

/* ---- writer ---- */
typedef struct LakesoulCWriter LakesoulCWriter;

LakesoulCWriter* lakesoul_c_writer_create(const char* path);
int lakesoul_c_writer_set_compression(LakesoulCWriter*, const char* codec, int level);
int lakesoul_c_writer_set_row_group_size(LakesoulCWriter*, int64_t rows);
/* schema: struct schema describing the columns (consumed, not released) */
int lakesoul_c_writer_set_schema(LakesoulCWriter*, struct ArrowSchema* schema);
/* write one batch (struct array matching the schema); data is copied */
int lakesoul_c_writer_write(LakesoulCWriter*, struct ArrowArray* batch);
/* finish the file; returns file size in bytes or -1 */
int64_t lakesoul_c_writer_close(LakesoulCWriter*);
void lakesoul_c_writer_abort(LakesoulCWriter*);

/* flush result (reference FlushOutput {path,size,rows,exist_cols},
 * async_writer/mod.rs:49; NativeIOWriter.java:240 decodes the same). */
typedef struct LakesoulCFlushResult {
  char* path;        /* malloc'd; free via lakesoul_c_flush_result_free */
  int64_t size;      /* file bytes */
  int64_t rows;      /* rows written */
  char* exist_cols;  /* comma-joined column names present in the file */
} LakesoulCFlushResult;

/* close the writer AND report what was written; returns 0 on success
 * (writer is consumed either way) */
int lakesoul_c_writer_flush(LakesoulCWriter*, LakesoulCFlushResult* out);
void lakesoul_c_flush_result_free(LakesoulCFlushResult*);

/* writer from config: compression/row-group options come from the map */
LakesoulCWriter* lakesoul_c_writer_create_from_config(const LakesoulCConfig*,
                                                      const char* path);

/* ---- spark murmur3 helpers (bit-exact with utils/hash) ---- */
uint32_t lakesoul_c_murmur3_bytes(const uint8_t* data, int64_t len, uint32_t seed);
uint32_t lakesoul_c_murmur3_i32(int32_t v, uint32_t seed);
uint32_t lakesoul_c_murmur3_i64(int64_t v, uint32_t seed);

#ifdef __cplusplus
}
#endif
#endif
