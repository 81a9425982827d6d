/* lakesoul_amd C ABI — engine-connector surface (torch-free).
 *
 * MI355X-native analog of the reference's rust/lakesoul-io-c
 * (lib.rs:113-1324): config-builder setters, reader create/next via the
 * Arrow C Data Interface, writer create/write/flush, spark-murmur3
 * helpers. JVM/other engines dlopen liblakesoul_amd_c.so and move
 * batches zero-copy through ArrowArray/ArrowSchema, exactly as the
 * reference's JNR-FFI binding does (native-io/lakesoul-io-java).
 *
 * Scope (round 1): per-bucket merge-on-read reads with UseLast dedup for
 * integer primary keys, pass-through reads, and sorted bucket writes.
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

/* ---- reader ---- */
/* config: file paths in snapshot order (oldest..newest), requested
 * column names, primary-key names (empty => pass-through concat). */
typedef struct LakesoulCReader LakesoulCReader;

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

/* ---- spark murmur3 helpers (bit-exact with utils/hash) ---- */
uint32_t lakesoul_c_murmur3_bytes(const uint8_t* data, int64_t len, uint32_t seed);
uint32_t lakesoul_c_murmur3_i32(int32_t v, uint32_t seed);
uint32_t lakesoul_c_murmur3_i64(int64_t v, uint32_t seed);

#ifdef __cplusplus
}
#endif
#endif
