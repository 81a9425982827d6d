/* C consumer smoke for liblakesoul_amd_c.so — stands in for the
 * reference's JNR-FFI Java consumer (native-io/lakesoul-io-java/.../
 * NativeIOWriter.java / NativeIOReader.java): a FOREIGN runtime dlopens
 * the library, builds Arrow C Data structures itself, writes two
 * overlapping PK files, and reads them back merged with a merge
 * operator, a filter, and the async callback API.
 *
 * Build: gcc -O2 capi_smoke.c -o capi_smoke -ldl -lpthread
 * Run:   ./capi_smoke /path/to/liblakesoul_amd_c.so /tmp/workdir
 */
#include <dlfcn.h>
#include <pthread.h>
#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

/* Arrow C Data Interface (stable ABI) */
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

typedef struct LakesoulCFlushResult {
  char* path;
  int64_t size;
  int64_t rows;
  char* exist_cols;
} FlushResult;

/* function pointers */
static void* (*cfg_create)(void);
static int (*cfg_add_file)(void*, const char*);
static int (*cfg_add_pk)(void*, const char*);
static int (*cfg_add_merge_op)(void*, const char*, const char*);
static int (*cfg_add_filter)(void*, const char*);
static int (*cfg_set_option)(void*, const char*, const char*);
static const char* (*cfg_get_option)(void*, const char*);
static void (*cfg_free)(void*);
static void* (*rd_from_cfg)(void*);
static int (*rd_start)(void*);
static int (*rd_next)(void*, struct ArrowArray*);
static int (*rd_next_async)(void*, struct ArrowArray*,
                            void (*)(void*, int, const char*), void*);
static void (*rd_close)(void*);
static void* (*wr_from_cfg)(void*, const char*);
static int (*wr_set_schema)(void*, struct ArrowSchema*);
static int (*wr_write)(void*, struct ArrowArray*);
static int (*wr_flush)(void*, FlushResult*);
static void (*fr_free)(FlushResult*);
static const char* (*last_err)(void);

#define DIE(msg)                                             \
  do {                                                       \
    fprintf(stderr, "FAIL %s:%d %s (%s)\n", __FILE__,        \
            __LINE__, msg, last_err ? last_err() : "");      \
    exit(1);                                                 \
  } while (0)

#define CHECK(cond, msg) \
  do {                   \
    if (!(cond)) DIE(msg); \
  } while (0)

static void noop_release_schema(struct ArrowSchema* s) { s->release = NULL; }
static void noop_release_array(struct ArrowArray* a) { a->release = NULL; }

/* build a 2-column struct schema {id:int64 "l", v:double "g"} */
static struct ArrowSchema* make_schema(void) {
  static struct ArrowSchema root, c0, c1;
  static struct ArrowSchema* kids[2];
  memset(&root, 0, sizeof root);
  memset(&c0, 0, sizeof c0);
  memset(&c1, 0, sizeof c1);
  c0.format = "l"; c0.name = "id"; c0.flags = 2; c0.release = noop_release_schema;
  c1.format = "g"; c1.name = "v"; c1.flags = 2; c1.release = noop_release_schema;
  kids[0] = &c0; kids[1] = &c1;
  root.format = "+s"; root.name = ""; root.n_children = 2; root.children = kids;
  root.release = noop_release_schema;
  return &root;
}

static void write_file(const char* path, const int64_t* ids, const double* vs,
                       int64_t n) {
  void* cfg = cfg_create();
  CHECK(cfg_set_option(cfg, "compression", "zstd") == 0, "set_option");
  void* w = wr_from_cfg(cfg, path);
  cfg_free(cfg);
  CHECK(w != NULL, "writer_create_from_config");
  CHECK(wr_set_schema(w, make_schema()) == 0, "set_schema");

  struct ArrowArray root, a0, a1;
  struct ArrowArray* kids[2] = {&a0, &a1};
  const void* b0[2] = {NULL, ids};
  const void* b1[2] = {NULL, vs};
  const void* br[1] = {NULL};
  memset(&root, 0, sizeof root);
  memset(&a0, 0, sizeof a0);
  memset(&a1, 0, sizeof a1);
  a0.length = n; a0.n_buffers = 2; a0.buffers = b0; a0.release = noop_release_array;
  a1.length = n; a1.n_buffers = 2; a1.buffers = b1; a1.release = noop_release_array;
  root.length = n; root.n_buffers = 1; root.buffers = br;
  root.n_children = 2; root.children = kids; root.release = noop_release_array;
  CHECK(wr_write(w, &root) == 0, "writer_write");

  FlushResult fr;
  CHECK(wr_flush(w, &fr) == 0, "writer_flush");
  CHECK(fr.size > 0, "flush size");
  CHECK(fr.rows == n, "flush rows");
  CHECK(strcmp(fr.exist_cols, "id,v") == 0, "flush exist_cols");
  CHECK(strcmp(fr.path, path) == 0, "flush path");
  fr_free(&fr);
}

struct AsyncState {
  pthread_mutex_t mu;
  pthread_cond_t cv;
  int done;
  int rc;
};

static void async_cb(void* user, int rc, const char* err) {
  struct AsyncState* st = (struct AsyncState*)user;
  (void)err;
  pthread_mutex_lock(&st->mu);
  st->rc = rc;
  st->done = 1;
  pthread_cond_signal(&st->cv);
  pthread_mutex_unlock(&st->mu);
}

int main(int argc, char** argv) {
  if (argc < 3) {
    fprintf(stderr, "usage: %s <libpath> <workdir>\n", argv[0]);
    return 2;
  }
  void* h = dlopen(argv[1], RTLD_NOW | RTLD_LOCAL);
  if (!h) {
    fprintf(stderr, "dlopen: %s\n", dlerror());
    return 2;
  }
#define LOAD(var, name)                 \
  do {                                  \
    *(void**)(&var) = dlsym(h, name);   \
    if (!var) DIE("dlsym " name);       \
  } while (0)
  LOAD(last_err, "lakesoul_c_last_error");
  LOAD(cfg_create, "lakesoul_c_config_create");
  LOAD(cfg_add_file, "lakesoul_c_config_add_file");
  LOAD(cfg_add_pk, "lakesoul_c_config_add_primary_key");
  LOAD(cfg_add_merge_op, "lakesoul_c_config_add_merge_op");
  LOAD(cfg_add_filter, "lakesoul_c_config_add_filter");
  LOAD(cfg_set_option, "lakesoul_c_config_set_option");
  LOAD(cfg_get_option, "lakesoul_c_config_get_option");
  LOAD(cfg_free, "lakesoul_c_config_free");
  LOAD(rd_from_cfg, "lakesoul_c_reader_create_from_config");
  LOAD(rd_start, "lakesoul_c_reader_start");
  LOAD(rd_next, "lakesoul_c_reader_next");
  LOAD(rd_next_async, "lakesoul_c_reader_next_async");
  LOAD(rd_close, "lakesoul_c_reader_close");
  LOAD(wr_from_cfg, "lakesoul_c_writer_create_from_config");
  LOAD(wr_set_schema, "lakesoul_c_writer_set_schema");
  LOAD(wr_write, "lakesoul_c_writer_write");
  LOAD(wr_flush, "lakesoul_c_writer_flush");
  LOAD(fr_free, "lakesoul_c_flush_result_free");

  char f1[1024], f2[1024];
  snprintf(f1, sizeof f1, "%s/base.parquet", argv[2]);
  snprintf(f2, sizeof f2, "%s/delta.parquet", argv[2]);

  /* base: ids 0..99 v=1.0; delta: even ids 0..98 v=10.0 (sorted by pk) */
  int64_t ids1[100]; double vs1[100];
  for (int i = 0; i < 100; i++) { ids1[i] = i; vs1[i] = 1.0; }
  int64_t ids2[50]; double vs2[50];
  for (int i = 0; i < 50; i++) { ids2[i] = 2 * i; vs2[i] = 10.0; }
  write_file(f1, ids1, vs1, 100);
  write_file(f2, ids2, vs2, 50);

  /* merged read: SumAll on v, filter id < 10 */
  void* cfg = cfg_create();
  cfg_add_file(cfg, f1);
  cfg_add_file(cfg, f2);
  cfg_add_pk(cfg, "id");
  CHECK(cfg_add_merge_op(cfg, "v", "SumAll") == 0, "add_merge_op");
  CHECK(cfg_add_merge_op(cfg, "v", "NoSuchOp") != 0, "bad op must fail");
  CHECK(cfg_add_merge_op(cfg, "v", "SumAll") == 0, "re-add merge_op");
  CHECK(cfg_add_filter(cfg, "lt(id, 10)") == 0, "add_filter");
  CHECK(cfg_set_option(cfg, "batch_size", "7") == 0, "batch_size opt");
  CHECK(strcmp(cfg_get_option(cfg, "batch_size"), "7") == 0, "get_option");
  void* r = rd_from_cfg(cfg);
  cfg_free(cfg);
  CHECK(r != NULL, "reader_create_from_config");
  CHECK(rd_start(r) == 0, "reader_start");

  int64_t seen = 0;
  double sum_v = 0;
  for (;;) {
    struct ArrowArray out;
    int rc = rd_next(r, &out);
    CHECK(rc >= 0, "reader_next");
    if (rc == 0) break;
    CHECK(out.n_children == 2, "n_children");
    const int64_t* ids = (const int64_t*)out.children[0]->buffers[1];
    const double* vs = (const double*)out.children[1]->buffers[1];
    for (int64_t i = 0; i < out.length; i++) {
      CHECK(ids[i] == seen, "merged ids must be 0..9 in order");
      double expect = (ids[i] % 2 == 0) ? 11.0 : 1.0; /* SumAll across files */
      CHECK(vs[i] == expect, "SumAll value");
      sum_v += vs[i];
      seen++;
    }
    out.release(&out);
  }
  CHECK(seen == 10, "filter id<10 row count");
  (void)sum_v;
  rd_close(r);

  /* async read path */
  cfg = cfg_create();
  cfg_add_file(cfg, f1);
  cfg_add_pk(cfg, "id");
  r = rd_from_cfg(cfg);
  cfg_free(cfg);
  CHECK(rd_start(r) == 0, "reader_start async");
  struct AsyncState st;
  pthread_mutex_init(&st.mu, NULL);
  pthread_cond_init(&st.cv, NULL);
  int64_t total_async = 0;
  for (;;) {
    struct ArrowArray out;
    st.done = 0;
    CHECK(rd_next_async(r, &out, async_cb, &st) == 0, "next_async");
    pthread_mutex_lock(&st.mu);
    while (!st.done) pthread_cond_wait(&st.cv, &st.mu);
    pthread_mutex_unlock(&st.mu);
    CHECK(st.rc >= 0, "async rc");
    if (st.rc == 0) break;
    total_async += out.length;
    out.release(&out);
  }
  CHECK(total_async == 100, "async total rows");
  rd_close(r);

  printf("capi_smoke OK\n");
  return 0;
}
