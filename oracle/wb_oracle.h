/* wb_oracle.h — CPU oracle for the rocksplicator slave-side WriteBatch apply path.
 *
 * TEST INFRASTRUCTURE ONLY. This library is the parity checker for the GPU
 * apply path: only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline
 * leg may call it. It is never the shipped or measured product path.
 *
 * PARITY STATUS: partially unpinned. The algorithm restated here follows
 *   - reference apply semantics: rocksdb_replicator/rocksdb_wrapper.cpp:13-28
 *     (re-wrap rep bytes, append LogData(8-byte timestamp), DB::Write) and
 *     rocksdb_replicator/replicated_db.cpp:369-383 (per-update loop),
 *   - seq accounting pinned by rocksdb_replicator/tests/
 *     rocksdb_assumption_test.cpp:136-187 (Put/Delete/Merge consume exactly one
 *     seq each; Write consumes Count() seqs; Get consumes none; seq starts at 0),
 *   - the WriteBatch rep byte layout of the UN-VENDORED third-party dependency
 *     rocksdb 5.7.fb @ cfaeb58 (docker/Dockerfile:259-284): db/write_batch.cc
 *     — 12-byte header (fixed64 LE seq + fixed32 LE count), then records
 *     tag(1B) + varint32-length-prefixed slices. The reference tree holds no
 *     golden byte vectors for this layout, so the layout itself is pinned only
 *     transitively: hand-built known-answer vectors in tests/golden/ +
 *     encoder/decoder round-trip + replay tests mirroring
 *     rocksdb_assumption_test.cpp:329-432.
 */
#ifndef WB_ORACLE_H
#define WB_ORACLE_H
#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* Record tags — rocksdb 5.7.fb db/dbformat.h ValueType (WAL-legal subset). */
enum {
  ORC_TYPE_DELETION = 0x00,
  ORC_TYPE_VALUE = 0x01,
  ORC_TYPE_MERGE = 0x02,
  ORC_TYPE_LOGDATA = 0x03,
  ORC_TYPE_CF_DELETION = 0x04,
  ORC_TYPE_CF_VALUE = 0x05,
  ORC_TYPE_CF_MERGE = 0x06,
  ORC_TYPE_SINGLE_DELETION = 0x07,
  ORC_TYPE_CF_SINGLE_DELETION = 0x08,
  ORC_TYPE_BEGIN_PREPARE = 0x09,
  ORC_TYPE_END_PREPARE = 0x0A,
  ORC_TYPE_COMMIT = 0x0B,
  ORC_TYPE_ROLLBACK = 0x0C,
  ORC_TYPE_NOOP = 0x0D,
  ORC_TYPE_CF_RANGE_DELETION = 0x0E,
  ORC_TYPE_RANGE_DELETION = 0x0F,
};

/* ---------- WriteBatch rep builder (leader-side encode) ---------- */
typedef struct OrcBatch OrcBatch;
OrcBatch *orc_wb_create(void);
void orc_wb_destroy(OrcBatch *b);
void orc_wb_clear(OrcBatch *b);
void orc_wb_put(OrcBatch *b, const void *key, size_t klen, const void *val, size_t vlen);
void orc_wb_delete(OrcBatch *b, const void *key, size_t klen);
void orc_wb_single_delete(OrcBatch *b, const void *key, size_t klen);
void orc_wb_merge(OrcBatch *b, const void *key, size_t klen, const void *val, size_t vlen);
void orc_wb_delete_range(OrcBatch *b, const void *bk, size_t bklen, const void *ek, size_t eklen);
void orc_wb_put_log_data(OrcBatch *b, const void *blob, size_t blen);
void orc_wb_set_seq(OrcBatch *b, uint64_t seq);
uint32_t orc_wb_count(const OrcBatch *b);
/* Pointer to rep bytes (valid until next mutation) + length. */
const uint8_t *orc_wb_data(const OrcBatch *b, size_t *len);

/* ---------- Record-level decode (parity target for the GPU decode kernel) ---------- */
typedef struct {
  uint8_t type;       /* ORC_TYPE_* */
  uint8_t consumes_seq; /* 1 for Put/Delete/SingleDelete/Merge/RangeDeletion (+CF) */
  uint32_t cf_id;     /* 0 for default column family */
  uint64_t seq;       /* header seq + index among seq-consuming records */
  uint32_t key_off, key_len; /* offsets into the rep blob */
  uint32_t val_off, val_len; /* value / range-end / logdata blob slice */
} OrcRecord;

/* Decode a rep blob. Returns 0 on success, nonzero on corruption (truncated
 * record, bad varint, count mismatch — write_batch.cc iterate contract).
 * On success *nrec = number of records (incl. LogData), *base_seq = header seq,
 * *count = header count. out may be NULL to validate/count only. */
int orc_decode(const uint8_t *rep, size_t len, OrcRecord *out, uint32_t cap,
               uint32_t *nrec, uint64_t *base_seq, uint32_t *count);

/* ---------- Memtable store (stands in for rocksdb::DB on the follower) ---------- */
typedef struct OrcStore OrcStore;
enum { ORC_MERGE_CONCAT = 0, ORC_MERGE_U64ADD = 1 };
OrcStore *orc_store_create(uint32_t nshards, int merge_op);
void orc_store_destroy(OrcStore *s);

/* Follower apply — semantics of RocksDbWrapper::HandleReplicateResponse
 * (rocksdb_wrapper.cpp:13-28): validate rep, (conceptually) append
 * LogData(timestamp), apply to memtable, advance seq by header count.
 * Returns 1 on success, 0 on failure (corruption); on failure nothing applied. */
int orc_apply(OrcStore *s, uint32_t shard, const uint8_t *rep, size_t len, int64_t ts);

uint64_t orc_latest_seq(const OrcStore *s, uint32_t shard);

/* Get: 0 = found (value in buf, *vlen set), 1 = not found, 2 = buffer too small,
 * 3 = bad shard. Merge folding: ORC_MERGE_CONCAT concatenates base+operands in
 * seq order separated by ','; ORC_MERGE_U64ADD sums little-endian u64 (missing
 * base = 0), result 8 bytes — the counter_service-style operator
 * (examples/counter_service). */
int orc_get(const OrcStore *s, uint32_t shard, const void *key, size_t klen,
            void *buf, size_t cap, size_t *vlen);

/* Number of live point entries across history (diagnostics). */
uint64_t orc_store_bytes(const OrcStore *s);

/* ---------- CPU baseline driver (bench.py cpu_baseline leg) ----------
 * Applies n pre-encoded updates (descs: shard, off, len into arena) on
 * nthreads host threads, one shard owned by exactly one thread (shard %
 * nthreads), mirroring the reference's per-shard sequential / cross-shard
 * concurrent executor model (rocksdb_replicator.cpp:41-67). Returns seconds. */
typedef struct {
  uint32_t shard;
  uint32_t len;
  uint64_t off;
  int64_t ts;
} OrcUpdateDesc;
double orc_cpu_apply_bench(OrcStore *s, const uint8_t *arena,
                           const OrcUpdateDesc *descs, uint64_t n, int nthreads);
/* WAL-on variant (rep + LogData trailer appended to a rotated log buffer
 * before the memtable insert — SURVEY §8d asks for both legs). */
double orc_cpu_apply_bench_wal(OrcStore *s, const uint8_t *arena,
                               const OrcUpdateDesc *descs, uint64_t n,
                               int nthreads);


/* ---------- Snappy block format (transport compression, config #5) ----------
 * Independent CPU restatement of the public Snappy format (raw block
 * format: varint32 uncompressed length, then literal/copy elements).
 * Checker-side codec for the GPU decompress stage. */
size_t orc_snappy_max_len(size_t n);
/* returns compressed size, 0 on overflow/cap */
size_t orc_snappy_compress(const uint8_t *src, size_t slen, uint8_t *dst, size_t dcap);
/* returns 0 ok (sets *dlen), nonzero on corruption */
int orc_snappy_decompress(const uint8_t *src, size_t slen, uint8_t *dst,
                          size_t dcap, size_t *dlen);
/* CPU baseline variant: decompress each update then apply (config #5). */
double orc_cpu_snappy_apply_bench(OrcStore *s, const uint8_t *arena,
                                  const OrcUpdateDesc *descs, uint64_t n,
                                  int nthreads);

/* Full-store content checksum for any-size parity ("checksum of
 * checksums"): per record FNV-1a over (seq LE8 | type | key_len LE4 |
 * val_len LE4 | key bytes | val bytes), combined per shard by unsigned
 * 64-bit ADDITION (order-independent, so run/representation layout does
 * not matter). Range tombstones hash with key=begin, val=end. Must match
 * the engine's gra_shard_checksum on identically applied streams. */
uint64_t orc_shard_checksum(const OrcStore *s, uint32_t shard);

#ifdef __cplusplus
}
#endif
#endif /* WB_ORACLE_H */
