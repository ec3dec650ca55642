/* rocksplicator_gpu.h — C-ABI boundary of the MI355X-native replication
 * apply path.
 *
 * This is the drop-in seam for pinterest/rocksplicator's follower-side
 * WriteBatch ingest. Each entry point states the reference interface it
 * replaces (file:line in /root/reference). A C++ adapter implementing
 * replicator::DbWrapper (rocksdb_replicator/db_wrapper.h:6-15) on top of
 * these calls is shown in INTEGRATION.md, so that
 * RocksDBReplicator::addDB/write (rocksdb_replicator/rocksdb_replicator.h:
 * 175-216) stay source-compatible.
 *
 * Threading contract (mirrors the reference executor model,
 * rocksdb_replicator.cpp:41-67): calls on ONE GraDb are sequential per
 * shard; calls across different GraDb handles may be concurrent.
 * gra_handle_replicate_response enqueues and completes asynchronously;
 * gra_flush is the drain barrier after which gra_latest_seq/gra_get are
 * linearizable per shard.
 *
 * The follower apply path is GPU-only BY DESIGN: gra_engine_create fails
 * loudly when no MI355X (HIP device) is present — there is no CPU fallback
 * for gra_handle_replicate_response.
 */
#ifndef ROCKSPLICATOR_GPU_H
#define ROCKSPLICATOR_GPU_H
#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef struct GraEngine GraEngine;   /* one per GPU; owns streams + device store */
typedef struct GraDb GraDb;           /* per-shard handle ≅ one replicator::DbWrapper */
typedef struct GraReplay GraReplay;   /* pre-uploaded replay stream (harness) */

enum { GRA_MERGE_CONCAT = 0, GRA_MERGE_U64ADD = 1 };

/* Status codes (gra_* return int unless noted): */
enum {
  GRA_OK = 0,
  GRA_NOT_FOUND = 1,      /* gra_get: key absent/deleted */
  GRA_BUF_TOO_SMALL = 2,  /* gra_get: value larger than cap */
  GRA_ERR = -1,           /* see gra_last_error() */
  GRA_CORRUPT = -2,       /* malformed WriteBatch rep */
  GRA_NO_GPU = -3,        /* no HIP device — the apply path refuses to run */
  GRA_FULL = -4,          /* staging or store arena exhausted */
};

typedef struct GraEngineOpts {
  uint32_t nshards;
  int device;              /* HIP device ordinal; -1 = current device */
  int merge_op;            /* GRA_MERGE_* — Get-time merge-operator fold */
  int store_ring;          /* 1: device run arena recycles oldest ticks (bench);
                              0: GRA_FULL when exhausted */
  uint64_t store_bytes;    /* device run-arena capacity (0 = 4 GiB default) */
  uint64_t staging_bytes;  /* pinned staging per buffer (0 = 256 MiB default) */
  uint32_t max_wb_records; /* per-batch record cap (0 = 1024 default) */
  int retain_log;          /* 1: retain applied/written rep blobs per shard so
                              this node can serve downstream pulls (the WAL-
                              retention analog; replicated_db.cpp:435-575) */
  uint64_t log_bytes;      /* retained-log cap across shards (0 = 256 MiB);
                              oldest batches evicted first (WAL_ttl analog,
                              performance.cpp:99) */
  int drain_host;          /* 1: eagerly drain each run's bytes to host
                              memtable buffers at flush (the literal
                              "drain to host memtables" mode; default 0 =
                              device-resident store with lazy fetch) */
} GraEngineOpts;

void gra_engine_opts_init(GraEngineOpts *opts); /* fill defaults */

/* Create/destroy the per-GPU engine. Returns GRA_OK or GRA_NO_GPU/GRA_ERR. */
int gra_engine_create(const GraEngineOpts *opts, GraEngine **out);
void gra_engine_destroy(GraEngine *e);

/* Thread-local message for the last GRA_ERR/GRA_CORRUPT/GRA_NO_GPU. */
const char *gra_last_error(void);

/* ---------------- per-shard handles (the DbWrapper seam) ---------------- */

/* ≅ constructing a RocksDbWrapper for one shard and registering it via
 * RocksDBReplicator::addDB (rocksdb_replicator.cpp:96-133). */
GraDb *gra_open(GraEngine *e, uint32_t shard_id);
void gra_close(GraDb *db);

/* ≅ DbWrapper::HandleReplicateResponse (db_wrapper.h:13; reference impl
 * rocksdb_wrapper.cpp:13-28): apply one Update's raw_data (WriteBatch rep
 * bytes) + timestamp to this shard. Returns 1 (true) when accepted, 0 on
 * failure — same bool contract the pull loop checks (replicated_db.cpp:378).
 * Acceptance is an enqueue; a later corruption detected on-GPU poisons the
 * shard: the NEXT call returns 0 and the shard's seq rolls back to the last
 * durable batch boundary, mirroring the reference's delayed re-pull
 * (replicated_db.cpp:378-382,412-431). */
int gra_handle_replicate_response(GraDb *db, const uint8_t *rep, size_t len,
                                  int64_t timestamp_ms);

/* ≅ DbWrapper::LatestSequenceNumber (db_wrapper.h:12; rocksdb_wrapper.cpp:4).
 * Returns the seq the pull loop should resume from: submitted seq while the
 * shard is healthy, last durable seq after a failure. */
uint64_t gra_latest_seq(GraDb *db);

/* ≅ DbWrapper::WriteToLeader (db_wrapper.h:8; rocksdb_wrapper.cpp:5-8):
 * leader-side local write of a WriteBatch rep. Host path (WAL+memtable is
 * host work in the reference; GPU leader batching is a later row — SURVEY
 * §8f f1). Fills *seq_out with the batch's last seq. */
int gra_write_leader(GraDb *db, const uint8_t *rep, size_t len, uint64_t *seq_out);

/* Parity probe ≅ rocksdb::DB::Get with the engine's merge operator folded.
 * GRA_OK (value in buf, *vlen set) / GRA_NOT_FOUND / GRA_BUF_TOO_SMALL. */
int gra_get(GraDb *db, const void *key, size_t klen, void *buf, size_t cap,
            size_t *vlen);

/* Drain barrier: every update submitted so far is applied and durable; after
 * this, gra_latest_seq and gra_get are linearizable per shard. */
int gra_flush(GraEngine *e);

/* drain_host mode setup: pre-pin `count` host arenas of `bytes` each so no
 * hipHostMalloc happens inside a timed region. */
int gra_drain_prewarm(GraEngine *e, size_t bytes, uint32_t count);

/* Batched point reads served FROM THE DEVICE STORE (followers serve reads
 * in rocksplicator deployments — ApplicationDB::Get routes to the local db;
 * here the memtable lives in HBM, so the search runs there too: one block
 * per query scanning the shard's runs newest->oldest). Each result value is
 * written at valbuf[q*val_stride] (vlen capped at val_stride). A query
 * whose outcome needs merge-operand folding reports GRA_GET_NEEDS_HOST and
 * is answered by the host path (gra_get) instead. Call gra_flush first for
 * linearizable results. */
enum { GRA_GET_FOUND = 0, GRA_GET_MISS = 1, GRA_GET_NEEDS_HOST = 2 };
typedef struct {
  uint32_t off, len; /* key slice in keybuf */
} GraKeyRef;
typedef struct {
  uint32_t status; /* GRA_GET_* */
  uint32_t vlen;
} GraGetResult;
int gra_multiget(GraDb *db, uint32_t nq, const GraKeyRef *keys,
                 const uint8_t *keybuf, size_t keybuf_len, uint8_t *valbuf,
                 uint32_t val_stride, GraGetResult *out);

/* Full-store content checksum for any-size parity ("checksum of
 * checksums"): per record FNV-1a over (seq LE8 | type | key_len LE4 |
 * val_len LE4 | key bytes | val bytes), combined per shard by u64
 * ADDITION — order/representation independent, so it can be compared
 * against the oracle's orc_shard_checksum on identically applied streams
 * at ANY size. Computed on-device over the run store (host-origin runs
 * folded on the host). Call gra_flush first. */
int gra_shard_checksum(GraDb *db, uint64_t *out);

/* ---------------- leader update-serving ----------------
 * ≅ ReplicatedDB::handleReplicateRequest batch read-out (replicated_db.cpp:
 * 435-575): serve retained batches with base seq > since_seq, up to
 * max_updates (the reference default is 50, replicated_db.cpp:42-43), as
 * (seq, timestamp, rep bytes) triples — the Update wire triple
 * (replicator.thrift:44-57). Blobs are copied into buf; out[i].off/len
 * index it. Requires opts.retain_log. Returns GRA_OK with *n_out = 0 when
 * caught up; GRA_ERR when since_seq predates the retained log (the
 * reference's WAL-gone case). */
typedef struct {
  uint64_t seq;   /* batch base seq (first record's seq) */
  int64_t ts;     /* timestamp the leader stamped (ms) */
  uint32_t off, len;
} GraServedUpdate;
/* requester_role: 0 = FOLLOWER (the request's seq_no is posted as the
 * confirmed ack), 1 = OBSERVER (no ack — replicated_db.cpp:452-456 ignores
 * observer sequence numbers). */
int gra_get_updates(GraDb *db, uint64_t since_seq, uint32_t max_updates,
                    GraServedUpdate *out, uint32_t *n_out, uint8_t *buf,
                    size_t cap, int requester_role);

/* Per-db counters ≅ the replicator stats hooks' per-db fan-out
 * (rocksdb_replicator/replicator_stats.cpp:33-102: replicator_in_bytes,
 * replicator_out_bytes, replicator_latency_ms, handle-response failures). */
typedef struct {
  uint64_t updates_applied, in_bytes, apply_failures;
  uint64_t updates_served, out_bytes;
  uint64_t latency_ms_sum, latency_samples;
  uint64_t latest_seq;
} GraDbCounters;
int gra_db_counters(GraDb *db, GraDbCounters *out);

/* ≅ MaxNumberBox::wait (max_number_box.h:38-83, .cpp:63) behind the 2-ACK
 * write modes (replicated_db.cpp:147-156): block until the downstream ack
 * reaches seq. confirmed=1 waits for follower-applied progress (mode 2 —
 * the pull request's seq_no, replicated_db.cpp:452-456); confirmed=0 waits
 * for served progress (mode 1, :543-546). GRA_OK, or GRA_NOT_FOUND on
 * timeout. */
int gra_wait_ack(GraDb *db, uint64_t seq, int confirmed, int timeout_ms);

/* ---------------- WriteBatch builder ----------------
 * The reference's callers construct updates with rocksdb::WriteBatch
 * (e.g. examples/counter_service/counter_handler.cpp:152-158); this is the
 * framework-native equivalent producing the identical rep byte layout. */
typedef struct GraBatch GraBatch;
GraBatch *gra_wb_create(void);
void gra_wb_destroy(GraBatch *b);
void gra_wb_clear(GraBatch *b);
void gra_wb_put(GraBatch *b, const void *k, size_t klen, const void *v, size_t vlen);
void gra_wb_delete(GraBatch *b, const void *k, size_t klen);
void gra_wb_single_delete(GraBatch *b, const void *k, size_t klen);
void gra_wb_merge(GraBatch *b, const void *k, size_t klen, const void *v, size_t vlen);
void gra_wb_delete_range(GraBatch *b, const void *bk, size_t bklen,
                         const void *ek, size_t eklen);
/* CF-prefixed record variants (varint cf id before the slices; cf 0 uses
 * the plain forms, as rocksdb does). */
void gra_wb_cf_put(GraBatch *b, uint32_t cf, const void *k, size_t klen,
                   const void *v, size_t vlen);
void gra_wb_cf_delete(GraBatch *b, uint32_t cf, const void *k, size_t klen);
void gra_wb_cf_single_delete(GraBatch *b, uint32_t cf, const void *k,
                             size_t klen);
void gra_wb_cf_merge(GraBatch *b, uint32_t cf, const void *k, size_t klen,
                     const void *v, size_t vlen);
void gra_wb_cf_delete_range(GraBatch *b, uint32_t cf, const void *bk,
                            size_t bklen, const void *ek, size_t eklen);
void gra_wb_put_log_data(GraBatch *b, const void *blob, size_t blen);
void gra_wb_set_seq(GraBatch *b, uint64_t seq);
uint32_t gra_wb_count(const GraBatch *b);
const uint8_t *gra_wb_data(const GraBatch *b, size_t *len);

/* ---------------- replay / measurement harness surface ----------------
 * Re-imagines rocksdb_replicator/performance.cpp:127-167 as a replay
 * harness: pre-encoded update streams are uploaded to HBM once (untimed),
 * then gra_replay_tick runs the apply pipeline with inputs already resident
 * in device memory. */
typedef struct {
  uint32_t shard;
  uint32_t len;    /* rep blob length */
  uint64_t off;    /* offset into the arena */
  int64_t ts;      /* Update.timestamp (ms) */
} GraUpdateDesc;

/* Pinned host allocation (for zero-copy staging by harnesses). */
int gra_pin_alloc(GraEngine *e, size_t bytes, uint8_t **ptr);
void gra_pin_free(GraEngine *e, uint8_t *ptr);

/* Upload a replay stream: arena blobs + descs, H2D once. Descs must be
 * grouped by shard within any tick window that will be replayed. Seqs are
 * assigned per shard at upload (header count field), continuing from the
 * shard's current seq. */
int gra_upload(GraEngine *e, const uint8_t *arena, size_t arena_bytes,
               const GraUpdateDesc *descs, uint64_t ndescs, GraReplay **out);
/* Same, but the blobs already live in device memory (e.g. the output tensor
 * of an RCCL all-to-all shard repartition — BASELINE config #4): zero-copy,
 * caller owns the arena (>=16 B readable slack required) and supplies each
 * batch's header record count (host can't read device headers). */
int gra_upload_dev(GraEngine *e, void *dev_arena, size_t arena_bytes,
                   const GraUpdateDesc *descs, uint64_t ndescs,
                   const uint32_t *counts, GraReplay **out);
/* Config #5: Snappy-compressed Update payloads. descs index the COMPRESSED
 * arena; ulens/counts carry per-update uncompressed size and batch record
 * count. A GPU pre-stage (k_snappy, lane-per-update) decompresses each tick
 * into a scratch arena feeding the normal decode pipeline. */
int gra_upload_snappy(GraEngine *e, const uint8_t *comp_arena,
                      size_t comp_bytes, const GraUpdateDesc *descs,
                      uint64_t ndescs, const uint32_t *ulens,
                      const uint32_t *counts, GraReplay **out);
/* Host-side Snappy codec (transport/leader side + harness). Returns
 * compressed length (0 = dst too small) / uncompressed length (UINT32_MAX =
 * corrupt). */
uint32_t gra_snappy_compress(const uint8_t *src, uint32_t slen, uint8_t *dst,
                             uint32_t dcap);
uint32_t gra_snappy_decompress(const uint8_t *src, uint32_t slen, uint8_t *dst,
                               uint32_t dcap);
void gra_replay_destroy(GraReplay *r);

/* Run one pipeline tick over descs [first, first+n): decode + emit +
 * partition-copy into the device run store. Asynchronous; gra_replay_sync
 * or gra_flush to drain. Updates applied this way are visible to gra_get /
 * gra_latest_seq like streamed ones. */
int gra_replay_tick(GraReplay *r, uint64_t first, uint64_t n);
int gra_replay_sync(GraReplay *r);
/* Pre-build + device-cache the plan for a tick window (group table, sorted
 * snappy task order) so the first tick of the window pays no allocation in
 * a timed region. Optional setup call. */
int gra_replay_prepare(GraReplay *r, uint64_t first, uint64_t n);

/* Streaming-ingest equivalent for PCIe-inclusive measurement: same tick, but
 * blobs start in the (pinned) host arena and are staged H2D inside the tick. */
int gra_replay_tick_h2d(GraReplay *r, uint64_t first, uint64_t n);

/* ---------------- stats / timing (HIP-event-measured, per kernel) -------- */
typedef struct {
  double h2d_ms, snappy_ms, decode_ms, scan_ms, emit_ms, copy_ms, runfix_ms,
      total_ms;
  uint64_t ticks, updates, records, blob_bytes, payload_bytes;
} GraStats;
void gra_stats(GraEngine *e, GraStats *out);
void gra_stats_reset(GraEngine *e);

/* ---------------- synthetic stream generator (harness) ----------------
 * Deterministic xoshiro256** streams, seed 0xR0CK5-style base + shard id
 * (SURVEY §8d). kind: 0 = uniform Put (config #2 shape), 1 = Zipf-0.99 keys
 * (config #3 shape), 2 = mixed Put/Delete/Merge 70/20/10 (config #5 shape).
 * Writes encoded update blobs into arena (cap bytes) and descs (cap n).
 * Returns GRA_OK and the consumed sizes, or GRA_FULL. Each update is one
 * record per batch, ≤50 updates per response window, mirroring
 * performance.cpp:139-142 / replicated_db.cpp:42-43. */
typedef struct {
  uint32_t nshards;
  uint32_t key_len;        /* e.g. 16 */
  uint32_t val_len;        /* e.g. 128 or 1024 */
  uint32_t kind;           /* 0 uniform put, 1 zipf, 2 mixed */
  uint64_t key_space;      /* e.g. 1<<24 */
  double zipf_s;           /* e.g. 0.99 */
  uint64_t seed;
  uint32_t compressible;   /* 1: low-entropy values (Snappy-compressible) */
  uint32_t _pad;
} GraGenOpts;
int gra_gen_stream(const GraGenOpts *g, uint64_t n_updates, uint8_t *arena,
                   size_t arena_cap, size_t *arena_used, GraUpdateDesc *descs,
                   int64_t ts);

#ifdef __cplusplus
}
#endif
#endif /* ROCKSPLICATOR_GPU_H */
