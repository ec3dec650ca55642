/* host_store.h — host-side run registry standing behind the DbWrapper seam.
 *
 * MI355X-first design: the follower "memtable" is a sequence of GPU-produced
 * sorted-by-seq runs resident in device HBM (288 GB/GPU); the host keeps only
 * run descriptors and lazily fetches run bytes for parity probes (gra_get).
 * Leader-side writes (WriteToLeader, rocksdb_wrapper.cpp:5-8) produce
 * host-built runs in the identical format.
 *
 * Get semantics must equal the oracle's (tests/test_oracle.py): walk runs
 * newest→oldest, entries newest→oldest; Put = base, Delete/SingleDelete =
 * tombstone, Merge = operand collected then folded oldest→newest with the
 * engine merge operator, RangeDeletion = tombstone for covered keys below
 * its seq.
 */
#pragma once
#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <cstring>
#include <deque>
#include <memory>
#include <mutex>
#include <string>
#include <vector>

#include "wb_format.h"

namespace gra {

struct Run {
  uint64_t base_seq = 0, last_seq = 0;
  uint32_t n_entries = 0;
  uint32_t payload_bytes = 0;
  /* device-resident location (absolute offsets into the store arena);
   * UINT64_MAX for host-origin (leader-write) runs */
  uint64_t hdr_cur = UINT64_MAX, payload_cur = UINT64_MAX;
  uint32_t pay_rel_base = 0; /* device runs: kv_off rebase at fetch time */
  /* host copies — always set for host runs, lazily fetched for device runs */
  std::vector<uint8_t> hdrs;     /* n_entries × wb::RecHdr */
  std::vector<uint8_t> payload;
  /* drain-host mode: runs are SPANS of a tick-shared pinned arena drained
   * by k_drain (no per-run host copy). kv_off stays tick-relative; pay_p
   * points at the tick's payload base so pay_p + kv_off is correct. */
  std::shared_ptr<uint8_t> hbuf;
  const uint8_t *hdr_p = nullptr, *pay_p = nullptr;
  bool kpref_built = false; /* device read index (k_kpref) done; guarded by
                               the engine mutex at build time */
  const uint8_t *hdrs_data() const { return hdr_p ? hdr_p : hdrs.data(); }
  const uint8_t *payload_data() const { return pay_p ? pay_p : payload.data(); }
  bool resident() const {
    return hdr_p != nullptr || !hdrs.empty() || n_entries == 0;
  }
};

/* One retained batch for downstream serving (leader update log — the WAL
 * retention analog; replicated_db.cpp:435-575). */
struct LogEnt {
  uint64_t base_seq = 0;
  uint32_t count = 0;
  int64_t ts = 0;
  std::vector<uint8_t> rep;
};

struct ShardState {
  mutable std::mutex mu;
  uint64_t durable_seq = 0;   /* last seq applied & synced on device */
  uint64_t next_seq = 1;      /* next seq to assign at submission */
  bool poisoned = false;      /* corrupt batch seen; next HRR returns false */
  std::vector<std::shared_ptr<Run>> runs; /* oldest .. newest */
  std::deque<LogEnt> log;     /* retained batches (retain_log mode) */
  uint64_t log_used = 0;      /* bytes retained in this shard's log */
  /* follower-ACK box ≅ MaxNumberBox (max_number_box.h:38-83): serving a
   * pull carries the follower's progress — the request's seq_no is the
   * confirmed ack (mode 2, replicated_db.cpp:452-456), and everything sent
   * is the sent ack (mode 1, :543-546). */
  uint64_t acked_sent = 0, acked_confirmed = 0;
  std::condition_variable ack_cv;
  /* per-db counters ≅ replicator_stats.cpp:33-102's per-db fan-out
   * (replicator_in_bytes / _out_bytes / latency / failure counters);
   * relaxed atomics so the hot ingest path never takes a lock for them */
  std::atomic<uint64_t> cnt_updates{0}, cnt_in_bytes{0}, cnt_failures{0};
  std::atomic<uint64_t> cnt_served{0}, cnt_out_bytes{0};
  std::atomic<uint64_t> lat_sum_ms{0}, lat_n{0};
};

/* Get over a run list (newest last). merge_op: 0 concat, 1 u64add.
 * Returns 0 found, 1 not found. Found value appended to out. */
int run_get(const std::vector<std::shared_ptr<Run>> &runs, const void *key,
            size_t klen, int merge_op, std::string *out);

/* Raw probe over a run list: max-seq terminator (Put/Delete/SingleDelete),
 * max merge-operand seq and max covering range-tombstone seq for `key` —
 * the same per-query verdict k_multiget computes, so a mixed shard can
 * merge the device and host halves by seq. */
struct ProbeResult {
  uint64_t term_seq = 0, merge_seq = 0, rd_seq = 0;
  uint8_t term_type = 0xFF;          /* wb tag of the terminator */
  const uint8_t *val = nullptr;      /* terminator value (kValue only) */
  uint32_t vlen = 0;
};
void run_probe(const std::vector<std::shared_ptr<Run>> &runs, const void *key,
               size_t klen, ProbeResult *out);

/* Host apply of one rep blob (leader write path): decodes with wb::walk and
 * builds a Run in the same format the GPU emits. Returns false on corrupt
 * rep. base_seq = first seq to assign. */
bool host_build_run(const uint8_t *rep, size_t len, uint64_t base_seq, Run *out);

} /* namespace gra */
