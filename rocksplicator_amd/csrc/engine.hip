/* engine.hip — MI355X-native follower apply engine.
 *
 * Replaces the body of RocksDbWrapper::HandleReplicateResponse
 * (rocksdb_wrapper.cpp:13-28) + the per-update apply loop
 * (replicated_db.cpp:369-383): update blobs for many shards are batched into
 * ticks; each tick runs a GPU pipeline over blobs resident in HBM:
 *
 *   K1 decode   — per-update sequential walk of the WriteBatch rep (varint
 *                 record-boundary discovery, validation, totals)
 *   K2 scan     — exclusive prefix sums over (records, payload bytes)
 *   K3 reserve  — bump-reserve tick space in the device run store (stream-
 *                 ordered single-thread kernel; ring or linear)
 *   K4 emit     — second walk emits 24-B record headers (seq/type/offsets)
 *                 and per-slice copy tasks
 *   K5 copy     — partition-copy key/value bytes into per-shard contiguous
 *                 run segments (16-lane groups, dword funnel for unaligned
 *                 sources)
 *   K6 rundesc  — per-shard run descriptors, D2H to the host run registry
 *
 * The follower "memtable" is the device run store (288 GB HBM3E); the host
 * keeps descriptors only. Per-shard seq order is preserved because updates
 * are shard-grouped per tick and record headers are emitted in stream order
 * (seq accounting per rocksdb_assumption_test.cpp:136-187).
 *
 * No CPU fallback exists for this path: engine creation fails loudly
 * without a HIP device.
 */
#include <hip/hip_runtime.h>

#define WB_UNALIGNED_OK 1 /* gfx950: misaligned global loads OK (micro_copy v1) */

#include <algorithm>
#include <atomic>
#include <chrono>
#include <cstdio>
#include <cstring>
#include <deque>
#include <map>
#include <memory>
#include <mutex>
#include <string>
#include <thread>
#include <vector>

#include "../../include/rocksplicator_gpu.h"
#include "host_store.h"
#include "snappy.h"
#include "wb_format.h"

/* ---------------- error plumbing ---------------- */
static thread_local std::string g_err;
extern "C" const char *gra_last_error(void) { return g_err.c_str(); }

#define HIP_TRY(x)                                                     \
  do {                                                                 \
    hipError_t _e = (x);                                               \
    if (_e != hipSuccess) {                                            \
      g_err = std::string(#x) + ": " + hipGetErrorString(_e);          \
      return GRA_ERR;                                                  \
    }                                                                  \
  } while (0)

namespace gra {

/* ---------------- device-side structs ---------------- */
struct UpdDesc {       /* one Update in device memory */
  uint64_t off;        /* blob offset in the blob arena */
  uint64_t base_seq;   /* follower-assigned base seq (host bookkeeping) */
  uint32_t len;
  uint32_t shard;
};
struct TickPlace {
  uint64_t cur;          /* monotonic cursor at reservation (16-aligned) */
  uint64_t hdr_off;      /* absolute offset of RecHdr region in store arena */
  uint64_t payload_off;  /* absolute offset of payload region */
  uint64_t payload_bytes;
  uint32_t total_rec;
  uint32_t overflow;
};
struct CopyTask {
  uint64_t src_off; /* into blob arena */
  uint32_t dst_rel; /* into tick payload region */
  uint32_t nbytes;
};
struct SnapTask { /* one compressed Update payload (config #5) */
  uint64_t comp_off;
  uint64_t out_off; /* scratch-arena slot (rides with the task so the launch
                       order can be length-sorted independently of descs) */
  uint32_t comp_len;
  uint32_t ulen;
};
struct GroupDesc {
  uint32_t shard, first, n_upds, _pad;
};
struct DevRunDesc {
  uint64_t base_seq, last_seq;
  uint64_t cur;         /* tick cursor (for ring pruning) */
  uint64_t hdr_off, payload_off;
  uint32_t n_entries, payload_bytes;
  uint32_t pay_rel_base; /* scan[first].y — kv_off values are tick-relative */
  uint32_t shard;
};

__host__ __device__ inline uint2 add2(uint2 a, uint2 b) {
  return make_uint2(a.x + b.x, a.y + b.y);
}

/* ---------------- kernels ---------------- */

/* K0 (config #5): per-update Snappy decompress, compressed arena ->
 * uncompressed scratch (= the tick's blob arena). Lane-per-update with the
 * compressed stream PRE-STAGED into LDS by independent vector loads —
 * the parse's dependent byte loads then hit LDS instead of L1/L2
 * (+76% over a straight-global parse, scripts/micro_snappy.hip; the
 * 16-lane cooperative variant ties global). +4 B row pad breaks the
 * all-threads-one-bank stride. Streams larger than the stage fall back to
 * the global parse. A failed stream poisons its slot header so the decode
 * walk rejects it. */
constexpr uint32_t kSnapStage = 508; /* 256 x (508+4) = 128 KiB LDS */

__global__ void __launch_bounds__(256) k_snappy(
    const uint8_t *__restrict__ comp, const SnapTask *__restrict__ tasks,
    uint32_t n, uint8_t *__restrict__ scratch,
    uint32_t *__restrict__ err_ring, uint32_t tick) {
  __shared__ uint8_t lds[256 * (kSnapStage + 4)];
  uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  SnapTask t = tasks[i];
  uint8_t *dst = scratch + t.out_off;
  uint32_t r;
  if (t.comp_len <= kSnapStage) {
    uint8_t *mine = lds + threadIdx.x * (kSnapStage + 4);
    const uint8_t *src = comp + t.comp_off;
    for (uint32_t b = 0; b < t.comp_len; b += 16)
      *(uint4 *)(mine + b) = *(const uint4 *)(src + b);
    r = snp::decompress(mine, t.comp_len, dst, t.ulen);
  } else {
    r = snp::decompress(comp + t.comp_off, t.comp_len, dst, t.ulen);
  }
  if (r != t.ulen) {
    for (int b = 0; b < 13; b++) dst[b] = 0xFF; /* force decode rejection */
    atomicAdd(&err_ring[tick % 64u], 1u);
  }
}

/* K1: decode walk fused with the first scan pass — each block decodes 256
 * updates, then block-scans (records, payload16) into exclusive partials +
 * a block sum. Errors accumulate in a 64-slot tick ring (no per-tick
 * memset; the fused scan2 kernel re-zeroes slot tick+32 ahead). */
constexpr uint32_t kErrRing = 64;

constexpr uint32_t kRecCache = 2; /* records cached per update by decode so
                                     emit can skip the second blob walk */

__global__ void k_decode(const uint8_t *__restrict__ blobs,
                         const UpdDesc *__restrict__ descs, uint32_t n,
                         wb::WalkTotals *__restrict__ totals, uint32_t max_rec,
                         uint32_t *__restrict__ err_ring, uint32_t tick,
                         uint2 *__restrict__ partial, uint2 *__restrict__ bsums,
                         wb::Rec *__restrict__ reccache,
                         uint8_t *__restrict__ ok_out) {
  __shared__ uint2 sh[256];
  uint32_t i = blockIdx.x * 256 + threadIdx.x;
  uint2 v = make_uint2(0, 0);
  if (i < n) {
    UpdDesc d = descs[i];
    wb::Rec *cache = reccache + (size_t)i * kRecCache;
    wb::WalkTotals t = wb::walk_f(blobs + d.off, d.len,
                                  [&](const wb::Rec &r, uint32_t idx) {
                                    if (idx < kRecCache) cache[idx] = r;
                                  });
    if (t.n_records > max_rec) t.ok = 0;
    totals[i] = t;
    ok_out[i] = (uint8_t)t.ok;
    if (!t.ok) atomicAdd(&err_ring[tick % kErrRing], 1u);
    if (t.ok) v = make_uint2(t.n_records, t.payload16);
  }
  sh[threadIdx.x] = v;
  __syncthreads();
  for (int ofs = 1; ofs < 256; ofs <<= 1) {
    uint2 a = sh[threadIdx.x];
    uint2 b = threadIdx.x >= (uint32_t)ofs ? sh[threadIdx.x - ofs] : make_uint2(0, 0);
    __syncthreads();
    sh[threadIdx.x] = add2(a, b);
    __syncthreads();
  }
  if (i < n) {
    uint2 inc = sh[threadIdx.x];
    partial[i] = make_uint2(inc.x - v.x, inc.y - v.y); /* exclusive-in-block */
  }
  if (threadIdx.x == 255) bsums[blockIdx.x] = sh[255];
}

__global__ void k_scan2(uint2 *__restrict__ bsums, uint32_t nblocks,
                        uint64_t *__restrict__ cursor,
                        TickPlace *__restrict__ place, uint64_t cap, int ring,
                        uint32_t task_cap, uint32_t *__restrict__ err_ring,
                        uint32_t tick) {
  /* single block of 256; sequential chunks with carry; exclusive in place;
   * bsums[nblocks] = grand total */
  __shared__ uint2 sh[256];
  __shared__ uint2 carry;
  if (threadIdx.x == 0) carry = make_uint2(0, 0);
  __syncthreads();
  for (uint32_t base = 0; base < nblocks; base += 256) {
    uint32_t i = base + threadIdx.x;
    uint2 v = i < nblocks ? bsums[i] : make_uint2(0, 0);
    sh[threadIdx.x] = v;
    __syncthreads();
    for (int ofs = 1; ofs < 256; ofs <<= 1) {
      uint2 a = sh[threadIdx.x];
      uint2 b = threadIdx.x >= (uint32_t)ofs ? sh[threadIdx.x - ofs] : make_uint2(0, 0);
      __syncthreads();
      sh[threadIdx.x] = add2(a, b);
      __syncthreads();
    }
    uint2 inc = add2(sh[threadIdx.x], carry);
    if (i < nblocks) bsums[i] = make_uint2(inc.x - v.x, inc.y - v.y);
    __syncthreads();
    if (threadIdx.x == 255) carry = inc; /* last of chunk = running total */
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    bsums[nblocks] = carry;
    /* fused reservation (single writer; ticks are stream-ordered) */
    uint2 tot = carry;
    uint64_t need_hdr = (((uint64_t)tot.x * sizeof(wb::RecHdr)) + 15) & ~15ULL;
    uint64_t need_pay = tot.y;
    uint64_t need = need_hdr + need_pay;
    TickPlace p = {};
    p.total_rec = tot.x;
    p.payload_bytes = need_pay;
    uint64_t cur = *cursor;
    if ((cur % cap) + need > cap) cur += cap - (cur % cap); /* no straddle */
    if (need > cap || (!ring && cur + need > cap) ||
        (uint64_t)tot.x * 2 > task_cap) {
      p.overflow = 1;
      p.total_rec = 0;
      atomicAdd(&err_ring[tick % kErrRing], 1u);
    } else {
      p.cur = cur;
      p.hdr_off = cur % cap;
      p.payload_off = (cur % cap) + need_hdr;
      *cursor = cur + need;
    }
    *place = p;
    err_ring[(tick + kErrRing / 2) % kErrRing] = 0; /* re-zero a future slot */
  }
}

__global__ void k_emit(const uint8_t *__restrict__ blobs,
                       const UpdDesc *__restrict__ descs, uint32_t n,
                       const wb::WalkTotals *__restrict__ totals,
                       const uint2 *__restrict__ partial,
                       const uint2 *__restrict__ bsums,
                       const TickPlace *__restrict__ place,
                       uint8_t *__restrict__ store,
                       CopyTask *__restrict__ tasks,
                       const wb::Rec *__restrict__ reccache) {
  uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  if (place->overflow) return;
  wb::WalkTotals t = totals[i];
  if (!t.ok || t.n_records == 0) return;
  UpdDesc d = descs[i];
  uint2 base = add2(partial[i], bsums[i >> 8]); /* exclusive scan, inline */
  wb::RecHdr *hdrs = (wb::RecHdr *)(store + place->hdr_off);
  uint8_t *pay_region = store + place->payload_off;
  uint32_t pay = base.y;
  uint32_t rec = base.x;
  auto emit_one = [&](const wb::Rec &r, uint32_t idx) {
    uint32_t cf4 = r.cf_id ? 4u : 0u;
    /* CF range tombstones prefix BOTH slices: the end key (value slice)
     * lives in the same cf-namespaced key space as the begin key, so
     * coverage comparisons against cf-prefixed query keys stay consistent
     * (mirrored in host_build_run and the oracle) */
    uint8_t btag = wb::base_tag(r.tag);
    uint32_t cfv = (cf4 && btag == wb::kRangeDeletion) ? 4u : 0u;
    wb::RecHdr h;
    h.seq = d.base_seq + idx;
    h.kv_off = pay;
    h.val_len = r.val_len + cfv;
    h.key_len = (uint16_t)(r.key_len + cf4);
    h.type = btag;
    h.flags = cf4 ? 1 : 0;
    h.kpref = 0; /* the read index is built LAZILY at first multiget
                    (k_kpref, one coalesced pass over the run) — every
                    apply-time fill variant measured 160-480 us/tick
                    against the headline (profiles/r02): k_copy scattered
                    header RMWs +480, emit cold gather +158, walk-side
                    fold +180 in k_decode. rocksdb's analog: filters are
                    built at flush, not per write. */
    hdrs[rec + idx] = h;
    if (cf4) { /* record start is 16-B aligned -> u32 store is aligned */
      *(uint32_t *)(pay_region + pay) = r.cf_id;
    }
    CopyTask tk;
    tk.src_off = d.off + r.key_off;
    tk.dst_rel = pay + cf4;
    tk.nbytes = r.key_len;
    tasks[2 * (rec + idx)] = tk;
    if (cfv) { /* unaligned position: byte stores */
      uint8_t *p = pay_region + pay + cf4 + r.key_len;
      p[0] = (uint8_t)r.cf_id;
      p[1] = (uint8_t)(r.cf_id >> 8);
      p[2] = (uint8_t)(r.cf_id >> 16);
      p[3] = (uint8_t)(r.cf_id >> 24);
    }
    tk.src_off = d.off + r.val_off;
    tk.dst_rel = pay + cf4 + r.key_len + cfv;
    tk.nbytes = r.val_len;
    tasks[2 * (rec + idx) + 1] = tk;
    pay += (cf4 + r.key_len + cfv + r.val_len + 15u) & ~15u;
  };
  if (t.n_records <= kRecCache) { /* decode cached these — no blob re-walk */
    const wb::Rec *cache = reccache + (size_t)i * kRecCache;
    for (uint32_t idx = 0; idx < t.n_records; idx++) emit_one(cache[idx], idx);
  } else {
    wb::walk_f(blobs + d.off, d.len, emit_one);
  }
}

/* G-lane-group cooperative copy, dword funnel for misaligned sources (blob
 * arena is over-allocated by 16 B so the +1 word read never faults).
 * Measured on gfx950 (scripts/micro_copy.hip): the dwordx4-store variant
 * below is +52%/+24% over this at 1KB/128B values. */
template <int G>
__device__ __forceinline__ void copy_dword(uint8_t *dst, const uint8_t *src,
                                           uint32_t n, uint32_t lane) {
  uint32_t head = (uint32_t)((0u - (uint32_t)(uintptr_t)dst) & 3u);
  if (head > n) head = n;
  if (lane < head) dst[lane] = src[lane];
  dst += head;
  src += head;
  n -= head;
  uint32_t nw = n >> 2;
  uint32_t r = (uint32_t)((uintptr_t)src & 3);
  const uint32_t *asrc = (const uint32_t *)(src - r);
  uint32_t *adst = (uint32_t *)dst;
  if (r == 0) {
    for (uint32_t w = lane; w < nw; w += G) adst[w] = asrc[w];
  } else {
    uint32_t sh = 8 * r;
    for (uint32_t w = lane; w < nw; w += G)
      adst[w] = (asrc[w] >> sh) | (asrc[w + 1] << (32 - sh));
  }
  uint32_t done = nw << 2, tail = n & 3;
  if (lane < tail) dst[done + lane] = src[done + lane];
}

/* dwordx4 stores + dword-funnel gather; needs 16B-aligned dst (record slots
 * are 16B-aligned; value slices at +klen are aligned for 16B keys). */
template <int G>
__device__ __forceinline__ void copy_dwordx4(uint8_t *dst, const uint8_t *src,
                                             uint32_t n, uint32_t lane) {
  if (((uintptr_t)dst & 15) != 0) {
    copy_dword<G>(dst, src, n, lane);
    return;
  }
  uint32_t r = (uint32_t)((uintptr_t)src & 3);
  const uint32_t *asrc = (const uint32_t *)(src - r);
  uint4 *d4 = (uint4 *)dst;
  uint32_t nc = n >> 4;
  if (r == 0) {
    for (uint32_t c = lane; c < nc; c += G) {
      uint32_t w = c * 4;
      d4[c] = make_uint4(asrc[w], asrc[w + 1], asrc[w + 2], asrc[w + 3]);
    }
  } else {
    uint32_t sh = 8 * r, ish = 32 - sh;
    for (uint32_t c = lane; c < nc; c += G) {
      uint32_t w = c * 4;
      uint32_t a0 = asrc[w], a1 = asrc[w + 1], a2 = asrc[w + 2],
               a3 = asrc[w + 3], a4 = asrc[w + 4];
      d4[c] = make_uint4((a0 >> sh) | (a1 << ish), (a1 >> sh) | (a2 << ish),
                         (a2 >> sh) | (a3 << ish), (a3 >> sh) | (a4 << ish));
    }
  }
  uint32_t done = nc << 4;
  for (uint32_t b = done + lane; b < n; b += G) dst[b] = src[b];
}

template <int G>
__global__ void __launch_bounds__(256) k_copy(const uint8_t *__restrict__ blobs,
                                              uint8_t *__restrict__ store,
                                              const TickPlace *__restrict__ place,
                                              const CopyTask *__restrict__ tasks) {
  if (place->overflow) return;
  uint32_t ntasks = place->total_rec * 2;
  uint32_t lane = threadIdx.x & (G - 1);
  uint32_t g = (blockIdx.x * blockDim.x + threadIdx.x) / G;
  uint32_t ngroups = (gridDim.x * blockDim.x) / G;
  uint8_t *pay_region = store + place->payload_off;
  for (uint32_t t = g; t < ntasks; t += ngroups) {
    CopyTask tk = tasks[t];
    if (tk.nbytes == 0) continue;
    copy_dwordx4<G>(pay_region + tk.dst_rel, blobs + tk.src_off, tk.nbytes, lane);
  }
}

/* Drain-host mode (the north star's literal "drain sorted runs back to the
 * host memtables"): stream the tick's header + payload regions from the
 * device store STRAIGHT INTO mapped pinned host memory. A kernel (not
 * hipMemcpyAsync) because only the device knows the tick's placement
 * (TickPlace is computed by k_scan2); grid-stride uint4 writes over PCIe.
 * Host arena layout: [hdrs, 16-aligned][payload]. */
__global__ void k_drain(const uint8_t *__restrict__ store,
                        const TickPlace *__restrict__ place,
                        uint8_t *__restrict__ hostbuf) {
  if (place->overflow) return;
  uint64_t hdr_bytes = (uint64_t)place->total_rec * sizeof(wb::RecHdr);
  uint64_t hb16 = (hdr_bytes + 15) & ~15ull;
  uint64_t pay = place->payload_bytes;
  const uint8_t *hsrc = store + place->hdr_off;
  const uint8_t *psrc = store + place->payload_off;
  size_t stride = (size_t)gridDim.x * blockDim.x * 16;
  size_t i0 = ((size_t)blockIdx.x * blockDim.x + threadIdx.x) * 16;
  for (size_t b = i0; b < hdr_bytes; b += stride)
    *(uint4 *)(hostbuf + b) = *(const uint4 *)(hsrc + b);
  uint8_t *pdst = hostbuf + hb16;
  for (size_t b = i0; b < pay; b += stride)
    *(uint4 *)(pdst + b) = *(const uint4 *)(psrc + b);
}

__global__ void k_rundesc(const GroupDesc *__restrict__ groups, uint32_t ngroups,
                          const UpdDesc *__restrict__ descs,
                          const wb::WalkTotals *__restrict__ totals,
                          const uint2 *__restrict__ partial,
                          const uint2 *__restrict__ bsums, uint32_t n,
                          uint32_t nblocks,
                          const TickPlace *__restrict__ place,
                          DevRunDesc *__restrict__ out) {
  uint32_t g = blockIdx.x * blockDim.x + threadIdx.x;
  if (g >= ngroups) return;
  GroupDesc gd = groups[g];
  auto scan_at = [&](uint32_t j) {
    return j >= n ? bsums[nblocks] : add2(partial[j], bsums[j >> 8]);
  };
  uint2 s0 = scan_at(gd.first);
  uint2 s1 = scan_at(gd.first + gd.n_upds);
  DevRunDesc rd = {};
  rd.shard = gd.shard;
  rd.n_entries = s1.x - s0.x;
  rd.payload_bytes = s1.y - s0.y;
  rd.pay_rel_base = s0.y;
  if (!place->overflow) {
    rd.cur = place->cur;
    rd.hdr_off = place->hdr_off + (uint64_t)s0.x * sizeof(wb::RecHdr);
    rd.payload_off = place->payload_off + s0.y;
    rd.base_seq = descs[gd.first].base_seq;
    const UpdDesc last = descs[gd.first + gd.n_upds - 1];
    rd.last_seq = last.base_seq + totals[gd.first + gd.n_upds - 1].hdr_count - 1;
  } else {
    rd.n_entries = 0;
  }
  out[g] = rd;
}

/* ---------------- device read serving (multiget) ----------------
 * One block per query: scan the shard's runs newest->oldest; per matching
 * entry track the max-seq terminator (Put/Delete/SingleDelete), whether any
 * Merge sits above it (=> host fold), and the max covering range-tombstone
 * seq. Followers serve reads in rocksplicator deployments; with the
 * memtable resident in HBM the lookup runs there too. */
struct RunView {
  uint64_t hdr_off, payload_off;
  uint32_t n_entries, pay_rel_base;
};

__device__ inline int dev_memcmp(const uint8_t *a, const uint8_t *b, uint32_t n) {
  for (uint32_t i = 0; i < n; i++) {
    if (a[i] != b[i]) return a[i] < b[i] ? -1 : 1;
  }
  return 0;
}

/* per-query raw seq info so mixed device/host shards can merge the device
 * verdict with a host-run probe (gra_multiget) */
struct MgExtra {
  uint64_t term_seq, merge_seq, rd_seq;
  uint32_t term_type, _pad;
};

__global__ void k_multiget(const uint8_t *__restrict__ store,
                           const RunView *__restrict__ runs, uint32_t nruns,
                           const GraKeyRef *__restrict__ keys,
                           const uint8_t *__restrict__ keybuf, uint32_t nq,
                           uint8_t *__restrict__ valbuf, uint32_t val_stride,
                           GraGetResult *__restrict__ out,
                           MgExtra *__restrict__ extra, int use_kpref) {
  __shared__ uint64_t sh_term_seq[256];
  __shared__ uint64_t sh_term_ref[256]; /* (run<<32)|entry, ~0 = none */
  __shared__ uint64_t sh_merge_seq[256];
  __shared__ uint64_t sh_rd_seq[256];
  uint32_t q = blockIdx.x;
  if (q >= nq) return;
  const uint8_t *key = keybuf + keys[q].off;
  uint32_t klen = keys[q].len;
  uint32_t qpref = wb::key_fnv_fold(wb::kFnvBasis32, key, klen);
  uint64_t term_seq = 0, term_ref = ~0ULL, merge_seq = 0, rd_seq = 0;
  for (uint32_t r = 0; r < nruns; r++) {
    RunView rv = runs[r];
    const wb::RecHdr *hdrs = (const wb::RecHdr *)(store + rv.hdr_off);
    const uint8_t *pay = store + rv.payload_off;
    for (uint32_t i = threadIdx.x; i < rv.n_entries; i += blockDim.x) {
      wb::RecHdr h = hdrs[i];
      if (h.type != wb::kRangeDeletion &&
          (h.key_len != klen || (use_kpref && h.kpref != qpref)))
        continue; /* header-only reject: no payload touch */
      uint32_t rel = h.kv_off - rv.pay_rel_base;
      if (h.type == wb::kRangeDeletion) {
        const uint8_t *b = pay + rel, *e2 = pay + rel + h.key_len;
        uint32_t bl = h.key_len, el = h.val_len;
        int c1 = dev_memcmp(b, key, bl < klen ? bl : klen);
        if (c1 > 0 || (c1 == 0 && bl > klen)) continue;
        int c2 = dev_memcmp(key, e2, klen < el ? klen : el);
        if (c2 > 0 || (c2 == 0 && klen >= el)) continue;
        if (h.seq > rd_seq) rd_seq = h.seq;
        continue;
      }
      if (dev_memcmp(pay + rel, key, klen) != 0) /* len+prefix pre-checked */
        continue;
      if (h.type == wb::kMerge) {
        if (h.seq > merge_seq) merge_seq = h.seq;
      } else if (h.seq > term_seq) {
        term_seq = h.seq;
        term_ref = ((uint64_t)r << 32) | i;
      }
    }
  }
  uint32_t tid = threadIdx.x;
  sh_term_seq[tid] = term_seq;
  sh_term_ref[tid] = term_ref;
  sh_merge_seq[tid] = merge_seq;
  sh_rd_seq[tid] = rd_seq;
  __syncthreads();
  for (uint32_t ofs = blockDim.x / 2; ofs > 0; ofs >>= 1) {
    if (tid < ofs) {
      if (sh_term_seq[tid + ofs] > sh_term_seq[tid]) {
        sh_term_seq[tid] = sh_term_seq[tid + ofs];
        sh_term_ref[tid] = sh_term_ref[tid + ofs];
      }
      if (sh_merge_seq[tid + ofs] > sh_merge_seq[tid])
        sh_merge_seq[tid] = sh_merge_seq[tid + ofs];
      if (sh_rd_seq[tid + ofs] > sh_rd_seq[tid])
        sh_rd_seq[tid] = sh_rd_seq[tid + ofs];
    }
    __syncthreads();
  }
  uint64_t T = sh_term_seq[0], M = sh_merge_seq[0], RD = sh_rd_seq[0];
  uint64_t ref = sh_term_ref[0];
  uint8_t term_type = 0xFF;
  wb::RecHdr h = {};
  RunView rv = {};
  if (ref != ~0ULL) {
    rv = runs[ref >> 32];
    h = ((const wb::RecHdr *)(store + rv.hdr_off))[(uint32_t)ref];
    term_type = h.type;
  }
  if (tid == 0 && extra) { /* raw verdict for host-side merging */
    extra[q].term_seq = T;
    extra[q].merge_seq = M;
    extra[q].rd_seq = RD;
    extra[q].term_type = term_type;
  }
  uint64_t floor_seq = T > RD ? T : RD;
  if (M > floor_seq) { /* live merge operands: fold on the host */
    if (tid == 0) {
      out[q].status = GRA_GET_NEEDS_HOST;
      out[q].vlen = 0;
    }
    return;
  }
  if (T == 0 || T <= RD || ref == ~0ULL) {
    if (tid == 0) {
      out[q].status = GRA_GET_MISS;
      out[q].vlen = 0;
    }
    return;
  }
  if (h.type != wb::kValue) { /* Delete/SingleDelete terminator */
    if (tid == 0) {
      out[q].status = GRA_GET_MISS;
      out[q].vlen = 0;
    }
    return;
  }
  uint32_t vlen = h.val_len < val_stride ? h.val_len : val_stride;
  const uint8_t *src = store + rv.payload_off + (h.kv_off - rv.pay_rel_base) +
                       h.key_len;
  uint8_t *dst = valbuf + (size_t)q * val_stride;
  for (uint32_t b = tid; b < vlen; b += blockDim.x) dst[b] = src[b];
  if (tid == 0) {
    out[q].status = GRA_GET_FOUND;
    out[q].vlen = h.val_len; /* true length (caller sees truncation) */
  }
}

/* Lazy read-index build: one coalesced pass over a run fills every
 * header's key fingerprint from the STORED key bytes (contiguous in the
 * payload). Launched at a run's first multiget — apply-time fills all
 * measured 160-480 us/tick against the headline (see k_emit). */
__global__ void k_kpref(uint8_t *__restrict__ store,
                        const RunView *__restrict__ runs) {
  RunView rv = runs[blockIdx.y];
  wb::RecHdr *hdrs = (wb::RecHdr *)(store + rv.hdr_off);
  const uint8_t *pay = store + rv.payload_off;
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < rv.n_entries;
       i += gridDim.x * blockDim.x) {
    wb::RecHdr h = hdrs[i];
    hdrs[i].kpref =
        (h.type == wb::kRangeDeletion)
            ? 0 /* tombstones are range-matched, not hash-matched */
            : wb::key_fnv_fold(wb::kFnvBasis32,
                               pay + (h.kv_off - rv.pay_rel_base), h.key_len);
  }
}

/* ---------------- hash-join multiget ----------------
 * The linear per-query scan is header-bandwidth bound (nq × entries × 24 B).
 * Invert it: ONE coalesced pass over all entries probes an open-addressed
 * query-fingerprint table (kpref -> query idx) and appends the rare
 * matches to a candidate list; tiny resolve kernels then pick per-query
 * winners by seq. Traffic: O(entries + matches) instead of O(nq × entries).
 * Range tombstones are collected in the same pass and tested per query
 * separately (they cover key RANGES — not hash-matchable). Overflow of
 * either list falls back to the per-query scan kernel. */
struct MgCand {
  uint32_t qidx, run, entry, type;
  uint64_t seq;
};
struct MgTomb {
  uint32_t run, entry, _pad0, _pad1;
  uint64_t seq;
};

__global__ void k_mg_scan(const uint8_t *__restrict__ store,
                          const RunView *__restrict__ runs,
                          const uint64_t *__restrict__ qtab, uint32_t qmask,
                          const GraKeyRef *__restrict__ keys,
                          const uint8_t *__restrict__ keybuf,
                          MgCand *__restrict__ cands,
                          uint32_t *__restrict__ ncand, uint32_t cand_cap,
                          MgTomb *__restrict__ tombs,
                          uint32_t *__restrict__ ntomb, uint32_t tomb_cap) {
  uint32_t r = blockIdx.y;
  RunView rv = runs[r];
  const wb::RecHdr *hdrs = (const wb::RecHdr *)(store + rv.hdr_off);
  const uint8_t *pay = store + rv.payload_off;
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < rv.n_entries;
       i += gridDim.x * blockDim.x) {
    wb::RecHdr h = hdrs[i];
    if (h.type == wb::kRangeDeletion) {
      uint32_t slot = atomicAdd(ntomb, 1u);
      if (slot < tomb_cap) tombs[slot] = {r, i, 0, 0, h.seq};
      continue;
    }
    uint32_t s = h.kpref & qmask;
    while (true) { /* linear probe; duplicates of a fingerprint chain on */
      uint64_t e = qtab[s];
      if (e == 0) break;
      if ((uint32_t)(e >> 32) == h.kpref) {
        uint32_t qidx = (uint32_t)e - 1;
        uint32_t klen = keys[qidx].len;
        if (h.key_len == klen &&
            dev_memcmp(pay + (h.kv_off - rv.pay_rel_base),
                       keybuf + keys[qidx].off, klen) == 0) {
          uint32_t slot = atomicAdd(ncand, 1u);
          if (slot < cand_cap) cands[slot] = {qidx, r, i, h.type, h.seq};
        }
      }
      s = (s + 1) & qmask;
    }
  }
}

__global__ void k_mg_resolve1(const MgCand *__restrict__ cands,
                              const uint32_t *__restrict__ ncand_p,
                              uint32_t cap,
                              unsigned long long *__restrict__ term_pack,
                              unsigned long long *__restrict__ merge_seq) {
  uint32_t n = *ncand_p < cap ? *ncand_p : cap; /* device-side bound: no
                                                   host sync between scan
                                                   and resolve */
  uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  MgCand c = cands[i];
  if (c.type == wb::kMerge)
    atomicMax(&merge_seq[c.qidx], (unsigned long long)c.seq);
  else /* seqs are unique per shard: the packed max has ONE winner */
    atomicMax(&term_pack[c.qidx],
              (unsigned long long)((c.seq << 8) | c.type));
}

__global__ void k_mg_tombs(const uint8_t *__restrict__ store,
                           const RunView *__restrict__ runs,
                           const MgTomb *__restrict__ tombs,
                           const uint32_t *__restrict__ ntomb_p, uint32_t cap,
                           const GraKeyRef *__restrict__ keys,
                           const uint8_t *__restrict__ keybuf, uint32_t nq,
                           unsigned long long *__restrict__ rd_seq) {
  uint32_t ntomb = *ntomb_p < cap ? *ntomb_p : cap;
  uint32_t q = blockIdx.x * blockDim.x + threadIdx.x;
  if (q >= nq) return;
  const uint8_t *key = keybuf + keys[q].off;
  uint32_t klen = keys[q].len;
  uint64_t best = 0;
  for (uint32_t t = 0; t < ntomb; t++) {
    MgTomb mt = tombs[t];
    RunView rv = runs[mt.run];
    const wb::RecHdr *hdrs = (const wb::RecHdr *)(store + rv.hdr_off);
    wb::RecHdr h = hdrs[mt.entry];
    const uint8_t *pay = store + rv.payload_off;
    uint32_t rel = h.kv_off - rv.pay_rel_base;
    const uint8_t *b = pay + rel, *e2 = pay + rel + h.key_len;
    uint32_t bl = h.key_len, el = h.val_len;
    int c1 = dev_memcmp(b, key, bl < klen ? bl : klen);
    if (c1 > 0 || (c1 == 0 && bl > klen)) continue;
    int c2 = dev_memcmp(key, e2, klen < el ? klen : el);
    if (c2 > 0 || (c2 == 0 && klen >= el)) continue;
    if (h.seq > best) best = h.seq;
  }
  rd_seq[q] = best;
}

__global__ void k_mg_resolve2(const MgCand *__restrict__ cands,
                              const uint32_t *__restrict__ ncand_p,
                              uint32_t cap,
                              const unsigned long long *__restrict__ term_pack,
                              unsigned long long *__restrict__ winner_ref) {
  uint32_t n = *ncand_p < cap ? *ncand_p : cap;
  uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  MgCand c = cands[i];
  if (c.type != wb::kMerge &&
      term_pack[c.qidx] == (unsigned long long)((c.seq << 8) | c.type))
    winner_ref[c.qidx] = ((unsigned long long)c.run << 32) | c.entry;
}

__global__ void k_mg_emit(const uint8_t *__restrict__ store,
                          const RunView *__restrict__ runs,
                          const unsigned long long *__restrict__ term_pack,
                          const unsigned long long *__restrict__ merge_seq,
                          const unsigned long long *__restrict__ rd_seq,
                          const unsigned long long *__restrict__ winner_ref,
                          uint32_t nq, uint8_t *__restrict__ valbuf,
                          uint32_t val_stride, GraGetResult *__restrict__ out,
                          MgExtra *__restrict__ extra) {
  uint32_t q = blockIdx.x;
  if (q >= nq) return;
  uint64_t tp = term_pack[q];
  uint64_t T = tp >> 8, M = merge_seq[q], RD = rd_seq[q];
  uint8_t ttype = tp ? (uint8_t)tp : 0xFF;
  if (threadIdx.x == 0 && extra) {
    extra[q].term_seq = T;
    extra[q].merge_seq = M;
    extra[q].rd_seq = RD;
    extra[q].term_type = ttype;
  }
  uint64_t fl = T > RD ? T : RD;
  if (M > fl) {
    if (threadIdx.x == 0) {
      out[q].status = GRA_GET_NEEDS_HOST;
      out[q].vlen = 0;
    }
    return;
  }
  if (T == 0 || T <= RD || ttype != wb::kValue) {
    if (threadIdx.x == 0) {
      out[q].status = GRA_GET_MISS;
      out[q].vlen = 0;
    }
    return;
  }
  unsigned long long ref = winner_ref[q];
  RunView rv = runs[ref >> 32];
  const wb::RecHdr *hdrs = (const wb::RecHdr *)(store + rv.hdr_off);
  wb::RecHdr h = hdrs[(uint32_t)ref];
  uint32_t vlen = h.val_len < val_stride ? h.val_len : val_stride;
  const uint8_t *src =
      store + rv.payload_off + (h.kv_off - rv.pay_rel_base) + h.key_len;
  uint8_t *dst = valbuf + (size_t)q * val_stride;
  for (uint32_t b = threadIdx.x; b < vlen; b += blockDim.x) dst[b] = src[b];
  if (threadIdx.x == 0) {
    out[q].status = GRA_GET_FOUND;
    out[q].vlen = h.val_len;
  }
}

/* ---- full-store checksum (parity at any size) ----
 * Same record hash as the oracle's orc_shard_checksum: FNV-1a over
 * (seq LE8 | type | key_len LE4 | val_len LE4 | key | val), u64-ADD
 * combined (order-independent). */
__host__ __device__ inline uint64_t rec_hash_cs(uint64_t seq, uint8_t type,
                                                uint32_t klen, uint32_t vlen,
                                                const uint8_t *key,
                                                const uint8_t *val) {
  uint64_t h = 1469598103934665603ULL;
#define GRA_FOLD(b) h = (h ^ (uint8_t)(b)) * 1099511628211ULL
  for (int i = 0; i < 8; i++) GRA_FOLD(seq >> (8 * i));
  GRA_FOLD(type);
  for (int i = 0; i < 4; i++) GRA_FOLD(klen >> (8 * i));
  for (int i = 0; i < 4; i++) GRA_FOLD(vlen >> (8 * i));
  for (uint32_t i = 0; i < klen; i++) GRA_FOLD(key[i]);
  for (uint32_t i = 0; i < vlen; i++) GRA_FOLD(val[i]);
#undef GRA_FOLD
  return h;
}

__global__ void k_checksum(const uint8_t *__restrict__ store,
                           const RunView *__restrict__ runs, uint32_t nruns,
                           unsigned long long *__restrict__ out) {
  uint64_t local = 0;
  for (uint32_t r = blockIdx.x; r < nruns; r += gridDim.x) {
    RunView rv = runs[r];
    const wb::RecHdr *hdrs = (const wb::RecHdr *)(store + rv.hdr_off);
    const uint8_t *pay = store + rv.payload_off;
    for (uint32_t i = threadIdx.x; i < rv.n_entries; i += blockDim.x) {
      wb::RecHdr h = hdrs[i];
      uint32_t rel = h.kv_off - rv.pay_rel_base;
      local += rec_hash_cs(h.seq, h.type, h.key_len, h.val_len, pay + rel,
                           pay + rel + h.key_len);
    }
  }
  /* wave-reduce then one atomic per wave */
  for (int ofs = 32; ofs > 0; ofs >>= 1)
    local += __shfl_down(local, ofs, 64);
  if ((threadIdx.x & 63) == 0 && local)
    atomicAdd(out, (unsigned long long)local);
}

static bool gra_check_staging() {
  static const bool on = [] {
    const char *v = getenv("GRA_CHECK_STAGING");
    return v && v[0] == '1';
  }();
  return on;
}

/* ---------------- host engine ---------------- */

struct Stats {
  double h2d_ms = 0, snappy_ms = 0, decode_ms = 0, scan_ms = 0, emit_ms = 0,
         copy_ms = 0, runfix_ms = 0, total_ms = 0;
  uint64_t ticks = 0, updates = 0, records = 0, blob_bytes = 0, payload_bytes = 0;
};

constexpr int kSlots = 8;
constexpr int kEventsPerTick = 12; /* 10 = decode start (prep stream),
                                      11 = decode done */

struct TickRec {
  int slot = -1;
  uint32_t n = 0, ngroups = 0;
  uint64_t blob_bytes = 0;
  bool h2d_timed = false;
  uint32_t evmask = 0; /* which ev[i] were recorded this tick */
  std::vector<uint32_t> counts; /* per-update record counts (error recovery) */
  std::shared_ptr<uint8_t> drain_buf; /* drain-host: this tick's pinned arena */
  hipEvent_t ev[kEventsPerTick];
};

struct Slot {
  GroupDesc *h_groups = nullptr;   /* pinned */
  TickPlace *h_place = nullptr;    /* pinned mirror of this tick's placement */
  DevRunDesc *h_rundescs = nullptr;
  DevRunDesc *d_rundescs = nullptr; /* per-slot device buffer so the D2H can
                                       overlap the next tick's kernels */
  uint8_t *d_ok = nullptr;         /* per-update validity from k_decode */
  uint8_t *h_ok = nullptr;         /* pinned mirror (read only on error) */
  uint32_t *h_err = nullptr;
  UpdDesc *h_descs = nullptr;      /* staging-path desc upload */
  bool busy = false;
};

} // namespace gra

using namespace gra;

struct GraEngine {
  GraEngineOpts opts;
  hipStream_t stream = nullptr;      /* pipeline kernels */
  hipStream_t copyout = nullptr;     /* run-descriptor D2H, overlapped */
  hipStream_t h2d = nullptr;         /* PCIe staging H2D, overlapped: tick
                                        N+1's copy runs under tick N's
                                        kernels (double-buffered device
                                        staging below) */
  hipStream_t prep = nullptr;        /* pre-stage kernels (k_snappy):
                                        tick N's decompress runs under tick
                                        N-1's decode..copy. Scratch slots
                                        are per-update (no aliasing across
                                        windows), so the only hazard is a
                                        WINDOW being re-decompressed while
                                        an older tick of the same window
                                        still reads it — fenced by the
                                        per-window consumed event the
                                        caller passes (TickPlan::
                                        consumed_ev), not by a global
                                        gate (a global gate serializes
                                        prep behind main and kills the
                                        overlap). */
  /* device store */
  uint8_t *d_store = nullptr;
  uint64_t *d_cursor = nullptr;
  /* tick scratch */
  uint32_t max_upd;          /* max updates per tick */
  uint32_t task_cap;
  uint32_t group_cap;        /* max contiguous shard-groups per tick */
  /* decode scratch is PARITY-DOUBLED: tick N+1's k_decode runs on the
   * prep stream under tick N's k_copy (decode is a sparse latency-bound
   * walk, copy is HBM-bound — they compose), so consecutive ticks must
   * not share totals/scans/reccache. scratch_used_ev[p] (recorded after
   * a tick's last scratch reader, k_rundesc) gates reuse. */
  wb::WalkTotals *d_totals_b[2] = {nullptr, nullptr};
  uint2 *d_partial_b[2] = {nullptr, nullptr};
  uint2 *d_bsums_b[2] = {nullptr, nullptr};
  wb::Rec *d_reccache_b[2] = {nullptr, nullptr};
  hipEvent_t scratch_used_ev[2] = {nullptr, nullptr};
  CopyTask *d_tasks = nullptr;
  TickPlace *d_place = nullptr; /* kSlots entries: per-tick placement slot
                                   (the drain D2H snapshot on copyout must
                                   not race the next tick's k_scan2) */
  GroupDesc *d_groups = nullptr;
  uint32_t *d_err_ring = nullptr; /* kErrRing slots, zeroed at init */
  uint32_t tick_id = 0;
  /* streaming-ingest staging: double-buffered pinned, LOCK-FREE writer
   * reservation (atomic byte/slot tickets; abandoned slots get len = 0 and
   * are filtered at tick build) so concurrent HandleReplicateResponse
   * callers never serialize on an engine lock */
  /* writer registration is STRIPED per thread slot: a shared writers
   * counter costs two seq_cst RMWs on one cacheline per update, which
   * capped the 16-thread C++ streaming ingest at a few M updates/s.
   * Each thread RMWs its own line; the builder's quiesce scans all
   * slots (same Dekker pairing per slot). */
  static constexpr int kWriterSlots = 256;
  struct WriterSlot {
    alignas(64) std::atomic<uint32_t> n{0};
  };
  struct StageBuf {
    uint8_t *pin = nullptr;
    GraUpdateDesc *descs = nullptr; /* plain host alloc, max_upd slots */
    alignas(64) std::atomic<uint64_t> pos{0};
    alignas(64) std::atomic<uint32_t> nslots{0};
    alignas(64) std::atomic<bool> closed{false};
    alignas(64) std::atomic<uint32_t> staged{0}; /* successful desc writes
        (cross-checked against the tick build's count) */
    WriterSlot writers[kWriterSlots];
    uint64_t epoch = 0; /* bumped at every swap: invalidates thread chunks */
    hipEvent_t free_ev = nullptr; /* recorded after this buffer's H2D */
  };
  StageBuf stage[2];
  std::atomic<StageBuf *> cur_stage{nullptr};
  /* double-buffered device staging: staged ticks alternate buffers so the
   * H2D into buffer B (h2d stream) overlaps the kernels still reading
   * buffer A (main stream). stage_used_ev[b] is recorded on the main
   * stream after the last kernel reading buffer b; the next H2D into b
   * waits on it. d_stage_blobs/d_stage_descs alias buffer 0 (capacity
   * checks + call-site compatibility). */
  uint8_t *d_stage_blobs_bufs[2] = {nullptr, nullptr};
  UpdDesc *d_stage_descs_bufs[2] = {nullptr, nullptr};
  hipEvent_t stage_used_ev[2] = {nullptr, nullptr};
  /* host-side submission gate: at most 2 staged copies queued on the h2d
   * stream. Measured on MI355X (profiles/r02): with >4 queued async
   * copies the submitting thread blocks INSIDE hipMemcpyAsync and the
   * in-flight SDMA transfers degrade 56 -> 37 GB/s; blocking the host in
   * hipEventSynchronize instead keeps them at full rate. */
  hipEvent_t h2d_gate_ev[2] = {nullptr, nullptr};
  uint8_t *d_stage_blobs = nullptr;
  UpdDesc *d_stage_descs = nullptr;
  /* slots + pending ticks */
  Slot slots[kSlots];
  std::deque<TickRec> pending;
  std::vector<hipEvent_t> event_pool;
  /* multiget staging cache (grow-only; guarded by mu) */
  struct MgCache {
    RunView *d_runs = nullptr;
    GraKeyRef *d_keys = nullptr;
    uint8_t *d_keybuf = nullptr, *d_valbuf = nullptr;
    GraGetResult *d_out = nullptr;
    void *d_extra = nullptr; /* MgExtra[] for mixed-shard merging */
    void *d_qtab = nullptr;  /* hash-join: query fingerprint table (u64) */
    void *d_cands = nullptr; /* MgCand[] matches */
    void *d_tombs = nullptr; /* MgTomb[] range tombstones seen in the scan */
    void *d_aux = nullptr;   /* 4 x nq u64: term_pack/merge/rd/winner */
    uint32_t *d_counts = nullptr; /* {ncand, ntomb} */
    uint32_t *h_counts = nullptr; /* pinned mirror */
    size_t runs_cap = 0, keys_cap = 0, keybuf_cap = 0, valbuf_cap = 0,
           out_cap = 0, extra_cap = 0, qtab_cap = 0, cands_cap = 0,
           tombs_cap = 0, aux_cap = 0;
  } mg;
  /* drain-host pinned arena pool. DECLARED BEFORE shards: runs hold
   * shared_ptrs whose deleter returns buffers here, so the pool must be
   * destroyed after the shards release them. */
  std::mutex drain_mu;
  std::vector<std::pair<uint8_t *, size_t>> drain_pool;
  bool closing = false; /* deleter frees directly during teardown */
  std::shared_ptr<uint8_t> drain_alloc(size_t need);
  /* shards */
  std::vector<ShardState> shards;
  Stats stats;
  std::mutex mu; /* engine-level: staging + tick enqueue + ingest */

  int init(const GraEngineOpts &o);
  ~GraEngine();
  int enqueue_tick(const uint8_t *d_blobs, const UpdDesc *d_descw, uint32_t n,
                   const std::vector<GroupDesc> &groups, uint64_t blob_bytes,
                   bool time_h2d, const void *h2d_src = nullptr,
                   size_t h2d_bytes = 0, uint8_t *d_h2d_dst = nullptr,
                   const UpdDesc *h_descs_src = nullptr,
                   const GroupDesc *d_groups_dev = nullptr,
                   const uint8_t *d_comp = nullptr,
                   const SnapTask *d_snaptasks = nullptr,
                   std::vector<uint32_t> &&counts = {},
                   hipEvent_t window_ev = nullptr);
  int ingest(bool wait_all);
  int ingest_one(TickRec &t, bool wait);
  int flush_locked();
  int stream_tick_locked();
  hipEvent_t get_event();
  void put_event(hipEvent_t e);
  int free_slot();
};

struct GraDb {
  GraEngine *e;
  uint32_t shard;
};

struct TickPlan {
  std::vector<GroupDesc> groups;
  uint64_t blob_bytes = 0;
  GroupDesc *d_groups = nullptr; /* device-cached copy (freed with replay) */
  UpdDesc *d_ud = nullptr;       /* tick_h2d: window descs REBASED to the
                                    staged window, cached on device — the
                                    per-step 20 MB desc rebase + H2D that
                                    throttled the staged leg to ~39 GB/s
                                    happens once per window instead */
  hipEvent_t consumed_ev = nullptr; /* snappy overlap: recorded after each
                                       tick of this window finishes reading
                                       its scratch slots (k_copy); the next
                                       k_snappy of the SAME window waits on
                                       it */
  SnapTask *d_snap = nullptr;    /* window's snap tasks, LENGTH-SORTED so a
                                    wave's lanes get similar-size streams
                                    (+15% k_snappy, scripts/micro_snappy.hip
                                    v2-lensorted; out_off rides with the
                                    task, results land identically) */
};

struct GraReplay {
  GraEngine *e = nullptr;
  uint8_t *d_blobs = nullptr;
  bool external_blobs = false; /* caller-owned device arena (gra_upload_dev) */
  UpdDesc *d_descs = nullptr;
  std::vector<UpdDesc> descs; /* host copy */
  std::vector<uint16_t> counts;
  size_t arena_bytes = 0;
  const uint8_t *h_arena = nullptr; /* for tick_h2d (must stay alive) */
  /* config #5: compressed transport — d_blobs becomes the uncompressed
   * scratch arena, blobs decompressed per tick by k_snappy */
  uint8_t *d_comp = nullptr;
  SnapTask *d_snaptasks = nullptr;
  std::vector<SnapTask> snap_tasks; /* host copy for per-window sorting */
  bool snappy = false;
  std::map<std::pair<uint64_t, uint64_t>, TickPlan> plans; /* window cache */
};

std::shared_ptr<uint8_t> GraEngine::drain_alloc(size_t need) {
  uint8_t *p = nullptr;
  size_t cap = 0;
  {
    std::lock_guard<std::mutex> lk(drain_mu);
    for (auto it = drain_pool.begin(); it != drain_pool.end(); ++it) {
      if (it->second >= need) {
        p = it->first;
        cap = it->second;
        drain_pool.erase(it);
        break;
      }
    }
  }
  if (!p) {
    cap = need + 64;
    if (hipHostMalloc(&p, cap) != hipSuccess) return nullptr;
  }
  GraEngine *e = this;
  return std::shared_ptr<uint8_t>(p, [e, cap](uint8_t *q) {
    std::lock_guard<std::mutex> lk(e->drain_mu);
    if (e->closing) {
      (void)hipHostFree(q);
    } else {
      e->drain_pool.push_back({q, cap});
    }
  });
}

hipEvent_t GraEngine::get_event() {
  if (!event_pool.empty()) {
    hipEvent_t e = event_pool.back();
    event_pool.pop_back();
    return e;
  }
  hipEvent_t e;
  (void)hipEventCreate(&e);
  return e;
}
void GraEngine::put_event(hipEvent_t e) { event_pool.push_back(e); }

int GraEngine::init(const GraEngineOpts &o) {
  opts = o;
  if (opts.store_bytes == 0) opts.store_bytes = 4ULL << 30;
  if (opts.staging_bytes == 0) opts.staging_bytes = 256ULL << 20;
  if (opts.max_wb_records == 0) opts.max_wb_records = 1024;
  max_upd = 1u << 20;
  task_cap = 4u << 20;
  group_cap = opts.nshards + 1 > 65536 ? opts.nshards + 1 : 65536;
  (void)hipGetLastError(); /* consume any sticky per-thread error a prior
                              (possibly intentionally failing) call left —
                              our launch checks would misattribute it */
  int ndev = 0;
  hipError_t de = hipGetDeviceCount(&ndev);
  if (de != hipSuccess || ndev == 0) {
    g_err = "no HIP device present — the GPU apply path refuses to run "
            "(no CPU fallback by design)";
    return GRA_NO_GPU;
  }
  if (opts.device >= 0) {
    hipError_t se = hipSetDevice(opts.device);
    if (se != hipSuccess) {
      g_err = std::string("hipSetDevice: ") + hipGetErrorString(se);
      (void)hipGetLastError(); /* clear the sticky error we just caused */
      return GRA_ERR;
    }
  }
  HIP_TRY(hipStreamCreate(&stream));
  HIP_TRY(hipStreamCreate(&copyout));
  HIP_TRY(hipStreamCreate(&h2d));
  HIP_TRY(hipStreamCreate(&prep));
  HIP_TRY(hipMalloc(&d_store, opts.store_bytes + 16));
  HIP_TRY(hipMalloc(&d_cursor, 8));
  HIP_TRY(hipMemset(d_cursor, 0, 8));
  for (int i = 0; i < 2; i++) {
    HIP_TRY(hipMalloc(&d_totals_b[i], (size_t)max_upd * sizeof(wb::WalkTotals)));
    HIP_TRY(hipMalloc(&d_partial_b[i], (size_t)max_upd * sizeof(uint2)));
    HIP_TRY(hipMalloc(&d_bsums_b[i], ((size_t)max_upd / 256 + 2) * sizeof(uint2)));
    HIP_TRY(hipMalloc(&d_reccache_b[i], (size_t)max_upd * 2 * sizeof(wb::Rec)));
    HIP_TRY(hipEventCreateWithFlags(&scratch_used_ev[i], hipEventDisableTiming));
    HIP_TRY(hipEventRecord(scratch_used_ev[i], stream));
  }
  HIP_TRY(hipMalloc(&d_tasks, (size_t)task_cap * sizeof(CopyTask)));
  HIP_TRY(hipMalloc(&d_place, kSlots * sizeof(TickPlace)));
  HIP_TRY(hipMalloc(&d_groups, (size_t)group_cap * sizeof(GroupDesc)));
  HIP_TRY(hipMalloc(&d_err_ring, kErrRing * 4));
  HIP_TRY(hipMemset(d_err_ring, 0, kErrRing * 4));
  for (int i = 0; i < 2; i++) {
    HIP_TRY(hipMalloc(&d_stage_blobs_bufs[i], opts.staging_bytes + 16));
    HIP_TRY(hipMalloc(&d_stage_descs_bufs[i], (size_t)max_upd * sizeof(UpdDesc)));
    HIP_TRY(hipEventCreate(&stage_used_ev[i]));
    HIP_TRY(hipEventRecord(stage_used_ev[i], stream));
    HIP_TRY(hipEventCreateWithFlags(&h2d_gate_ev[i], hipEventDisableTiming));
    HIP_TRY(hipEventRecord(h2d_gate_ev[i], h2d));
  }
  d_stage_blobs = d_stage_blobs_bufs[0];
  d_stage_descs = d_stage_descs_bufs[0];
  /* epochs must be unique ACROSS engine instances: a thread-local staging
   * chunk could otherwise match a freshly created engine reusing the same
   * heap address (ABA) */
  static std::atomic<uint64_t> g_epoch{1};
  /* staging pinned memory is written by MANY host threads and only read
   * by SDMA H2D: allocate it NON-COHERENT (cacheable on the CPU side) —
   * the default fine-grained pinned memory takes uncached CPU writes on
   * these hosts and capped the streaming fill around 20 GB/s.
   * GRA_STAGE_COHERENT=1 restores the default for A/B. */
  static const unsigned stage_flags = [] {
    const char *v = getenv("GRA_STAGE_COHERENT");
    return (v && v[0] == '1') ? hipHostMallocDefault
                              : hipHostMallocNonCoherent;
  }();
  for (int i = 0; i < 2; i++) {
    HIP_TRY(hipHostMalloc(&stage[i].pin, opts.staging_bytes + 16,
                          stage_flags));
    stage[i].descs = (GraUpdateDesc *)malloc((size_t)max_upd * sizeof(GraUpdateDesc));
    stage[i].epoch = g_epoch.fetch_add(1u << 20);
    HIP_TRY(hipEventCreate(&stage[i].free_ev));
    HIP_TRY(hipEventRecord(stage[i].free_ev, stream));
  }
  cur_stage.store(&stage[0]);
  for (int i = 0; i < kSlots; i++) {
    Slot &s = slots[i];
    HIP_TRY(hipHostMalloc(&s.h_groups, (size_t)group_cap * sizeof(GroupDesc)));
    HIP_TRY(hipHostMalloc(&s.h_place, sizeof(TickPlace)));
    HIP_TRY(hipHostMalloc(&s.h_rundescs, (size_t)group_cap * sizeof(DevRunDesc)));
    HIP_TRY(hipMalloc(&s.d_rundescs, (size_t)group_cap * sizeof(DevRunDesc)));
    HIP_TRY(hipMalloc(&s.d_ok, (size_t)max_upd));
    HIP_TRY(hipHostMalloc(&s.h_ok, (size_t)max_upd));
    HIP_TRY(hipHostMalloc(&s.h_err, 4));
    HIP_TRY(hipHostMalloc(&s.h_descs, (size_t)max_upd * sizeof(UpdDesc)));
  }
  if (opts.drain_host && opts.store_ring &&
      opts.store_bytes < 10 * opts.staging_bytes) {
    /* k_drain reads a tick's store region on the copyout stream AFTER the
     * main stream has moved on; a ring that wraps within the pipeline
     * depth could overwrite it first. Enforce the safe sizing rather than
     * risk a silent corruption (kSlots in-flight ticks + slack; a tick is
     * bounded by staging_bytes of input). */
    g_err = "drain_host with store_ring needs store_bytes >= 10x "
            "staging_bytes (ring wrap inside the drain pipeline)";
    return GRA_ERR;
  }
  shards = std::vector<ShardState>(opts.nshards);
  /* pre-create the event pool for a full pipeline (hipEventCreate mid-run
   * showed up as ~1 ms hiccups in kernel traces) */
  event_pool.reserve(kSlots * kEventsPerTick);
  for (int i = 0; i < kSlots * kEventsPerTick; i++) {
    hipEvent_t ev;
    HIP_TRY(hipEventCreate(&ev));
    event_pool.push_back(ev);
  }
  return GRA_OK;
}

GraEngine::~GraEngine() {
  {
    std::lock_guard<std::mutex> lk(drain_mu);
    closing = true; /* run releases during/after teardown free directly */
  }
  (void)hipStreamSynchronize(stream);
  (void)hipStreamSynchronize(copyout); /* pending rundesc/ok D2H target the
                                          pinned slot buffers freed below */
  (void)hipStreamSynchronize(h2d);     /* pending H2D reads pinned staging */
  for (auto &t : pending)
    for (int i = 0; i < kEventsPerTick; i++)
      if (t.ev[i]) (void)hipEventDestroy(t.ev[i]);
  for (auto e : event_pool) (void)hipEventDestroy(e);
  for (int i = 0; i < 2; i++) {
    if (stage[i].pin) (void)hipHostFree(stage[i].pin);
    free(stage[i].descs);
    if (stage[i].free_ev) (void)hipEventDestroy(stage[i].free_ev);
  }
  for (int i = 0; i < kSlots; i++) {
    Slot &s = slots[i];
    if (s.h_groups) (void)hipHostFree(s.h_groups);
    if (s.h_place) (void)hipHostFree(s.h_place);
    if (s.h_rundescs) (void)hipHostFree(s.h_rundescs);
    if (s.d_rundescs) (void)hipFree(s.d_rundescs);
    if (s.d_ok) (void)hipFree(s.d_ok);
    if (s.h_ok) (void)hipHostFree(s.h_ok);
    if (s.h_err) (void)hipHostFree(s.h_err);
    if (s.h_descs) (void)hipHostFree(s.h_descs);
  }
  for (void *p : {(void *)mg.d_runs, (void *)mg.d_keys, (void *)mg.d_keybuf,
                  (void *)mg.d_valbuf, (void *)mg.d_out, mg.d_extra,
                  mg.d_qtab, mg.d_cands, mg.d_tombs, mg.d_aux,
                  (void *)mg.d_counts})
    if (p) (void)hipFree(p);
  if (mg.h_counts) (void)hipHostFree(mg.h_counts);
  for (int i = 0; i < 2; i++) {
    if (d_stage_blobs_bufs[i]) (void)hipFree(d_stage_blobs_bufs[i]);
    if (d_stage_descs_bufs[i]) (void)hipFree(d_stage_descs_bufs[i]);
    if (stage_used_ev[i]) (void)hipEventDestroy(stage_used_ev[i]);
    if (h2d_gate_ev[i]) (void)hipEventDestroy(h2d_gate_ev[i]);
  }
  for (int i = 0; i < 2; i++) {
    for (void *p : {(void *)d_totals_b[i], (void *)d_partial_b[i],
                    (void *)d_bsums_b[i], (void *)d_reccache_b[i]})
      if (p) (void)hipFree(p);
    if (scratch_used_ev[i]) (void)hipEventDestroy(scratch_used_ev[i]);
  }
  for (void *p : {(void *)d_store, (void *)d_cursor, (void *)d_tasks,
                  (void *)d_place, (void *)d_groups, (void *)d_err_ring})
    if (p) (void)hipFree(p);
  shards.clear(); /* release run arenas before draining the pool */
  pending.clear();
  {
    std::lock_guard<std::mutex> lk(drain_mu);
    for (auto &pr : drain_pool) (void)hipHostFree(pr.first);
    drain_pool.clear();
  }
  if (stream) (void)hipStreamDestroy(stream);
  if (copyout) (void)hipStreamDestroy(copyout);
  if (h2d) (void)hipStreamDestroy(h2d);
  if (prep) (void)hipStreamDestroy(prep);
}

int GraEngine::free_slot() {
  for (;;) {
    for (int i = 0; i < kSlots; i++)
      if (!slots[i].busy) return i;
    /* all slots in flight — ingest the oldest pending tick (waiting) */
    int rc = ingest(false);
    if (rc != GRA_OK) return -1;
    if (!pending.empty()) {
      if (ingest_one(pending.front(), true) != GRA_OK) return -1;
      pending.pop_front();
    }
  }
}

int GraEngine::enqueue_tick(const uint8_t *d_blobs, const UpdDesc *d_descw,
                            uint32_t n, const std::vector<GroupDesc> &groups,
                            uint64_t blob_bytes, bool time_h2d,
                            const void *h2d_src, size_t h2d_bytes,
                            uint8_t *d_h2d_dst, const UpdDesc *h_descs_src,
                            const GroupDesc *d_groups_dev,
                            const uint8_t *d_comp,
                            const SnapTask *d_snaptasks,
                            std::vector<uint32_t> &&counts,
                            hipEvent_t window_ev) {
  if (n == 0) return GRA_OK;
  if (n > max_upd) {
    g_err = "tick exceeds max updates per tick";
    return GRA_ERR;
  }
  if (opts.drain_host && opts.store_ring && blob_bytes * 10 > opts.store_bytes) {
    /* same wrap hazard as the init-time guard, for replay ticks whose
     * size isn't bounded by staging_bytes */
    g_err = "drain_host with store_ring needs store_bytes >= 10x the tick "
            "size (ring wrap inside the drain pipeline)";
    return GRA_ERR;
  }
  if (groups.size() > group_cap) {
    g_err = "tick exceeds max shard-groups per tick";
    return GRA_ERR;
  }
  int si = free_slot();
  if (si < 0) return GRA_ERR;
  Slot &sl = slots[si];
  sl.busy = true;
  uint32_t ngroups = (uint32_t)groups.size();
  /* the host copy is ALWAYS filled: ingest's corruption recovery walks
   * sl.h_groups to find each group's update range (r01 bug: with a
   * device-cached plan the host copy stayed stale and the error path read
   * garbage groups — replay-tick corruption went undetected) */
  memcpy(sl.h_groups, groups.data(), ngroups * sizeof(GroupDesc));
  const GroupDesc *groups_for_kernel = d_groups_dev;
  if (!groups_for_kernel) groups_for_kernel = d_groups;
  uint32_t tick = tick_id++;

  TickRec t;
  t.slot = si;
  t.n = n;
  t.ngroups = ngroups;
  t.blob_bytes = blob_bytes;
  t.h2d_timed = time_h2d;
  t.counts = std::move(counts);
  for (int i = 0; i < kEventsPerTick; i++) t.ev[i] = get_event();
  static const bool detailed = [] {
    const char *v = getenv("GRA_DETAILED_EVENTS");
    return v && v[0] == '1';
  }();
  auto rec = [&](int i) {
    t.evmask |= 1u << i;
    return hipEventRecord(t.ev[i], stream);
  };

  if (!d_groups_dev) {
    HIP_TRY(hipMemcpyAsync(d_groups, sl.h_groups, ngroups * sizeof(GroupDesc),
                           hipMemcpyHostToDevice, stream));
  }
  HIP_TRY(rec(0)); /* tick start */
  int stage_buf = -1;
  if (h2d_src != nullptr) { /* PCIe-inclusive path: stage blobs (+descs) on
                             * the dedicated h2d stream, double-buffered —
                             * this tick's copy overlaps the previous tick's
                             * kernels (which read the other buffer) */
    stage_buf = (int)(tick & 1u);
    if (d_h2d_dst == d_stage_blobs_bufs[0]) { /* engine-managed staging */
      d_h2d_dst = d_stage_blobs_bufs[stage_buf];
      d_blobs = d_stage_blobs_bufs[stage_buf];
      if (h_descs_src) d_descw = d_stage_descs_bufs[stage_buf];
    }
    /* host gate: the copy that last used this buffer (2 staged ticks ago)
     * must have completed — bounds the SDMA queue to <=2 pending copies
     * (see h2d_gate_ev above for why) */
    HIP_TRY(hipEventSynchronize(h2d_gate_ev[stage_buf]));
    /* previous reader of this buffer must be done before overwrite */
    HIP_TRY(hipStreamWaitEvent(h2d, stage_used_ev[stage_buf], 0));
    t.evmask |= 1u << 9; /* h2d leg timed on its own stream: ev9 -> ev1 */
    HIP_TRY(hipEventRecord(t.ev[9], h2d));
    HIP_TRY(hipMemcpyAsync(d_h2d_dst, h2d_src, h2d_bytes,
                           hipMemcpyHostToDevice, h2d));
    if (h_descs_src) {
      memcpy(sl.h_descs, h_descs_src, (size_t)n * sizeof(UpdDesc));
      HIP_TRY(hipMemcpyAsync((void *)d_descw, sl.h_descs,
                             (size_t)n * sizeof(UpdDesc),
                             hipMemcpyHostToDevice, h2d));
    }
    t.evmask |= 1u << 1;
    HIP_TRY(hipEventRecord(t.ev[1], h2d));
    HIP_TRY(hipEventRecord(h2d_gate_ev[stage_buf], h2d));
    HIP_TRY(hipStreamWaitEvent(stream, t.ev[1], 0)); /* kernels gate on data */
  }
  uint32_t nb = (n + 255) / 256;
  static const bool snappy_overlap = [] { /* A/B escape hatch */
    const char *v = getenv("GRA_SNAPPY_OVERLAP");
    return !v || v[0] != '0';
  }();
  if (d_snaptasks) { /* config #5 pre-stage: decompress into the blob arena
                      * on the prep stream, overlapped with the previous
                      * tick's decode..copy. window_ev (recorded after the
                      * LAST tick of this same window finished reading the
                      * scratch) fences window reuse without serializing
                      * prep behind main. Both kernels are CU-bound, so the
                      * overlap mostly timeslices — kept because it never
                      * loses and wins when the main tick has SDMA phases. */
    hipStream_t ps = snappy_overlap ? prep : stream;
    if (window_ev && snappy_overlap)
      HIP_TRY(hipStreamWaitEvent(prep, window_ev, 0));
    t.evmask |= 1u << 9; /* snappy timed on its own stream: ev9 -> ev8 */
    HIP_TRY(hipEventRecord(t.ev[9], ps));
    hipLaunchKernelGGL(k_snappy, dim3(nb), dim3(256), 0, ps, d_comp,
                       d_snaptasks, n, (uint8_t *)d_blobs, d_err_ring, tick);
    HIP_TRY(hipGetLastError());
    t.evmask |= 1u << 8;
    HIP_TRY(hipEventRecord(t.ev[8], ps));
    if (snappy_overlap)
      HIP_TRY(hipStreamWaitEvent(stream, t.ev[8], 0)); /* decode gates on it */
  }
  /* decode runs on the PREP stream, overlapped with the PREVIOUS tick's
   * emit/copy on main (decode is a sparse latency-bound walk over record
   * headers; copy is HBM-bound — they compose). Parity scratch + the
   * scratch_used gate make consecutive ticks independent; main's scan2
   * gates on decode-done (ev11). GRA_DECODE_OVERLAP=0 restores the serial
   * order for A/B. */
  static const bool decode_overlap = [] {
    /* default OFF: the A/B measured +1% (noise) on 1 KB records and -9%
     * on 128 B ones (the cross-stream ev-wait costs more than the
     * overlap returns; see DESIGN §10.5b zero-sum finding) */
    const char *v = getenv("GRA_DECODE_OVERLAP");
    return v && v[0] == '1';
  }();
  int par = (int)(tick & 1u);
  wb::WalkTotals *t_totals = d_totals_b[par];
  uint2 *t_partial = d_partial_b[par];
  uint2 *t_bsums = d_bsums_b[par];
  wb::Rec *t_reccache = d_reccache_b[par];
  hipStream_t ds = decode_overlap ? prep : stream;
  if (decode_overlap) {
    /* previous user of this parity's scratch must have finished reading */
    HIP_TRY(hipStreamWaitEvent(prep, scratch_used_ev[par], 0));
    if (stage_buf >= 0) /* staged blobs land via the h2d stream */
      HIP_TRY(hipStreamWaitEvent(prep, t.ev[1], 0));
    t.evmask |= 1u << 10;
    HIP_TRY(hipEventRecord(t.ev[10], prep));
  }
  hipLaunchKernelGGL(k_decode, dim3(nb), dim3(256), 0, ds, d_blobs, d_descw,
                     n, t_totals, opts.max_wb_records, d_err_ring, tick,
                     t_partial, t_bsums, t_reccache, sl.d_ok);
  HIP_TRY(hipGetLastError());
  if (decode_overlap) {
    t.evmask |= 1u << 11;
    HIP_TRY(hipEventRecord(t.ev[11], prep));
    HIP_TRY(hipStreamWaitEvent(stream, t.ev[11], 0));
  } else if (detailed) {
    HIP_TRY(rec(2)); /* after decode(+scan1) */
  }
  TickPlace *place_slot = d_place + si;
  hipLaunchKernelGGL(k_scan2, dim3(1), dim3(256), 0, stream, t_bsums, nb,
                     d_cursor, place_slot, opts.store_bytes, opts.store_ring,
                     task_cap, d_err_ring, tick);
  HIP_TRY(hipGetLastError());
  if (detailed) HIP_TRY(rec(3)); /* after scan(+reserve) */
  hipLaunchKernelGGL(k_emit, dim3(nb), dim3(256), 0, stream, d_blobs, d_descw,
                     n, t_totals, t_partial, t_bsums, place_slot, d_store,
                     d_tasks, t_reccache);
  HIP_TRY(hipGetLastError());
  HIP_TRY(rec(4)); /* after emit (pre-copy) */
  /* group width by average update size (micro_copy.hip: g32 wins >=512B).
   * Grid: 2048 blocks = 8/CU = every wave slot — the next tick's decode
   * can only overlap if the copy leaves wave headroom, hence the
   * GRA_COPY_GRID knob for the A/B (profiles/r02). */
  static const uint32_t copy_grid = [] {
    const char *v = getenv("GRA_COPY_GRID");
    uint32_t g = v ? (uint32_t)atoi(v) : 2048;
    return g ? g : 2048;
  }();
  if (blob_bytes / (n ? n : 1) >= 512) {
    hipLaunchKernelGGL((k_copy<32>), dim3(copy_grid), dim3(256), 0, stream,
                       d_blobs, d_store, place_slot, d_tasks);
  } else {
    hipLaunchKernelGGL((k_copy<16>), dim3(copy_grid), dim3(256), 0, stream,
                       d_blobs, d_store, place_slot, d_tasks);
  }
  HIP_TRY(hipGetLastError());
  HIP_TRY(rec(5)); /* after copy */
  if (window_ev) /* this window's scratch fully consumed */
    HIP_TRY(hipEventRecord(window_ev, stream));
  hipLaunchKernelGGL(k_rundesc, dim3((ngroups + 255) / 256), dim3(256), 0,
                     stream, groups_for_kernel, ngroups, d_descw, t_totals,
                     t_partial, t_bsums, n, nb, place_slot, sl.d_rundescs);
  HIP_TRY(hipGetLastError());
  HIP_TRY(rec(6)); /* main-stream tick end */
  HIP_TRY(hipEventRecord(scratch_used_ev[par], stream)); /* scratch free */
  if (stage_buf >= 0) /* last reader of this staging buffer has retired */
    HIP_TRY(hipEventRecord(stage_used_ev[stage_buf], stream));
  /* publish run descriptors on the copyout stream, overlapped with the next
   * tick's kernels (per-slot device buffer: no hazard with slot reuse —
   * ingest waits on ev[7], which gates sl.busy) */
  HIP_TRY(hipStreamWaitEvent(copyout, t.ev[6], 0));
  HIP_TRY(hipMemcpyAsync(sl.h_rundescs, sl.d_rundescs,
                         (size_t)ngroups * sizeof(DevRunDesc),
                         hipMemcpyDeviceToHost, copyout));
  HIP_TRY(hipMemcpyAsync(sl.h_err, d_err_ring + (tick % kErrRing), 4,
                         hipMemcpyDeviceToHost, copyout));
  HIP_TRY(hipMemcpyAsync(sl.h_ok, sl.d_ok, n, hipMemcpyDeviceToHost, copyout));
  HIP_TRY(hipMemcpyAsync(sl.h_place, place_slot, sizeof(TickPlace),
                         hipMemcpyDeviceToHost, copyout));
  if (opts.drain_host && !t.counts.empty()) {
    /* drain the whole tick into a pinned host arena on the copyout stream,
     * overlapped with the next tick's kernels. Size bound: payload <=
     * blob + 7/record (cf re-prefix), + 15/record align pad + 24/record
     * headers. Without counts (no record total) ingest falls back to the
     * eager per-run fetch. */
    uint64_t recs = 0;
    for (uint32_t c : t.counts) recs += c;
    size_t need = (size_t)blob_bytes + 46 * recs + 80;
    t.drain_buf = drain_alloc(need);
    if (t.drain_buf) {
      hipLaunchKernelGGL(k_drain, dim3(1024), dim3(256), 0, copyout, d_store,
                         place_slot, t.drain_buf.get());
      HIP_TRY(hipGetLastError());
    }
  }
  HIP_TRY(hipEventRecord(t.ev[7], copyout)); /* publication done */
  pending.push_back(std::move(t));
  return GRA_OK;
}

static int fetch_run_impl(GraEngine *e, Run &r);

int GraEngine::ingest_one(TickRec &t, bool wait) {
  if (wait) {
    HIP_TRY(hipEventSynchronize(t.ev[7]));
  } else {
    hipError_t q = hipEventQuery(t.ev[7]);
    if (q == hipErrorNotReady) return GRA_NOT_FOUND; /* not done yet */
    if (q != hipSuccess) {
      g_err = std::string("tick event: ") + hipGetErrorString(q);
      return GRA_ERR;
    }
  }
  Slot &sl = slots[t.slot];
  /* stats: chained deltas over whichever events were recorded (fine-grained
   * stage events are opt-in via GRA_DETAILED_EVENTS; coarse mode lumps
   * decode+scan+emit into emit_ms) */
  float ms;
  auto has = [&](int i) { return (t.evmask >> i) & 1u; };
  auto dt = [&](int a, int b) {
    if (!has(a) || !has(b)) return 0.0;
    (void)hipEventElapsedTime(&ms, t.ev[a], t.ev[b]);
    return (double)ms;
  };
  int prev = 0;
  if (has(1)) {
    stats.h2d_ms += dt(has(9) ? 9 : 0, 1); /* ev9->ev1 = copy on the h2d
                                              stream (overlapped path) */
    prev = 1;
  }
  if (has(8)) {
    /* snappy runs on the prep stream, overlapped with the previous tick's
     * main-stream kernels: ev9->ev8 is its own duration; the main chain
     * (decode..copy) is unaffected by it */
    stats.snappy_ms += dt(has(9) ? 9 : prev, 8);
  }
  if (has(10) && has(11)) {
    /* decode overlapped on the prep stream: its own duration, main chain
     * unaffected */
    stats.decode_ms += dt(10, 11);
  } else if (has(2)) {
    stats.decode_ms += dt(prev, 2);
    prev = 2;
  }
  if (has(3)) {
    stats.scan_ms += dt(prev, 3);
    prev = 3;
  }
  stats.emit_ms += dt(prev, 4);
  stats.copy_ms += dt(4, 5);
  stats.runfix_ms += dt(5, 6);
  stats.total_ms += dt(0, 6);
  stats.ticks += 1;
  stats.updates += t.n;
  stats.blob_bytes += t.blob_bytes;
  uint32_t err = *sl.h_err;
  /* drain-host: back a run with its span of the tick's pinned arena
   * (k_drain streamed the whole tick there; kv_off stays tick-relative,
   * pay_p = tick payload base). Falls back to the eager per-run fetch when
   * the arena is absent (no counts / allocation failure). */
  auto attach_drain = [&](Run &run, const DevRunDesc &rd) -> bool {
    if (!t.drain_buf || sl.h_place->overflow) return false;
    uint64_t hb16 =
        ((uint64_t)sl.h_place->total_rec * sizeof(wb::RecHdr) + 15) & ~15ull;
    run.hbuf = t.drain_buf;
    run.hdr_p = t.drain_buf.get() + (rd.hdr_off - sl.h_place->hdr_off);
    run.pay_p = t.drain_buf.get() + hb16;
    return true;
  };
  if (err != 0) {
    /* Precise corruption recovery: per-update validity came back with the
     * run descriptors (sl.h_ok). Per shard, keep the prefix of records from
     * updates BEFORE its first corrupt one (their seqs are final), drop the
     * rest (their host-assigned seqs assumed the corrupt batch applied),
     * roll the shard's seq back to the kept boundary and poison it — the
     * reference's failed-apply -> re-pull-from-LatestSequenceNumber cadence
     * (replicated_db.cpp:378-382). Groups are in tick order, so a later
     * group of an already-poisoned shard is dropped whole. */
    const bool have_counts = t.counts.size() == t.n;
    for (uint32_t g = 0; g < t.ngroups; g++) {
      DevRunDesc &rd = sl.h_rundescs[g];
      const GroupDesc &gd = sl.h_groups[g];
      ShardState &ss = shards[rd.shard];
      std::lock_guard<std::mutex> lk(ss.mu);
      if (rd.base_seq == 0) {
        /* place overflow (store full in non-ring mode): NOTHING of this tick
         * was applied — roll back to the durable boundary and poison; the
         * condition persists until the operator gives the store room, like
         * the reference's repeated DB::Write failures on a full disk
         * (base_seq==0 is the discriminator: real seqs start at 1). */
        ss.poisoned = true;
        ss.next_seq = ss.durable_seq + 1;
        ss.cnt_failures++;
        while (!ss.log.empty() && ss.log.back().base_seq > ss.durable_seq) {
          ss.log_used -= ss.log.back().rep.size();
          ss.log.pop_back();
        }
        continue;
      }
      if (ss.poisoned) {
        /* shard already failed (earlier this tick or a previous tick):
         * everything staged after the failure is stale — drop the group
         * without touching durable_seq (rd.base_seq-1 may sit past a seq
         * hole), keep next_seq at the durable boundary, prune the log */
        ss.next_seq = ss.durable_seq + 1;
        while (!ss.log.empty() && ss.log.back().base_seq > ss.durable_seq) {
          ss.log_used -= ss.log.back().rep.size();
          ss.log.pop_back();
        }
        continue;
      }
      uint32_t keep_recs = 0;
      uint64_t keep_seq = rd.base_seq - 1;
      bool bad = false;
      for (uint32_t i = 0; i < gd.n_upds; i++) {
        if (!sl.h_ok[gd.first + i]) {
          bad = true;
          break;
        }
        uint32_t c = have_counts ? t.counts[gd.first + i] : 0;
        keep_recs += c;
        keep_seq += c;
      }
      if (!bad) { /* whole group clean */
        auto run = std::make_shared<Run>();
        run->base_seq = rd.base_seq;
        run->last_seq = rd.last_seq;
        run->n_entries = rd.n_entries;
        run->payload_bytes = rd.payload_bytes;
        run->hdr_cur = rd.hdr_off;
        run->payload_cur = rd.payload_off;
        run->pay_rel_base = rd.pay_rel_base;
        if (opts.drain_host && !attach_drain(*run, rd))
          (void)fetch_run_impl(this, *run);
        if (rd.n_entries > 0 && !opts.store_ring) ss.runs.push_back(std::move(run));
        if (rd.last_seq > ss.durable_seq) ss.durable_seq = rd.last_seq;
        stats.records += rd.n_entries;
        stats.payload_bytes += rd.payload_bytes;
        continue;
      }
      if (keep_recs > 0 && have_counts && !opts.store_ring) {
        /* truncated run: the leading clean records are contiguous */
        auto run = std::make_shared<Run>();
        run->base_seq = rd.base_seq;
        run->last_seq = keep_seq;
        run->n_entries = keep_recs;
        run->payload_bytes = rd.payload_bytes; /* upper bound; fetch uses hdrs */
        run->hdr_cur = rd.hdr_off;
        run->payload_cur = rd.payload_off;
        run->pay_rel_base = rd.pay_rel_base;
        if (opts.drain_host && !attach_drain(*run, rd))
          (void)fetch_run_impl(this, *run);
        ss.runs.push_back(std::move(run));
        stats.records += keep_recs;
      }
      if (keep_seq > ss.durable_seq) ss.durable_seq = keep_seq;
      ss.poisoned = true;
      ss.next_seq = ss.durable_seq + 1;
      ss.cnt_failures++; /* ≅ kReplicatorHandleResponseFailure */
      /* drop retained-log entries past the rolled-back boundary: a batch
       * that failed validation must never be re-served downstream (the
       * reference's WAL only ever contains accepted batches) */
      while (!ss.log.empty() && ss.log.back().base_seq > ss.durable_seq) {
        ss.log_used -= ss.log.back().rep.size();
        ss.log.pop_back();
      }
    }
    for (int i = 0; i < kEventsPerTick; i++) put_event(t.ev[i]);
    sl.busy = false;
    return GRA_OK;
  }
  /* Stale-group guard for the err==0 paths: a tick staged BEFORE a corrupt
   * batch's failure was detected carries base_seqs that assumed the corrupt
   * batch applied. Such ticks always ingest while the shard is still
   * poisoned (gra_handle_replicate_response's failure report drains pending
   * ticks before clearing the flag), so drop the group, keep next_seq at
   * the durable boundary and prune retained-log entries past it — the same
   * handling as the error path's 'already failed earlier' case. Without
   * this, durable_seq re-advances past the rolled-back boundary and the
   * failed batch is silently skipped on re-pull (follower divergence). */
  auto drop_stale_locked = [&](ShardState &ss) {
    ss.next_seq = ss.durable_seq + 1;
    while (!ss.log.empty() && ss.log.back().base_seq > ss.durable_seq) {
      ss.log_used -= ss.log.back().rep.size();
      ss.log.pop_back();
    }
  };
  if (opts.store_ring) {
    /* throughput store: runs are recycled by the ring; keep only seq/stats
     * bookkeeping (Get over recycled regions is undefined by contract) */
    for (uint32_t g = 0; g < t.ngroups; g++) {
      DevRunDesc &rd = sl.h_rundescs[g];
      ShardState &ss = shards[rd.shard];
      std::lock_guard<std::mutex> lk(ss.mu);
      if (ss.poisoned) {
        drop_stale_locked(ss);
        continue;
      }
      if (opts.drain_host && rd.n_entries > 0) {
        /* ring + drain = the production drain shape: the device store is a
         * staging ring, the host arena keeps the durable run (host-only:
         * device bytes recycle, so reads route through the host path) */
        auto run = std::make_shared<Run>();
        run->base_seq = rd.base_seq;
        run->last_seq = rd.last_seq;
        run->n_entries = rd.n_entries;
        run->payload_bytes = rd.payload_bytes;
        run->pay_rel_base = rd.pay_rel_base;
        if (attach_drain(*run, rd)) ss.runs.push_back(std::move(run));
      }
      if (rd.last_seq > ss.durable_seq) ss.durable_seq = rd.last_seq;
      stats.records += rd.n_entries;
      stats.payload_bytes += rd.payload_bytes;
    }
    for (int i = 0; i < kEventsPerTick; i++) put_event(t.ev[i]);
    sl.busy = false;
    return GRA_OK;
  }
  for (uint32_t g = 0; g < t.ngroups; g++) {
    DevRunDesc &rd = sl.h_rundescs[g];
    ShardState &ss = shards[rd.shard];
    std::lock_guard<std::mutex> lk(ss.mu);
    if (ss.poisoned) {
      drop_stale_locked(ss);
      continue;
    }
    auto run = std::make_shared<Run>();
    run->base_seq = rd.base_seq;
    run->last_seq = rd.last_seq;
    run->n_entries = rd.n_entries;
    run->payload_bytes = rd.payload_bytes;
    run->hdr_cur = rd.hdr_off;      /* absolute offsets into d_store */
    run->payload_cur = rd.payload_off;
    run->pay_rel_base = rd.pay_rel_base;
    if (rd.n_entries > 0) {
      if (opts.drain_host && !attach_drain(*run, rd))
        (void)fetch_run_impl(this, *run); /* eager D2H fallback */
      ss.runs.push_back(std::move(run));
    }
    ss.durable_seq = rd.last_seq > ss.durable_seq ? rd.last_seq : ss.durable_seq;
    stats.records += rd.n_entries;
    stats.payload_bytes += rd.payload_bytes;
  }
  for (int i = 0; i < kEventsPerTick; i++) put_event(t.ev[i]);
  sl.busy = false;
  return GRA_OK;
}

int GraEngine::ingest(bool wait_all) {
  while (!pending.empty()) {
    int rc = ingest_one(pending.front(), wait_all);
    if (rc == GRA_NOT_FOUND) return GRA_OK; /* front not done; stop */
    if (rc != GRA_OK) return rc;
    pending.pop_front();
  }
  return GRA_OK;
}

int GraEngine::stream_tick_locked() {
  StageBuf *old = cur_stage.load(std::memory_order_acquire);
  if (old->nslots.load(std::memory_order_relaxed) == 0) return GRA_OK;
  StageBuf *next = old == &stage[0] ? &stage[1] : &stage[0];
  /* the next buffer must have drained its previous H2D before writers
   * reuse it */
  HIP_TRY(hipEventSynchronize(next->free_ev));
  next->pos.store(0, std::memory_order_relaxed);
  next->nslots.store(0, std::memory_order_relaxed);
  next->staged.store(0, std::memory_order_relaxed);
  next->epoch++; /* invalidates every thread-local chunk of the previous
                    generation (sequenced before the release publish below;
                    without this, stale chunks write into recycled slots —
                    lost updates under flush churn) */
  next->closed.store(false, std::memory_order_release);
  /* swap: new writers land in `next`; then quiesce `old`.
   * The close/quiesce pair is a Dekker handshake with the writers'
   * (writers++ ; closed.load) pair: with a plain release store the
   * builder's closed=true can reorder after its writers read (x86
   * StoreLoad), letting quiesce pass while a writer that never saw
   * `closed` stages one more update — observed as rare single-update
   * loss under flush churn. seq_cst on both sides closes it. */
  old->closed.store(true, std::memory_order_seq_cst);
  cur_stage.store(next, std::memory_order_release);
  for (int s = 0; s < kWriterSlots; s++)
    while (old->writers[s].n.load(std::memory_order_seq_cst) != 0)
      std::this_thread::yield();
  uint32_t nall = old->nslots.load(std::memory_order_relaxed);
  if (nall > max_upd) nall = max_upd;
  uint64_t fill = old->pos.load(std::memory_order_relaxed);
  if (fill > opts.staging_bytes) fill = opts.staging_bytes;
  /* counting-sort by shard, filtering abandoned (len==0) slots */
  std::vector<uint32_t> cnt(opts.nshards + 1, 0);
  for (uint32_t i = 0; i < nall; i++)
    if (old->descs[i].len) cnt[old->descs[i].shard + 1]++;
  for (uint32_t s = 0; s < opts.nshards; s++) cnt[s + 1] += cnt[s];
  uint32_t n = cnt[opts.nshards];
  if (gra_check_staging()) {
    /* invariant: every successful HandleReplicateResponse of this buffer
     * generation must surface exactly once in the tick build */
    uint32_t expect = old->staged.load(std::memory_order_relaxed);
    if (n != expect)
      fprintf(stderr,
              "[gra] BUG: tick build found %u staged updates, expected %u\n",
              n, expect);
  }
  if (n == 0) return GRA_OK;
  std::vector<UpdDesc> ud(n);
  std::vector<GroupDesc> groups;
  uint64_t blob_bytes = 0;
  {
    std::vector<uint32_t> pos = cnt;
    for (uint32_t i = 0; i < nall; i++) {
      const GraUpdateDesc &d = old->descs[i];
      if (!d.len) continue;
      uint32_t j = pos[d.shard]++;
      ud[j].off = d.off;
      ud[j].len = d.len;
      ud[j].shard = d.shard;
      ud[j].base_seq = (uint64_t)d.ts; /* base_seq stashed at submit */
      blob_bytes += d.len;
    }
    for (uint32_t s = 0; s < opts.nshards; s++)
      if (cnt[s + 1] > cnt[s])
        groups.push_back({s, cnt[s], cnt[s + 1] - cnt[s], 0});
  }
  /* per-update counts from consecutive base seqs are unavailable here;
   * recover them from the staged batch headers (4B read per update) */
  std::vector<uint32_t> counts(n);
  for (uint32_t i = 0; i < n; i++)
    counts[i] = wb::fixed32_le(old->pin + ud[i].off + 8);
  int rc = enqueue_tick(d_stage_blobs, d_stage_descs, n, groups, blob_bytes,
                        true, old->pin, fill, d_stage_blobs, ud.data(), nullptr,
                        nullptr, nullptr, std::move(counts));
  if (rc != GRA_OK) return rc;
  /* mark the old pinned buffer reusable once its H2D completed (the copy
   * runs on the dedicated h2d stream now) */
  HIP_TRY(hipEventRecord(old->free_ev, h2d));
  return GRA_OK;
}

int GraEngine::flush_locked() {
  int rc = stream_tick_locked();
  if (rc != GRA_OK) return rc;
  HIP_TRY(hipStreamSynchronize(stream));
  return ingest(true);
}

/* ---------------- C ABI ---------------- */
extern "C" {

void gra_engine_opts_init(GraEngineOpts *o) {
  memset(o, 0, sizeof(*o));
  o->device = -1;
  o->nshards = 1;
}

int gra_engine_create(const GraEngineOpts *opts, GraEngine **out) {
  auto *e = new GraEngine();
  int rc = e->init(*opts);
  if (rc != GRA_OK) {
    delete e;
    return rc;
  }
  *out = e;
  return GRA_OK;
}

void gra_engine_destroy(GraEngine *e) { delete e; }

GraDb *gra_open(GraEngine *e, uint32_t shard) {
  if (shard >= e->opts.nshards) {
    g_err = "shard id out of range";
    return nullptr;
  }
  return new GraDb{e, shard};
}
void gra_close(GraDb *db) { delete db; }

/* Append a batch to the shard's retained update log (under ss.mu), evicting
 * oldest entries past the per-shard byte budget — the WAL-retention analog
 * (bounded like WAL_ttl_seconds bounds the reference's serving window). */
static void retain_locked(GraEngine *e, ShardState &ss, uint64_t base_seq,
                          uint32_t count, int64_t ts, const uint8_t *rep,
                          size_t len) {
  uint64_t per_shard = (e->opts.log_bytes ? e->opts.log_bytes : 256ULL << 20) /
                       (e->opts.nshards ? e->opts.nshards : 1);
  LogEnt ent;
  ent.base_seq = base_seq;
  ent.count = count;
  ent.ts = ts;
  ent.rep.assign(rep, rep + len);
  ss.log_used += len;
  ss.log.push_back(std::move(ent));
  while (ss.log_used > per_shard && ss.log.size() > 1) {
    ss.log_used -= ss.log.front().rep.size();
    ss.log.pop_front();
  }
}

int gra_handle_replicate_response(GraDb *db, const uint8_t *rep, size_t len,
                                  int64_t ts) {
  GraEngine *e = db->e;
  ShardState &ss = e->shards[db->shard];
  uint64_t base;
  uint32_t count;
  bool was_poisoned = false;
  {
    std::lock_guard<std::mutex> lk(ss.mu);
    if (ss.poisoned) {
      was_poisoned = true;
    } else {
      if (len < wb::kHeaderBytes) return 0;
      count = wb::fixed32_le(rep + 8);
      base = ss.next_seq;
      ss.next_seq += count; /* optimistic; rolled back on ANY failure below */
    }
  }
  if (was_poisoned) {
    /* Reference cadence: fail once, caller re-pulls from
     * LatestSequenceNumber (replicated_db.cpp:378-382). Before clearing the
     * flag, drain every pending tick: updates staged between the corrupt
     * batch and its detection carry stale base_seqs, and the ingest paths
     * drop their groups only while the shard is still poisoned. Error path
     * only — never taken on healthy shards. */
    {
      std::lock_guard<std::mutex> lk(e->mu);
      (void)e->flush_locked();
    }
    std::lock_guard<std::mutex> lk(ss.mu);
    ss.poisoned = false;
    return 0;
  }
  auto rollback = [&] {
    std::lock_guard<std::mutex> lk(ss.mu);
    /* per-shard calls are sequential (the seam contract), so the optimistic
     * advance is still the tail and can be undone */
    if (ss.next_seq == base + count) ss.next_seq = base;
  };
  if (len + 16 > e->opts.staging_bytes) {
    /* a single Update larger than the staging buffer cannot be staged;
     * refuse it (the reference's responses are bounded by max_updates x
     * batch size — this is a misconfiguration, not a data error) */
    rollback();
    return 0;
  }
  /* lock-free staging with per-thread CHUNK reservation: a thread grabs a
   * ~256 KiB byte range + a run of desc slots in one pair of atomic adds
   * and bump-allocates locally, so the shared cachelines are touched once
   * per ~chunk instead of once per update. Reserved-but-unused desc slots
   * are pre-zeroed (len = 0 -> filtered at tick build). */
  struct Chunk {
    GraEngine *e = nullptr;
    GraEngine::StageBuf *sb = nullptr;
    uint64_t epoch = 0;
    uint64_t base = 0, used = 0, cap = 0;
    uint32_t slot = 0, slots_left = 0;
  };
  thread_local Chunk ck;
  constexpr uint64_t kChunkBytes = 256 << 10;
  /* striped writer slot: each thread RMWs its own cacheline (see
   * WriterSlot); >kWriterSlots threads share slots benignly */
  static std::atomic<uint32_t> g_wslot{0};
  thread_local const uint32_t wslot =
      g_wslot.fetch_add(1, std::memory_order_relaxed) %
      GraEngine::kWriterSlots;
  bool staged = false;
  for (int attempt = 0; attempt < 100000 && !staged; attempt++) {
    GraEngine::StageBuf *sb = e->cur_stage.load(std::memory_order_acquire);
    auto &w = sb->writers[wslot].n;
    w.fetch_add(1, std::memory_order_seq_cst);
    if (sb->closed.load(std::memory_order_seq_cst)) { /* Dekker pair with the
                                                         builder's close+quiesce */
      w.fetch_sub(1, std::memory_order_acq_rel);
      std::this_thread::yield();
      continue;
    }
    bool chunk_ok = ck.e == e && ck.sb == sb && ck.epoch == sb->epoch &&
                    ck.used + len <= ck.cap && ck.slots_left > 0;
    if (!chunk_ok) {
      uint64_t want = len > kChunkBytes ? (uint64_t)len : kChunkBytes;
      /* slots sized for THIS call's update size so neither resource
       * exhausts far ahead of the other (a fixed slot count abandoned
       * ~half the byte range at 1 KB updates, and the H2D copies the
       * whole [0, pos) range including dead space) */
      uint32_t cslots = (uint32_t)(want / (len ? len : 1)) + 8;
      if (cslots > 4096) cslots = 4096;
      uint64_t off = sb->pos.fetch_add(want, std::memory_order_relaxed);
      uint32_t slot = sb->nslots.fetch_add(cslots, std::memory_order_relaxed);
      if (off + want + 16 > e->opts.staging_bytes ||
          slot + cslots > e->max_upd) {
        /* buffer full: abandon the reserved slots (len = 0) and tick */
        for (uint32_t i = slot; i < slot + cslots && i < e->max_upd; i++)
          sb->descs[i].len = 0;
        ck.sb = nullptr;
        w.fetch_sub(1, std::memory_order_acq_rel);
        /* double-checked: when a buffer fills, every writer lands here —
         * only the first may take e->mu and build the tick (~15 ms for a
         * 1M-update buffer); the rest must NOT queue on the mutex behind
         * it (the swap publishes the next buffer before the build, so
         * they can continue writing immediately) */
        if (e->cur_stage.load(std::memory_order_acquire) == sb) {
          std::lock_guard<std::mutex> lk(e->mu);
          if (e->cur_stage.load(std::memory_order_acquire) == sb) {
            if (e->stream_tick_locked() != GRA_OK) {
              rollback();
              return 0;
            }
          }
        }
        continue;
      }
      for (uint32_t i = 0; i < cslots; i++) sb->descs[slot + i].len = 0;
      ck = {e, sb, sb->epoch, off, 0, want, slot, cslots};
    }
    uint64_t off = ck.base + ck.used;
    memcpy(sb->pin + off, rep, len);
    GraUpdateDesc d;
    d.shard = db->shard;
    d.len = (uint32_t)len;
    d.off = off;
    d.ts = (int64_t)base; /* staging path smuggles base_seq here */
    sb->descs[ck.slot] = d;
    ck.used += len;
    ck.slot++;
    ck.slots_left--;
    if (gra_check_staging()) /* invariant counter is a hot shared RMW:
                                debug builds/soaks only (GRA_CHECK_STAGING=1) */
      sb->staged.fetch_add(1, std::memory_order_relaxed);
    w.fetch_sub(1, std::memory_order_release);
    staged = true;
  }
  if (!staged) {
    g_err = "staging contention: could not reserve a slot";
    rollback();
    return 0;
  }
  /* success: bookkeeping + retention (AFTER staging so a failure above
   * never leaves a retained-log entry for an un-applied batch); counters
   * are relaxed atomics — no lock on the hot path */
  if (e->opts.retain_log) {
    std::lock_guard<std::mutex> lk(ss.mu);
    retain_locked(e, ss, base, count, ts, rep, len);
  }
  ss.cnt_updates.fetch_add(1, std::memory_order_relaxed);
  ss.cnt_in_bytes.fetch_add(len, std::memory_order_relaxed);
  /* ≅ kReplicatorInBytes (replicated_db.cpp:409) */
  if (ts != 0) { /* ≅ kReplicatorLatency (replicated_db.cpp:370-374) */
    int64_t now_ms = (int64_t)(std::chrono::duration_cast<std::chrono::milliseconds>(
        std::chrono::system_clock::now().time_since_epoch()).count());
    ss.lat_sum_ms.fetch_add((uint64_t)(now_ms > ts ? now_ms - ts : 0),
                            std::memory_order_relaxed);
    ss.lat_n.fetch_add(1, std::memory_order_relaxed);
  }
  return 1;
}

uint64_t gra_latest_seq(GraDb *db) {
  ShardState &ss = db->e->shards[db->shard];
  std::lock_guard<std::mutex> lk(ss.mu);
  return ss.poisoned ? ss.durable_seq : ss.next_seq - 1;
}

int gra_write_leader(GraDb *db, const uint8_t *rep, size_t len,
                     uint64_t *seq_out) {
  GraEngine *e = db->e;
  ShardState &ss = e->shards[db->shard];
  std::lock_guard<std::mutex> lk(ss.mu);
  auto run = std::make_shared<Run>();
  uint64_t base = ss.next_seq;
  if (!host_build_run(rep, len, base, run.get())) {
    g_err = "corrupt WriteBatch rep";
    return GRA_CORRUPT;
  }
  uint32_t count = wb::fixed32_le(rep + 8);
  ss.next_seq += count;
  ss.durable_seq = ss.next_seq - 1;
  if (run->n_entries > 0) ss.runs.push_back(std::move(run));
  if (e->opts.retain_log) {
    /* the leader stamps writes with now-ms (replicated_db.cpp:115-117's
     * PutLogData breadcrumb); serving returns it as Update.timestamp */
    int64_t now_ms = (int64_t)(std::chrono::duration_cast<std::chrono::milliseconds>(
        std::chrono::system_clock::now().time_since_epoch()).count());
    retain_locked(e, ss, base, count, now_ms, rep, len);
  }
  if (seq_out) *seq_out = ss.durable_seq;
  return GRA_OK;
}

int gra_get_updates(GraDb *db, uint64_t since_seq, uint32_t max_updates,
                    GraServedUpdate *out, uint32_t *n_out, uint8_t *buf,
                    size_t cap, int requester_role) {
  GraEngine *e = db->e;
  if (!e->opts.retain_log) {
    g_err = "retain_log disabled on this engine";
    return GRA_ERR;
  }
  ShardState &ss = e->shards[db->shard];
  std::lock_guard<std::mutex> lk(ss.mu);
  /* the pull request's seq_no IS the follower's confirmed progress
   * (mode-2 ack); an OBSERVER's progress is ignored
   * (replicated_db.cpp:452-456) */
  if (requester_role == 0 && since_seq > ss.acked_confirmed)
    ss.acked_confirmed = since_seq;
  /* NOTE: the C-ABI contract is that `out` holds max_updates entries, so
   * max_updates==0 returns nothing HERE; the wire layers translate the
   * IDL's 0-means-unlimited (replicator.thrift:36-38) into a bounded
   * allocation before calling down (ffi.Db.get_updates). */
  *n_out = 0;
  if (!ss.log.empty() && since_seq + 1 < ss.log.front().base_seq) {
    /* reference analog: WAL no longer reaches back that far */
    g_err = "retained log truncated before requested seq";
    return GRA_ERR;
  }
  size_t off = 0;
  for (const LogEnt &ent : ss.log) {
    if (*n_out >= max_updates) break;
    if (ent.base_seq <= since_seq) continue;
    if (off + ent.rep.size() > cap) break;
    memcpy(buf + off, ent.rep.data(), ent.rep.size());
    out[*n_out].seq = ent.base_seq;
    out[*n_out].ts = ent.ts;
    out[*n_out].off = (uint32_t)off;
    out[*n_out].len = (uint32_t)ent.rep.size();
    off += ent.rep.size();
    (*n_out)++;
    ss.cnt_served++;
    ss.cnt_out_bytes += ent.rep.size();
  }
  if (*n_out > 0 && requester_role == 0) {
    /* mode-1 ack: acked once sent (replicated_db.cpp:543-546) */
    uint64_t last = out[*n_out - 1].seq;
    const LogEnt *le = nullptr;
    for (const LogEnt &ent : ss.log)
      if (ent.base_seq == last) le = &ent;
    uint64_t sent = le ? le->base_seq + le->count - 1 : last;
    if (sent > ss.acked_sent) ss.acked_sent = sent;
  }
  ss.ack_cv.notify_all();
  return GRA_OK;
}

int gra_db_counters(GraDb *db, GraDbCounters *out) {
  ShardState &ss = db->e->shards[db->shard];
  std::lock_guard<std::mutex> lk(ss.mu);
  out->updates_applied = ss.cnt_updates.load(std::memory_order_relaxed);
  out->in_bytes = ss.cnt_in_bytes.load(std::memory_order_relaxed);
  out->apply_failures = ss.cnt_failures.load(std::memory_order_relaxed);
  out->updates_served = ss.cnt_served.load(std::memory_order_relaxed);
  out->out_bytes = ss.cnt_out_bytes.load(std::memory_order_relaxed);
  out->latency_ms_sum = ss.lat_sum_ms.load(std::memory_order_relaxed);
  out->latency_samples = ss.lat_n.load(std::memory_order_relaxed);
  out->latest_seq = ss.poisoned ? ss.durable_seq : ss.next_seq - 1;
  return GRA_OK;
}

/* ≅ MaxNumberBox::wait (max_number_box.cpp:63) behind ReplicatedDB::Write's
 * 2-ACK modes (replicated_db.cpp:147-156): block until the downstream ack
 * (confirmed: the follower applied it; else: it was served) reaches seq, or
 * timeout. Returns GRA_OK / GRA_NOT_FOUND on timeout. */
int gra_wait_ack(GraDb *db, uint64_t seq, int confirmed, int timeout_ms) {
  ShardState &ss = db->e->shards[db->shard];
  std::unique_lock<std::mutex> lk(ss.mu);
  auto reached = [&] {
    return (confirmed ? ss.acked_confirmed : ss.acked_sent) >= seq;
  };
  if (ss.ack_cv.wait_for(lk, std::chrono::milliseconds(timeout_ms), reached))
    return GRA_OK;
  return GRA_NOT_FOUND;
}

int gra_flush(GraEngine *e) {
  std::lock_guard<std::mutex> lk(e->mu);
  return e->flush_locked();
}

/* Pre-pin `count` drain arenas of `bytes` each into the pool (harness
 * setup: pinning ~1 GB costs ~100 ms — keep it out of timed regions). */
int gra_drain_prewarm(GraEngine *e, size_t bytes, uint32_t count) {
  std::vector<std::shared_ptr<uint8_t>> held;
  held.reserve(count);
  for (uint32_t i = 0; i < count; i++) {
    auto p = e->drain_alloc(bytes);
    if (!p) {
      g_err = "gra_drain_prewarm: pinned allocation failed";
      return GRA_ERR;
    }
    held.push_back(std::move(p));
  }
  return GRA_OK; /* releasing `held` returns every arena to the pool */
}

static int fetch_run_impl(GraEngine *e, Run &r) {
  if (r.resident()) return GRA_OK;
  r.hdrs.resize((size_t)r.n_entries * sizeof(wb::RecHdr));
  r.payload.resize(r.payload_bytes);
  HIP_TRY(hipMemcpy(r.hdrs.data(), e->d_store + r.hdr_cur, r.hdrs.size(),
                    hipMemcpyDeviceToHost));
  HIP_TRY(hipMemcpy(r.payload.data(), e->d_store + r.payload_cur,
                    r.payload.size(), hipMemcpyDeviceToHost));
  /* kv_off values are tick-relative; rebase to run-relative */
  wb::RecHdr *h = (wb::RecHdr *)r.hdrs.data();
  for (uint32_t i = 0; i < r.n_entries; i++) h[i].kv_off -= r.pay_rel_base;
  return GRA_OK;
}

int gra_get(GraDb *db, const void *key, size_t klen, void *buf, size_t cap,
            size_t *vlen) {
  GraEngine *e = db->e;
  ShardState &ss = e->shards[db->shard];
  std::vector<std::shared_ptr<Run>> runs;
  {
    std::lock_guard<std::mutex> lk(ss.mu);
    runs = ss.runs;
    for (auto &r : runs) { /* lazy fetch under the shard lock */
      int rc = fetch_run_impl(e, *r);
      if (rc != GRA_OK) return rc;
    }
  }
  std::string out;
  int rc = run_get(runs, key, klen, e->opts.merge_op, &out);
  if (rc == 1) return GRA_NOT_FOUND;
  if (out.size() > cap) return GRA_BUF_TOO_SMALL;
  memcpy(buf, out.data(), out.size());
  if (vlen) *vlen = out.size();
  return GRA_OK;
}

int gra_multiget(GraDb *db, uint32_t nq, const GraKeyRef *keys,
                 const uint8_t *keybuf, size_t keybuf_len, uint8_t *valbuf,
                 uint32_t val_stride, GraGetResult *out) {
  GraEngine *e = db->e;
  if (nq == 0) return GRA_OK;
  ShardState &ss = e->shards[db->shard];
  /* snapshot the run list: device-resident runs feed the kernel; host-only
   * runs (leader writes, ring+drain arenas) are probed on the host and the
   * two halves merge by seq — a mixed shard no longer forces the whole
   * call to the host path */
  std::vector<RunView> views;
  std::vector<std::shared_ptr<Run>> host_runs, unbuilt;
  {
    std::lock_guard<std::mutex> lk(ss.mu);
    views.reserve(ss.runs.size());
    for (auto it = ss.runs.rbegin(); it != ss.runs.rend(); ++it) {
      const Run &r = **it;
      if (r.n_entries == 0) continue;
      if (r.hdr_cur == UINT64_MAX) {
        host_runs.push_back(*it); /* oldest..newest order irrelevant: probe
                                     tracks max seqs */
      } else {
        views.push_back({r.hdr_cur, r.payload_cur, r.n_entries, r.pay_rel_base});
        if (!r.kpref_built) unbuilt.push_back(*it); /* racy pre-filter:
                                     re-checked under e->mu (rebuilds are
                                     idempotent anyway) */
      }
    }
  }
  if (views.empty()) {
    if (host_runs.empty()) {
      for (uint32_t q = 0; q < nq; q++) {
        out[q].status = GRA_GET_MISS;
        out[q].vlen = 0;
      }
    } else { /* purely host-resident shard: answer via the host path */
      for (uint32_t q = 0; q < nq; q++) out[q].status = GRA_GET_NEEDS_HOST;
    }
    return GRA_OK;
  }
  /* grow-only device staging cached on the engine (serialized with other
   * engine ops by mu; a serving deployment would shard this per stream) */
  std::lock_guard<std::mutex> lk_engine(e->mu);
  auto grow = [](auto **p, size_t *cap, size_t need) {
    if (*cap >= need) return true;
    if (*p) (void)hipFree(*p);
    *p = nullptr;
    *cap = 0;
    size_t want = need + need / 2 + 64;
    if (hipMalloc(p, want) != hipSuccess) return false;
    *cap = want;
    return true;
  };
  size_t vb = (size_t)nq * val_stride;
  auto &mg = e->mg;
  if (!grow(&mg.d_runs, &mg.runs_cap, views.size() * sizeof(RunView)) ||
      !grow(&mg.d_keys, &mg.keys_cap, nq * sizeof(GraKeyRef)) ||
      !grow(&mg.d_keybuf, &mg.keybuf_cap, keybuf_len + 16) ||
      !grow(&mg.d_valbuf, &mg.valbuf_cap, vb + 16) ||
      !grow(&mg.d_out, &mg.out_cap, nq * sizeof(GraGetResult))) {
    g_err = "gra_multiget: allocation failed";
    return GRA_ERR;
  }
  RunView *d_runs = mg.d_runs;
  GraKeyRef *d_keys = mg.d_keys;
  uint8_t *d_keybuf = mg.d_keybuf, *d_valbuf = mg.d_valbuf;
  GraGetResult *d_out = mg.d_out;
  const bool mixed = !host_runs.empty();
  std::vector<MgExtra> h_extra;
  /* hash-join path setup: query fingerprint table + candidate buffers */
  uint32_t qtab_size = 64;
  while (qtab_size < 2 * nq) qtab_size <<= 1;
  uint32_t cand_cap = 4 * nq + 1024, tomb_cap = 4096;
  static const bool force_scan = [] { /* debug: bypass the hash join */
    const char *v = getenv("GRA_MG_FORCE_SCAN");
    return v && v[0] == '1';
  }();
  bool hashjoin = views.size() <= 65535 && !force_scan;
  if ((mixed && !grow(&mg.d_extra, &mg.extra_cap, nq * sizeof(MgExtra))) ||
      (hashjoin &&
       (!grow(&mg.d_qtab, &mg.qtab_cap, qtab_size * 8) ||
        !grow(&mg.d_cands, &mg.cands_cap, (size_t)cand_cap * sizeof(MgCand)) ||
        !grow(&mg.d_tombs, &mg.tombs_cap, (size_t)tomb_cap * sizeof(MgTomb)) ||
        !grow(&mg.d_aux, &mg.aux_cap, (size_t)nq * 8 * 4)))) {
    g_err = "gra_multiget: allocation failed";
    return GRA_ERR;
  }
  if (!mg.d_counts) {
    if (hipMalloc(&mg.d_counts, 8) != hipSuccess) {
      g_err = "gra_multiget: allocation failed";
      return GRA_ERR;
    }
    if (hipHostMalloc(&mg.h_counts, 8) != hipSuccess) {
      (void)hipFree(mg.d_counts);
      mg.d_counts = nullptr;
      g_err = "gra_multiget: allocation failed";
      return GRA_ERR;
    }
  }
  /* lazy read-index build for first-served runs (single pass per run,
   * ordered before this call's lookup kernels on the same stream;
   * concurrent calls serialize on e->mu and re-check the flag) */
  std::vector<RunView> bviews;
  for (auto &rp : unbuilt) {
    if (!rp->kpref_built) {
      bviews.push_back(
          {rp->hdr_cur, rp->payload_cur, rp->n_entries, rp->pay_rel_base});
      rp->kpref_built = true;
    }
  }
  if (!bviews.empty()) {
    size_t need = (views.size() > bviews.size() ? views.size()
                                                : bviews.size()) *
                  sizeof(RunView);
    if (!grow(&mg.d_runs, &mg.runs_cap, need) ||
        hipMemcpyAsync(mg.d_runs, bviews.data(),
                       bviews.size() * sizeof(RunView),
                       hipMemcpyHostToDevice, e->stream) != hipSuccess) {
      g_err = "gra_multiget: index build failed";
      return GRA_ERR;
    }
    uint64_t maxe = 0;
    for (auto &v : bviews)
      maxe = v.n_entries > maxe ? v.n_entries : maxe;
    uint32_t bx = (uint32_t)((maxe + 255) / 256);
    if (bx > 256) bx = 256;
    hipLaunchKernelGGL(k_kpref, dim3(bx, (uint32_t)bviews.size()), dim3(256),
                       0, e->stream, e->d_store, mg.d_runs);
    if (hipGetLastError() != hipSuccess) {
      g_err = "gra_multiget: index build launch failed";
      return GRA_ERR;
    }
    d_runs = mg.d_runs; /* grow may have reallocated */
  }
  std::vector<uint64_t> qtab;
  if (hashjoin) { /* host-built open-addressed fingerprint table */
    qtab.assign(qtab_size, 0);
    uint32_t mask = qtab_size - 1;
    for (uint32_t q = 0; q < nq; q++) {
      uint32_t fp =
          wb::key_fnv_fold(wb::kFnvBasis32, keybuf + keys[q].off, keys[q].len);
      uint32_t s = fp & mask;
      while (qtab[s] != 0) s = (s + 1) & mask;
      qtab[s] = ((uint64_t)fp << 32) | (q + 1);
    }
  }
  int rc = GRA_ERR;
  for (int attempt = 0; attempt < 2 && rc != GRA_OK; attempt++) {
    /* attempt 0: hash-join, fully async, ONE sync at the end. attempt 1
     * (only when the candidate/tombstone caps overflowed): the per-query
     * scan kernel. */
    bool use_hj = hashjoin && attempt == 0;
    do {
      if (hipMemcpyAsync(d_runs, views.data(),
                         views.size() * sizeof(RunView),
                         hipMemcpyHostToDevice, e->stream) != hipSuccess ||
          hipMemcpyAsync(d_keys, keys, nq * sizeof(GraKeyRef),
                         hipMemcpyHostToDevice, e->stream) != hipSuccess ||
          hipMemcpyAsync(d_keybuf, keybuf, keybuf_len,
                         hipMemcpyHostToDevice, e->stream) != hipSuccess)
        break;
      if (use_hj) {
        uint32_t nruns32 = (uint32_t)views.size();
        if (hipMemcpyAsync(mg.d_qtab, qtab.data(), qtab_size * 8,
                           hipMemcpyHostToDevice, e->stream) != hipSuccess ||
            hipMemsetAsync(mg.d_counts, 0, 8, e->stream) != hipSuccess ||
            hipMemsetAsync(mg.d_aux, 0, (size_t)nq * 8 * 4, e->stream) !=
                hipSuccess)
          break;
        uint64_t max_entries = 0;
        for (auto &v : views)
          max_entries = v.n_entries > max_entries ? v.n_entries : max_entries;
        uint32_t bx = (uint32_t)((max_entries + 255) / 256);
        if (bx > 1024) bx = 1024;
        if (bx == 0) bx = 1;
        hipLaunchKernelGGL(k_mg_scan, dim3(bx, nruns32), dim3(256), 0,
                           e->stream, e->d_store, d_runs,
                           (const uint64_t *)mg.d_qtab, qtab_size - 1, d_keys,
                           d_keybuf, (MgCand *)mg.d_cands, mg.d_counts,
                           cand_cap, (MgTomb *)mg.d_tombs, mg.d_counts + 1,
                           tomb_cap);
        if (hipGetLastError() != hipSuccess) break;
        unsigned long long *aux = (unsigned long long *)mg.d_aux;
        unsigned long long *term_pack = aux, *mergeq = aux + nq,
                           *rdq = aux + 2 * nq, *winner = aux + 3 * nq;
        uint32_t rb = (cand_cap + 255) / 256;
        hipLaunchKernelGGL(k_mg_resolve1, dim3(rb), dim3(256), 0, e->stream,
                           (MgCand *)mg.d_cands, mg.d_counts, cand_cap,
                           term_pack, mergeq);
        hipLaunchKernelGGL(k_mg_tombs, dim3((nq + 255) / 256), dim3(256), 0,
                           e->stream, e->d_store, d_runs,
                           (MgTomb *)mg.d_tombs, mg.d_counts + 1, tomb_cap,
                           d_keys, d_keybuf, nq, rdq);
        hipLaunchKernelGGL(k_mg_resolve2, dim3(rb), dim3(256), 0, e->stream,
                           (MgCand *)mg.d_cands, mg.d_counts, cand_cap,
                           term_pack, winner);
        hipLaunchKernelGGL(k_mg_emit, dim3(nq), dim3(64), 0, e->stream,
                           e->d_store, d_runs, term_pack, mergeq, rdq, winner,
                           nq, d_valbuf, val_stride, d_out,
                           mixed ? (MgExtra *)mg.d_extra : nullptr);
        if (hipGetLastError() != hipSuccess) break;
        if (hipMemcpyAsync(mg.h_counts, mg.d_counts, 8,
                           hipMemcpyDeviceToHost, e->stream) != hipSuccess)
          break;
      } else {
        static const int use_kpref = [] { /* debug: bypass the filter */
          const char *v = getenv("GRA_MG_NO_KPREF");
          return (v && v[0] == '1') ? 0 : 1;
        }();
        hipLaunchKernelGGL(k_multiget, dim3(nq), dim3(256), 0, e->stream,
                           e->d_store, d_runs, (uint32_t)views.size(), d_keys,
                           d_keybuf, nq, d_valbuf, val_stride, d_out,
                           mixed ? (MgExtra *)mg.d_extra : nullptr, use_kpref);
        if (hipGetLastError() != hipSuccess) break;
      }
      if (hipMemcpyAsync(out, d_out, nq * sizeof(GraGetResult),
                         hipMemcpyDeviceToHost, e->stream) != hipSuccess ||
          hipMemcpyAsync(valbuf, d_valbuf, vb, hipMemcpyDeviceToHost,
                         e->stream) != hipSuccess)
        break;
      if (mixed) {
        h_extra.resize(nq);
        if (hipMemcpyAsync(h_extra.data(), mg.d_extra, nq * sizeof(MgExtra),
                           hipMemcpyDeviceToHost, e->stream) != hipSuccess)
          break;
      }
      if (hipStreamSynchronize(e->stream) != hipSuccess) break;
      if (use_hj &&
          (mg.h_counts[0] > cand_cap || mg.h_counts[1] > tomb_cap))
        break; /* caps overflowed: results invalid, retry via scan kernel */
      rc = GRA_OK;
    } while (0);
    if (rc != GRA_OK && use_hj) (void)hipStreamSynchronize(e->stream);
  }
  if (rc != GRA_OK) {
    g_err = "gra_multiget: device op failed";
    return rc;
  }
  if (mixed) {
    /* merge the device verdict with a host-run probe, by seq — exactly the
     * k_multiget decision re-run over the union */
    for (uint32_t q = 0; q < nq; q++) {
      if (out[q].status == GRA_GET_NEEDS_HOST) continue; /* full fold */
      ProbeResult hp;
      run_probe(host_runs, keybuf + keys[q].off, keys[q].len, &hp);
      const MgExtra &de = h_extra[q];
      uint64_t T = de.term_seq, M = de.merge_seq, RD = de.rd_seq;
      bool host_wins = hp.term_seq > T;
      if (host_wins) T = hp.term_seq;
      if (hp.merge_seq > M) M = hp.merge_seq;
      if (hp.rd_seq > RD) RD = hp.rd_seq;
      uint64_t fl = T > RD ? T : RD;
      if (M > fl) {
        out[q].status = GRA_GET_NEEDS_HOST; /* cross-half fold */
        out[q].vlen = 0;
        continue;
      }
      if (T == 0 || T <= RD) {
        out[q].status = GRA_GET_MISS;
        out[q].vlen = 0;
        continue;
      }
      if (host_wins) {
        if (hp.term_type != wb::kValue) {
          out[q].status = GRA_GET_MISS;
          out[q].vlen = 0;
        } else {
          uint32_t v = hp.vlen < val_stride ? hp.vlen : val_stride;
          memcpy(valbuf + (size_t)q * val_stride, hp.val, v);
          out[q].status = GRA_GET_FOUND;
          out[q].vlen = hp.vlen;
        }
      } else if (de.term_type != wb::kValue || T == 0) {
        out[q].status = GRA_GET_MISS;
        out[q].vlen = 0;
      } /* else: the device verdict (+ value already in valbuf) stands */
    }
  }
  return rc;
}

int gra_shard_checksum(GraDb *db, uint64_t *out) {
  GraEngine *e = db->e;
  ShardState &ss = e->shards[db->shard];
  std::vector<RunView> views;
  uint64_t host_sum = 0;
  {
    std::lock_guard<std::mutex> lk(ss.mu);
    for (const auto &rp : ss.runs) {
      const Run &r = *rp;
      if (r.n_entries == 0) continue;
      if (r.hdr_cur == UINT64_MAX) { /* host-origin/drained: fold on host */
        const wb::RecHdr *h = (const wb::RecHdr *)r.hdrs_data();
        const uint8_t *pay = r.payload_data();
        for (uint32_t i = 0; i < r.n_entries; i++)
          host_sum += rec_hash_cs(h[i].seq, h[i].type, h[i].key_len,
                                  h[i].val_len, pay + h[i].kv_off,
                                  pay + h[i].kv_off + h[i].key_len);
      } else {
        views.push_back({r.hdr_cur, r.payload_cur, r.n_entries, r.pay_rel_base});
      }
    }
  }
  if (views.empty()) {
    *out = host_sum;
    return GRA_OK;
  }
  std::lock_guard<std::mutex> lk(e->mu);
  RunView *d_runs = nullptr;
  unsigned long long *d_sum = nullptr;
  int rc = GRA_ERR;
  do {
    if (hipMalloc(&d_runs, views.size() * sizeof(RunView)) != hipSuccess ||
        hipMalloc(&d_sum, 8) != hipSuccess)
      break;
    if (hipMemcpy(d_runs, views.data(), views.size() * sizeof(RunView),
                  hipMemcpyHostToDevice) != hipSuccess ||
        hipMemset(d_sum, 0, 8) != hipSuccess)
      break;
    uint32_t nb = views.size() < 1024 ? (uint32_t)views.size() : 1024;
    hipLaunchKernelGGL(k_checksum, dim3(nb), dim3(256), 0, e->stream, e->d_store,
                       d_runs, (uint32_t)views.size(), d_sum);
    if (hipGetLastError() != hipSuccess) break;
    unsigned long long sum = 0;
    if (hipStreamSynchronize(e->stream) != hipSuccess ||
        hipMemcpy(&sum, d_sum, 8, hipMemcpyDeviceToHost) != hipSuccess)
      break;
    *out = host_sum + (uint64_t)sum;
    rc = GRA_OK;
  } while (0);
  if (rc != GRA_OK) g_err = "gra_shard_checksum: device op failed";
  if (d_runs) (void)hipFree(d_runs);
  if (d_sum) (void)hipFree(d_sum);
  return rc;
}

int gra_pin_alloc(GraEngine *e, size_t bytes, uint8_t **ptr) {
  (void)e;
  HIP_TRY(hipHostMalloc(ptr, bytes + 16));
  return GRA_OK;
}
void gra_pin_free(GraEngine *e, uint8_t *ptr) {
  (void)e;
  (void)hipHostFree(ptr);
}

extern "C" void gra_replay_destroy(GraReplay *r);

/* host_counts: batch record counts read from a HOST copy of the headers
 * (needed because with an external device arena the host cannot read the
 * blob bytes); pass nullptr when `arena` is host-readable. */
static int upload_common(GraEngine *e, GraReplay *r, const uint8_t *arena,
                         size_t arena_bytes, const GraUpdateDesc *descs,
                         uint64_t n, const uint32_t *host_counts) {
  if (hipMalloc(&r->d_descs, (size_t)n * sizeof(UpdDesc)) != hipSuccess) {
    g_err = "gra_upload: desc allocation failed";
    return GRA_ERR;
  }
  r->descs.resize(n);
  r->counts.resize(n);
  /* validate everything BEFORE touching shard seq state: a failed upload
   * must leave next_seq untouched for the caller to retry */
  for (uint64_t i = 0; i < n; i++) {
    const GraUpdateDesc &d = descs[i];
    if (d.shard >= e->opts.nshards || d.len > arena_bytes ||
        d.off > arena_bytes - d.len || d.len < wb::kHeaderBytes) {
      g_err = "gra_upload: bad desc";
      return GRA_ERR;
    }
  }
  for (uint64_t i = 0; i < n; i++) {
    const GraUpdateDesc &d = descs[i];
    uint32_t count = host_counts ? host_counts[i]
                                 : wb::fixed32_le(arena + d.off + 8);
    ShardState &ss = e->shards[d.shard];
    UpdDesc u;
    u.off = d.off;
    u.len = d.len;
    u.shard = d.shard;
    u.base_seq = ss.next_seq;
    ss.next_seq += count;
    r->descs[i] = u;
    r->counts[i] = (uint16_t)count;
  }
  if (hipMemcpy(r->d_descs, r->descs.data(), (size_t)n * sizeof(UpdDesc),
                hipMemcpyHostToDevice) != hipSuccess) {
    g_err = "gra_upload: desc H2D failed";
    return GRA_ERR;
  }
  return GRA_OK;
}

int gra_upload(GraEngine *e, const uint8_t *arena, size_t arena_bytes,
               const GraUpdateDesc *descs, uint64_t n, GraReplay **out) {
  auto *r = new GraReplay();
  r->e = e;
  r->arena_bytes = arena_bytes;
  r->h_arena = arena;
  if (hipMalloc(&r->d_blobs, arena_bytes + 16) != hipSuccess ||
      hipMemcpy(r->d_blobs, arena, arena_bytes, hipMemcpyHostToDevice) !=
          hipSuccess) {
    g_err = "gra_upload: blob allocation/H2D failed";
    gra_replay_destroy(r);
    return GRA_ERR;
  }
  int rc = upload_common(e, r, arena, arena_bytes, descs, n, nullptr);
  if (rc != GRA_OK) {
    gra_replay_destroy(r);
    return rc;
  }
  *out = r;
  return GRA_OK;
}

/* Device-resident arena (e.g. an RCCL all-to-all output tensor): the engine
 * reads blobs in place, zero-copy. The caller owns dev_arena (must outlive
 * the replay and include >=16 B of readable slack past arena_bytes) and
 * passes batch record counts explicitly. */
int gra_upload_dev(GraEngine *e, void *dev_arena, size_t arena_bytes,
                   const GraUpdateDesc *descs, uint64_t n,
                   const uint32_t *counts, GraReplay **out) {
  auto *r = new GraReplay();
  r->e = e;
  r->arena_bytes = arena_bytes;
  r->d_blobs = (uint8_t *)dev_arena;
  r->external_blobs = true;
  int rc = upload_common(e, r, nullptr, arena_bytes, descs, n, counts);
  if (rc != GRA_OK) {
    gra_replay_destroy(r);
    return rc;
  }
  *out = r;
  return GRA_OK;
}

/* Config #5 upload: Update payloads are Snappy-compressed in transit.
 * comp descs give (off,len) into the compressed arena; ulen[i] is each
 * update's uncompressed size (transport metadata); counts[i] the batch
 * record count (headers unreadable on host while compressed). The engine
 * decompresses per tick on-GPU (k_snappy) into a scratch arena that then
 * feeds the normal decode pipeline. */
int gra_upload_snappy(GraEngine *e, const uint8_t *comp_arena,
                      size_t comp_bytes, const GraUpdateDesc *descs,
                      uint64_t n, const uint32_t *ulens,
                      const uint32_t *counts, GraReplay **out) {
  auto *r = new GraReplay();
  r->e = e;
  r->snappy = true;
  /* scratch layout: per-update slot, 16-B aligned, +8 B chunk-write slack */
  std::vector<UpdDesc> tmp; /* not used; offsets computed inline */
  std::vector<GraUpdateDesc> udescs(n);
  std::vector<SnapTask> tasks(n);
  uint64_t scratch = 0;
  for (uint64_t i = 0; i < n; i++) {
    if (descs[i].len > comp_bytes || descs[i].off > comp_bytes - descs[i].len) {
      g_err = "gra_upload_snappy: comp desc out of range";
      gra_replay_destroy(r);
      return GRA_ERR;
    }
    tasks[i] = {descs[i].off, scratch, descs[i].len, ulens[i]};
    udescs[i].shard = descs[i].shard;
    udescs[i].len = ulens[i];
    udescs[i].off = scratch;
    udescs[i].ts = descs[i].ts;
    scratch += ((uint64_t)ulens[i] + 16 + 15) & ~15ULL; /* 16B chunk slack */
  }
  r->arena_bytes = scratch;
  if (hipMalloc(&r->d_blobs, scratch + 16) != hipSuccess ||
      hipMalloc(&r->d_comp, comp_bytes + 16) != hipSuccess ||
      hipMalloc(&r->d_snaptasks, n * sizeof(SnapTask)) != hipSuccess) {
    g_err = "gra_upload_snappy: allocation failed";
    gra_replay_destroy(r);
    return GRA_ERR;
  }
  if (hipMemcpy(r->d_comp, comp_arena, comp_bytes, hipMemcpyHostToDevice) !=
          hipSuccess ||
      hipMemcpy(r->d_snaptasks, tasks.data(), n * sizeof(SnapTask),
                hipMemcpyHostToDevice) != hipSuccess) {
    g_err = "gra_upload_snappy: H2D failed";
    gra_replay_destroy(r);
    return GRA_ERR;
  }
  int rc = upload_common(e, r, nullptr, scratch, udescs.data(), n, counts);
  if (rc != GRA_OK) {
    gra_replay_destroy(r);
    return rc;
  }
  r->snap_tasks = std::move(tasks);
  *out = r;
  return GRA_OK;
}

void gra_replay_destroy(GraReplay *r) {
  if (!r) return;
  if (r->d_blobs && !r->external_blobs) (void)hipFree(r->d_blobs);
  if (r->d_comp) (void)hipFree(r->d_comp);
  if (r->d_snaptasks) (void)hipFree(r->d_snaptasks);
  if (r->d_descs) (void)hipFree(r->d_descs);
  for (auto &kv : r->plans) {
    if (kv.second.d_groups) (void)hipFree(kv.second.d_groups);
    if (kv.second.d_snap) (void)hipFree(kv.second.d_snap);
    if (kv.second.d_ud) (void)hipFree(kv.second.d_ud);
    if (kv.second.consumed_ev) (void)hipEventDestroy(kv.second.consumed_ev);
  }
  delete r;
}

static TickPlan &plan_for(GraReplay *r, uint64_t first, uint64_t n) {
  auto key = std::make_pair(first, n);
  auto it = r->plans.find(key);
  if (it != r->plans.end()) return it->second;
  TickPlan plan;
  uint64_t bb = 0;
  uint32_t cur_shard = UINT32_MAX;
  /* each contiguous same-shard range becomes one run; a shard may appear in
   * several ranges per tick (runs are ingested in tick order, so per-shard
   * seq order is preserved) */
  for (uint64_t i = 0; i < n; i++) {
    const UpdDesc &d = r->descs[first + i];
    bb += d.len;
    if (d.shard != cur_shard) {
      plan.groups.push_back({d.shard, (uint32_t)i, 1, 0});
      cur_shard = d.shard;
    } else {
      plan.groups.back().n_upds++;
    }
  }
  plan.blob_bytes = bb;
  if (hipMalloc(&plan.d_groups, plan.groups.size() * sizeof(GroupDesc)) ==
          hipSuccess &&
      hipMemcpy(plan.d_groups, plan.groups.data(),
                plan.groups.size() * sizeof(GroupDesc),
                hipMemcpyHostToDevice) == hipSuccess) {
    /* cached on device: per-tick group upload is skipped */
  } else {
    plan.d_groups = nullptr; /* fall back to per-tick upload */
  }
  static const bool sort_snap = [] { /* A/B escape hatch */
    const char *v = getenv("GRA_SNAPPY_SORT");
    return !v || v[0] != '0';
  }();
  if (r->snappy && sort_snap && first + n <= r->snap_tasks.size()) {
    /* length-sorted launch order for k_snappy (see TickPlan::d_snap) */
    std::vector<SnapTask> sorted(r->snap_tasks.begin() + first,
                                 r->snap_tasks.begin() + first + n);
    std::stable_sort(sorted.begin(), sorted.end(),
                     [](const SnapTask &a, const SnapTask &b) {
                       return a.comp_len < b.comp_len;
                     });
    if (hipMalloc(&plan.d_snap, n * sizeof(SnapTask)) == hipSuccess &&
        hipMemcpy(plan.d_snap, sorted.data(), n * sizeof(SnapTask),
                  hipMemcpyHostToDevice) == hipSuccess) {
      /* cached */
    } else {
      if (plan.d_snap) (void)hipFree(plan.d_snap);
      plan.d_snap = nullptr; /* fall back to the unsorted full array */
    }
  }
  return r->plans.emplace(key, std::move(plan)).first->second;
}

/* Pre-build (and device-cache) the tick plan for a replay window so the
 * first replayed tick of that window pays no hipMalloc/H2D inside a timed
 * region. Harness setup call; gra_replay_tick works without it. */
int gra_replay_prepare(GraReplay *r, uint64_t first, uint64_t n) {
  if (first + n > r->descs.size()) {
    g_err = "replay window out of range";
    return GRA_ERR;
  }
  std::lock_guard<std::mutex> lk(r->e->mu);
  (void)plan_for(r, first, n);
  return GRA_OK;
}

int gra_replay_tick(GraReplay *r, uint64_t first, uint64_t n) {
  GraEngine *e = r->e;
  if (first + n > r->descs.size()) {
    g_err = "replay window out of range";
    return GRA_ERR;
  }
  TickPlan &plan = plan_for(r, first, n);
  std::vector<uint32_t> counts(r->counts.begin() + first,
                               r->counts.begin() + first + n);
  std::lock_guard<std::mutex> lk(e->mu);
  if (r->snappy && !plan.consumed_ev)
    (void)hipEventCreateWithFlags(&plan.consumed_ev, hipEventDisableTiming);
  return e->enqueue_tick(r->d_blobs, r->d_descs + first, (uint32_t)n,
                         plan.groups, plan.blob_bytes, false, nullptr, 0,
                         nullptr, nullptr, plan.d_groups, r->d_comp,
                         !r->snappy ? nullptr
                         : plan.d_snap ? plan.d_snap
                                       : r->d_snaptasks + first,
                         std::move(counts), plan.consumed_ev);
}

int gra_replay_tick_h2d(GraReplay *r, uint64_t first, uint64_t n) {
  GraEngine *e = r->e;
  if (first + n > r->descs.size()) {
    g_err = "replay window out of range";
    return GRA_ERR;
  }
  if (r->snappy || r->h_arena == nullptr) {
    g_err = "tick_h2d needs a host-resident uncompressed arena "
            "(use gra_replay_tick for snappy/device uploads)";
    return GRA_ERR;
  }
  /* window blobs must be contiguous in the arena (generator layout) */
  uint64_t lo = r->descs[first].off;
  uint64_t hi = r->descs[first + n - 1].off + r->descs[first + n - 1].len;
  if (hi - lo > e->opts.staging_bytes) {
    g_err = "h2d window exceeds staging";
    return GRA_ERR;
  }
  TickPlan &plan = plan_for(r, first, n);
  std::lock_guard<std::mutex> lk(e->mu);
  if (!plan.d_ud) {
    /* rebased descs (blob offsets relative to the staged window), uploaded
     * once per window and reused every step */
    std::vector<UpdDesc> ud(n);
    for (uint64_t i = 0; i < n; i++) {
      ud[i] = r->descs[first + i];
      ud[i].off -= lo;
    }
    if (hipMalloc(&plan.d_ud, n * sizeof(UpdDesc)) != hipSuccess ||
        hipMemcpy(plan.d_ud, ud.data(), n * sizeof(UpdDesc),
                  hipMemcpyHostToDevice) != hipSuccess) {
      if (plan.d_ud) (void)hipFree(plan.d_ud);
      plan.d_ud = nullptr;
      g_err = "tick_h2d: desc cache allocation failed";
      return GRA_ERR;
    }
  }
  std::vector<uint32_t> counts(r->counts.begin() + first,
                               r->counts.begin() + first + n);
  return e->enqueue_tick(e->d_stage_blobs, plan.d_ud, (uint32_t)n,
                         plan.groups, plan.blob_bytes, true, r->h_arena + lo,
                         hi - lo, e->d_stage_blobs, nullptr, plan.d_groups,
                         nullptr, nullptr, std::move(counts));
}

int gra_replay_sync(GraReplay *r) { return gra_flush(r->e); }

void gra_stats(GraEngine *e, GraStats *out) {
  std::lock_guard<std::mutex> lk(e->mu);
  (void)e->ingest(false);
  GraStats s;
  s.h2d_ms = e->stats.h2d_ms;
  s.snappy_ms = e->stats.snappy_ms;
  s.decode_ms = e->stats.decode_ms;
  s.scan_ms = e->stats.scan_ms;
  s.emit_ms = e->stats.emit_ms;
  s.copy_ms = e->stats.copy_ms;
  s.runfix_ms = e->stats.runfix_ms;
  s.total_ms = e->stats.total_ms;
  s.ticks = e->stats.ticks;
  s.updates = e->stats.updates;
  s.records = e->stats.records;
  s.blob_bytes = e->stats.blob_bytes;
  s.payload_bytes = e->stats.payload_bytes;
  *out = s;
}
void gra_stats_reset(GraEngine *e) {
  std::lock_guard<std::mutex> lk(e->mu);
  e->stats = Stats();
}

} /* extern "C" */
