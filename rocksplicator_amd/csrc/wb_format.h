/* wb_format.h — WriteBatch rep byte layout, shared by host C++ and HIP device
 * code. Restates the layout of the reference's un-vendored dependency rocksdb
 * 5.7.fb (docker/Dockerfile:259-284): db/write_batch.cc / db/dbformat.h —
 * 12-byte header (fixed64 LE seq + fixed32 LE count), then records of
 * tag(1B) + varint32-length-prefixed slices. Decode contract pinned by
 * rocksdb_replicator/tests/rocksdb_assumption_test.cpp (seq accounting) and
 * tests/golden/writebatch_vectors.json in this repo.
 */
#pragma once
#include <stdint.h>

#ifdef __HIP__ /* compiling in HIP mode (-x hip) */
#define WB_HD __host__ __device__ inline
#define WB_HD_OP __host__ __device__
#else
#define WB_HD static inline
#define WB_HD_OP
#endif

namespace wb {

enum Tag : uint8_t {
  kDeletion = 0x00,
  kValue = 0x01,
  kMerge = 0x02,
  kLogData = 0x03,
  kCfDeletion = 0x04,
  kCfValue = 0x05,
  kCfMerge = 0x06,
  kSingleDeletion = 0x07,
  kCfSingleDeletion = 0x08,
  kBeginPrepare = 0x09,
  kEndPrepare = 0x0A,
  kCommit = 0x0B,
  kRollback = 0x0C,
  kNoop = 0x0D,
  kCfRangeDeletion = 0x0E,
  kRangeDeletion = 0x0F,
  kMaxTag = 0x0F,
};

constexpr uint32_t kHeaderBytes = 12;

WB_HD uint64_t fixed64_le(const uint8_t *p) {
#if defined(__HIP_DEVICE_COMPILE__) && defined(WB_UNALIGNED_OK)
  /* gfx950 global loads tolerate misaligned multi-dword accesses (verified
   * by scripts/micro_copy.hip v1); blob arenas are over-allocated by 16 B */
  return *(const uint64_t *)p;
#else
  uint64_t v = 0;
  for (int i = 0; i < 8; i++) v |= (uint64_t)p[i] << (8 * i);
  return v;
#endif
}
WB_HD uint32_t fixed32_le(const uint8_t *p) {
#if defined(__HIP_DEVICE_COMPILE__) && defined(WB_UNALIGNED_OK)
  return *(const uint32_t *)p;
#else
  return (uint32_t)p[0] | ((uint32_t)p[1] << 8) | ((uint32_t)p[2] << 16) |
         ((uint32_t)p[3] << 24);
#endif
}

/* varint32 decode; returns bytes consumed (1..5), 0 on error/overrun.
 * p must have at least `avail` readable bytes. */
WB_HD uint32_t varint32(const uint8_t *p, uint32_t avail, uint32_t *out) {
  uint32_t result = 0;
#pragma unroll
  for (uint32_t i = 0; i < 5; i++) {
    if (i >= avail) return 0;
    uint32_t byte = p[i];
    if (byte & 0x80) {
      result |= (byte & 0x7F) << (7 * i);
    } else {
      if (i == 4 && byte > 0x0F) return 0; /* >32 bits */
      result |= byte << (7 * i);
      *out = result;
      return i + 1;
    }
  }
  return 0;
}

/* 32-bit FNV-1a fingerprint of the STORED key bytes ([cf LE4 | key] for cf
 * records) — the kpref filter value. A raw prefix was useless for real key
 * families (shared "user..."-style prefixes reject nothing); a hash keeps
 * ~2^-32 false-accept regardless of key structure. Seeded fold helper so
 * the cf prefix can be folded first. */
constexpr uint32_t kFnvBasis32 = 2166136261u;
WB_HD uint32_t key_fnv_fold(uint32_t h, const uint8_t *p, uint32_t n) {
  for (uint32_t i = 0; i < n; i++) h = (h ^ p[i]) * 16777619u;
  return h;
}

/* Does this record tag consume a sequence number?
 * Put/Delete/SingleDelete/Merge/RangeDeletion (+CF variants) do; LogData,
 * Noop and 2PC markers do not (assumption test :136-187). */
WB_HD bool consumes_seq(uint8_t tag) {
  switch (tag) {
    case kValue: case kDeletion: case kMerge: case kSingleDeletion:
    case kRangeDeletion: case kCfValue: case kCfDeletion: case kCfMerge:
    case kCfSingleDeletion: case kCfRangeDeletion:
      return true;
    default:
      return false;
  }
}
WB_HD bool has_cf_prefix(uint8_t tag) {
  return tag == kCfValue || tag == kCfDeletion || tag == kCfMerge ||
         tag == kCfSingleDeletion || tag == kCfRangeDeletion;
}
WB_HD uint8_t base_tag(uint8_t tag) {
  switch (tag) {
    case kCfValue: return kValue;
    case kCfDeletion: return kDeletion;
    case kCfMerge: return kMerge;
    case kCfSingleDeletion: return kSingleDeletion;
    case kCfRangeDeletion: return kRangeDeletion;
    default: return tag;
  }
}
/* slice shape per base tag: 0 = none, 1 = key only, 2 = key + value */
WB_HD int nslices(uint8_t btag) {
  switch (btag) {
    case kValue: case kMerge: case kRangeDeletion: return 2;
    case kDeletion: case kSingleDeletion: return 1;
    case kLogData: return 2 - 1; /* one slice (blob) */
    case kEndPrepare: case kCommit: case kRollback: return 1; /* xid */
    case kBeginPrepare: case kNoop: return 0;
    default: return -1; /* unknown tag */
  }
}

/* One decoded record (device pipeline form; offsets into the blob). */
struct Rec {
  uint32_t key_off, key_len;
  uint32_t val_off, val_len;
  uint8_t tag;       /* original tag */
  uint8_t consumes;  /* 1 if seq-consuming */
  uint16_t _pad;
  uint32_t cf_id;
};

/* Walk result for one blob. */
struct WalkTotals {
  uint32_t n_records;     /* seq-consuming, memtable-relevant records
                             (excludes LogData/markers) */
  uint32_t payload_bytes; /* sum over emitted records of klen(+4 if cf)+vlen */
  uint32_t payload16;     /* same, each record rounded up to 16 B (store form) */
  uint32_t hdr_count;     /* header count field */
  uint32_t ok;            /* 1 iff well-formed and consumed == hdr_count */
  uint32_t _pad;
  uint64_t hdr_seq;       /* header seq field */
};

/* Sequential walk over one rep blob, calling f(rec, index) for every
 * seq-consuming record in order. Corruption (truncated slice, bad varint,
 * unknown tag, count mismatch — write_batch.cc Iterate contract) => ok = 0
 * (f may have been called for a prefix; callers gate on ok). */
template <class F>
WB_HD WalkTotals walk_f(const uint8_t *rep, uint32_t len, F &&f) {
  WalkTotals t = {0, 0, 0, 0, 0, 0, 0};
  if (len < kHeaderBytes) return t;
  t.hdr_seq = fixed64_le(rep);
  t.hdr_count = fixed32_le(rep + 8);
  uint32_t pos = kHeaderBytes;
  uint32_t consumed = 0, nrec = 0, payload = 0, payload16 = 0;
  while (pos < len) {
    uint8_t tag = rep[pos++];
    uint8_t bt = base_tag(tag);
    int ns = nslices(bt);
    if (ns < 0) return t; /* unknown tag */
    uint32_t cf = 0;
    if (has_cf_prefix(tag)) {
      uint32_t c = varint32(rep + pos, len - pos, &cf);
      if (c == 0) return t;
      pos += c;
    }
    uint32_t off[2] = {0, 0}, slen[2] = {0, 0};
    for (int s = 0; s < ns; s++) {
      uint32_t n, c = varint32(rep + pos, len - pos, &n);
      if (c == 0) return t;
      pos += c;
      if (pos + n > len || n > len) return t;
      off[s] = pos;
      slen[s] = n;
      pos += n;
    }
    if (consumes_seq(tag)) {
      Rec r;
      r.key_off = off[0];
      r.key_len = slen[0];
      r.val_off = off[1];
      r.val_len = slen[1];
      r.tag = tag;
      r.consumes = 1;
      r._pad = 0;
      r.cf_id = cf;
      f(r, nrec);
      /* payload stored per record: [cf_id?4B] key [cf_id?4B-for-range-end]
       * value, 16-B aligned. CF range tombstones prefix BOTH slices (begin
       * AND end key live in the cf-namespaced key space) so covering
       * comparisons against cf-prefixed query keys stay consistent. */
      uint32_t cf4 = cf ? 4u : 0u;
      uint32_t cfx = (cf && bt == kRangeDeletion) ? 8u : cf4;
      payload += slen[0] + slen[1] + cfx;
      payload16 += (slen[0] + slen[1] + cfx + 15u) & ~15u;
      nrec++;
      consumed++;
    }
  }
  if (consumed != t.hdr_count) return t; /* "WriteBatch has wrong count" */
  t.n_records = nrec;
  t.payload_bytes = payload;
  t.payload16 = payload16;
  t.ok = 1;
  return t;
}

struct NullEmit {
  WB_HD_OP void operator()(const Rec &, uint32_t) const {}
};

/* Array-emitting wrapper: emit == nullptr for totals only. */
WB_HD WalkTotals walk(const uint8_t *rep, uint32_t len, Rec *emit, uint32_t cap) {
  if (!emit) return walk_f(rep, len, NullEmit{});
  return walk_f(rep, len, [&](const Rec &r, uint32_t i) {
    if (i < cap) emit[i] = r;
  });
}

/* ---- device store record header (one per applied record, 24 B) ---- */
struct RecHdr {
  uint64_t seq;
  uint32_t kv_off;   /* offset of key bytes within the tick payload region;
                        value bytes follow the key immediately */
  uint32_t val_len;
  uint16_t key_len;  /* includes 4-byte cf prefix when cf_id != 0 */
  uint8_t type;      /* base tag */
  uint8_t flags;     /* bit0: cf-prefixed key */
  uint32_t kpref;    /* first min(4,key_len) key bytes, zero-padded — lets
                        point lookups filter without touching the payload */
};

WB_HD uint32_t key_prefix4(const uint8_t *key, uint32_t klen) {
  uint32_t p = 0;
  for (uint32_t i = 0; i < 4 && i < klen; i++) p |= (uint32_t)key[i] << (8 * i);
  return p;
}
static_assert(sizeof(RecHdr) == 24, "RecHdr must be 24 bytes");

} /* namespace wb */
