/* host_store.cpp — run-format Get + leader-side host apply.
 * Semantics cites: rocksdb_wrapper.cpp:5-8 (WriteToLeader = DB::Write),
 * rocksdb_assumption_test.cpp:136-187 (seq accounting). */
#include "host_store.h"

namespace gra {

namespace {

/* bytewise comparator (rocksdb default): begin <= key < end */
bool range_covers(const uint8_t *b, size_t bl, const uint8_t *e, size_t el,
                  const uint8_t *k, size_t kl) {
  int c1 = memcmp(b, k, bl < kl ? bl : kl);
  if (c1 > 0 || (c1 == 0 && bl > kl)) return false;
  int c2 = memcmp(k, e, kl < el ? kl : el);
  if (c2 > 0 || (c2 == 0 && kl >= el)) return false;
  return true;
}

} /* namespace */

int run_get(const std::vector<std::shared_ptr<Run>> &runs, const void *key_,
            size_t klen, int merge_op, std::string *out) {
  const uint8_t *key = (const uint8_t *)key_;
  /* operand collection: entry pointers into run payloads (runs stay alive
   * for the duration of this call — caller holds shared_ptrs) */
  std::vector<std::pair<const uint8_t *, uint32_t>> ops; /* newest..oldest */
  const uint8_t *base = nullptr;
  uint32_t base_len = 0;
  bool have_base = false, stopped = false;

  for (auto it = runs.rbegin(); it != runs.rend() && !stopped; ++it) {
    const Run &r = **it;
    if (r.n_entries == 0) continue;
    const wb::RecHdr *hdrs = (const wb::RecHdr *)r.hdrs_data();
    const uint8_t *pay = r.payload_data();
    for (int32_t i = (int32_t)r.n_entries - 1; i >= 0; i--) {
      const wb::RecHdr &h = hdrs[i];
      if (h.type == wb::kRangeDeletion) {
        /* covers key? begin = key slice, end = value slice */
        if (range_covers(pay + h.kv_off, h.key_len,
                         pay + h.kv_off + h.key_len, h.val_len, key, klen)) {
          stopped = true; /* everything below this seq is deleted */
          break;
        }
        continue;
      }
      if (h.key_len != klen || memcmp(pay + h.kv_off, key, klen) != 0) continue;
      if (h.type == wb::kMerge) {
        ops.emplace_back(pay + h.kv_off + h.key_len, h.val_len);
        continue;
      }
      if (h.type == wb::kValue) {
        base = pay + h.kv_off + h.key_len;
        base_len = h.val_len;
        have_base = true;
      }
      stopped = true; /* Value / Deletion / SingleDeletion end the walk */
      break;
    }
  }

  if (!have_base && ops.empty()) return 1;
  if (ops.empty()) {
    /* no operands above the base: the Put's value verbatim (rocksdb
     * FullMerge only runs when merge records are newer — mirrored in the
     * oracle and the device path) */
    out->assign((const char *)base, base_len);
    return 0;
  }
  if (merge_op == 1 /* u64add */) {
    uint64_t acc = 0;
    if (have_base) memcpy(&acc, base, base_len < 8 ? base_len : 8);
    for (auto it = ops.rbegin(); it != ops.rend(); ++it) {
      uint64_t v = 0;
      memcpy(&v, it->first, it->second < 8 ? it->second : 8);
      acc += v;
    }
    out->assign((const char *)&acc, 8);
  } else { /* concat with ',' oldest→newest */
    out->clear();
    bool first = true;
    if (have_base) {
      out->append((const char *)base, base_len);
      first = false;
    }
    for (auto it = ops.rbegin(); it != ops.rend(); ++it) {
      if (!first) out->push_back(',');
      first = false;
      out->append((const char *)it->first, it->second);
    }
  }
  return 0;
}

void run_probe(const std::vector<std::shared_ptr<Run>> &runs, const void *key_,
               size_t klen, ProbeResult *out) {
  const uint8_t *key = (const uint8_t *)key_;
  *out = ProbeResult();
  for (const auto &rp : runs) {
    const Run &r = *rp;
    if (r.n_entries == 0) continue;
    const wb::RecHdr *hdrs = (const wb::RecHdr *)r.hdrs_data();
    const uint8_t *pay = r.payload_data();
    for (uint32_t i = 0; i < r.n_entries; i++) {
      const wb::RecHdr &h = hdrs[i];
      if (h.type == wb::kRangeDeletion) {
        if (range_covers(pay + h.kv_off, h.key_len,
                         pay + h.kv_off + h.key_len, h.val_len, key, klen) &&
            h.seq > out->rd_seq)
          out->rd_seq = h.seq;
        continue;
      }
      if (h.key_len != klen || memcmp(pay + h.kv_off, key, klen) != 0)
        continue;
      if (h.type == wb::kMerge) {
        if (h.seq > out->merge_seq) out->merge_seq = h.seq;
      } else if (h.seq > out->term_seq) {
        out->term_seq = h.seq;
        out->term_type = h.type;
        out->val = pay + h.kv_off + h.key_len;
        out->vlen = h.val_len;
      }
    }
  }
}

bool host_build_run(const uint8_t *rep, size_t len, uint64_t base_seq, Run *out) {
  wb::WalkTotals tot = wb::walk(rep, (uint32_t)len, nullptr, 0);
  if (!tot.ok) return false;
  out->n_entries = tot.n_records;
  out->base_seq = base_seq;
  out->last_seq = base_seq + tot.hdr_count - 1;
  out->hdrs.resize((size_t)tot.n_records * sizeof(wb::RecHdr));
  /* same 16-byte record alignment the GPU emit kernel uses */
  size_t pay_need = 0;
  {
    uint32_t i = 0;
    std::vector<wb::Rec> recs(tot.n_records);
    wb::WalkTotals t2 = wb::walk(rep, (uint32_t)len, recs.data(), tot.n_records);
    (void)t2;
    /* CF range tombstones prefix BOTH slices (begin and end key live in the
     * cf-namespaced key space) — mirrors k_emit and the oracle */
    auto cfv_of = [](const wb::Rec &rc) {
      return (rc.cf_id && wb::base_tag(rc.tag) == wb::kRangeDeletion) ? 4u : 0u;
    };
    for (i = 0; i < tot.n_records; i++) {
      const wb::Rec &rc = recs[i];
      uint32_t cf4 = rc.cf_id ? 4u : 0u;
      pay_need += (cf4 + rc.key_len + cfv_of(rc) + rc.val_len + 15u) & ~15u;
    }
    out->payload.resize(pay_need);
    out->payload_bytes = (uint32_t)pay_need;
    wb::RecHdr *hd = (wb::RecHdr *)out->hdrs.data();
    uint32_t off = 0;
    for (i = 0; i < tot.n_records; i++) {
      const wb::Rec &rc = recs[i];
      uint32_t cf4 = rc.cf_id ? 4u : 0u;
      uint32_t cfv = cfv_of(rc);
      wb::RecHdr h;
      h.seq = base_seq + i;
      h.kv_off = off;
      h.val_len = rc.val_len + cfv;
      h.key_len = (uint16_t)(rc.key_len + cf4);
      h.type = wb::base_tag(rc.tag);
      h.flags = cf4 ? 1 : 0;
      hd[i] = h;
      uint8_t *p = out->payload.data() + off;
      if (cf4) memcpy(p, &rc.cf_id, 4);
      memcpy(p + cf4, rep + rc.key_off, rc.key_len);
      /* fingerprint over the STORED key bytes just written (host runs pay
       * it eagerly — CPU cost is trivial; device runs build theirs lazily
       * at first multiget via k_kpref) */
      hd[i].kpref = wb::key_fnv_fold(wb::kFnvBasis32, p, cf4 + rc.key_len);
      if (cfv) memcpy(p + cf4 + rc.key_len, &rc.cf_id, 4);
      memcpy(p + cf4 + rc.key_len + cfv, rep + rc.val_off, rc.val_len);
      off += (cf4 + rc.key_len + cfv + rc.val_len + 15u) & ~15u;
    }
  }
  return true;
}

} /* namespace gra */
