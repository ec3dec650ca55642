/* wb_oracle.c — CPU oracle for the slave-side WriteBatch apply path.
 * See wb_oracle.h header comment for parity-pinning status and reference cites.
 * TEST INFRASTRUCTURE ONLY — never the shipped or measured product path.
 */
#include "wb_oracle.h"
#include <pthread.h>
#include <stdlib.h>
#include <string.h>
#include <time.h>

/* ================= varint32 (LEB128) — rocksdb util/coding.cc ================= */
static size_t varint32_encode(uint8_t *dst, uint32_t v) {
  size_t n = 0;
  while (v >= 0x80) {
    dst[n++] = (uint8_t)(v | 0x80);
    v >>= 7;
  }
  dst[n++] = (uint8_t)v;
  return n;
}
/* returns bytes consumed, 0 on error */
static size_t varint32_decode(const uint8_t *p, const uint8_t *end, uint32_t *out) {
  uint32_t result = 0;
  for (uint32_t shift = 0; shift <= 28 && p + (shift / 7) < end; shift += 7) {
    uint32_t byte = p[shift / 7];
    if (byte & 0x80) {
      result |= (byte & 0x7F) << shift;
    } else {
      result |= byte << shift;
      *out = result;
      return shift / 7 + 1;
    }
  }
  return 0;
}

/* ================= batch builder ================= */
struct OrcBatch {
  uint8_t *buf;
  size_t len, cap;
  uint32_t count;
};

#define WB_HEADER 12

static void wb_reserve(OrcBatch *b, size_t extra) {
  if (b->len + extra > b->cap) {
    size_t ncap = b->cap * 2 > b->len + extra ? b->cap * 2 : b->len + extra + 64;
    b->buf = (uint8_t *)realloc(b->buf, ncap);
    b->cap = ncap;
  }
}
static void wb_append(OrcBatch *b, const void *p, size_t n) {
  wb_reserve(b, n);
  memcpy(b->buf + b->len, p, n);
  b->len += n;
}
static void wb_append_byte(OrcBatch *b, uint8_t v) { wb_append(b, &v, 1); }
static void wb_append_lps(OrcBatch *b, const void *p, size_t n) { /* length-prefixed slice */
  uint8_t tmp[5];
  size_t c = varint32_encode(tmp, (uint32_t)n);
  wb_append(b, tmp, c);
  wb_append(b, p, n);
}
static void wb_set_count(OrcBatch *b, uint32_t c) {
  b->count = c;
  b->buf[8] = (uint8_t)c;
  b->buf[9] = (uint8_t)(c >> 8);
  b->buf[10] = (uint8_t)(c >> 16);
  b->buf[11] = (uint8_t)(c >> 24);
}

OrcBatch *orc_wb_create(void) {
  OrcBatch *b = (OrcBatch *)calloc(1, sizeof(OrcBatch));
  b->cap = 256;
  b->buf = (uint8_t *)calloc(1, b->cap);
  b->len = WB_HEADER; /* zeroed seq + count */
  return b;
}
void orc_wb_destroy(OrcBatch *b) {
  if (b) {
    free(b->buf);
    free(b);
  }
}
void orc_wb_clear(OrcBatch *b) {
  memset(b->buf, 0, WB_HEADER);
  b->len = WB_HEADER;
  b->count = 0;
}
void orc_wb_put(OrcBatch *b, const void *k, size_t kl, const void *v, size_t vl) {
  wb_append_byte(b, ORC_TYPE_VALUE);
  wb_append_lps(b, k, kl);
  wb_append_lps(b, v, vl);
  wb_set_count(b, b->count + 1);
}
void orc_wb_delete(OrcBatch *b, const void *k, size_t kl) {
  wb_append_byte(b, ORC_TYPE_DELETION);
  wb_append_lps(b, k, kl);
  wb_set_count(b, b->count + 1);
}
void orc_wb_single_delete(OrcBatch *b, const void *k, size_t kl) {
  wb_append_byte(b, ORC_TYPE_SINGLE_DELETION);
  wb_append_lps(b, k, kl);
  wb_set_count(b, b->count + 1);
}
void orc_wb_merge(OrcBatch *b, const void *k, size_t kl, const void *v, size_t vl) {
  wb_append_byte(b, ORC_TYPE_MERGE);
  wb_append_lps(b, k, kl);
  wb_append_lps(b, v, vl);
  wb_set_count(b, b->count + 1);
}
void orc_wb_delete_range(OrcBatch *b, const void *bk, size_t bkl, const void *ek, size_t ekl) {
  wb_append_byte(b, ORC_TYPE_RANGE_DELETION);
  wb_append_lps(b, bk, bkl);
  wb_append_lps(b, ek, ekl);
  wb_set_count(b, b->count + 1);
}
void orc_wb_put_log_data(OrcBatch *b, const void *blob, size_t bl) {
  wb_append_byte(b, ORC_TYPE_LOGDATA);
  wb_append_lps(b, blob, bl); /* LogData consumes no seq: count unchanged */
}
void orc_wb_set_seq(OrcBatch *b, uint64_t seq) {
  for (int i = 0; i < 8; i++) b->buf[i] = (uint8_t)(seq >> (8 * i));
}
uint32_t orc_wb_count(const OrcBatch *b) { return b->count; }
const uint8_t *orc_wb_data(const OrcBatch *b, size_t *len) {
  if (len) *len = b->len;
  return b->buf;
}

/* ================= decode ================= */
static uint64_t fixed64_le(const uint8_t *p) {
  uint64_t v;
  memcpy(&v, p, 8); /* x86/host LE */
  return v;
}
static uint32_t fixed32_le(const uint8_t *p) {
  uint32_t v;
  memcpy(&v, p, 4);
  return v;
}

/* read a length-prefixed slice; returns 0 on corruption */
static int read_lps(const uint8_t *rep, size_t len, size_t *pos, uint32_t *off, uint32_t *slen) {
  uint32_t n;
  size_t c = varint32_decode(rep + *pos, rep + len, &n);
  if (c == 0) return 0;
  *pos += c;
  if (*pos + n > len) return 0;
  *off = (uint32_t)*pos;
  *slen = n;
  *pos += n;
  return 1;
}

int orc_decode(const uint8_t *rep, size_t len, OrcRecord *out, uint32_t cap,
               uint32_t *nrec, uint64_t *base_seq, uint32_t *count) {
  if (len < WB_HEADER) return 1;
  uint64_t seq = fixed64_le(rep);
  uint32_t cnt = fixed32_le(rep + 8);
  if (base_seq) *base_seq = seq;
  if (count) *count = cnt;
  size_t pos = WB_HEADER;
  uint32_t n = 0, consumed = 0;
  while (pos < len) {
    uint8_t tag = rep[pos++];
    OrcRecord r;
    memset(&r, 0, sizeof(r));
    r.type = tag;
    uint8_t base_tag = tag;
    /* CF-prefixed variants carry a varint32 column-family id first
     * (write_batch.cc ReadRecordFromWriteBatch, 5.7.fb). */
    if (tag == ORC_TYPE_CF_VALUE || tag == ORC_TYPE_CF_DELETION ||
        tag == ORC_TYPE_CF_MERGE || tag == ORC_TYPE_CF_SINGLE_DELETION ||
        tag == ORC_TYPE_CF_RANGE_DELETION) {
      uint32_t cf;
      size_t c = varint32_decode(rep + pos, rep + len, &cf);
      if (c == 0) return 2;
      pos += c;
      r.cf_id = cf;
      switch (tag) {
        case ORC_TYPE_CF_VALUE: base_tag = ORC_TYPE_VALUE; break;
        case ORC_TYPE_CF_DELETION: base_tag = ORC_TYPE_DELETION; break;
        case ORC_TYPE_CF_MERGE: base_tag = ORC_TYPE_MERGE; break;
        case ORC_TYPE_CF_SINGLE_DELETION: base_tag = ORC_TYPE_SINGLE_DELETION; break;
        case ORC_TYPE_CF_RANGE_DELETION: base_tag = ORC_TYPE_RANGE_DELETION; break;
      }
    }
    switch (base_tag) {
      case ORC_TYPE_VALUE:
      case ORC_TYPE_MERGE:
      case ORC_TYPE_RANGE_DELETION: /* key = begin, val = end */
        if (!read_lps(rep, len, &pos, &r.key_off, &r.key_len)) return 3;
        if (!read_lps(rep, len, &pos, &r.val_off, &r.val_len)) return 3;
        r.consumes_seq = 1;
        break;
      case ORC_TYPE_DELETION:
      case ORC_TYPE_SINGLE_DELETION:
        if (!read_lps(rep, len, &pos, &r.key_off, &r.key_len)) return 3;
        r.consumes_seq = 1;
        break;
      case ORC_TYPE_LOGDATA:
        if (!read_lps(rep, len, &pos, &r.val_off, &r.val_len)) return 3;
        break;
      case ORC_TYPE_BEGIN_PREPARE:
      case ORC_TYPE_NOOP:
        break; /* marker, no payload, no seq */
      case ORC_TYPE_END_PREPARE:
      case ORC_TYPE_COMMIT:
      case ORC_TYPE_ROLLBACK:
        if (!read_lps(rep, len, &pos, &r.key_off, &r.key_len)) return 3; /* xid */
        break;
      default:
        return 4; /* unknown tag */
    }
    if (r.consumes_seq) r.seq = seq + consumed++;
    if (out) {
      if (n >= cap) return 5;
      out[n] = r;
    }
    n++;
  }
  if (consumed != cnt) return 6; /* "WriteBatch has wrong count" */
  if (nrec) *nrec = n;
  return 0;
}

/* ================= memtable store ================= */
/* Per shard: open-addressing hash map key -> head index of a backward-linked
 * history chain in an append-only arena; plus a range-tombstone list.
 * Mirrors what rocksdb memtable insertion gives the replicator's parity
 * checks (per-key Get equality + latest seq) without LSM mechanics. */

typedef struct {
  uint64_t seq;
  uint32_t prev;  /* index+1 of previous entry for same key, 0 = none */
  uint32_t key_off, val_off;
  uint32_t key_len, val_len;
  uint8_t type; /* base tag */
} Entry;

typedef struct {
  uint64_t seq;
  uint32_t b_off, b_len, e_off, e_len;
} RangeTomb;

typedef struct {
  /* hash table: slot -> entry index+1 */
  uint32_t *slots;
  uint32_t nslots, nkeys;
  Entry *entries;
  uint32_t nentries, cap_entries;
  uint8_t *arena;
  size_t arena_len, arena_cap;
  RangeTomb *tombs;
  uint32_t ntombs, cap_tombs;
  uint64_t latest_seq;
} ShardTable;

struct OrcStore {
  uint32_t nshards;
  int merge_op;
  ShardTable *shards;
};

static uint64_t fnv1a(const uint8_t *p, size_t n) {
  uint64_t h = 1469598103934665603ULL;
  for (size_t i = 0; i < n; i++) h = (h ^ p[i]) * 1099511628211ULL;
  return h;
}

OrcStore *orc_store_create(uint32_t nshards, int merge_op) {
  OrcStore *s = (OrcStore *)calloc(1, sizeof(OrcStore));
  s->nshards = nshards;
  s->merge_op = merge_op;
  s->shards = (ShardTable *)calloc(nshards, sizeof(ShardTable));
  for (uint32_t i = 0; i < nshards; i++) {
    s->shards[i].nslots = 64;
    s->shards[i].slots = (uint32_t *)calloc(64, sizeof(uint32_t));
  }
  return s;
}
void orc_store_destroy(OrcStore *s) {
  if (!s) return;
  for (uint32_t i = 0; i < s->nshards; i++) {
    free(s->shards[i].slots);
    free(s->shards[i].entries);
    free(s->shards[i].arena);
    free(s->shards[i].tombs);
  }
  free(s->shards);
  free(s);
}

static uint32_t arena_add(ShardTable *t, const uint8_t *p, size_t n) {
  if (t->arena_len + n > t->arena_cap) {
    size_t nc = t->arena_cap * 2 > t->arena_len + n ? t->arena_cap * 2 : t->arena_len + n + 4096;
    t->arena = (uint8_t *)realloc(t->arena, nc);
    t->arena_cap = nc;
  }
  memcpy(t->arena + t->arena_len, p, n);
  uint32_t off = (uint32_t)t->arena_len;
  t->arena_len += n;
  return off;
}

static void table_grow(ShardTable *t) {
  uint32_t nn = t->nslots * 2;
  uint32_t *ns = (uint32_t *)calloc(nn, sizeof(uint32_t));
  for (uint32_t i = 0; i < t->nslots; i++) {
    uint32_t e = t->slots[i];
    if (!e) continue;
    Entry *en = &t->entries[e - 1];
    uint64_t h = fnv1a(t->arena + en->key_off, en->key_len);
    uint32_t j = (uint32_t)h & (nn - 1);
    while (ns[j]) j = (j + 1) & (nn - 1);
    ns[j] = e;
  }
  free(t->slots);
  t->slots = ns;
  t->nslots = nn;
}

/* find slot for key; returns slot index; *found = entry index+1 or 0 */
static uint32_t table_find(ShardTable *t, const uint8_t *key, size_t klen, uint32_t *found) {
  uint64_t h = fnv1a(key, klen);
  uint32_t j = (uint32_t)h & (t->nslots - 1);
  for (;;) {
    uint32_t e = t->slots[j];
    if (!e) {
      *found = 0;
      return j;
    }
    Entry *en = &t->entries[e - 1];
    if (en->key_len == klen && memcmp(t->arena + en->key_off, key, klen) == 0) {
      *found = e;
      return j;
    }
    j = (j + 1) & (t->nslots - 1);
  }
}

static void shard_insert(ShardTable *t, uint8_t type, uint64_t seq, const uint8_t *key,
                         size_t klen, const uint8_t *val, size_t vlen) {
  if (t->nentries == t->cap_entries) {
    t->cap_entries = t->cap_entries ? t->cap_entries * 2 : 64;
    t->entries = (Entry *)realloc(t->entries, t->cap_entries * sizeof(Entry));
  }
  if ((t->nkeys + 1) * 4 > t->nslots * 3) table_grow(t);
  uint32_t found, slot = table_find(t, key, klen, &found);
  Entry *e = &t->entries[t->nentries];
  e->type = type;
  e->seq = seq;
  e->prev = found; /* chain to previous entry for this key (0 if first) */
  e->key_len = (uint32_t)klen;
  e->val_len = (uint32_t)vlen;
  if (found) {
    e->key_off = t->entries[found - 1].key_off; /* reuse interned key bytes */
  } else {
    e->key_off = arena_add(t, key, klen);
    t->nkeys++;
  }
  e->val_off = vlen ? arena_add(t, val, vlen) : 0;
  t->nentries++;
  t->slots[slot] = t->nentries; /* head = newest */
}

int orc_apply(OrcStore *s, uint32_t shard, const uint8_t *rep, size_t len, int64_t ts) {
  (void)ts; /* the appended LogData(timestamp) record (rocksdb_wrapper.cpp:19-20)
             * consumes no seq and never reaches the memtable — WAL-only. */
  if (shard >= s->nshards) return 0;
  ShardTable *t = &s->shards[shard];
  /* two-pass: validate fully, then apply (a corrupt batch applies nothing —
   * DB::Write fails before memtable insert on iterate corruption).
   * Stack storage for typical batches: per-call malloc would handicap the
   * CPU-baseline leg (the reference's memtable insert allocates from an
   * arena, not malloc-per-op). */
  uint32_t nrec = 0, cnt = 0;
  uint64_t hdr_seq = 0;
  if (orc_decode(rep, len, NULL, 0, &nrec, &hdr_seq, &cnt) != 0) return 0;
  OrcRecord stack_recs[64];
  OrcRecord *recs = nrec <= 64 ? stack_recs
                               : (OrcRecord *)malloc(nrec * sizeof(OrcRecord));
  if (orc_decode(rep, len, recs, nrec, &nrec, &hdr_seq, &cnt) != 0) {
    if (recs != stack_recs) free(recs);
    return 0;
  }
  /* Follower assigns its own seqs: base = latest+1
   * (rocksdb_assumption_test.cpp:136-187; replay equality :329-359). */
  uint64_t base = t->latest_seq + 1;
  uint32_t consumed = 0;
  for (uint32_t i = 0; i < nrec; i++) {
    OrcRecord *r = &recs[i];
    if (!r->consumes_seq) continue;
    uint64_t seq = base + consumed++;
    uint8_t bt = r->type;
    if (bt >= ORC_TYPE_CF_DELETION && bt <= ORC_TYPE_CF_MERGE) bt -= 4; /* 4,5,6 -> 0,1,2 */
    else if (bt == ORC_TYPE_CF_SINGLE_DELETION) bt = ORC_TYPE_SINGLE_DELETION;
    else if (bt == ORC_TYPE_CF_RANGE_DELETION) bt = ORC_TYPE_RANGE_DELETION;
    if (bt == ORC_TYPE_RANGE_DELETION) {
      if (t->ntombs == t->cap_tombs) {
        t->cap_tombs = t->cap_tombs ? t->cap_tombs * 2 : 8;
        t->tombs = (RangeTomb *)realloc(t->tombs, t->cap_tombs * sizeof(RangeTomb));
      }
      RangeTomb *rt = &t->tombs[t->ntombs++];
      rt->seq = seq;
      if (r->cf_id != 0) {
        /* CF range tombstones are namespaced like CF point keys: prefix the
         * cf id to BOTH begin and end so covering comparisons against
         * cf-prefixed query keys stay consistent (mirrors the engine's
         * k_emit/host_build_run) */
        rt->b_len = 4 + r->key_len;
        rt->e_len = 4 + r->val_len;
        rt->b_off = arena_add(t, (const uint8_t *)&r->cf_id, 4);
        (void)arena_add(t, rep + r->key_off, r->key_len);
        rt->e_off = arena_add(t, (const uint8_t *)&r->cf_id, 4);
        (void)arena_add(t, rep + r->val_off, r->val_len);
      } else {
        rt->b_len = r->key_len;
        rt->e_len = r->val_len;
        rt->b_off = arena_add(t, rep + r->key_off, r->key_len);
        rt->e_off = arena_add(t, rep + r->val_off, r->val_len);
      }
    } else {
      /* CF-qualified keys are namespaced by prefixing the cf id (our store has
       * no column families; synthetic streams use cf 0). */
      if (r->cf_id != 0) {
        uint8_t tmp[4 + 65536];
        if (r->key_len > 65536) {
          if (recs != stack_recs) free(recs);
          return 0;
        }
        memcpy(tmp, &r->cf_id, 4);
        memcpy(tmp + 4, rep + r->key_off, r->key_len);
        shard_insert(t, bt, seq, tmp, 4 + r->key_len, rep + r->val_off, r->val_len);
      } else {
        shard_insert(t, bt, seq, rep + r->key_off, r->key_len, rep + r->val_off, r->val_len);
      }
    }
  }
  if (recs != stack_recs) free(recs);
  t->latest_seq += cnt; /* Write consumes Count() seqs (assumption test :179-187) */
  return 1;
}

uint64_t orc_latest_seq(const OrcStore *s, uint32_t shard) {
  return shard < s->nshards ? s->shards[shard].latest_seq : 0;
}

/* max seq of a range tombstone covering key with seq > floor */
static uint64_t tomb_cover(const ShardTable *t, const uint8_t *key, size_t klen) {
  uint64_t best = 0;
  for (uint32_t i = 0; i < t->ntombs; i++) {
    const RangeTomb *rt = &t->tombs[i];
    /* begin <= key < end, bytewise (rocksdb default comparator) */
    const uint8_t *b = t->arena + rt->b_off, *e = t->arena + rt->e_off;
    size_t bl = rt->b_len, el = rt->e_len;
    int c1 = memcmp(b, key, bl < klen ? bl : klen);
    if (c1 > 0 || (c1 == 0 && bl > klen)) continue; /* begin > key */
    int c2 = memcmp(key, e, klen < el ? klen : el);
    if (c2 > 0 || (c2 == 0 && klen >= el)) continue; /* key >= end */
    if (rt->seq > best) best = rt->seq;
  }
  return best;
}

int orc_get(const OrcStore *s, uint32_t shard, const void *key_, size_t klen,
            void *buf, size_t cap, size_t *vlen) {
  if (shard >= s->nshards) return 3;
  const uint8_t *key = (const uint8_t *)key_;
  ShardTable *t = &s->shards[shard];
  uint32_t found, slot;
  (void)slot;
  slot = table_find(t, key, klen, &found);
  uint64_t floor_seq = tomb_cover(t, key, klen);
  /* walk newest -> oldest collecting merge operands until base */
  enum { MAX_OPS = 4096 };
  uint32_t ops[MAX_OPS];
  uint32_t nops = 0;
  int have_base = 0;
  uint32_t base_entry = 0;
  uint32_t e = found;
  while (e) {
    Entry *en = &t->entries[e - 1];
    if (en->seq <= floor_seq) break; /* range-deleted below here */
    if (en->type == ORC_TYPE_MERGE) {
      if (nops < MAX_OPS) ops[nops++] = e;
      e = en->prev;
      continue;
    }
    if (en->type == ORC_TYPE_VALUE) {
      have_base = 1;
      base_entry = e;
    }
    break; /* VALUE / DELETION / SINGLE_DELETION all stop the walk */
  }
  if (!have_base && nops == 0) return 1; /* not found / deleted */
  if (nops == 0) {
    /* No merge operands above the base: rocksdb returns the Put's value
     * VERBATIM — the merge operator (FullMerge) only runs when merge
     * records are newer than the base (db/memtable.cc semantics). The
     * earlier restatement folded plain Puts to 8 bytes under u64add,
     * which the round-2 GPU fuzz soak exposed as a deviation. */
    Entry *be = &t->entries[base_entry - 1];
    if (cap < be->val_len) return 2;
    memcpy(buf, t->arena + be->val_off, be->val_len);
    if (vlen) *vlen = be->val_len;
    return 0;
  }
  /* fold operator, oldest -> newest */
  if (s->merge_op == ORC_MERGE_U64ADD) {
    uint64_t acc = 0;
    if (have_base) {
      Entry *be = &t->entries[base_entry - 1];
      memcpy(&acc, t->arena + be->val_off, be->val_len < 8 ? be->val_len : 8);
    }
    for (uint32_t i = nops; i > 0; i--) {
      Entry *oe = &t->entries[ops[i - 1] - 1];
      uint64_t v = 0;
      memcpy(&v, t->arena + oe->val_off, oe->val_len < 8 ? oe->val_len : 8);
      acc += v;
    }
    if (cap < 8) return 2;
    memcpy(buf, &acc, 8);
    if (vlen) *vlen = 8;
    return 0;
  } else {
    size_t need = 0;
    if (have_base) need += t->entries[base_entry - 1].val_len;
    for (uint32_t i = 0; i < nops; i++)
      need += t->entries[ops[i] - 1].val_len + 1;
    if (!have_base && nops) need -= 1;
    if (cap < need) return 2;
    uint8_t *o = (uint8_t *)buf;
    int first = 1;
    if (have_base) {
      Entry *be = &t->entries[base_entry - 1];
      memcpy(o, t->arena + be->val_off, be->val_len);
      o += be->val_len;
      first = 0;
    }
    for (uint32_t i = nops; i > 0; i--) {
      Entry *oe = &t->entries[ops[i - 1] - 1];
      if (!first) *o++ = ',';
      first = 0;
      memcpy(o, t->arena + oe->val_off, oe->val_len);
      o += oe->val_len;
    }
    if (vlen) *vlen = (size_t)(o - (uint8_t *)buf);
    return 0;
  }
}

uint64_t orc_store_bytes(const OrcStore *s) {
  uint64_t n = 0;
  for (uint32_t i = 0; i < s->nshards; i++) n += s->shards[i].arena_len;
  return n;
}

/* ================= CPU baseline bench ================= */
typedef struct {
  OrcStore *s;
  const uint8_t *arena;
  const OrcUpdateDesc *descs;
  const uint64_t *idx; /* this thread's pre-partitioned update indices */
  uint64_t n;
  int tid, nthreads, wal;
} BenchArg;

static void *bench_worker(void *p) {
  BenchArg *a = (BenchArg *)p;
  /* WAL-on variant: the reference's DB::Write appends the rep (+LogData
   * trailer) to the WAL before the memtable insert (rocksdb_wrapper.cpp:22);
   * modeled as a per-thread log append (no fsync — group-commit steady
   * state), reported alongside the WAL-less variant per SURVEY §8d. */
  uint8_t *wal_buf = NULL;
  size_t wal_len = 0, wal_cap = 0;
  /* per-shard sequential, cross-shard concurrent: thread owns shard %
   * nthreads; indices pre-partitioned so no thread scans the full stream */
  for (uint64_t k = 0; k < a->n; k++) {
    const OrcUpdateDesc *d = &a->descs[a->idx[k]];
    if (a->wal) {
      if (wal_len + d->len + 10 > wal_cap) {
        wal_cap = wal_cap ? wal_cap * 2 : (1 << 20);
        if (wal_cap < wal_len + d->len + 10) wal_cap = wal_len + d->len + 10;
        wal_buf = (uint8_t *)realloc(wal_buf, wal_cap);
      }
      memcpy(wal_buf + wal_len, a->arena + d->off, d->len);
      wal_len += d->len + 10; /* + LogData(ts) trailer bytes */
      if (wal_len > (64u << 20)) wal_len = 0; /* rotated segment */
    }
    orc_apply(a->s, d->shard, a->arena + d->off, d->len, d->ts);
  }
  free(wal_buf);
  return NULL;
}

static double cpu_bench(OrcStore *s, const uint8_t *arena,
                        const OrcUpdateDesc *descs, uint64_t n, int nthreads,
                        int wal) {
  if (nthreads < 1) nthreads = 1;
  pthread_t th[256];
  BenchArg args[256];
  if (nthreads > 256) nthreads = 256;
  /* partition updates per owning thread up front (work distribution is the
   * harness's job, as in the reference's executor; not timed) */
  uint64_t *cnt = (uint64_t *)calloc(nthreads + 1, sizeof(uint64_t));
  for (uint64_t i = 0; i < n; i++) cnt[descs[i].shard % nthreads + 1]++;
  for (int t = 0; t < nthreads; t++) cnt[t + 1] += cnt[t];
  uint64_t *idx = (uint64_t *)malloc(n * sizeof(uint64_t));
  uint64_t *pos = (uint64_t *)malloc(nthreads * sizeof(uint64_t));
  memcpy(pos, cnt, nthreads * sizeof(uint64_t));
  for (uint64_t i = 0; i < n; i++) idx[pos[descs[i].shard % nthreads]++] = i;
  struct timespec t0, t1;
  clock_gettime(CLOCK_MONOTONIC, &t0);
  for (int i = 0; i < nthreads; i++) {
    args[i] = (BenchArg){s,    arena,        descs, idx + cnt[i],
                         cnt[i + 1] - cnt[i], i,    nthreads, wal};
    pthread_create(&th[i], NULL, bench_worker, &args[i]);
  }
  for (int i = 0; i < nthreads; i++) pthread_join(th[i], NULL);
  clock_gettime(CLOCK_MONOTONIC, &t1);
  free(cnt);
  free(idx);
  free(pos);
  return (t1.tv_sec - t0.tv_sec) + (t1.tv_nsec - t0.tv_nsec) * 1e-9;
}

double orc_cpu_apply_bench(OrcStore *s, const uint8_t *arena,
                           const OrcUpdateDesc *descs, uint64_t n, int nthreads) {
  return cpu_bench(s, arena, descs, n, nthreads, 0);
}

double orc_cpu_apply_bench_wal(OrcStore *s, const uint8_t *arena,
                               const OrcUpdateDesc *descs, uint64_t n,
                               int nthreads) {
  return cpu_bench(s, arena, descs, n, nthreads, 1);
}

/* ================= Snappy block format ================= */
size_t orc_snappy_max_len(size_t n) { return 32 + n + n / 6; }

static size_t snp_varint(uint8_t *dst, uint32_t v) {
  size_t i = 0;
  while (v >= 0x80) {
    dst[i++] = (uint8_t)(v | 0x80);
    v >>= 7;
  }
  dst[i++] = (uint8_t)v;
  return i;
}

static size_t snp_emit_literal(uint8_t *dst, const uint8_t *src, size_t len) {
  size_t i = 0;
  size_t n = len - 1;
  if (n < 60) {
    dst[i++] = (uint8_t)(n << 2);
  } else if (n < (1u << 8)) {
    dst[i++] = 60 << 2;
    dst[i++] = (uint8_t)n;
  } else if (n < (1u << 16)) {
    dst[i++] = 61 << 2;
    dst[i++] = (uint8_t)n;
    dst[i++] = (uint8_t)(n >> 8);
  } else if (n < (1u << 24)) {
    dst[i++] = 62 << 2;
    dst[i++] = (uint8_t)n;
    dst[i++] = (uint8_t)(n >> 8);
    dst[i++] = (uint8_t)(n >> 16);
  } else {
    dst[i++] = 63 << 2;
    dst[i++] = (uint8_t)n;
    dst[i++] = (uint8_t)(n >> 8);
    dst[i++] = (uint8_t)(n >> 16);
    dst[i++] = (uint8_t)(n >> 24);
  }
  memcpy(dst + i, src, len);
  return i + len;
}

size_t orc_snappy_compress(const uint8_t *src, size_t slen, uint8_t *dst,
                           size_t dcap) {
  if (dcap < orc_snappy_max_len(slen)) return 0;
  size_t o = snp_varint(dst, (uint32_t)slen);
  enum { HBITS = 13, HSIZE = 1 << HBITS };
  uint32_t tab[HSIZE];
  memset(tab, 0xFF, sizeof(tab));
  size_t pos = 0, lit = 0;
  while (pos + 4 <= slen) {
    uint32_t cur;
    memcpy(&cur, src + pos, 4);
    uint32_t h = (cur * 0x1e35a7bdu) >> (32 - HBITS);
    uint32_t cand = tab[h];
    tab[h] = (uint32_t)pos;
    uint32_t c4;
    if (cand != 0xFFFFFFFFu && pos - cand <= 0xFFFF &&
        (memcpy(&c4, src + cand, 4), c4 == cur)) {
      if (pos > lit) o += snp_emit_literal(dst + o, src + lit, pos - lit);
      size_t len = 4, maxlen = slen - pos;
      if (maxlen > 64) maxlen = 64;
      while (len < maxlen && src[cand + len] == src[pos + len]) len++;
      uint32_t off = (uint32_t)(pos - cand);
      dst[o++] = (uint8_t)(((len - 1) << 2) | 2); /* 2-byte-offset copy */
      dst[o++] = (uint8_t)off;
      dst[o++] = (uint8_t)(off >> 8);
      pos += len;
      lit = pos;
    } else {
      pos++;
    }
  }
  if (slen > lit) o += snp_emit_literal(dst + o, src + lit, slen - lit);
  return o;
}

int orc_snappy_decompress(const uint8_t *src, size_t slen, uint8_t *dst,
                          size_t dcap, size_t *dlen) {
  uint32_t ulen;
  size_t ip = varint32_decode(src, src + slen, &ulen);
  if (ip == 0 || ulen > dcap) return 1;
  size_t op = 0;
  while (ip < slen) {
    uint8_t tag = src[ip++];
    if ((tag & 3) == 0) { /* literal */
      uint32_t len = (tag >> 2) + 1;
      if (len > 60) {
        uint32_t nb = len - 60;
        if (ip + nb > slen) return 2;
        len = 0;
        for (uint32_t b = 0; b < nb; b++) len |= (uint32_t)src[ip + b] << (8 * b);
        len += 1;
        ip += nb;
      }
      if (ip + len > slen || op + len > ulen) return 3;
      memcpy(dst + op, src + ip, len);
      ip += len;
      op += len;
    } else {
      uint32_t len, off;
      if ((tag & 3) == 1) {
        len = ((tag >> 2) & 7) + 4;
        if (ip + 1 > slen) return 4;
        off = ((uint32_t)(tag >> 5) << 8) | src[ip];
        ip += 1;
      } else if ((tag & 3) == 2) {
        len = (tag >> 2) + 1;
        if (ip + 2 > slen) return 4;
        off = (uint32_t)src[ip] | ((uint32_t)src[ip + 1] << 8);
        ip += 2;
      } else {
        len = (tag >> 2) + 1;
        if (ip + 4 > slen) return 4;
        off = (uint32_t)src[ip] | ((uint32_t)src[ip + 1] << 8) |
              ((uint32_t)src[ip + 2] << 16) | ((uint32_t)src[ip + 3] << 24);
        ip += 4;
      }
      if (off == 0 || off > op || op + len > ulen) return 5;
      for (uint32_t b = 0; b < len; b++) dst[op + b] = dst[op + b - off];
      op += len;
    }
  }
  if (op != ulen) return 6;
  *dlen = op;
  return 0;
}

static void *snap_bench_worker(void *p) {
  BenchArg *a = (BenchArg *)p;
  uint8_t *scratch = (uint8_t *)malloc(1 << 20);
  for (uint64_t k = 0; k < a->n; k++) {
    const OrcUpdateDesc *d = &a->descs[a->idx[k]];
    size_t ulen;
    if (orc_snappy_decompress(a->arena + d->off, d->len, scratch, 1 << 20,
                              &ulen) == 0)
      orc_apply(a->s, d->shard, scratch, ulen, d->ts);
  }
  free(scratch);
  return NULL;
}

double orc_cpu_snappy_apply_bench(OrcStore *s, const uint8_t *arena,
                                  const OrcUpdateDesc *descs, uint64_t n,
                                  int nthreads) {
  if (nthreads < 1) nthreads = 1;
  if (nthreads > 256) nthreads = 256;
  pthread_t th[256];
  BenchArg args[256];
  uint64_t *cnt = (uint64_t *)calloc(nthreads + 1, sizeof(uint64_t));
  for (uint64_t i = 0; i < n; i++) cnt[descs[i].shard % nthreads + 1]++;
  for (int t = 0; t < nthreads; t++) cnt[t + 1] += cnt[t];
  uint64_t *idx = (uint64_t *)malloc(n * sizeof(uint64_t));
  uint64_t *pos = (uint64_t *)malloc(nthreads * sizeof(uint64_t));
  memcpy(pos, cnt, nthreads * sizeof(uint64_t));
  for (uint64_t i = 0; i < n; i++) idx[pos[descs[i].shard % nthreads]++] = i;
  struct timespec t0, t1;
  clock_gettime(CLOCK_MONOTONIC, &t0);
  for (int i = 0; i < nthreads; i++) {
    args[i] = (BenchArg){s,    arena,        descs, idx + cnt[i],
                         cnt[i + 1] - cnt[i], i,    nthreads, 0};
    pthread_create(&th[i], NULL, snap_bench_worker, &args[i]);
  }
  for (int i = 0; i < nthreads; i++) pthread_join(th[i], NULL);
  clock_gettime(CLOCK_MONOTONIC, &t1);
  free(cnt);
  free(idx);
  free(pos);
  return (t1.tv_sec - t0.tv_sec) + (t1.tv_nsec - t0.tv_nsec) * 1e-9;
}

/* ================= full-store checksum (parity at any size) ================= */
static uint64_t rec_hash(uint64_t seq, uint8_t type, uint32_t klen,
                         uint32_t vlen, const uint8_t *key, const uint8_t *val) {
  uint64_t h = 1469598103934665603ULL;
#define FOLD(b) h = (h ^ (uint8_t)(b)) * 1099511628211ULL
  for (int i = 0; i < 8; i++) FOLD(seq >> (8 * i));
  FOLD(type);
  for (int i = 0; i < 4; i++) FOLD(klen >> (8 * i));
  for (int i = 0; i < 4; i++) FOLD(vlen >> (8 * i));
  for (uint32_t i = 0; i < klen; i++) FOLD(key[i]);
  for (uint32_t i = 0; i < vlen; i++) FOLD(val[i]);
#undef FOLD
  return h;
}

uint64_t orc_shard_checksum(const OrcStore *s, uint32_t shard) {
  if (shard >= s->nshards) return 0;
  const ShardTable *t = &s->shards[shard];
  uint64_t sum = 0;
  for (uint32_t i = 0; i < t->nentries; i++) {
    const Entry *e = &t->entries[i];
    sum += rec_hash(e->seq, e->type, e->key_len, e->val_len,
                    t->arena + e->key_off, t->arena + e->val_off);
  }
  for (uint32_t i = 0; i < t->ntombs; i++) {
    const RangeTomb *rt = &t->tombs[i];
    sum += rec_hash(rt->seq, ORC_TYPE_RANGE_DELETION, rt->b_len, rt->e_len,
                    t->arena + rt->b_off, t->arena + rt->e_off);
  }
  return sum;
}
