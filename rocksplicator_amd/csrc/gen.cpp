/* gen.cpp — deterministic synthetic replication-stream generator (harness).
 * Shapes follow BASELINE.json configs / SURVEY §8d: per-shard xoshiro256**
 * streams (seed base + shard), 1 record per WriteBatch (mirrors
 * performance.cpp:139-142), shard-grouped within each ≤50-update response
 * window (replicated_db.cpp:42-43). Zipf key skew via the YCSB-style
 * constant-time zipfian generator. */
#include <cmath>
#include <cstring>
#include <mutex>
#include <vector>

#include "../../include/rocksplicator_gpu.h"
#include "snappy.h"
#include "wb_format.h"

namespace {

struct Xoshiro {
  uint64_t s[4];
  static uint64_t rotl(uint64_t x, int k) { return (x << k) | (x >> (64 - k)); }
  explicit Xoshiro(uint64_t seed) {
    /* splitmix64 init */
    uint64_t z = seed;
    for (int i = 0; i < 4; i++) {
      z += 0x9e3779b97f4a7c15ULL;
      uint64_t t = z;
      t = (t ^ (t >> 30)) * 0xbf58476d1ce4e5b9ULL;
      t = (t ^ (t >> 27)) * 0x94d049bb133111ebULL;
      s[i] = t ^ (t >> 31);
    }
  }
  uint64_t next() {
    uint64_t r = rotl(s[1] * 5, 7) * 9;
    uint64_t t = s[1] << 17;
    s[2] ^= s[0];
    s[3] ^= s[1];
    s[1] ^= s[2];
    s[0] ^= s[3];
    s[2] ^= t;
    s[3] = rotl(s[3], 45);
    return r;
  }
  double unit() { return (next() >> 11) * 0x1.0p-53; }
};

/* YCSB ZipfianGenerator constants for item count N, skew s */
struct Zipf {
  uint64_t n;
  double theta, zeta_n, zeta2, alpha, eta;
  Zipf(uint64_t n_, double s) : n(n_), theta(s) {
    zeta2 = zeta(2, theta);
    zeta_n = zeta(n, theta);
    alpha = 1.0 / (1.0 - theta);
    eta = (1 - std::pow(2.0 / (double)n, 1 - theta)) / (1 - zeta2 / zeta_n);
  }
  static double zeta(uint64_t n, double theta) {
    double z = 0;
    for (uint64_t i = 1; i <= n; i++) z += 1.0 / std::pow((double)i, theta);
    return z;
  }
  uint64_t next(Xoshiro &rng) const {
    double u = rng.unit();
    double uz = u * zeta_n;
    if (uz < 1.0) return 0;
    if (uz < 1.0 + std::pow(0.5, theta)) return 1;
    return (uint64_t)((double)n * std::pow(eta * u - eta + 1, alpha));
  }
};

/* zeta(2^24, .99) is ~seconds to compute; memoize per (n, s) */
const Zipf &zipf_for(uint64_t n, double s) {
  static std::mutex mu;
  static std::vector<std::pair<std::pair<uint64_t, double>, Zipf *>> cache;
  std::lock_guard<std::mutex> g(mu);
  for (auto &e : cache)
    if (e.first.first == n && e.first.second == s) return *e.second;
  cache.push_back({{n, s}, new Zipf(n, s)});
  return *cache.back().second;
}

inline size_t put_varint(uint8_t *p, uint32_t v) {
  size_t i = 0;
  while (v >= 0x80) {
    p[i++] = (uint8_t)(v | 0x80);
    v >>= 7;
  }
  p[i++] = (uint8_t)v;
  return i;
}

} /* namespace */

extern "C" int gra_gen_stream(const GraGenOpts *g, uint64_t n_updates,
                              uint8_t *arena, size_t arena_cap,
                              size_t *arena_used, GraUpdateDesc *descs,
                              int64_t ts) {
  const uint32_t kl = g->key_len, vl = g->val_len;
  const uint64_t kspace = g->key_space ? g->key_space : (1ULL << 24);
  const Zipf *zipf = g->kind == 1 ? &zipf_for(kspace, g->zipf_s ? g->zipf_s : 0.99) : nullptr;
  std::vector<Xoshiro> rngs;
  rngs.reserve(g->nshards);
  for (uint32_t s = 0; s < g->nshards; s++)
    rngs.emplace_back((g->seed ? g->seed : 0xB0CC5EEDULL) + s);

  size_t off = 0;
  uint64_t u = 0;
  /* shard-grouped response windows: each shard emits up to 50 consecutive
   * updates (one pull response), round-robin over shards */
  const uint64_t kWindow = 50;
  uint32_t shard = 0;
  while (u < n_updates) {
    uint64_t in_window = n_updates - u < kWindow ? n_updates - u : kWindow;
    Xoshiro &rng = rngs[shard];
    for (uint64_t w = 0; w < in_window; w++, u++) {
      /* record type: kind 2 = 70/20/10 put/delete/merge, else put */
      uint32_t tag = wb::kValue;
      if (g->kind == 2) {
        double r = rng.unit();
        tag = r < 0.7 ? wb::kValue : (r < 0.9 ? wb::kDeletion : wb::kMerge);
      }
      uint32_t this_vl = tag == wb::kDeletion ? 0 : vl;
      /* worst-case size: 12 hdr + 1 tag + 5 + kl + 5 + vl */
      if (off + 23 + kl + this_vl + 16 > arena_cap) return GRA_FULL;
      uint8_t *p = arena + off;
      /* header: seq (leader-side; follower reassigns identically) + count=1 */
      memset(p, 0, wb::kHeaderBytes);
      p[8] = 1;
      size_t pos = wb::kHeaderBytes;
      p[pos++] = (uint8_t)tag;
      pos += put_varint(p + pos, kl);
      /* key bytes: 8B key id (uniform or zipf over key_space) + filler */
      uint64_t kid = zipf ? zipf->next(rng) : (rng.next() % kspace);
      uint8_t kb[8];
      memcpy(kb, &kid, 8);
      for (uint32_t i = 0; i < kl; i++) p[pos + i] = i < 8 ? kb[i] : (uint8_t)(i * 131 + shard);
      pos += kl;
      if (tag != wb::kDeletion) {
        pos += put_varint(p + pos, this_vl);
        if (g->compressible) {
          /* low-entropy values (config #5 transport-compression shape):
           * 64-byte blocks drawn from 4 seed-derived patterns */
          for (uint32_t i = 0; i < this_vl; i += 64) {
            uint32_t c = (uint32_t)(rng.next() & 3);
            uint32_t run = this_vl - i < 64 ? this_vl - i : 64;
            for (uint32_t j = 0; j < run; j++)
              p[pos + i + j] = (uint8_t)(c * 67 + (j & 15) * 13 + 7);
          }
        } else {
          uint32_t i = 0;
          for (; i + 8 <= this_vl; i += 8) {
            uint64_t x = rng.next();
            memcpy(p + pos + i, &x, 8);
          }
          if (i < this_vl) {
            uint64_t x = rng.next();
            memcpy(p + pos + i, &x, this_vl - i);
          }
        }
        pos += this_vl;
      }
      descs[u].shard = shard;
      descs[u].len = (uint32_t)pos;
      descs[u].off = off;
      descs[u].ts = ts;
      off += pos;
    }
    shard = (shard + 1) % g->nshards;
  }
  if (arena_used) *arena_used = off;
  return GRA_OK;
}

extern "C" uint32_t gra_snappy_compress(const uint8_t *src, uint32_t slen,
                                        uint8_t *dst, uint32_t dcap) {
  return snp::compress(src, slen, dst, dcap);
}
extern "C" uint32_t gra_snappy_decompress(const uint8_t *src, uint32_t slen,
                                          uint8_t *dst, uint32_t dcap) {
  return snp::decompress(src, slen, dst, dcap);
}

#include <thread>

/* Parallel host compression of a generated stream (harness transport side).
 * Slot layout uses worst-case per-update offsets so threads write
 * independently; out_descs carry the real compressed lengths. */
extern "C" int gra_snappy_compress_stream(const uint8_t *arena,
                                          const GraUpdateDesc *descs,
                                          uint64_t n, uint8_t *out,
                                          size_t out_cap, size_t *out_used,
                                          GraUpdateDesc *out_descs,
                                          uint32_t *ulens, int nthreads) {
  std::vector<uint64_t> off(n + 1);
  off[0] = 0;
  for (uint64_t i = 0; i < n; i++)
    off[i + 1] = off[i] + snp::max_compressed_len(descs[i].len);
  if (off[n] > out_cap) return GRA_FULL;
  if (nthreads < 1) nthreads = 1;
  std::vector<std::thread> ths;
  for (int t = 0; t < nthreads; t++) {
    ths.emplace_back([&, t] {
      for (uint64_t i = t; i < n; i += (uint64_t)nthreads) {
        const GraUpdateDesc &d = descs[i];
        uint32_t clen = snp::compress(arena + d.off, d.len, out + off[i],
                                      (uint32_t)(off[i + 1] - off[i]));
        out_descs[i].shard = d.shard;
        out_descs[i].len = clen;
        out_descs[i].off = off[i];
        out_descs[i].ts = d.ts;
        ulens[i] = d.len;
      }
    });
  }
  for (auto &th : ths) th.join();
  if (out_used) *out_used = off[n];
  return GRA_OK;
}
