/* test_cpp_chain.cpp — native C++ end-to-end replication over the C-ABI:
 * leader writes (modes 0 and 2) replicate through GpuReplicator pull
 * threads to a follower and a chained tail; verified by seq catch-up +
 * per-key Get equality (the reference's own verification pattern).
 * Exit 0 = pass. Built/run by the GPU test suite (test_cpp_native.py).
 *
 * Build: hipcc -O3 -std=c++17 scripts/test_cpp_chain.cpp -Iinclude \
 *          -Lrocksplicator_amd -lgra -Wl,-rpath,'$ORIGIN/../rocksplicator_amd' \
 *          -o build/test_cpp_chain
 */
#include <cassert>
#include <cstdio>
#include <cstring>
#include <random>
#include <string>
#include <vector>

#include "../include/rocksplicator_gpu.h"
#include "../include/rocksplicator_replicator.hpp"

static std::vector<uint8_t> put_batch(const std::string &k, const std::string &v) {
  GraBatch *b = gra_wb_create();
  gra_wb_put(b, k.data(), k.size(), v.data(), v.size());
  size_t len;
  const uint8_t *p = gra_wb_data(b, &len);
  std::vector<uint8_t> out(p, p + len);
  gra_wb_destroy(b);
  return out;
}

static std::string get_or_miss(GraDb *db, const std::string &k) {
  char buf[4096];
  size_t vlen = 0;
  int rc = gra_get(db, k.data(), k.size(), buf, sizeof(buf), &vlen);
  if (rc == GRA_NOT_FOUND) return "<miss>";
  if (rc != GRA_OK) return std::string("<err:") + gra_last_error() + ">";
  return std::string(buf, vlen);
}

int main() {
  GraEngineOpts opts;
  gra_engine_opts_init(&opts);
  opts.nshards = 8;
  opts.retain_log = 1;
  GraEngine *el = nullptr, *em = nullptr, *et = nullptr;
  if (gra_engine_create(&opts, &el) != GRA_OK ||
      gra_engine_create(&opts, &em) != GRA_OK ||
      gra_engine_create(&opts, &et) != GRA_OK) {
    fprintf(stderr, "engine: %s\n", gra_last_error());
    return 1;
  }
  {
    gra::GpuReplicator leader(el), mid(em), tail(et);
    GraDb *ldb = leader.add_db("db0", gra::Role::LEADER);
    GraDb *mdb = mid.add_db("db0", gra::Role::FOLLOWER, gra::local_upstream(ldb));
    GraDb *tdb = tail.add_db("db0", gra::Role::FOLLOWER, gra::local_upstream(mdb));

    std::mt19937_64 rng(7);
    const int N = 2000;
    for (int i = 0; i < N; i++) {
      std::string k = "key" + std::to_string(rng() % 64);
      std::string v = "v" + std::to_string(i);
      auto rep = put_batch(k, v);
      /* every 10th write waits for the follower-confirmed ACK (mode 2) */
      leader.write("db0", rep.data(), rep.size(), i % 10 == 9 ? 2 : 0);
    }
    /* wait for the chain to converge */
    for (int spin = 0; spin < 20000; spin++) {
      if (gra_latest_seq(tdb) == (uint64_t)N) break;
      std::this_thread::sleep_for(std::chrono::milliseconds(1));
    }
    mid.flush();
    tail.flush();
    assert(gra_latest_seq(ldb) == (uint64_t)N);
    assert(gra_latest_seq(mdb) == (uint64_t)N);
    assert(gra_latest_seq(tdb) == (uint64_t)N);
    for (int i = 0; i < 64; i++) {
      std::string k = "key" + std::to_string(i);
      std::string a = get_or_miss(ldb, k), b = get_or_miss(mdb, k),
                  c = get_or_miss(tdb, k);
      if (a != b || b != c) {
        fprintf(stderr, "MISMATCH %s: %s / %s / %s\n", k.c_str(), a.c_str(),
                b.c_str(), c.c_str());
        return 2;
      }
    }
    /* role transition: promote mid, write through it, tail keeps following */
    mid.change_role("db0", gra::Role::LEADER);
    auto rep = put_batch("promoted", "yes");
    mid.write("db0", rep.data(), rep.size());
    for (int spin = 0; spin < 20000; spin++) {
      if (gra_latest_seq(tdb) == (uint64_t)N + 1) break;
      std::this_thread::sleep_for(std::chrono::milliseconds(1));
    }
    tail.flush();
    assert(gra_latest_seq(tdb) == (uint64_t)N + 1);
    assert(get_or_miss(tdb, "promoted") == "yes");
    /* WRITE_TO_SLAVE on the tail */
    bool threw = false;
    try {
      tail.write("db0", rep.data(), rep.size());
    } catch (const std::exception &) {
      threw = true;
    }
    assert(threw);
    /* per-db counters flowed */
    GraDbCounters c;
    gra_db_counters(mdb, &c);
    assert(c.updates_applied == (uint64_t)N);
    assert(c.updates_served >= (uint64_t)N); /* mid re-served to tail */
  }
  gra_engine_destroy(el);
  gra_engine_destroy(em);
  gra_engine_destroy(et);
  printf("cpp chain OK\n");
  return 0;
}
