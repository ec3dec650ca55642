/* bench_stream.cpp — C++ streaming-ingest benchmark: the INTEGRATION-
 * realistic path (gra_handle_replicate_response from concurrent threads,
 * pinned staging + H2D inside the pipeline), as a DbWrapper adapter would
 * drive it. PCIe staging bounds this mode (side figure; the HBM-resident
 * replay path is the headline metric — see DESIGN.md).
 *
 * Build: hipcc -O3 -std=c++17 scripts/bench_stream.cpp -Lrocksplicator_amd -lgra \
 *          -Wl,-rpath,'$ORIGIN/../rocksplicator_amd' -o build/bench_stream
 * Run:   ./build/bench_stream [nshards] [nthreads] [val_len] [seconds]
 */
#include <atomic>
#include <chrono>
#include <cstdio>
#include <cstring>
#include <thread>
#include <vector>

#include "../include/rocksplicator_gpu.h"

int main(int argc, char **argv) {
  uint32_t nshards = argc > 1 ? atoi(argv[1]) : 1024;
  int nthreads = argc > 2 ? atoi(argv[2]) : 16;
  uint32_t vlen = argc > 3 ? atoi(argv[3]) : 1024;
  double seconds = argc > 4 ? atof(argv[4]) : 5.0;
  uint32_t staging_mb = argc > 5 ? atoi(argv[5]) : 0; /* 0 = engine default */
  int64_t ts = argc > 6 ? atoll(argv[6]) : 1; /* 0 disables the per-call
                                                 latency clock sample */

  GraEngineOpts opts;
  gra_engine_opts_init(&opts);
  opts.nshards = nshards;
  opts.store_ring = 1;
  opts.store_bytes = 8ULL << 30;
  if (staging_mb) opts.staging_bytes = (uint64_t)staging_mb << 20;
  GraEngine *e = nullptr;
  if (gra_engine_create(&opts, &e) != GRA_OK) {
    fprintf(stderr, "engine: %s\n", gra_last_error());
    return 1;
  }

  /* one pre-built blob per shard (16B key + vlen value, 1-record batch) */
  GraBatch *b = gra_wb_create();
  std::vector<uint8_t> key(16, 0x2A), val(vlen, 0x5C);
  gra_wb_put(b, key.data(), key.size(), val.data(), val.size());
  size_t blen;
  const uint8_t *bdata = gra_wb_data(b, &blen);

  std::atomic<uint64_t> total{0};
  std::atomic<bool> stop{false};
  std::vector<std::thread> ths;
  auto t0 = std::chrono::steady_clock::now();
  for (int t = 0; t < nthreads; t++) {
    ths.emplace_back([&, t] {
      /* thread owns shards t, t+nthreads, ... — sequential per shard,
       * concurrent across shards (the reference executor contract) */
      std::vector<GraDb *> dbs;
      for (uint32_t s = t; s < nshards; s += nthreads)
        dbs.push_back(gra_open(e, s));
      uint64_t n = 0;
      while (!stop.load(std::memory_order_relaxed)) {
        for (GraDb *db : dbs) {
          if (!gra_handle_replicate_response(db, bdata, blen, ts)) {
            fprintf(stderr, "apply refused\n");
            stop = true;
            break;
          }
          n++;
        }
      }
      total += n;
      for (GraDb *db : dbs) gra_close(db);
    });
  }
  std::this_thread::sleep_for(std::chrono::duration<double>(seconds));
  stop = true;
  for (auto &th : ths) th.join();
  gra_flush(e);
  auto t1 = std::chrono::steady_clock::now();
  double secs = std::chrono::duration<double>(t1 - t0).count();
  GraStats st;
  gra_stats(e, &st);
  printf("{\"mode\": \"streaming-cpp\", \"updates_per_s\": %.0f, "
         "\"nthreads\": %d, \"nshards\": %u, \"val_len\": %u, "
         "\"blob_GBps\": %.2f, \"gpu_busy_ms_per_tick\": %.3f, "
         "\"ticks\": %llu}\n",
         (double)total / secs, nthreads, nshards, vlen,
         (double)st.blob_bytes / secs / 1e9,
         st.ticks ? st.total_ms / st.ticks : 0.0,
         (unsigned long long)st.ticks);
  gra_wb_destroy(b);
  gra_engine_destroy(e);
  return 0;
}
