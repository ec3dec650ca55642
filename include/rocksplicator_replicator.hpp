/* rocksplicator_replicator.hpp — native C++ replicator registry over the
 * C-ABI (header-only; link -lgra).
 *
 * ≅ replicator::RocksDBReplicator (rocksdb_replicator/rocksdb_replicator.h:
 * 160-256) re-imagined for the GPU engine: a per-process registry of named
 * replicated shards, per-shard roles with transitions, one pull thread per
 * FOLLOWER/OBSERVER (the pullFromUpstream control flow,
 * replicated_db.cpp:314-433), write modes 0/1/2 over the engine's ACK box
 * (MaxNumberBox equivalent) and the reference's write-degradation cadence
 * (replicated_db.cpp:236-273). The upstream is any callable returning
 * (seq, ts, rep) triples — an in-process leader handle, or a TCP/fbthrift
 * client in a deployment.
 */
#pragma once
#include <atomic>
#include <chrono>
#include <cstring>
#include <functional>
#include <map>
#include <memory>
#include <mutex>
#include <stdexcept>
#include <string>
#include <thread>
#include <vector>

#include "rocksplicator_gpu.h"

namespace gra {

struct WireUpdate {
  uint64_t seq;
  int64_t ts;
  std::vector<uint8_t> rep;
};

/* Upstream source: updates with base seq > since, at most max_updates
 * (≅ Replicator.replicate(), replicator.thrift:89-92). */
using UpstreamFn =
    std::function<std::vector<WireUpdate>(uint64_t since, uint32_t max_updates)>;

/* Serve an in-process leader GraDb as an upstream (test/demo wiring).
 * observer: the request carries the observer role (no ACK posted). */
inline UpstreamFn local_upstream(GraDb *db, bool observer = false) {
  return [db, observer](uint64_t since, uint32_t max_updates) {
    std::vector<GraServedUpdate> out(max_updates);
    std::vector<uint8_t> buf(4u << 20);
    uint32_t n = 0;
    if (gra_get_updates(db, since, max_updates, out.data(), &n, buf.data(),
                        buf.size(), observer ? 1 : 0) != GRA_OK)
      throw std::runtime_error(gra_last_error());
    std::vector<WireUpdate> ups(n);
    for (uint32_t i = 0; i < n; i++) {
      ups[i].seq = out[i].seq;
      ups[i].ts = out[i].ts;
      ups[i].rep.assign(buf.data() + out[i].off,
                        buf.data() + out[i].off + out[i].len);
    }
    return ups;
  };
}

enum class Role { LEADER, FOLLOWER, OBSERVER };

class GpuReplicator {
 public:
  /* Reference cadence constants (replicated_db.cpp:36-90). */
  int ack_timeout_ms = 2000;
  int degraded_timeout_ms = 10;
  int degrade_after_misses = 100;
  int pull_idle_ms = 2;

  explicit GpuReplicator(GraEngine *engine) : engine_(engine) {}
  ~GpuReplicator() { close(); }

  /* ≅ RocksDBReplicator::addDB (rocksdb_replicator.cpp:96-133): FOLLOWER
   * starts pulling immediately. */
  GraDb *add_db(const std::string &name, Role role,
                UpstreamFn upstream = nullptr) {
    std::lock_guard<std::mutex> lk(mu_);
    if (dbs_.count(name)) throw std::runtime_error("db exists: " + name);
    auto rs = std::make_shared<Shard>();
    rs->db = gra_open(engine_, next_shard_++);
    rs->role = role;
    rs->upstream = std::move(upstream);
    dbs_[name] = rs;
    if (role != Role::LEADER && rs->upstream) start_pull(rs);
    return rs->db;
  }

  /* ≅ RocksDBReplicator::removeDB (rocksdb_replicator.cpp:135-154). */
  void remove_db(const std::string &name) {
    std::shared_ptr<Shard> rs;
    {
      std::lock_guard<std::mutex> lk(mu_);
      rs = dbs_.at(name);
      dbs_.erase(name);
    }
    stop_pull(*rs);
    gra_close(rs->db);
  }

  /* ≅ changeDBRoleAndUpstream-driven transition. */
  void change_role(const std::string &name, Role role,
                   UpstreamFn upstream = nullptr) {
    std::shared_ptr<Shard> rs;
    {
      std::lock_guard<std::mutex> lk(mu_);
      rs = dbs_.at(name);
    }
    stop_pull(*rs);
    rs->role = role;
    rs->upstream = std::move(upstream);
    if (role != Role::LEADER && rs->upstream) start_pull(rs);
  }

  GraDb *get(const std::string &name) {
    std::lock_guard<std::mutex> lk(mu_);
    return dbs_.at(name)->db;
  }

  /* ≅ RocksDBReplicator::write → ReplicatedDB::Write
   * (replicated_db.cpp:103-166): throws on follower (WRITE_TO_SLAVE),
   * modes 1/2 wait on the downstream ACK with degradation. */
  uint64_t write(const std::string &name, const uint8_t *rep, size_t len,
                 int mode = 0) {
    std::shared_ptr<Shard> rs;
    {
      std::lock_guard<std::mutex> lk(mu_);
      rs = dbs_.at(name);
    }
    if (rs->role != Role::LEADER)
      throw std::runtime_error("WRITE_TO_SLAVE: " + name);
    uint64_t seq = 0;
    int rc = gra_write_leader(rs->db, rep, len, &seq);
    if (rc != GRA_OK) throw std::runtime_error(gra_last_error());
    if (mode == 1 || mode == 2) {
      /* degradation state is PER shard, as the reference keeps it per
       * ReplicatedDB (replicated_db.cpp:236-273) — one shard's healthy
       * acks must not reset another's degradation */
      int timeout = rs->consecutive_misses.load() >= degrade_after_misses
                        ? degraded_timeout_ms
                        : ack_timeout_ms;
      if (gra_wait_ack(rs->db, seq, mode == 2 ? 1 : 0, timeout) == GRA_OK)
        rs->consecutive_misses.store(0);
      else
        rs->consecutive_misses.fetch_add(1);
    }
    return seq;
  }

  void flush() { gra_flush(engine_); }

  void close() {
    std::map<std::string, std::shared_ptr<Shard>> dbs;
    {
      std::lock_guard<std::mutex> lk(mu_);
      dbs.swap(dbs_);
    }
    for (auto &kv : dbs) {
      stop_pull(*kv.second);
      gra_close(kv.second->db);
    }
  }

 private:
  struct Shard {
    GraDb *db = nullptr;
    Role role = Role::FOLLOWER;
    UpstreamFn upstream;
    std::thread thread;
    std::atomic<bool> stop{false};
    std::atomic<int> consecutive_misses{0}; /* write-degradation, per db */
  };

  /* The pullFromUpstream control flow (replicated_db.cpp:314-433): request
   * from LatestSequenceNumber, apply in order, failed apply → re-pull from
   * the (rolled back) latest seq; idle/error → short delay. */
  void start_pull(const std::shared_ptr<Shard> &rs) {
    rs->stop.store(false);
    GpuReplicator *self = this;
    rs->thread = std::thread([self, rs] {
      while (!rs->stop.load()) {
        bool progressed = false;
        try {
          uint64_t since = gra_latest_seq(rs->db);
          auto ups = rs->upstream(since, 50);
          for (auto &u : ups) {
            if (!gra_handle_replicate_response(rs->db, u.rep.data(),
                                               u.rep.size(), u.ts))
              break;
            progressed = true;
          }
          if (progressed) gra_flush(self->engine_);
        } catch (...) {
          /* retry after delay (randomized in the reference, :412-431) */
        }
        if (!progressed && !rs->stop.load())
          std::this_thread::sleep_for(
              std::chrono::milliseconds(self->pull_idle_ms));
      }
    });
  }

  void stop_pull(Shard &rs) {
    if (rs.thread.joinable()) {
      rs.stop.store(true);
      rs.thread.join();
    }
  }

  GraEngine *engine_;
  std::mutex mu_;
  std::map<std::string, std::shared_ptr<Shard>> dbs_;
  uint32_t next_shard_ = 0;
};

} /* namespace gra */
