/* gpu_apply_db_wrapper.h — the DbWrapper adapter a rocksplicator maintainer
 * adds to drop the GPU apply path in (INTEGRATION.md shows this file in
 * context). In the reference tree it includes
 * "rocksdb_replicator/db_wrapper.h"; here the same include resolves to the
 * compile-proof mock (tools/mock_seam/rocksdb_replicator/db_wrapper.h) so
 * `make tools` proves the adapter against the exact 4-method seam
 * (db_wrapper.h:6-15) without rocksdb/folly being installable.
 */
#pragma once
#include <memory>

#include "rocksdb_replicator/db_wrapper.h"
#include "rocksplicator_gpu.h" /* -lgra */

namespace replicator {

/* One GraEngine per GPU per process (shared across shards), created at
 * service start:  gra_engine_opts_init(&o); o.nshards = ...;
 * gra_engine_create(&o, &engine); */
class GpuApplyDbWrapper : public DbWrapper {
 public:
  GpuApplyDbWrapper(GraEngine* engine, uint32_t shard_id,
                    std::shared_ptr<rocksdb::DB> leader_db /*leader role only*/)
      : db_(gra_open(engine, shard_id)), leader_db_(std::move(leader_db)) {}
  ~GpuApplyDbWrapper() override { gra_close(db_); }

  /* ≅ RocksDbWrapper::HandleReplicateResponse (rocksdb_wrapper.cpp:13-28).
   * Same bool contract the pull loop checks (replicated_db.cpp:378):
   * false ⇒ counter + delayed re-pull from LatestSequenceNumber(). */
  bool HandleReplicateResponse(Update* update) override {
    auto range = update->raw_data.coalesce();
    return gra_handle_replicate_response(db_, range.data(), range.size(),
                                         update->timestamp) != 0;
  }

  /* ≅ RocksDbWrapper::LatestSequenceNumber (rocksdb_wrapper.cpp:4). */
  uint64_t LatestSequenceNumber() override { return gra_latest_seq(db_); }

  /* Leader-side paths (rocksdb_wrapper.cpp:5-12): local write routes to
   * gra_write_leader; update-serving can stay on a retained local DB. */
  rocksdb::Status WriteToLeader(const rocksdb::WriteOptions& options,
                                rocksdb::WriteBatch* updates) override {
    (void)options;
    uint64_t seq = 0;
    int rc = gra_write_leader(
        db_, reinterpret_cast<const uint8_t*>(updates->Data().data()),
        updates->GetDataSize(), &seq);
    return rc == GRA_OK ? rocksdb::Status::OK()
                        : rocksdb::Status::Corruption(gra_last_error());
  }

  rocksdb::Status GetUpdatesFromLeader(
      rocksdb::SequenceNumber seq,
      std::unique_ptr<rocksdb::TransactionLogIterator>* iter) override {
    /* Leader update-serving via gra_get_updates is wired in the native
     * control layer (rocksplicator_replicator.hpp); a leader that keeps a
     * local rocksdb::DB for the stock iterator path forwards here. */
    return leader_db_ ? leader_db_->GetUpdatesSince(seq, iter)
                      : rocksdb::Status::NotSupported("follower-only");
  }

 private:
  GraDb* db_;
  std::shared_ptr<rocksdb::DB> leader_db_;
};

} /* namespace replicator */
