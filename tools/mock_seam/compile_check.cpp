/* compile_check.cpp — proves INTEGRATION.md's GpuApplyDbWrapper compiles
 * against the exact DbWrapper virtual seam (db_wrapper.h:6-15, restated by
 * the mock header) and links against libgra.so's C-ABI. Built by
 * `make tools`; runs on CPU (constructs nothing that needs a GPU — engine
 * creation is the GPU-gated step, by design). */
#include <cstdio>

#include "gpu_apply_db_wrapper.h"

/* Never called at runtime; fully type-checked + linked. The compiler must
 * accept the adapter as a concrete DbWrapper (all 4 pure virtuals
 * overridden with the reference's exact signatures). */
replicator::DbWrapper* make_adapter(GraEngine* engine, uint32_t shard) {
  return new replicator::GpuApplyDbWrapper(engine, shard, nullptr);
}

/* Exercise the seam types end-to-end at compile time. */
static bool drive(replicator::DbWrapper* w) {
  replicator::Update u;
  u.raw_data = folly::IOBuf(std::string(12, '\0'));
  u.timestamp = 123;
  bool ok = w->HandleReplicateResponse(&u);
  rocksdb::WriteOptions wo;
  rocksdb::WriteBatch wb;
  rocksdb::Status s = w->WriteToLeader(wo, &wb);
  std::unique_ptr<rocksdb::TransactionLogIterator> it;
  rocksdb::Status s2 = w->GetUpdatesFromLeader(w->LatestSequenceNumber(), &it);
  return ok && s.ok() && s2.ok();
}

int main() {
  /* link-time proof only: verify the adapter factory and the C-ABI symbols
   * resolve, without creating an engine (needs a GPU). */
  std::printf("adapter compile check OK: make=%p drive=%p gra_last_error=%p\n",
              (void*)&make_adapter, (void*)&drive, (void*)&gra_last_error);
  return 0;
}
