/* COMPILE-PROOF MOCK — test infrastructure only, never shipped.
 *
 * Restates the reference seam `rocksdb_replicator/db_wrapper.h:6-15`
 * (pinterest/rocksplicator) plus the minimal slice of rocksdb / thrift /
 * folly types its signatures name, so INTEGRATION.md's GpuApplyDbWrapper
 * compiles against the exact 4-method virtual interface in a container
 * that has no rocksdb/folly/fbthrift headers. Behaviorally
 * inert: the rocksdb::DB surface is the two calls the adapter forwards.
 */
#pragma once
#include <cstdint>
#include <cstring>
#include <memory>
#include <string>

namespace folly {
/* IOBuf mock: the adapter only calls coalesce() and reads data()/size(). */
class IOBuf {
 public:
  struct Range {
    const uint8_t *data_;
    size_t size_;
    const uint8_t *data() const { return data_; }
    size_t size() const { return size_; }
  };
  IOBuf() = default;
  explicit IOBuf(std::string bytes) : bytes_(std::move(bytes)) {}
  Range coalesce() {
    return Range{reinterpret_cast<const uint8_t *>(bytes_.data()),
                 bytes_.size()};
  }

 private:
  std::string bytes_;
};
} /* namespace folly */

namespace rocksdb {

using SequenceNumber = uint64_t;

class Status {
 public:
  static Status OK() { return Status(0, ""); }
  static Status Corruption(const std::string &msg) { return Status(2, msg); }
  static Status NotSupported(const std::string &msg) { return Status(3, msg); }
  bool ok() const { return code_ == 0; }
  int code() const { return code_; }
  const std::string &message() const { return msg_; }

 private:
  Status(int c, std::string m) : code_(c), msg_(std::move(m)) {}
  int code_ = 0;
  std::string msg_;
};

struct WriteOptions {
  bool sync = false;
};

/* WriteBatch surface the adapter touches: rep bytes + size. */
class WriteBatch {
 public:
  explicit WriteBatch(std::string rep = std::string(12, '\0'))
      : rep_(std::move(rep)) {}
  const std::string &Data() const { return rep_; }
  size_t GetDataSize() const { return rep_.size(); }

 private:
  std::string rep_;
};

class TransactionLogIterator {
 public:
  virtual ~TransactionLogIterator() = default;
};

/* DB surface the adapter forwards to (leader role only). */
class DB {
 public:
  virtual ~DB() = default;
  virtual Status GetUpdatesSince(SequenceNumber,
                                 std::unique_ptr<TransactionLogIterator> *) {
    return Status::NotSupported("mock");
  }
  virtual SequenceNumber GetLatestSequenceNumber() const { return 0; }
};

} /* namespace rocksdb */

namespace replicator {

/* Update wire triple ≅ replicator.thrift:44-57 (raw_data as folly::IOBuf,
 * timestamp ms, optional leader seq_no). */
struct Update {
  folly::IOBuf raw_data;
  int64_t timestamp = 0;
  int64_t seq_no = 0;
};

/* The seam itself — signatures restated 1:1 from db_wrapper.h:6-15. */
class DbWrapper {
 public:
  virtual ~DbWrapper() = default;
  virtual rocksdb::Status WriteToLeader(const rocksdb::WriteOptions &options,
                                        rocksdb::WriteBatch *updates) = 0;
  virtual rocksdb::Status GetUpdatesFromLeader(
      rocksdb::SequenceNumber seq_number,
      std::unique_ptr<rocksdb::TransactionLogIterator> *iter) = 0;
  virtual uint64_t LatestSequenceNumber() = 0;
  virtual bool HandleReplicateResponse(Update *update) = 0;
};

} /* namespace replicator */
