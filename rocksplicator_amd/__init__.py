"""rocksplicator_amd — MI355X-native replication apply path for
pinterest/rocksplicator's follower-side WriteBatch ingest.

The hot path: fbthrift ReplicateResponse payloads (rocksdb
WriteBatch rep blobs, one stream per shard) are staged into HBM and applied
by hand-written CDNA4 HIP kernels — varint record-boundary walk, prefix-sum
placement, partition-copy into per-shard device-resident runs — behind a
C-ABI (include/rocksplicator_gpu.h) that mirrors the reference's
replicator::DbWrapper seam (rocksdb_replicator/db_wrapper.h:6-15).
"""
from .ffi import (  # noqa: F401
    Batch,
    Db,
    Engine,
    Replay,
    gen_stream,
    load,
    GRA_OK,
    GRA_NOT_FOUND,
    GRA_NO_GPU,
    MERGE_CONCAT,
    MERGE_U64ADD,
)

__version__ = "0.1.0"
