"""Kafka-modality ingest — the second producer of
WriteBatch payloads, re-imagined over the same engine path the replication
ingest uses (common/kafka's consumer pool + rocksdb_admin's
StartMessageIngestion, admin_handler.h:126-127, feed messages whose values
are serialized WriteBatches; the watcher tracks per-partition offsets so
restarts resume without re-applying).

This adapter provides the offset bookkeeping / dedup / resume semantics
over `gra_handle_replicate_response`; an actual librdkafka consumer would
sit in front of `consume()` (no Kafka broker exists in this environment —
the modality's apply-side contract is what's implemented and tested).
"""
import threading


class KafkaIngestor:
    """One topic's ingestion into an engine: partition -> shard mapping,
    monotonic offset tracking with at-least-once dedup, checkpointable."""

    def __init__(self, engine, partition_to_shard):
        self.engine = engine
        self._mu = threading.Lock()
        self._dbs = {}
        self._committed = {}  # partition -> highest applied offset
        for part, shard in partition_to_shard.items():
            self._dbs[part] = engine.open(shard)
            self._committed[part] = -1

    def consume(self, partition, offset, payload, ts=0):
        """Apply one message. Returns True if applied, False if it was a
        duplicate (offset <= committed) or the apply was refused (caller
        should back off and re-consume from committed_offset()+1)."""
        db = self._dbs[partition]
        with self._mu:
            if offset <= self._committed[partition]:
                return False  # duplicate delivery (at-least-once)
            if offset != self._committed[partition] + 1:
                raise ValueError(
                    f"offset gap on partition {partition}: "
                    f"expected {self._committed[partition] + 1}, got {offset}")
            if not db.handle_replicate_response(payload, ts):
                return False
            self._committed[partition] = offset
        return True

    def committed_offset(self, partition):
        with self._mu:
            return self._committed[partition]

    def checkpoint(self):
        """Offsets to persist (resume = re-create ingestor, then
        restore_checkpoint; replay from offset+1 is idempotent by dedup)."""
        with self._mu:
            return dict(self._committed)

    def restore_checkpoint(self, committed):
        with self._mu:
            self._committed.update(committed)

    def flush(self):
        self.engine.flush()

    def close(self):
        for db in self._dbs.values():
            db.close()
