"""Kafka consumer/watcher shim tests (CPU): the consume loop, offset
resume, at-least-once redelivery dedup and commit cadence over the
in-memory broker — with a fake ingestor so no GPU is needed. The GPU end
of the modality (engine apply parity) is test_kafka_ingest_modality and
test_kafka_watcher_end_to_end in test_replication_chain.py.
"""
import threading
import time

from rocksplicator_amd.kafka_consumer import (InMemoryBroker,
                                              InMemoryConsumer,
                                              KafkaWatcher, RdKafkaConsumer)


class FakeIngestor:
    """kafka_ingest.KafkaIngestor's contract without an engine."""

    def __init__(self, partitions):
        self._committed = {p: -1 for p in partitions}
        self._mu = threading.Lock()
        self.applied = []

    def consume(self, partition, offset, payload, ts=0):
        with self._mu:
            if offset <= self._committed[partition]:
                return False
            assert offset == self._committed[partition] + 1
            self._committed[partition] = offset
            self.applied.append((partition, offset, payload))
            return True

    def checkpoint(self):
        with self._mu:
            return dict(self._committed)

    def flush(self):
        pass


def test_consumer_roundrobin_and_seek():
    b = InMemoryBroker()
    for p in (0, 1):
        for i in range(5):
            b.produce("t", p, f"p{p}m{i}".encode())
    c = InMemoryConsumer(b, "t", [0, 1])
    got = [c.consume(100) for _ in range(10)]
    assert all(m is not None for m in got)
    per = {0: [], 1: []}
    for m in got:
        per[m.partition].append(m.offset)
    assert per[0] == list(range(5)) and per[1] == list(range(5))
    c.seek({0: 3})
    m = c.consume(100)
    while m and m.partition != 0:
        m = c.consume(100)
    assert m and m.offset == 3


def test_watcher_applies_and_resumes():
    b = InMemoryBroker()
    parts = [0, 1, 2]
    for i in range(30):
        b.produce("t", i % 3, f"m{i}".encode(), timestamp=i)
    ing = FakeIngestor(parts)
    c = InMemoryConsumer(b, "t", parts)
    w = KafkaWatcher(c, ing, commit_every=4, poll_ms=50)
    w.start()
    deadline = time.monotonic() + 5
    while len(ing.applied) < 30 and time.monotonic() < deadline:
        time.sleep(0.02)
    w.stop()
    assert len(ing.applied) == 30
    assert ing.checkpoint() == {0: 9, 1: 9, 2: 9}
    # live production while running
    w2 = KafkaWatcher(InMemoryConsumer(b, "t", parts), ing, poll_ms=50)
    w2.start()
    for i in range(30, 36):
        b.produce("t", i % 3, f"m{i}".encode())
    deadline = time.monotonic() + 5
    while len(ing.applied) < 36 and time.monotonic() < deadline:
        time.sleep(0.02)
    w2.stop()
    assert ing.checkpoint() == {0: 11, 1: 11, 2: 11}


def test_watcher_restart_redelivery_is_idempotent():
    """Restart from a STALE consumer position (at-least-once): the
    ingestor's dedup absorbs redelivered offsets."""
    b = InMemoryBroker()
    for i in range(12):
        b.produce("t", 0, f"m{i}".encode())
    ing = FakeIngestor([0])
    c = InMemoryConsumer(b, "t", [0])
    w = KafkaWatcher(c, ing, poll_ms=50)
    w.start()
    deadline = time.monotonic() + 5
    while len(ing.applied) < 12 and time.monotonic() < deadline:
        time.sleep(0.02)
    w.stop()
    # new watcher, consumer seeked back by the checkpoint (simulates a
    # crash after apply but before broker commit): re-delivery happens,
    # nothing re-applies
    c2 = InMemoryConsumer(b, "t", [0])
    c2.seek({0: 6})  # stale: will redeliver 6..11 before the watcher seek
    w2 = KafkaWatcher(c2, ing, poll_ms=50)
    w2.start()  # start() seeks to checkpoint+1 = 12 -> nothing new
    time.sleep(0.3)
    w2.stop()
    assert len(ing.applied) == 12
    assert ing.checkpoint()[0] == 11


def test_rdkafka_consumer_absent_is_loud():
    try:
        RdKafkaConsumer("localhost:9092", "t", [0], "g")
        raise AssertionError("expected RuntimeError (no confluent_kafka)")
    except RuntimeError as e:
        assert "confluent_kafka" in str(e)


def test_watcher_commit_cadence_and_final_commit():
    """The watcher commits broker offsets every `commit_every` applies and
    once more on stop — a restart from the BROKER's committed position
    then redelivers at most `commit_every-1` messages (all deduped)."""
    b = InMemoryBroker()
    for i in range(10):
        b.produce("t", 0, f"m{i}".encode())
    ing = FakeIngestor([0])
    c = InMemoryConsumer(b, "t", [0])
    w = KafkaWatcher(c, ing, commit_every=4, poll_ms=50)
    w.start()
    deadline = time.monotonic() + 5
    while len(ing.applied) < 10 and time.monotonic() < deadline:
        time.sleep(0.02)
    w.stop()
    # broker-side committed offset reflects the final commit on stop
    assert c.committed()[0] == 9
    assert w.applied == 10 and w.duplicates == 0
