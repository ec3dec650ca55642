"""Kafka consumer shim (SURVEY §8f row f4's missing half, VERDICT r01
missing-#4): the consumer/watcher pair that sits in FRONT of
`kafka_ingest.KafkaIngestor`, shaped after the reference's
common/kafka stack — `kafka::KafkaConsumer` (Consume/Seek/Commit,
kafka_consumer.h:27-80) driven by a `KafkaWatcher` loop
(kafka_watcher.h:39's message callback) as wired by
`StartMessageIngestion` (admin_handler.h:126-127).

Two consumer implementations behind one interface:
  * InMemoryConsumer / InMemoryBroker — a broker in this process, used by
    the tests (no Kafka broker exists in this environment);
  * RdKafkaConsumer — the librdkafka-backed one via confluent_kafka,
    constructed lazily and raising a clear error when the library is
    absent (it is not installed in this image; the class exists so a
    deployment with librdkafka wires in without touching the watcher).
"""
import threading
import time


class Message:
    """≅ RdKafka::Message surface the watcher uses."""

    __slots__ = ("topic", "partition", "offset", "timestamp", "value")

    def __init__(self, topic, partition, offset, timestamp, value):
        self.topic = topic
        self.partition = partition
        self.offset = offset
        self.timestamp = timestamp
        self.value = value


class InMemoryBroker:
    """Per-(topic, partition) append-only logs + a condition for long
    polls — the test stand-in for a Kafka cluster."""

    def __init__(self):
        self._logs = {}
        self._cond = threading.Condition()

    def produce(self, topic, partition, value, timestamp=0):
        with self._cond:
            log = self._logs.setdefault((topic, partition), [])
            off = len(log)
            log.append(Message(topic, partition, off, timestamp, value))
            self._cond.notify_all()
            return off

    def fetch(self, topic, partition, offset, timeout_s):
        deadline = time.monotonic() + timeout_s
        with self._cond:
            while True:
                log = self._logs.get((topic, partition), [])
                if offset < len(log):
                    return log[offset]
                rem = deadline - time.monotonic()
                if rem <= 0:
                    return None
                self._cond.wait(rem)

    def wait_any(self, topic, wants, timeout_s):
        """Block until ANY (partition, next_offset) in `wants` has data, or
        timeout. Returns True if something is available."""
        deadline = time.monotonic() + timeout_s
        with self._cond:
            while True:
                for p, off in wants.items():
                    if off < len(self._logs.get((topic, p), [])):
                        return True
                rem = deadline - time.monotonic()
                if rem <= 0:
                    return False
                self._cond.wait(rem)

    def high_watermark(self, topic, partition):
        with self._cond:
            return len(self._logs.get((topic, partition), []))


class InMemoryConsumer:
    """≅ kafka::KafkaConsumer over the in-memory broker: one consumer per
    partition set, round-robin Consume across assigned partitions,
    Seek-to-offsets, committed-offset tracking."""

    def __init__(self, broker, topic, partition_ids):
        self._broker = broker
        self._topic = topic
        self._parts = sorted(partition_ids)
        self._next = {p: 0 for p in self._parts}
        self._committed = {p: -1 for p in self._parts}
        self._rr = 0

    def is_healthy(self):
        return True

    def seek(self, offsets):
        """offsets: {partition: next offset to consume} ≅ Seek(last_offsets)"""
        for p, off in offsets.items():
            if p in self._next:
                self._next[p] = off

    def consume(self, timeout_ms):
        deadline = time.monotonic() + timeout_ms / 1e3
        while True:
            for _ in range(len(self._parts)):
                p = self._parts[self._rr % len(self._parts)]
                self._rr += 1
                m = self._broker.fetch(self._topic, p, self._next[p], 0)
                if m is not None:
                    self._next[p] = m.offset + 1
                    return m
            rem = deadline - time.monotonic()
            if rem <= 0:
                return None
            # park until ANY assigned partition has data (a single-partition
            # wait could sleep through another partition's arrival)
            self._broker.wait_any(self._topic, dict(self._next), rem)

    def commit(self, message):
        self._committed[message.partition] = message.offset

    def committed(self):
        return dict(self._committed)


class RdKafkaConsumer:
    """librdkafka-backed consumer via confluent_kafka, same surface.
    Not constructible in this image (no broker, no library) — the point
    is that a real deployment swaps it in without touching KafkaWatcher."""

    def __init__(self, broker_list, topic, partition_ids, group_id):
        try:
            import confluent_kafka  # noqa: F401
        except ImportError as e:
            raise RuntimeError(
                "confluent_kafka (librdkafka) is not installed in this "
                "environment; use InMemoryConsumer for tests or install "
                "the library in deployment") from e
        from confluent_kafka import Consumer, TopicPartition
        self._TopicPartition = TopicPartition
        self._topic = topic
        self._c = Consumer({
            "bootstrap.servers": broker_list,
            "group.id": group_id,
            "enable.auto.commit": False,
            "auto.offset.reset": "earliest",
        })
        self._c.assign([TopicPartition(topic, p) for p in partition_ids])

    def is_healthy(self):
        return True

    def seek(self, offsets):
        for p, off in offsets.items():
            self._c.seek(self._TopicPartition(self._topic, p, off))

    def consume(self, timeout_ms):
        m = self._c.poll(timeout_ms / 1e3)
        if m is None or m.error():
            return None
        return Message(m.topic(), m.partition(), m.offset(),
                       (m.timestamp() or (0, 0))[1], m.value())

    def commit(self, message):
        self._c.commit(offsets=[self._TopicPartition(
            message.topic, message.partition, message.offset + 1)],
            asynchronous=True)


class KafkaWatcher:
    """≅ KafkaWatcher's consume loop feeding StartMessageIngestion's
    callback: pulls messages, applies via the ingestor, commits applied
    offsets, resumes from the ingestor's checkpoint on start (replays
    from committed+1; the ingestor's dedup makes redelivery idempotent)."""

    def __init__(self, consumer, ingestor, commit_every=64, poll_ms=200):
        self._consumer = consumer
        self._ingestor = ingestor
        self._commit_every = commit_every
        self._poll_ms = poll_ms
        self._stop = threading.Event()
        self._thread = None
        self.applied = 0
        self.duplicates = 0

    def start(self):
        # resume: consume from the ingestor's committed offsets + 1
        self._consumer.seek({p: off + 1 for p, off in
                             self._ingestor.checkpoint().items()})
        self._stop.clear()
        self._thread = threading.Thread(target=self._loop, daemon=True,
                                        name="kafka-watcher")
        self._thread.start()

    def _loop(self):
        since_commit = 0
        last = None
        while not self._stop.is_set():
            m = self._consumer.consume(self._poll_ms)
            if m is None:
                continue
            if self._ingestor.consume(m.partition, m.offset, m.value,
                                      m.timestamp):
                self.applied += 1
            else:
                self.duplicates += 1
            last = m
            since_commit += 1
            if since_commit >= self._commit_every:
                self._consumer.commit(last)
                since_commit = 0
        if last is not None:
            self._consumer.commit(last)

    def stop(self):
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=10)
            self._thread = None
        self._ingestor.flush()
