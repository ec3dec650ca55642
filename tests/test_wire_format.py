"""Wire-framing known-answer tests (CPU): the ReplicateRequest/Update/
ReplicateResponse triple restated from replicator.thrift:21-70."""
import io
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from rocksplicator_amd import wire


class FakeSock:
    def __init__(self, data):
        self.b = io.BytesIO(data)

    def recv(self, n):
        return self.b.read(n)


def test_wire_format_roundtrip():
    """Framing known-answer: request/response encode/decode symmetry."""
    req = wire.encode_request(12345, "shard_07", 10000, 50, wire.ROLE_FOLLOWER)
    seq, name, wait, maxu, role = wire.decode_request(FakeSock(req))
    assert (seq, name, wait, maxu, role) == (12345, "shard_07", 10000, 50, 0)
    ups = [(1, 99, b"\x00" * 12), (2, 100, b"payload-bytes")]
    frame = wire.encode_response(ups)
    assert wire.decode_response(FakeSock(frame)) == ups




def test_wire_server_concurrent_pullers_no_gaps():
    """CPU-only stress of the transport: 4 concurrent pullers long-polling
    one UpdateServer over a mock in-memory log while a writer appends.
    Every puller must see the complete, gapless, in-order stream (the
    framing and the long-poll wakeup must not drop or reorder batches)."""
    import threading
    import time

    from rocksplicator_amd import wire

    class MockDb:
        def __init__(self):
            self.mu = threading.Lock()
            self.log = []  # (seq, ts, rep)

        def append(self, seq, rep):
            with self.mu:
                self.log.append((seq, -1, rep))

        def get_updates(self, since, max_updates, observer=False):
            with self.mu:
                out = [u for u in self.log if u[0] > since]
            return out[:max_updates]

    db = MockDb()
    srv = wire.UpdateServer()
    srv.register("shard0", db)

    TOTAL = 300
    results = {}

    def puller(idx):
        got = []
        cli = wire.RemoteUpstream("127.0.0.1", srv.port, "shard0")
        try:
            while (not got or got[-1][0] < TOTAL):
                since = got[-1][0] if got else 0
                ups = cli.get_updates(since, 40)
                got.extend(ups)
        finally:
            cli.close()
        results[idx] = got

    threads = [threading.Thread(target=puller, args=(i,)) for i in range(4)]
    for t in threads:
        t.start()
    for seq in range(1, TOTAL + 1):
        db.append(seq, b"rep%d" % seq)
        srv.notify_write()
        if seq % 50 == 0:
            time.sleep(0.005)  # let long-pollers drain
    for t in threads:
        t.join(timeout=30)
        assert not t.is_alive(), "puller hung"
    for idx, got in results.items():
        assert [u[0] for u in got] == list(range(1, TOTAL + 1)), idx
        assert all(rep == b"rep%d" % seq for seq, _ts, rep in got)
    srv.close()
