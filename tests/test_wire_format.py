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


