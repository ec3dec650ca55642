"""fbthrift-compatible framing tests: hand-derived byte vectors for the
compact protocol + THeader frame (the pinning artifact — no fbthrift
exists in this container, so the published specs + these vectors are the
parity anchor, exactly like the WriteBatch layout), plus encode/decode
round trips and a live TCP server/client exchange.
"""
import struct
import threading

from rocksplicator_amd import thrift_wire as tw


def test_hand_kat_replicate_request():
    # ReplicateRequest{seq_no=5, db_name=b"db", max_wait_ms=0,
    #                  max_updates=10, role=FOLLOWER(1)} — bytes derived BY
    # HAND from the Thrift compact spec (field delta<<4|type; zigzag varints)
    expect = bytes.fromhex("160a" "18026462" "1500" "1514" "1502" "00")
    got = tw.encode_replicate_request(5, b"db", 0, 10, tw.ROLE_FOLLOWER)
    assert got == expect, got.hex()
    back = tw.decode_replicate_request(tw.Reader(expect))
    assert back == {"seq_no": 5, "db_name": b"db", "max_wait_ms": 0,
                    "max_updates": 10, "role": 1}


def test_hand_kat_update_and_response():
    # Update{raw_data=b"X", timestamp=7, seq_no=3}
    upd = bytes.fromhex("180158" "160e" "1606" "00")
    w = tw.StructWriter()
    tw.encode_update(w, b"X", 7, 3)
    assert bytes(w.out) == upd, bytes(w.out).hex()
    # ReplicateResponse{updates=[^], role=LEADER(2)}
    expect = bytes.fromhex("19" "1c" + upd.hex() + "1504" "00")
    got = tw.encode_replicate_response([(3, 7, b"X")], role=tw.ROLE_LEADER)
    assert got == expect, got.hex()
    back = tw.decode_replicate_response(tw.Reader(expect))
    assert back["role"] == 2
    assert back["updates"] == [{"raw_data": b"X", "timestamp": 7,
                                "seq_no": 3}]


def test_hand_kat_message_envelope():
    # compact message: 0x82 | version 1, CALL<<5 | varint seq 1 |
    # varint len 9 | "replicate" | args{1: struct ...}
    req = tw.encode_replicate_request(5, b"db", 0, 10, tw.ROLE_FOLLOWER)
    expect = (bytes.fromhex("8221" "01" "09") + b"replicate"
              + bytes.fromhex("1c") + req + bytes.fromhex("00"))
    got = tw.encode_call_replicate(1, 5, b"db", 0, 10, tw.ROLE_FOLLOWER)
    assert got == expect, got.hex()
    mt, name, seq, r = tw.decode_message(expect)
    assert (mt, name, seq) == (tw.CALL, "replicate", 1)


def test_hand_kat_theader_frame():
    payload = b"\xAA\xBB"
    f = tw.frame(payload, seq_id=9)
    # header block: varint proto 2, varint ntransforms 0, padded to 4 -> 1 word
    body_expect = (struct.pack(">HHIH", 0x0FFF, 0, 9, 1)
                   + bytes.fromhex("02000000") + payload)
    assert f == struct.pack(">I", len(body_expect)) + body_expect, f.hex()


def test_frame_roundtrip_via_socketpair():
    import socket
    a, b = socket.socketpair()
    try:
        payload = tw.encode_call_replicate(4, 2, b"shard7", 100, 50)
        a.sendall(tw.frame(payload, 4))
        seq, got = tw.read_frame(b)
        assert seq == 4 and got == payload
    finally:
        a.close()
        b.close()


def test_frame_reader_skips_info_keyvalue():
    # a peer may append INFO_KEYVALUE headers; the reader must skip them
    payload = b"\x01\x02\x03"
    hdr = bytearray()
    tw.write_varint(hdr, tw.PROTO_COMPACT)
    tw.write_varint(hdr, 0)
    tw.write_varint(hdr, tw.INFO_KEYVALUE)
    tw.write_varint(hdr, 1)
    for s in (b"client_timeout", b"1000"):
        tw.write_varint(hdr, len(s))
        hdr += s
    while len(hdr) % 4:
        hdr.append(0)
    body = (struct.pack(">HHIH", 0x0FFF, 0, 3, len(hdr) // 4)
            + bytes(hdr) + payload)
    import socket
    a, b = socket.socketpair()
    try:
        a.sendall(struct.pack(">I", len(body)) + body)
        seq, got = tw.read_frame(b)
        assert seq == 3 and got == payload
    finally:
        a.close()
        b.close()


def test_exception_path_roundtrip():
    reply = tw.encode_reply_replicate(7, [], exc=("no such db", 1))
    mt, name, seq, r = tw.decode_message(reply)
    assert (mt, name, seq) == (tw.REPLY, "replicate", 7)
    out = {}

    def f(fid, ct, rr):
        if fid == 1 and ct == tw.CT_STRUCT:
            out.update(tw.decode_replicate_exception(rr))
        else:
            tw.skip_field(rr, ct)

    tw.read_struct_fields(r, f)
    assert out == {"msg": b"no such db", "code": 1}


class _FakeDb:
    """get_updates surface a leader db exposes (wire.py contract)."""

    def __init__(self):
        self.log = []  # (seq, ts, rep)

    def get_updates(self, since, max_updates=50, observer=False):
        ups = [u for u in self.log if u[0] > since]
        return ups[:max_updates] if max_updates else ups


def test_server_client_end_to_end():
    srv = tw.ThriftUpdateServer()
    try:
        db = _FakeDb()
        db.log = [(1, 11, b"rep-one"), (2, 22, b"rep-two"),
                  (3, 33, b"rep-three")]
        srv.register("shard0", db)
        cli = tw.ThriftRemoteUpstream("127.0.0.1", srv.port, "shard0")
        try:
            ups = cli.get_updates(1, 50)
            assert ups == [(2, 22, b"rep-two"), (3, 33, b"rep-three")]
            assert cli.get_updates(3, 50) == []
            # unknown db -> ReplicateException(SOURCE_NOT_FOUND)
            cli2 = tw.ThriftRemoteUpstream("127.0.0.1", srv.port, "nope")
            try:
                try:
                    cli2.get_updates(0, 1)
                    assert False, "expected ReplicateError"
                except tw.ReplicateError as e:
                    assert e.code == 1
            finally:
                cli2.close()
        finally:
            cli.close()
    finally:
        srv.close()


def test_long_poll_wakeup():
    srv = tw.ThriftUpdateServer()
    try:
        db = _FakeDb()
        srv.register("s", db)
        cli = tw.ThriftRemoteUpstream("127.0.0.1", srv.port, "s",
                                      max_wait_ms=3000)
        got = []

        def puller():
            got.extend(cli.get_updates(0, 10))

        t = threading.Thread(target=puller)
        t.start()
        import time
        time.sleep(0.15)  # puller is parked in the long poll
        db.log.append((1, 5, b"woke"))
        srv.notify_write()
        t.join(timeout=5)
        assert got == [(1, 5, b"woke")]
        cli.close()
    finally:
        srv.close()


def test_long_form_list_roundtrip():
    """>=15 updates exercises the compact long-form list header
    (0xF0|etype + varint size) — the shape every full pull response
    (max_updates=50) uses."""
    ups = [(i + 1, 10 * i, bytes([i]) * (i + 1)) for i in range(50)]
    enc = tw.encode_replicate_response(ups, role=tw.ROLE_LEADER)
    assert enc[1] == 0xF0 | tw.CT_STRUCT  # long-form list header
    back = tw.decode_replicate_response(tw.Reader(enc))
    assert [(u["seq_no"], u["timestamp"], u["raw_data"])
            for u in back["updates"]] == ups


def test_large_binary_and_field_long_form():
    # 70 KB raw_data (3-byte varint length) + a long-form field id jump
    big = bytes(range(256)) * 280
    w = tw.StructWriter()
    w.begin_struct()
    w.binary(1, big)
    w.i64(200, 7)  # delta > 15 -> long-form header with zigzag fid
    w.end_struct()
    got = {}

    def f(fid, ct, rr):
        if fid == 1:
            got["b"] = rr.binary()
        elif fid == 200:
            got["x"] = rr.zz()
        else:
            tw.skip_field(rr, ct)

    tw.read_struct_fields(tw.Reader(bytes(w.out)), f)
    assert got == {"b": big, "x": 7}


def test_transport_equivalence_with_framework_wire():
    """The thrift framing and the framework's length-prefixed framing must
    deliver IDENTICAL update streams from the same db (transport choice
    cannot change replication content)."""
    from rocksplicator_amd import wire

    db = _FakeDb()
    db.log = [(i + 1, 100 + i, bytes([i % 251]) * (i % 37 + 1))
              for i in range(120)]
    tsrv = tw.ThriftUpdateServer()
    wsrv = wire.UpdateServer()
    try:
        tsrv.register("d", db)
        wsrv.register("d", db)
        tcli = tw.ThriftRemoteUpstream("127.0.0.1", tsrv.port, "d")
        wcli = wire.RemoteUpstream("127.0.0.1", wsrv.port, "d")
        try:
            for since in (0, 1, 57, 119, 120, 500):
                for mx in (1, 14, 50, 0):
                    a = tcli.get_updates(since, mx)
                    b = wcli.get_updates(since, mx)
                    assert a == b, (since, mx)
        finally:
            tcli.close()
            wcli.close()
    finally:
        tsrv.close()
        wsrv.close()
