"""Generates tests/golden/writebatch_vectors.json — known-answer vectors for
the WriteBatch rep layout, built with the pure-Python encoder (tests/pywb.py),
independent of the C oracle. Run: python tests/gen_golden_vectors.py
The two simplest vectors are additionally hand-checked hex literals in
tests/test_oracle.py.
"""
import json
import os

from pywb import PyBatch


def vectors():
    v = []

    b = PyBatch().put(b"key1", b"value1")
    v.append({"name": "single_put_seq0", "ops": [["put", "key1", "value1"]], "seq": 0,
              "count": 1, "hex": b.data().hex()})

    b = PyBatch(seq=5).delete(b"k")
    v.append({"name": "delete_seq5", "ops": [["delete", "k"]], "seq": 5,
              "count": 1, "hex": b.data().hex()})

    b = PyBatch().merge(b"counter", b"\x01\x00\x00\x00\x00\x00\x00\x00")
    v.append({"name": "merge_u64_1", "ops": [["merge", "counter", "01"]], "seq": 0,
              "count": 1, "hex": b.data().hex()})

    b = PyBatch().log_data(b"\x11\x22\x33\x44\x55\x66\x77\x88")
    v.append({"name": "logdata_only", "ops": [["log_data", "8B"]], "seq": 0,
              "count": 0, "hex": b.data().hex()})

    # mirrors rocksdb_assumption_test.cpp:179-187 batch shape
    b = (PyBatch().delete(b"key1").put(b"key2", b"value2")
         .put(b"key2", b"value2").merge(b"key1", b"value1"))
    v.append({"name": "assumption_batch4", "ops": "del,put,put,merge", "seq": 0,
              "count": 4, "hex": b.data().hex()})

    # 2-byte varint value length (200 B)
    b = PyBatch().put(b"k" * 16, bytes(range(200 % 256)) + b"")
    v.append({"name": "put_200B_varint2", "ops": [["put", "16B", "200B"]], "seq": 0,
              "count": 1, "hex": b.data().hex()})

    # single delete + delete range
    b = PyBatch(seq=42).single_delete(b"sd").delete_range(b"a", b"z")
    v.append({"name": "sdel_rangedel_seq42", "ops": "sdel,rangedel", "seq": 42,
              "count": 2, "hex": b.data().hex()})

    # column-family-prefixed put (cf 2)
    b = PyBatch().cf_put(2, b"cfkey", b"cfval")
    v.append({"name": "cf2_put", "ops": [["cf_put", 2, "cfkey", "cfval"]], "seq": 0,
              "count": 1, "hex": b.data().hex()})

    # a 1 KB value (config #3 shape) with 16 B key
    b = PyBatch(seq=123456789).put(b"0123456789abcdef", bytes(i % 251 for i in range(1024)))
    v.append({"name": "put_16B_1KB", "ops": [["put", "16B", "1KB"]], "seq": 123456789,
              "count": 1, "hex": b.data().hex()})

    # follower timestamp trailer: what HandleReplicateResponse appends
    # (rocksdb_wrapper.cpp:19-20): LogData(8-byte ts) = tag 0x03 + varint 8 + 8B
    b = PyBatch().put(b"key1", b"value1").log_data((1234567890123).to_bytes(8, "little"))
    v.append({"name": "put_with_ts_trailer", "ops": "put,logdata_ts", "seq": 0,
              "count": 1, "hex": b.data().hex()})

    return v


if __name__ == "__main__":
    here = os.path.dirname(os.path.abspath(__file__))
    out = os.path.join(here, "golden", "writebatch_vectors.json")
    with open(out, "w") as f:
        json.dump(vectors(), f, indent=1)
    print(f"wrote {out}")
