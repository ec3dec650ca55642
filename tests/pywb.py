"""Pure-Python WriteBatch rep encoder — an independent restatement of the
rocksdb 5.7.fb rep layout (db/write_batch.cc, un-vendored third-party dep of
the reference), used only to generate/check golden vectors
against the C oracle. TEST INFRASTRUCTURE ONLY.

Layout: 8-byte fixed64 LE seq + 4-byte fixed32 LE count, then records of
tag(1B) + varint32-length-prefixed slices. LogData consumes no count.
"""
import struct

TYPE_DELETION = 0x00
TYPE_VALUE = 0x01
TYPE_MERGE = 0x02
TYPE_LOGDATA = 0x03
TYPE_CF_DELETION = 0x04
TYPE_CF_VALUE = 0x05
TYPE_CF_MERGE = 0x06
TYPE_SINGLE_DELETION = 0x07
TYPE_CF_SINGLE_DELETION = 0x08
TYPE_NOOP = 0x0D
TYPE_CF_RANGE_DELETION = 0x0E
TYPE_RANGE_DELETION = 0x0F


def varint32(v):
    out = bytearray()
    while v >= 0x80:
        out.append((v & 0x7F) | 0x80)
        v >>= 7
    out.append(v)
    return bytes(out)


def lps(b):
    return varint32(len(b)) + b


class PyBatch:
    def __init__(self, seq=0):
        self.seq = seq
        self.count = 0
        self.body = b""

    def put(self, k, v):
        self.body += bytes([TYPE_VALUE]) + lps(k) + lps(v)
        self.count += 1
        return self

    def delete(self, k):
        self.body += bytes([TYPE_DELETION]) + lps(k)
        self.count += 1
        return self

    def single_delete(self, k):
        self.body += bytes([TYPE_SINGLE_DELETION]) + lps(k)
        self.count += 1
        return self

    def merge(self, k, v):
        self.body += bytes([TYPE_MERGE]) + lps(k) + lps(v)
        self.count += 1
        return self

    def delete_range(self, bk, ek):
        self.body += bytes([TYPE_RANGE_DELETION]) + lps(bk) + lps(ek)
        self.count += 1
        return self

    def log_data(self, blob):
        self.body += bytes([TYPE_LOGDATA]) + lps(blob)
        return self

    def cf_put(self, cf, k, v):
        self.body += bytes([TYPE_CF_VALUE]) + varint32(cf) + lps(k) + lps(v)
        self.count += 1
        return self

    def cf_delete(self, cf, k):
        self.body += bytes([TYPE_CF_DELETION]) + varint32(cf) + lps(k)
        self.count += 1
        return self

    def cf_merge(self, cf, k, v):
        self.body += bytes([TYPE_CF_MERGE]) + varint32(cf) + lps(k) + lps(v)
        self.count += 1
        return self

    def cf_single_delete(self, cf, k):
        self.body += bytes([TYPE_CF_SINGLE_DELETION]) + varint32(cf) + lps(k)
        self.count += 1
        return self

    def cf_delete_range(self, cf, bk, ek):
        self.body += (bytes([TYPE_CF_RANGE_DELETION]) + varint32(cf)
                      + lps(bk) + lps(ek))
        self.count += 1
        return self

    def noop(self):
        self.body += bytes([TYPE_NOOP])  # consumes no seq
        return self

    def begin_prepare(self):
        self.body += bytes([0x09])  # marker, no payload, no seq
        return self

    def commit_xid(self, xid):
        self.body += bytes([0x0B]) + lps(xid)  # xid slice, no seq
        return self

    def data(self):
        return struct.pack("<QI", self.seq, self.count) + self.body
