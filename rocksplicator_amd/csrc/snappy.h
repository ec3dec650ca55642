/* snappy.h — Snappy raw-block codec, product restatement (host + device).
 *
 * Format (public Snappy format description): varint32 uncompressed length,
 * then elements tagged by the low 2 bits of the first byte —
 *   00 literal (len-1 in bits 2..7; 60..63 => 1..4 extra LE length bytes)
 *   01 copy, 11-bit offset (len-4 in bits 2..4, offset = bits5..7 <<8 | byte)
 *   10 copy, 2-byte LE offset (len-1 in bits 2..7)
 *   11 copy, 4-byte LE offset (len-1 in bits 2..7)
 * Copies may overlap (offset < len repeats the pattern).
 *
 * Config #5: Update payloads are Snappy-compressed in transit; the follower
 * decompresses before the WriteBatch decode (GPU stage k_snappy). The
 * oracle's codec (oracle/wb_oracle.c) is an independent restatement used as
 * the checker.
 */
#pragma once
#include <stdint.h>
#ifdef __HIP__
#include <hip/hip_runtime.h> /* uint4 for the cooperative device variant */
#endif

#include "wb_format.h" /* WB_HD */

namespace snp {

WB_HD uint32_t max_compressed_len(uint32_t n) { return 32 + n + n / 6; }

/* Decompress src[0..slen) into dst (capacity dcap). Returns uncompressed
 * length, or UINT32_MAX on corruption. dst regions may be written in wide
 * chunks up to 15 bytes past the uncompressed length — callers provide
 * >= 16 bytes of slack per output slot. */
WB_HD uint32_t decompress(const uint8_t *src, uint32_t slen, uint8_t *dst,
                          uint32_t dcap) {
  uint32_t ulen = 0;
  uint32_t ip = wb::varint32(src, slen, &ulen);
  if (ip == 0 || ulen > dcap) return UINT32_MAX;
  uint32_t op = 0;
  while (ip < slen) {
    uint8_t tag = src[ip++];
    if ((tag & 3) == 0) { /* literal */
      uint32_t len = (tag >> 2) + 1;
      if (__builtin_expect(len > 60, 0)) {
        /* extended length: math + bounds in 64-bit — a 4-extra-byte length
         * near 2^32 must FAIL the check, not wrap it (u32 ip+len / op+len
         * would pass and the copy loop would scribble ~4 GB OOB) */
        uint32_t nb = len - 60;
        if ((uint64_t)ip + nb > slen) return UINT32_MAX;
        uint64_t len64 = 0;
        for (uint32_t b = 0; b < nb; b++)
          len64 |= (uint64_t)src[ip + b] << (8 * b);
        len64 += 1;
        ip += nb;
        if ((uint64_t)ip + len64 > slen || (uint64_t)op + len64 > ulen)
          return UINT32_MAX;
        len = (uint32_t)len64; /* checked: fits the remaining space */
      } else if (ip + len > slen || op + len > ulen) {
        /* u32 is exact here: len <= 61 and ip <= slen, op <= ulen, so a
         * wrap would need slen/ulen > 2^32-62 — i.e. >4 GB of real backing
         * per slot, within which any access stays in-arena anyway */
        return UINT32_MAX;
      }
      /* literals never overlap: 16-byte chunks (arena + slot slack cover
       * the over-read/over-write; gfx950 tolerates misaligned dwordx4) */
      for (uint32_t b = 0; b < len; b += 16) {
#if defined(__HIP_DEVICE_COMPILE__) && defined(WB_UNALIGNED_OK)
        *(uint4 *)(dst + op + b) = *(const uint4 *)(src + ip + b);
#else
        for (uint32_t j = b; j < b + 16 && j < len; j++) dst[op + j] = src[ip + j];
#endif
      }
      ip += len;
      op += len;
    } else {
      uint32_t len, off;
      if ((tag & 3) == 1) {
        len = ((tag >> 2) & 7) + 4;
        if (ip + 1 > slen) return UINT32_MAX;
        off = ((uint32_t)(tag >> 5) << 8) | src[ip];
        ip += 1;
      } else if ((tag & 3) == 2) {
        len = (tag >> 2) + 1;
        if (ip + 2 > slen) return UINT32_MAX;
        off = (uint32_t)src[ip] | ((uint32_t)src[ip + 1] << 8);
        ip += 2;
      } else {
        len = (tag >> 2) + 1;
        if (ip + 4 > slen) return UINT32_MAX;
        off = (uint32_t)src[ip] | ((uint32_t)src[ip + 1] << 8) |
              ((uint32_t)src[ip + 2] << 16) | ((uint32_t)src[ip + 3] << 24);
        ip += 4;
      }
      /* u32 exact: len <= 64 from the tag; wrap needs ulen > 2^32-65 */
      if (off == 0 || off > op || op + len > ulen) return UINT32_MAX;
      if (off >= 16) { /* chunked forward copy is overlap-safe at off>=chunk */
        for (uint32_t b = 0; b < len; b += 16) {
#if defined(__HIP_DEVICE_COMPILE__) && defined(WB_UNALIGNED_OK)
          *(uint4 *)(dst + op + b) = *(const uint4 *)(dst + op - off + b);
#else
          for (uint32_t j = b; j < b + 16 && j < len; j++)
            dst[op + j] = dst[op + j - off];
#endif
        }
      } else if (off >= 8) {
        for (uint32_t b = 0; b < len; b += 8) {
#if defined(__HIP_DEVICE_COMPILE__) && defined(WB_UNALIGNED_OK)
          *(uint64_t *)(dst + op + b) = *(const uint64_t *)(dst + op - off + b);
#else
          for (uint32_t j = b; j < b + 8 && j < len; j++)
            dst[op + j] = dst[op + j - off];
#endif
        }
      } else {
        for (uint32_t b = 0; b < len; b++) dst[op + b] = dst[op + b - off];
      }
      op += len;
    }
  }
  return op == ulen ? op : UINT32_MAX;
}

/* Host-side greedy compressor (leader/transport side; any valid Snappy
 * stream is acceptable — decoders define the format contract). Plain inline
 * (host-only in HIP mode; parsed but never codegen'd for device). */
inline uint32_t compress(const uint8_t *src, uint32_t slen, uint8_t *dst,
                         uint32_t dcap) {
  if (dcap < max_compressed_len(slen)) return 0;
  uint32_t o = 0;
  { /* varint32 uncompressed length */
    uint32_t v = slen;
    while (v >= 0x80) {
      dst[o++] = (uint8_t)(v | 0x80);
      v >>= 7;
    }
    dst[o++] = (uint8_t)v;
  }
  auto emit_lit = [&](uint32_t start, uint32_t len) {
    uint32_t n = len - 1;
    if (n < 60) {
      dst[o++] = (uint8_t)(n << 2);
    } else if (n < (1u << 8)) {
      dst[o++] = 60 << 2;
      dst[o++] = (uint8_t)n;
    } else if (n < (1u << 16)) {
      dst[o++] = 61 << 2;
      dst[o++] = (uint8_t)n;
      dst[o++] = (uint8_t)(n >> 8);
    } else {
      dst[o++] = 62 << 2;
      dst[o++] = (uint8_t)n;
      dst[o++] = (uint8_t)(n >> 8);
      dst[o++] = (uint8_t)(n >> 16);
    }
    __builtin_memcpy(dst + o, src + start, len);
    o += len;
  };
  constexpr uint32_t HBITS = 13, HSIZE = 1u << HBITS;
  uint32_t tab[HSIZE];
  __builtin_memset(tab, 0xFF, sizeof(tab));
  uint32_t pos = 0, lit = 0;
  while (pos + 4 <= slen) {
    uint32_t cur;
    __builtin_memcpy(&cur, src + pos, 4);
    uint32_t h = (cur * 0x1e35a7bdu) >> (32 - HBITS);
    uint32_t cand = tab[h];
    tab[h] = pos;
    uint32_t c4 = 0;
    if (cand != UINT32_MAX && pos - cand <= 0xFFFF) __builtin_memcpy(&c4, src + cand, 4);
    if (cand != UINT32_MAX && pos - cand <= 0xFFFF && c4 == cur) {
      if (pos > lit) emit_lit(lit, pos - lit);
      uint32_t len = 4, maxlen = slen - pos;
      if (maxlen > 64) maxlen = 64;
      while (len < maxlen && src[cand + len] == src[pos + len]) len++;
      uint32_t off = pos - cand;
      dst[o++] = (uint8_t)(((len - 1) << 2) | 2);
      dst[o++] = (uint8_t)off;
      dst[o++] = (uint8_t)(off >> 8);
      pos += len;
      lit = pos;
    } else {
      pos++;
    }
  }
  if (slen > lit) emit_lit(lit, slen - lit);
  return o;
}
} /* namespace snp */

#ifdef __HIP__
namespace snp {

/* Cooperative decompress: 16 lanes share ONE stream. Every lane parses the
 * element headers redundantly (identical data -> converged control flow
 * within the group; cross-group divergence in a wave drops from 64-way to
 * 4-way), and the element payload moves with 16 parallel 16-byte chunks
 * (coalesced within the output slot) instead of one lane's serial loop.
 * Overlapping copies (off < len) read only bytes before the element start
 * via the period-off pattern, so chunk order doesn't matter.
 * Same format/validation as snp::decompress; same >=16 B slot slack. */
__device__ inline uint32_t decompress_coop16(const uint8_t *__restrict__ src,
                                             uint32_t slen,
                                             uint8_t *__restrict__ dst,
                                             uint32_t dcap, uint32_t lane) {
  uint32_t ulen = 0;
  uint32_t ip = wb::varint32(src, slen, &ulen);
  if (ip == 0 || ulen > dcap) return UINT32_MAX;
  uint32_t op = 0;
  while (ip < slen) {
    uint8_t tag = src[ip++];
    if ((tag & 3) == 0) { /* literal */
      uint32_t len = (tag >> 2) + 1;
      if (__builtin_expect(len > 60, 0)) {
        /* extended length in 64-bit — same overflow hazard as decompress() */
        uint32_t nb = len - 60;
        if ((uint64_t)ip + nb > slen) return UINT32_MAX;
        uint64_t len64 = 0;
        for (uint32_t b = 0; b < nb; b++)
          len64 |= (uint64_t)src[ip + b] << (8 * b);
        len64 += 1;
        ip += nb;
        if ((uint64_t)ip + len64 > slen || (uint64_t)op + len64 > ulen)
          return UINT32_MAX;
        len = (uint32_t)len64;
      } else if (ip + len > slen || op + len > ulen) {
        return UINT32_MAX; /* u32 exact: len <= 61 (see decompress()) */
      }
      for (uint32_t b = lane * 16; b < len; b += 16 * 16)
        *(uint4 *)(dst + op + b) = *(const uint4 *)(src + ip + b);
      ip += len;
      op += len;
    } else {
      uint32_t len, off;
      if ((tag & 3) == 1) {
        len = ((tag >> 2) & 7) + 4;
        if (ip + 1 > slen) return UINT32_MAX;
        off = ((uint32_t)(tag >> 5) << 8) | src[ip];
        ip += 1;
      } else if ((tag & 3) == 2) {
        len = (tag >> 2) + 1;
        if (ip + 2 > slen) return UINT32_MAX;
        off = (uint32_t)src[ip] | ((uint32_t)src[ip + 1] << 8);
        ip += 2;
      } else {
        len = (tag >> 2) + 1;
        if (ip + 4 > slen) return UINT32_MAX;
        off = (uint32_t)src[ip] | ((uint32_t)src[ip + 1] << 8) |
              ((uint32_t)src[ip + 2] << 16) | ((uint32_t)src[ip + 3] << 24);
        ip += 4;
      }
      if (off == 0 || off > op || op + len > ulen) return UINT32_MAX;
      if (off >= len) { /* no overlap: parallel 16B chunks */
        for (uint32_t b = lane * 16; b < len; b += 16 * 16)
          *(uint4 *)(dst + op + b) = *(const uint4 *)(dst + op - off + b);
      } else { /* periodic pattern: read only pre-element bytes */
        for (uint32_t j = lane; j < len; j += 16)
          dst[op + j] = dst[op - off + (j % off)];
      }
      op += len;
    }
  }
  return op == ulen ? op : UINT32_MAX;
}

} /* namespace snp */
#endif /* __HIP__ */
