/* builder.cpp — framework-native WriteBatch rep builder (gra_wb_*).
 * Produces the byte layout of rocksdb 5.7.fb WriteBatch rep (see
 * wb_format.h header comment); the reference's callers build these with
 * rocksdb::WriteBatch (examples/counter_service/counter_handler.cpp:152-158,
 * performance.cpp:139-142). */
#include <cstdlib>
#include <cstring>
#include <vector>

#include "../../include/rocksplicator_gpu.h"
#include "wb_format.h"

struct GraBatch {
  std::vector<uint8_t> buf;
  uint32_t count = 0;
  GraBatch() : buf(wb::kHeaderBytes, 0) {}
  void set_count(uint32_t c) {
    count = c;
    buf[8] = (uint8_t)c;
    buf[9] = (uint8_t)(c >> 8);
    buf[10] = (uint8_t)(c >> 16);
    buf[11] = (uint8_t)(c >> 24);
  }
  void varint(uint32_t v) {
    while (v >= 0x80) {
      buf.push_back((uint8_t)(v | 0x80));
      v >>= 7;
    }
    buf.push_back((uint8_t)v);
  }
  void slice(const void *p, size_t n) {
    varint((uint32_t)n);
    buf.insert(buf.end(), (const uint8_t *)p, (const uint8_t *)p + n);
  }
};

extern "C" {

GraBatch *gra_wb_create(void) { return new GraBatch(); }
void gra_wb_destroy(GraBatch *b) { delete b; }
void gra_wb_clear(GraBatch *b) {
  b->buf.assign(wb::kHeaderBytes, 0);
  b->count = 0;
}
void gra_wb_put(GraBatch *b, const void *k, size_t kl, const void *v, size_t vl) {
  b->buf.push_back(wb::kValue);
  b->slice(k, kl);
  b->slice(v, vl);
  b->set_count(b->count + 1);
}
void gra_wb_delete(GraBatch *b, const void *k, size_t kl) {
  b->buf.push_back(wb::kDeletion);
  b->slice(k, kl);
  b->set_count(b->count + 1);
}
void gra_wb_single_delete(GraBatch *b, const void *k, size_t kl) {
  b->buf.push_back(wb::kSingleDeletion);
  b->slice(k, kl);
  b->set_count(b->count + 1);
}
void gra_wb_merge(GraBatch *b, const void *k, size_t kl, const void *v, size_t vl) {
  b->buf.push_back(wb::kMerge);
  b->slice(k, kl);
  b->slice(v, vl);
  b->set_count(b->count + 1);
}
void gra_wb_delete_range(GraBatch *b, const void *bk, size_t bkl, const void *ek,
                         size_t ekl) {
  b->buf.push_back(wb::kRangeDeletion);
  b->slice(bk, bkl);
  b->slice(ek, ekl);
  b->set_count(b->count + 1);
}
/* CF-prefixed variants (tags 0x04-0x06/0x08/0x0E: varint32 cf id before
 * the slices). cf_id 0 is the default family — callers use the plain
 * forms for it, as rocksdb does. */
void gra_wb_cf_put(GraBatch *b, uint32_t cf, const void *k, size_t kl,
                   const void *v, size_t vl) {
  b->buf.push_back(wb::kCfValue);
  b->varint(cf);
  b->slice(k, kl);
  b->slice(v, vl);
  b->set_count(b->count + 1);
}
void gra_wb_cf_delete(GraBatch *b, uint32_t cf, const void *k, size_t kl) {
  b->buf.push_back(wb::kCfDeletion);
  b->varint(cf);
  b->slice(k, kl);
  b->set_count(b->count + 1);
}
void gra_wb_cf_single_delete(GraBatch *b, uint32_t cf, const void *k,
                             size_t kl) {
  b->buf.push_back(wb::kCfSingleDeletion);
  b->varint(cf);
  b->slice(k, kl);
  b->set_count(b->count + 1);
}
void gra_wb_cf_merge(GraBatch *b, uint32_t cf, const void *k, size_t kl,
                     const void *v, size_t vl) {
  b->buf.push_back(wb::kCfMerge);
  b->varint(cf);
  b->slice(k, kl);
  b->slice(v, vl);
  b->set_count(b->count + 1);
}
void gra_wb_cf_delete_range(GraBatch *b, uint32_t cf, const void *bk,
                            size_t bkl, const void *ek, size_t ekl) {
  b->buf.push_back(wb::kCfRangeDeletion);
  b->varint(cf);
  b->slice(bk, bkl);
  b->slice(ek, ekl);
  b->set_count(b->count + 1);
}
void gra_wb_put_log_data(GraBatch *b, const void *blob, size_t bl) {
  b->buf.push_back(wb::kLogData);
  b->slice(blob, bl); /* consumes no count */
}
void gra_wb_set_seq(GraBatch *b, uint64_t seq) {
  for (int i = 0; i < 8; i++) b->buf[i] = (uint8_t)(seq >> (8 * i));
}
uint32_t gra_wb_count(const GraBatch *b) { return b->count; }
const uint8_t *gra_wb_data(const GraBatch *b, size_t *len) {
  if (len) *len = b->buf.size();
  return b->buf.data();
}

} /* extern "C" */
