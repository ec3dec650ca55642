/* micro_copy.hip — A/B copy-variant microbench for the partition-copy kernel.
 * Tests correctness + bandwidth of byte-granular copies with misaligned
 * sources (the apply path's payload gather): dword funnel vs unaligned
 * vector loads, at several lane-group widths.
 * Build: hipcc --offload-arch=gfx950 -O3 scripts/micro_copy.hip -o build/micro_copy
 */
#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <vector>

#define CHECK(x)                                                      \
  do {                                                                \
    hipError_t e = (x);                                               \
    if (e != hipSuccess) {                                            \
      printf("ERR %s: %s\n", #x, hipGetErrorString(e));               \
      exit(1);                                                        \
    }                                                                 \
  } while (0)

struct Task {
  uint64_t src_off;
  uint32_t dst_off;
  uint32_t nbytes;
};

/* v0: dword funnel (current engine implementation), group G */
template <int G>
__device__ void copy_v0(uint8_t *dst, const uint8_t *src, uint32_t n, uint32_t lane) {
  uint32_t head = (uint32_t)((0u - (uint32_t)(uintptr_t)dst) & 3u);
  if (head > n) head = n;
  if (lane < head) dst[lane] = src[lane];
  dst += head;
  src += head;
  n -= head;
  uint32_t nw = n >> 2;
  uint32_t r = (uint32_t)((uintptr_t)src & 3);
  const uint32_t *asrc = (const uint32_t *)(src - r);
  uint32_t *adst = (uint32_t *)dst;
  if (r == 0) {
    for (uint32_t w = lane; w < nw; w += G) adst[w] = asrc[w];
  } else {
    uint32_t sh = 8 * r;
    for (uint32_t w = lane; w < nw; w += G)
      adst[w] = (asrc[w] >> sh) | (asrc[w + 1] << (32 - sh));
  }
  uint32_t done = nw << 2, tail = n & 3;
  if (lane < tail) dst[done + lane] = src[done + lane];
}

/* v1: unaligned uint4 loads via reinterpret (probes whether gfx950 global
 * loads tolerate misaligned dwordx4), 16B/lane when dst is 16B-aligned */
template <int G>
__device__ void copy_v1(uint8_t *dst, const uint8_t *src, uint32_t n, uint32_t lane) {
  if (((uintptr_t)dst & 15) == 0) {
    uint32_t nc = n >> 4;
    const uint4 *s4 = (const uint4 *)src; /* possibly misaligned! */
    uint4 *d4 = (uint4 *)dst;
    for (uint32_t c = lane; c < nc; c += G) d4[c] = s4[c];
    uint32_t done = nc << 4;
    for (uint32_t b = done + lane; b < n; b += G) dst[b] = src[b];
  } else {
    copy_v0<G>(dst, src, n, lane);
  }
}

/* v2: dwordx4 store + dword-funnel gather (aligned loads, wide stores) */
template <int G>
__device__ void copy_v2(uint8_t *dst, const uint8_t *src, uint32_t n, uint32_t lane) {
  if (((uintptr_t)dst & 15) == 0) {
    uint32_t r = (uint32_t)((uintptr_t)src & 3);
    const uint32_t *asrc = (const uint32_t *)(src - r);
    uint4 *d4 = (uint4 *)dst;
    uint32_t nc = n >> 4;
    if (r == 0) {
      for (uint32_t c = lane; c < nc; c += G) {
        uint32_t w = c * 4;
        d4[c] = make_uint4(asrc[w], asrc[w + 1], asrc[w + 2], asrc[w + 3]);
      }
    } else {
      uint32_t sh = 8 * r, ish = 32 - sh;
      for (uint32_t c = lane; c < nc; c += G) {
        uint32_t w = c * 4;
        uint32_t a0 = asrc[w], a1 = asrc[w + 1], a2 = asrc[w + 2],
                 a3 = asrc[w + 3], a4 = asrc[w + 4];
        d4[c] = make_uint4((a0 >> sh) | (a1 << ish), (a1 >> sh) | (a2 << ish),
                           (a2 >> sh) | (a3 << ish), (a3 >> sh) | (a4 << ish));
      }
    }
    uint32_t done = nc << 4;
    for (uint32_t b = done + lane; b < n; b += G) dst[b] = src[b];
  } else {
    copy_v0<G>(dst, src, n, lane);
  }
}

/* v3: 32B per lane (2x dwordx4 stores), funnel gather */
template <int G>
__device__ void copy_v3(uint8_t *dst, const uint8_t *src, uint32_t n, uint32_t lane) {
  if (((uintptr_t)dst & 15) == 0) {
    uint32_t r = (uint32_t)((uintptr_t)src & 3);
    const uint32_t *asrc = (const uint32_t *)(src - r);
    uint4 *d4 = (uint4 *)dst;
    uint32_t nc2 = n >> 5; /* 32B chunks */
    uint32_t sh = 8 * r, ish = 32 - sh;
    for (uint32_t c2 = lane; c2 < nc2; c2 += G) {
      uint32_t w = c2 * 8;
      uint32_t a[9];
#pragma unroll
      for (int j = 0; j < 9; j++) a[j] = asrc[w + j];
      if (r == 0) {
        d4[2 * c2] = make_uint4(a[0], a[1], a[2], a[3]);
        d4[2 * c2 + 1] = make_uint4(a[4], a[5], a[6], a[7]);
      } else {
        d4[2 * c2] = make_uint4((a[0] >> sh) | (a[1] << ish), (a[1] >> sh) | (a[2] << ish),
                                (a[2] >> sh) | (a[3] << ish), (a[3] >> sh) | (a[4] << ish));
        d4[2 * c2 + 1] = make_uint4((a[4] >> sh) | (a[5] << ish), (a[5] >> sh) | (a[6] << ish),
                                    (a[6] >> sh) | (a[7] << ish), (a[7] >> sh) | (a[8] << ish));
      }
    }
    uint32_t done = nc2 << 5;
    for (uint32_t b = done + lane; b < n; b += G) dst[b] = src[b];
  } else {
    copy_v0<G>(dst, src, n, lane);
  }
}

template <int V, int G, int STEP = 1, int OFS = 0>
__global__ void __launch_bounds__(256) k_bench(const uint8_t *__restrict__ src,
                                               uint8_t *__restrict__ dst,
                                               const Task *__restrict__ tasks,
                                               uint32_t ntasks) {
  uint32_t lane = threadIdx.x & (G - 1);
  uint32_t g = (blockIdx.x * blockDim.x + threadIdx.x) / G;
  uint32_t ngroups = (gridDim.x * blockDim.x) / G;
  for (uint32_t t = g * STEP + OFS; t < ntasks; t += ngroups * STEP) {
    Task tk = tasks[t];
    if (V == 0) copy_v0<G>(dst + tk.dst_off, src + tk.src_off, tk.nbytes, lane);
    if (V == 1) copy_v1<G>(dst + tk.dst_off, src + tk.src_off, tk.nbytes, lane);
    if (V == 2) copy_v2<G>(dst + tk.dst_off, src + tk.src_off, tk.nbytes, lane);
    if (V == 3) copy_v3<G>(dst + tk.dst_off, src + tk.src_off, tk.nbytes, lane);
  }
}

int main(int argc, char **argv) {
  /* task mix mirroring config #3: per record a 16B key task (src misaligned)
   * + a 1024B value task; dst 16B-aligned record slots */
  uint32_t nrec = argc > 1 ? atoi(argv[1]) : 200000;
  uint32_t vlen = argc > 2 ? atoi(argv[2]) : 1024;
  uint32_t klen = 16;
  uint64_t src_bytes = (uint64_t)nrec * (klen + vlen + 23) + 64;
  uint64_t rec16 = (klen + vlen + 15) & ~15u;
  uint64_t dst_bytes = (uint64_t)nrec * rec16 + 64;
  std::vector<uint8_t> h_src(src_bytes);
  for (uint64_t i = 0; i < src_bytes; i++) h_src[i] = (uint8_t)(i * 131 + 7);
  std::vector<Task> h_tasks;
  uint64_t soff = 0;
  for (uint32_t i = 0; i < nrec; i++) {
    soff += 3 + (i % 5); /* varint-ish misalignment */
    h_tasks.push_back({soff, (uint32_t)(i * rec16), klen});
    soff += klen + 2;
    h_tasks.push_back({soff, (uint32_t)(i * rec16 + klen), vlen});
    soff += vlen;
  }
  uint8_t *d_src, *d_dst;
  Task *d_tasks;
  CHECK(hipMalloc(&d_src, src_bytes + 16));
  CHECK(hipMalloc(&d_dst, dst_bytes + 16));
  CHECK(hipMalloc(&d_tasks, h_tasks.size() * sizeof(Task)));
  CHECK(hipMemcpy(d_src, h_src.data(), src_bytes, hipMemcpyHostToDevice));
  CHECK(hipMemcpy(d_tasks, h_tasks.data(), h_tasks.size() * sizeof(Task),
                  hipMemcpyHostToDevice));
  std::vector<uint8_t> ref(dst_bytes, 0), got(dst_bytes);
  for (auto &t : h_tasks) memcpy(ref.data() + t.dst_off, h_src.data() + t.src_off, t.nbytes);

  double total_bytes = 0;
  for (auto &t : h_tasks) total_bytes += 2.0 * t.nbytes;
  hipEvent_t e0, e1;
  CHECK(hipEventCreate(&e0));
  CHECK(hipEventCreate(&e1));

  auto run = [&](const char *name, auto kern) {
    CHECK(hipMemset(d_dst, 0, dst_bytes));
    kern(); /* warmup + correctness */
    CHECK(hipMemcpy(got.data(), d_dst, dst_bytes, hipMemcpyDeviceToHost));
    bool ok = memcmp(got.data(), ref.data(), dst_bytes) == 0;
    CHECK(hipEventRecord(e0));
    for (int i = 0; i < 20; i++) kern();
    CHECK(hipEventRecord(e1));
    CHECK(hipEventSynchronize(e1));
    float ms;
    CHECK(hipEventElapsedTime(&ms, e0, e1));
    printf("%-10s ok=%d  %8.1f GB/s  (%.1f us/launch)\n", name, ok,
           total_bytes * 20 / (ms * 1e-3) / 1e9, ms * 1000 / 20);
  };

  uint32_t nt = (uint32_t)h_tasks.size();
  dim3 blk(256);
#define RUNG(V, G, NB) run("v" #V "/g" #G "/b" #NB, [&] { \
    hipLaunchKernelGGL((k_bench<V, G>), dim3(NB), blk, 0, 0, d_src, d_dst, d_tasks, nt); })
  RUNG(2, 16, 2048); RUNG(2, 32, 2048);
  RUNG(3, 16, 2048); RUNG(3, 32, 2048); RUNG(3, 64, 2048);
  RUNG(2, 32, 1024); RUNG(2, 32, 4096); RUNG(2, 32, 8192);
  RUNG(3, 32, 1024); RUNG(3, 32, 4096); RUNG(3, 32, 8192);
  RUNG(3, 16, 4096);
  /* split streams: value-only tasks (odd) at g32, key-only (even) at g16 —
   * does de-mixing the task sizes in a wave buy anything? */
  run("vals/g32", [&] {
    hipLaunchKernelGGL((k_bench<2, 32, 2, 1>), dim3(2048), blk, 0, 0, d_src,
                       d_dst, d_tasks, nt);
  });
  run("keys/g16", [&] {
    hipLaunchKernelGGL((k_bench<2, 16, 2, 0>), dim3(2048), blk, 0, 0, d_src,
                       d_dst, d_tasks, nt);
  });
  run("split-both", [&] {
    hipLaunchKernelGGL((k_bench<2, 32, 2, 1>), dim3(2048), blk, 0, 0, d_src,
                       d_dst, d_tasks, nt);
    hipLaunchKernelGGL((k_bench<2, 16, 2, 0>), dim3(2048), blk, 0, 0, d_src,
                       d_dst, d_tasks, nt);
  });
  return 0;
}
