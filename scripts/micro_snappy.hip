/* micro_snappy.hip — A/B variants for the Snappy decompress stage (config #5).
 * v0: lane-per-update, parse straight from global (dependent byte loads)
 * v1: 16-lane cooperative groups (current engine kernel)
 * v2: lane-per-update with the compressed stream PRE-STAGED into LDS by
 *     independent vector loads, parse from LDS (dependent loads become
 *     ~cheap); +4B row pad to dodge the all-threads-same-bank stride.
 * Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 scripts/micro_snappy.hip -o build/micro_snappy
 */
#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <vector>

#define WB_UNALIGNED_OK 1
#include "../rocksplicator_amd/csrc/snappy.h"

#define CHECK(x)                                                      \
  do {                                                                \
    hipError_t e = (x);                                               \
    if (e != hipSuccess) {                                            \
      printf("ERR %s: %s\n", #x, hipGetErrorString(e));               \
      exit(1);                                                        \
    }                                                                 \
  } while (0)

struct Task {
  uint64_t comp_off;
  uint32_t comp_len;
  uint32_t ulen;
  uint64_t out_off;
};

__global__ void k_v0(const uint8_t *__restrict__ comp,
                     const Task *__restrict__ tasks, uint32_t n,
                     uint8_t *__restrict__ out) {
  uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  Task t = tasks[i];
  snp::decompress(comp + t.comp_off, t.comp_len, out + t.out_off, t.ulen);
}

__global__ void k_v1(const uint8_t *__restrict__ comp,
                     const Task *__restrict__ tasks, uint32_t n,
                     uint8_t *__restrict__ out) {
  uint32_t lane = threadIdx.x & 15u;
  uint32_t i = (blockIdx.x * blockDim.x + threadIdx.x) >> 4;
  if (i >= n) return;
  Task t = tasks[i];
  snp::decompress_coop16(comp + t.comp_off, t.comp_len, out + t.out_off,
                         t.ulen, lane);
}

template <int MAXC> /* max staged compressed bytes per stream */
__global__ void __launch_bounds__(256) k_v2(const uint8_t *__restrict__ comp,
                                            const Task *__restrict__ tasks,
                                            uint32_t n,
                                            uint8_t *__restrict__ out) {
  constexpr int STRIDE = MAXC + 4; /* +4B pad: break the 64-bank stride */
  __shared__ uint8_t lds[256 * STRIDE];
  uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  Task t = tasks[i];
  uint8_t *mine = lds + threadIdx.x * STRIDE;
  if (t.comp_len <= MAXC) {
    /* independent vector prefetch: ~comp_len/16 non-dependent loads */
    const uint8_t *src = comp + t.comp_off;
    for (uint32_t b = 0; b < t.comp_len; b += 16)
      *(uint4 *)(mine + b) = *(const uint4 *)(src + b);
    snp::decompress(mine, t.comp_len, out + t.out_off, t.ulen);
  } else {
    snp::decompress(comp + t.comp_off, t.comp_len, out + t.out_off, t.ulen);
  }
}

int main(int argc, char **argv) {
  uint32_t n = argc > 1 ? atoi(argv[1]) : 400000;
  uint32_t vlen = argc > 2 ? atoi(argv[2]) : 1024;
  /* build compressible payloads like the generator's config-#5 values */
  std::vector<uint8_t> plain(vlen);
  std::vector<Task> tasks(n);
  std::vector<uint8_t> comp;
  std::vector<uint8_t> ref;
  uint64_t coff = 0, ooff = 0;
  srand(7);
  std::vector<uint8_t> cbuf(snp::max_compressed_len(vlen) + 16);
  for (uint32_t i = 0; i < n; i++) {
    for (uint32_t b = 0; b < vlen; b += 64) {
      uint8_t c = (uint8_t)(rand() & 3);
      for (uint32_t j = 0; j < 64 && b + j < vlen; j++)
        plain[b + j] = (uint8_t)(c * 67 + (j & 15) * 13 + 7);
    }
    uint32_t clen = snp::compress(plain.data(), vlen, cbuf.data(),
                                  (uint32_t)cbuf.size());
    tasks[i] = {coff, clen, vlen, ooff};
    comp.insert(comp.end(), cbuf.data(), cbuf.data() + clen);
    ref.insert(ref.end(), plain.begin(), plain.end());
    coff += clen;
    ooff += (vlen + 15) & ~15u;
    tasks[i].out_off = ooff - ((vlen + 15) & ~15u);
  }
  printf("n=%u vlen=%u comp ratio %.2fx (avg clen %.0f)\n", n, vlen,
         (double)n * vlen / comp.size(), (double)comp.size() / n);
  uint8_t *d_comp, *d_out;
  Task *d_tasks;
  CHECK(hipMalloc(&d_comp, comp.size() + 16));
  CHECK(hipMalloc(&d_out, ooff + 16));
  CHECK(hipMalloc(&d_tasks, n * sizeof(Task)));
  CHECK(hipMemcpy(d_comp, comp.data(), comp.size(), hipMemcpyHostToDevice));
  CHECK(hipMemcpy(d_tasks, tasks.data(), n * sizeof(Task),
                  hipMemcpyHostToDevice));
  std::vector<uint8_t> got(ooff);
  hipEvent_t e0, e1;
  CHECK(hipEventCreate(&e0));
  CHECK(hipEventCreate(&e1));
  double ubytes = (double)n * vlen;

  auto run = [&](const char *name, auto kern) {
    CHECK(hipMemset(d_out, 0, ooff));
    kern();
    CHECK(hipMemcpy(got.data(), d_out, ooff, hipMemcpyDeviceToHost));
    bool ok = true;
    for (uint32_t i = 0; i < n && ok; i += 997)
      ok = memcmp(got.data() + tasks[i].out_off, ref.data() + (uint64_t)i * vlen,
                  vlen) == 0;
    CHECK(hipEventRecord(e0));
    for (int r = 0; r < 10; r++) kern();
    CHECK(hipEventRecord(e1));
    CHECK(hipEventSynchronize(e1));
    float ms;
    CHECK(hipEventElapsedTime(&ms, e0, e1));
    printf("%-12s ok=%d  %7.1f GB/s out  (%.0f us/launch)\n", name, ok,
           ubytes * 10 / (ms * 1e-3) / 1e9, ms * 100);
  };

  run("v0-global", [&] {
    hipLaunchKernelGGL(k_v0, dim3((n + 255) / 256), dim3(256), 0, 0, d_comp,
                       d_tasks, n, d_out);
  });
  run("v1-coop16", [&] {
    hipLaunchKernelGGL(k_v1, dim3((n * 16 + 255) / 256), dim3(256), 0, 0,
                       d_comp, d_tasks, n, d_out);
  });
  run("v2-lds512", [&] {
    hipLaunchKernelGGL(k_v2<508>, dim3((n + 255) / 256), dim3(256), 0, 0,
                       d_comp, d_tasks, n, d_out);
  });
  run("v2-lds256", [&] {
    hipLaunchKernelGGL(k_v2<252>, dim3((n + 255) / 256), dim3(256), 0, 0,
                       d_comp, d_tasks, n, d_out);
  });
  run("v0-global", [&] {
    hipLaunchKernelGGL(k_v0, dim3((n + 255) / 256), dim3(256), 0, 0, d_comp,
                       d_tasks, n, d_out);
  });
  return 0;
}
