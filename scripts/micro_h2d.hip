/* micro_h2d.hip — bisect the staged-leg H2D gap: the engine's
 * hipMemcpyAsync gets ~43 GB/s where torch's copy_ gets 57. Replicates the
 * engine's exact pattern, then strips pieces. */
#include <hip/hip_runtime.h>

#include <chrono>
#include <cstdio>
#include <vector>
#include <thread>

#define CK(x)                                                      \
  do {                                                             \
    hipError_t e = (x);                                            \
    if (e != hipSuccess) {                                         \
      printf("FAIL %s: %s\n", #x, hipGetErrorString(e));           \
      return 1;                                                    \
    }                                                              \
  } while (0)

__global__ void k_touch(uint8_t *p, size_t n) { /* stand-in kernel */
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) p[i] ^= 1;
}

/* ~engine-sized kernel: streams ~2.4 GB through HBM (≈0.4 ms at 6 TB/s) */
__global__ void k_heavy(const uint4 *__restrict__ src, uint4 *__restrict__ dst,
                        size_t n4) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t j = i; j < n4; j += stride) dst[j] = src[j];
}

int main() {
  const size_t WIN = 856ull << 20;
  const size_t PINB = 2 * WIN + (1 << 20);
  uint8_t *pin;
  CK(hipHostMalloc(&pin, PINB));
  uint8_t *dA, *dB;
  CK(hipMalloc(&dA, WIN + 16));
  CK(hipMalloc(&dB, WIN + 16));
  hipStream_t h2d, main_s, copyout;
  CK(hipStreamCreate(&h2d));
  CK(hipStreamCreate(&main_s));
  CK(hipStreamCreate(&copyout));
  hipEvent_t ev9, ev1, usedA, usedB;
  CK(hipEventCreate(&ev9));
  CK(hipEventCreate(&ev1));
  CK(hipEventCreate(&usedA));
  CK(hipEventCreate(&usedB));
  CK(hipEventRecord(usedA, main_s));
  CK(hipEventRecord(usedB, main_s));

  auto bench = [&](const char *label, bool events, bool crosswait,
                   bool kernel, size_t win_off) {
    CK(hipDeviceSynchronize());
    const int K = 12;
    // warmup
    CK(hipMemcpyAsync(dA, pin, WIN, hipMemcpyHostToDevice, h2d));
    CK(hipStreamSynchronize(h2d));
    auto t0 = std::chrono::steady_clock::now();
    for (int i = 0; i < K; i++) {
      uint8_t *dst = (i & 1) ? dB : dA;
      hipEvent_t used = (i & 1) ? usedB : usedA;
      const uint8_t *src = pin + (i & 1) * (WIN + win_off);
      if (crosswait) CK(hipStreamWaitEvent(h2d, used, 0));
      if (events) CK(hipEventRecord(ev9, h2d));
      CK(hipMemcpyAsync(dst, src, WIN, hipMemcpyHostToDevice, h2d));
      if (events) CK(hipEventRecord(ev1, h2d));
      if (crosswait) {
        CK(hipStreamWaitEvent(main_s, ev1, 0));
        if (kernel)
          hipLaunchKernelGGL(k_touch, dim3(1024), dim3(256), 0, main_s, dst,
                             1 << 20);
        CK(hipEventRecord(used, main_s));
      }
    }
    CK(hipDeviceSynchronize());
    double s = std::chrono::duration<double>(std::chrono::steady_clock::now() - t0).count();
    printf("%-44s %.1f GB/s (%.2f ms/copy)\n", label, K * (double)WIN / s / 1e9,
           s / K * 1e3);
    return 0;
  };

  bench("plain alternating memcpyasync", false, false, false, 0);
  bench("+ timing events", true, false, false, 0);
  bench("+ cross-stream wait/record (engine shape)", true, true, false, 0);
  bench("+ tiny kernel on main stream", true, true, true, 0);
  bench("odd window offset (1045B)", true, true, true, 1045);

  /* full engine replica: heavy kernels (~0.4 ms HBM-bound), 5 timing
   * events on main per tick, copyout-stream D2H of ~1 MB, deep host-side
   * queue (all ticks submitted up front like the bench loop) */
  uint8_t *work, *work2, *h_small;
  CK(hipMalloc(&work, 1200ull << 20));
  CK(hipMalloc(&work2, 1200ull << 20));
  CK(hipHostMalloc(&h_small, 1 << 20));
  std::vector<hipEvent_t> pool(256);
  for (auto &pev : pool) CK(hipEventCreate(&pev));
  hipEvent_t used2[2];
  CK(hipEventCreate(&used2[0]));
  CK(hipEventCreate(&used2[1]));
  CK(hipEventRecord(used2[0], main_s));
  CK(hipEventRecord(used2[1], main_s));
  {
    const int K = 12;
    // warmup copy
    CK(hipMemcpyAsync(dA, pin, WIN, hipMemcpyHostToDevice, h2d));
    CK(hipStreamSynchronize(h2d));
    auto t0 = std::chrono::steady_clock::now();
    int pe = 0;
    for (int i = 0; i < K; i++) {
      uint8_t *dst = (i & 1) ? dB : dA;
      hipEvent_t used = used2[i & 1];
      auto nev = [&]() { return pool[pe++ % 256]; };
      hipEvent_t e0 = nev(), e9 = nev(), e1 = nev(), e4 = nev(), e5 = nev(),
                 e6 = nev(), e7 = nev();
      CK(hipEventRecord(e0, main_s));
      CK(hipStreamWaitEvent(h2d, used, 0));
      CK(hipEventRecord(e9, h2d));
      CK(hipMemcpyAsync(dst, pin + (i & 1) * WIN, WIN, hipMemcpyHostToDevice, h2d));
      CK(hipEventRecord(e1, h2d));
      CK(hipStreamWaitEvent(main_s, e1, 0));
      /* ~3 engine-shaped kernels */
      hipLaunchKernelGGL(k_heavy, dim3(4096), dim3(256), 0, main_s,
                         (const uint4 *)work, (uint4 *)work2, (300ull << 20) / 16);
      CK(hipEventRecord(e4, main_s));
      hipLaunchKernelGGL(k_heavy, dim3(4096), dim3(256), 0, main_s,
                         (const uint4 *)work2, (uint4 *)work, (300ull << 20) / 16);
      CK(hipEventRecord(e5, main_s));
      hipLaunchKernelGGL(k_touch, dim3(64), dim3(256), 0, main_s, work, 1 << 14);
      CK(hipEventRecord(e6, main_s));
      CK(hipEventRecord(used, main_s));
      CK(hipStreamWaitEvent(copyout, e6, 0));
      CK(hipMemcpyAsync(h_small, work, 1 << 20, hipMemcpyDeviceToHost, copyout));
      CK(hipEventRecord(e7, copyout));
    }
    CK(hipDeviceSynchronize());
    double s = std::chrono::duration<double>(std::chrono::steady_clock::now() - t0).count();
    printf("%-44s %.1f GB/s (%.2f ms/tick)\n", "full engine replica",
           K * (double)WIN / s / 1e9, s / K * 1e3);
  }

  /* the bench loop's real shape: 8-deep pipeline, then the host BLOCKS on a
   * per-tick event while copies are in flight (the engine's free_slot →
   * hipEventSynchronize). Hypothesis: the blocked-host wait mode is what
   * drags SDMA from 57 to ~40 GB/s. Variants: spin (hipEventSynchronize on
   * a default event), blocking-sync event, sleep-poll via hipEventQuery. */
  for (int mode = 0; mode < 3; mode++) {
    const int K = 16, DEPTH = 6;
    std::vector<hipEvent_t> done(K);
    for (int i = 0; i < K; i++) {
      unsigned fl = (mode == 1) ? hipEventBlockingSync | hipEventDisableTiming
                                : hipEventDefault;
      CK(hipEventCreateWithFlags(&done[i], fl));
    }
    CK(hipMemcpyAsync(dA, pin, WIN, hipMemcpyHostToDevice, h2d));
    CK(hipStreamSynchronize(h2d));
    auto t0 = std::chrono::steady_clock::now();
    for (int i = 0; i < K; i++) {
      uint8_t *dst = (i & 1) ? dB : dA;
      CK(hipMemcpyAsync(dst, pin + (i & 1) * WIN, WIN, hipMemcpyHostToDevice, h2d));
      CK(hipEventRecord(done[i], h2d));
      if (i >= DEPTH) { /* host blocks like free_slot does */
        if (mode == 2) {
          while (hipEventQuery(done[i - DEPTH]) == hipErrorNotReady)
            std::this_thread::sleep_for(std::chrono::microseconds(100));
        } else {
          CK(hipEventSynchronize(done[i - DEPTH]));
        }
      }
    }
    CK(hipDeviceSynchronize());
    double s = std::chrono::duration<double>(std::chrono::steady_clock::now() - t0).count();
    const char *names[] = {"host spins in hipEventSynchronize",
                           "blocking-sync (interrupt) event wait",
                           "sleep-poll hipEventQuery"};
    printf("%-44s %.1f GB/s (%.2f ms/copy)\n", names[mode],
           K * (double)WIN / s / 1e9, s / K * 1e3);
    for (auto &d : done) (void)hipEventDestroy(d);
  }
  return 0;
}
