"""Measure this box's raw pinned-host -> HBM copy bandwidth at the staged
leg's transfer size, to separate 'engine overhead' from 'PCIe ceiling'.
Scratch tool."""
import time

import torch

N = 856 * 1024 * 1024
src = torch.empty(N, dtype=torch.uint8, pin_memory=True)
dst = torch.empty(N, dtype=torch.uint8, device="cuda")
s = torch.cuda.Stream()
# warmup
with torch.cuda.stream(s):
    for _ in range(3):
        dst.copy_(src, non_blocking=True)
torch.cuda.synchronize()
t0 = time.perf_counter()
K = 12
with torch.cuda.stream(s):
    for _ in range(K):
        dst.copy_(src, non_blocking=True)
torch.cuda.synchronize()
dt = time.perf_counter() - t0
print(f"H2D {N/1e6:.0f} MB x{K}: {K*N/dt/1e9:.1f} GB/s ({dt/K*1e3:.2f} ms/copy)")

# D2H for the drain leg ceiling
t0 = time.perf_counter()
with torch.cuda.stream(s):
    for _ in range(K):
        src.copy_(dst, non_blocking=True)
torch.cuda.synchronize()
dt = time.perf_counter() - t0
print(f"D2H: {K*N/dt/1e9:.1f} GB/s")

# bidirectional (drain overlapped with staging)
src2 = torch.empty(N, dtype=torch.uint8, pin_memory=True)
dst2 = torch.empty(N, dtype=torch.uint8, device="cuda")
s2 = torch.cuda.Stream()
t0 = time.perf_counter()
with torch.cuda.stream(s):
    for _ in range(K):
        dst.copy_(src, non_blocking=True)
with torch.cuda.stream(s2):
    for _ in range(K):
        src2.copy_(dst2, non_blocking=True)
torch.cuda.synchronize()
dt = time.perf_counter() - t0
print(f"bidir: {2*K*N/dt/1e9:.1f} GB/s aggregate")

# --- variants mirroring the engine's staged-tick pattern ---
dstA = torch.empty(N, dtype=torch.uint8, device="cuda")
dstB = torch.empty(N, dtype=torch.uint8, device="cuda")

def timed(label, fn, k=12):
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    fn(k)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"{label}: {k*N/dt/1e9:.1f} GB/s ({dt/k*1e3:.2f} ms/copy)")

def alt(k):
    with torch.cuda.stream(s):
        for i in range(k):
            (dstA if i % 2 == 0 else dstB).copy_(src, non_blocking=True)
timed("alternating-2-dst", alt)

def alt_ev(k):
    evs = [torch.cuda.Event(enable_timing=True) for _ in range(2 * k)]
    with torch.cuda.stream(s):
        for i in range(k):
            evs[2 * i].record(s)
            (dstA if i % 2 == 0 else dstB).copy_(src, non_blocking=True)
            evs[2 * i + 1].record(s)
timed("alternating + 2 timing events/copy", alt_ev)

# window from a larger pinned buffer at alternating offsets (like 2 windows)
big = torch.empty(2 * N, dtype=torch.uint8, pin_memory=True)
def win(k):
    with torch.cuda.stream(s):
        for i in range(k):
            d = dstA if i % 2 == 0 else dstB
            d.copy_(big[(i % 2) * N:(i % 2) * N + N], non_blocking=True)
timed("2 pinned windows -> 2 dst", win)

# unaligned source offset (engine windows start at arbitrary byte offsets)
def unal(k):
    with torch.cuda.stream(s):
        for i in range(k):
            d = dstA if i % 2 == 0 else dstB
            off = 123457 + (i % 2) * 1045
            d[:N - 4096].copy_(big[off:off + N - 4096], non_blocking=True)
timed("unaligned pinned src offset", unal)

def al4k(k):
    with torch.cuda.stream(s):
        for i in range(k):
            d = dstA if i % 2 == 0 else dstB
            off = 4096 * 300 + (i % 2) * 4096
            d[:N - 4096].copy_(big[off:off + N - 4096], non_blocking=True)
timed("4K-aligned pinned src offset", al4k)

# unaligned DEVICE dst as well
def unal_dst(k):
    with torch.cuda.stream(s):
        for i in range(k):
            d = dstA if i % 2 == 0 else dstB
            d[13:13 + N - 4096].copy_(big[123457:123457 + N - 4096],
                                      non_blocking=True)
timed("unaligned src+dst", unal_dst)
