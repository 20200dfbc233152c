"""Diagnose the ~100ms GPU-idle stalls before the big SELL fills at 512^3:
is a warm torch re-allocation of multi-GB blocks fast (cache hit) or slow?"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

def t(fn):
    torch.cuda.synchronize(); t0 = time.perf_counter()
    r = fn(); torch.cuda.synchronize()
    return r, time.perf_counter() - t0

sizes = [17_000_000_000, 5_700_000_000, 11_300_000_000]
for s in sizes:
    a, dt = t(lambda: torch.empty(s, dtype=torch.uint8, device="cuda"))
    print(f"cold alloc {s/1e9:5.1f} GB: {dt*1e3:8.1f} ms")
    del a; torch.cuda.synchronize()
    a, dt = t(lambda: torch.empty(s, dtype=torch.uint8, device="cuda"))
    print(f"warm alloc {s/1e9:5.1f} GB: {dt*1e3:8.1f} ms")
    del a

# simulate the bench pattern: build-free-build with interleaved other allocs
def hierarchy_like():
    blobs = [torch.empty(n, dtype=torch.uint8, device="cuda")
             for n in (11_300_000_000, 5_700_000_000, 2_000_000_000,
                       1_000_000_000, 17_000_000_000)]
    return blobs
b, dt = t(hierarchy_like); print(f"pass1 alloc set: {dt*1e3:8.1f} ms")
del b; torch.cuda.synchronize()
b, dt = t(hierarchy_like); print(f"pass2 alloc set: {dt*1e3:8.1f} ms")
del b
stats = torch.cuda.memory_stats()
print("num_device_alloc:", stats.get("num_device_alloc"))
