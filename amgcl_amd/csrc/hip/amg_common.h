// amgcl_amd — shared device helpers for the gfx950 kernels.
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>

#define WAVE 64

// XCD-aware logical block id (guide cdna_hip_programming.md §1): the command
// processor round-robins workgroups across the 8 XCDs, so blocks with
// blockIdx%8==k run on XCD k.  Remapping gives each XCD one contiguous slice
// of the logical index space: the x-gather bands of neighboring CSR rows
// then land in a single XCD's L2 instead of being replicated in all eight.
// Affects speed only, never correctness.
__device__ static inline int64_t amg_logical_block() {
#ifndef AMGCL_NO_SWIZZLE
    unsigned g = gridDim.x;
    if ((g & 7u) == 0u)
        return (int64_t)(blockIdx.x & 7u) * (int64_t)(g >> 3) + (blockIdx.x >> 3);
#endif
    return blockIdx.x;
}

// Streaming loads for the matrix arrays (val/col are read exactly once per
// SpMV): the nontemporal hint keeps them from evicting the x vector, which
// is the only array with reuse.
#ifndef AMGCL_NO_NT
#define AMG_STREAM_LD(p) __builtin_nontemporal_load(p)
#else
#define AMG_STREAM_LD(p) (*(p))
#endif

// Launch geometry for memory-bound grid-stride kernels (guide §6 G11): cap
// at ~8 blocks/CU and grid-stride the rest; rounded to a multiple of 8 so
// the XCD swizzle divides evenly.
static inline int amg_nblocks(int64_t work, int block = 256, int cap = 2048) {
    int64_t b = (work + block - 1) / block;
    if (b < 1) b = 1;
    if (b > cap) b = cap;
    if (b > 8) b = (b + 7) & ~(int64_t)7;
    return (int)b;
}
