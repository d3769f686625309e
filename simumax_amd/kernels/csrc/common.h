// Common device helpers for the simumax_amd CDNA4 (gfx950) kernels.
// Wave = 64 lanes; bf16 traffic is vectorized 8-wide (16 B/lane) per
// /opt/skills guide G13 (scalar bf16 loads are ~2x slower).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define DEV __device__ __forceinline__

typedef unsigned short bf16raw;

// 8 bf16 elements = 16 bytes, one vector load per lane
struct bf16x8 {
    uint4 raw;
    DEV float get(int i) const {
        const bf16raw *p = reinterpret_cast<const bf16raw *>(&raw);
        __hip_bfloat16 h;
        *reinterpret_cast<bf16raw *>(&h) = p[i];
        return __bfloat162float(h);
    }
    DEV void set(int i, float v) {
        bf16raw *p = reinterpret_cast<bf16raw *>(&raw);
        __hip_bfloat16 h = __float2bfloat16(v);
        p[i] = *reinterpret_cast<bf16raw *>(&h);
    }
};

DEV bf16x8 load8(const bf16raw *ptr) {
    bf16x8 v;
    v.raw = *reinterpret_cast<const uint4 *>(ptr);
    return v;
}

DEV void store8(bf16raw *ptr, const bf16x8 &v) {
    *reinterpret_cast<uint4 *>(ptr) = v.raw;
}

DEV float bf2f(bf16raw r) {
    __hip_bfloat16 h;
    *reinterpret_cast<bf16raw *>(&h) = r;
    return __bfloat162float(h);
}

DEV bf16raw f2bf(float v) {
    __hip_bfloat16 h = __float2bfloat16(v);
    return *reinterpret_cast<bf16raw *>(&h);
}

// ---- wave(64) reductions via xor shuffles ----
DEV float wave_sum(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
    return v;
}

DEV float wave_max(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
    return v;
}

// ---- block reduction (<= 1024 threads) ----
template <int BLOCK>
DEV float block_sum(float v, float *lds /* >= BLOCK/WAVE floats */) {
    v = wave_sum(v);
    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    constexpr int NW = BLOCK / WAVE;
    if (lane == 0) lds[wid] = v;
    __syncthreads();
    float r = (lane < NW) ? lds[lane] : 0.f;
    r = wave_sum(r);  // wasteful but trivial at NW<=16
    return r;
}

template <int BLOCK>
DEV float block_max(float v, float *lds) {
    v = wave_max(v);
    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    constexpr int NW = BLOCK / WAVE;
    if (lane == 0) lds[wid] = v;
    __syncthreads();
    float r = (lane < NW) ? lds[lane] : -INFINITY;
    r = wave_max(r);
    return r;
}

#define CDIV(a, b) (((a) + (b) - 1) / (b))
