// One-shot probe of the v_mfma_f32_16x16x32_bf16 fragment layouts.
// Three experiments in one launch (wave 0 of one workgroup):
//  mode 0: A,B loaded from global with the ASSUMED layouts
//          (A: row=lane&15, k=(lane>>4)*8+j; B: col=lane&15, same k),
//          C stored with assumed C layout (col=lane&15, row=(lane>>4)*4+j)
//          -> compare against torch A@B.
//  mode 1: a_frag[j] = lane + 64*j (IDs), B = all-ones -> C row sums expose
//          the true A mapping.
//  mode 2: A = all-ones, b_frag[j] = lane + 64*j -> C col sums expose B.
// Raw per-lane output c_raw[lane][reg] is always stored too.
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8v;
typedef __attribute__((ext_vector_type(4))) float f32x4;

extern "C" __global__ void mfma_probe_kernel(
    const bf16raw *__restrict__ A,  // [16][32] row-major
    const bf16raw *__restrict__ B,  // [32][16] row-major
    float *__restrict__ C_mapped,   // [16][16]
    float *__restrict__ C_raw,      // [64][4]
    int mode) {
    const int lane = threadIdx.x;
    if (lane >= WAVE) return;
    const int col = lane & 15;
    const int kgrp = lane >> 4;

    bf16x8v a, b;
    bf16x8 atmp, btmp;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
        float av, bv;
        const int kk = kgrp * 8 + j;
        if (mode == 1) {
            av = (float)(lane + 64 * j) * (1.0f / 64.f);
            bv = 1.f;
        } else if (mode == 2) {
            av = 1.f;
            bv = (float)(lane + 64 * j) * (1.0f / 64.f);
        } else {
            av = bf2f(A[col * 32 + kk]);        // A[row][k]
            bv = bf2f(B[kk * 16 + col]);        // B[k][col]
        }
        atmp.set(j, av);
        btmp.set(j, bv);
    }
    a = *reinterpret_cast<bf16x8v *>(&atmp.raw);
    b = *reinterpret_cast<bf16x8v *>(&btmp.raw);
    f32x4 c = f32x4{0, 0, 0, 0};
    c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
        C_raw[lane * 4 + j] = c[j];
        C_mapped[(kgrp * 4 + j) * 16 + col] = c[j];
    }
}

extern "C" void mfma_probe_launch(const void *A, const void *B, void *C_mapped,
                                  void *C_raw, int mode, hipStream_t stream) {
    hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream,
                       (const bf16raw *)A, (const bf16raw *)B,
                       (float *)C_mapped, (float *)C_raw, mode);
}
