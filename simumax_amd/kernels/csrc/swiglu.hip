// Fused SwiGLU fwd/bwd for gfx950: y = silu(gate) * up with
// gate = x[:, :I], up = x[:, I:]. Memory-bound, bf16x8 vector traffic.
#include "common.h"

#define BLOCK 256

extern "C" __global__ void swiglu_fwd_kernel(
    const bf16raw *__restrict__ x, bf16raw *__restrict__ y,
    long rows, int I) {
    const long total = rows * (long)I;
    for (long idx = ((long)blockIdx.x * BLOCK + threadIdx.x) * 8; idx < total;
         idx += (long)gridDim.x * BLOCK * 8) {
        const long r = idx / I;
        const int c = idx % I;
        bf16x8 g = load8(x + r * 2 * I + c);
        bf16x8 u = load8(x + r * 2 * I + I + c);
        bf16x8 o;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float gv = g.get(j);
            float sig = 1.f / (1.f + __expf(-gv));
            o.set(j, gv * sig * u.get(j));
        }
        store8(y + idx, o);
    }
}

extern "C" __global__ void swiglu_bwd_kernel(
    const bf16raw *__restrict__ dy, const bf16raw *__restrict__ x,
    bf16raw *__restrict__ dx, long rows, int I) {
    const long total = rows * (long)I;
    for (long idx = ((long)blockIdx.x * BLOCK + threadIdx.x) * 8; idx < total;
         idx += (long)gridDim.x * BLOCK * 8) {
        const long r = idx / I;
        const int c = idx % I;
        bf16x8 g = load8(x + r * 2 * I + c);
        bf16x8 u = load8(x + r * 2 * I + I + c);
        bf16x8 d = load8(dy + idx);
        bf16x8 dg, du;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float gv = g.get(j);
            float dv = d.get(j);
            float sig = 1.f / (1.f + __expf(-gv));
            float silu = gv * sig;
            dg.set(j, dv * u.get(j) * (sig + silu * (1.f - sig)));
            du.set(j, dv * silu);
        }
        store8(dx + r * 2 * I + c, dg);
        store8(dx + r * 2 * I + I + c, du);
    }
}

extern "C" void swiglu_fwd_launch(const void *x, void *y, long rows, int I,
                                  hipStream_t stream) {
    long grid = CDIV(rows * (long)I, BLOCK * 8);
    if (grid > 2048) grid = 2048;
    hipLaunchKernelGGL(swiglu_fwd_kernel, dim3((int)grid), dim3(BLOCK), 0,
                       stream, (const bf16raw *)x, (bf16raw *)y, rows, I);
}

extern "C" void swiglu_bwd_launch(const void *dy, const void *x, void *dx,
                                  long rows, int I, hipStream_t stream) {
    long grid = CDIV(rows * (long)I, BLOCK * 8);
    if (grid > 2048) grid = 2048;
    hipLaunchKernelGGL(swiglu_bwd_kernel, dim3((int)grid), dim3(BLOCK), 0,
                       stream, (const bf16raw *)dy, (const bf16raw *)x,
                       (bf16raw *)dx, rows, I);
}
