// Fused rotary position embedding (GPT-NeoX half-rotation convention) for
// gfx950. cos/sin tables are precomputed on host (guide App. B: on-device
// trig turns a memory-bound op VALU-bound). Operates on a [tokens, heads, D]
// view; backward is the same rotation with the sine negated, so one kernel
// serves both (sign argument) and nothing is saved for autograd.
#include "common.h"

#define BLOCK 256

// x: [rows, heads, D] bf16 (row = b*s flattened, seq position = pos[row])
// cs: [max_pos, D/2, 2] fp32 interleaved (cos, sin)
extern "C" __global__ void rope_kernel(
    const bf16raw *__restrict__ x, bf16raw *__restrict__ y,
    const float *__restrict__ cs, const int *__restrict__ pos,
    int rows, int heads, int D, float sign) {
    const int half = D / 2;
    const long total_pairs = (long)rows * heads * half;
    for (long idx = (long)blockIdx.x * BLOCK + threadIdx.x; idx < total_pairs;
         idx += (long)gridDim.x * BLOCK) {
        const int d = idx % half;
        const long rh = idx / half;
        const int row = rh / heads;
        const long base = rh * D;
        const float2 c = reinterpret_cast<const float2 *>(cs)[(long)pos[row] * half + d];
        const float x1 = bf2f(x[base + d]);
        const float x2 = bf2f(x[base + d + half]);
        const float s = c.y * sign;
        y[base + d] = f2bf(x1 * c.x - x2 * s);
        y[base + d + half] = f2bf(x2 * c.x + x1 * s);
    }
}

extern "C" void rope_launch(const void *x, void *y, const void *cs,
                            const void *pos, int rows, int heads, int D,
                            float sign, hipStream_t stream) {
    long pairs = (long)rows * heads * (D / 2);
    long grid = CDIV(pairs, BLOCK);
    if (grid > 4096) grid = 4096;
    hipLaunchKernelGGL(rope_kernel, dim3((int)grid), dim3(BLOCK), 0, stream,
                       (const bf16raw *)x, (bf16raw *)y, (const float *)cs,
                       (const int *)pos, rows, heads, D, sign);
}
