// Fused RMSNorm fwd/bwd for gfx950. Memory-bound: one 256-thread workgroup
// per row, bf16x8 vector traffic, fp32 accumulation, rstd saved for bwd.
// dw uses an LDS-resident fp32 partial per workgroup + one global atomic
// add per element per workgroup (H*4 B <= 64 KiB LDS up to H=16384).
#include "common.h"

#define BLOCK 256

extern "C" __global__ void rmsnorm_fwd_kernel(
    const bf16raw *__restrict__ x, const bf16raw *__restrict__ w,
    bf16raw *__restrict__ y, float *__restrict__ rstd,
    int rows, int H, float eps) {
    __shared__ float red[BLOCK / WAVE];
    for (int r = blockIdx.x; r < rows; r += gridDim.x) {
        const bf16raw *xr = x + (long)r * H;
        bf16raw *yr = y + (long)r * H;
        float ss = 0.f;
        for (int i = threadIdx.x * 8; i < H; i += BLOCK * 8) {
            bf16x8 v = load8(xr + i);
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                float f = v.get(j);
                ss += f * f;
            }
        }
        float total = block_sum<BLOCK>(ss, red);
        float rs = rsqrtf(total / H + eps);
        if (threadIdx.x == 0) rstd[r] = rs;
        for (int i = threadIdx.x * 8; i < H; i += BLOCK * 8) {
            bf16x8 v = load8(xr + i);
            bf16x8 wv = load8(w + i);
            bf16x8 o;
#pragma unroll
            for (int j = 0; j < 8; ++j) o.set(j, v.get(j) * rs * wv.get(j));
            store8(yr + i, o);
        }
        __syncthreads();
    }
}

// dx = rstd * w * dy - x * rstd^3 / H * sum_j(dy_j * w_j * x_j)
// dw_partial[H] accumulated in LDS across this workgroup's rows, then one
// atomicAdd per element into dw (fp32).
extern "C" __global__ void rmsnorm_bwd_kernel(
    const bf16raw *__restrict__ dy, const bf16raw *__restrict__ x,
    const bf16raw *__restrict__ w, const float *__restrict__ rstd,
    bf16raw *__restrict__ dx, float *__restrict__ dw,
    int rows, int H) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    float *dw_part = reinterpret_cast<float *>(smem);          // H floats
    float *red = dw_part + H;                                  // BLOCK/WAVE
    for (int i = threadIdx.x; i < H; i += BLOCK) dw_part[i] = 0.f;
    __syncthreads();

    for (int r = blockIdx.x; r < rows; r += gridDim.x) {
        const bf16raw *dyr = dy + (long)r * H;
        const bf16raw *xr = x + (long)r * H;
        bf16raw *dxr = dx + (long)r * H;
        const float rs = rstd[r];
        float dot = 0.f;
        for (int i = threadIdx.x * 8; i < H; i += BLOCK * 8) {
            bf16x8 vdy = load8(dyr + i);
            bf16x8 vx = load8(xr + i);
            bf16x8 vw = load8(w + i);
#pragma unroll
            for (int j = 0; j < 8; ++j)
                dot += vdy.get(j) * vw.get(j) * vx.get(j);
        }
        float total = block_sum<BLOCK>(dot, red);
        const float k = total * rs * rs * rs / H;
        for (int i = threadIdx.x * 8; i < H; i += BLOCK * 8) {
            bf16x8 vdy = load8(dyr + i);
            bf16x8 vx = load8(xr + i);
            bf16x8 vw = load8(w + i);
            bf16x8 o;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                float xv = vx.get(j);
                float dyv = vdy.get(j);
                o.set(j, rs * vw.get(j) * dyv - xv * k);
                dw_part[i + j] += dyv * xv * rs;
            }
            store8(dxr + i, o);
        }
        __syncthreads();
    }
    for (int i = threadIdx.x; i < H; i += BLOCK)
        atomicAdd(&dw[i], dw_part[i]);
}

extern "C" void rmsnorm_fwd_launch(const void *x, const void *w, void *y,
                                   void *rstd, int rows, int H, float eps,
                                   hipStream_t stream) {
    int grid = rows < 2048 ? rows : 2048;
    hipLaunchKernelGGL(rmsnorm_fwd_kernel, dim3(grid), dim3(BLOCK), 0, stream,
                       (const bf16raw *)x, (const bf16raw *)w, (bf16raw *)y,
                       (float *)rstd, rows, H, eps);
}

extern "C" void rmsnorm_bwd_launch(const void *dy, const void *x, const void *w,
                                   const void *rstd, void *dx, void *dw,
                                   int rows, int H, hipStream_t stream) {
    int grid = rows < 1024 ? rows : 1024;
    size_t smem = (size_t)H * 4 + (BLOCK / WAVE) * 4;
    hipLaunchKernelGGL(rmsnorm_bwd_kernel, dim3(grid), dim3(BLOCK), smem, stream,
                       (const bf16raw *)dy, (const bf16raw *)x,
                       (const bf16raw *)w, (const float *)rstd, (bf16raw *)dx,
                       (float *)dw, rows, H);
}
