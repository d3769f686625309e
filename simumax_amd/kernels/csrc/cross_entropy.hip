// Fused cross-entropy fwd/bwd for gfx950 (single-rank vocab; the
// vocab-parallel variant layers TP all_reduces in Python on top of the
// per-shard pieces). One 256-thread workgroup per row, online max+sumexp
// (one read pass), bf16x8 traffic; saves (max, sumexp) fp32 for bwd.
#include "common.h"

#define BLOCK 256

extern "C" __global__ void ce_fwd_kernel(
    const bf16raw *__restrict__ logits, const long *__restrict__ labels,
    float *__restrict__ loss, float *__restrict__ row_max,
    float *__restrict__ row_sum, long rows, int V) {
    __shared__ float redm[BLOCK / WAVE];
    __shared__ float reds[BLOCK / WAVE];
    for (long r = blockIdx.x; r < rows; r += gridDim.x) {
        const bf16raw *lr = logits + r * V;
        float m = -INFINITY, l = 0.f, label_logit = 0.f;
        const long lbl = labels[r];
        for (int i = threadIdx.x * 8; i < V; i += BLOCK * 8) {
            bf16x8 v = load8(lr + i);
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                float f = v.get(j);
                if (i + j == lbl) label_logit = f;
                if (f > m) {
                    l *= __expf(m - f);
                    m = f;
                }
                l += __expf(f - m);
            }
        }
        // combine lanes/waves: m_tot then rescaled sums
        __syncthreads();
        float m_tot = block_max<BLOCK>(m, redm);
        __syncthreads();
        float l_scaled = l * __expf(m - m_tot);
        float l_tot = block_sum<BLOCK>(l_scaled, reds);
        // label_logit is non-zero on exactly one thread; sum broadcasts it
        __syncthreads();
        float ll = block_sum<BLOCK>(label_logit, redm);
        if (threadIdx.x == 0) {
            row_max[r] = m_tot;
            row_sum[r] = l_tot;
            loss[r] = __logf(l_tot) + m_tot - ll;
        }
        __syncthreads();
    }
}

extern "C" __global__ void ce_bwd_kernel(
    const bf16raw *__restrict__ logits, const long *__restrict__ labels,
    const float *__restrict__ dloss, const float *__restrict__ row_max,
    const float *__restrict__ row_sum, bf16raw *__restrict__ dlogits,
    long rows, int V) {
    for (long r = blockIdx.x; r < rows; r += gridDim.x) {
        const bf16raw *lr = logits + r * V;
        bf16raw *dr = dlogits + r * V;
        const float m = row_max[r], inv = 1.f / row_sum[r], d = dloss[r];
        const long lbl = labels[r];
        for (int i = threadIdx.x * 8; i < V; i += BLOCK * 8) {
            bf16x8 v = load8(lr + i);
            bf16x8 o;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                float p = __expf(v.get(j) - m) * inv;
                if (i + j == lbl) p -= 1.f;
                o.set(j, p * d);
            }
            store8(dr + i, o);
        }
    }
}

extern "C" void ce_fwd_launch(const void *logits, const void *labels,
                              void *loss, void *row_max, void *row_sum,
                              long rows, int V, hipStream_t stream) {
    int grid = rows < 2048 ? (int)rows : 2048;
    hipLaunchKernelGGL(ce_fwd_kernel, dim3(grid), dim3(BLOCK), 0, stream,
                       (const bf16raw *)logits, (const long *)labels,
                       (float *)loss, (float *)row_max, (float *)row_sum,
                       rows, V);
}

extern "C" void ce_bwd_launch(const void *logits, const void *labels,
                              const void *dloss, const void *row_max,
                              const void *row_sum, void *dlogits, long rows,
                              int V, hipStream_t stream) {
    int grid = rows < 2048 ? (int)rows : 2048;
    hipLaunchKernelGGL(ce_bwd_kernel, dim3(grid), dim3(BLOCK), 0, stream,
                       (const bf16raw *)logits, (const long *)labels,
                       (const float *)dloss, (const float *)row_max,
                       (const float *)row_sum, (bf16raw *)dlogits, rows, V);
}
