// Batched grouped GEMM for MoE expert layers (gfx950, MFMA 16x16x32 bf16).
//
// One launch covers ALL experts — the per-expert torch.mm loop is
// host-launch-bound at DeepSeek expert counts (E=160), and both
// hipBLASLt strided-batched bmm backward and torch._grouped_mm
// memory-fault on this ROCm stack (scripts/grouped_mm_probe.py), so the
// three training GEMMs are hand-written here:
//
//   fwd:   C[e][M,N]  = A[e][M,K]   @ W[e][N,K]^T      (bf16 out)
//   dgrad: dX[e][M,K] = dOut[e][M,N] @ W[e][N,K]       (bf16 out)
//   wgrad: G[e][N,K] += dOut[e][M,N]^T @ A[e][M,K]     (fp32 accumulate)
//
// Weights live in the natural torch Linear layout [E, out, in] = [E,N,K]:
// with C = A.W^T both MFMA fragments are plain contiguous 16-byte LDS
// reads (A-frag: row l&15, k-offset (l>>4)*8; B-frag: W row l&15, same
// offset) — no transposes anywhere in the fwd hot loop. dgrad stages the
// W tile as a ds_read_b64_tr_b16 image (same v_img_row permutation as
// attention.hip's V image); wgrad stages both dOut and X tiles as images.
//
// Tiling: 256 threads = 4 waves; workgroup tile 128x128, reduction step
// 32; wave w owns output rows [w*32, w*32+32) = 2 row-subtiles x 8
// col-subtiles of f32x4 accumulators. M (tokens per expert = capacity)
// is guarded; N and K must be multiples of 128 / 32 (asserted host-side
// — true for every registered MoE config).
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8v;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4v;
typedef __attribute__((address_space(3))) bf16x4v lds_b64_t;

#define GG_BLOCK 256
#define GG_TM 128
#define GG_TN 128
#define GG_RED 32
#define GG_PAD 36              // LDS row stride for [row][32] tiles
#define GG_SUB 520             // image subtile stride (32x16 + 8 pad)

DEV int gg_img_row(int r) {    // same permutation as attention's V image
    const int kg = r >> 3, j = r & 7;
    return kg * 4 + (j & 3) + ((j >> 2) << 4);
}

DEV bf16x8v gg_tr_frag(const bf16raw *sub_base, int lane) {
    bf16x4v lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
        (lds_b64_t *)(sub_base + lane * 4));
    bf16x4v hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
        (lds_b64_t *)(sub_base + 256 + lane * 4));
    bf16x8v r;
#pragma unroll
    for (int i = 0; i < 4; ++i) { r[i] = lo[i]; r[i + 4] = hi[i]; }
    return r;
}

DEV bf16x8v gg_frag(const bf16raw *p) {
    uint4 r = *reinterpret_cast<const uint4 *>(p);
    return *reinterpret_cast<bf16x8v *>(&r);
}

// ---------------------------------------------------------------------
// fwd: C[e][M,N] = A[e][M,K] @ W[e][N,K]^T
//
// K-step 128 (not 32): each thread stages one 256-byte contiguous run
// per tile per step — the 32-deep variant issued 2.5x more 64 B global
// transactions and its waves sat 3x longer in memory stalls (PMC study:
// SQ_WAIT_ANY 3.6e10 vs dgrad 1.3e10 at identical MFMA counts).
// ---------------------------------------------------------------------
#define GG_KSTEP 128
#define GG_KPAD 140            // row stride (elems): 70-dword rows, 16
                               // frag rows hit 16 distinct bank starts

__global__ __launch_bounds__(GG_BLOCK, 2)
void gg_fwd_kernel(const bf16raw *__restrict__ A, const bf16raw *__restrict__ W,
                   bf16raw *__restrict__ C, int E, int M, int N, int K) {
    __shared__ __attribute__((aligned(16))) bf16raw a_lds[GG_TM * GG_KPAD];
    __shared__ __attribute__((aligned(16))) bf16raw w_lds[GG_TN * GG_KPAD];
    const int tiles_m = (M + GG_TM - 1) / GG_TM;
    const int m_tile = blockIdx.x % tiles_m;
    const int n_tile = blockIdx.x / tiles_m;
    const int e = blockIdx.y;
    const bf16raw *Ae = A + (long)e * M * K + (long)m_tile * GG_TM * K;
    const bf16raw *We = W + (long)e * N * K + (long)n_tile * GG_TN * K;
    bf16raw *Ce = C + (long)e * M * N + (long)m_tile * GG_TM * N
                  + (long)n_tile * GG_TN;
    const int mrem = M - m_tile * GG_TM;   // rows valid in this tile

    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int tid = threadIdx.x;
    // staging: 2 threads per row, 128 B (8 elems x 8) contiguous each
    const int srow = tid / 2;              // 0..127
    const int scol = (tid % 2) * 64;       // 0 or 64

    f32x4 acc[2][8];
#pragma unroll
    for (int rs = 0; rs < 2; ++rs)
#pragma unroll
        for (int cs = 0; cs < 8; ++cs) acc[rs][cs] = f32x4{0.f, 0.f, 0.f, 0.f};

    // T14 split staging: tile k0+GG_KSTEP's loads are issued into
    // registers while tile k0 computes (r1 PMC: the synchronous
    // stage->barrier->compute loop left fwd 3x more wait-parked than
    // dgrad at identical MFMA work)
    bf16x8 st_a[8], st_w[8];
    auto stage_load = [&](int k0) {
        if (k0 >= K) return;
        const int arow = srow < mrem ? srow : 0;   // clamp; zeroed on write
#pragma unroll
        for (int c = 0; c < 8; ++c)
            st_a[c] = load8(Ae + (long)arow * K + k0 + scol + c * 8);
#pragma unroll
        for (int c = 0; c < 8; ++c)
            st_w[c] = load8(We + (long)srow * K + k0 + scol + c * 8);
    };
    auto stage_write = [&]() {
        if (srow < mrem) {
#pragma unroll
            for (int c = 0; c < 8; ++c)
                store8(a_lds + srow * GG_KPAD + scol + c * 8, st_a[c]);
        } else {
            const uint4 z{0, 0, 0, 0};
#pragma unroll
            for (int c = 0; c < 8; ++c)
                *reinterpret_cast<uint4 *>(a_lds + srow * GG_KPAD + scol + c * 8)
                    = z;
        }
#pragma unroll
        for (int c = 0; c < 8; ++c)
            store8(w_lds + srow * GG_KPAD + scol + c * 8, st_w[c]);
    };
    stage_load(0);
    stage_write();
    __syncthreads();
    for (int k0 = 0; k0 < K; k0 += GG_KSTEP) {
        stage_load(k0 + GG_KSTEP);
#pragma unroll
        for (int kc = 0; kc < GG_KSTEP / 32; ++kc) {
            bf16x8v a0 = gg_frag(a_lds + (wave * 32 + (lane & 15)) * GG_KPAD
                                 + kc * 32 + (lane >> 4) * 8);
            bf16x8v a1 = gg_frag(a_lds + (wave * 32 + 16 + (lane & 15)) * GG_KPAD
                                 + kc * 32 + (lane >> 4) * 8);
#pragma unroll
            for (int cs = 0; cs < 8; ++cs) {
                bf16x8v b = gg_frag(w_lds + (cs * 16 + (lane & 15)) * GG_KPAD
                                    + kc * 32 + (lane >> 4) * 8);
                acc[0][cs] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a0, b, acc[0][cs], 0, 0, 0);
                acc[1][cs] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a1, b, acc[1][cs], 0, 0, 0);
            }
        }
        __syncthreads();   // all waves done reading this k tile
        stage_write();
        __syncthreads();   // next tile visible
    }
#pragma unroll
    for (int rs = 0; rs < 2; ++rs)
#pragma unroll
        for (int cs = 0; cs < 8; ++cs)
#pragma unroll
            for (int i = 0; i < 4; ++i) {
                const int r = wave * 32 + rs * 16 + (lane >> 4) * 4 + i;
                if (r < mrem)
                    Ce[(long)r * N + cs * 16 + (lane & 15)] = f2bf(acc[rs][cs][i]);
            }
}

// ---------------------------------------------------------------------
// dgrad: dX[e][M,K] = dOut[e][M,N] @ W[e][N,K]   (reduction over n)
// ---------------------------------------------------------------------
__global__ __launch_bounds__(GG_BLOCK, 2)
void gg_dgrad_kernel(const bf16raw *__restrict__ dOut,
                     const bf16raw *__restrict__ W,
                     bf16raw *__restrict__ dX, int E, int M, int N, int K) {
    __shared__ __attribute__((aligned(16))) bf16raw a_lds[GG_TM * GG_PAD];
    __shared__ __attribute__((aligned(16))) bf16raw w_img[8 * GG_SUB];
    const int tiles_m = (M + GG_TM - 1) / GG_TM;
    const int m_tile = blockIdx.x % tiles_m;
    const int k_tile = blockIdx.x / tiles_m;    // output-column tile of K
    const int e = blockIdx.y;
    const bf16raw *De = dOut + (long)e * M * N + (long)m_tile * GG_TM * N;
    const bf16raw *We = W + (long)e * N * K + (long)k_tile * GG_TN;
    bf16raw *Xe = dX + (long)e * M * K + (long)m_tile * GG_TM * K
                  + (long)k_tile * GG_TN;
    const int mrem = M - m_tile * GG_TM;

    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int tid = threadIdx.x;
    const int srow = tid / 4;
    const int scol = (tid % 4) * 8;
    // image staging: 32 n-rows x 128 k-cols; thread -> (n = tid/32, two
    // 8-elem chunks at cols (tid%32... use: each thread writes 16 elems:
    // rows tid/8 (32 rows), cols (tid%8)*16 .. +16 in two store8
    const int irow = tid / 8;
    const int icol0 = (tid % 8) * 16;

    f32x4 acc[2][8];
#pragma unroll
    for (int rs = 0; rs < 2; ++rs)
#pragma unroll
        for (int cs = 0; cs < 8; ++cs) acc[rs][cs] = f32x4{0.f, 0.f, 0.f, 0.f};

    for (int n0 = 0; n0 < N; n0 += GG_RED) {
        __syncthreads();
#pragma unroll
        for (int h = 0; h < 2; ++h) {
            const int r = srow + h * 64;
            bf16x8 dv;
            if (r < mrem) dv = load8(De + (long)r * N + n0 + scol);
            else dv.raw = uint4{0, 0, 0, 0};
            store8(a_lds + r * GG_PAD + scol, dv);
        }
        // W image: rows = n (32), cols = k (128)
#pragma unroll
        for (int c = 0; c < 2; ++c) {
            const int d0 = icol0 + c * 8;
            store8(w_img + (d0 >> 4) * GG_SUB + gg_img_row(irow) * 16 + (d0 & 15),
                   load8(We + (long)(n0 + irow) * K + d0));
        }
        __syncthreads();
        bf16x8v a0 = gg_frag(a_lds + (wave * 32 + (lane & 15)) * GG_PAD
                             + (lane >> 4) * 8);
        bf16x8v a1 = gg_frag(a_lds + (wave * 32 + 16 + (lane & 15)) * GG_PAD
                             + (lane >> 4) * 8);
#pragma unroll
        for (int cs = 0; cs < 8; ++cs) {
            bf16x8v b = gg_tr_frag(w_img + cs * GG_SUB, lane);
            acc[0][cs] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a0, b, acc[0][cs], 0, 0, 0);
            acc[1][cs] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a1, b, acc[1][cs], 0, 0, 0);
        }
    }
#pragma unroll
    for (int rs = 0; rs < 2; ++rs)
#pragma unroll
        for (int cs = 0; cs < 8; ++cs)
#pragma unroll
            for (int i = 0; i < 4; ++i) {
                const int r = wave * 32 + rs * 16 + (lane >> 4) * 4 + i;
                if (r < mrem)
                    Xe[(long)r * K + cs * 16 + (lane & 15)] = f2bf(acc[rs][cs][i]);
            }
}

// ---------------------------------------------------------------------
// wgrad: G[e][N,K] += dOut[e][M,N]^T @ A[e][M,K]  (fp32 accumulate,
// reduction over m — both operands staged as tr images)
// ---------------------------------------------------------------------
__global__ __launch_bounds__(GG_BLOCK, 2)
void gg_wgrad_kernel(const bf16raw *__restrict__ dOut,
                     const bf16raw *__restrict__ A,
                     float *__restrict__ G, int E, int M, int N, int K) {
    __shared__ __attribute__((aligned(16))) bf16raw d_img[8 * GG_SUB];
    __shared__ __attribute__((aligned(16))) bf16raw a_img[8 * GG_SUB];
    const int tiles_n = N / GG_TN;
    const int n_tile = blockIdx.x % tiles_n;
    const int k_tile = blockIdx.x / tiles_n;
    const int e = blockIdx.y;
    const bf16raw *De = dOut + (long)e * M * N + (long)n_tile * GG_TN;
    const bf16raw *Ae = A + (long)e * M * K + (long)k_tile * GG_TN;
    float *Ge = G + (long)e * N * K + (long)n_tile * GG_TN * K
                + (long)k_tile * GG_TN;

    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int tid = threadIdx.x;
    const int irow = tid / 8;             // m row within the 32-step
    const int icol0 = (tid % 8) * 16;

    f32x4 acc[2][8];
#pragma unroll
    for (int rs = 0; rs < 2; ++rs)
#pragma unroll
        for (int cs = 0; cs < 8; ++cs) acc[rs][cs] = f32x4{0.f, 0.f, 0.f, 0.f};

    for (int m0 = 0; m0 < M; m0 += GG_RED) {
        __syncthreads();
        const int mr = m0 + irow;
#pragma unroll
        for (int c = 0; c < 2; ++c) {
            const int d0 = icol0 + c * 8;
            bf16x8 dv, av;
            if (mr < M) {
                dv = load8(De + (long)mr * N + d0);
                av = load8(Ae + (long)mr * K + d0);
            } else {
                dv.raw = uint4{0, 0, 0, 0};
                av.raw = uint4{0, 0, 0, 0};
            }
            store8(d_img + (d0 >> 4) * GG_SUB + gg_img_row(irow) * 16 + (d0 & 15), dv);
            store8(a_img + (d0 >> 4) * GG_SUB + gg_img_row(irow) * 16 + (d0 & 15), av);
        }
        __syncthreads();
        // A'-frag (rows = n): lane supplies dOut^T[n=l&15][m=(l>>4)*8+j]
        // from the dOut image; B-frag: A[m][k] from the A image. Output
        // row-subtiles rs of wave -> n sub-indices wave*2+{0,1}.
        bf16x8v a0 = gg_tr_frag(d_img + (wave * 2) * GG_SUB, lane);
        bf16x8v a1 = gg_tr_frag(d_img + (wave * 2 + 1) * GG_SUB, lane);
#pragma unroll
        for (int cs = 0; cs < 8; ++cs) {
            bf16x8v b = gg_tr_frag(a_img + cs * GG_SUB, lane);
            acc[0][cs] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a0, b, acc[0][cs], 0, 0, 0);
            acc[1][cs] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a1, b, acc[1][cs], 0, 0, 0);
        }
    }
#pragma unroll
    for (int rs = 0; rs < 2; ++rs)
#pragma unroll
        for (int cs = 0; cs < 8; ++cs)
#pragma unroll
            for (int i = 0; i < 4; ++i) {
                const int n = (wave * 2 + rs) * 16 + (lane >> 4) * 4 + i;
                float *gp = Ge + (long)n * K + cs * 16 + (lane & 15);
                *gp += acc[rs][cs][i];
            }
}

// ---------------------------------------------------------------------
// host wrappers
// ---------------------------------------------------------------------
extern "C" {

void gg_fwd(const void *A, const void *W, void *C, int E, int M, int N,
            int K, hipStream_t stream) {
    const int tiles = ((M + GG_TM - 1) / GG_TM) * (N / GG_TN);
    hipLaunchKernelGGL(gg_fwd_kernel, dim3(tiles, E), dim3(GG_BLOCK), 0,
                       stream, (const bf16raw *)A, (const bf16raw *)W,
                       (bf16raw *)C, E, M, N, K);
}

void gg_dgrad(const void *dOut, const void *W, void *dX, int E, int M,
              int N, int K, hipStream_t stream) {
    const int tiles = ((M + GG_TM - 1) / GG_TM) * (K / GG_TN);
    hipLaunchKernelGGL(gg_dgrad_kernel, dim3(tiles, E), dim3(GG_BLOCK), 0,
                       stream, (const bf16raw *)dOut, (const bf16raw *)W,
                       (bf16raw *)dX, E, M, N, K);
}

void gg_wgrad(const void *dOut, const void *A, void *G, int E, int M,
              int N, int K, hipStream_t stream) {
    const int tiles = (N / GG_TN) * (K / GG_TN);
    hipLaunchKernelGGL(gg_wgrad_kernel, dim3(tiles, E), dim3(GG_BLOCK), 0,
                       stream, (const bf16raw *)dOut, (const bf16raw *)A,
                       (float *)G, E, M, N, K);
}

}  // extern "C"
