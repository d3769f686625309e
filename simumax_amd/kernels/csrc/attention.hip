// Flash attention (fwd + bwd) for gfx950, bf16, GQA, causal, D=128.
//
// v2 forward geometry: one 256-thread workgroup (4 waves) owns 128 q rows
// of one (batch, head); each wave owns 32 q rows (two 16-row MFMA subtiles).
// K[32][136] (row-padded: a 256-B-stride row-major tile read with
// ds_read_b128 by 16-lane groups is an up-to-16-way bank conflict —
// cdna_hip_programming.md G4) and V^T[128][40] tiles are staged in LDS,
// double-buffered so tile t+1's global loads overlap tile t's compute with
// ONE barrier per tile. QK^T and PV use v_mfma_f32_16x16x32_bf16; the
// softmax'd P crosses C-layout -> A-layout through a padded per-wave LDS
// buffer. Online softmax in registers; saves LSE fp32 for backward.
//
// Backward: one workgroup (4 waves) owns a 128-key block (32 keys/wave),
// loops over 32-row q tiles staged in (padded) LDS. dK/dV accumulate in
// registers; dQ via fp32 atomics.
//
// Reference behavior target: sdp_fwd/sdp_bwd priced by
// simumax/core/transformer/dense_module.py:1061-1605.
#include "common.h"
#include <cstdlib>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8v;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(2))) int v2i;

#define FA_BLOCK 256
#define QTILE 32            // q rows per wave (two 16-row subtiles)
#define KVTILE 32
#define DHEAD 128
#define WAVES 4
#define KS 144              // padded K row stride (elems); stride%128==16
// makes (col*stride + kgrp*8) distinct banks within every ds_read_b128
// 16-lane group (a +8 pad leaves a 2-way conflict: stride dwords % 16 == 4)
#define VTS 48              // padded V^T row stride

DEV bf16x8v ld_frag(const bf16raw *p) {
    uint4 r = *reinterpret_cast<const uint4 *>(p);
    return *reinterpret_cast<bf16x8v *>(&r);
}

// LDS (fwd): 2 x double-buffered { K rows, tr-readable V image }; P stays
// in registers (swapped-QK^T layout)
// V is staged as a tr-read image: 8 d-subtiles of [32 permuted key rows][16],
// each padded to VSUB elems; B-fragments come from two ds_read_b64_tr_b16
// per MFMA (guide T10: lane l elem j reads lds[(l&15) + j*16 + (l>>4)*64]).
// Image row perm: key = kgrp*8+j  ->  row = kgrp*4 + (j&3) + (j>=4)*16.
#define VSUB 520
#define FWD_BUF_ELEMS (KVTILE * KS + 8 * VSUB)

typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4v;
typedef __attribute__((address_space(3))) bf16x4v lds_b64_t;

DEV int v_img_row(int key) {
    const int kg = key >> 3, j = key & 7;
    return kg * 4 + (j & 3) + ((j >> 2) << 4);
}

DEV bf16x8v tr_frag(const bf16raw *sub_base, int lane) {
    // two transpose reads: rows [0,16) then [16,32) of one d-subtile.
    // Per-lane address = base + lane*8B; the instruction redistributes so
    // lane l receives elems (l&15) + j*16 + (l>>4)*64 (guide T10).
    bf16x4v lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
        (lds_b64_t *)(sub_base + lane * 4));
    bf16x4v hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
        (lds_b64_t *)(sub_base + 256 + lane * 4));
    bf16x8v r;
#pragma unroll
    for (int i = 0; i < 4; ++i) { r[i] = lo[i]; r[i + 4] = hi[i]; }
    return r;
}

// v3 forward geometry (guide-structure: 8-wave/512-thread workgroup,
// 2 waves per SIMD, one workgroup per CU):
//   * the workgroup owns a 256-row q block; wave w owns TWO 16-row
//     subtiles, w and 15-w — with causal masking each wave then sees the
//     same total kv work (load balance across the 2-wave SIMD pairing)
//   * 64-key kv tiles (half the barriers of v2's 32), double-buffered with
//     T14 split staging (issue loads early, LDS writes after compute)
//   * static s_setprio(1) for the younger wave half (guide T5 static form)
//   * V (and the second half of K's tr image) keep the v2 ds_read_b64_tr_b16
//     image: two 32-key groups per 64-key tile
// DQK: q/k head dim (128 dense/GQA, 192 MLA); v/o stay 128 (DHEAD).
#define FWD_BLOCK 512
#define FWD_WAVES 8
#define KVT2 64             // kv tile (two 32-key tr-image groups)

template <int DQK>
__global__ __launch_bounds__(FWD_BLOCK, 2)
void fa_fwd_kernel(const bf16raw *__restrict__ q, const bf16raw *__restrict__ k,
                   const bf16raw *__restrict__ v, bf16raw *__restrict__ o,
                   float *__restrict__ lse, int B, int S, int Hq, int Hkv,
                   int causal) {
    constexpr int NQS = 2;                // two causal-balanced subtiles
    constexpr int KC = DQK / 32;          // 32-deep k chunks of QK^T
    constexpr int KS_T = DQK + 16;        // padded K row stride
    constexpr int BUF_ELEMS = KVT2 * KS_T + 2 * 8 * VSUB;
    extern __shared__ __attribute__((aligned(16))) char smem[];
    bf16raw *buf0 = reinterpret_cast<bf16raw *>(smem);

    const int qblk = blockIdx.x;            // 256-row q block
    const int h = blockIdx.y;
    const int b = blockIdx.z;
    const int hkv = h / (Hq / Hkv);
    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int col = lane & 15;
    const int kgrp = lane >> 4;

    // wave w owns subtiles w (low rows) and 15-w (high rows)
    const int sub[NQS] = {wave, 2 * FWD_WAVES - 1 - wave};
    const int blk_rows = FWD_WAVES * 2 * 16;   // 256
    int qsb[NQS];
#pragma unroll
    for (int qs = 0; qs < NQS; ++qs) qsb[qs] = qblk * blk_rows + sub[qs] * 16;
    const float scale = rsqrtf((float)DQK);

    const long q_row = (long)Hq * DQK;        // q layout [B,S,Hq,DQK]
    const long k_row = (long)Hkv * DQK;       // k layout [B,S,Hkv,DQK]
    const long v_row = (long)Hkv * DHEAD;     // v layout [B,S,Hkv,128]
    const long o_row = (long)Hq * DHEAD;      // o layout [B,S,Hq,128]
    const bf16raw *qp = q + ((long)b * S) * q_row + (long)h * DQK;
    const bf16raw *kp = k + ((long)b * S) * k_row + (long)hkv * DQK;
    const bf16raw *vp = v + ((long)b * S) * v_row + (long)hkv * DHEAD;

    bf16x8v a_q[NQS][KC];
#pragma unroll
    for (int qs = 0; qs < NQS; ++qs) {
        const int qrow = qsb[qs] + col;
#pragma unroll
        for (int kc = 0; kc < KC; ++kc) {
            bf16x8 raw = load8(qp + (long)min(qrow, S - 1) * q_row + kc * 32 + kgrp * 8);
            bf16x8 sc;
#pragma unroll
            for (int j = 0; j < 8; ++j) sc.set(j, raw.get(j) * scale);
            a_q[qs][kc] = *reinterpret_cast<bf16x8v *>(&sc.raw);
        }
    }

    // swapped-QK^T layout (guide T12): S^T = mfma(A=K, B=Q) puts one q
    // COLUMN per lane, so the softmax row stats are (nearly) lane-local
    // and P never round-trips through LDS. m/l are per-lane scalars.
    float m[NQS], l[NQS];
    f32x4 acc[NQS][8];        // O^T: lane holds col q, rows d
#pragma unroll
    for (int qs = 0; qs < NQS; ++qs) {
        m[qs] = -INFINITY;
        l[qs] = 0.f;
    }
#pragma unroll
    for (int qs = 0; qs < NQS; ++qs)
#pragma unroll
        for (int dt = 0; dt < 8; ++dt) acc[qs][dt] = f32x4{0.f, 0.f, 0.f, 0.f};

    const int kv_limit = causal ? min(S, qblk * blk_rows + blk_rows) : S;
    const int n_tiles = CDIV(kv_limit, KVT2);

    // T14 split staging over the 512-thread workgroup
    constexpr int KCHUNKS = KVT2 * DQK / (FWD_BLOCK * 8);   // per-thread K pieces
    constexpr int VCHUNKS = KVT2 * DHEAD / (FWD_BLOCK * 8);
    bf16x8 st_k[KCHUNKS], st_v[VCHUNKS];
    auto stage_load = [&](int t) {
        if (t >= n_tiles) return;
        const int kv = t * KVT2;
        const int tid = threadIdx.x;
#pragma unroll
        for (int c = 0; c < KCHUNKS; ++c) {
            const int e = tid * 8 + c * FWD_BLOCK * 8;
            const int kvr = e / DQK, d0 = e % DQK;
            const int src = min(kv + kvr, S - 1);
            st_k[c] = load8(kp + (long)src * k_row + d0);
        }
#pragma unroll
        for (int c = 0; c < VCHUNKS; ++c) {
            const int e = tid * 8 + c * FWD_BLOCK * 8;
            const int kvr = e / DHEAD, d0 = e % DHEAD;
            const int src = min(kv + kvr, S - 1);
            st_v[c] = load8(vp + (long)src * v_row + d0);
        }
    };
    auto stage_write = [&](int t) {
        if (t >= n_tiles) return;
        bf16raw *K_lds = buf0 + (t & 1) * BUF_ELEMS;
        bf16raw *V_img = K_lds + KVT2 * KS_T;
        const int tid = threadIdx.x;
#pragma unroll
        for (int c = 0; c < KCHUNKS; ++c) {
            const int e = tid * 8 + c * FWD_BLOCK * 8;
            const int kvr = e / DQK, d0 = e % DQK;
            store8(K_lds + kvr * KS_T + d0, st_k[c]);
        }
#pragma unroll
        for (int c = 0; c < VCHUNKS; ++c) {
            const int e = tid * 8 + c * FWD_BLOCK * 8;
            const int kvr = e / DHEAD, d0 = e % DHEAD;
            // 32-key group (kvr>>5), subtile d0/16, row perm(kvr&31)
            store8(V_img + (kvr >> 5) * 8 * VSUB + (d0 >> 4) * VSUB
                       + v_img_row(kvr & 31) * 16 + (d0 & 15),
                   st_v[c]);
        }
    };

    stage_load(0);
    stage_write(0);
    __syncthreads();
    stage_load(1);
    // static priority for the younger wave half (T5 static form): the
    // condition must be provably wave-uniform or s_setprio goes under exec
    if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= FWD_BLOCK / 2)
        __builtin_amdgcn_s_setprio(1);

    for (int t = 0; t < n_tiles; ++t) {
        const int kv = t * KVT2;
        bf16raw *K_lds = buf0 + (t & 1) * BUF_ELEMS;
        bf16raw *V_img = K_lds + KVT2 * KS_T;

        // subtile 1 (high rows) is always active (its last row is the
        // block's last row >= kv_limit); subtile 0 (low rows) goes inactive
        // once the causal diagonal passes it. In the common both-active case
        // the K and V LDS fragments are read ONCE and feed both subtiles'
        // MFMA chains (halves the LDS read traffic).
        const bool act0 = !(causal && kv >= qsb[0] + 16);

        f32x4 sq[NQS][4];
#pragma unroll
        for (int qs = 0; qs < NQS; ++qs)
#pragma unroll
            for (int ks = 0; ks < 4; ++ks) sq[qs][ks] = f32x4{0, 0, 0, 0};
        // swapped QK^T: A = K subtile fragments (same bytes as the old
        // B-fragments), B = the pre-scaled Q fragments (same registers) ->
        // S^T[kv][q]: lane holds col q = lane&15, rows kv = ks*16+kgrp*4+j
        if (act0) {
#pragma unroll
            for (int kc = 0; kc < KC; ++kc)
#pragma unroll
                for (int ks = 0; ks < 4; ++ks) {
                    bf16x8v bk = ld_frag(K_lds + (ks * 16 + col) * KS_T + kc * 32 + kgrp * 8);
                    sq[0][ks] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(bk, a_q[0][kc], sq[0][ks], 0, 0, 0);
                    sq[1][ks] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(bk, a_q[1][kc], sq[1][ks], 0, 0, 0);
                }
        } else {
#pragma unroll
            for (int kc = 0; kc < KC; ++kc)
#pragma unroll
                for (int ks = 0; ks < 4; ++ks) {
                    bf16x8v bk = ld_frag(K_lds + (ks * 16 + col) * KS_T + kc * 32 + kgrp * 8);
                    sq[1][ks] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(bk, a_q[1][kc], sq[1][ks], 0, 0, 0);
                }
        }

        // online softmax (lane-local over 16 values + 2 xor-shuffles
        // across the 4 kgrp lanes of each q column) + in-register P via
        // cvt_pk + the permlane32/16 butterfly -> PV B-fragments
        bf16x8v b_p[NQS][2];
        auto softmax_p = [&](int qs) {
            const bool clean = (qsb[qs] + 16 <= S) && (kv + KVT2 <= S)
                && (!causal || kv + KVT2 - 1 <= qsb[qs]);
            if (!clean) {
                const int qrow = qsb[qs] + col;
#pragma unroll
                for (int ks = 0; ks < 4; ++ks)
#pragma unroll
                    for (int j = 0; j < 4; ++j) {
                        const int c0 = kv + ks * 16 + kgrp * 4 + j;
                        if (qrow >= S || c0 >= S || (causal && c0 > qrow))
                            sq[qs][ks][j] = -INFINITY;
                    }
            }
            float mx = sq[qs][0][0];
#pragma unroll
            for (int ks = 0; ks < 4; ++ks)
#pragma unroll
                for (int j = 0; j < 4; ++j) mx = fmaxf(mx, sq[qs][ks][j]);
            mx = fmaxf(mx, __shfl_xor(mx, 16, WAVE));
            mx = fmaxf(mx, __shfl_xor(mx, 32, WAVE));
            const float mn = fmaxf(m[qs], mx);
            // exp(-inf - mn) = 0, so the masked entries cost nothing extra
            const float alpha = __expf(m[qs] - mn);
            m[qs] = mn;
            float ps = 0.f;
#pragma unroll
            for (int ks = 0; ks < 4; ++ks)
#pragma unroll
                for (int j = 0; j < 4; ++j) {
                    float pv = __expf(sq[qs][ks][j] - mn);
                    sq[qs][ks][j] = pv;
                    ps += pv;
                }
            // l kept per-lane (per kgrp); reduced once in the epilogue
            l[qs] = l[qs] * alpha + ps;
#pragma unroll
            for (int dt = 0; dt < 8; ++dt)
#pragma unroll
                for (int j = 0; j < 4; ++j) acc[qs][dt][j] *= alpha;
            // pack P to bf16 and butterfly into PV B-fragments: target
            // lane kgrp k' holds kv 32t + 8k' + [0,8) of its q column
#pragma unroll
            for (int t2 = 0; t2 < 2; ++t2) {
                int E0, E1, F0, F1;
                asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(E0)
                    : "v"(sq[qs][2 * t2][0]), "v"(sq[qs][2 * t2][1]));
                asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(E1)
                    : "v"(sq[qs][2 * t2][2]), "v"(sq[qs][2 * t2][3]));
                asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(F0)
                    : "v"(sq[qs][2 * t2 + 1][0]), "v"(sq[qs][2 * t2 + 1][1]));
                asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(F1)
                    : "v"(sq[qs][2 * t2 + 1][2]), "v"(sq[qs][2 * t2 + 1][3]));
                v2i u0 = __builtin_amdgcn_permlane32_swap(E0, F0, false, false);
                v2i u1 = __builtin_amdgcn_permlane32_swap(E1, F1, false, false);
                v2i d02 = __builtin_amdgcn_permlane16_swap(u0[0], u0[1], false, false);
                v2i d13 = __builtin_amdgcn_permlane16_swap(u1[0], u1[1], false, false);
                int frag[4] = {d02[0], d13[0], d02[1], d13[1]};
                b_p[qs][t2] = *reinterpret_cast<bf16x8v *>(frag);
            }
        };
        if (act0) softmax_p(0);
        softmax_p(1);

        // PV swapped: O^T += mfma(A = V^T tr-fragments (same bytes as the
        // old B side), B = in-register P^T fragments)
#pragma unroll
        for (int g = 0; g < 2; ++g) {
            if (act0) {
#pragma unroll
                for (int dt = 0; dt < 8; ++dt) {
                    bf16x8v b_v = tr_frag(V_img + g * 8 * VSUB + dt * VSUB, lane);
                    acc[0][dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        b_v, b_p[0][g], acc[0][dt], 0, 0, 0);
                    acc[1][dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        b_v, b_p[1][g], acc[1][dt], 0, 0, 0);
                }
            } else {
#pragma unroll
                for (int dt = 0; dt < 8; ++dt) {
                    bf16x8v b_v = tr_frag(V_img + g * 8 * VSUB + dt * VSUB, lane);
                    acc[1][dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        b_v, b_p[1][g], acc[1][dt], 0, 0, 0);
                }
            }
        }
        stage_write(t + 1);
        __syncthreads();
        stage_load(t + 2);
    }

    bf16raw *op = o + ((long)b * S) * o_row + (long)h * DHEAD;
    // O^T epilogue: lane owns q column `col`; its 32 acc values are d =
    // dt*16 + kgrp*4 + [0,4) — contiguous, so each dt is ONE 8-byte store
#pragma unroll
    for (int qs = 0; qs < NQS; ++qs) {
        float lt = l[qs];
        lt += __shfl_xor(lt, 16, WAVE);
        lt += __shfl_xor(lt, 32, WAVE);
        const int row = qsb[qs] + col;
        if (row >= S) continue;
        const float inv = (lt > 0.f) ? 1.f / lt : 0.f;
#pragma unroll
        for (int dt = 0; dt < 8; ++dt) {
            const unsigned w0 =
                (unsigned)f2bf(acc[qs][dt][0] * inv) |
                ((unsigned)f2bf(acc[qs][dt][1] * inv) << 16);
            const unsigned w1 =
                (unsigned)f2bf(acc[qs][dt][2] * inv) |
                ((unsigned)f2bf(acc[qs][dt][3] * inv) << 16);
            *reinterpret_cast<uint2 *>(
                op + (long)row * o_row + dt * 16 + kgrp * 4) = uint2{w0, w1};
        }
        if (kgrp == 0)
            lse[((long)b * Hq + h) * S + row] =
                (lt > 0.f) ? m[qs] + __logf(lt) : -INFINITY;
    }
}

// ==========================================================================
// backward — FA2-style split (no inner-loop atomics):
//   1. fa_bwd_pre:  D[b,h,s] = rowsum(dO * O)
//   2. fa_bwd_dkv:  kv-parallel; each wave owns 16 keys, accumulates dK/dV
//      in registers over all q tiles, one atomic add per element at the end
//      (atomics only because GQA q-heads share a kv head)
//   3. fa_bwd_dq:   q-parallel like the forward; dQ written directly (bf16)
// ==========================================================================
#define BW_QT 32
#define QS2 144   // padded [32][128] row stride
#define BKV 16    // keys per wave in dkv kernel

extern "C" __global__ __launch_bounds__(FA_BLOCK)
void fa_bwd_pre_kernel(const bf16raw *__restrict__ dout,
                       const bf16raw *__restrict__ o, float *__restrict__ D,
                       int B, int S, int Hq) {
    // one wave per (b,h,s) row: 128 elems, 2 per lane
    const long rows = (long)B * Hq * S;
    const long row0 = ((long)blockIdx.x * (FA_BLOCK / WAVE) + threadIdx.x / WAVE);
    const int lane = threadIdx.x % WAVE;
    for (long r = row0; r < rows; r += (long)gridDim.x * (FA_BLOCK / WAVE)) {
        // r = (b*Hq + h)*S + s ; dout layout [B,S,Hq,D]
        const long s_ = r % S;
        const long bh = r / S;
        const long h = bh % Hq;
        const long b = bh / Hq;
        const long base = (((long)b * S + s_) * Hq + h) * DHEAD;
        // load 2 bf16 from dout and o (4 bytes each)
        const unsigned du = *reinterpret_cast<const unsigned *>(&dout[base + lane * 2]);
        const unsigned ou = *reinterpret_cast<const unsigned *>(&o[base + lane * 2]);
        float acc = bf2f((bf16raw)(du & 0xffff)) * bf2f((bf16raw)(ou & 0xffff)) +
                    bf2f((bf16raw)(du >> 16)) * bf2f((bf16raw)(ou >> 16));
        acc = wave_sum(acc);
        if (lane == 0) D[r] = acc;
    }
}

// ---- dK/dV: wave owns BKV keys; q-side tiles staged per WG ----
template <int DQK>
__global__ __launch_bounds__(FA_BLOCK)
void fa_bwd_dkv_kernel(const bf16raw *__restrict__ dout,
                       const bf16raw *__restrict__ q,
                       const bf16raw *__restrict__ k,
                       const bf16raw *__restrict__ v,
                       const float *__restrict__ lse,
                       const float *__restrict__ Dsum,
                       float *__restrict__ dkv,  // [B,S,Hkv,2,D] fp32
                       int B, int S, int Hq, int Hkv, int causal) {
    constexpr int KC = DQK / 32;
    constexpr int KS_T = DQK + 16;
    constexpr int QSUB = DQK / 16;     // q-side tr-image d subtiles for Q
    constexpr int QS_T = DQK + 16;     // padded Q row stride
    extern __shared__ __attribute__((aligned(16))) char smem[];
    bf16raw *Q_lds = reinterpret_cast<bf16raw *>(smem);      // [32][QS_T]
    bf16raw *dO_lds = Q_lds + BW_QT * QS_T;                  // [32][QS2]
    bf16raw *Q_img = dO_lds + BW_QT * QS2;                   // tr image QSUB*VSUB
    bf16raw *dO_img = Q_img + QSUB * VSUB;                   // tr image 8*VSUB
    float *lse_lds = reinterpret_cast<float *>(dO_img + 8 * VSUB);
    float *D_lds = lse_lds + BW_QT;
    bf16raw *wbase = reinterpret_cast<bf16raw *>(D_lds + BW_QT);
    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int col = lane & 15;
    const int kgrp = lane >> 4;
    const int WSZ = BKV * KS_T;
    bf16raw *K_l = wbase + wave * WSZ;         // [16][KS_T] scaled

    const int kvblk = blockIdx.x;
    const int h = blockIdx.y;
    const int b = blockIdx.z;
    const int hkv = h / (Hq / Hkv);
    const int kvbase = kvblk * (WAVES * BKV) + wave * BKV;

    const long q_row = (long)Hq * DQK;
    const long k_row = (long)Hkv * DQK;
    const long v_row = (long)Hkv * DHEAD;
    const long o_row = (long)Hq * DHEAD;
    const bf16raw *qp = q + ((long)b * S) * q_row + (long)h * DQK;
    const bf16raw *kp = k + ((long)b * S) * k_row + (long)hkv * DQK;
    const bf16raw *vp = v + ((long)b * S) * v_row + (long)hkv * DHEAD;
    const bf16raw *dop = dout + ((long)b * S) * o_row + (long)h * DHEAD;
    const float scale = rsqrtf((float)DQK);

    // stage this wave's scaled K; K and V B-fragments held in REGISTERS
    // across the whole q loop (kv-tile-invariant; K_l LDS keeps the tr
    // image source for dK)
    bf16x8v b_v[4], b_k[KC];
    {
        for (int e = lane * 8; e < BKV * DQK; e += WAVE * 8) {
            const int kvr = e / DQK, d0 = e % DQK;
            const int src = min(kvbase + kvr, S - 1);
            bf16x8 kk = load8(kp + (long)src * k_row + d0);
            bf16x8 ks;
#pragma unroll
            for (int j = 0; j < 8; ++j) ks.set(j, kk.get(j) * scale);
            store8(K_l + kvr * KS_T + d0, ks);
        }
        const int src = min(kvbase + col, S - 1);
#pragma unroll
        for (int kc = 0; kc < 4; ++kc)
            b_v[kc] = ld_frag(vp + (long)src * v_row + kc * 32 + kgrp * 8);
        __syncthreads();   // K_l visible (same wave wrote it, but be strict)
#pragma unroll
        for (int kc = 0; kc < KC; ++kc)
            b_k[kc] = ld_frag(K_l + col * KS_T + kc * 32 + kgrp * 8);
    }

    f32x4 dv_acc[8], dk_acc[QSUB];
#pragma unroll
    for (int dt = 0; dt < 8; ++dt) dv_acc[dt] = f32x4{0, 0, 0, 0};
#pragma unroll
    for (int dt = 0; dt < QSUB; ++dt) dk_acc[dt] = f32x4{0, 0, 0, 0};

    const int q_start = causal
        ? (kvblk * (WAVES * BKV) / BW_QT) * BW_QT : 0;
    // T14 split staging: tile qt+BW_QT's global loads are issued into
    // registers while tile qt computes; the LDS write happens between the
    // two barriers (r1 PMC: the serial stage->sync->compute loop left the
    // q-tile loads fully exposed at 1 block/CU)
    constexpr int QCH = BW_QT * DQK / (FA_BLOCK * 8);
    constexpr int OCH = BW_QT * DHEAD / (FA_BLOCK * 8);
    bf16x8 st_q[QCH], st_do[OCH];
    float st_lse = 0.f, st_D = 0.f;
    auto q_stage_load = [&](int qt) {
        if (qt >= S) return;
        const int tid = threadIdx.x;
#pragma unroll
        for (int c = 0; c < QCH; ++c) {
            const int e = tid * 8 + c * FA_BLOCK * 8;
            const int r = e / DQK, d0 = e % DQK;
            st_q[c] = load8(qp + (long)min(qt + r, S - 1) * q_row + d0);
        }
#pragma unroll
        for (int c = 0; c < OCH; ++c) {
            const int e = tid * 8 + c * FA_BLOCK * 8;
            const int r = e / DHEAD, d0 = e % DHEAD;
            st_do[c] = load8(dop + (long)min(qt + r, S - 1) * o_row + d0);
        }
        if (tid < BW_QT) {
            const int src = min(qt + tid, S - 1);
            st_lse = lse[((long)b * Hq + h) * S + src];
            st_D = Dsum[((long)b * Hq + h) * S + src];
        }
    };
    auto q_stage_write = [&](int qt) {
        if (qt >= S) return;
        const int tid = threadIdx.x;
#pragma unroll
        for (int c = 0; c < QCH; ++c) {
            const int e = tid * 8 + c * FA_BLOCK * 8;
            const int r = e / DQK, d0 = e % DQK;
            store8(Q_lds + r * QS_T + d0, st_q[c]);
            store8(Q_img + (d0 >> 4) * VSUB + v_img_row(r) * 16 + (d0 & 15),
                   st_q[c]);
        }
#pragma unroll
        for (int c = 0; c < OCH; ++c) {
            const int e = tid * 8 + c * FA_BLOCK * 8;
            const int r = e / DHEAD, d0 = e % DHEAD;
            store8(dO_lds + r * QS2 + d0, st_do[c]);
            store8(dO_img + (d0 >> 4) * VSUB + v_img_row(r) * 16 + (d0 & 15),
                   st_do[c]);
        }
        if (tid < BW_QT) {
            lse_lds[tid] = st_lse;
            D_lds[tid] = st_D;
        }
    };
    q_stage_load(q_start);
    q_stage_write(q_start);
    __syncthreads();
    q_stage_load(q_start + BW_QT);
    for (int qt = q_start; qt < S; qt += BW_QT) {

        f32x4 p_r[2], ds_r[2];
#pragma unroll
        for (int qs = 0; qs < 2; ++qs) {
            f32x4 s0 = f32x4{0, 0, 0, 0};
            f32x4 dp0 = f32x4{0, 0, 0, 0};
#pragma unroll
            for (int kc = 0; kc < KC; ++kc) {
                bf16x8v a_qf = ld_frag(Q_lds + (qs * 16 + col) * QS_T + kc * 32 + kgrp * 8);
                s0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_qf, b_k[kc], s0, 0, 0, 0);
            }
#pragma unroll
            for (int kc = 0; kc < 4; ++kc) {
                bf16x8v a_do = ld_frag(dO_lds + (qs * 16 + col) * QS2 + kc * 32 + kgrp * 8);
                dp0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_do, b_v[kc], dp0, 0, 0, 0);
            }
            // interior tiles: every row/col real and strictly below the
            // diagonal -> no per-element masking (PMC: mask VALU-bound)
            const bool clean = (qt + qs * 16 + 16 <= S) && (kvbase + BKV <= S)
                && (!causal || kvbase + BKV - 1 <= qt + qs * 16);
#pragma unroll
            for (int j = 0; j < 4; ++j) {
                const float ls = lse_lds[qs * 16 + kgrp * 4 + j];
                const float Dv = D_lds[qs * 16 + kgrp * 4 + j];
                float p0;
                if (clean) {
                    p0 = __expf(s0[j] - ls);
                } else {
                    const int row = qt + qs * 16 + kgrp * 4 + j;
                    const int c0 = kvbase + col;
                    p0 = (row < S && c0 < S && (!causal || c0 <= row)
                          && ls != -INFINITY) ? __expf(s0[j] - ls) : 0.f;
                }
                p_r[qs][j] = p0;
                ds_r[qs][j] = p0 * (dp0[j] - Dv) * scale;
            }
        }
        // dV += P^T @ dO ; dK += dS^T @ Q(unscaled via Q_img). P^T and
        // dS^T come straight from the C tiles via the cvt_pk + permlane
        // butterfly (the output registers read as an A-fragment ARE the
        // transpose) — no element-wise LDS transpose round trip.
        {
            auto butterfly = [&](const f32x4 &e, const f32x4 &f) {
                int E0, E1, F0, F1;
                asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(E0)
                    : "v"(e[0]), "v"(e[1]));
                asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(E1)
                    : "v"(e[2]), "v"(e[3]));
                asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(F0)
                    : "v"(f[0]), "v"(f[1]));
                asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(F1)
                    : "v"(f[2]), "v"(f[3]));
                v2i u0 = __builtin_amdgcn_permlane32_swap(E0, F0, false, false);
                v2i u1 = __builtin_amdgcn_permlane32_swap(E1, F1, false, false);
                v2i d02 = __builtin_amdgcn_permlane16_swap(u0[0], u0[1], false, false);
                v2i d13 = __builtin_amdgcn_permlane16_swap(u1[0], u1[1], false, false);
                int frag[4] = {d02[0], d13[0], d02[1], d13[1]};
                return *reinterpret_cast<bf16x8v *>(frag);
            };
            bf16x8v a_pt = butterfly(p_r[0], p_r[1]);
            bf16x8v a_dst = butterfly(ds_r[0], ds_r[1]);
#pragma unroll
            for (int dt = 0; dt < 8; ++dt) {
                bf16x8v b_do = tr_frag(dO_img + dt * VSUB, lane);
                dv_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a_pt, b_do, dv_acc[dt], 0, 0, 0);
            }
#pragma unroll
            for (int dt = 0; dt < QSUB; ++dt) {
                bf16x8v b_q = tr_frag(Q_img + dt * VSUB, lane);
                dk_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a_dst, b_q, dk_acc[dt], 0, 0, 0);
            }
        }
        __syncthreads();          // all waves done reading this q tile
        q_stage_write(qt + BW_QT);
        __syncthreads();          // next tile visible
        q_stage_load(qt + 2 * BW_QT);
    }

    // dkv layout: [B, S, Hkv, DQK + DHEAD] fp32 (dk then dv)
#pragma unroll
    for (int j = 0; j < 4; ++j) {
        const int krow = kvbase + kgrp * 4 + j;
        if (krow >= S) continue;
        const long base = (((long)b * S + krow) * Hkv + hkv) * (DQK + DHEAD);
#pragma unroll
        for (int dt = 0; dt < QSUB; ++dt)
            atomicAdd(&dkv[base + dt * 16 + col], dk_acc[dt][j]);
#pragma unroll
        for (int dt = 0; dt < 8; ++dt)
            atomicAdd(&dkv[base + DQK + dt * 16 + col], dv_acc[dt][j]);
    }
}

// ---- dQ: q-parallel, fwd-like; K/V staged double-buffered + K tr-image ----
template <int DQK, int NQS>
__global__ __launch_bounds__(FA_BLOCK, 2)
void fa_bwd_dq_kernel(const bf16raw *__restrict__ dout,
                      const bf16raw *__restrict__ q,
                      const bf16raw *__restrict__ k,
                      const bf16raw *__restrict__ v,
                      const float *__restrict__ lse,
                      const float *__restrict__ Dsum,
                      bf16raw *__restrict__ dq,
                      int B, int S, int Hq, int Hkv, int causal) {
    constexpr int KC = DQK / 32;
    constexpr int KS_T = DQK + 16;
    constexpr int QSUB = DQK / 16;
    constexpr int QT = NQS * 16;
    constexpr int BUF = KVTILE * KS_T + KVTILE * KS + QSUB * VSUB;
    extern __shared__ __attribute__((aligned(16))) char smem[];
    bf16raw *buf0 = reinterpret_cast<bf16raw *>(smem);

    const int qblk = blockIdx.x;
    const int h = blockIdx.y;
    const int b = blockIdx.z;
    const int hkv = h / (Hq / Hkv);
    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int col = lane & 15;
    const int kgrp = lane >> 4;
    const int qbase = qblk * (WAVES * QT) + wave * QT;
    const float scale = rsqrtf((float)DQK);

    const long q_row = (long)Hq * DQK;
    const long k_row = (long)Hkv * DQK;
    const long v_row = (long)Hkv * DHEAD;
    const long o_row = (long)Hq * DHEAD;
    const bf16raw *qp = q + ((long)b * S) * q_row + (long)h * DQK;
    const bf16raw *kp = k + ((long)b * S) * k_row + (long)hkv * DQK;
    const bf16raw *vp = v + ((long)b * S) * v_row + (long)hkv * DHEAD;
    const bf16raw *dop = dout + ((long)b * S) * o_row + (long)h * DHEAD;

    // per-wave q-side registers: scaled Q frags, dO frags, lse, D — all
    // kv-tile-invariant, so they are loaded ONCE (r1 PMC: re-reading dO
    // from global inside the kv loop parked the wave 51.5% of its cycles).
    // Swapped layout (T12, like the forward): the wave's q rows become
    // per-lane COLUMNS, so lse/D are per-lane scalars.
    bf16x8v a_q[NQS][KC], a_do[NQS][4];
    float lse_r[NQS], D_r[NQS];
#pragma unroll
    for (int qs = 0; qs < NQS; ++qs) {
        const int qrow = qbase + qs * 16 + col;
#pragma unroll
        for (int kc = 0; kc < KC; ++kc) {
            bf16x8 raw = load8(qp + (long)min(qrow, S - 1) * q_row + kc * 32 + kgrp * 8);
            bf16x8 sc;
#pragma unroll
            for (int j = 0; j < 8; ++j) sc.set(j, raw.get(j) * scale);
            a_q[qs][kc] = *reinterpret_cast<bf16x8v *>(&sc.raw);
        }
#pragma unroll
        for (int kc = 0; kc < 4; ++kc)
            a_do[qs][kc] = ld_frag(
                dop + (long)min(qrow, S - 1) * o_row + kc * 32 + kgrp * 8);
        const long idx = ((long)b * Hq + h) * S + min(qrow, S - 1);
        lse_r[qs] = lse[idx];
        D_r[qs] = Dsum[idx];
    }

    f32x4 dq_acc[NQS][QSUB];
#pragma unroll
    for (int qs = 0; qs < NQS; ++qs)
#pragma unroll
        for (int dt = 0; dt < QSUB; ++dt) dq_acc[qs][dt] = f32x4{0, 0, 0, 0};

    const int blk_rows = WAVES * QT;
    const int kv_limit = causal ? min(S, qblk * blk_rows + blk_rows) : S;
    const int n_tiles = CDIV(kv_limit, KVTILE);

    constexpr int KCHUNKS = KVTILE * DQK / (FA_BLOCK * 8);
    bf16x8 st_k[KCHUNKS], st_v[2];
    auto stage_load = [&](int t) {
        if (t >= n_tiles) return;
        const int kv = t * KVTILE;
        const int tid = threadIdx.x;
#pragma unroll
        for (int c = 0; c < KCHUNKS; ++c) {
            const int e = tid * 8 + c * FA_BLOCK * 8;
            const int kvr = e / DQK, d0 = e % DQK;
            const int src = min(kv + kvr, S - 1);
            st_k[c] = load8(kp + (long)src * k_row + d0);
        }
#pragma unroll
        for (int c = 0; c < 2; ++c) {
            const int e = tid * 8 + c * FA_BLOCK * 8;
            const int kvr = e / DHEAD, d0 = e % DHEAD;
            const int src = min(kv + kvr, S - 1);
            st_v[c] = load8(vp + (long)src * v_row + d0);
        }
    };
    auto stage_write = [&](int t) {
        if (t >= n_tiles) return;
        bf16raw *K_lds = buf0 + (t & 1) * BUF;               // raw K rows
        bf16raw *V_lds = K_lds + KVTILE * KS_T;              // V rows [32][KS]
        bf16raw *K_img = V_lds + KVTILE * KS;                // tr image (raw)
        const int tid = threadIdx.x;
#pragma unroll
        for (int c = 0; c < KCHUNKS; ++c) {
            const int e = tid * 8 + c * FA_BLOCK * 8;
            const int kvr = e / DQK, d0 = e % DQK;
            // K kept raw: S reuses the pre-scaled Q fragments (a_q)
            store8(K_lds + kvr * KS_T + d0, st_k[c]);
            store8(K_img + (d0 >> 4) * VSUB + v_img_row(kvr) * 16 + (d0 & 15),
                   st_k[c]);
        }
#pragma unroll
        for (int c = 0; c < 2; ++c) {
            const int e = tid * 8 + c * FA_BLOCK * 8;
            const int kvr = e / DHEAD, d0 = e % DHEAD;
            store8(V_lds + kvr * KS + d0, st_v[c]);
        }
    };

    stage_load(0);
    stage_write(0);
    __syncthreads();
    stage_load(1);

    for (int t = 0; t < n_tiles; ++t) {
        const int kv = t * KVTILE;
        bf16raw *K_lds = buf0 + (t & 1) * BUF;
        bf16raw *V_lds = K_lds + KVTILE * KS_T;
        bf16raw *K_img = V_lds + KVTILE * KS;

        if (!(causal && kv >= qbase + QT)) {
#pragma unroll
            for (int qs = 0; qs < NQS; ++qs) {
                // swapped: S^T[kv][q], dP^T[kv][q] — lane col = q, rows kv
                f32x4 s_t[2] = {f32x4{0, 0, 0, 0}, f32x4{0, 0, 0, 0}};
                f32x4 dp_t[2] = {f32x4{0, 0, 0, 0}, f32x4{0, 0, 0, 0}};
#pragma unroll
                for (int kc = 0; kc < KC; ++kc) {
                    bf16x8v bk0 = ld_frag(K_lds + col * KS_T + kc * 32 + kgrp * 8);
                    bf16x8v bk1 = ld_frag(K_lds + (16 + col) * KS_T + kc * 32 + kgrp * 8);
                    s_t[0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(bk0, a_q[qs][kc], s_t[0], 0, 0, 0);
                    s_t[1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(bk1, a_q[qs][kc], s_t[1], 0, 0, 0);
                }
#pragma unroll
                for (int kc = 0; kc < 4; ++kc) {
                    bf16x8v bv0 = ld_frag(V_lds + col * KS + kc * 32 + kgrp * 8);
                    bf16x8v bv1 = ld_frag(V_lds + (16 + col) * KS + kc * 32 + kgrp * 8);
                    dp_t[0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(bv0, a_do[qs][kc], dp_t[0], 0, 0, 0);
                    dp_t[1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(bv1, a_do[qs][kc], dp_t[1], 0, 0, 0);
                }
                const bool clean = (qbase + qs * 16 + 16 <= S)
                    && (kv + KVTILE <= S)
                    && (!causal || kv + KVTILE - 1 <= qbase + qs * 16);
                const float ls = lse_r[qs];
                const float Dv = D_r[qs];
                const int qcol = qbase + qs * 16 + col;
#pragma unroll
                for (int ks = 0; ks < 2; ++ks)
#pragma unroll
                    for (int j = 0; j < 4; ++j) {
                        float pv;
                        if (clean) {
                            pv = __expf(s_t[ks][j] - ls);
                        } else {
                            const int c0 = kv + ks * 16 + kgrp * 4 + j;
                            pv = (qcol < S && c0 < S
                                  && (!causal || c0 <= qcol)
                                  && ls != -INFINITY)
                                     ? __expf(s_t[ks][j] - ls) : 0.f;
                        }
                        s_t[ks][j] = pv * (dp_t[ks][j] - Dv) * scale;
                    }
                // dS^T -> B-fragment in registers (cvt_pk + permlane
                // butterfly, same mapping as the forward's P^T)
                int E0, E1, F0, F1;
                asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(E0)
                    : "v"(s_t[0][0]), "v"(s_t[0][1]));
                asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(E1)
                    : "v"(s_t[0][2]), "v"(s_t[0][3]));
                asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(F0)
                    : "v"(s_t[1][0]), "v"(s_t[1][1]));
                asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(F1)
                    : "v"(s_t[1][2]), "v"(s_t[1][3]));
                v2i u0 = __builtin_amdgcn_permlane32_swap(E0, F0, false, false);
                v2i u1 = __builtin_amdgcn_permlane32_swap(E1, F1, false, false);
                v2i d02 = __builtin_amdgcn_permlane16_swap(u0[0], u0[1], false, false);
                v2i d13 = __builtin_amdgcn_permlane16_swap(u1[0], u1[1], false, false);
                int frag[4] = {d02[0], d13[0], d02[1], d13[1]};
                bf16x8v b_ds = *reinterpret_cast<bf16x8v *>(frag);
                // dq^T = mfma(A = K^T tr-fragments, B = dS^T)
#pragma unroll
                for (int dt = 0; dt < QSUB; ++dt) {
                    bf16x8v a_k = tr_frag(K_img + dt * VSUB, lane);
                    dq_acc[qs][dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a_k, b_ds, dq_acc[qs][dt], 0, 0, 0);
                }
            }
        }
        stage_write(t + 1);
        __syncthreads();
        stage_load(t + 2);
    }

    bf16raw *dqp = dq + ((long)b * S) * q_row + (long)h * DQK;
    // dq^T epilogue: lane owns q column; 4 contiguous d per (dt) -> one
    // 8-byte store each (same shape as the forward's O^T epilogue)
#pragma unroll
    for (int qs = 0; qs < NQS; ++qs) {
        const int row = qbase + qs * 16 + col;
        if (row >= S) continue;
#pragma unroll
        for (int dt = 0; dt < QSUB; ++dt) {
            const unsigned w0 =
                (unsigned)f2bf(dq_acc[qs][dt][0]) |
                ((unsigned)f2bf(dq_acc[qs][dt][1]) << 16);
            const unsigned w1 =
                (unsigned)f2bf(dq_acc[qs][dt][2]) |
                ((unsigned)f2bf(dq_acc[qs][dt][3]) << 16);
            *reinterpret_cast<uint2 *>(
                dqp + (long)row * q_row + dt * 16 + kgrp * 4) = uint2{w0, w1};
        }
    }
}

extern "C" void fa_fwd_launch(const void *q, const void *k, const void *v,
                              void *o, void *lse, int B, int S, int Hq,
                              int Hkv, int dqk, int causal,
                              hipStream_t stream) {
    if (dqk == 128) {
        dim3 grid(CDIV(S, 256), Hq, B);
        size_t smem = 2 * (KVT2 * (128 + 16) + 2 * 8 * VSUB) * sizeof(bf16raw);
        hipLaunchKernelGGL((fa_fwd_kernel<128>), grid, dim3(FWD_BLOCK),
                           smem, stream, (const bf16raw *)q,
                           (const bf16raw *)k, (const bf16raw *)v,
                           (bf16raw *)o, (float *)lse, B, S, Hq, Hkv, causal);
    } else if (dqk == 192) {
        dim3 grid(CDIV(S, 256), Hq, B);
        size_t smem = 2 * (KVT2 * (192 + 16) + 2 * 8 * VSUB) * sizeof(bf16raw);
        hipLaunchKernelGGL((fa_fwd_kernel<192>), grid, dim3(FWD_BLOCK),
                           smem, stream, (const bf16raw *)q,
                           (const bf16raw *)k, (const bf16raw *)v,
                           (bf16raw *)o, (float *)lse, B, S, Hq, Hkv, causal);
    } else {
        abort();
    }
}

extern "C" void fa_bwd_launch(const void *dout, const void *q, const void *k,
                              const void *v, const void *o, const void *lse,
                              void *dq_bf16, void *dkv, void *dsum,
                              int B, int S, int Hq, int Hkv, int dqk,
                              int causal, hipStream_t stream) {
    // 1. D = rowsum(dO*O)
    {
        long rows = (long)B * Hq * S;
        long grid = CDIV(rows, FA_BLOCK / WAVE);
        if (grid > 4096) grid = 4096;
        hipLaunchKernelGGL(fa_bwd_pre_kernel, dim3((int)grid), dim3(FA_BLOCK),
                           0, stream, (const bf16raw *)dout,
                           (const bf16raw *)o, (float *)dsum, B, S, Hq);
    }
    // 2. dK/dV (kv-parallel)
    {
        dim3 grid(CDIV(S, WAVES * BKV), Hq, B);
        size_t smem;
        if (dqk == 128) {
            smem = (BW_QT * (128 + 16) + BW_QT * QS2 + (128 / 16 + 8) * VSUB)
                       * sizeof(bf16raw)
                   + 2 * BW_QT * sizeof(float)
                   + WAVES * BKV * (128 + 16) * sizeof(bf16raw);
            hipLaunchKernelGGL((fa_bwd_dkv_kernel<128>), grid, dim3(FA_BLOCK),
                               smem, stream, (const bf16raw *)dout,
                               (const bf16raw *)q, (const bf16raw *)k,
                               (const bf16raw *)v, (const float *)lse,
                               (const float *)dsum, (float *)dkv, B, S, Hq,
                               Hkv, causal);
        } else if (dqk == 192) {
            smem = (BW_QT * (192 + 16) + BW_QT * QS2 + (192 / 16 + 8) * VSUB)
                       * sizeof(bf16raw)
                   + 2 * BW_QT * sizeof(float)
                   + WAVES * BKV * (192 + 16) * sizeof(bf16raw);
            hipLaunchKernelGGL((fa_bwd_dkv_kernel<192>), grid, dim3(FA_BLOCK),
                               smem, stream, (const bf16raw *)dout,
                               (const bf16raw *)q, (const bf16raw *)k,
                               (const bf16raw *)v, (const float *)lse,
                               (const float *)dsum, (float *)dkv, B, S, Hq,
                               Hkv, causal);
        } else {
            abort();
        }
    }
    // 3. dQ (q-parallel)
    if (dqk == 128) {
        constexpr int QT = 16;   // NQS=1: occupancy over per-wave work
        dim3 grid(CDIV(S, WAVES * QT), Hq, B);
        size_t smem = 2 * (KVTILE * (128 + 16) + KVTILE * KS + 8 * VSUB)
                      * sizeof(bf16raw);
        hipLaunchKernelGGL((fa_bwd_dq_kernel<128, 1>), grid, dim3(FA_BLOCK),
                           smem, stream, (const bf16raw *)dout,
                           (const bf16raw *)q, (const bf16raw *)k,
                           (const bf16raw *)v, (const float *)lse,
                           (const float *)dsum, (bf16raw *)dq_bf16, B, S, Hq,
                           Hkv, causal);
    } else {
        constexpr int QT = 16;
        dim3 grid(CDIV(S, WAVES * QT), Hq, B);
        size_t smem = 2 * (KVTILE * (192 + 16) + KVTILE * KS + 12 * VSUB)
                      * sizeof(bf16raw);
        hipLaunchKernelGGL((fa_bwd_dq_kernel<192, 1>), grid, dim3(FA_BLOCK),
                           smem, stream, (const bf16raw *)dout,
                           (const bf16raw *)q, (const bf16raw *)k,
                           (const bf16raw *)v, (const float *)lse,
                           (const float *)dsum, (bf16raw *)dq_bf16, B, S, Hq,
                           Hkv, causal);
    }
}
