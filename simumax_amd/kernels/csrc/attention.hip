// Flash attention (fwd + bwd) for gfx950, bf16, GQA, causal, D=128.
//
// Forward geometry: one 256-thread workgroup (4 waves) owns 64 q rows of one
// (batch, head); each wave owns 16 q rows. K[32][128] and V^T[128][32] tiles
// are staged in LDS per 32-key step and shared by the 4 waves. QK^T and PV
// both use v_mfma_f32_16x16x32_bf16; the softmax'd P tile crosses from the
// MFMA C-layout to the A-layout through a per-wave LDS buffer. Online
// softmax (running m, l) in registers; saves LSE fp32 for backward.
//
// Backward: one workgroup (4 waves) owns a 128-key block (32 keys/wave) of
// one (batch, head) and loops over 32-row q tiles staged in LDS
// (Q, Q^T, dO, dO^T + lse + D = rowsum(dO*O)). dK/dV accumulate in
// registers; dQ accumulates via fp32 atomics (cast by the host wrapper).
//
// Reference behavior target: the sdp_fwd/sdp_bwd ops priced by
// simumax/core/transformer/dense_module.py:1061-1605.
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8v;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define FA_BLOCK 256
#define QTILE 16
#define KVTILE 32
#define DHEAD 128
#define WAVES 4

DEV bf16x8v ld_frag(const bf16raw *p) {
    uint4 r = *reinterpret_cast<const uint4 *>(p);
    return *reinterpret_cast<bf16x8v *>(&r);
}

// reduce val across the 16 lanes of the C-fragment column group
DEV float group16_max(float v) {
#pragma unroll
    for (int off = 1; off < 16; off <<= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
    return v;
}

DEV float group16_sum(float v) {
#pragma unroll
    for (int off = 1; off < 16; off <<= 1) v += __shfl_xor(v, off, WAVE);
    return v;
}

// LDS layout (fwd):
//  K_lds  [KVTILE][DHEAD]      row-major   8 KiB
//  Vt_lds [DHEAD][KVTILE]      transposed  8 KiB
//  P_lds  [WAVES][QTILE][KVTILE]           4 KiB
extern "C" __global__ __launch_bounds__(FA_BLOCK)
void fa_fwd_kernel(const bf16raw *__restrict__ q, const bf16raw *__restrict__ k,
                   const bf16raw *__restrict__ v, bf16raw *__restrict__ o,
                   float *__restrict__ lse, int B, int S, int Hq, int Hkv,
                   int causal) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    bf16raw *K_lds = reinterpret_cast<bf16raw *>(smem);
    bf16raw *Vt_lds = K_lds + KVTILE * DHEAD;
    bf16raw *P_lds = Vt_lds + DHEAD * KVTILE;

    const int qblk = blockIdx.x;           // 64-row q block
    const int h = blockIdx.y;
    const int b = blockIdx.z;
    const int hkv = h / (Hq / Hkv);
    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int col = lane & 15;             // fragment col / A-row
    const int kgrp = lane >> 4;            // fragment k-group

    const int qbase = qblk * (WAVES * QTILE) + wave * QTILE;
    const float scale = rsqrtf((float)DHEAD);

    // strides for [B, S, H, D] layout
    const long q_row = (long)Hq * DHEAD;
    const long kv_row = (long)Hkv * DHEAD;
    const bf16raw *qp = q + ((long)b * S) * q_row + (long)h * DHEAD;
    const bf16raw *kp = k + ((long)b * S) * kv_row + (long)hkv * DHEAD;
    const bf16raw *vp = v + ((long)b * S) * kv_row + (long)hkv * DHEAD;

    // Q fragments, pre-scaled: a_q[kc] covers d in [kc*32, kc*32+32)
    bf16x8v a_q[4];
    const int qrow = qbase + col;
#pragma unroll
    for (int kc = 0; kc < 4; ++kc) {
        bf16x8 raw = load8(qp + (long)min(qrow, S - 1) * q_row + kc * 32 + kgrp * 8);
        bf16x8 sc;
#pragma unroll
        for (int j = 0; j < 8; ++j) sc.set(j, raw.get(j) * scale);
        a_q[kc] = *reinterpret_cast<bf16x8v *>(&sc.raw);
    }

    float m[4], l[4];
    f32x4 acc[8];
#pragma unroll
    for (int j = 0; j < 4; ++j) { m[j] = -INFINITY; l[j] = 0.f; }
#pragma unroll
    for (int dt = 0; dt < 8; ++dt) acc[dt] = f32x4{0.f, 0.f, 0.f, 0.f};

    const int kv_limit = causal ? min(S, qblk * (WAVES * QTILE) + WAVES * QTILE)
                                : S;

    for (int kv = 0; kv < kv_limit; kv += KVTILE) {
        // ---- stage K row-major + V transposed (all 256 threads) ----
        {
            // K: 32 rows x 128 cols = 4096 elems; each thread moves 16 elems
            const int tid = threadIdx.x;
            for (int e = tid * 8; e < KVTILE * DHEAD; e += FA_BLOCK * 8) {
                const int kvr = e / DHEAD, d0 = e % DHEAD;
                const int src = min(kv + kvr, S - 1);
                bf16x8 kk = load8(kp + (long)src * kv_row + d0);
                store8(K_lds + kvr * DHEAD + d0, kk);
                bf16x8 vv = load8(vp + (long)src * kv_row + d0);
#pragma unroll
                for (int j = 0; j < 8; ++j)
                    Vt_lds[(d0 + j) * KVTILE + kvr] = f2bf(vv.get(j));
            }
        }
        __syncthreads();

        // ---- QK^T: two 16-col subtiles ----
        f32x4 s0 = f32x4{0, 0, 0, 0}, s1 = f32x4{0, 0, 0, 0};
#pragma unroll
        for (int kc = 0; kc < 4; ++kc) {
            bf16x8v b0 = ld_frag(K_lds + (0 * 16 + col) * DHEAD + kc * 32 + kgrp * 8);
            bf16x8v b1 = ld_frag(K_lds + (1 * 16 + col) * DHEAD + kc * 32 + kgrp * 8);
            s0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_q[kc], b0, s0, 0, 0, 0);
            s1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_q[kc], b1, s1, 0, 0, 0);
        }
        // mask: rows beyond S, cols beyond S or causal-future
        float tile_max[4];
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            const int row = qbase + kgrp * 4 + j;
            const int c0 = kv + col, c1 = kv + 16 + col;
            if (row >= S || c0 >= S || (causal && c0 > row)) s0[j] = -INFINITY;
            if (row >= S || c1 >= S || (causal && c1 > row)) s1[j] = -INFINITY;
            tile_max[j] = group16_max(fmaxf(s0[j], s1[j]));
        }
        // ---- online softmax ----
        float alpha[4];
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            float mn = fmaxf(m[j], tile_max[j]);
            alpha[j] = (m[j] == -INFINITY) ? 0.f : __expf(m[j] - mn);
            m[j] = mn;
            float p0 = (s0[j] == -INFINITY) ? 0.f : __expf(s0[j] - mn);
            float p1 = (s1[j] == -INFINITY) ? 0.f : __expf(s1[j] - mn);
            s0[j] = p0; s1[j] = p1;
            l[j] = l[j] * alpha[j] + group16_sum(p0 + p1);
        }
        // rescale O
#pragma unroll
        for (int dt = 0; dt < 8; ++dt)
#pragma unroll
            for (int j = 0; j < 4; ++j) acc[dt][j] *= alpha[j];
        // ---- P to LDS (C-layout -> row-major) ----
        bf16raw *pw = P_lds + wave * QTILE * KVTILE;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            pw[(kgrp * 4 + j) * KVTILE + col] = f2bf(s0[j]);
            pw[(kgrp * 4 + j) * KVTILE + 16 + col] = f2bf(s1[j]);
        }
        // ---- PV ----
        bf16x8v a_p = ld_frag(pw + col * KVTILE + kgrp * 8);
#pragma unroll
        for (int dt = 0; dt < 8; ++dt) {
            bf16x8v b_v = ld_frag(Vt_lds + (dt * 16 + col) * KVTILE + kgrp * 8);
            acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_p, b_v, acc[dt], 0, 0, 0);
        }
        __syncthreads();
    }

    // ---- epilogue ----
    bf16raw *op = o + ((long)b * S) * q_row + (long)h * DHEAD;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
        const int row = qbase + kgrp * 4 + j;
        if (row >= S) continue;
        const float inv = (l[j] > 0.f) ? 1.f / l[j] : 0.f;
#pragma unroll
        for (int dt = 0; dt < 8; ++dt)
            op[(long)row * q_row + dt * 16 + col] = f2bf(acc[dt][j] * inv);
        if (col == 0) {
            // lse [B, Hq, S]
            lse[((long)b * Hq + h) * S + row] =
                (l[j] > 0.f) ? m[j] + __logf(l[j]) : -INFINITY;
        }
    }
}

// ==========================================================================
// backward
// ==========================================================================
// LDS (shared by the workgroup, per 32-row q tile):
//  Q_lds   [32][128]  8 KiB   (pre-scaled by 1/sqrt(D))
//  Qt_lds  [128][32]  8 KiB
//  dO_lds  [32][128]  8 KiB
//  dOt_lds [128][32]  8 KiB
//  lse/Dv  [32] + [32] fp32
// per wave (4 waves, 32 keys each):
//  K_l [32][128] row-major, Kt_l [128][32], V_l [32][128]
//  Pt_l [32][32], dSr_l [32][32], dSt_l [32][32]
#define BW_QT 32

extern "C" __global__ __launch_bounds__(FA_BLOCK)
void fa_bwd_kernel(const bf16raw *__restrict__ dout,
                   const bf16raw *__restrict__ q, const bf16raw *__restrict__ k,
                   const bf16raw *__restrict__ v, const bf16raw *__restrict__ o,
                   const float *__restrict__ lse, float *__restrict__ dq,
                   float *__restrict__ dkv,  // [B, S, Hkv, 2, D] fp32 atomics
                   int B, int S, int Hq, int Hkv, int causal) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    bf16raw *Q_lds = reinterpret_cast<bf16raw *>(smem);          // 32*128
    bf16raw *Qt_lds = Q_lds + BW_QT * DHEAD;                     // 128*32
    bf16raw *dO_lds = Qt_lds + DHEAD * BW_QT;                    // 32*128
    bf16raw *dOt_lds = dO_lds + BW_QT * DHEAD;                   // 128*32
    float *lse_lds = reinterpret_cast<float *>(dOt_lds + DHEAD * BW_QT);
    float *D_lds = lse_lds + BW_QT;
    bf16raw *wbase = reinterpret_cast<bf16raw *>(D_lds + BW_QT);
    // per-wave carve
    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int col = lane & 15;
    const int kgrp = lane >> 4;
    const int WSZ = KVTILE * DHEAD * 2 + KVTILE * KVTILE + BW_QT * KVTILE * 2;
    bf16raw *K_l = wbase + wave * WSZ;            // [32][128]
    bf16raw *Kt_l = K_l + KVTILE * DHEAD;         // [128][32]
    bf16raw *Pt_l = Kt_l + DHEAD * KVTILE;        // [32k][32q]
    bf16raw *dSr_l = Pt_l + KVTILE * BW_QT;       // [32q][32k]
    bf16raw *dSt_l = dSr_l + BW_QT * KVTILE;      // [32k][32q]

    const int kvblk = blockIdx.x;                 // 128-key block
    const int h = blockIdx.y;
    const int b = blockIdx.z;
    const int hkv = h / (Hq / Hkv);
    const int kvbase = kvblk * (WAVES * KVTILE) + wave * KVTILE;

    const long q_row = (long)Hq * DHEAD;
    const long kv_row = (long)Hkv * DHEAD;
    const bf16raw *qp = q + ((long)b * S) * q_row + (long)h * DHEAD;
    const bf16raw *kp = k + ((long)b * S) * kv_row + (long)hkv * DHEAD;
    const bf16raw *vp = v + ((long)b * S) * kv_row + (long)hkv * DHEAD;
    const bf16raw *dop = dout + ((long)b * S) * q_row + (long)h * DHEAD;
    const bf16raw *op = o + ((long)b * S) * q_row + (long)h * DHEAD;
    const float scale = rsqrtf((float)DHEAD);

    // stage this wave's K (scaled) + Kt; V stays in registers as B-fragments
    bf16x8v b_v[8];  // V fragments for dP: B[k=d, col=key]
    {
        // each lane: handles rows kvbase..; cooperative within wave
        for (int e = lane * 8; e < KVTILE * DHEAD; e += WAVE * 8) {
            const int kvr = e / DHEAD, d0 = e % DHEAD;
            const int src = min(kvbase + kvr, S - 1);
            bf16x8 kk = load8(kp + (long)src * kv_row + d0);
            bf16x8 ks;
#pragma unroll
            for (int j = 0; j < 8; ++j) ks.set(j, kk.get(j) * scale);
            store8(K_l + kvr * DHEAD + d0, ks);
#pragma unroll
            for (int j = 0; j < 8; ++j)
                Kt_l[(d0 + j) * KVTILE + kvr] = f2bf(kk.get(j));  // unscaled
        }
        // V fragments: for dP, B[k = d-chunk, col=key]: lane reads
        // V[key=col][kc*32 + kgrp*8 + j] directly from global
        const int src = min(kvbase + col, S - 1);
#pragma unroll
        for (int kc = 0; kc < 4; ++kc) {
            b_v[kc] = ld_frag(vp + (long)src * kv_row + kc * 32 + kgrp * 8);
        }
    }

    // accumulators: dK [32k][128d] and dV [32k][128d] as C-fragments
    // C layout: col = key... we compute dV^T? Keep dV as 8 d-tiles of
    // C[16 rows=k, 16 cols=d]? -> we accumulate dV[k,d] with A=Pt, B=dO:
    // C col = lane&15 = d, row = kgrp*4+j = k (within 16) -> need 2 k-sub
    // tiles x 8 d-tiles = 16 frags each for dK and dV.
    f32x4 dv_acc[2][8], dk_acc[2][8];
#pragma unroll
    for (int a = 0; a < 2; ++a)
#pragma unroll
        for (int dt = 0; dt < 8; ++dt) {
            dv_acc[a][dt] = f32x4{0, 0, 0, 0};
            dk_acc[a][dt] = f32x4{0, 0, 0, 0};
        }

    const int q_start = causal ? (kvblk * (WAVES * KVTILE) / BW_QT) * BW_QT : 0;
    for (int qt = q_start; qt < S; qt += BW_QT) {
        // ---- stage q-side tiles (whole block) ----
        {
            const int tid = threadIdx.x;
            for (int e = tid * 8; e < BW_QT * DHEAD; e += FA_BLOCK * 8) {
                const int r = e / DHEAD, d0 = e % DHEAD;
                const int src = min(qt + r, S - 1);
                bf16x8 qq = load8(qp + (long)src * q_row + d0);
                store8(Q_lds + r * DHEAD + d0, qq);
                bf16x8 dd = load8(dop + (long)src * q_row + d0);
                store8(dO_lds + r * DHEAD + d0, dd);
#pragma unroll
                for (int j = 0; j < 8; ++j) {
                    Qt_lds[(d0 + j) * BW_QT + r] = f2bf(qq.get(j));
                    dOt_lds[(d0 + j) * BW_QT + r] = f2bf(dd.get(j));
                }
            }
            // lse + D = rowsum(dO * O)
            for (int r = tid; r < BW_QT; r += FA_BLOCK) {
                const int src = min(qt + r, S - 1);
                lse_lds[r] = lse[((long)b * Hq + h) * S + src];
                float dsum = 0.f;
                for (int d0 = 0; d0 < DHEAD; d0 += 8) {
                    bf16x8 dd = load8(dop + (long)src * q_row + d0);
                    bf16x8 oo = load8(op + (long)src * q_row + d0);
#pragma unroll
                    for (int j = 0; j < 8; ++j) dsum += dd.get(j) * oo.get(j);
                }
                D_lds[r] = dsum;
            }
        }
        __syncthreads();

        // ---- per wave: two 16-q subtiles ----
#pragma unroll
        for (int qs = 0; qs < 2; ++qs) {
            // S^T tile? compute S[16q, 32k] like fwd: A=Q frag, B=K_l
            f32x4 s0 = f32x4{0, 0, 0, 0}, s1 = f32x4{0, 0, 0, 0};
            f32x4 dp0 = f32x4{0, 0, 0, 0}, dp1 = f32x4{0, 0, 0, 0};
#pragma unroll
            for (int kc = 0; kc < 4; ++kc) {
                bf16x8v a_q = ld_frag(Q_lds + (qs * 16 + col) * DHEAD + kc * 32 + kgrp * 8);
                bf16x8v b0 = ld_frag(K_l + (0 * 16 + col) * DHEAD + kc * 32 + kgrp * 8);
                bf16x8v b1 = ld_frag(K_l + (1 * 16 + col) * DHEAD + kc * 32 + kgrp * 8);
                s0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_q, b0, s0, 0, 0, 0);
                s1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_q, b1, s1, 0, 0, 0);
                // dP = dO @ V^T
                bf16x8v a_do = ld_frag(dO_lds + (qs * 16 + col) * DHEAD + kc * 32 + kgrp * 8);
                dp0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_do, b_v[kc], dp0, 0, 0, 0);
            }
            // second key subtile of dP needs V fragments of keys 16..31:
            // those live in lanes via b_v only for col keys; B col selects
            // key: b_v's col = lane&15 = key within THIS wave's 32 keys for
            // subtile 0 (keys 0-15). For keys 16-31 load fresh fragments.
            {
                const int src = min(kvbase + 16 + col, S - 1);
#pragma unroll
                for (int kc = 0; kc < 4; ++kc) {
                    bf16x8v a_do = ld_frag(dO_lds + (qs * 16 + col) * DHEAD + kc * 32 + kgrp * 8);
                    bf16x8v bv1 = ld_frag(vp + (long)src * kv_row + kc * 32 + kgrp * 8);
                    dp1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_do, bv1, dp1, 0, 0, 0);
                }
            }
            // P = exp(S - lse); dS = P * (dP - D)
#pragma unroll
            for (int j = 0; j < 4; ++j) {
                const int row = qt + qs * 16 + kgrp * 4 + j;
                const int c0 = kvbase + col, c1 = kvbase + 16 + col;
                const float ls = lse_lds[qs * 16 + kgrp * 4 + j];
                const float Dv = D_lds[qs * 16 + kgrp * 4 + j];
                float p0 = (row < S && c0 < S && (!causal || c0 <= row) && ls != -INFINITY)
                               ? __expf(s0[j] - ls) : 0.f;
                float p1 = (row < S && c1 < S && (!causal || c1 <= row) && ls != -INFINITY)
                               ? __expf(s1[j] - ls) : 0.f;
                float ds0 = p0 * (dp0[j] - Dv) * scale;
                float ds1 = p1 * (dp1[j] - Dv) * scale;
                const int qrow = qs * 16 + kgrp * 4 + j;
                Pt_l[(col)*BW_QT + qrow] = f2bf(p0);
                Pt_l[(16 + col) * BW_QT + qrow] = f2bf(p1);
                dSr_l[qrow * KVTILE + col] = f2bf(ds0);
                dSr_l[qrow * KVTILE + 16 + col] = f2bf(ds1);
                // dK = (dS_raw*scale)^T @ Q with UNSCALED Q staged in Qt_lds,
                // so dSt carries the same scale factor as dSr
                dSt_l[(col)*BW_QT + qrow] = f2bf(ds0);
                dSt_l[(16 + col) * BW_QT + qrow] = f2bf(ds1);
            }
        }
        // wave-local LDS now has Pt[32k][32q], dSr[32q][32k], dSt[32k][32q]
        // ---- dV += P^T @ dO ; dK += dS^T @ (Q*scale -> use unscaled Q) ----
#pragma unroll
        for (int a = 0; a < 2; ++a) {
            bf16x8v a_pt = ld_frag(Pt_l + (a * 16 + col) * BW_QT + kgrp * 8);
            bf16x8v a_dst = ld_frag(dSt_l + (a * 16 + col) * BW_QT + kgrp * 8);
#pragma unroll
            for (int dt = 0; dt < 8; ++dt) {
                bf16x8v b_do = ld_frag(dOt_lds + (dt * 16 + col) * BW_QT + kgrp * 8);
                dv_acc[a][dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a_pt, b_do, dv_acc[a][dt], 0, 0, 0);
                bf16x8v b_q = ld_frag(Qt_lds + (dt * 16 + col) * BW_QT + kgrp * 8);
                dk_acc[a][dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a_dst, b_q, dk_acc[a][dt], 0, 0, 0);
            }
        }
        // ---- dQ += dS @ K (unscaled K = Kt_l); atomics into fp32 dq ----
#pragma unroll
        for (int qs = 0; qs < 2; ++qs) {
            bf16x8v a_ds = ld_frag(dSr_l + (qs * 16 + col) * KVTILE + kgrp * 8);
            const int row = qt + qs * 16 + kgrp * 4;
#pragma unroll
            for (int dt = 0; dt < 8; ++dt) {
                bf16x8v b_k = ld_frag(Kt_l + (dt * 16 + col) * KVTILE + kgrp * 8);
                f32x4 dq_t = f32x4{0, 0, 0, 0};
                dq_t = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_ds, b_k, dq_t, 0, 0, 0);
#pragma unroll
                for (int j = 0; j < 4; ++j) {
                    const int r = row + j;
                    if (r < S)
                        atomicAdd(&dq[(((long)b * S + r) * Hq + h) * DHEAD + dt * 16 + col],
                                  dq_t[j]);
                }
            }
        }
        __syncthreads();
    }

    // ---- write dK/dV via atomics (GQA heads collide on dkv) ----
#pragma unroll
    for (int a = 0; a < 2; ++a)
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            const int krow = kvbase + a * 16 + kgrp * 4 + j;
            if (krow >= S) continue;
            const long base = (((long)b * S + krow) * Hkv + hkv) * 2 * DHEAD;
#pragma unroll
            for (int dt = 0; dt < 8; ++dt) {
                atomicAdd(&dkv[base + dt * 16 + col], dk_acc[a][dt][j]);
                atomicAdd(&dkv[base + DHEAD + dt * 16 + col], dv_acc[a][dt][j]);
            }
        }
}

extern "C" void fa_fwd_launch(const void *q, const void *k, const void *v,
                              void *o, void *lse, int B, int S, int Hq,
                              int Hkv, int causal, hipStream_t stream) {
    dim3 grid(CDIV(S, WAVES * QTILE), Hq, B);
    size_t smem = (KVTILE * DHEAD + DHEAD * KVTILE +
                   WAVES * QTILE * KVTILE) * sizeof(bf16raw);
    hipLaunchKernelGGL(fa_fwd_kernel, grid, dim3(FA_BLOCK), smem, stream,
                       (const bf16raw *)q, (const bf16raw *)k,
                       (const bf16raw *)v, (bf16raw *)o, (float *)lse, B, S,
                       Hq, Hkv, causal);
}

extern "C" void fa_bwd_launch(const void *dout, const void *q, const void *k,
                              const void *v, const void *o, const void *lse,
                              void *dq, void *dkv, int B, int S, int Hq,
                              int Hkv, int causal, hipStream_t stream) {
    dim3 grid(CDIV(S, WAVES * KVTILE), Hq, B);
    size_t q_side = (2 * BW_QT * DHEAD + 2 * DHEAD * BW_QT) * sizeof(bf16raw) +
                    2 * BW_QT * sizeof(float);
    size_t per_wave = (KVTILE * DHEAD * 2 + KVTILE * KVTILE +
                       BW_QT * KVTILE * 2) * sizeof(bf16raw);
    size_t smem = q_side + WAVES * per_wave;
    hipLaunchKernelGGL(fa_bwd_kernel, grid, dim3(FA_BLOCK), smem, stream,
                       (const bf16raw *)dout, (const bf16raw *)q,
                       (const bf16raw *)k, (const bf16raw *)v,
                       (const bf16raw *)o, (const float *)lse, (float *)dq,
                       (float *)dkv, B, S, Hq, Hkv, causal);
}
