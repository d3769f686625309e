// Fused bf16 -> fp8 cast with scale + running-amax update (TE-style
// delayed scaling for the fp8 training path: this call's amax feeds the
// NEXT call's scale, so the cast is ONE pass instead of the two-pass
// dynamic recipe). gfx950-native OCP conversion via v_cvt_pk_fp8_f32 /
// v_cvt_pk_bf8_f32 (e4m3 / e5m2).
#include "common.h"

#define CAST_BLOCK 256

typedef __attribute__((ext_vector_type(2))) short v2s;

__global__ __launch_bounds__(CAST_BLOCK)
void fp8_cast_kernel(const bf16raw *__restrict__ x,
                     unsigned char *__restrict__ out,
                     float *__restrict__ amax,
                     const float *__restrict__ scale_p,
                     long n, int e5m2, float fmax) {
    const float scale = *scale_p;
    float local = 0.f;
    const long stride = (long)gridDim.x * CAST_BLOCK * 8;
    for (long i = ((long)blockIdx.x * CAST_BLOCK + threadIdx.x) * 8; i < n;
         i += stride) {
        bf16x8 v = load8(x + i);
        unsigned char o[8];
#pragma unroll
        for (int p = 0; p < 4; ++p) {
            float a = v.get(2 * p), b = v.get(2 * p + 1);
            local = fmaxf(local, fmaxf(fabsf(a), fabsf(b)));
            // clamp: with delayed scaling a tensor can exceed last call's
            // amax; saturate instead of overflowing the fp8 range
            float as = fminf(fmaxf(a * scale, -fmax), fmax);
            float bs = fminf(fmaxf(b * scale, -fmax), fmax);
            int packed;
            if (e5m2)
                packed = __builtin_amdgcn_cvt_pk_bf8_f32(as, bs, 0, false);
            else
                packed = __builtin_amdgcn_cvt_pk_fp8_f32(as, bs, 0, false);
            o[2 * p] = packed & 0xff;
            o[2 * p + 1] = (packed >> 8) & 0xff;
        }
        *reinterpret_cast<uint2 *>(out + i) =
            *reinterpret_cast<uint2 *>(o);
    }
    // wave-reduce then one atomic per wave (positive floats order as ints)
    local = wave_max(local);
    if ((threadIdx.x % WAVE) == 0 && local > 0.f)
        atomicMax(reinterpret_cast<int *>(amax), __float_as_int(local));
}

extern "C" void fp8_cast_launch(const void *x, void *out, void *amax,
                                const void *scale, long n, int e5m2,
                                float fmax, hipStream_t stream) {
    long blocks = (n / 8 + CAST_BLOCK - 1) / CAST_BLOCK;
    if (blocks > 4096) blocks = 4096;
    if (blocks < 1) blocks = 1;
    hipLaunchKernelGGL(fp8_cast_kernel, dim3((int)blocks), dim3(CAST_BLOCK),
                       0, stream, (const bf16raw *)x, (unsigned char *)out,
                       (float *)amax, (const float *)scale, n, e5m2, fmax);
}
