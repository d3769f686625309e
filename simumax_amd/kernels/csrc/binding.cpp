// PyTorch bindings for the simumax_amd gfx950 kernels.
// Compiled directly with hipcc (no hipify pass) by kernels/build.py.
#include <torch/extension.h>

#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#define CHECK_IN(t)                                                     \
    TORCH_CHECK((t).is_cuda(), #t " must be on GPU");                   \
    TORCH_CHECK((t).is_contiguous(), #t " must be contiguous")

static hipStream_t cur_stream() {
    return c10::hip::getCurrentHIPStream().stream();
}

extern "C" {
void rmsnorm_fwd_launch(const void *, const void *, void *, void *, int, int,
                        float, hipStream_t);
void rmsnorm_bwd_launch(const void *, const void *, const void *, const void *,
                        void *, void *, int, int, hipStream_t);
void rope_launch(const void *, void *, const void *, const void *, int, int,
                 int, float, hipStream_t);
void swiglu_fwd_launch(const void *, void *, long, int, hipStream_t);
void swiglu_bwd_launch(const void *, const void *, void *, long, int,
                       hipStream_t);
void ce_fwd_launch(const void *, const void *, void *, void *, void *, long,
                   int, hipStream_t);
void ce_bwd_launch(const void *, const void *, const void *, const void *,
                   const void *, void *, long, int, hipStream_t);
void fp8_cast_launch(const void *, void *, void *, const void *, long, int,
                     float, hipStream_t);
void fp8_cast_t_launch(const void *, void *, void *, void *, const void *,
                       int, int, int, float, hipStream_t);
}

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w,
                                       double eps) {
    CHECK_IN(x);
    CHECK_IN(w);
    TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "x must be bf16");
    const int H = x.size(-1);
    const long rows = x.numel() / H;
    TORCH_CHECK(H % 8 == 0, "H must be a multiple of 8");
    auto y = torch::empty_like(x);
    auto rstd = torch::empty({rows}, x.options().dtype(torch::kFloat32));
    rmsnorm_fwd_launch(x.data_ptr(), w.data_ptr(), y.data_ptr(),
                       rstd.data_ptr(), (int)rows, H, (float)eps,
                       cur_stream());
    return {y, rstd};
}

std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor w, torch::Tensor rstd) {
    CHECK_IN(dy);
    CHECK_IN(x);
    CHECK_IN(w);
    CHECK_IN(rstd);
    const int H = x.size(-1);
    const long rows = x.numel() / H;
    auto dx = torch::empty_like(x);
    auto dw = torch::zeros({H}, x.options().dtype(torch::kFloat32));
    rmsnorm_bwd_launch(dy.data_ptr(), x.data_ptr(), w.data_ptr(),
                       rstd.data_ptr(), dx.data_ptr(), dw.data_ptr(),
                       (int)rows, H, cur_stream());
    return {dx, dw};
}

torch::Tensor rope(torch::Tensor x, torch::Tensor cs, torch::Tensor pos,
                   double sign) {
    CHECK_IN(x);
    CHECK_IN(cs);
    CHECK_IN(pos);
    TORCH_CHECK(x.dim() == 3, "x must be [tokens, heads, D]");
    const int D = x.size(2);
    const int heads = x.size(1);
    const int rows = x.size(0);
    TORCH_CHECK(D % 2 == 0);
    auto y = torch::empty_like(x);
    rope_launch(x.data_ptr(), y.data_ptr(), cs.data_ptr(), pos.data_ptr(),
                rows, heads, D, (float)sign, cur_stream());
    return y;
}

torch::Tensor swiglu_fwd(torch::Tensor x) {
    CHECK_IN(x);
    const int I2 = x.size(-1);
    TORCH_CHECK(I2 % 16 == 0, "last dim must be a multiple of 16");
    const long rows = x.numel() / I2;
    auto sizes = x.sizes().vec();
    sizes.back() = I2 / 2;
    auto y = torch::empty(sizes, x.options());
    swiglu_fwd_launch(x.data_ptr(), y.data_ptr(), rows, I2 / 2, cur_stream());
    return y;
}

torch::Tensor swiglu_bwd(torch::Tensor dy, torch::Tensor x) {
    CHECK_IN(dy);
    CHECK_IN(x);
    const int I2 = x.size(-1);
    const long rows = x.numel() / I2;
    auto dx = torch::empty_like(x);
    swiglu_bwd_launch(dy.data_ptr(), x.data_ptr(), dx.data_ptr(), rows, I2 / 2,
                      cur_stream());
    return dx;
}

std::vector<torch::Tensor> ce_fwd(torch::Tensor logits, torch::Tensor labels) {
    CHECK_IN(logits);
    CHECK_IN(labels);
    TORCH_CHECK(labels.scalar_type() == torch::kInt64);
    const int V = logits.size(-1);
    TORCH_CHECK(V % 8 == 0, "fused CE requires vocab % 8 == 0 (pad the vocab)");
    const long rows = logits.numel() / V;
    auto opts = logits.options().dtype(torch::kFloat32);
    auto loss = torch::empty({rows}, opts);
    auto row_max = torch::empty({rows}, opts);
    auto row_sum = torch::empty({rows}, opts);
    ce_fwd_launch(logits.data_ptr(), labels.data_ptr(), loss.data_ptr(),
                  row_max.data_ptr(), row_sum.data_ptr(), rows, V,
                  cur_stream());
    return {loss, row_max, row_sum};
}

torch::Tensor ce_bwd(torch::Tensor logits, torch::Tensor labels,
                     torch::Tensor dloss, torch::Tensor row_max,
                     torch::Tensor row_sum) {
    CHECK_IN(logits);
    const int V = logits.size(-1);
    const long rows = logits.numel() / V;
    auto dlogits = torch::empty_like(logits);
    ce_bwd_launch(logits.data_ptr(), labels.data_ptr(), dloss.data_ptr(),
                  row_max.data_ptr(), row_sum.data_ptr(), dlogits.data_ptr(),
                  rows, V, cur_stream());
    return dlogits;
}

extern "C" {
void mfma_probe_launch(const void *, const void *, void *, void *, int,
                       hipStream_t);
void fa_fwd_launch(const void *, const void *, const void *, void *, void *,
                   int, int, int, int, int, int, hipStream_t);
void fa_bwd_launch(const void *, const void *, const void *, const void *,
                   const void *, const void *, void *, void *, void *, int,
                   int, int, int, int, int, hipStream_t);
}

std::vector<torch::Tensor> mfma_probe(torch::Tensor A, torch::Tensor B,
                                      long mode) {
    CHECK_IN(A);
    CHECK_IN(B);
    auto f32 = A.options().dtype(torch::kFloat32);
    auto cm = torch::zeros({16, 16}, f32);
    auto cr = torch::zeros({64, 4}, f32);
    mfma_probe_launch(A.data_ptr(), B.data_ptr(), cm.data_ptr(), cr.data_ptr(),
                      (int)mode, cur_stream());
    return {cm, cr};
}

std::vector<torch::Tensor> fa_fwd(torch::Tensor q, torch::Tensor k,
                                  torch::Tensor v, bool causal) {
    CHECK_IN(q);
    CHECK_IN(k);
    CHECK_IN(v);
    TORCH_CHECK(q.dim() == 4, "q must be [B,S,Hq,Dqk]");
    const int B = q.size(0), S = q.size(1), Hq = q.size(2), Dqk = q.size(3);
    const int Hkv = k.size(2), Dv = v.size(3);
    TORCH_CHECK(Dqk == 128 || Dqk == 192, "fa kernel supports Dqk in {128,192}");
    TORCH_CHECK(Dv == 128, "fa kernel supports Dv=128");
    TORCH_CHECK(Hq % Hkv == 0);
    auto o = torch::empty({B, S, Hq, Dv}, q.options());
    auto lse = torch::empty({B, Hq, S}, q.options().dtype(torch::kFloat32));
    fa_fwd_launch(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                  lse.data_ptr(), B, S, Hq, Hkv, Dqk, causal ? 1 : 0,
                  cur_stream());
    return {o, lse};
}

std::vector<torch::Tensor> fa_bwd(torch::Tensor dout, torch::Tensor q,
                                  torch::Tensor k, torch::Tensor v,
                                  torch::Tensor o, torch::Tensor lse,
                                  bool causal) {
    CHECK_IN(dout);
    CHECK_IN(q);
    const int B = q.size(0), S = q.size(1), Hq = q.size(2), Dqk = q.size(3);
    const int Hkv = k.size(2), Dv = v.size(3);
    auto f32 = q.options().dtype(torch::kFloat32);
    auto dq = torch::empty({B, S, Hq, Dqk}, q.options());
    auto dkv = torch::zeros({B, S, Hkv, Dqk + Dv}, f32);
    auto dsum = torch::empty({B, Hq, S}, f32);
    fa_bwd_launch(dout.data_ptr(), q.data_ptr(), k.data_ptr(), v.data_ptr(),
                  o.data_ptr(), lse.data_ptr(), dq.data_ptr(), dkv.data_ptr(),
                  dsum.data_ptr(), B, S, Hq, Hkv, Dqk, causal ? 1 : 0,
                  cur_stream());
    auto dk = dkv.narrow(3, 0, Dqk).to(torch::kBFloat16);
    auto dv = dkv.narrow(3, Dqk, Dv).to(torch::kBFloat16);
    return {dq, dk.contiguous(), dv.contiguous()};
}

void wgrad_accum(torch::Tensor dout, torch::Tensor x, torch::Tensor main_grad);

extern "C" {
void gg_fwd(const void *, const void *, void *, int, int, int, int,
            hipStream_t);
void gg_dgrad(const void *, const void *, void *, int, int, int, int,
              hipStream_t);
void gg_wgrad(const void *, const void *, void *, int, int, int, int,
              hipStream_t);
}

// grouped expert GEMMs (see grouped_gemm.hip); weights are [E, N, K]
torch::Tensor grouped_fwd(torch::Tensor x, torch::Tensor w) {
    CHECK_IN(x);
    CHECK_IN(w);
    TORCH_CHECK(x.dim() == 3 && w.dim() == 3, "x [E,M,K], w [E,N,K]");
    const int E = x.size(0), M = x.size(1), K = x.size(2), N = w.size(1);
    TORCH_CHECK(w.size(0) == E && (int)w.size(2) == K);
    TORCH_CHECK(N % 128 == 0 && K % 128 == 0, "grouped fwd needs N%128, K%128");
    auto c = torch::empty({E, M, N}, x.options());
    gg_fwd(x.data_ptr(), w.data_ptr(), c.data_ptr(), E, M, N, K,
           cur_stream());
    return c;
}

torch::Tensor grouped_dgrad(torch::Tensor dout, torch::Tensor w) {
    CHECK_IN(dout);
    CHECK_IN(w);
    const int E = dout.size(0), M = dout.size(1), N = dout.size(2);
    const int K = w.size(2);
    TORCH_CHECK((int)w.size(1) == N);
    TORCH_CHECK(K % 128 == 0 && N % 32 == 0, "grouped dgrad needs K%128, N%32");
    auto dx = torch::empty({E, M, K}, dout.options());
    gg_dgrad(dout.data_ptr(), w.data_ptr(), dx.data_ptr(), E, M, N, K,
             cur_stream());
    return dx;
}

void grouped_wgrad(torch::Tensor dout, torch::Tensor x, torch::Tensor g) {
    CHECK_IN(dout);
    CHECK_IN(x);
    CHECK_IN(g);
    const int E = dout.size(0), M = dout.size(1), N = dout.size(2);
    const int K = x.size(2);
    TORCH_CHECK(g.scalar_type() == torch::kFloat32, "main_grad must be fp32");
    TORCH_CHECK((int)g.size(1) == N && (int)g.size(2) == K);
    TORCH_CHECK(N % 128 == 0 && K % 128 == 0, "grouped wgrad needs N%128, K%128");
    gg_wgrad(dout.data_ptr(), x.data_ptr(), g.data_ptr(), E, M, N, K,
             cur_stream());
}

// fused bf16 -> fp8 cast with delayed-scaling amax update; out must be a
// float8 tensor of x's shape, amax a zeroed fp32 [1], scale fp32 [1]
void fp8_cast(torch::Tensor x, torch::Tensor out, torch::Tensor amax,
              torch::Tensor scale, bool e5m2, double fmax) {
    CHECK_IN(x);
    TORCH_CHECK(out.is_cuda() && out.is_contiguous(), "out");
    TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "x must be bf16");
    TORCH_CHECK(x.numel() % 8 == 0, "numel % 8");
    TORCH_CHECK(out.numel() == x.numel(), "out size");
    fp8_cast_launch(x.data_ptr(), out.data_ptr(), amax.data_ptr(),
                    scale.data_ptr(), (long)x.numel(), e5m2 ? 1 : 0,
                    (float)fmax, cur_stream());
}

// cast + transpose in one pass: out [M,N] fp8 + out_t [N,M] fp8
void fp8_cast_t(torch::Tensor x, torch::Tensor out, torch::Tensor out_t,
                torch::Tensor amax, torch::Tensor scale, bool e5m2,
                double fmax) {
    CHECK_IN(x);
    TORCH_CHECK(x.dim() == 2 && x.scalar_type() == torch::kBFloat16, "x");
    const int M = x.size(0), N = x.size(1);
    TORCH_CHECK(M % 64 == 0 && N % 64 == 0, "dims must be multiples of 64");
    TORCH_CHECK(out.numel() == x.numel() && out_t.numel() == x.numel(),
                "out sizes");
    fp8_cast_t_launch(x.data_ptr(), out.data_ptr(), out_t.data_ptr(),
                      amax.data_ptr(), scale.data_ptr(), M, N,
                      e5m2 ? 1 : 0, (float)fmax, cur_stream());
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("fp8_cast", &fp8_cast,
          "fused bf16->fp8 cast + running amax (delayed scaling)");
    m.def("fp8_cast_t", &fp8_cast_t,
          "fused bf16->fp8 cast + transpose + running amax");
    m.def("wgrad_accum", &wgrad_accum,
          "main_grad(fp32) += dout^T @ x (bf16 in, hipBLAS GemmEx)");
    m.def("mfma_probe", &mfma_probe, "16x16x32 bf16 fragment-layout probe");
    m.def("fa_fwd", &fa_fwd, "flash attention forward (gfx950 MFMA)");
    m.def("fa_bwd", &fa_bwd, "flash attention backward (gfx950 MFMA)");
    m.def("rmsnorm_fwd", &rmsnorm_fwd, "fused RMSNorm forward (gfx950)");
    m.def("rmsnorm_bwd", &rmsnorm_bwd, "fused RMSNorm backward (gfx950)");
    m.def("rope", &rope, "fused RoPE (sign=+1 fwd, -1 bwd) (gfx950)");
    m.def("swiglu_fwd", &swiglu_fwd, "fused SwiGLU forward (gfx950)");
    m.def("swiglu_bwd", &swiglu_bwd, "fused SwiGLU backward (gfx950)");
    m.def("ce_fwd", &ce_fwd, "fused cross-entropy forward (gfx950)");
    m.def("ce_bwd", &ce_bwd, "fused cross-entropy backward (gfx950)");
    m.def("grouped_fwd", &grouped_fwd,
          "grouped expert GEMM fwd: [E,M,K]x[E,N,K]^T -> [E,M,N] (MFMA)");
    m.def("grouped_dgrad", &grouped_dgrad,
          "grouped expert GEMM dgrad: [E,M,N]x[E,N,K] -> [E,M,K] (MFMA)");
    m.def("grouped_wgrad", &grouped_wgrad,
          "grouped expert wgrad: fp32 G[E,N,K] += dout^T @ x (MFMA)");
}
