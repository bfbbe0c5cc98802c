// Python bindings for the shallowspeed_amd HIP/CDNA4 kernels.
//
// Thin torch-extension layer: shape/dtype/contiguity checks + launch on
// the current torch stream.  All compute lives in gemm.hip /
// elementwise.hip (raw HIP, extern "C" launchers).

#include <torch/extension.h>

#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>

extern "C" {
void ss_gemm_nt(const void*, const void*, const void*, const void*, void*,
                int, int, int, bool, hipStream_t);
void ss_wgrad_tn(const void*, const void*, const void*, void*, void*, int,
                 int, int, int, hipStream_t);
void ss_wgrad_tn_multi(const void*, int, bool, void*, void*, int, int, int,
                       int, hipStream_t);
void ss_colsum(const void*, const void*, void*, int, int, hipStream_t);
void ss_relu_fwd(const void*, void*, long, hipStream_t);
void ss_relu_bwd(const void*, const void*, void*, long, hipStream_t);
void ss_softmax_fwd(const void*, void*, int, int, hipStream_t);
void ss_softmax_bwd(const void*, const void*, void*, int, int, hipStream_t);
void ss_head_mse_bwd(const void*, const void*, void*, int, int, float,
                     hipStream_t);
void ss_head_xent_bwd(const void*, const void*, void*, long, float,
                      hipStream_t);
void ss_transpose_bf16(const void*, void*, int, int, hipStream_t);
void ss_sgd_multi2(const void*, const void*, int, float, float, float,
                   hipStream_t);
void ss_adamw_multi2(const void*, const void*, int, float, float, float,
                     float, float, float, float, hipStream_t);
void ss_adamw_multi(const void*, int, long, float, float, float, float,
                    float, float, float, hipStream_t);
void ss_sgd_multi(const void*, int, long, float, float, float,
                  hipStream_t);
void ss_ln_fwd(const void*, const void*, const void*, void*, void*, void*,
               int, int, float, hipStream_t);
void ss_ln_bwd_dx(const void*, const void*, const void*, const void*,
                  const void*, void*, int, int, hipStream_t);
void ss_ln_bwd_dparam(const void*, const void*, const void*, const void*,
                      void*, void*, int, int, hipStream_t);
void ss_gelu_fwd(const void*, void*, long, hipStream_t);
void ss_gelu_bwd(const void*, const void*, void*, long, hipStream_t);
void ss_row_argmax(const void*, void*, int, int, hipStream_t);
}
// C++-linkage (gemm256.hip / wgrad256.hip / fused_mlp.hip)
bool ss_fused_mlp_fwd(const void*, const void*, int, int, int, int, void*,
                      hipStream_t);
bool ss_gemm_nt_256w(const void*, const void*, const void*, void*, int, int,
                     int, bool, hipStream_t);
bool ss_gemm_nt_256(const void*, const void*, const void*, void*, int, int,
                    int, bool, hipStream_t);
bool ss_wgrad_tn_256(const void*, const void*, void*, int, int, int,
                     hipStream_t);
void ss_fp8_quantize(const void*, void*, void*, int, int, hipStream_t);
bool ss_gemm_nt_f8(const void*, const void*, const void*, const void*,
                   const void*, void*, int, int, int, bool, hipStream_t);
bool ss_gemm_nt_f8_q(const void*, const void*, const void*, const void*,
                     const void*, void*, void*, int, int, int, bool,
                     hipStream_t);

namespace {

hipStream_t cur_stream() {
    return (hipStream_t)at::cuda::getCurrentCUDAStream().stream();
}

void check_bf16(const torch::Tensor& t, const char* name) {
    TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
    TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
    TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

void check_f32(const torch::Tensor& t, const char* name) {
    TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
    TORCH_CHECK(t.scalar_type() == torch::kFloat, name, " must be f32");
    TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

bool has(const torch::Tensor& t) { return t.defined() && t.numel() > 0; }

torch::Tensor gemm_nt(torch::Tensor a, torch::Tensor b, torch::Tensor bias,
                      torch::Tensor mask, bool relu) {
    check_bf16(a, "a");
    check_bf16(b, "b");
    TORCH_CHECK(a.dim() == 2 && b.dim() == 2, "a/b must be 2-D");
    const int M = a.size(0), K = a.size(1), N = b.size(0);
    TORCH_CHECK(b.size(1) == K, "K mismatch: ", K, " vs ", b.size(1));
    const void* bias_p = nullptr;
    if (has(bias)) {
        check_bf16(bias, "bias");
        TORCH_CHECK(bias.numel() == N, "bias size");
        bias_p = bias.data_ptr();
    }
    const void* mask_p = nullptr;
    if (has(mask)) {
        check_bf16(mask, "mask");
        TORCH_CHECK(mask.sizes() == a.sizes(), "mask must match A");
        mask_p = mask.data_ptr();
    }
    auto c = torch::empty({M, N}, a.options());
    ss_gemm_nt(a.data_ptr(), b.data_ptr(), bias_p, mask_p, c.data_ptr(), M, N,
               K, relu, cur_stream());
    return c;
}

torch::Tensor gemm_nt_256(torch::Tensor a, torch::Tensor b,
                          torch::Tensor bias, bool relu) {
    check_bf16(a, "a");
    check_bf16(b, "b");
    const int M = a.size(0), K = a.size(1), N = b.size(0);
    TORCH_CHECK(b.size(1) == K, "K mismatch");
    const void* bias_p = nullptr;
    if (has(bias)) {
        check_bf16(bias, "bias");
        bias_p = bias.data_ptr();
    }
    auto c = torch::empty({M, N}, a.options());
    TORCH_CHECK(ss_gemm_nt_256(a.data_ptr(), b.data_ptr(), bias_p,
                               c.data_ptr(), M, N, K, relu, cur_stream()),
                "shape outside the 256-tile tier: ", M, "x", N, "x", K);
    return c;
}

torch::Tensor gemm_nt_256w(torch::Tensor a, torch::Tensor b,
                           torch::Tensor bias, bool relu) {
    check_bf16(a, "a");
    check_bf16(b, "b");
    const int M = a.size(0), K = a.size(1), N = b.size(0);
    TORCH_CHECK(b.size(1) == K, "K mismatch");
    const void* bias_p = nullptr;
    if (has(bias)) {
        check_bf16(bias, "bias");
        bias_p = bias.data_ptr();
    }
    auto c = torch::empty({M, N}, a.options());
    TORCH_CHECK(ss_gemm_nt_256w(a.data_ptr(), b.data_ptr(), bias_p,
                                c.data_ptr(), M, N, K, relu, cur_stream()),
                "shape outside the 256w tier: ", M, "x", N, "x", K);
    return c;
}

std::vector<torch::Tensor> fp8_quantize(torch::Tensor x) {
    check_bf16(x, "x");
    TORCH_CHECK(x.dim() == 2 && x.size(1) % 128 == 0,
                "x must be 2-D with K % 128 == 0");
    const int R = x.size(0), K = x.size(1);
    auto q = torch::empty({R, K}, x.options().dtype(torch::kUInt8));
    auto s = torch::empty({K / 128, R, 4}, x.options().dtype(torch::kUInt8));
    ss_fp8_quantize(x.data_ptr(), q.data_ptr(), s.data_ptr(), R, K,
                    cur_stream());
    return {q, s};
}

torch::Tensor gemm_nt_f8(torch::Tensor a, torch::Tensor asc, torch::Tensor b,
                         torch::Tensor bsc, torch::Tensor bias, bool relu) {
    TORCH_CHECK(a.scalar_type() == torch::kUInt8 &&
                b.scalar_type() == torch::kUInt8, "a/b must be u8 (e4m3)");
    TORCH_CHECK(a.is_contiguous() && b.is_contiguous() &&
                asc.is_contiguous() && bsc.is_contiguous(), "contiguous");
    const int M = a.size(0), K = a.size(1), N = b.size(0);
    TORCH_CHECK(b.size(1) == K, "K mismatch");
    const void* bias_p = nullptr;
    if (has(bias)) {
        check_bf16(bias, "bias");
        bias_p = bias.data_ptr();
    }
    auto c = torch::empty({M, N},
                          a.options().dtype(torch::kBFloat16));
    TORCH_CHECK(ss_gemm_nt_f8(a.data_ptr(), asc.data_ptr(), b.data_ptr(),
                              bsc.data_ptr(), bias_p, c.data_ptr(), M, N, K,
                              relu, cur_stream()),
                "shape outside the fp8 tier: ", M, "x", N, "x", K);
    return c;
}

std::vector<torch::Tensor> gemm_nt_f8_q(torch::Tensor a, torch::Tensor asc,
                                         torch::Tensor b, torch::Tensor bsc,
                                         torch::Tensor bias, bool relu) {
    TORCH_CHECK(a.scalar_type() == torch::kUInt8 &&
                b.scalar_type() == torch::kUInt8, "a/b must be u8 (e4m3)");
    const int M = a.size(0), K = a.size(1), N = b.size(0);
    TORCH_CHECK(b.size(1) == K, "K mismatch");
    const void* bias_p = nullptr;
    if (has(bias)) {
        check_bf16(bias, "bias");
        bias_p = bias.data_ptr();
    }
    auto cq = torch::empty({M, N}, a.options());
    auto cs = torch::empty({N / 128, M, 4}, a.options());
    TORCH_CHECK(ss_gemm_nt_f8_q(a.data_ptr(), asc.data_ptr(), b.data_ptr(),
                                bsc.data_ptr(), bias_p, cq.data_ptr(),
                                cs.data_ptr(), M, N, K, relu, cur_stream()),
                "shape outside the fp8 tier: ", M, "x", N, "x", K);
    return {cq, cs};
}

void wgrad_tn_256(torch::Tensor dy, torch::Tensor x, torch::Tensor gw) {
    check_bf16(dy, "dy");
    check_bf16(x, "x");
    check_f32(gw, "gw");
    const int Kb = dy.size(0), Mo = dy.size(1), N = x.size(1);
    TORCH_CHECK(x.size(0) == Kb, "batch mismatch");
    TORCH_CHECK(gw.size(0) == Mo && gw.size(1) == N, "gw shape");
    TORCH_CHECK(ss_wgrad_tn_256(dy.data_ptr(), x.data_ptr(), gw.data_ptr(),
                                Mo, N, Kb, cur_stream()),
                "shape outside the 256-tile wgrad tier");
}

void wgrad_tn(torch::Tensor dy, torch::Tensor x, torch::Tensor gw,
              torch::Tensor gb, torch::Tensor mask, int64_t split_k) {
    check_bf16(dy, "dy");
    check_bf16(x, "x");
    check_f32(gw, "gw");
    const int Kb = dy.size(0), Mo = dy.size(1), N = x.size(1);
    TORCH_CHECK(x.size(0) == Kb, "batch mismatch");
    TORCH_CHECK(gw.size(0) == Mo && gw.size(1) == N, "gw shape");
    const void* mask_p = nullptr;
    if (has(mask)) {
        check_bf16(mask, "mask");
        TORCH_CHECK(mask.sizes() == dy.sizes(), "mask must match dy");
        mask_p = mask.data_ptr();
    }
    void* gb_p = nullptr;
    if (has(gb)) {
        check_f32(gb, "gb");
        TORCH_CHECK(gb.numel() == Mo, "gb size");
        gb_p = gb.data_ptr();
    }
    // bias grad is FUSED into the wgrad kernel (n-tile-0 blocks sum
    // the already-masked LDS dY tile) — no separate colsum launch or
    // extra dY read on the hot path.
    ss_wgrad_tn(dy.data_ptr(), x.data_ptr(), mask_p, gw.data_ptr(), gb_p, Mo,
                N, Kb, (int)split_k, cur_stream());
}

void wgrad_tn_multi(torch::Tensor chunk_table, int64_t nchunks,
                    bool has_mask, torch::Tensor gw, torch::Tensor gb,
                    int64_t Mo, int64_t N, int64_t kb_chunk,
                    int64_t split_k) {
    // chunk_table: CUDA int64 [nchunks, 3] = {dY*, X*, mask*} per
    // µbatch; all chunks share (Kb, Mo, N).  Deferred-µbatch wgrad:
    // one launch accumulates every chunk into gw/gb.
    TORCH_CHECK(chunk_table.is_cuda() &&
                    chunk_table.scalar_type() == torch::kLong &&
                    chunk_table.is_contiguous() &&
                    chunk_table.dim() == 2 && chunk_table.size(1) == 3 &&
                    chunk_table.size(0) == nchunks,
                "chunk_table must be CUDA int64 [nchunks,3]");
    check_f32(gw, "gw");
    TORCH_CHECK(gw.size(0) == Mo && gw.size(1) == N, "gw shape");
    void* gb_p = nullptr;
    if (has(gb)) {
        check_f32(gb, "gb");
        TORCH_CHECK(gb.numel() == Mo, "gb size");
        gb_p = gb.data_ptr();
    }
    ss_wgrad_tn_multi(chunk_table.data_ptr(), (int)nchunks, has_mask,
                      gw.data_ptr(), gb_p, (int)Mo, (int)N, (int)kb_chunk,
                      (int)split_k, cur_stream());
}

torch::Tensor colsum(torch::Tensor dy, torch::Tensor mask) {
    check_bf16(dy, "dy");
    const void* mask_p = nullptr;
    if (has(mask)) {
        check_bf16(mask, "mask");
        mask_p = mask.data_ptr();
    }
    auto gb = torch::zeros({dy.size(1)},
                           dy.options().dtype(torch::kFloat));
    ss_colsum(dy.data_ptr(), mask_p, gb.data_ptr(), dy.size(0), dy.size(1),
              cur_stream());
    return gb;
}

torch::Tensor relu_fwd(torch::Tensor x) {
    check_bf16(x, "x");
    auto y = torch::empty_like(x);
    ss_relu_fwd(x.data_ptr(), y.data_ptr(), x.numel(), cur_stream());
    return y;
}

torch::Tensor relu_bwd(torch::Tensor dy, torch::Tensor y) {
    check_bf16(dy, "dy");
    check_bf16(y, "y");
    TORCH_CHECK(dy.sizes() == y.sizes(), "shape mismatch");
    auto dx = torch::empty_like(dy);
    ss_relu_bwd(dy.data_ptr(), y.data_ptr(), dx.data_ptr(), dy.numel(),
                cur_stream());
    return dx;
}

torch::Tensor softmax_fwd(torch::Tensor x) {
    check_bf16(x, "x");
    TORCH_CHECK(x.dim() == 2, "x must be 2-D");
    auto s = torch::empty_like(x);
    ss_softmax_fwd(x.data_ptr(), s.data_ptr(), x.size(0), x.size(1),
                   cur_stream());
    return s;
}

torch::Tensor softmax_bwd(torch::Tensor dy, torch::Tensor s) {
    check_bf16(dy, "dy");
    check_bf16(s, "s");
    TORCH_CHECK(dy.sizes() == s.sizes(), "shape mismatch");
    auto dx = torch::empty_like(dy);
    ss_softmax_bwd(dy.data_ptr(), s.data_ptr(), dx.data_ptr(), dy.size(0),
                   dy.size(1), cur_stream());
    return dx;
}

torch::Tensor head_mse_bwd(torch::Tensor s, torch::Tensor t,
                           double global_batch) {
    check_bf16(s, "probs");
    check_bf16(t, "target");
    TORCH_CHECK(s.sizes() == t.sizes(), "shape mismatch");
    auto dz = torch::empty_like(s);
    ss_head_mse_bwd(s.data_ptr(), t.data_ptr(), dz.data_ptr(), s.size(0),
                    s.size(1), (float)(1.0 / global_batch), cur_stream());
    return dz;
}

torch::Tensor head_xent_bwd(torch::Tensor s, torch::Tensor t,
                            double global_batch) {
    check_bf16(s, "probs");
    check_bf16(t, "target");
    TORCH_CHECK(s.sizes() == t.sizes(), "shape mismatch");
    auto dz = torch::empty_like(s);
    ss_head_xent_bwd(s.data_ptr(), t.data_ptr(), dz.data_ptr(), s.numel(),
                     (float)(1.0 / global_batch), cur_stream());
    return dz;
}

std::vector<torch::Tensor> ln_fwd(torch::Tensor x, torch::Tensor gamma,
                                  torch::Tensor beta, double eps) {
    check_bf16(x, "x");
    check_bf16(gamma, "gamma");
    check_bf16(beta, "beta");
    TORCH_CHECK(x.dim() == 2, "x must be 2-D");
    const int B = x.size(0), C = x.size(1);
    TORCH_CHECK(gamma.numel() == C && beta.numel() == C, "param size");
    auto y = torch::empty_like(x);
    auto mean = torch::empty({B}, x.options().dtype(torch::kFloat));
    auto rstd = torch::empty({B}, x.options().dtype(torch::kFloat));
    ss_ln_fwd(x.data_ptr(), gamma.data_ptr(), beta.data_ptr(), y.data_ptr(),
              mean.data_ptr(), rstd.data_ptr(), B, C, (float)eps,
              cur_stream());
    return {y, mean, rstd};
}

torch::Tensor ln_bwd_dx(torch::Tensor dy, torch::Tensor x,
                        torch::Tensor gamma, torch::Tensor mean,
                        torch::Tensor rstd) {
    check_bf16(dy, "dy");
    check_bf16(x, "x");
    check_bf16(gamma, "gamma");
    check_f32(mean, "mean");
    check_f32(rstd, "rstd");
    auto dx = torch::empty_like(dy);
    ss_ln_bwd_dx(dy.data_ptr(), x.data_ptr(), gamma.data_ptr(),
                 mean.data_ptr(), rstd.data_ptr(), dx.data_ptr(), dy.size(0),
                 dy.size(1), cur_stream());
    return dx;
}

void ln_bwd_dparam(torch::Tensor dy, torch::Tensor x, torch::Tensor mean,
                   torch::Tensor rstd, torch::Tensor dgamma,
                   torch::Tensor dbeta) {
    check_bf16(dy, "dy");
    check_bf16(x, "x");
    check_f32(dgamma, "dgamma");
    check_f32(dbeta, "dbeta");
    ss_ln_bwd_dparam(dy.data_ptr(), x.data_ptr(), mean.data_ptr(),
                     rstd.data_ptr(), dgamma.data_ptr(), dbeta.data_ptr(),
                     dy.size(0), dy.size(1), cur_stream());
}

torch::Tensor gelu_fwd(torch::Tensor z) {
    check_bf16(z, "z");
    auto y = torch::empty_like(z);
    ss_gelu_fwd(z.data_ptr(), y.data_ptr(), z.numel(), cur_stream());
    return y;
}

torch::Tensor gelu_bwd(torch::Tensor dy, torch::Tensor z) {
    check_bf16(dy, "dy");
    check_bf16(z, "z");
    auto dz = torch::empty_like(dy);
    ss_gelu_bwd(dy.data_ptr(), z.data_ptr(), dz.data_ptr(), dy.numel(),
                cur_stream());
    return dz;
}

torch::Tensor row_argmax(torch::Tensor x) {
    check_bf16(x, "x");
    TORCH_CHECK(x.dim() == 2, "x must be 2-D");
    auto out = torch::empty({x.size(0)}, x.options().dtype(torch::kInt));
    ss_row_argmax(x.data_ptr(), out.data_ptr(), x.size(0), x.size(1),
                  cur_stream());
    return out;
}

void sgd_multi(torch::Tensor desc, double lr, int64_t total,
               double momentum, double weight_decay) {
    // total passed by the caller (cached host-side) — no device sync.
    TORCH_CHECK(desc.is_cuda() && desc.scalar_type() == torch::kLong &&
                    desc.is_contiguous() && desc.dim() == 2 &&
                    desc.size(1) == 8,
                "desc must be CUDA int64 [T,8]");
    ss_sgd_multi(desc.data_ptr(), (int)desc.size(0), (long)total, (float)lr,
                 (float)momentum, (float)weight_decay, cur_stream());
}

void adamw_multi(torch::Tensor desc, double lr, int64_t total, double beta1,
                 double beta2, double eps, double weight_decay,
                 double inv_bc1, double inv_bc2) {
    TORCH_CHECK(desc.is_cuda() && desc.scalar_type() == torch::kLong &&
                    desc.is_contiguous() && desc.dim() == 2 &&
                    desc.size(1) == 9,
                "desc must be CUDA int64 [T,9]");
    ss_adamw_multi(desc.data_ptr(), (int)desc.size(0), (long)total,
                   (float)lr, (float)beta1, (float)beta2, (float)eps,
                   (float)weight_decay, (float)inv_bc1, (float)inv_bc2,
                   cur_stream());
}

void sgd_multi2(torch::Tensor desc, torch::Tensor bmap, double lr,
                double momentum, double weight_decay) {
    TORCH_CHECK(desc.is_cuda() && desc.scalar_type() == torch::kLong &&
                    desc.is_contiguous() && desc.dim() == 2 &&
                    desc.size(1) == 8,
                "desc must be CUDA int64 [T,8]");
    TORCH_CHECK(bmap.is_cuda() && bmap.scalar_type() == torch::kLong &&
                    bmap.is_contiguous() && bmap.dim() == 2 &&
                    bmap.size(1) == 2,
                "bmap must be CUDA int64 [B,2]");
    ss_sgd_multi2(desc.data_ptr(), bmap.data_ptr(), (int)bmap.size(0),
                  (float)lr, (float)momentum, (float)weight_decay,
                  cur_stream());
}

void adamw_multi2(torch::Tensor desc, torch::Tensor bmap, double lr,
                  double beta1, double beta2, double eps,
                  double weight_decay, double inv_bc1, double inv_bc2) {
    TORCH_CHECK(desc.is_cuda() && desc.scalar_type() == torch::kLong &&
                    desc.is_contiguous() && desc.dim() == 2 &&
                    desc.size(1) == 9,
                "desc must be CUDA int64 [T,9]");
    TORCH_CHECK(bmap.is_cuda() && bmap.scalar_type() == torch::kLong &&
                    bmap.is_contiguous() && bmap.dim() == 2 &&
                    bmap.size(1) == 2,
                "bmap must be CUDA int64 [B,2]");
    ss_adamw_multi2(desc.data_ptr(), bmap.data_ptr(), (int)bmap.size(0),
                    (float)lr, (float)beta1, (float)beta2, (float)eps,
                    (float)weight_decay, (float)inv_bc1, (float)inv_bc2,
                    cur_stream());
}

torch::Tensor fused_mlp_argmax(torch::Tensor x, torch::Tensor desc,
                               int64_t in_dim, int64_t nhidden,
                               int64_t cout) {
    check_bf16(x, "x");
    TORCH_CHECK(desc.is_cuda() && desc.scalar_type() == torch::kLong &&
                    desc.is_contiguous() &&
                    desc.numel() == (nhidden + 1) * 2,
                "desc must be CUDA int64 [(nhidden+1)*2]");
    const int M = x.size(0);
    TORCH_CHECK(x.size(1) == in_dim, "in_dim mismatch");
    auto out = torch::empty({M}, x.options().dtype(torch::kInt32));
    TORCH_CHECK(ss_fused_mlp_fwd(x.data_ptr(), desc.data_ptr(), M,
                                 (int)in_dim, (int)nhidden, (int)cout,
                                 out.data_ptr(), cur_stream()),
                "shape outside the fused-MLP tier");
    return out;
}

void transpose_bf16(torch::Tensor src, torch::Tensor dst) {
    check_bf16(src, "src");
    check_bf16(dst, "dst");
    const int rows = src.size(0), cols = src.size(1);
    TORCH_CHECK(src.dim() == 2 && dst.dim() == 2 &&
                    dst.size(0) == cols && dst.size(1) == rows &&
                    rows % 64 == 0 && cols % 64 == 0,
                "transpose_bf16 needs 64-aligned 2-D shapes");
    ss_transpose_bf16(src.data_ptr(), dst.data_ptr(), rows, cols,
                      cur_stream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("gemm_nt", &gemm_nt, "C = A @ B^T (+bias)(+relu) with optional A-mask");
    m.def("gemm_nt_256", &gemm_nt_256,
          "256-tile 8-phase GEMM (M%256==N%256==K%128==0)");
    m.def("gemm_nt_256w", &gemm_nt_256w,
          "256-tile 8-phase GEMM, 4-wave 128x128-wave-tile variant");
    m.def("wgrad_tn_256", &wgrad_tn_256,
          "256-tile 8-phase wgrad: gw += dy^T @ x (no mask/bias)");
    m.def("fp8_quantize", &fp8_quantize,
          "bf16 [R,K] -> (e4m3 u8 [R,K], e8m0 scales [K/128,R,4])");
    m.def("gemm_nt_f8", &gemm_nt_f8,
          "MX-fp8 256-tile GEMM: C=bf16(A@B^T) (+bias)(+relu)");
    m.def("gemm_nt_f8_q", &gemm_nt_f8_q,
          "MX-fp8 GEMM with FUSED output quantization -> (e4m3, scales)");
    m.def("wgrad_tn", &wgrad_tn, "gW += (dy⊙mask)^T @ x; gb += colsum (fused)");
    m.def("colsum", &colsum, "standalone column sum (bias grad)");
    m.def("wgrad_tn_multi", &wgrad_tn_multi,
          "chunked (deferred-µbatch) wgrad: one launch, many (dy,x) pairs");
    m.def("relu_fwd", &relu_fwd);
    m.def("relu_bwd", &relu_bwd);
    m.def("softmax_fwd", &softmax_fwd);
    m.def("softmax_bwd", &softmax_bwd);
    m.def("head_mse_bwd", &head_mse_bwd);
    m.def("head_xent_bwd", &head_xent_bwd);
    m.def("sgd_multi", &sgd_multi);
    m.def("adamw_multi", &adamw_multi);
    m.def("transpose_bf16", &transpose_bf16);
    m.def("sgd_multi2", &sgd_multi2);
    m.def("adamw_multi2", &adamw_multi2);
    m.def("fused_mlp_argmax", &fused_mlp_argmax,
          "persistent fused MLP forward + per-row argmax (serving)");
    m.def("ln_fwd", &ln_fwd);
    m.def("ln_bwd_dx", &ln_bwd_dx);
    m.def("ln_bwd_dparam", &ln_bwd_dparam);
    m.def("gelu_fwd", &gelu_fwd);
    m.def("gelu_bwd", &gelu_bwd);
    m.def("row_argmax", &row_argmax);
}
