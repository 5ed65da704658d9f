// pybind11 bindings for the maggy_amd HIP kernels (MI355X / gfx950).
//
// The chunk/tensor metadata tables are packed in Python (ops/fused_adam.py)
// and live in device memory as uint8 tensors; these bindings only validate,
// fetch the current stream and call the launch wrappers defined in
// maggy_kernels.hip.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

extern "C" void launch_multi_l2norm_sq(const void*, int, const void*, int,
                                       float*, hipStream_t);
extern "C" void launch_multi_fused_adam(const void*, int, const void*, int,
                                        float, float, float, float, float,
                                        float, float, const float*, float,
                                        float, hipStream_t);
extern "C" void launch_multi_fused_sgd(const void*, int, const void*, int,
                                       float, float, float, float, int, int,
                                       const float*, float, float,
                                       hipStream_t);
extern "C" void launch_bn_fwd(const void*, const void*, void*, long long,
                              int, const void*, const void*, void*, void*,
                              float, float, int, int, void*, void*,
                              hipStream_t);
extern "C" void launch_bn_bwd(const void*, const void*, const void*, void*,
                              void*, long long, int, const void*,
                              const void*, const void*, int, void*, void*,
                              void*, void*, hipStream_t);
extern "C" void launch_rms_fwd(const void*, const void*, void*, void*,
                               long long, int, float, hipStream_t);
extern "C" void launch_rms_bwd(const void*, const void*, const void*,
                               const void*, void*, void*, void*, long long,
                               int, hipStream_t);
extern "C" void launch_swiglu_fwd(const void*, const void*, void*,
                                  long long, hipStream_t);
extern "C" void launch_swiglu_bwd(const void*, const void*, const void*,
                                  void*, void*, long long, hipStream_t);
extern "C" void launch_reduce_sum_f32(const float*, long long, float*,
                                      hipStream_t);
extern "C" void launch_reduce_sum_bf16(const void*, long long, float*,
                                       hipStream_t);
extern "C" void launch_reduce_max_f32(const float*, long long, float*,
                                      hipStream_t);
extern "C" void launch_gemm_tn_bf16(const void*, const void*, void*, int,
                                    int, int, hipStream_t);
extern "C" void launch_gemm_tn_bf16_swiglu(const void*, const void*, void*,
                                           const void*, void*, int, int,
                                           int, hipStream_t);
extern "C" void launch_transpose_bf16(const void*, void*, int, int,
                                      hipStream_t);
extern "C" void launch_rope(const void*, void*, const void*, const void*,
                            long long, int, int, int, int, float,
                            hipStream_t);

namespace {

hipStream_t current_stream() {
  return at::cuda::getCurrentCUDAStream().stream();
}

void check_table(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be a CUDA tensor");
  TORCH_CHECK(t.dtype() == torch::kUInt8, name, " must be uint8");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

void multi_l2norm_sq(torch::Tensor chunks, int64_t n_chunks,
                     torch::Tensor tensors, bool grad_bf16,
                     torch::Tensor norm_sq) {
  check_table(chunks, "chunks");
  check_table(tensors, "tensors");
  TORCH_CHECK(norm_sq.is_cuda() && norm_sq.dtype() == torch::kFloat32);
  launch_multi_l2norm_sq(chunks.data_ptr(), (int)n_chunks,
                         tensors.data_ptr(), grad_bf16 ? 1 : 0,
                         norm_sq.data_ptr<float>(), current_stream());
}

void multi_fused_adam(torch::Tensor chunks, int64_t n_chunks,
                      torch::Tensor tensors, bool grad_bf16, double lr,
                      double beta1, double beta2, double eps,
                      double weight_decay, double bc1, double bc2,
                      c10::optional<torch::Tensor> norm_sq, double max_norm,
                      double grad_scale_inv) {
  check_table(chunks, "chunks");
  check_table(tensors, "tensors");
  const float* ns = nullptr;
  if (norm_sq.has_value()) ns = norm_sq->data_ptr<float>();
  launch_multi_fused_adam(chunks.data_ptr(), (int)n_chunks,
                          tensors.data_ptr(), grad_bf16 ? 1 : 0, (float)lr,
                          (float)beta1, (float)beta2, (float)eps,
                          (float)weight_decay, (float)bc1, (float)bc2, ns,
                          (float)max_norm, (float)grad_scale_inv,
                          current_stream());
}

void multi_fused_sgd(torch::Tensor chunks, int64_t n_chunks,
                     torch::Tensor tensors, bool grad_bf16, double lr,
                     double momentum, double weight_decay, double dampening,
                     bool nesterov, bool first_step,
                     c10::optional<torch::Tensor> norm_sq, double max_norm,
                     double grad_scale_inv) {
  check_table(chunks, "chunks");
  check_table(tensors, "tensors");
  const float* ns = nullptr;
  if (norm_sq.has_value()) ns = norm_sq->data_ptr<float>();
  launch_multi_fused_sgd(chunks.data_ptr(), (int)n_chunks,
                         tensors.data_ptr(), grad_bf16 ? 1 : 0, (float)lr,
                         (float)momentum, (float)weight_decay,
                         (float)dampening, nesterov ? 1 : 0,
                         first_step ? 1 : 0, ns, (float)max_norm,
                         (float)grad_scale_inv, current_stream());
}

constexpr int64_t kBnNB = 1024;  // must match BN_NB in fused_bn.hip

void bn_fwd(torch::Tensor x, c10::optional<torch::Tensor> residual,
            torch::Tensor y, int64_t M, int64_t C, torch::Tensor gamma,
            torch::Tensor beta, torch::Tensor running_mean,
            torch::Tensor running_var, double momentum, double eps,
            bool training, bool relu, torch::Tensor workspace,
            c10::optional<torch::Tensor> partials) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16);
  TORCH_CHECK(C >= 8 && C % 8 == 0 && (C & (C - 1)) == 0 && C <= 2048,
              "fused BN requires power-of-2 C in [8, 2048]");
  TORCH_CHECK(workspace.numel() >= 6 * C);
  void* part = nullptr;
  if (training) {
    TORCH_CHECK(partials.has_value() &&
                partials->numel() >= 2 * C * kBnNB);
    part = partials->data_ptr();
  }
  const void* res = residual.has_value() ? residual->data_ptr() : nullptr;
  launch_bn_fwd(x.data_ptr(), res, y.data_ptr(), M, (int)C,
                gamma.data_ptr(), beta.data_ptr(), running_mean.data_ptr(),
                running_var.data_ptr(), (float)momentum, (float)eps,
                training ? 1 : 0, relu ? 1 : 0, workspace.data_ptr(),
                part, current_stream());
}

void bn_bwd(torch::Tensor dy, torch::Tensor x, torch::Tensor y,
            torch::Tensor dx, c10::optional<torch::Tensor> dres, int64_t M,
            int64_t C, torch::Tensor gamma, torch::Tensor fwd_ws,
            bool relu, torch::Tensor dgamma, torch::Tensor dbeta,
            torch::Tensor bwd_ws, torch::Tensor partials) {
  TORCH_CHECK(dy.is_cuda() && dy.dtype() == torch::kBFloat16);
  TORCH_CHECK(bwd_ws.numel() >= 5 * C);
  TORCH_CHECK(partials.numel() >= 2 * C * kBnNB);
  const float* ws = fwd_ws.data_ptr<float>();
  void* dres_p = dres.has_value() ? dres->data_ptr() : nullptr;
  launch_bn_bwd(dy.data_ptr(), x.data_ptr(), y.data_ptr(), dx.data_ptr(),
                dres_p, M, (int)C, gamma.data_ptr(), (const void*)(ws + 2 * C),
                (const void*)(ws + 3 * C), relu ? 1 : 0, dgamma.data_ptr(),
                dbeta.data_ptr(), bwd_ws.data_ptr(), partials.data_ptr(),
                current_stream());
}

constexpr int64_t kRmsNB = 1024;  // must match RMS_NB in fused_rms.hip

void rms_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor y,
             torch::Tensor inv_rms, int64_t R, int64_t D, double eps) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16);
  TORCH_CHECK(w.dtype() == torch::kFloat32);
  TORCH_CHECK(D % 2048 == 0 && D / 2048 <= 4,
              "fused RMSNorm requires D a multiple of 2048, <= 8192");
  launch_rms_fwd(x.data_ptr(), w.data_ptr(), y.data_ptr(),
                 inv_rms.data_ptr(), R, (int)D, (float)eps,
                 current_stream());
}

void rms_bwd(torch::Tensor dy, torch::Tensor x, torch::Tensor w,
             torch::Tensor inv_rms, torch::Tensor dx,
             torch::Tensor dw_partials, torch::Tensor dw, int64_t R,
             int64_t D) {
  TORCH_CHECK(dy.is_cuda() && dy.dtype() == torch::kBFloat16);
  TORCH_CHECK(dw_partials.numel() >= D * kRmsNB);
  launch_rms_bwd(dy.data_ptr(), x.data_ptr(), w.data_ptr(),
                 inv_rms.data_ptr(), dx.data_ptr(), dw_partials.data_ptr(),
                 dw.data_ptr(), R, (int)D, current_stream());
}

void swiglu_fwd(torch::Tensor g, torch::Tensor u, torch::Tensor out) {
  TORCH_CHECK(g.is_cuda() && g.dtype() == torch::kBFloat16);
  TORCH_CHECK(g.numel() % 8 == 0);
  launch_swiglu_fwd(g.data_ptr(), u.data_ptr(), out.data_ptr(), g.numel(),
                    current_stream());
}

void swiglu_bwd(torch::Tensor dy, torch::Tensor g, torch::Tensor u,
                torch::Tensor dg, torch::Tensor du) {
  TORCH_CHECK(dy.is_cuda() && dy.dtype() == torch::kBFloat16);
  launch_swiglu_bwd(dy.data_ptr(), g.data_ptr(), u.data_ptr(),
                    dg.data_ptr(), du.data_ptr(), dy.numel(),
                    current_stream());
}

void reduce_sum(torch::Tensor in, torch::Tensor out) {
  TORCH_CHECK(in.is_cuda() && in.is_contiguous());
  TORCH_CHECK(out.is_cuda() && out.dtype() == torch::kFloat32);
  if (in.dtype() == torch::kFloat32) {
    launch_reduce_sum_f32(in.data_ptr<float>(), in.numel(),
                          out.data_ptr<float>(), current_stream());
  } else if (in.dtype() == torch::kBFloat16) {
    launch_reduce_sum_bf16(in.data_ptr(), in.numel(),
                           out.data_ptr<float>(), current_stream());
  } else {
    TORCH_CHECK(false, "reduce_sum: dtype must be float32 or bfloat16");
  }
}

void reduce_max(torch::Tensor in, torch::Tensor out) {
  TORCH_CHECK(in.is_cuda() && in.is_contiguous());
  TORCH_CHECK(in.dtype() == torch::kFloat32,
              "reduce_max: float32 only");
  launch_reduce_max_f32(in.data_ptr<float>(), in.numel(),
                        out.data_ptr<float>(), current_stream());
}

void check_gemm_operand(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be a CUDA tensor");
  TORCH_CHECK(t.dtype() == torch::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.dim() == 2, name, " must be 2-D");
}

// C[M][N] = A[M][K] @ W[N][K]^T (torch F.linear layout), bf16, fp32 acc
torch::Tensor gemm_tn(torch::Tensor a, torch::Tensor w) {
  check_gemm_operand(a, "a");
  check_gemm_operand(w, "w");
  const int64_t M = a.size(0), K = a.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K, "gemm_tn: inner dims mismatch");
  TORCH_CHECK(M % 256 == 0 && N % 256 == 0 && K % 64 == 0,
              "gemm_tn needs M,N % 256 == 0 and K % 64 == 0; got ", M, "x",
              N, "x", K);
  auto c = torch::empty({M, N}, a.options());
  launch_gemm_tn_bf16(a.data_ptr(), w.data_ptr(), c.data_ptr(), (int)M,
                      (int)N, (int)K, current_stream());
  return c;
}

// y3 = A @ W3^T with fused h = silu(y1) * y3 epilogue; returns (y3, h)
std::vector<torch::Tensor> gemm_tn_swiglu(torch::Tensor a, torch::Tensor w3,
                                          torch::Tensor y1) {
  check_gemm_operand(a, "a");
  check_gemm_operand(w3, "w3");
  check_gemm_operand(y1, "y1");
  const int64_t M = a.size(0), K = a.size(1), N = w3.size(0);
  TORCH_CHECK(w3.size(1) == K, "gemm_tn_swiglu: inner dims mismatch");
  TORCH_CHECK(y1.size(0) == M && y1.size(1) == N,
              "gemm_tn_swiglu: y1 must be [M][N]");
  TORCH_CHECK(M % 256 == 0 && N % 256 == 0 && K % 64 == 0,
              "gemm_tn_swiglu needs M,N % 256 == 0 and K % 64 == 0");
  auto y3 = torch::empty({M, N}, a.options());
  auto h = torch::empty({M, N}, a.options());
  launch_gemm_tn_bf16_swiglu(a.data_ptr(), w3.data_ptr(), y3.data_ptr(),
                             y1.data_ptr(), h.data_ptr(), (int)M, (int)N,
                             (int)K, current_stream());
  return {y3, h};
}

// out[c][r] = in[r][c], bf16, R and C multiples of 64
torch::Tensor transpose2d(torch::Tensor in) {
  check_gemm_operand(in, "in");
  const int64_t R = in.size(0), C = in.size(1);
  TORCH_CHECK(R % 64 == 0 && C % 64 == 0,
              "transpose2d needs dims % 64 == 0; got ", R, "x", C);
  auto out = torch::empty({C, R}, in.options());
  launch_transpose_bf16(in.data_ptr(), out.data_ptr(), (int)R, (int)C,
                        current_stream());
  return out;
}

// RoPE on [B, T, H, D] bf16 (t = dim 1), fp32 cos/sin [>=pos+T, D/2];
// sign=+1 forward rotation, -1 backward
torch::Tensor rope(torch::Tensor x, torch::Tensor cost, torch::Tensor sint,
                   int64_t pos, double sign) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16 &&
              x.is_contiguous() && x.dim() == 4);
  TORCH_CHECK(cost.dtype() == torch::kFloat32 && cost.is_contiguous());
  TORCH_CHECK(sint.dtype() == torch::kFloat32 && sint.is_contiguous());
  const int64_t B = x.size(0), T = x.size(1), H = x.size(2), D = x.size(3);
  TORCH_CHECK(D % 4 == 0, "rope needs head_dim % 4 == 0");
  TORCH_CHECK(cost.size(-1) == D / 2 && sint.size(-1) == D / 2,
              "cos/sin table width must be D/2");
  TORCH_CHECK(cost.size(0) >= pos + T && sint.size(0) >= pos + T,
              "rope tables too short: need pos+T rows");
  auto out = torch::empty_like(x);
  launch_rope(x.data_ptr(), out.data_ptr(), cost.data_ptr(),
              sint.data_ptr(), B * T * H, (int)D, (int)H, (int)T, (int)pos,
              (float)sign, current_stream());
  return out;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("multi_l2norm_sq", &multi_l2norm_sq,
        "sum of squares over the chunk table -> norm_sq[0]");
  m.def("multi_fused_adam", &multi_fused_adam,
        "fused multi-tensor AdamW step with in-kernel grad clip");
  m.def("multi_fused_sgd", &multi_fused_sgd,
        "fused multi-tensor SGD(+momentum) step with in-kernel grad clip");
  m.def("bn_fwd", &bn_fwd,
        "fused NHWC bf16 BatchNorm fwd (+residual +relu)");
  m.def("bn_bwd", &bn_bwd,
        "fused NHWC bf16 BatchNorm bwd (+relu mask +residual grad)");
  m.def("rms_fwd", &rms_fwd, "fused bf16 RMSNorm forward");
  m.def("rms_bwd", &rms_bwd, "fused bf16 RMSNorm backward");
  m.def("swiglu_fwd", &swiglu_fwd, "fused silu(g)*u forward");
  m.def("swiglu_bwd", &swiglu_bwd, "fused silu(g)*u backward");
  m.def("reduce_sum", &reduce_sum, "scalar sum reduction");
  m.def("reduce_max", &reduce_max, "scalar max reduction");
  m.def("gemm_tn", &gemm_tn,
        "bf16 MFMA GEMM C = A @ W^T (torch Linear layout)");
  m.def("gemm_tn_swiglu", &gemm_tn_swiglu,
        "bf16 MFMA GEMM y3 = A @ W3^T with fused h = silu(y1)*y3");
  m.def("transpose2d", &transpose2d, "bf16 2-D transpose");
  m.def("rope", &rope, "fused rotary embedding (bf16, fp32 tables)");
}
