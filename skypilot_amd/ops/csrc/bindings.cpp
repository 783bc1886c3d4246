// Python bindings for the skypilot_amd CDNA4 kernels.
//
// Compiled with hipcc against libtorch; kernels live in the .hip TUs and
// are reached through extern "C" launch functions so this TU only needs
// the HIP runtime + torch headers.
#include <torch/extension.h>

#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

#include <vector>

#define CHECK_GPU(x) TORCH_CHECK(x.is_cuda(), #x " must be on GPU")
#define CHECK_CONTIG(x) TORCH_CHECK(x.is_contiguous(), #x " must be contiguous")
#define CHECK_BF16(x) \
  TORCH_CHECK(x.scalar_type() == at::kBFloat16, #x " must be bf16")

static hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

// ---- extern "C" launchers from the .hip files -----------------------------
extern "C" {
void rmsnorm_fwd_launch(const void*, const void*, void*, float*, long long,
                        int, float, hipStream_t);
void rmsnorm_bwd_launch(const void*, const void*, const void*, const float*,
                        void*, float*, long long, int, hipStream_t);
void rope_launch(const void*, void*, const float*, const float*, const int*,
                 long long, int, int, bool, hipStream_t);
void adamw_launch(void*, float*, const void*, float*, float*, long long,
                  float, float, float, float, float, int, float, hipStream_t);
void cross_entropy_launch(void*, const int*, float*, long long, int, float,
                          int, hipStream_t);
void attn_fwd_launch(const void*, const void*, const void*, void*, float*,
                     int, int, int, int, float, bool, hipStream_t);
void attn_bwd_launch(const void*, const void*, const void*, const void*,
                     const void*, const float*, float*, void*, void*, void*,
                     int, int, int, int, float, bool, hipStream_t);
void mfma_probe_launch(const void*, const void*, float*, hipStream_t);
void mfma32_probe_launch(const void*, const void*, float*, hipStream_t);
void trb16_probe_launch(float*, int, hipStream_t);
void permlane_probe_launch(float*, hipStream_t);
void skinny_gemm_launch(const void*, const void*, void*, int, int, int,
                        hipStream_t);
void skinny_gemm_swiglu_launch(const void*, const void*, void*, int, int,
                               int, hipStream_t);
void transpose_bf16_launch(const void*, void*, int, int, hipStream_t);
void skinny_gemm_fp8_launch(const void*, const float*, const void*, void*,
                            int, int, int, hipStream_t);
void gemm_nt_launch(const void*, const void*, void*, int, int, long long,
                    hipStream_t);
void adamw_mt_launch(const void*, const void*, const void*, const void*,
                     const void*, long long, float, float, float, float,
                     int, float, hipStream_t);
void decode_advance_launch(const void*, long long, void*, long long,
                           void*, const void*, int, hipStream_t);
void rmsnorm_res_launch(const void*, const void*, const void*, void*, void*,
                        int, int, float, hipStream_t);
void rope_kvwrite_launch(const void*, void*, void*, void*, const void*,
                         const void*, const void*, const void*, int, int,
                         int, int, int, hipStream_t);
void swiglu_fwd_launch(const void*, void*, long long, int, hipStream_t);
void swiglu_bwd_launch(const void*, const void*, void*, long long, int,
                       hipStream_t);
void skinny_gemm_fp8_norm_launch(const void*, const float*, const void*,
                                 const void*, const void*, void*, float,
                                 void*, int, int, int, hipStream_t);
void attn_decode_qkv_launch(const void*, void*, void*, void*, const void*,
                            const void*, const int*, const int*,
                            const int*, int, int, int, int, float,
                            hipStream_t);
void attn_decode_launch(const void*, const void*, const void*, void*,
                        const int*, const int*, int, int, int, int, float,
                        hipStream_t);
}

// ---- RMSNorm --------------------------------------------------------------
torch::Tensor rmsnorm_fwd(torch::Tensor x, torch::Tensor w,
                          torch::Tensor inv_rms, double eps) {
  CHECK_GPU(x); CHECK_CONTIG(x); CHECK_BF16(x);
  CHECK_GPU(w); CHECK_CONTIG(w); CHECK_BF16(w);
  const int H = (int)x.size(-1);
  TORCH_CHECK(H % 8 == 0, "hidden dim must be a multiple of 8");
  long long rows = x.numel() / H;
  auto y = torch::empty_like(x);
  rmsnorm_fwd_launch(x.data_ptr(), w.data_ptr(), y.data_ptr(),
                     inv_rms.data_ptr<float>(), rows, H, (float)eps,
                     cur_stream());
  return y;
}

std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor x, torch::Tensor w,
                                       torch::Tensor dy,
                                       torch::Tensor inv_rms) {
  CHECK_GPU(x); CHECK_CONTIG(x); CHECK_CONTIG(dy);
  const int H = (int)x.size(-1);
  long long rows = x.numel() / H;
  auto dx = torch::empty_like(x);
  auto dw = torch::zeros({(long)H},
                         x.options().dtype(at::kFloat));
  rmsnorm_bwd_launch(x.data_ptr(), w.data_ptr(), dy.data_ptr(),
                     inv_rms.data_ptr<float>(), dx.data_ptr(),
                     dw.data_ptr<float>(), rows, H, cur_stream());
  return {dx, dw};
}

// ---- RoPE -----------------------------------------------------------------
torch::Tensor rope(torch::Tensor x, torch::Tensor cos_tab,
                   torch::Tensor sin_tab, torch::Tensor positions,
                   bool backward) {
  CHECK_GPU(x); CHECK_CONTIG(x); CHECK_BF16(x);
  TORCH_CHECK(positions.scalar_type() == at::kInt);
  const int D = (int)x.size(-1);
  const int H = (int)x.size(-2);
  long long n_tokens = x.numel() / ((long long)H * D);
  auto y = torch::empty_like(x);
  rope_launch(x.data_ptr(), y.data_ptr(), cos_tab.data_ptr<float>(),
              sin_tab.data_ptr<float>(), positions.data_ptr<int>(), n_tokens,
              H, D, backward, cur_stream());
  return y;
}

// ---- AdamW ----------------------------------------------------------------
void adamw_step(std::vector<torch::Tensor> params_bf16,
                std::vector<torch::Tensor> params_master,
                std::vector<torch::Tensor> grads,
                std::vector<torch::Tensor> exp_avg,
                std::vector<torch::Tensor> exp_avg_sq, double lr, double beta1,
                double beta2, double eps, double weight_decay, int64_t step,
                double grad_scale, std::vector<bool> decay_mask) {
  auto stream = cur_stream();
  for (size_t i = 0; i < params_bf16.size(); ++i) {
    auto& p = params_bf16[i];
    long long n = p.numel();
    float wd = decay_mask.empty() || decay_mask[i] ? (float)weight_decay : 0.f;
    adamw_launch(p.data_ptr(), params_master[i].data_ptr<float>(),
                 grads[i].data_ptr(), exp_avg[i].data_ptr<float>(),
                 exp_avg_sq[i].data_ptr<float>(), n, (float)lr, (float)beta1,
                 (float)beta2, (float)eps, wd, (int)step, (float)grad_scale,
                 stream);
  }
}

// ---- Cross entropy --------------------------------------------------------
torch::Tensor cross_entropy_fused(torch::Tensor logits, torch::Tensor targets,
                                  double grad_scale, int64_t ignore_index) {
  CHECK_GPU(logits); CHECK_CONTIG(logits); CHECK_BF16(logits);
  TORCH_CHECK(targets.scalar_type() == at::kInt);
  const int V = (int)logits.size(-1);
  long long N = logits.numel() / V;
  auto loss = torch::empty({(long)N}, logits.options().dtype(at::kFloat));
  cross_entropy_launch(logits.data_ptr(), targets.data_ptr<int>(),
                       loss.data_ptr<float>(), N, V, (float)grad_scale,
                       (int)ignore_index, cur_stream());
  return loss;
}

// ---- Attention ------------------------------------------------------------
std::vector<torch::Tensor> attn_fwd(torch::Tensor Q, torch::Tensor K,
                                    torch::Tensor V, double scale,
                                    bool causal) {
  CHECK_GPU(Q); CHECK_CONTIG(Q); CHECK_CONTIG(K); CHECK_CONTIG(V);
  CHECK_BF16(Q);
  const int B = (int)Q.size(0), S = (int)Q.size(1), Hq = (int)Q.size(2),
            D = (int)Q.size(3);
  const int Hkv = (int)K.size(2);
  TORCH_CHECK(D == 128, "attention kernel requires head_dim=128");
  TORCH_CHECK(S % 64 == 0, "seq len must be a multiple of 64");
  TORCH_CHECK(Hq % Hkv == 0);
  auto O = torch::empty_like(Q);
  auto lse = torch::empty({B, Hq, S}, Q.options().dtype(at::kFloat));
  attn_fwd_launch(Q.data_ptr(), K.data_ptr(), V.data_ptr(), O.data_ptr(),
                  lse.data_ptr<float>(), B, S, Hq, Hkv, (float)scale, causal,
                  cur_stream());
  return {O, lse};
}

std::vector<torch::Tensor> attn_bwd(torch::Tensor Q, torch::Tensor K,
                                    torch::Tensor V, torch::Tensor O,
                                    torch::Tensor dO, torch::Tensor lse,
                                    double scale, bool causal) {
  CHECK_GPU(Q); CHECK_CONTIG(dO);
  const int B = (int)Q.size(0), S = (int)Q.size(1), Hq = (int)Q.size(2);
  const int Hkv = (int)K.size(2);
  auto dQ = torch::empty_like(Q);
  auto dK = torch::empty_like(K);
  auto dV = torch::empty_like(V);
  auto Dvec = torch::empty({(long)B * S * Hq}, Q.options().dtype(at::kFloat));
  attn_bwd_launch(Q.data_ptr(), K.data_ptr(), V.data_ptr(), O.data_ptr(),
                  dO.data_ptr(), lse.data_ptr<float>(),
                  Dvec.data_ptr<float>(), dQ.data_ptr(), dK.data_ptr(),
                  dV.data_ptr(), B, S, Hq, Hkv, (float)scale, causal,
                  cur_stream());
  return {dQ, dK, dV};
}

torch::Tensor attn_decode(torch::Tensor Q, torch::Tensor Kc,
                          torch::Tensor Vc, torch::Tensor kv_lens,
                          torch::Tensor slot_ids, double scale) {
  CHECK_GPU(Q); CHECK_CONTIG(Q); CHECK_BF16(Q);
  TORCH_CHECK(kv_lens.scalar_type() == at::kInt);
  TORCH_CHECK(slot_ids.scalar_type() == at::kInt);
  const int B = (int)Q.size(0), Hq = (int)Q.size(1), D = (int)Q.size(2);
  const int S_max = (int)Kc.size(1), Hkv = (int)Kc.size(2);
  TORCH_CHECK(D == 128);
  auto O = torch::empty_like(Q);
  attn_decode_launch(Q.data_ptr(), Kc.data_ptr(), Vc.data_ptr(),
                     O.data_ptr(), kv_lens.data_ptr<int>(),
                     slot_ids.data_ptr<int>(), B, S_max, Hq,
                     Hkv, (float)scale, cur_stream());
  return O;
}

torch::Tensor attn_decode_qkv(torch::Tensor qkv, torch::Tensor Kc,
                              torch::Tensor Vc, torch::Tensor cos_tab,
                              torch::Tensor sin_tab,
                              torch::Tensor positions,
                              torch::Tensor kv_lens,
                              torch::Tensor slot_ids, int64_t Hq,
                              int64_t Hkv, double scale) {
  // fused rope + cache-write + decode attention (attention_decode.hip)
  CHECK_GPU(qkv); CHECK_CONTIG(qkv); CHECK_BF16(qkv);
  TORCH_CHECK(positions.scalar_type() == at::kInt);
  TORCH_CHECK(kv_lens.scalar_type() == at::kInt);
  TORCH_CHECK(slot_ids.scalar_type() == at::kInt);
  const int B = (int)qkv.size(0);
  const int S_max = (int)Kc.size(1);
  const int D = (int)Kc.size(3);
  TORCH_CHECK(D == 128, "attn_decode_qkv: head_dim must be 128");
  TORCH_CHECK(qkv.size(1) == (Hq + 2 * Hkv) * D,
              "attn_decode_qkv: qkv width mismatch");
  auto O = torch::empty({B, Hq, (long)D}, qkv.options());
  attn_decode_qkv_launch(qkv.data_ptr(), Kc.data_ptr(), Vc.data_ptr(),
                         O.data_ptr(), cos_tab.data_ptr(),
                         sin_tab.data_ptr(), positions.data_ptr<int>(),
                         kv_lens.data_ptr<int>(),
                         slot_ids.data_ptr<int>(), B, S_max, (int)Hq,
                         (int)Hkv, (float)scale, cur_stream());
  return O;
}

std::vector<torch::Tensor> skinny_gemm_fp8_norm(
    torch::Tensor x, torch::Tensor res, torch::Tensor nw, double eps,
    torch::Tensor W8, torch::Tensor scale, bool want_xout) {
  // norm-fused fp8 GEMV (skinny_gemm.hip): y = rmsnorm(x+res)*nw @ W8^T
  CHECK_GPU(x); CHECK_CONTIG(x); CHECK_BF16(x);
  const int N = (int)x.size(0), I = (int)x.size(1);
  const int O = (int)W8.size(0);
  TORCH_CHECK(I <= 4096 && I % 1024 == 0,
              "fp8_norm: I must be <= 4096 and 1024-aligned");
  TORCH_CHECK(N >= 1 && N <= 2, "fp8_norm: N <= 2");
  TORCH_CHECK(O % 2 == 0, "fp8_norm: O must be even");
  const bool has_res = res.defined() && res.numel() > 0;
  auto y = torch::empty({N, O}, x.options());
  torch::Tensor xout;
  if (want_xout) xout = torch::empty_like(x);
  skinny_gemm_fp8_norm_launch(
      W8.data_ptr(), scale.data_ptr<float>(), x.data_ptr(),
      has_res ? res.data_ptr() : nullptr, nw.data_ptr(),
      want_xout ? xout.data_ptr() : nullptr, (float)eps, y.data_ptr(),
      N, I, O, cur_stream());
  if (!want_xout) xout = x;
  return {xout, y};
}

void adamw_step_mt(torch::Tensor ptrs, torch::Tensor wd_arr,
                   torch::Tensor sizes, torch::Tensor chunk_tensor,
                   torch::Tensor chunk_start, double lr, double beta1,
                   double beta2, double eps, int64_t step,
                   double grad_scale) {
  // One launch for the whole parameter list (adamw.hip multi-tensor).
  CHECK_GPU(ptrs); CHECK_GPU(chunk_tensor);
  adamw_mt_launch(ptrs.data_ptr(), wd_arr.data_ptr(), sizes.data_ptr(),
                  chunk_tensor.data_ptr(), chunk_start.data_ptr(),
                  (long long)chunk_tensor.numel(), (float)lr, (float)beta1,
                  (float)beta2, (float)eps, (int)step, (float)grad_scale,
                  cur_stream());
}

torch::Tensor skinny_gemm_swiglu(torch::Tensor GU, torch::Tensor W) {
  // Y[N,O] = silu(G)*U @ W^T where GU = [N, 2M] packed gate|up.
  CHECK_GPU(GU); CHECK_CONTIG(GU); CHECK_BF16(GU);
  CHECK_GPU(W); CHECK_CONTIG(W); CHECK_BF16(W);
  const int N = (int)GU.size(0), M = (int)GU.size(1) / 2;
  const int O = (int)W.size(0);
  TORCH_CHECK(W.size(1) == M, "skinny_gemm_swiglu: dims mismatch");
  TORCH_CHECK(N >= 1 && N <= 4, "skinny_gemm_swiglu: N must be 1..4");
  TORCH_CHECK(M % 512 == 0, "skinny_gemm_swiglu: M %% 512 != 0");
  auto y = torch::empty({N, O}, GU.options());
  skinny_gemm_swiglu_launch(W.data_ptr(), GU.data_ptr(), y.data_ptr(), N,
                            M, O, cur_stream());
  return y;
}

// Transpose-then-NT weight-gradient GEMM (wgrad_gemm.hip).
torch::Tensor transpose_bf16(torch::Tensor X) {
  CHECK_GPU(X); CHECK_CONTIG(X); CHECK_BF16(X);
  int R = X.size(0), C = X.size(1);
  TORCH_CHECK(R % 64 == 0 && C % 64 == 0, "transpose_bf16: dims % 64");
  auto Y = torch::empty({C, R}, X.options());
  transpose_bf16_launch(X.data_ptr(), Y.data_ptr(), R, C, cur_stream());
  return Y;
}

torch::Tensor gemm_nt(torch::Tensor A, torch::Tensor B) {
  // C[i,j] = sum_m A[i,m] * B[j,m]
  CHECK_GPU(A); CHECK_CONTIG(A); CHECK_BF16(A);
  CHECK_GPU(B); CHECK_CONTIG(B); CHECK_BF16(B);
  long long M = A.size(1);
  int I = A.size(0), J = B.size(0);
  TORCH_CHECK(B.size(1) == M, "gemm_nt: inner dims mismatch");
  TORCH_CHECK(M % 64 == 0, "gemm_nt: M % 64 != 0");
  TORCH_CHECK(I % 128 == 0 && J % 256 == 0, "gemm_nt: I%128/J%256");
  auto C = torch::empty({I, J}, A.options());
  gemm_nt_launch(A.data_ptr(), B.data_ptr(), C.data_ptr(), I, J, M,
                 cur_stream());
  return C;
}

torch::Tensor wgrad_tn(torch::Tensor dy, torch::Tensor x) {
  // dW[i,j] = sum_m dy[m,i] * x[m,j]  (nn.Linear weight grad).
  CHECK_GPU(dy); CHECK_CONTIG(dy); CHECK_BF16(dy);
  CHECK_GPU(x); CHECK_CONTIG(x); CHECK_BF16(x);
  long long M = dy.size(0);
  int I = dy.size(1), J = x.size(1);
  TORCH_CHECK(x.size(0) == M, "wgrad_tn: token dims mismatch");
  auto dyT = transpose_bf16(dy);
  auto xT = transpose_bf16(x);
  return gemm_nt(dyT, xT);
}

torch::Tensor skinny_gemm_fp8(torch::Tensor X, torch::Tensor W8,
                              torch::Tensor scale) {
  // Y[N,O] = X[N,I] @ (scale[o] * W8[O,I])^T ; W8 = OCP e4m3fn bytes.
  CHECK_GPU(X); CHECK_CONTIG(X); CHECK_BF16(X);
  CHECK_GPU(W8); CHECK_CONTIG(W8);
  TORCH_CHECK(W8.scalar_type() == at::kByte ||
              W8.scalar_type() == at::kFloat8_e4m3fn,
              "W8 must be uint8/float8_e4m3fn");
  CHECK_GPU(scale); CHECK_CONTIG(scale);
  TORCH_CHECK(scale.scalar_type() == at::kFloat, "scale must be fp32");
  int N = X.size(0), I = X.size(1), O = W8.size(0);
  TORCH_CHECK(W8.size(1) == I, "skinny_gemm_fp8: inner dims mismatch");
  TORCH_CHECK(scale.numel() == O, "skinny_gemm_fp8: scale size");
  TORCH_CHECK(N >= 1 && N <= 8, "skinny_gemm_fp8: N must be 1..8");
  TORCH_CHECK(I % 16 == 0, "skinny_gemm_fp8: I must be 16-aligned");
  auto y = torch::empty({N, O}, X.options());
  skinny_gemm_fp8_launch(W8.data_ptr(), scale.data_ptr<float>(),
                         X.data_ptr(), y.data_ptr(), N, I, O,
                         cur_stream());
  return y;
}

torch::Tensor skinny_gemm(torch::Tensor X, torch::Tensor W) {
  // Y[N,O] = X[N,I] @ W[O,I]^T (decode GEMV; see skinny_gemm.hip)
  CHECK_GPU(X); CHECK_CONTIG(X); CHECK_BF16(X);
  CHECK_GPU(W); CHECK_CONTIG(W); CHECK_BF16(W);
  const int N = (int)X.size(0), I = (int)X.size(1), O = (int)W.size(0);
  TORCH_CHECK(W.size(1) == I, "skinny_gemm: inner dims mismatch");
  TORCH_CHECK(N >= 1 && N <= 8, "skinny_gemm: N must be 1..8");
  TORCH_CHECK(I % 512 == 0, "skinny_gemm: I must be a multiple of 512");
  auto y = torch::empty({N, O}, X.options());
  skinny_gemm_launch(W.data_ptr(), X.data_ptr(), y.data_ptr(), N, I, O,
                     cur_stream());
  return y;
}

std::vector<torch::Tensor> rmsnorm_res(torch::Tensor x, torch::Tensor res,
                                       torch::Tensor w, double eps) {
  // (x + res, rmsnorm(x + res) * w) in one kernel (decode_fused.hip)
  CHECK_GPU(x); CHECK_CONTIG(x); CHECK_BF16(x);
  CHECK_GPU(res); CHECK_CONTIG(res); CHECK_BF16(res);
  const int H = (int)x.size(-1);
  long long rows = x.numel() / H;
  TORCH_CHECK(res.sizes() == x.sizes(), "rmsnorm_res: shape mismatch");
  TORCH_CHECK(H % 256 == 0 && H <= 8192, "rmsnorm_res: H must be a "
              "multiple of 256 and <= 8192");
  auto x_out = torch::empty_like(x);
  auto h_out = torch::empty_like(x);
  rmsnorm_res_launch(x.data_ptr(), res.data_ptr(), w.data_ptr(),
                     x_out.data_ptr(), h_out.data_ptr(), (int)rows, H,
                     (float)eps, cur_stream());
  return {x_out, h_out};
}

void decode_advance(torch::Tensor logits, torch::Tensor stage,
                    torch::Tensor ring, torch::Tensor ctr) {
  // fused greedy argmax + in-graph decode state bump (decode_fused.hip)
  CHECK_GPU(logits); CHECK_CONTIG(logits); CHECK_BF16(logits);
  CHECK_GPU(stage); CHECK_CONTIG(stage);
  CHECK_GPU(ring); CHECK_CONTIG(ring);
  TORCH_CHECK(stage.dtype() == torch::kInt32, "decode_advance: stage int32");
  TORCH_CHECK(ring.dtype() == torch::kInt64, "decode_advance: ring int64");
  TORCH_CHECK(ctr.dtype() == torch::kInt64, "decode_advance: ctr int64");
  const long long V = logits.size(-1);
  const long long b = logits.numel() / V;
  TORCH_CHECK(stage.size(0) == 6 && stage.size(1) == b,
              "decode_advance: stage must be [6, b]");
  TORCH_CHECK(ring.size(1) == b, "decode_advance: ring must be [chunk, b]");
  decode_advance_launch(logits.data_ptr(), V, stage.data_ptr(), b,
                        ring.data_ptr(), ctr.data_ptr(),
                        (int)ring.size(0), cur_stream());
}

torch::Tensor rope_kvwrite(torch::Tensor qkv, torch::Tensor kc,
                           torch::Tensor vc, torch::Tensor cos_tab,
                           torch::Tensor sin_tab, torch::Tensor positions,
                           torch::Tensor slot_ids, int64_t Hq,
                           int64_t Hkv) {
  // packed-qkv rope + KV-cache scatter (decode_fused.hip)
  CHECK_GPU(qkv); CHECK_CONTIG(qkv); CHECK_BF16(qkv);
  const int D = (int)kc.size(3), S_max = (int)kc.size(1);
  const int n = (int)qkv.size(0);
  TORCH_CHECK(D == 128, "rope_kvwrite: head_dim must be 128");
  TORCH_CHECK(qkv.size(1) == (Hq + 2 * Hkv) * D,
              "rope_kvwrite: packed width mismatch");
  auto q = torch::empty({n, (long)Hq, (long)D}, qkv.options());
  rope_kvwrite_launch(qkv.data_ptr(), q.data_ptr(), kc.data_ptr(),
                      vc.data_ptr(), cos_tab.data_ptr(),
                      sin_tab.data_ptr(), positions.data_ptr(),
                      slot_ids.data_ptr(), n, (int)Hq, (int)Hkv, D, S_max,
                      cur_stream());
  return q;
}

torch::Tensor swiglu_fwd(torch::Tensor gu) {
  CHECK_GPU(gu); CHECK_CONTIG(gu); CHECK_BF16(gu);
  const int M2 = (int)gu.size(-1);
  long long rows = gu.numel() / M2;
  auto sizes = gu.sizes().vec();
  sizes.back() = M2 / 2;
  auto y = torch::empty(sizes, gu.options());
  swiglu_fwd_launch(gu.data_ptr(), y.data_ptr(), rows, M2 / 2,
                    cur_stream());
  return y;
}

torch::Tensor swiglu_bwd(torch::Tensor gu, torch::Tensor dy) {
  CHECK_GPU(gu); CHECK_CONTIG(dy);
  const int M2 = (int)gu.size(-1);
  long long rows = gu.numel() / M2;
  auto dgu = torch::empty_like(gu);
  swiglu_bwd_launch(gu.data_ptr(), dy.data_ptr(), dgu.data_ptr(), rows,
                    M2 / 2, cur_stream());
  return dgu;
}

// ---- MFMA layout probe (used by tests/test_gpu_mfma.py) -------------------
torch::Tensor mfma_probe(torch::Tensor A, torch::Tensor B) {
  CHECK_GPU(A); CHECK_BF16(A); CHECK_CONTIG(A); CHECK_CONTIG(B);
  TORCH_CHECK(A.size(0) == 16 && A.size(1) == 32);
  TORCH_CHECK(B.size(0) == 32 && B.size(1) == 16);
  auto C = torch::zeros({16, 16}, A.options().dtype(at::kFloat));
  mfma_probe_launch(A.data_ptr(), B.data_ptr(), C.data_ptr<float>(),
                    cur_stream());
  return C;
}

torch::Tensor mfma32_probe(torch::Tensor A, torch::Tensor B) {
  CHECK_GPU(A); CHECK_BF16(A); CHECK_CONTIG(A); CHECK_CONTIG(B);
  TORCH_CHECK(A.size(0) == 32 && A.size(1) == 16);
  TORCH_CHECK(B.size(0) == 16 && B.size(1) == 32);
  auto C = torch::zeros({32, 32}, A.options().dtype(at::kFloat));
  mfma32_probe_launch(A.data_ptr(), B.data_ptr(), C.data_ptr<float>(),
                      cur_stream());
  return C;
}

torch::Tensor trb16_probe(int64_t mode, torch::Tensor ref_dev) {
  auto out = torch::zeros({64, 4}, ref_dev.options().dtype(at::kFloat));
  trb16_probe_launch(out.data_ptr<float>(), (int)mode, cur_stream());
  return out;
}

torch::Tensor permlane_probe(torch::Tensor ref_dev) {
  auto out = torch::zeros({64, 2}, ref_dev.options().dtype(at::kFloat));
  permlane_probe_launch(out.data_ptr<float>(), cur_stream());
  return out;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm_fwd", &rmsnorm_fwd);
  m.def("rmsnorm_bwd", &rmsnorm_bwd);
  m.def("rope", &rope);
  m.def("adamw_step", &adamw_step);
  m.def("cross_entropy_fused", &cross_entropy_fused);
  m.def("attn_fwd", &attn_fwd);
  m.def("attn_bwd", &attn_bwd);
  m.def("attn_decode", &attn_decode);
  m.def("skinny_gemm", &skinny_gemm);
  m.def("skinny_gemm_fp8", &skinny_gemm_fp8);
  m.def("transpose_bf16", &transpose_bf16);
  m.def("gemm_nt", &gemm_nt);
  m.def("wgrad_tn", &wgrad_tn);
  m.def("adamw_step_mt", &adamw_step_mt);
  m.def("skinny_gemm_swiglu", &skinny_gemm_swiglu);
  m.def("rmsnorm_res", &rmsnorm_res);
  m.def("rope_kvwrite", &rope_kvwrite);
  m.def("decode_advance", &decode_advance);
  m.def("attn_decode_qkv", &attn_decode_qkv);
  m.def("skinny_gemm_fp8_norm", &skinny_gemm_fp8_norm);
  m.def("swiglu_fwd", &swiglu_fwd);
  m.def("swiglu_bwd", &swiglu_bwd);
  m.def("mfma_probe", &mfma_probe);
  m.def("mfma32_probe", &mfma32_probe);
  m.def("trb16_probe", &trb16_probe);
  m.def("permlane_probe", &permlane_probe);
}
