// vescale_amd C++/HIP extension — single TU: kernels + torch bindings.
//
// Built in-tree for gfx950 only (no multi-backend dispatch, no CUDA shims).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "rmsnorm.hip"
#include "rope.hip"
#include "rope_qkv.hip"
#include "swiglu.hip"
#include "adamw.hip"
#include "cross_entropy.hip"
#include "philox_random.hip"
#include "gemm.hip"
#include "gemm8.hip"
#include "attention_bwd.hip"
#include "attention_fwd.hip"
#include "attention_bwd2.hip"

#include <vector>

namespace {

inline hipStream_t cur_stream() {
  return (hipStream_t)at::cuda::getCurrentCUDAStream().stream();
}

inline void check_bf16_contig(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == at::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.is_cuda(), name, " must be on device");
}

// ------------------------------ RMSNorm ------------------------------
std::vector<at::Tensor> rmsnorm_fwd(at::Tensor x, at::Tensor w, double eps) {
  check_bf16_contig(x, "x");
  check_bf16_contig(w, "w");
  int hidden = (int)x.size(-1);
  TORCH_CHECK(hidden % 8 == 0, "hidden must be a multiple of 8");
  int64_t rows = x.numel() / hidden;
  auto out = at::empty_like(x);
  auto rrms = at::empty({rows}, x.options().dtype(at::kFloat));
  int grid = (int)std::min<int64_t>(rows, 2048);
  hipLaunchKernelGGL(rmsnorm_fwd_bf16, dim3(grid), dim3(256), 0, cur_stream(),
                     (const unsigned short*)x.data_ptr(),
                     (const unsigned short*)w.data_ptr(),
                     (unsigned short*)out.data_ptr(), rrms.data_ptr<float>(),
                     rows, hidden, (float)eps);
  return {out, rrms};
}

std::vector<at::Tensor> rmsnorm_bwd(at::Tensor dy, at::Tensor x, at::Tensor w,
                                    at::Tensor rrms,
                                    c10::optional<at::Tensor> dres) {
  check_bf16_contig(dy, "dy");
  check_bf16_contig(x, "x");
  int hidden = (int)x.size(-1);
  TORCH_CHECK(hidden <= 8192, "rmsnorm_bwd supports hidden <= 8192");
  int64_t rows = x.numel() / hidden;
  auto dx = at::empty_like(x);
  auto dw = at::zeros({hidden}, x.options().dtype(at::kFloat));
  int grid = (int)std::min<int64_t>(rows, 1024);
  const unsigned short* dres_p = nullptr;
  at::Tensor dres_c;
  if (dres.has_value()) {
    dres_c = dres->contiguous();
    dres_p = (const unsigned short*)dres_c.data_ptr();
  }
  hipLaunchKernelGGL(rmsnorm_bwd_bf16, dim3(grid), dim3(256), 0, cur_stream(),
                     (const unsigned short*)dy.data_ptr(),
                     (const unsigned short*)x.data_ptr(),
                     (const unsigned short*)w.data_ptr(),
                     rrms.data_ptr<float>(), dres_p,
                     (unsigned short*)dx.data_ptr(),
                     dw.data_ptr<float>(), rows, hidden);
  return {dx, dw};
}

std::vector<at::Tensor> rmsnorm_res_fwd(at::Tensor x,
                                        c10::optional<at::Tensor> res,
                                        at::Tensor w, double eps) {
  // fused residual add + rmsnorm: returns {norm_out, res_new, rrms}
  check_bf16_contig(x, "x");
  check_bf16_contig(w, "w");
  int hidden = (int)x.size(-1);
  TORCH_CHECK(hidden % 8 == 0, "hidden must be a multiple of 8");
  int64_t rows = x.numel() / hidden;
  auto out = at::empty_like(x);
  auto res_out = at::empty_like(x);
  auto rrms = at::empty({rows}, x.options().dtype(at::kFloat));
  const unsigned short* res_p = nullptr;
  at::Tensor res_c;
  if (res.has_value()) {
    res_c = res->contiguous();
    res_p = (const unsigned short*)res_c.data_ptr();
  }
  int grid = (int)std::min<int64_t>(rows, 2048);
  hipLaunchKernelGGL(rmsnorm_res_fwd_bf16, dim3(grid), dim3(256), 0,
                     cur_stream(), (const unsigned short*)x.data_ptr(), res_p,
                     (const unsigned short*)w.data_ptr(),
                     (unsigned short*)out.data_ptr(),
                     (unsigned short*)res_out.data_ptr(),
                     rrms.data_ptr<float>(), rows, hidden, (float)eps);
  return {out, res_out, rrms};
}

// ------------------------------ RoPE ------------------------------
at::Tensor rope(at::Tensor x, at::Tensor table, int64_t pos_offset,
                bool backward, bool inplace) {
  // x: [B, S, H, D] bf16, table: [S_max, D/2, 2] f32
  check_bf16_contig(x, "x");
  TORCH_CHECK(table.scalar_type() == at::kFloat && table.is_contiguous());
  TORCH_CHECK(x.dim() == 4, "rope expects [B,S,H,D]");
  int B = (int)x.size(0), S = (int)x.size(1), H = (int)x.size(2), D = (int)x.size(3);
  TORCH_CHECK(D % 2 == 0);
  auto out = inplace ? x : at::empty_like(x);
  int64_t n_tokens = (int64_t)B * S * H;
  int64_t pairs = n_tokens * (D / 2);
  int grid = grid_for(pairs / 2, 256);
  hipLaunchKernelGGL(rope_fwd_bf16, dim3(grid), dim3(256), 0, cur_stream(),
                     (const unsigned short*)x.data_ptr(),
                     (unsigned short*)out.data_ptr(), table.data_ptr<float>(),
                     n_tokens, H, S, D, (int)pos_offset, backward ? 1 : 0);
  return out;
}

std::vector<at::Tensor> rope_qkv_fwd(at::Tensor qkv, at::Tensor table,
                                     int64_t B, int64_t S, int64_t Hq,
                                     int64_t Hkv, int64_t D,
                                     int64_t pos_offset) {
  check_bf16_contig(qkv, "qkv");
  int64_t T = B * S;
  auto opt = qkv.options();
  auto q = at::empty({B, Hq, S, D}, opt);
  auto k = at::empty({B, Hkv, S, D}, opt);
  auto v = at::empty({B, Hkv, S, D}, opt);
  int64_t total = T * (Hq + 2 * Hkv) * (D / 2);
  int grid = grid_for(total, 256);
  hipLaunchKernelGGL(rope_qkv_fwd_bf16, dim3(grid), dim3(256), 0,
                     cur_stream(), (const unsigned short*)qkv.data_ptr(),
                     (unsigned short*)q.data_ptr(),
                     (unsigned short*)k.data_ptr(),
                     (unsigned short*)v.data_ptr(), table.data_ptr<float>(),
                     T, (int)S, (int)Hq, (int)Hkv, (int)D, (int)pos_offset);
  return {q, k, v};
}

at::Tensor rope_qkv_bwd(at::Tensor dq, at::Tensor dk, at::Tensor dv,
                        at::Tensor table, int64_t B, int64_t S, int64_t Hq,
                        int64_t Hkv, int64_t D, int64_t pos_offset) {
  int64_t T = B * S;
  auto dqkv = at::empty({B, S, (Hq + 2 * Hkv) * D}, dq.options());
  int64_t total = T * (Hq + 2 * Hkv) * (D / 2);
  int grid = grid_for(total, 256);
  hipLaunchKernelGGL(rope_qkv_bwd_bf16, dim3(grid), dim3(256), 0,
                     cur_stream(),
                     (const unsigned short*)dq.contiguous().data_ptr(),
                     (const unsigned short*)dk.contiguous().data_ptr(),
                     (const unsigned short*)dv.contiguous().data_ptr(),
                     (unsigned short*)dqkv.data_ptr(), table.data_ptr<float>(),
                     T, (int)S, (int)Hq, (int)Hkv, (int)D, (int)pos_offset);
  return dqkv;
}

// ------------------------------ SwiGLU ------------------------------
at::Tensor swiglu_fwd(at::Tensor gate, at::Tensor up) {
  check_bf16_contig(gate, "gate");
  check_bf16_contig(up, "up");
  auto out = at::empty_like(gate);
  int64_t n = gate.numel();
  int grid = grid_for(n / 8, 256);
  hipLaunchKernelGGL(swiglu_fwd_bf16, dim3(grid), dim3(256), 0, cur_stream(),
                     (const unsigned short*)gate.data_ptr(),
                     (const unsigned short*)up.data_ptr(),
                     (unsigned short*)out.data_ptr(), n);
  return out;
}

at::Tensor swiglu_packed_fwd(at::Tensor gu) {
  check_bf16_contig(gu, "gu");
  int F2 = (int)gu.size(-1);
  TORCH_CHECK(F2 % 16 == 0, "packed ffn dim must be multiple of 16");
  int F = F2 / 2;
  int64_t rows = gu.numel() / F2;
  auto sizes = gu.sizes().vec();
  sizes.back() = F;
  auto out = at::empty(sizes, gu.options());
  int grid = grid_for(rows * (F / 8), 256);
  hipLaunchKernelGGL(swiglu_packed_fwd_bf16, dim3(grid), dim3(256), 0,
                     cur_stream(), (const unsigned short*)gu.data_ptr(),
                     (unsigned short*)out.data_ptr(), rows, F);
  return out;
}

at::Tensor swiglu_packed_bwd(at::Tensor dy, at::Tensor gu) {
  int F2 = (int)gu.size(-1);
  int F = F2 / 2;
  int64_t rows = gu.numel() / F2;
  auto dgu = at::empty_like(gu);
  int grid = grid_for(rows * (F / 8), 256);
  hipLaunchKernelGGL(swiglu_packed_bwd_bf16, dim3(grid), dim3(256), 0,
                     cur_stream(), (const unsigned short*)dy.contiguous().data_ptr(),
                     (const unsigned short*)gu.data_ptr(),
                     (unsigned short*)dgu.data_ptr(), rows, F);
  return dgu;
}

std::vector<at::Tensor> swiglu_bwd(at::Tensor dy, at::Tensor gate, at::Tensor up) {
  auto dgate = at::empty_like(gate);
  auto dup = at::empty_like(up);
  int64_t n = gate.numel();
  int grid = grid_for(n / 8, 256);
  hipLaunchKernelGGL(swiglu_bwd_bf16, dim3(grid), dim3(256), 0, cur_stream(),
                     (const unsigned short*)dy.data_ptr(),
                     (const unsigned short*)gate.data_ptr(),
                     (const unsigned short*)up.data_ptr(),
                     (unsigned short*)dgate.data_ptr(),
                     (unsigned short*)dup.data_ptr(), n);
  return {dgate, dup};
}

// ------------------------------ Cross entropy ------------------------------
std::vector<at::Tensor> ce_fwd(at::Tensor logits, at::Tensor target,
                               int64_t ignore_index) {
  check_bf16_contig(logits, "logits");
  TORCH_CHECK(target.scalar_type() == at::kLong && target.is_contiguous());
  int vocab = (int)logits.size(-1);
  TORCH_CHECK(vocab % 8 == 0, "vocab must be a multiple of 8");
  int64_t rows = logits.numel() / vocab;
  auto loss = at::empty({rows}, logits.options().dtype(at::kFloat));
  auto lse = at::empty({rows}, logits.options().dtype(at::kFloat));
  int grid = (int)std::min<int64_t>(rows, 2048);
  hipLaunchKernelGGL(ce_fwd_bf16, dim3(grid), dim3(256), 0, cur_stream(),
                     (const unsigned short*)logits.data_ptr(),
                     target.data_ptr<int64_t>(), loss.data_ptr<float>(),
                     lse.data_ptr<float>(), rows, vocab, ignore_index);
  return {loss, lse};
}

at::Tensor ce_bwd(at::Tensor logits, at::Tensor target, at::Tensor lse,
                  at::Tensor dloss, int64_t ignore_index, bool inplace) {
  int vocab = (int)logits.size(-1);
  int64_t rows = logits.numel() / vocab;
  auto dlogits = inplace ? logits : at::empty_like(logits);
  int grid = (int)std::min<int64_t>(rows, 2048);
  hipLaunchKernelGGL(ce_bwd_bf16, dim3(grid), dim3(256), 0, cur_stream(),
                     (const unsigned short*)logits.data_ptr(),
                     (unsigned short*)dlogits.data_ptr(),
                     target.data_ptr<int64_t>(), lse.data_ptr<float>(),
                     dloss.data_ptr<float>(), rows, vocab, ignore_index);
  return dlogits;
}

// ------------------------------ AdamW ------------------------------
void adamw_step(at::Tensor param, c10::optional<at::Tensor> master,
                at::Tensor grad, at::Tensor m, at::Tensor v, double lr,
                double beta1, double beta2, double eps, double weight_decay,
                int64_t step, double grad_scale,
                c10::optional<at::Tensor> clip_scale) {
  check_bf16_contig(param, "param");
  int64_t n = param.numel();
  float bc1 = 1.f - powf((float)beta1, (float)step);
  float bc2 = 1.f - powf((float)beta2, (float)step);
  const unsigned short* gb = nullptr;
  const float* gf = nullptr;
  if (grad.scalar_type() == at::kBFloat16) gb = (const unsigned short*)grad.data_ptr();
  else gf = grad.data_ptr<float>();
  float* mp = master.has_value() ? master->data_ptr<float>() : nullptr;
  const float* cp = clip_scale.has_value() ? clip_scale->data_ptr<float>() : nullptr;
  int grid = grid_for(n / 8, 256);
  hipLaunchKernelGGL(adamw_flat_bf16, dim3(grid), dim3(256), 0, cur_stream(),
                     (unsigned short*)param.data_ptr(), mp, gb, gf,
                     m.data_ptr<float>(), v.data_ptr<float>(), n, (float)lr,
                     (float)beta1, (float)beta2, (float)eps,
                     (float)weight_decay, bc1, bc2, (float)grad_scale, cp);
}

at::Tensor l2norm_sq(at::Tensor x) {
  auto out = at::zeros({1}, x.options().dtype(at::kFloat));
  int64_t n = x.numel();
  const unsigned short* xb = nullptr;
  const float* xf = nullptr;
  if (x.scalar_type() == at::kBFloat16) xb = (const unsigned short*)x.data_ptr();
  else xf = x.data_ptr<float>();
  int grid = grid_for(n / 8, 256);
  hipLaunchKernelGGL(l2norm_sq_flat, dim3(grid), dim3(256), 0, cur_stream(),
                     xb, xf, out.data_ptr<float>(), n);
  return out;
}

void scale_(at::Tensor x, c10::optional<at::Tensor> scale_t, double scale_c) {
  int64_t n = x.numel();
  unsigned short* xb = nullptr;
  float* xf = nullptr;
  if (x.scalar_type() == at::kBFloat16) xb = (unsigned short*)x.data_ptr();
  else xf = x.data_ptr<float>();
  const float* sp = scale_t.has_value() ? scale_t->data_ptr<float>() : nullptr;
  int grid = grid_for(n / 8, 256);
  hipLaunchKernelGGL(scale_flat, dim3(grid), dim3(256), 0, cur_stream(), xb,
                     xf, sp, (float)scale_c, n);
}

// ------------------------------ MFMA GEMM ------------------------------
at::Tensor gemm_tn(at::Tensor a, at::Tensor b, int64_t variant) {
  // C[M,N] = a[M,K] @ b[N,K]^T
  check_bf16_contig(a, "a");
  check_bf16_contig(b, "b");
  int64_t M = a.size(0), K = a.size(1), N = b.size(0);
  TORCH_CHECK(b.size(1) == K, "K mismatch");
  int BM = variant == 1 ? 128 : (variant == 3 ? 128 : (variant == 4 ? 512 : 256));
  int BN = variant == 3 ? 128 : 256;
  int BK = variant == 1 ? 64 : 32;
  int threads = variant == 3 ? 256 : (variant == 4 ? 1024 : 512);
  TORCH_CHECK(M % BM == 0 && N % BN == 0 && K % BK == 0,
              "gemm_tn tile divisibility violated");
  auto c = at::empty({M, N}, a.options());
  int grid = (int)((M / BM) * (N / BN));
  auto kern = gemm_tn_bf16_v0;
  if (variant == 1) kern = gemm_tn_bf16_v1;
  if (variant == 2) kern = gemm_tn_bf16_v2;
  if (variant == 3) kern = gemm_tn_bf16_v3;
  if (variant == 4) kern = gemm_tn_bf16_v4;
  hipLaunchKernelGGL(kern, dim3(grid), dim3(threads), 0, cur_stream(),
                     (const unsigned short*)a.data_ptr(),
                     (const unsigned short*)b.data_ptr(),
                     (unsigned short*)c.data_ptr(), (int)M, (int)N, (int)K);
  return c;
}

at::Tensor gemm_tn8(at::Tensor a, at::Tensor b, int64_t mode) {
  // C[M,N] = A[M,K] . B[N,K]^T via the 8-phase 256x256 template (gemm8.hip)
  check_bf16_contig(a, "a");
  check_bf16_contig(b, "b");
  int64_t M = a.size(0), K = a.size(1), N = b.size(0);
  TORCH_CHECK(b.size(1) == K, "K mismatch");
  TORCH_CHECK(M % 256 == 0 && N % 256 == 0 && K % 64 == 0,
              "gemm_tn8 requires M,N % 256 == 0 and K % 64 == 0");
  auto c = at::empty({M, N}, a.options());
  dim3 grid((unsigned)(M / 256), (unsigned)(N / 256), 1);
  TORCH_CHECK(mode != 5 || (K / 64) % 2 == 0, "mode 5 needs K % 128 == 0");
  TORCH_CHECK((mode != 7 && mode != 8) || ((K / 64) % 2 == 0 && K / 64 >= 4),
              "modes 7/8 need an even K/64 >= 4");
  auto kern = mode == 8   ? gemm8_tn_bf16_rot9
              : mode == 7 ? gemm8_tn_bf16_rot8
              : mode == 6 ? gemm8_tn_bf16_rot3ra
              : mode == 5 ? gemm8_tn_bf16_rot3u2
              : mode == 4 ? gemm8_tn_bf16_rot3np
              : mode == 3 ? gemm8_tn_bf16_rot3
              : mode == 2 ? gemm8_tn_bf16_rot
              : mode == 1 ? gemm8_tn_bf16_lockstep
                          : gemm8_tn_bf16;
  hipLaunchKernelGGL(kern, grid, dim3(G8_THREADS), 0, cur_stream(),
                     (const unsigned short*)a.data_ptr(),
                     (const unsigned short*)b.data_ptr(),
                     (unsigned short*)c.data_ptr(), (int)M, (int)N, (int)K);
  return c;
}

// ------------------------------ flash-attention backward -------------------
std::vector<at::Tensor> fa_bwd(at::Tensor q, at::Tensor k, at::Tensor v,
                               at::Tensor o, at::Tensor dout, at::Tensor lse,
                               double scale) {
  // q,o,dout: [B,Hq,S,D]; k,v: [B,Hkv,S,D]; lse: [B,Hq,S] fp32; causal
  check_bf16_contig(q, "q");
  check_bf16_contig(k, "k");
  check_bf16_contig(v, "v");
  check_bf16_contig(o, "o");
  int B = (int)q.size(0), Hq = (int)q.size(1), S = (int)q.size(2), D = (int)q.size(3);
  int Hkv = (int)k.size(1);
  TORCH_CHECK(D == FA_D, "fa_bwd supports head_dim 128");
  TORCH_CHECK(S % FA_BLK == 0, "seq must be a multiple of 64");
  TORCH_CHECK(lse.scalar_type() == at::kFloat);
  auto lse_c = lse.contiguous();
  auto dout_c = dout.contiguous();
  auto delta = at::empty({B, Hq, S}, q.options().dtype(at::kFloat));
  int64_t rows = (int64_t)B * Hq * S;
  hipLaunchKernelGGL(fa_bwd_preprocess, dim3(grid_for(rows, 1, 4096)), dim3(256),
                     0, cur_stream(),
                     (const unsigned short*)dout_c.data_ptr(),
                     (const unsigned short*)o.data_ptr(),
                     delta.data_ptr<float>(), rows);
  auto dq = at::empty_like(q);
  auto dk_part = at::empty({B, Hq, S, D}, q.options());
  auto dv_part = at::empty({B, Hq, S, D}, q.options());
  dim3 gridkv(S / FA_BLK, Hq, B);
  hipLaunchKernelGGL(fa_bwd_dkdv, gridkv, dim3(FA_THREADS), 0, cur_stream(),
                     (const unsigned short*)q.data_ptr(),
                     (const unsigned short*)k.data_ptr(),
                     (const unsigned short*)v.data_ptr(),
                     (const unsigned short*)dout_c.data_ptr(),
                     lse_c.data_ptr<float>(), delta.data_ptr<float>(),
                     (unsigned short*)dk_part.data_ptr(),
                     (unsigned short*)dv_part.data_ptr(), B, Hq, Hkv, S,
                     (float)scale);
  hipLaunchKernelGGL(fa_bwd_dq, gridkv, dim3(FA_THREADS), 0, cur_stream(),
                     (const unsigned short*)q.data_ptr(),
                     (const unsigned short*)k.data_ptr(),
                     (const unsigned short*)v.data_ptr(),
                     (const unsigned short*)dout_c.data_ptr(),
                     lse_c.data_ptr<float>(), delta.data_ptr<float>(),
                     (unsigned short*)dq.data_ptr(), B, Hq, Hkv, S,
                     (float)scale);
  at::Tensor dk, dv;
  if (Hkv == Hq) {
    dk = dk_part;
    dv = dv_part;
  } else {
    dk = at::empty({B, Hkv, S, D}, q.options());
    dv = at::empty({B, Hkv, S, D}, q.options());
    int64_t SD = (int64_t)S * D;
    int grid = grid_for((int64_t)B * Hkv * SD / 8, 256);
    hipLaunchKernelGGL(fa_bwd_reduce_gqa, dim3(grid), dim3(256), 0,
                       cur_stream(),
                       (const unsigned short*)dk_part.data_ptr(),
                       (unsigned short*)dk.data_ptr(), B, Hq, Hkv, SD);
    hipLaunchKernelGGL(fa_bwd_reduce_gqa, dim3(grid), dim3(256), 0,
                       cur_stream(),
                       (const unsigned short*)dv_part.data_ptr(),
                       (unsigned short*)dv.data_ptr(), B, Hq, Hkv, SD);
  }
  return {dq, dk, dv};
}

// ------------------------------ flash-attention forward --------------------
extern "C" __global__ void permlane_probe_kernel(int* out) {
  int x = (int)threadIdx.x;
  auto r = __builtin_amdgcn_permlane32_swap(x, x, false, false);
  out[threadIdx.x] = (r[0] << 16) | (r[1] & 0xFFFF);
}

// mfma_f32_32x32x16_bf16 layout probe: builds A/B frags per the ASSUMED
// layout (operand row/col = lane&31, k = (lane>>5)*8 + e) and dumps D for
// three tests; see tools/fa_fwd_check.py for decoding.
extern "C" __global__ void mfma32_probe_kernel(float* out) {
  typedef short sx8 __attribute__((ext_vector_type(8)));
  typedef float fx16 __attribute__((ext_vector_type(16)));
  int lane = threadIdx.x & 63;
  int l31 = lane & 31, half = lane >> 5;
  sx8 af, bf, a1, bj, ak, bk;
  for (int e = 0; e < 8; ++e) {
    int k = half * 8 + e;
    af[e] = (short)f32_to_bf16((float)((k == 0) ? l31 : 0));
    bf[e] = (short)f32_to_bf16((float)((k == 0) ? 1 : 0));
    a1[e] = bf[e];
    bj[e] = (short)f32_to_bf16((float)((k == 0) ? l31 : 0));
    ak[e] = (short)f32_to_bf16((float)((k == 5) ? 1 : 0));
    bk[e] = (short)f32_to_bf16((float)((k == 5) ? l31 : 0));
  }
  fx16 z = (fx16)(0.f);
  fx16 d1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, z, 0, 0, 0);
  fx16 d2 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, bj, z, 0, 0, 0);
  fx16 d3 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ak, bk, z, 0, 0, 0);
  for (int r = 0; r < 16; ++r) {
    out[0 * 64 * 16 + lane * 16 + r] = d1[r];
    out[1 * 64 * 16 + lane * 16 + r] = d2[r];
    out[2 * 64 * 16 + lane * 16 + r] = d3[r];
  }
}

at::Tensor mfma32_probe() {
  auto out = at::empty({3, 64, 16}, at::TensorOptions().dtype(at::kFloat).device(at::kCUDA));
  hipLaunchKernelGGL(mfma32_probe_kernel, dim3(1), dim3(64), 0, cur_stream(),
                     out.data_ptr<float>());
  return out;
}

// tr-read addressing probe for the 4x4-group transpose scheme: stage a
// pattern (value = kv*128 + d) through the SAME swizzled store the kernel
// uses, then perform the kernel's exact PV reads; host checks lane l elem
// e == (ks*16 + half*8 + e)*128 + ds*32 + l31 (tools/tr_probe_check.py).
extern "C" __global__ void tr_probe_kernel(unsigned short* out) {
  __shared__ unsigned short lvp[FF_KV * FF_D];
  int tid = threadIdx.x;
  for (int i = tid * 8; i < FF_KV * FF_D; i += 64 * 8) {
    ff_shortx8 vv;
#pragma unroll
    for (int x = 0; x < 8; ++x) vv[x] = (short)(i + x);
    *reinterpret_cast<ff_shortx8*>((char*)lvp + ff_kswz(i * 2)) = vv;
  }
  __syncthreads();
  int lane = tid & 63;
  int l31 = lane & 31;
  int half = lane >> 5;
  for (int ds = 0; ds < 4; ++ds) {
    const int col2 = (ds * 32 + (l31 & 16) + (lane & 3) * 4) * 2;
    const int row0 = half * 8 + ((lane & 15) >> 2);
    const int a0 = (row0 * 256 + col2) ^ ((row0 & 7) << 4);
    const int row1 = row0 + 4;
    const int a1 = (row1 * 256 + col2) ^ ((row1 & 7) << 4);
    ff_lds_p b0 = (ff_lds_p)((const char*)lvp + a0);
    ff_lds_p b1 = (ff_lds_p)((const char*)lvp + a1);
    ff_shortx4 t0, t1;
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      switch (ks) {
        case 0: asm volatile("ds_read_b64_tr_b16 %0, %2 offset:0\n\tds_read_b64_tr_b16 %1, %3 offset:0" : "=v"(t0), "=v"(t1) : "v"(b0), "v"(b1)); break;
        case 1: asm volatile("ds_read_b64_tr_b16 %0, %2 offset:4096\n\tds_read_b64_tr_b16 %1, %3 offset:4096" : "=v"(t0), "=v"(t1) : "v"(b0), "v"(b1)); break;
        case 2: asm volatile("ds_read_b64_tr_b16 %0, %2 offset:8192\n\tds_read_b64_tr_b16 %1, %3 offset:8192" : "=v"(t0), "=v"(t1) : "v"(b0), "v"(b1)); break;
        default: asm volatile("ds_read_b64_tr_b16 %0, %2 offset:12288\n\tds_read_b64_tr_b16 %1, %3 offset:12288" : "=v"(t0), "=v"(t1) : "v"(b0), "v"(b1)); break;
      }
      asm volatile("s_waitcnt lgkmcnt(0)" : "+v"(t0), "+v"(t1));
      unsigned short* o = out + ((ds * 4 + ks) * 64 + lane) * 8;
#pragma unroll
      for (int x = 0; x < 4; ++x) { o[x] = (unsigned short)t0[x]; o[4 + x] = (unsigned short)t1[x]; }
    }
  }
}

at::Tensor tr_probe() {
  auto out = at::empty({4, 4, 64, 8},
                       at::TensorOptions().dtype(at::kShort).device(at::kCUDA));
  hipLaunchKernelGGL(tr_probe_kernel, dim3(1), dim3(64), 0, cur_stream(),
                     (unsigned short*)out.data_ptr());
  return out;
}

at::Tensor permlane_probe() {
  auto out = at::empty({64}, at::TensorOptions().dtype(at::kInt).device(at::kCUDA));
  hipLaunchKernelGGL(permlane_probe_kernel, dim3(1), dim3(64), 0, cur_stream(),
                     out.data_ptr<int>());
  return out;
}

std::vector<at::Tensor> fa_fwd(at::Tensor q, at::Tensor k, at::Tensor v,
                               double scale) {
  // q: [B,Hq,S,D]; k,v: [B,Hkv,S,D]; causal, D=128, S % 256 == 0
  check_bf16_contig(q, "q");
  check_bf16_contig(k, "k");
  check_bf16_contig(v, "v");
  int B = (int)q.size(0), Hq = (int)q.size(1), S = (int)q.size(2), D = (int)q.size(3);
  int Hkv = (int)k.size(1);
  TORCH_CHECK(D == FF_D, "fa_fwd supports head_dim 128");
  TORCH_CHECK(S % FF_QTILE == 0, "seq must be a multiple of 256");
  TORCH_CHECK(Hq % Hkv == 0, "Hq must be a multiple of Hkv");
  auto out = at::empty_like(q);
  auto lse = at::empty({B, Hq, S}, q.options().dtype(at::kFloat));
  dim3 grid(S / FF_QTILE, Hq, B);
  hipLaunchKernelGGL(fa_fwd_bf16, grid, dim3(FF_THREADS), 0, cur_stream(),
                     (const unsigned short*)q.data_ptr(),
                     (const unsigned short*)k.data_ptr(),
                     (const unsigned short*)v.data_ptr(),
                     (unsigned short*)out.data_ptr(), lse.data_ptr<float>(),
                     B, Hq, Hkv, S, (float)scale);
  return {out, lse};
}

std::vector<at::Tensor> fa_fwd_ablate(at::Tensor q, at::Tensor k, at::Tensor v,
                                      double scale, int64_t mode) {
  // ablation-timing variants of fa_fwd (NOT numerically meaningful for
  // mode != 0); see attention_fwd.hip template MODE docs
  int B = (int)q.size(0), Hq = (int)q.size(1), S = (int)q.size(2);
  int Hkv = (int)k.size(1);
  auto out = at::empty_like(q);
  auto lse = at::empty({B, Hq, S}, q.options().dtype(at::kFloat));
  dim3 grid(S / FF_QTILE, Hq, B);
  auto kern = mode == 1 ? fa_fwd_bf16_ab1 : mode == 2 ? fa_fwd_bf16_ab2
              : mode == 3 ? fa_fwd_bf16_ab3 : mode == 4 ? fa_fwd_bf16_ab4
              : mode == 5 ? fa_fwd_bf16_ab5 : fa_fwd_bf16;
  hipLaunchKernelGGL(kern, grid, dim3(FF_THREADS), 0, cur_stream(),
                     (const unsigned short*)q.data_ptr(),
                     (const unsigned short*)k.data_ptr(),
                     (const unsigned short*)v.data_ptr(),
                     (unsigned short*)out.data_ptr(), lse.data_ptr<float>(),
                     B, Hq, Hkv, S, (float)scale);
  return {out, lse};
}

// ------------------------------ flash-attention backward v2 ---------------
std::vector<at::Tensor> fa_bwd2(at::Tensor q, at::Tensor k, at::Tensor v,
                                at::Tensor o, at::Tensor dout, at::Tensor lse,
                                double scale, bool fused_dvdk) {
  // 32x32-MFMA backward (attention_bwd2.hip): preprocess + dq + dv + dk
  // (+ GQA reduction). q,o,dout: [B,Hq,S,D]; k,v: [B,Hkv,S,D]; lse ln-dom.
  check_bf16_contig(q, "q");
  check_bf16_contig(k, "k");
  check_bf16_contig(v, "v");
  check_bf16_contig(o, "o");
  int B = (int)q.size(0), Hq = (int)q.size(1), S = (int)q.size(2), D = (int)q.size(3);
  int Hkv = (int)k.size(1);
  TORCH_CHECK(D == FB_D, "fa_bwd2 supports head_dim 128");
  TORCH_CHECK(S % FB_TILE == 0, "seq must be a multiple of 128");
  TORCH_CHECK(lse.scalar_type() == at::kFloat);
  auto lse_c = lse.contiguous();
  auto dout_c = dout.contiguous();
  auto delta = at::empty({B, Hq, S}, q.options().dtype(at::kFloat));
  int64_t rows = (int64_t)B * Hq * S;
  hipLaunchKernelGGL(fa_bwd_preprocess, dim3(grid_for(rows, 1, 4096)), dim3(256),
                     0, cur_stream(),
                     (const unsigned short*)dout_c.data_ptr(),
                     (const unsigned short*)o.data_ptr(),
                     delta.data_ptr<float>(), rows);
  auto dq = at::empty_like(q);
  dim3 grid(S / FB_TILE, Hq, B);
  hipLaunchKernelGGL(fa2_dq_bf16, grid, dim3(FB_THREADS), 0, cur_stream(),
                     (const unsigned short*)q.data_ptr(),
                     (const unsigned short*)k.data_ptr(),
                     (const unsigned short*)v.data_ptr(),
                     (const unsigned short*)dout_c.data_ptr(),
                     lse_c.data_ptr<float>(), delta.data_ptr<float>(),
                     (unsigned short*)dq.data_ptr(), B, Hq, Hkv, S,
                     (float)scale);
  auto dv_part = at::empty({B, Hq, S, D}, q.options());
  auto dk_part = at::empty({B, Hq, S, D}, q.options());
  if (fused_dvdk) {
    // 4-wave blocks over 128-row kv tiles (see kernel comment): 2x the
    // blocks of the split kernels' 256-row tiles.
    dim3 gridf(S / FD_TILE, Hq, B);
    hipLaunchKernelGGL(fa2_dvdk_bf16, gridf, dim3(FD_THREADS), 0, cur_stream(),
                       (const unsigned short*)q.data_ptr(),
                       (const unsigned short*)k.data_ptr(),
                       (const unsigned short*)v.data_ptr(),
                       (const unsigned short*)dout_c.data_ptr(),
                       lse_c.data_ptr<float>(), delta.data_ptr<float>(),
                       (unsigned short*)dv_part.data_ptr(),
                       (unsigned short*)dk_part.data_ptr(), B, Hq, Hkv, S,
                       (float)scale);
  } else {
    hipLaunchKernelGGL(fa2_dv_bf16, grid, dim3(FB_THREADS), 0, cur_stream(),
                       (const unsigned short*)q.data_ptr(),
                       (const unsigned short*)k.data_ptr(),
                       (const unsigned short*)dout_c.data_ptr(),
                       lse_c.data_ptr<float>(),
                       (unsigned short*)dv_part.data_ptr(), B, Hq, Hkv, S,
                       (float)scale);
    hipLaunchKernelGGL(fa2_dk_bf16, grid, dim3(FB_THREADS), 0, cur_stream(),
                       (const unsigned short*)q.data_ptr(),
                       (const unsigned short*)k.data_ptr(),
                       (const unsigned short*)v.data_ptr(),
                       (const unsigned short*)dout_c.data_ptr(),
                       lse_c.data_ptr<float>(), delta.data_ptr<float>(),
                       (unsigned short*)dk_part.data_ptr(), B, Hq, Hkv, S,
                       (float)scale);
  }
  at::Tensor dk, dv;
  if (Hkv == Hq) {
    dk = dk_part;
    dv = dv_part;
  } else {
    dk = at::empty({B, Hkv, S, D}, q.options());
    dv = at::empty({B, Hkv, S, D}, q.options());
    int64_t SD = (int64_t)S * D;
    int rgrid = grid_for((int64_t)B * Hkv * SD / 8, 256);
    hipLaunchKernelGGL(fa_bwd_reduce_gqa, dim3(rgrid), dim3(256), 0,
                       cur_stream(),
                       (const unsigned short*)dk_part.data_ptr(),
                       (unsigned short*)dk.data_ptr(), B, Hq, Hkv, SD);
    hipLaunchKernelGGL(fa_bwd_reduce_gqa, dim3(rgrid), dim3(256), 0,
                       cur_stream(),
                       (const unsigned short*)dv_part.data_ptr(),
                       (unsigned short*)dv.data_ptr(), B, Hq, Hkv, SD);
  }
  return {dq, dk, dv};
}

// ------------------------------ philox random ------------------------------
ShardDesc make_desc(const std::vector<int64_t>& gshape,
                    const std::vector<int64_t>& lshape,
                    const std::vector<int64_t>& offset, int64_t flat_offset,
                    bool is_flat) {
  ShardDesc d{};
  d.ndim = (int)gshape.size();
  TORCH_CHECK(d.ndim <= MAXD);
  for (int i = 0; i < d.ndim; ++i) {
    d.gshape[i] = gshape[i];
    d.lshape[i] = lshape[i];
    d.offset[i] = offset[i];
  }
  d.flat_offset = flat_offset;
  d.is_flat = is_flat ? 1 : 0;
  return d;
}

void philox_uniform_(at::Tensor out, std::vector<int64_t> gshape,
                     std::vector<int64_t> lshape, std::vector<int64_t> offset,
                     int64_t flat_offset, bool is_flat, int64_t seed,
                     int64_t philox_offset, double lo, double hi) {
  auto d = make_desc(gshape, lshape, offset, flat_offset, is_flat);
  int64_t n = out.numel();
  int grid = grid_for(n, 256);
  if (out.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(philox_uniform_bf16, dim3(grid), dim3(256), 0,
                       cur_stream(), (unsigned short*)out.data_ptr(), d, n,
                       (uint64_t)seed, (uint64_t)philox_offset, (float)lo,
                       (float)hi);
  } else {
    hipLaunchKernelGGL(philox_uniform_f32, dim3(grid), dim3(256), 0,
                       cur_stream(), out.data_ptr<float>(), d, n,
                       (uint64_t)seed, (uint64_t)philox_offset, (float)lo,
                       (float)hi);
  }
}

void philox_normal_(at::Tensor out, std::vector<int64_t> gshape,
                    std::vector<int64_t> lshape, std::vector<int64_t> offset,
                    int64_t flat_offset, bool is_flat, int64_t seed,
                    int64_t philox_offset, double mean, double std) {
  auto d = make_desc(gshape, lshape, offset, flat_offset, is_flat);
  int64_t n = out.numel();
  int grid = grid_for(n, 256);
  if (out.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(philox_normal_bf16, dim3(grid), dim3(256), 0,
                       cur_stream(), (unsigned short*)out.data_ptr(), d, n,
                       (uint64_t)seed, (uint64_t)philox_offset, (float)mean,
                       (float)std);
  } else {
    hipLaunchKernelGGL(philox_normal_f32, dim3(grid), dim3(256), 0,
                       cur_stream(), out.data_ptr<float>(), d, n,
                       (uint64_t)seed, (uint64_t)philox_offset, (float)mean,
                       (float)std);
  }
}

std::vector<at::Tensor> philox_dropout(at::Tensor x, std::vector<int64_t> gshape,
                                       std::vector<int64_t> lshape,
                                       std::vector<int64_t> offset,
                                       int64_t flat_offset, bool is_flat,
                                       int64_t seed, int64_t philox_offset,
                                       double p, bool need_mask) {
  check_bf16_contig(x, "x");
  auto d = make_desc(gshape, lshape, offset, flat_offset, is_flat);
  auto out = at::empty_like(x);
  at::Tensor mask;
  unsigned char* mp = nullptr;
  if (need_mask) {
    mask = at::empty(x.sizes(), x.options().dtype(at::kByte));
    mp = mask.data_ptr<unsigned char>();
  }
  int64_t n = x.numel();
  int grid = grid_for(n, 256);
  hipLaunchKernelGGL(philox_dropout_bf16, dim3(grid), dim3(256), 0,
                     cur_stream(), (const unsigned short*)x.data_ptr(),
                     (unsigned short*)out.data_ptr(), mp, d, n, (uint64_t)seed,
                     (uint64_t)philox_offset, (float)p);
  if (need_mask) return {out, mask};
  return {out};
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm_fwd", &rmsnorm_fwd);
  m.def("rmsnorm_bwd", &rmsnorm_bwd, pybind11::arg("dy"), pybind11::arg("x"),
        pybind11::arg("w"), pybind11::arg("rrms"),
        pybind11::arg("dres") = pybind11::none());
  m.def("rmsnorm_res_fwd", &rmsnorm_res_fwd);
  m.def("rope", &rope);
  m.def("rope_qkv_fwd", &rope_qkv_fwd);
  m.def("rope_qkv_bwd", &rope_qkv_bwd);
  m.def("swiglu_fwd", &swiglu_fwd);
  m.def("swiglu_bwd", &swiglu_bwd);
  m.def("swiglu_packed_fwd", &swiglu_packed_fwd);
  m.def("swiglu_packed_bwd", &swiglu_packed_bwd);
  m.def("ce_fwd", &ce_fwd);
  m.def("ce_bwd", &ce_bwd);
  m.def("adamw_step", &adamw_step);
  m.def("l2norm_sq", &l2norm_sq);
  m.def("scale_", &scale_);
  m.def("gemm_tn", &gemm_tn, pybind11::arg("a"), pybind11::arg("b"), pybind11::arg("variant") = 0);
  m.def("gemm_tn8", &gemm_tn8, pybind11::arg("a"), pybind11::arg("b"), pybind11::arg("mode") = 0);
  m.def("fa_bwd", &fa_bwd);
  m.def("fa_bwd2", &fa_bwd2);
  m.def("fa_fwd", &fa_fwd);
  m.def("fa_fwd_ablate", &fa_fwd_ablate);
  m.def("permlane_probe", &permlane_probe);
  m.def("mfma32_probe", &mfma32_probe);
  m.def("tr_probe", &tr_probe);
  m.def("philox_uniform_", &philox_uniform_);
  m.def("philox_normal_", &philox_normal_);
  m.def("philox_dropout", &philox_dropout);
}
