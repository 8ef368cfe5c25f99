// Fused QKV-split + RoPE — CDNA4.
//
// Consumes the packed output of the fused wqkv GEMM [B, S, (Hq+2*Hkv)*D]
// and emits contiguous q/k/v in [B, H, S, D] layout (the attention
// kernels' native layout — saves three transpose-copies per layer) with
// RoPE applied to q and k — one read + one write instead of (3
// slice-copies + 2 rope passes + 3 transposes).  Backward packs dq/dk/dv
// back (inverse rotation on dq/dk).
#include "common.h"

#define BLOCK 256

// forward: qkv [T, (Hq+2Hkv)*D] -> q [B,Hq,S,D] (roped), k [B,Hkv,S,D]
// (roped), v [B,Hkv,S,D].  T = B*S tokens; table [S, D/2, 2].
extern "C" __global__ void __launch_bounds__(BLOCK)
rope_qkv_fwd_bf16(const unsigned short* __restrict__ qkv,
                  unsigned short* __restrict__ q,
                  unsigned short* __restrict__ k,
                  unsigned short* __restrict__ v,
                  const float* __restrict__ table,
                  int64_t n_tokens, int seq, int Hq, int Hkv, int D,
                  int pos_offset) {
  const int rot = D / 2;
  const int row_in = (Hq + 2 * Hkv) * D;
  // one (token, head) pair handles D/2 rotation pairs; v heads copied raw.
  const int heads_total = Hq + 2 * Hkv;
  int64_t total = n_tokens * heads_total * rot;
  int64_t i0 = blockIdx.x * BLOCK + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * BLOCK;
  for (int64_t t = i0; t < total; t += stride) {
    int d = (int)(t % rot);
    int64_t th = t / rot;
    int h = (int)(th % heads_total);
    int64_t tok = th / heads_total;
    int64_t s = tok % seq;
    int64_t b = tok / seq;
    const unsigned short* src = qkv + tok * row_in + (int64_t)h * D;
    float x0 = bf16_to_f32(src[d]);
    float x1 = bf16_to_f32(src[d + rot]);
    unsigned short* dst;
    bool do_rope = true;
    if (h < Hq) {
      dst = q + (((b * Hq + h) * seq) + s) * D;
    } else if (h < Hq + Hkv) {
      dst = k + (((b * Hkv + (h - Hq)) * seq) + s) * D;
    } else {
      dst = v + (((b * Hkv + (h - Hq - Hkv)) * seq) + s) * D;
      do_rope = false;
    }
    if (do_rope) {
      const float* tb = table + ((s + pos_offset) * (int64_t)rot + d) * 2;
      float c = tb[0], sn = tb[1];
      dst[d] = f32_to_bf16(x0 * c - x1 * sn);
      dst[d + rot] = f32_to_bf16(x1 * c + x0 * sn);
    } else {
      dst[d] = f32_to_bf16(x0);
      dst[d + rot] = f32_to_bf16(x1);
    }
  }
}

// backward: dq/dk/dv -> dqkv packed; inverse rotation (negated sin) on dq,dk
extern "C" __global__ void __launch_bounds__(BLOCK)
rope_qkv_bwd_bf16(const unsigned short* __restrict__ dq,
                  const unsigned short* __restrict__ dk,
                  const unsigned short* __restrict__ dv,
                  unsigned short* __restrict__ dqkv,
                  const float* __restrict__ table,
                  int64_t n_tokens, int seq, int Hq, int Hkv, int D,
                  int pos_offset) {
  const int rot = D / 2;
  const int row_out = (Hq + 2 * Hkv) * D;
  const int heads_total = Hq + 2 * Hkv;
  int64_t total = n_tokens * heads_total * rot;
  int64_t i0 = blockIdx.x * BLOCK + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * BLOCK;
  for (int64_t t = i0; t < total; t += stride) {
    int d = (int)(t % rot);
    int64_t th = t / rot;
    int h = (int)(th % heads_total);
    int64_t tok = th / heads_total;
    int64_t s = tok % seq;
    int64_t b = tok / seq;
    const unsigned short* src;
    bool do_rope = true;
    if (h < Hq) {
      src = dq + (((b * Hq + h) * seq) + s) * D;
    } else if (h < Hq + Hkv) {
      src = dk + (((b * Hkv + (h - Hq)) * seq) + s) * D;
    } else {
      src = dv + (((b * Hkv + (h - Hq - Hkv)) * seq) + s) * D;
      do_rope = false;
    }
    float x0 = bf16_to_f32(src[d]);
    float x1 = bf16_to_f32(src[d + rot]);
    unsigned short* dst = dqkv + tok * row_out + (int64_t)h * D;
    if (do_rope) {
      const float* tb = table + ((s + pos_offset) * (int64_t)rot + d) * 2;
      float c = tb[0], sn = -tb[1];
      dst[d] = f32_to_bf16(x0 * c - x1 * sn);
      dst[d + rot] = f32_to_bf16(x1 * c + x0 * sn);
    } else {
      dst[d] = f32_to_bf16(x0);
      dst[d + rot] = f32_to_bf16(x1);
    }
  }
}
