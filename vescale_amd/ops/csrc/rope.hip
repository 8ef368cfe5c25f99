// RoPE (rotary position embedding), NeoX rotate-half layout — CDNA4.
//
// Memory-bound elementwise; cos/sin table PRECOMPUTED on host (guide App. B:
// on-device sinf/cosf turns memory-bound into VALU-bound).
// q/k layout: [B, S, H, D] bf16 (contiguous head-last).  table: [S, D/2] f32
// pairs (cos, sin) interleaved as float2.
#include "common.h"

#define BLOCK 256

// applies in-place-capable: out may alias x.
// rot = D/2.  For token (b,s,h): out[d]      = x[d]*cos[d] - x[d+rot]*sin[d]
//                                out[d+rot]  = x[d+rot]*cos[d] + x[d]*sin[d]
extern "C" __global__ void __launch_bounds__(BLOCK)
rope_fwd_bf16(const unsigned short* __restrict__ x,
              unsigned short* __restrict__ out,
              const float* __restrict__ table,  // [S, rot, 2] (cos,sin)
              int64_t n_tokens,                  // B*S*H
              int n_heads, int seq, int dim, int pos_offset, int backward) {
  const int rot = dim / 2;
  const int vec = 2;  // process 2 rotation pairs per iter (4 bf16 loads)
  int64_t total_pairs = n_tokens * rot;
  int64_t idx0 = (int64_t)(blockIdx.x * BLOCK + threadIdx.x) * vec;
  int64_t stride = (int64_t)gridDim.x * BLOCK * vec;
  for (int64_t p = idx0; p < total_pairs; p += stride) {
#pragma unroll
    for (int v = 0; v < vec; ++v) {
      int64_t pp = p + v;
      if (pp >= total_pairs) break;
      int64_t tok = pp / rot;
      int d = (int)(pp % rot);
      int64_t s = (tok / n_heads) % seq;  // token layout [B,S,H]
      const float* tb = table + ((s + pos_offset) * (int64_t)rot + d) * 2;
      float c = tb[0], sn = tb[1];
      if (backward) sn = -sn;
      int64_t base = tok * dim;
      float x0 = bf16_to_f32(x[base + d]);
      float x1 = bf16_to_f32(x[base + d + rot]);
      out[base + d] = f32_to_bf16(x0 * c - x1 * sn);
      out[base + d + rot] = f32_to_bf16(x1 * c + x0 * sn);
    }
  }
}
