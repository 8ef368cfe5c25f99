// Fused cross-entropy over a large vocab — CDNA4.
//
// Replaces log_softmax + nll (reference: vocab-parallel CE at
// legacy/vescale/model/patch/vp_cross_entropy.py:43-147 is the sharded
// variant; this is the dense fast path).  One 256-thread block per row,
// single pass max+sumexp (online), bf16x8 loads.  Backward recomputes
// softmax from the saved logsumexp and can write grads IN-PLACE over the
// logits buffer (a [tokens, 128k] bf16 tensor is GBs — avoid a second one).
#include "common.h"

#define BLOCK 256

// logits: [N, V] bf16; target: [N] int64; loss/lse: [N] f32
extern "C" __global__ void __launch_bounds__(BLOCK)
ce_fwd_bf16(const unsigned short* __restrict__ logits,
            const int64_t* __restrict__ target,
            float* __restrict__ loss, float* __restrict__ lse_out,
            int64_t n_rows, int vocab, int64_t ignore_index) {
  __shared__ float lds[BLOCK / WAVE];
  const int vec = 8;
  const int iters = (vocab / vec + BLOCK - 1) / BLOCK;
  for (int64_t row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const unsigned short* xr = logits + row * vocab;
    int64_t tgt = target[row];
    // online max + sumexp in one pass
    float mx = -INFINITY, sum = 0.f;
    for (int it = 0; it < iters; ++it) {
      int i = (it * BLOCK + threadIdx.x) * vec;
      if (i < vocab) {
        short8v v = *reinterpret_cast<const short8v*>(xr + i);
#pragma unroll
        for (int j = 0; j < vec; ++j) {
          int col = i + j;
          float f = (col < vocab) ? bf16_to_f32((unsigned short)v[j]) : -INFINITY;
          float m2 = fmaxf(mx, f);
          sum = sum * __expf(mx - m2) + ((col < vocab) ? __expf(f - m2) : 0.f);
          mx = m2;
        }
      }
    }
    // block combine (max then rescaled sums)
    float gmax = block_reduce_max<BLOCK>(mx, lds);
    float gsum = block_reduce_sum<BLOCK>(sum * __expf(mx - gmax), lds);
    float lse = gmax + __logf(gsum);
    if (threadIdx.x == 0) {
      if (tgt == ignore_index) {
        loss[row] = 0.f;
        lse_out[row] = lse;
      } else {
        float xt = bf16_to_f32(xr[tgt]);
        loss[row] = lse - xt;
        lse_out[row] = lse;
      }
    }
  }
}

// dlogits[r, c] = scale[r] * (softmax - onehot); scale[r] = dloss[r] (0 for
// ignored rows).  dlogits MAY alias logits.
extern "C" __global__ void __launch_bounds__(BLOCK)
ce_bwd_bf16(const unsigned short* __restrict__ logits,
            unsigned short* __restrict__ dlogits,
            const int64_t* __restrict__ target,
            const float* __restrict__ lse,
            const float* __restrict__ dloss,   // [N] upstream grad per row
            int64_t n_rows, int vocab, int64_t ignore_index) {
  const int vec = 8;
  const int iters = (vocab / vec + BLOCK - 1) / BLOCK;
  for (int64_t row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const unsigned short* xr = logits + row * vocab;
    unsigned short* dr = dlogits + row * vocab;
    int64_t tgt = target[row];
    float l = lse[row];
    float sc = (tgt == ignore_index) ? 0.f : dloss[row];
    for (int it = 0; it < iters; ++it) {
      int i = (it * BLOCK + threadIdx.x) * vec;
      if (i < vocab) {
        short8v v = *reinterpret_cast<const short8v*>(xr + i);
        short8v o;
#pragma unroll
        for (int j = 0; j < vec; ++j) {
          int col = i + j;
          float f = bf16_to_f32((unsigned short)v[j]);
          float p = __expf(f - l);
          if (col == (int)tgt) p -= 1.f;
          o[j] = (short)f32_to_bf16(p * sc);
        }
        *reinterpret_cast<short8v*>(dr + i) = o;
      }
    }
  }
}
