// Fused AdamW over flat (ragged-sharded) buffers — CDNA4.
//
// The FSDP engine keeps params/grads/opt-state as FLAT per-rank shards
// (RaggedShard layout), so the optimizer is a single fused kernel per
// buffer — no multi-tensor-apply bookkeeping needed (SURVEY.md §2.7
// "_fused_adamw_ ... over ragged local shards").  bf16 params + fp32
// master weights + fp32 m/v; decoupled weight decay; bias correction.
#include "common.h"

#define BLOCK 256

extern "C" __global__ void __launch_bounds__(BLOCK)
adamw_flat_bf16(unsigned short* __restrict__ param_bf16,
                float* __restrict__ master,      // fp32 master (may be null)
                const unsigned short* __restrict__ grad_bf16,
                const float* __restrict__ grad_f32,  // one of the two grads
                float* __restrict__ m,
                float* __restrict__ v,
                int64_t n, float lr, float beta1, float beta2, float eps,
                float weight_decay, float bc1, float bc2,  // 1-beta^t
                float grad_scale) {
  const int vec = 4;
  int64_t i0 = (int64_t)(blockIdx.x * BLOCK + threadIdx.x) * vec;
  int64_t stride = (int64_t)gridDim.x * BLOCK * vec;
  const float step_size = lr / bc1;
  for (int64_t i = i0; i < n; i += stride) {
#pragma unroll
    for (int j = 0; j < vec; ++j) {
      int64_t k = i + j;
      if (k >= n) break;
      float g = grad_f32 ? grad_f32[k] : bf16_to_f32(grad_bf16[k]);
      g *= grad_scale;
      float p = master ? master[k] : bf16_to_f32(param_bf16[k]);
      p -= lr * weight_decay * p;
      float mk = beta1 * m[k] + (1.f - beta1) * g;
      float vk = beta2 * v[k] + (1.f - beta2) * g * g;
      m[k] = mk;
      v[k] = vk;
      float denom = sqrtf(vk / bc2) + eps;
      p -= step_size * mk / denom;
      if (master) master[k] = p;
      param_bf16[k] = f32_to_bf16(p);
    }
  }
}

// L2-norm-squared of a flat bf16 or fp32 buffer (for grad clipping):
// partial per block -> atomicAdd into out[0] (fp32, zero-init by caller).
extern "C" __global__ void __launch_bounds__(BLOCK)
l2norm_sq_flat(const unsigned short* __restrict__ x_bf16,
               const float* __restrict__ x_f32,
               float* __restrict__ out, int64_t n) {
  __shared__ float lds[BLOCK / WAVE];
  const int vec = 8;
  int64_t i0 = (int64_t)(blockIdx.x * BLOCK + threadIdx.x) * vec;
  int64_t stride = (int64_t)gridDim.x * BLOCK * vec;
  float ss = 0.f;
  for (int64_t i = i0; i < n; i += stride) {
#pragma unroll
    for (int j = 0; j < vec; ++j) {
      int64_t k = i + j;
      if (k >= n) break;
      float f = x_f32 ? x_f32[k] : bf16_to_f32(x_bf16[k]);
      ss += f * f;
    }
  }
  float total = block_reduce_sum<BLOCK>(ss, lds);
  if (threadIdx.x == 0) atomicAdd(out, total);
}

// in-place scale of a flat buffer (grad clip / loss scale)
extern "C" __global__ void __launch_bounds__(BLOCK)
scale_flat(unsigned short* __restrict__ x_bf16, float* __restrict__ x_f32,
           const float* __restrict__ scale_ptr, float scale_const, int64_t n) {
  const int vec = 8;
  float s = scale_ptr ? *scale_ptr : scale_const;
  int64_t i0 = (int64_t)(blockIdx.x * BLOCK + threadIdx.x) * vec;
  int64_t stride = (int64_t)gridDim.x * BLOCK * vec;
  for (int64_t i = i0; i < n; i += stride) {
#pragma unroll
    for (int j = 0; j < vec; ++j) {
      int64_t k = i + j;
      if (k >= n) break;
      if (x_f32) x_f32[k] *= s;
      else x_bf16[k] = f32_to_bf16(bf16_to_f32(x_bf16[k]) * s);
    }
  }
}
