// Sharded-philox random fills — the MI355X replacement for the reference's
// patched CUDA DistributionTemplates / Dropout.cu (SURVEY.md §2.6 #2,#3).
//
// Parity contract: every element's randomness is keyed by its GLOBAL
// element index (ctr = g/4, component g%4 of philox4x32-10), so a sharded
// fill is bitwise-identical to our own single-GPU fill regardless of shard
// shape, launch geometry, or wavefront width.  The shard descriptor maps
// local row-major indices to global ones.
#include "common.h"

#define BLOCK 256
#define MAXD 8

struct ShardDesc {
  int64_t gshape[MAXD];
  int64_t lshape[MAXD];
  int64_t offset[MAXD];
  int ndim;
  int64_t flat_offset;  // fast path: local is a contiguous global range
  int is_flat;
};

DEV int64_t local_to_global(const ShardDesc d, int64_t li) {
  if (d.is_flat) return li + d.flat_offset;
  int64_t g = 0, rem = li;
  int64_t gstride = 1;
  // compute global strides on the fly (row-major): need from last dim back
  int64_t coords[MAXD];
#pragma unroll
  for (int k = MAXD - 1; k >= 0; --k) {
    if (k < d.ndim) {
      coords[k] = rem % d.lshape[k];
      rem /= d.lshape[k];
    }
  }
  int64_t stride = 1;
#pragma unroll
  for (int k = MAXD - 1; k >= 0; --k) {
    if (k < d.ndim) {
      g += (coords[k] + d.offset[k]) * stride;
      stride *= d.gshape[k];
    }
  }
  return g;
}

DEV float philox_uniform_at(uint64_t seed, uint64_t offset, int64_t g) {
  Philox4 p = philox4x32_10(seed, (uint64_t)(g >> 2) + offset, 0u);
  uint32_t u;
  switch (g & 3) {
    case 0: u = p.x; break;
    case 1: u = p.y; break;
    case 2: u = p.z; break;
    default: u = p.w; break;
  }
  return uint32_to_uniform(u);
}

DEV float philox_normal_at(uint64_t seed, uint64_t offset, int64_t g) {
  Philox4 p = philox4x32_10(seed, (uint64_t)(g >> 2) + offset, 1u);
  float n0, n1, n2, n3;
  box_muller(uint32_to_uniform(p.x), uint32_to_uniform(p.y), &n0, &n1);
  box_muller(uint32_to_uniform(p.z), uint32_to_uniform(p.w), &n2, &n3);
  switch (g & 3) {
    case 0: return n0;
    case 1: return n1;
    case 2: return n2;
    default: return n3;
  }
}

extern "C" __global__ void __launch_bounds__(BLOCK)
philox_uniform_bf16(unsigned short* __restrict__ out, ShardDesc d,
                    int64_t n, uint64_t seed, uint64_t offset,
                    float lo, float hi) {
  int64_t i0 = blockIdx.x * BLOCK + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * BLOCK;
  for (int64_t i = i0; i < n; i += stride) {
    float u = philox_uniform_at(seed, offset, local_to_global(d, i));
    out[i] = f32_to_bf16(lo + u * (hi - lo));
  }
}

extern "C" __global__ void __launch_bounds__(BLOCK)
philox_uniform_f32(float* __restrict__ out, ShardDesc d, int64_t n,
                   uint64_t seed, uint64_t offset, float lo, float hi) {
  int64_t i0 = blockIdx.x * BLOCK + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * BLOCK;
  for (int64_t i = i0; i < n; i += stride) {
    float u = philox_uniform_at(seed, offset, local_to_global(d, i));
    out[i] = lo + u * (hi - lo);
  }
}

extern "C" __global__ void __launch_bounds__(BLOCK)
philox_normal_bf16(unsigned short* __restrict__ out, ShardDesc d, int64_t n,
                   uint64_t seed, uint64_t offset, float mean, float std) {
  int64_t i0 = blockIdx.x * BLOCK + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * BLOCK;
  for (int64_t i = i0; i < n; i += stride) {
    float z = philox_normal_at(seed, offset, local_to_global(d, i));
    out[i] = f32_to_bf16(mean + z * std);
  }
}

extern "C" __global__ void __launch_bounds__(BLOCK)
philox_normal_f32(float* __restrict__ out, ShardDesc d, int64_t n,
                  uint64_t seed, uint64_t offset, float mean, float std) {
  int64_t i0 = blockIdx.x * BLOCK + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * BLOCK;
  for (int64_t i = i0; i < n; i += stride) {
    float z = philox_normal_at(seed, offset, local_to_global(d, i));
    out[i] = mean + z * std;
  }
}

// fused dropout: out = keep ? x/(1-p) : 0;  mask byte = keep
extern "C" __global__ void __launch_bounds__(BLOCK)
philox_dropout_bf16(const unsigned short* __restrict__ x,
                    unsigned short* __restrict__ out,
                    unsigned char* __restrict__ mask, ShardDesc d, int64_t n,
                    uint64_t seed, uint64_t offset, float p) {
  const float scale = 1.0f / (1.0f - p);
  int64_t i0 = blockIdx.x * BLOCK + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * BLOCK;
  for (int64_t i = i0; i < n; i += stride) {
    float u = philox_uniform_at(seed, offset, local_to_global(d, i));
    bool keep = u > p;
    float f = bf16_to_f32(x[i]);
    out[i] = f32_to_bf16(keep ? f * scale : 0.f);
    if (mask) mask[i] = keep ? 1 : 0;
  }
}
