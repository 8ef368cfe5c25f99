// Common device helpers for vescale_amd CDNA4 (gfx950) kernels.
//
// Written MI355X-first per /opt/skills/guides/cdna_hip_programming.md:
//   - wave = 64 lanes (hard-coded; NOT warp-32)
//   - bf16 loads vectorized as short4/short8 (Guideline 13)
//   - grid-stride memory-bound launches capped at ~2048 blocks (Guideline 11)
//   - philox4x32-10 counter-based RNG with GLOBAL virtual indexing so a
//     sharded fill is bitwise-identical to the single-GPU fill (the MI355X
//     replacement for the reference's patched CUDA DistributionTemplates —
//     SURVEY.md §2.6 #2)
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <cstdint>

#define WAVE 64
#define DEV __device__ __forceinline__

typedef short short4v __attribute__((ext_vector_type(4)));
typedef short short8v __attribute__((ext_vector_type(8)));
typedef float float4v __attribute__((ext_vector_type(4)));

DEV float bf16_to_f32(unsigned short u) {
  union { unsigned int i; float f; } v;
  v.i = ((unsigned int)u) << 16;
  return v.f;
}

DEV unsigned short f32_to_bf16(float f) {
  union { unsigned int i; float f; } v;
  v.f = f;
  // round-to-nearest-even
  unsigned int lsb = (v.i >> 16) & 1;
  v.i += 0x7fffu + lsb;
  return (unsigned short)(v.i >> 16);
}

// ---------------- wave / block reductions ----------------
DEV float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) x += __shfl_down(x, off, WAVE);
  return x;  // valid in lane 0
}

DEV float wave_reduce_max(float x) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) x = fmaxf(x, __shfl_down(x, off, WAVE));
  return x;
}

DEV float wave_allreduce_sum(float x) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE);
  return x;
}

DEV float wave_allreduce_max(float x) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) x = fmaxf(x, __shfl_xor(x, off, WAVE));
  return x;
}

// Block reduction via LDS (block size multiple of 64, <= 1024).
template <int BLOCK>
DEV float block_reduce_sum(float x, float* lds /* BLOCK/WAVE floats */) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  x = wave_reduce_sum(x);
  if (lane == 0) lds[wid] = x;
  __syncthreads();
  constexpr int NW = BLOCK / WAVE;
  float v = (threadIdx.x < NW) ? lds[threadIdx.x] : 0.f;
  if (wid == 0) {
#pragma unroll
    for (int off = NW / 2; off > 0; off >>= 1) v += __shfl_down(v, off, WAVE);
    if (lane == 0) lds[0] = v;
  }
  __syncthreads();
  float r = lds[0];
  __syncthreads();
  return r;
}

template <int BLOCK>
DEV float block_reduce_max(float x, float* lds) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  x = wave_reduce_max(x);
  if (lane == 0) lds[wid] = x;
  __syncthreads();
  constexpr int NW = BLOCK / WAVE;
  float v = (threadIdx.x < NW) ? lds[threadIdx.x] : -INFINITY;
  if (wid == 0) {
#pragma unroll
    for (int off = NW / 2; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off, WAVE));
    if (lane == 0) lds[0] = v;
  }
  __syncthreads();
  float r = lds[0];
  __syncthreads();
  return r;
}

// ---------------- philox4x32-10 ----------------
// Counter-based: out = philox(key=(seed), ctr=(idx, stream...)).  We index
// the counter by the GLOBAL element id so shards reproduce the single-GPU
// stream bitwise regardless of launch geometry.
struct Philox4 {
  uint32_t x, y, z, w;
};

DEV uint32_t mulhilo(uint32_t a, uint32_t b, uint32_t* hip) {
  uint64_t p = (uint64_t)a * (uint64_t)b;
  *hip = (uint32_t)(p >> 32);
  return (uint32_t)p;
}

DEV Philox4 philox4x32_10(uint64_t seed, uint64_t ctr64, uint32_t ctr_hi) {
  const uint32_t kPhiloxW32A = 0x9E3779B9u, kPhiloxW32B = 0xBB67AE85u;
  const uint32_t kPhiloxM4x32A = 0xD2511F53u, kPhiloxM4x32B = 0xCD9E8D57u;
  uint32_t key0 = (uint32_t)seed, key1 = (uint32_t)(seed >> 32);
  uint32_t c0 = (uint32_t)ctr64, c1 = (uint32_t)(ctr64 >> 32), c2 = ctr_hi, c3 = 0;
#pragma unroll
  for (int i = 0; i < 10; ++i) {
    uint32_t hi0, hi1;
    uint32_t lo0 = mulhilo(kPhiloxM4x32A, c0, &hi0);
    uint32_t lo1 = mulhilo(kPhiloxM4x32B, c2, &hi1);
    uint32_t n0 = hi1 ^ c1 ^ key0;
    uint32_t n1 = lo1;
    uint32_t n2 = hi0 ^ c3 ^ key1;
    uint32_t n3 = lo0;
    c0 = n0; c1 = n1; c2 = n2; c3 = n3;
    key0 += kPhiloxW32A;
    key1 += kPhiloxW32B;
  }
  return {c0, c1, c2, c3};
}

DEV float uint32_to_uniform(uint32_t v) {
  // (0,1]: match curand convention closely enough for our own parity
  constexpr float k = 1.0f / 4294967296.0f;
  return ((float)v + 1.0f) * k;
}

// Box-Muller pair from two uniforms
DEV void box_muller(float u1, float u2, float* n1, float* n2) {
  float r = sqrtf(-2.0f * logf(u1));
  float s, c;
  __sincosf(6.2831853071795864769f * u2, &s, &c);
  *n1 = r * c;
  *n2 = r * s;
}

// ---------------- launch helpers (host) ----------------
static inline int grid_for(int64_t n, int block, int cap = 2048) {
  int64_t g = (n + block - 1) / block;
  return (int)(g < cap ? (g > 0 ? g : 1) : cap);
}

#define HIP_CHECK(cmd)                                                        \
  do {                                                                        \
    hipError_t e = (cmd);                                                     \
    if (e != hipSuccess) {                                                    \
      throw std::runtime_error(std::string("HIP error: ") +                   \
                               hipGetErrorString(e));                         \
    }                                                                         \
  } while (0)
