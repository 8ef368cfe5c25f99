// Fused SwiGLU (silu(gate) * up) forward/backward — CDNA4 memory-bound.
//
// Fusing removes one full read+write of the ffn activation vs separate
// silu+mul (HBM3E-bound op: fuse elementwise work into the producer).
// gate/up are separate [N, F] bf16 tensors (the two halves of the fused
// w13 matmul output views).
#include "common.h"

#define BLOCK 256

// PACKED variant: gu = [N, 2F] (gate||up per row, the fused-w13 GEMM
// output) -> out [N, F]; avoids the split()+contiguous()+cat() round trip
// torch autograd would otherwise insert (profiled at ~50 ms/step on 8B).
extern "C" __global__ void __launch_bounds__(BLOCK)
swiglu_packed_fwd_bf16(const unsigned short* __restrict__ gu,
                       unsigned short* __restrict__ out,
                       int64_t n_rows, int F) {
  const int vec = 8;
  int64_t total = n_rows * (int64_t)(F / vec);
  int64_t i0 = blockIdx.x * BLOCK + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * BLOCK;
  for (int64_t t = i0; t < total; t += stride) {
    int64_t row = t / (F / vec);
    int64_t col = (t % (F / vec)) * vec;
    const unsigned short* base = gu + row * 2 * F;
    short8v g = *reinterpret_cast<const short8v*>(base + col);
    short8v u = *reinterpret_cast<const short8v*>(base + F + col);
    short8v o;
#pragma unroll
    for (int j = 0; j < vec; ++j) {
      float gf = bf16_to_f32((unsigned short)g[j]);
      float uf = bf16_to_f32((unsigned short)u[j]);
      o[j] = (short)f32_to_bf16(gf / (1.0f + __expf(-gf)) * uf);
    }
    *reinterpret_cast<short8v*>(out + row * F + col) = o;
  }
}

extern "C" __global__ void __launch_bounds__(BLOCK)
swiglu_packed_bwd_bf16(const unsigned short* __restrict__ dy,
                       const unsigned short* __restrict__ gu,
                       unsigned short* __restrict__ dgu,
                       int64_t n_rows, int F) {
  const int vec = 8;
  int64_t total = n_rows * (int64_t)(F / vec);
  int64_t i0 = blockIdx.x * BLOCK + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * BLOCK;
  for (int64_t t = i0; t < total; t += stride) {
    int64_t row = t / (F / vec);
    int64_t col = (t % (F / vec)) * vec;
    const unsigned short* base = gu + row * 2 * F;
    short8v d = *reinterpret_cast<const short8v*>(dy + row * F + col);
    short8v g = *reinterpret_cast<const short8v*>(base + col);
    short8v u = *reinterpret_cast<const short8v*>(base + F + col);
    short8v og, ou;
#pragma unroll
    for (int j = 0; j < vec; ++j) {
      float df = bf16_to_f32((unsigned short)d[j]);
      float gf = bf16_to_f32((unsigned short)g[j]);
      float uf = bf16_to_f32((unsigned short)u[j]);
      float sig = 1.0f / (1.0f + __expf(-gf));
      og[j] = (short)f32_to_bf16(df * uf * sig * (1.0f + gf * (1.0f - sig)));
      ou[j] = (short)f32_to_bf16(df * gf * sig);
    }
    unsigned short* dbase = dgu + row * 2 * F;
    *reinterpret_cast<short8v*>(dbase + col) = og;
    *reinterpret_cast<short8v*>(dbase + F + col) = ou;
  }
}

extern "C" __global__ void __launch_bounds__(BLOCK)
swiglu_fwd_bf16(const unsigned short* __restrict__ gate,
                const unsigned short* __restrict__ up,
                unsigned short* __restrict__ out, int64_t n) {
  const int vec = 8;
  int64_t i0 = (int64_t)(blockIdx.x * BLOCK + threadIdx.x) * vec;
  int64_t stride = (int64_t)gridDim.x * BLOCK * vec;
  for (int64_t i = i0; i + vec <= n; i += stride) {
    short8v g = *reinterpret_cast<const short8v*>(gate + i);
    short8v u = *reinterpret_cast<const short8v*>(up + i);
    short8v o;
#pragma unroll
    for (int j = 0; j < vec; ++j) {
      float gf = bf16_to_f32((unsigned short)g[j]);
      float uf = bf16_to_f32((unsigned short)u[j]);
      float s = gf / (1.0f + __expf(-gf));
      o[j] = (short)f32_to_bf16(s * uf);
    }
    *reinterpret_cast<short8v*>(out + i) = o;
  }
  // tail (n not multiple of 8)
  int64_t tail_start = (n / vec) * vec;
  int64_t t = tail_start + blockIdx.x * BLOCK + threadIdx.x;
  if (blockIdx.x == 0 && t < n && t >= tail_start) {
    float gf = bf16_to_f32(gate[t]);
    float uf = bf16_to_f32(up[t]);
    out[t] = f32_to_bf16(gf / (1.0f + __expf(-gf)) * uf);
  }
}

// dgate = dy * up * silu'(gate);  dup = dy * silu(gate)
// silu'(x) = sig(x) * (1 + x * (1 - sig(x)))
extern "C" __global__ void __launch_bounds__(BLOCK)
swiglu_bwd_bf16(const unsigned short* __restrict__ dy,
                const unsigned short* __restrict__ gate,
                const unsigned short* __restrict__ up,
                unsigned short* __restrict__ dgate,
                unsigned short* __restrict__ dup, int64_t n) {
  const int vec = 8;
  int64_t i0 = (int64_t)(blockIdx.x * BLOCK + threadIdx.x) * vec;
  int64_t stride = (int64_t)gridDim.x * BLOCK * vec;
  for (int64_t i = i0; i + vec <= n; i += stride) {
    short8v d = *reinterpret_cast<const short8v*>(dy + i);
    short8v g = *reinterpret_cast<const short8v*>(gate + i);
    short8v u = *reinterpret_cast<const short8v*>(up + i);
    short8v og, ou;
#pragma unroll
    for (int j = 0; j < vec; ++j) {
      float df = bf16_to_f32((unsigned short)d[j]);
      float gf = bf16_to_f32((unsigned short)g[j]);
      float uf = bf16_to_f32((unsigned short)u[j]);
      float sig = 1.0f / (1.0f + __expf(-gf));
      float silu = gf * sig;
      og[j] = (short)f32_to_bf16(df * uf * sig * (1.0f + gf * (1.0f - sig)));
      ou[j] = (short)f32_to_bf16(df * silu);
    }
    *reinterpret_cast<short8v*>(dgate + i) = og;
    *reinterpret_cast<short8v*>(dup + i) = ou;
  }
  int64_t tail_start = (n / vec) * vec;
  int64_t t = tail_start + blockIdx.x * BLOCK + threadIdx.x;
  if (blockIdx.x == 0 && t < n && t >= tail_start) {
    float df = bf16_to_f32(dy[t]);
    float gf = bf16_to_f32(gate[t]);
    float uf = bf16_to_f32(up[t]);
    float sig = 1.0f / (1.0f + __expf(-gf));
    dgate[t] = f32_to_bf16(df * uf * sig * (1.0f + gf * (1.0f - sig)));
    dup[t] = f32_to_bf16(df * gf * sig);
  }
}
