// RMSNorm forward/backward — CDNA4, memory-bound (target: HBM BW ceiling).
//
// Replaces the reference's layer-norm call sites for Llama (SURVEY.md §2.7
// "layer_norm fwd/bwd ... RMSNorm for Llama in new build").  bf16 I/O,
// fp32 accumulation, bf16x8 vectorized loads (guide Guideline 13: scalar
// bf16 loads cost ~2x), one 256-thread block per row for H<=8192, fp32
// rrms stashed for backward.
#include "common.h"

#define BLOCK 256

// x: [N, H] bf16;  w: [H] bf16;  out: [N, H] bf16;  rrms: [N] f32
extern "C" __global__ void __launch_bounds__(BLOCK)
rmsnorm_fwd_bf16(const unsigned short* __restrict__ x,
                 const unsigned short* __restrict__ w,
                 unsigned short* __restrict__ out,
                 float* __restrict__ rrms_out,
                 int64_t n_rows, int hidden, float eps) {
  __shared__ float lds[BLOCK / WAVE];
  const int vec = 8;
  const int per_row_iters = (hidden / vec + BLOCK - 1) / BLOCK;
  for (int64_t row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const unsigned short* xr = x + row * hidden;
    unsigned short* orow = out + row * hidden;
    float ss = 0.f;
    // pass 1: sum of squares
    for (int it = 0; it < per_row_iters; ++it) {
      int i = (it * BLOCK + threadIdx.x) * vec;
      if (i < hidden) {
        short8v v = *reinterpret_cast<const short8v*>(xr + i);
#pragma unroll
        for (int j = 0; j < vec; ++j) {
          float f = bf16_to_f32((unsigned short)v[j]);
          ss += f * f;
        }
      }
    }
    float total = block_reduce_sum<BLOCK>(ss, lds);
    float rrms = rsqrtf(total / (float)hidden + eps);
    if (threadIdx.x == 0 && rrms_out) rrms_out[row] = rrms;
    // pass 2: normalize + scale (x row is in L1/L2 now)
    for (int it = 0; it < per_row_iters; ++it) {
      int i = (it * BLOCK + threadIdx.x) * vec;
      if (i < hidden) {
        short8v v = *reinterpret_cast<const short8v*>(xr + i);
        short8v wv = *reinterpret_cast<const short8v*>(w + i);
        short8v o;
#pragma unroll
        for (int j = 0; j < vec; ++j) {
          float f = bf16_to_f32((unsigned short)v[j]);
          float g = bf16_to_f32((unsigned short)wv[j]);
          o[j] = (short)f32_to_bf16(f * rrms * g);
        }
        *reinterpret_cast<short8v*>(orow + i) = o;
      }
    }
  }
}

// fused residual-add + rmsnorm forward: res_new = x + res (written out),
// out = rmsnorm(res_new) * w.  Fuses the transformer residual-stream add
// into the norm's read pass — saves a full read+write of the hidden state
// per call vs a separate elementwise add kernel (the adds were ~2% of the
// Llama-8B step).  res may be null (start of the stream): res_new = x.
extern "C" __global__ void __launch_bounds__(BLOCK)
rmsnorm_res_fwd_bf16(const unsigned short* __restrict__ x,
                     const unsigned short* __restrict__ res,
                     const unsigned short* __restrict__ w,
                     unsigned short* __restrict__ out,
                     unsigned short* __restrict__ res_out,
                     float* __restrict__ rrms_out,
                     int64_t n_rows, int hidden, float eps) {
  __shared__ float lds[BLOCK / WAVE];
  const int vec = 8;
  const int per_row_iters = (hidden / vec + BLOCK - 1) / BLOCK;
  for (int64_t row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const unsigned short* xr = x + row * hidden;
    const unsigned short* rr = res ? res + row * hidden : nullptr;
    unsigned short* orow = out + row * hidden;
    unsigned short* rout = res_out + row * hidden;
    float ss = 0.f;
    for (int it = 0; it < per_row_iters; ++it) {
      int i = (it * BLOCK + threadIdx.x) * vec;
      if (i < hidden) {
        short8v v = *reinterpret_cast<const short8v*>(xr + i);
        short8v o;
        if (rr) {
          short8v rv = *reinterpret_cast<const short8v*>(rr + i);
#pragma unroll
          for (int j = 0; j < vec; ++j) {
            float f = bf16_to_f32((unsigned short)v[j]) +
                      bf16_to_f32((unsigned short)rv[j]);
            o[j] = (short)f32_to_bf16(f);
            // accumulate from the ROUNDED value: pass 2 normalizes the
            // bf16 res_new, so rrms must be computed over the same data
            float fr = bf16_to_f32((unsigned short)o[j]);
            ss += fr * fr;
          }
        } else {
#pragma unroll
          for (int j = 0; j < vec; ++j) {
            float f = bf16_to_f32((unsigned short)v[j]);
            o[j] = (short)v[j];
            ss += f * f;
          }
        }
        *reinterpret_cast<short8v*>(rout + i) = o;
      }
    }
    float total = block_reduce_sum<BLOCK>(ss, lds);
    float rrms = rsqrtf(total / (float)hidden + eps);
    if (threadIdx.x == 0 && rrms_out) rrms_out[row] = rrms;
    for (int it = 0; it < per_row_iters; ++it) {
      int i = (it * BLOCK + threadIdx.x) * vec;
      if (i < hidden) {
        short8v v = *reinterpret_cast<const short8v*>(rout + i);  // L1-hot
        short8v wv = *reinterpret_cast<const short8v*>(w + i);
        short8v o;
#pragma unroll
        for (int j = 0; j < vec; ++j) {
          float f = bf16_to_f32((unsigned short)v[j]);
          float g = bf16_to_f32((unsigned short)wv[j]);
          o[j] = (short)f32_to_bf16(f * rrms * g);
        }
        *reinterpret_cast<short8v*>(orow + i) = o;
      }
    }
  }
}

// backward:
//   dx = rrms * w * dy - x * rrms^3/H * sum_j(dy_j * w_j * x_j)
//   dw = sum_rows(dy * x * rrms)
// dw accumulated per-block in registers over this block's rows, then one
// atomicAdd per column into fp32 dw buffer (caller zero-inits).
extern "C" __global__ void __launch_bounds__(BLOCK)
rmsnorm_bwd_bf16(const unsigned short* __restrict__ dy,
                 const unsigned short* __restrict__ x,
                 const unsigned short* __restrict__ w,
                 const float* __restrict__ rrms_in,
                 const unsigned short* __restrict__ dres,  // nullable:
                 // residual-stream grad added into dx (fused add-backward)
                 unsigned short* __restrict__ dx,
                 float* __restrict__ dw,  // [H] fp32, zero-init
                 int64_t n_rows, int hidden) {
  __shared__ float lds[BLOCK / WAVE];
  const int vec = 8;
  // per-thread register cache of the row data: pass 1 reads dy/x/w from
  // global ONCE; pass 2 consumes the registers (hidden <= 8192 at BLOCK
  // 256 -> up to 4 iters x 8 elems live per tensor)
  const int per_row_iters = (hidden / vec + BLOCK - 1) / BLOCK;
  float dw_acc[32];
#pragma unroll
  for (int j = 0; j < 32; ++j) dw_acc[j] = 0.f;
  // w is row-invariant: load once per thread
  float wreg[32];
  #pragma unroll
  for (int it = 0; it < 4; ++it) {
    int i = (it * BLOCK + threadIdx.x) * vec;
    if (i < hidden) {
      short8v wv = *reinterpret_cast<const short8v*>(w + i);
#pragma unroll
      for (int j = 0; j < vec; ++j) wreg[it * vec + j] = bf16_to_f32((unsigned short)wv[j]);
    }
  }

  for (int64_t row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const unsigned short* dyr = dy + row * hidden;
    const unsigned short* xr = x + row * hidden;
    unsigned short* dxr = dx + row * hidden;
    float rrms = rrms_in[row];
    float dot = 0.f;
    float dyreg[32], xreg[32];
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      int i = (it * BLOCK + threadIdx.x) * vec;
      if (i < hidden) {
        short8v dv = *reinterpret_cast<const short8v*>(dyr + i);
        short8v xv = *reinterpret_cast<const short8v*>(xr + i);
#pragma unroll
        for (int j = 0; j < vec; ++j) {
          float dyf = bf16_to_f32((unsigned short)dv[j]);
          float xf = bf16_to_f32((unsigned short)xv[j]);
          dyreg[it * vec + j] = dyf;
          xreg[it * vec + j] = xf;
          dot += dyf * wreg[it * vec + j] * xf;
        }
      }
    }
    float tdot = block_reduce_sum<BLOCK>(dot, lds);
    float coef = tdot * rrms * rrms * rrms / (float)hidden;
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      int i = (it * BLOCK + threadIdx.x) * vec;
      if (i < hidden) {
        short8v o;
        short8v drv;
        if (dres)
          drv = *reinterpret_cast<const short8v*>(dres + row * hidden + i);
#pragma unroll
        for (int j = 0; j < vec; ++j) {
          float dyf = dyreg[it * vec + j];
          float xf = xreg[it * vec + j];
          float g = rrms * wreg[it * vec + j] * dyf - xf * coef;
          if (dres) g += bf16_to_f32((unsigned short)drv[j]);
          o[j] = (short)f32_to_bf16(g);
          dw_acc[it * vec + j] += dyf * xf * rrms;
        }
        *reinterpret_cast<short8v*>(dxr + i) = o;
      }
    }
  }
  // flush dw
  #pragma unroll
  for (int it = 0; it < 4; ++it) {
    int i = (it * BLOCK + threadIdx.x) * vec;
    if (i < hidden) {
#pragma unroll
      for (int j = 0; j < vec; ++j) atomicAdd(dw + i + j, dw_acc[it * vec + j]);
    }
  }
}
