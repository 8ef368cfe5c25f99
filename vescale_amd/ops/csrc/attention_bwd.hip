// Flash-attention BACKWARD — hand-written CDNA4 MFMA kernels.
//
// The forward stays on the library flash kernel (it returns the logsumexp);
// these kernels replace the slow stock backward (profiled at ~255 TF
// effective on MI355X) with the standard FA2 split:
//   1) fa_bwd_preprocess: delta[b,h,q] = rowsum(dO * O)
//   2) fa_bwd_dkdv: per KV-tile: recompute P^T, dP^T; accumulate dV, dK
//      (per Q-head partials; GQA groups reduced by kernel 4)
//   3) fa_bwd_dq:   per Q-tile: recompute P, dP; accumulate dQ
//   4) fa_bwd_reduce_gqa: sum dK/dV partials over the query-head group
//
// Every GEMM inside is expressed in the TN form C[i][j] = sum_k X[i][k] *
// Y[j][k] with both operands row-major over the contraction dim, so each
// mfma_f32_16x16x32_bf16 operand is a contiguous 8-elem LDS read (operand
// fragment: row = lane&15, k-chunk = lane>>4; D-layout: col = lane&15,
// row = 4*(lane>>4)+r — guide §3, m89-verified).  Tensors transposed on
// stage where the TN form needs it (Q^T/dO^T/K^T), with +8-elem row pad
// against bank conflicts.  Correctness-first structure: __syncthreads
// stages, no deep pipelining yet.
//
// Layout: q,k,v,o,do_,dq: [B, H, S, D] contiguous, D = 128.  causal only.
#include "common.h"

#define FA_D 128
#define FA_BLK 64              // tile rows (both q and kv tiles)
#define FA_THREADS 256         // 4 waves; wave w owns output rows [16w,16w+16)
#define FA_PAD 8               // LDS row pad (elems)

typedef float fa_floatx4 __attribute__((ext_vector_type(4)));
typedef short fa_shortx8 __attribute__((ext_vector_type(8)));

// ---------------------------------------------------------------------------
// 1) delta = rowsum(dO * O)   [B*H*S rows of D elems]
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256)
fa_bwd_preprocess(const unsigned short* __restrict__ dout,
                  const unsigned short* __restrict__ o,
                  float* __restrict__ delta, int64_t rows) {
  // 16 lanes per row (8 elems each), 16 rows per 256-thread block per
  // iteration; fully coalesced loads, shfl-xor reduce inside each 16-lane
  // group, no barriers — HBM-roofline for this pure-bandwidth pass.
  const int sub = threadIdx.x & 15;         // lane within row
  int64_t row = (int64_t)blockIdx.x * 16 + (threadIdx.x >> 4);
  const int64_t stride = (int64_t)gridDim.x * 16;
  for (; row < rows; row += stride) {
    const unsigned short* pd = dout + row * FA_D + sub * 8;
    const unsigned short* po = o + row * FA_D + sub * 8;
    fa_shortx8 a = *reinterpret_cast<const fa_shortx8*>(pd);
    fa_shortx8 b = *reinterpret_cast<const fa_shortx8*>(po);
    float s = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      s += bf16_to_f32((unsigned short)a[j]) * bf16_to_f32((unsigned short)b[j]);
#pragma unroll
    for (int off = 8; off >= 1; off >>= 1)
      s += __shfl_xor(s, off, 64);
    if (sub == 0) delta[row] = s;
  }
}

// ---------------------------------------------------------------------------
// shared tile helpers
// ---------------------------------------------------------------------------
#define ROWS_N (FA_BLK + FA_PAD)     // padded row length for [*, 64] tiles
#define ROWS_D (FA_D + FA_PAD)       // padded row length for [*, 128] tiles

// stage [FA_BLK, D] global tile as-is into LDS [FA_BLK][ROWS_D]
DEV void stage_rows(const unsigned short* g, unsigned short* lds_t, int tid) {
  // 64 x 128 elems = 8192; 256 threads x 32 elems (4 x short8)
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    int e = (tid + j * FA_THREADS) * 8;
    int r = e / FA_D, c = e % FA_D;
    *reinterpret_cast<fa_shortx8*>(lds_t + r * ROWS_D + c) =
        *reinterpret_cast<const fa_shortx8*>(g + r * FA_D + c);
  }
}

// stage [FA_BLK, D] global tile TRANSPOSED into LDS [D][ROWS_N]
DEV void stage_rows_t(const unsigned short* g, unsigned short* lds_t, int tid) {
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    int e = (tid + j * FA_THREADS) * 8;
    int r = e / FA_D, c = e % FA_D;
    fa_shortx8 v = *reinterpret_cast<const fa_shortx8*>(g + r * FA_D + c);
#pragma unroll
    for (int x = 0; x < 8; ++x) lds_t[(c + x) * ROWS_N + r] = (unsigned short)v[x];
  }
}

// operand fragment read: row-major LDS tile with row stride `stride` elems;
// frag (rf = 16-row block index, kc = 32-elem contraction chunk)
DEV fa_shortx8 frag(const unsigned short* t, int stride, int rf, int kc, int lane) {
  int row = rf * 16 + (lane & 15);
  int col = kc * 32 + (lane >> 4) * 8;
  return *reinterpret_cast<const fa_shortx8*>(t + row * stride + col);
}

// ---------------------------------------------------------------------------
// 2) dK/dV (per query head; GQA partials)
//    grid: (S/FA_BLK, Hq, B); block: 256 threads
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(FA_THREADS)
fa_bwd_dkdv(const unsigned short* __restrict__ q,
            const unsigned short* __restrict__ k,
            const unsigned short* __restrict__ v,
            const unsigned short* __restrict__ dout,
            const float* __restrict__ lse,     // [B,Hq,S]
            const float* __restrict__ delta,   // [B,Hq,S]
            unsigned short* __restrict__ dk_part,  // [B,Hq,S,D]
            unsigned short* __restrict__ dv_part,  // [B,Hq,S,D]
            int B, int Hq, int Hkv, int S, float scale) {
  // Each q-tile buffer serves two lives: rows [64][136] for the ST/dP^T
  // GEMMs, then (rewritten from the registers that staged it) transposed
  // [128][72] for the dV/dK GEMMs.  Sized for the larger layout.
  __shared__ unsigned short lq[FA_D * ROWS_N];     // Q rows -> Q^T
  __shared__ unsigned short ldo[FA_D * ROWS_N];    // dO rows -> dO^T
  __shared__ unsigned short lpt[FA_BLK * ROWS_N];  // P^T  [64][72]
  __shared__ unsigned short lds_t[FA_BLK * ROWS_N];// dS^T [64][72]
  __shared__ float lse_s[FA_BLK];
  __shared__ float dlt_s[FA_BLK];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;                 // output rows [16w, 16w+16)
  const int jkv = blockIdx.x;                // kv tile
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = h / (Hq / Hkv);

  const int64_t qbase = (((int64_t)b * Hq + h) * S) * FA_D;
  const int64_t kbase = (((int64_t)b * Hkv + hkv) * S) * FA_D;
  const int64_t lbase = ((int64_t)b * Hq + h) * S;

  fa_shortx8 kfrag[4], vfrag[4];
  {
    const unsigned short* kg = k + kbase + (int64_t)(jkv * FA_BLK) * FA_D;
    const unsigned short* vg = v + kbase + (int64_t)(jkv * FA_BLK) * FA_D;
    int row = wave * 16 + (lane & 15);
#pragma unroll
    for (int kc = 0; kc < 4; ++kc) {
      int col = kc * 32 + (lane >> 4) * 8;
      kfrag[kc] = *reinterpret_cast<const fa_shortx8*>(kg + row * FA_D + col);
      vfrag[kc] = *reinterpret_cast<const fa_shortx8*>(vg + row * FA_D + col);
    }
  }

  fa_floatx4 accK[8], accV[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    accK[i] = (fa_floatx4){0.f, 0.f, 0.f, 0.f};
    accV[i] = (fa_floatx4){0.f, 0.f, 0.f, 0.f};
  }

  const int n_qtiles = S / FA_BLK;
  for (int iq = jkv; iq < n_qtiles; ++iq) {
    const unsigned short* qg = q + qbase + (int64_t)(iq * FA_BLK) * FA_D;
    const unsigned short* dog = dout + qbase + (int64_t)(iq * FA_BLK) * FA_D;
    // stage rows from global ONCE; keep values in registers for the later
    // in-LDS transpose
    fa_shortx8 regq[4], regdo[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int e = (tid + j * FA_THREADS) * 8;
      int r = e / FA_D, c = e % FA_D;
      regq[j] = *reinterpret_cast<const fa_shortx8*>(qg + r * FA_D + c);
      regdo[j] = *reinterpret_cast<const fa_shortx8*>(dog + r * FA_D + c);
      *reinterpret_cast<fa_shortx8*>(lq + r * ROWS_D + c) = regq[j];
      *reinterpret_cast<fa_shortx8*>(ldo + r * ROWS_D + c) = regdo[j];
    }
    if (tid < FA_BLK) {
      lse_s[tid] = lse[lbase + iq * FA_BLK + tid];
      dlt_s[tid] = delta[lbase + iq * FA_BLK + tid];
    }
    __syncthreads();

    // ST[kv][q] = K . Q^T; dPT[kv][q] = V . dO^T  (Y operands = rows)
    fa_floatx4 st[4], dpt[4];
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      st[f] = (fa_floatx4){0.f, 0.f, 0.f, 0.f};
      dpt[f] = (fa_floatx4){0.f, 0.f, 0.f, 0.f};
    }
#pragma unroll
    for (int kc = 0; kc < 4; ++kc) {
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        fa_shortx8 bq = frag(lq, ROWS_D, f, kc, lane);
        st[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(kfrag[kc], bq, st[f], 0, 0, 0);
        fa_shortx8 bd = frag(ldo, ROWS_D, f, kc, lane);
        dpt[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(vfrag[kc], bd, dpt[f], 0, 0, 0);
      }
    }

    // P^T = exp(ST*scale - lse[q]); dS^T = P^T * (dPT - delta[q]) * scale
    const int kvrow = jkv * FA_BLK + wave * 16 + 4 * (lane >> 4);
    __syncthreads();  // rows layouts fully consumed before the rewrite
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      int qcol = f * 16 + (lane & 15);
      float l = lse_s[qcol];
      float dl = dlt_s[qcol];
      int qidx = iq * FA_BLK + qcol;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        bool valid = qidx >= (kvrow + r);
        float p = valid ? __expf(st[f][r] * scale - l) : 0.f;
        float ds = p * (dpt[f][r] - dl) * scale;
        int row = wave * 16 + 4 * (lane >> 4) + r;
        lpt[row * ROWS_N + qcol] = f32_to_bf16(p);
        lds_t[row * ROWS_N + qcol] = f32_to_bf16(ds);
      }
    }
    // rewrite lq/ldo as TRANSPOSED [128][72] from the staging registers
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int e = (tid + j * FA_THREADS) * 8;
      int r = e / FA_D, c = e % FA_D;
#pragma unroll
      for (int x = 0; x < 8; ++x) {
        lq[(c + x) * ROWS_N + r] = (unsigned short)regq[j][x];
        ldo[(c + x) * ROWS_N + r] = (unsigned short)regdo[j][x];
      }
    }
    __syncthreads();

    // dV += P^T . dO (Y = dO^T); dK += dS^T . Q (Y = Q^T)
#pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      fa_shortx8 xp = frag(lpt, ROWS_N, wave, kc, lane);
      fa_shortx8 xs = frag(lds_t, ROWS_N, wave, kc, lane);
#pragma unroll
      for (int f = 0; f < 8; ++f) {
        fa_shortx8 yd = frag(ldo, ROWS_N, f, kc, lane);
        accV[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(xp, yd, accV[f], 0, 0, 0);
        fa_shortx8 yq = frag(lq, ROWS_N, f, kc, lane);
        accK[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(xs, yq, accK[f], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  unsigned short* dkg = dk_part + qbase + (int64_t)(jkv * FA_BLK) * FA_D;
  unsigned short* dvg = dv_part + qbase + (int64_t)(jkv * FA_BLK) * FA_D;
#pragma unroll
  for (int f = 0; f < 8; ++f) {
    int col = f * 16 + (lane & 15);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = wave * 16 + 4 * (lane >> 4) + r;
      dkg[(int64_t)row * FA_D + col] = f32_to_bf16(accK[f][r]);
      dvg[(int64_t)row * FA_D + col] = f32_to_bf16(accV[f][r]);
    }
  }
}

// ---------------------------------------------------------------------------
// 3) dQ   grid: (S/FA_BLK, Hq, B)
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(FA_THREADS)
fa_bwd_dq(const unsigned short* __restrict__ q,
          const unsigned short* __restrict__ k,
          const unsigned short* __restrict__ v,
          const unsigned short* __restrict__ dout,
          const float* __restrict__ lse,
          const float* __restrict__ delta,
          unsigned short* __restrict__ dq,
          int B, int Hq, int Hkv, int S, float scale) {
  __shared__ unsigned short lk[FA_D * ROWS_N];     // K rows -> K^T
  __shared__ unsigned short lv[FA_BLK * ROWS_D];   // V rows [64][136]
  __shared__ unsigned short lds_s[FA_BLK * ROWS_N];// dS [64q][72]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int iq = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = h / (Hq / Hkv);

  const int64_t qbase = (((int64_t)b * Hq + h) * S) * FA_D;
  const int64_t kbase = (((int64_t)b * Hkv + hkv) * S) * FA_D;
  const int64_t lbase = ((int64_t)b * Hq + h) * S;

  fa_shortx8 qfrag[4], dofrag[4];
  {
    const unsigned short* qg = q + qbase + (int64_t)(iq * FA_BLK) * FA_D;
    const unsigned short* dog = dout + qbase + (int64_t)(iq * FA_BLK) * FA_D;
    int row = wave * 16 + (lane & 15);
#pragma unroll
    for (int kc = 0; kc < 4; ++kc) {
      int col = kc * 32 + (lane >> 4) * 8;
      qfrag[kc] = *reinterpret_cast<const fa_shortx8*>(qg + row * FA_D + col);
      dofrag[kc] = *reinterpret_cast<const fa_shortx8*>(dog + row * FA_D + col);
    }
  }
  float lse_r[4], dlt_r[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int row = iq * FA_BLK + wave * 16 + 4 * (lane >> 4) + r;
    lse_r[r] = lse[lbase + row];
    dlt_r[r] = delta[lbase + row];
  }

  fa_floatx4 accQ[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) accQ[i] = (fa_floatx4){0.f, 0.f, 0.f, 0.f};

  for (int jkv = 0; jkv <= iq; ++jkv) {
    const unsigned short* kg = k + kbase + (int64_t)(jkv * FA_BLK) * FA_D;
    const unsigned short* vg = v + kbase + (int64_t)(jkv * FA_BLK) * FA_D;
    fa_shortx8 regk[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int e = (tid + j * FA_THREADS) * 8;
      int r = e / FA_D, c = e % FA_D;
      regk[j] = *reinterpret_cast<const fa_shortx8*>(kg + r * FA_D + c);
      *reinterpret_cast<fa_shortx8*>(lk + r * ROWS_D + c) = regk[j];
      *reinterpret_cast<fa_shortx8*>(lv + r * ROWS_D + c) =
          *reinterpret_cast<const fa_shortx8*>(vg + r * FA_D + c);
    }
    __syncthreads();

    // S[q][kv] = Q . K^T (Y = K rows); dP[q][kv] = dO . V^T (Y = V rows)
    fa_floatx4 sacc[4], dp[4];
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      sacc[f] = (fa_floatx4){0.f, 0.f, 0.f, 0.f};
      dp[f] = (fa_floatx4){0.f, 0.f, 0.f, 0.f};
    }
#pragma unroll
    for (int kc = 0; kc < 4; ++kc) {
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        fa_shortx8 yk = frag(lk, ROWS_D, f, kc, lane);
        sacc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[kc], yk, sacc[f], 0, 0, 0);
        fa_shortx8 yv = frag(lv, ROWS_D, f, kc, lane);
        dp[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dofrag[kc], yv, dp[f], 0, 0, 0);
      }
    }
    __syncthreads();  // rows consumed; lk may be rewritten transposed

    // dS[q][kv] = P * (dP - delta[q]) * scale
    const int qrow0 = iq * FA_BLK + wave * 16 + 4 * (lane >> 4);
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      int kvcol = jkv * FA_BLK + f * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        bool valid = (qrow0 + r) >= kvcol;
        float p = valid ? __expf(sacc[f][r] * scale - lse_r[r]) : 0.f;
        float ds = p * (dp[f][r] - dlt_r[r]) * scale;
        int row = wave * 16 + 4 * (lane >> 4) + r;
        lds_s[row * ROWS_N + f * 16 + (lane & 15)] = f32_to_bf16(ds);
      }
    }
    // rewrite lk as K^T [128][72] from the staging registers
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int e = (tid + j * FA_THREADS) * 8;
      int r = e / FA_D, c = e % FA_D;
#pragma unroll
      for (int x = 0; x < 8; ++x) lk[(c + x) * ROWS_N + r] = (unsigned short)regk[j][x];
    }
    __syncthreads();

    // dQ += dS . K (Y = K^T)
#pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      fa_shortx8 xs = frag(lds_s, ROWS_N, wave, kc, lane);
#pragma unroll
      for (int f = 0; f < 8; ++f) {
        fa_shortx8 yk = frag(lk, ROWS_N, f, kc, lane);
        accQ[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(xs, yk, accQ[f], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  unsigned short* dqg = dq + qbase + (int64_t)(iq * FA_BLK) * FA_D;
#pragma unroll
  for (int f = 0; f < 8; ++f) {
    int col = f * 16 + (lane & 15);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = wave * 16 + 4 * (lane >> 4) + r;
      dqg[(int64_t)row * FA_D + col] = f32_to_bf16(accQ[f][r]);
    }
  }
}

// ---------------------------------------------------------------------------
// 4) reduce GQA partials: dk[b,hkv,s,d] = sum over group of dk_part[b,h,s,d]
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256)
fa_bwd_reduce_gqa(const unsigned short* __restrict__ part,  // [B,Hq,S,D]
                  unsigned short* __restrict__ out,         // [B,Hkv,S,D]
                  int B, int Hq, int Hkv, int64_t SD) {
  int group = Hq / Hkv;
  int64_t total = (int64_t)B * Hkv * SD;
  int64_t i0 = (int64_t)(blockIdx.x * 256 + threadIdx.x) * 8;
  int64_t stride = (int64_t)gridDim.x * 256 * 8;
  for (int64_t i = i0; i < total; i += stride) {
    int64_t bh = i / SD;
    int64_t off = i % SD;
    int64_t b = bh / Hkv;
    int64_t hk = bh % Hkv;
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll 4
    for (int g = 0; g < group; ++g) {
      const unsigned short* p =
          part + ((b * Hq + hk * group + g) * SD) + off;
      fa_shortx8 v = *reinterpret_cast<const fa_shortx8*>(p);
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] += bf16_to_f32((unsigned short)v[j]);
    }
    fa_shortx8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = (short)f32_to_bf16(acc[j]);
    *reinterpret_cast<fa_shortx8*>(out + i) = o;
  }
}
