// Flash-attention BACKWARD v2 — CDNA4 MFMA 32x32x16, causal, GQA, D=128.
//
// Replaces the 16x16-MFMA attention_bwd.hip hot kernels with the patterns
// proven out in attention_fwd.hip (same helpers: ff_kswz staging swizzle,
// ff_swap_other permlane relayout, 4x4-group ds_read_b64_tr_b16 transpose
// reads — all probe-verified on hardware, tools/tr_probe_check.py):
//
//   fa2_dq : grid over q-tiles (fwd-shaped; wave owns 32 q rows).
//            Per 64-kv tile:
//              ST' = mfma(A=K_lds, B=Q_regs)          -> D[kv][q], col=q
//              dP' = mfma(A=V_lds, B=dO_lds B-frags)  -> D[kv][q]
//              P'  = exp2(st*scale*log2e - lse2[q])   (lse known: no online
//                    max bookkeeping; masked entries -> exp2(-inf) = 0)
//              dS' = P' * (dP' - delta[q])            (lane-local, col=q)
//              dQacc[d][q] += mfma(A=K^T tr-frags, B=relayout(dS'))
//            Epilogue dQ[q][d] = scale * acc — the fwd's O-epilogue shape.
//   fa2_dv : grid over kv-tiles (wave owns 32 kv rows). Per 64-q tile
//            (diagonal..S):
//              ST = mfma(A=Q_lds, B=K_regs)           -> D[q][kv], col=kv
//              P  = exp2(st*scale2 - lse2[row q])     (LDS broadcast reads)
//              dVacc[d][kv] += mfma(A=dO^T tr-frags, B=relayout(P))
//   fa2_dk : same grid; adds
//              dP = mfma(A=dO_lds, B=V_regs)          -> D[q][kv]
//              dS = P * (dP - delta[row q])
//              dKacc[d][kv] += mfma(A=Q^T tr-frags, B=relayout(dS))
//            Epilogue dK[kv][d] = scale * acc.
//
// dV/dK write per-Hq partials; the existing fa_bwd_reduce_gqa kernel sums
// the GQA groups, and fa_bwd_preprocess provides delta = rowsum(dO*O).
//
// Reference parity: the reference framework uses the stock flash kernels;
// this replaces the train step's heaviest kernel (the library backward is
// 30% of the Llama-8B step — profiles/bench_kernel_stats_r1_final.txt).
#include "common.h"

#define FB_D 128
#define FB_OWN 32             // rows owned per wave (q for dq; kv for dv/dk)
#define FB_WAVES 8
#define FB_TILE (FB_OWN * FB_WAVES)  // 128 own-rows per block
#define FB_OTH 64             // other-side tile (kv for dq; q for dv/dk)
#define FB_THREADS (FB_WAVES * 64)
#define FB_LOG2E 1.44269504088896340736f

typedef float fb_floatx16 __attribute__((ext_vector_type(16)));
typedef short fb_shortx4 __attribute__((ext_vector_type(4)));
typedef short fb_shortx8 __attribute__((ext_vector_type(8)));
typedef int fb_intx4 __attribute__((ext_vector_type(4)));

// stage a 64x128 bf16 tile row-major + ff_kswz swizzle (block-cooperative)
#define FB_CHUNKS (FB_OTH * FB_D / (FB_THREADS * 8))
DEV void fb_stage64(const unsigned short* __restrict__ g,
                    unsigned short* l, int tid) {
#pragma unroll
  for (int j = 0; j < FB_CHUNKS; ++j) {
    int e = (tid + j * FB_THREADS) * 8;
    fb_shortx8 v = *reinterpret_cast<const fb_shortx8*>(g + e);
    *reinterpret_cast<fb_shortx8*>((char*)l + ff_kswz(e * 2)) = v;
  }
}

// A-operand fragment: rows s2*32 + l31 of a swizzled [*x128] LDS tile,
// k = ks*16 + half*8 + e
DEV fb_shortx8 fb_afrag(const unsigned short* l, int s2, int ks, int l31,
                        int half) {
  int byte = ((s2 * 32 + l31) * FB_D + ks * 16 + half * 8) * 2;
  return *reinterpret_cast<const fb_shortx8*>((const char*)l + ff_kswz(byte));
}

// B-operand fragment of row `row` from a swizzled LDS tile:
// B[j=row][k = ks*16 + half*8 + e] (row index may exceed 64; swizzle is
// row-periodic mod 8 so any row count works)
DEV fb_shortx8 fb_bfrag(const unsigned short* l, int row, int ks, int half) {
  int byte = (row * FB_D + ks * 16 + half * 8) * 2;
  return *reinterpret_cast<const fb_shortx8*>((const char*)l + ff_kswz(byte));
}

// relayout a D-layout [rows][cols=l31] fp32 register block (2 subtiles of
// 32 rows) into 4 operand fragments with lane-dim = former cols and
// contraction = former rows — identical to the fwd's P->pb relayout.
DEV void fb_relayout(const fb_floatx16 st[2], fb_shortx8 pb[4], int half) {
#pragma unroll
  for (int s2 = 0; s2 < 2; ++s2) {
#pragma unroll
    for (int g = 0; g < 2; ++g) {
      unsigned lo0, lo1, hi0, hi1;
      asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(lo0)
                   : "v"(st[s2][g * 8 + 0]), "v"(st[s2][g * 8 + 1]));
      asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(lo1)
                   : "v"(st[s2][g * 8 + 2]), "v"(st[s2][g * 8 + 3]));
      asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(hi0)
                   : "v"(st[s2][g * 8 + 4]), "v"(st[s2][g * 8 + 5]));
      asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(hi1)
                   : "v"(st[s2][g * 8 + 6]), "v"(st[s2][g * 8 + 7]));
      int p0 = ff_swap_other((int)lo0, half);
      int p1 = ff_swap_other((int)lo1, half);
      int p2 = ff_swap_other((int)hi0, half);
      int p3 = ff_swap_other((int)hi1, half);
      fb_intx4 frag;
      if (half == 0) {
        frag = (fb_intx4){(int)lo0, (int)lo1, p0, p1};
      } else {
        frag = (fb_intx4){p2, p3, (int)hi0, (int)hi1};
      }
      pb[s2 * 2 + g] = *reinterpret_cast<fb_shortx8*>(&frag);
    }
  }
}

// 8 tr reads (4 ks x 2) of a transposed operand from a swizzled 64x128
// tile: fragment[lane-dim = ds*32 + l31][k = ks*16 + half*8 + e] — the
// fwd's probe-verified 4x4-group transpose addressing.
#define FB_TR_READS(lsrc, t, ds)                                            \
  {                                                                         \
    const int col2 = ((ds) * 32 + (l31 & 16) + (lane & 3) * 4) * 2;         \
    const int row0 = half * 8 + ((lane & 15) >> 2);                         \
    const int a0 = (row0 * 256 + col2) ^ ((row0 & 7) << 4);                 \
    const int row1 = row0 + 4;                                              \
    const int a1 = (row1 * 256 + col2) ^ ((row1 & 7) << 4);                 \
    ff_lds_p b0 = (ff_lds_p)((const char*)(lsrc) + a0);                     \
    ff_lds_p b1 = (ff_lds_p)((const char*)(lsrc) + a1);                     \
    /* single asm: 8 reads + drain (see attention_fwd.hip note on the   \
       interposed-copy race; earlyclobber keeps dsts off b0/b1) */         \
    asm volatile(                                                           \
        "ds_read_b64_tr_b16 %0, %8 offset:0\n\t"                           \
        "ds_read_b64_tr_b16 %1, %9 offset:0\n\t"                           \
        "ds_read_b64_tr_b16 %2, %8 offset:4096\n\t"                        \
        "ds_read_b64_tr_b16 %3, %9 offset:4096\n\t"                        \
        "ds_read_b64_tr_b16 %4, %8 offset:8192\n\t"                        \
        "ds_read_b64_tr_b16 %5, %9 offset:8192\n\t"                        \
        "ds_read_b64_tr_b16 %6, %8 offset:12288\n\t"                       \
        "ds_read_b64_tr_b16 %7, %9 offset:12288\n\t"                       \
        "s_waitcnt lgkmcnt(0)"                                              \
        : "=&v"(t[0][0]), "=&v"(t[0][1]), "=&v"(t[1][0]), "=&v"(t[1][1]),   \
          "=&v"(t[2][0]), "=&v"(t[2][1]), "=&v"(t[3][0]), "=&v"(t[3][1])    \
        : "v"(b0), "v"(b1));                                                \
  }

DEV fb_shortx8 fb_cat(const fb_shortx4 a, const fb_shortx4 b) {
  fb_shortx8 r;
#pragma unroll
  for (int x = 0; x < 4; ++x) { r[x] = a[x]; r[4 + x] = b[x]; }
  return r;
}

// ---------------------------------------------------------------------------
// fa2_dq: dQ = scale * K^T-contract(dS'); grid (S/128, Hq, B)
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(FB_THREADS, 2)
fa2_dq_bf16(const unsigned short* __restrict__ q,
            const unsigned short* __restrict__ k,
            const unsigned short* __restrict__ v,
            const unsigned short* __restrict__ dout,
            const float* __restrict__ lse,
            const float* __restrict__ delta,
            unsigned short* __restrict__ dq,
            int B, int Hq, int Hkv, int S, float scale) {
  __shared__ unsigned short lk[FB_OTH * FB_D];    // K kv-tile (swz)
  __shared__ unsigned short lv[FB_OTH * FB_D];    // V kv-tile (swz)
  __shared__ unsigned short ldo[FB_TILE * FB_D];  // block's dO q rows (swz)

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int l31 = lane & 31;
  const int half = lane >> 5;

  int bx = blockIdx.x, by = blockIdx.y, bz = blockIdx.z;
  ff_xcd_remap(bx, by, bz);
  const int qt = bx;
  const int h = by;
  const int b = bz;
  const int hkv = h / (Hq / Hkv);
  const int64_t qbase = (((int64_t)b * Hq + h) * S) * FB_D;
  const int64_t kbase = (((int64_t)b * Hkv + hkv) * S) * FB_D;
  const int q0 = qt * FB_TILE + wave * FB_OWN;
  const int my_q = q0 + l31;
  const float scale2 = scale * FB_LOG2E;

  // lane-local per-q constants (col = q = l31)
  const int64_t lrow = ((int64_t)b * Hq + h) * S + my_q;
  const float lse2 = lse[lrow] * FB_LOG2E;
  const float dlt = delta[lrow];

  // Q fragments (B-operand, like the fwd's qf)
  fb_shortx8 qf[8];
  {
    const unsigned short* qg = q + qbase + (int64_t)my_q * FB_D;
#pragma unroll
    for (int ks = 0; ks < 8; ++ks)
      qf[ks] = *reinterpret_cast<const fb_shortx8*>(qg + ks * 16 + half * 8);
  }
  // stage the block's FB_TILE dO rows once (swizzled)
  {
    const unsigned short* dg = dout + qbase + (int64_t)qt * FB_TILE * FB_D;
#pragma unroll
    for (int j = 0; j < FB_TILE * FB_D / (FB_THREADS * 8); ++j) {
      int e = (tid + j * FB_THREADS) * 8;
      fb_shortx8 vv = *reinterpret_cast<const fb_shortx8*>(dg + e);
      *reinterpret_cast<fb_shortx8*>((char*)ldo + ff_kswz(e * 2)) = vv;
    }
  }

  fb_floatx16 dqacc[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) dqacc[i] = (fb_floatx16)(0.f);

  const int last_kv = min((qt + 1) * (FB_TILE / FB_OTH), S / FB_OTH);
  const unsigned short* kg = k + kbase;
  const unsigned short* vg = v + kbase;

  for (int jkv = 0; jkv < last_kv; ++jkv) {
    const int kv0 = jkv * FB_OTH;
    const bool active = (kv0 <= q0 + FB_OWN - 1);
    const bool need_mask = (kv0 + FB_OTH - 1 > q0);
    __syncthreads();  // previous tile's readers done
    fb_stage64(kg + (int64_t)kv0 * FB_D, lk, tid);
    fb_stage64(vg + (int64_t)kv0 * FB_D, lv, tid);
    __syncthreads();

    if (active) {
      // ST' = K.Q^T and dP' = V.dO^T — both D[kv][q], col = q = l31
      fb_floatx16 st[2], dp[2];
#pragma unroll
      for (int s2 = 0; s2 < 2; ++s2) {
        st[s2] = (fb_floatx16)(0.f);
        dp[s2] = (fb_floatx16)(0.f);
      }
#pragma unroll
      for (int s2 = 0; s2 < 2; ++s2)
#pragma unroll
        for (int ks = 0; ks < 8; ++ks) {
          fb_shortx8 kfr = fb_afrag(lk, s2, ks, l31, half);
          st[s2] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kfr, qf[ks], st[s2], 0, 0, 0);
          fb_shortx8 vfr = fb_afrag(lv, s2, ks, l31, half);
          fb_shortx8 dof = fb_bfrag(ldo, wave * FB_OWN + l31, ks, half);
          dp[s2] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vfr, dof, dp[s2], 0, 0, 0);
        }
      // dS' = P' * (dP' - delta); P' = exp2(st*scale2 - lse2); mask kv > q
#pragma unroll
      for (int s2 = 0; s2 < 2; ++s2)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int kv = kv0 + s2 * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
          float x = st[s2][r] * scale2 - lse2;
          if (need_mask && kv > my_q) x = -INFINITY;
          float p = __builtin_amdgcn_exp2f(x);
          st[s2][r] = p * (dp[s2][r] - dlt);
        }
      fb_shortx8 pb[4];
      fb_relayout(st, pb, half);  // B-frags [q][kv-contraction]
      // dQacc[d][q] += K^T . dS' (fwd's PV shape: A via tr, B = relayout)
#pragma unroll
      for (int ds = 0; ds < 4; ++ds) {
        fb_shortx4 t[4][2];
        FB_TR_READS(lk, t, ds);
#pragma unroll
        for (int ks = 0; ks < 4; ++ks)
          dqacc[ds] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              fb_cat(t[ks][0], t[ks][1]), pb[ks], dqacc[ds], 0, 0, 0);
      }
    }
  }

  // epilogue: dQ[q][d] = scale * acc (D-layout: rows = d, col = q = l31)
  unsigned short* og = dq + qbase + (int64_t)my_q * FB_D;
#pragma unroll
  for (int ds = 0; ds < 4; ++ds)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int d = ds * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
      og[d] = f32_to_bf16(dqacc[ds][r] * scale);
    }
}

// ---------------------------------------------------------------------------
// fa2_dv: dV[kv][d] = sum_q P dO ; grid (S/128 kv-tiles, Hq, B); Hq partials
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(FB_THREADS, 2)
fa2_dv_bf16(const unsigned short* __restrict__ q,
            const unsigned short* __restrict__ k,
            const unsigned short* __restrict__ dout,
            const float* __restrict__ lse,
            unsigned short* __restrict__ dv_part,
            int B, int Hq, int Hkv, int S, float scale) {
  __shared__ unsigned short lq[FB_OTH * FB_D];    // Q q-tile (swz)
  __shared__ unsigned short ldo[FB_OTH * FB_D];   // dO q-tile (swz)
  __shared__ float llse[FB_OTH];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int l31 = lane & 31;
  const int half = lane >> 5;

  int bx = blockIdx.x, by = blockIdx.y, bz = blockIdx.z;
  ff_xcd_remap(bx, by, bz);
  const int kt = bx;
  const int h = by;
  const int b = bz;
  const int hkv = h / (Hq / Hkv);
  const int64_t qbase = (((int64_t)b * Hq + h) * S) * FB_D;
  const int64_t kbase = (((int64_t)b * Hkv + hkv) * S) * FB_D;
  const int kvB = kt * FB_TILE;
  const int kv0w = kvB + wave * FB_OWN;
  const int my_kv = kv0w + l31;
  const float scale2 = scale * FB_LOG2E;

  // K fragments (B-operand): lane holds K[my_kv][ks*16 + half*8 + e]
  fb_shortx8 kf[8];
  {
    const unsigned short* kg = k + kbase + (int64_t)my_kv * FB_D;
#pragma unroll
    for (int ks = 0; ks < 8; ++ks)
      kf[ks] = *reinterpret_cast<const fb_shortx8*>(kg + ks * 16 + half * 8);
  }

  fb_floatx16 dvacc[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) dvacc[i] = (fb_floatx16)(0.f);

  const int jq0 = kvB / FB_OTH;       // diagonal q tile, block-uniform
  const int n_jq = S / FB_OTH;
  const float* lseg = lse + ((int64_t)b * Hq + h) * S;

  for (int jq = jq0; jq < n_jq; ++jq) {
    const int qt0 = jq * FB_OTH;
    const bool active = (qt0 + FB_OTH - 1 >= kv0w);
    const bool need_mask = (qt0 < kv0w + FB_OWN - 1);
    __syncthreads();
    fb_stage64(q + qbase + (int64_t)qt0 * FB_D, lq, tid);
    fb_stage64(dout + qbase + (int64_t)qt0 * FB_D, ldo, tid);
    if (tid < FB_OTH) llse[tid] = lseg[qt0 + tid] * FB_LOG2E;
    __syncthreads();

    if (active) {
      // ST = Q.K^T -> D[q][kv], col = kv = l31
      fb_floatx16 st[2];
#pragma unroll
      for (int s2 = 0; s2 < 2; ++s2) st[s2] = (fb_floatx16)(0.f);
#pragma unroll
      for (int s2 = 0; s2 < 2; ++s2)
#pragma unroll
        for (int ks = 0; ks < 8; ++ks) {
          fb_shortx8 qfr = fb_afrag(lq, s2, ks, l31, half);
          st[s2] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qfr, kf[ks], st[s2], 0, 0, 0);
        }
      // P = exp2(st*scale2 - lse2[q row]); mask q < kv (broadcast reads)
#pragma unroll
      for (int s2 = 0; s2 < 2; ++s2)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int qrow = s2 * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
          float x = st[s2][r] * scale2 - llse[qrow];
          if (need_mask && qt0 + qrow < my_kv) x = -INFINITY;
          st[s2][r] = __builtin_amdgcn_exp2f(x);
        }
      fb_shortx8 pb[4];
      fb_relayout(st, pb, half);  // B-frags [kv][q-contraction]
      // dVacc[d][kv] += dO^T . P (A via tr reads of dO, B = relayout(P))
#pragma unroll
      for (int ds = 0; ds < 4; ++ds) {
        fb_shortx4 t[4][2];
        FB_TR_READS(ldo, t, ds);
#pragma unroll
        for (int ks = 0; ks < 4; ++ks)
          dvacc[ds] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              fb_cat(t[ks][0], t[ks][1]), pb[ks], dvacc[ds], 0, 0, 0);
      }
    }
  }

  // epilogue: dV[kv][d] (rows = d over regs, col = kv = l31 -> row write)
  unsigned short* og = dv_part + qbase + (int64_t)my_kv * FB_D;
#pragma unroll
  for (int ds = 0; ds < 4; ++ds)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int d = ds * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
      og[d] = f32_to_bf16(dvacc[ds][r]);
    }
}

// ---------------------------------------------------------------------------
// fa2_dvdk: fused dV + dK.  The two split kernels share the Q/dO staging,
// the ST = Q.K^T GEMM, the lse/delta loads and a barrier pair per q-tile;
// fusing removes that duplicated work plus one kernel launch.
//
// Geometry: 4 waves / 256 threads, 128-kv-row block tile (HALF the split
// kernels' 8-wave/256-row blocks).  Reason: a wave needs two fp32
// accumulator banks (dV, dK: 64 VGPRs each) + st/dp + operands ~ 320 regs.
// A gfx950 wave can hold up to 512 (256 arch VGPRs + AGPR overflow) but
// only at 1 wave/SIMD — so an 8-wave block (which forces >= 2 waves/SIMD)
// pins the budget at 256 and spills 264-484 B/lane to scratch (measured
// via -Rpass-analysis).  At 4 waves/block the block fits one-per-CU with
// each wave on its own SIMD and the full 512-reg budget.
//
// K and V live in LDS (staged ONCE - the block's own kv-tile never
// changes), read as B-fragments with fb_bfrag, the same pattern fa2_dq
// uses for dO.  lse/delta travel as one wave register each (lane i holds
// row qt0+i) broadcast per-element with ds_bpermute.  st[] is rewritten in
// place P -> dS between the two MFMA phases so relayout/tr temporaries are
// shared.  LDS 96 KB/block.  Replaces fa2_dv_bf16 + fa2_dk_bf16 (kept
// above for A/B).
// ---------------------------------------------------------------------------
#define FD_WAVES 4
#define FD_THREADS (FD_WAVES * 64)
#define FD_TILE (FD_WAVES * FB_OWN)  // 128 kv rows per block

// stage a 64x128 bf16 tile with 256 threads (fb_stage64 assumes 512)
DEV void fd_stage64(const unsigned short* __restrict__ g,
                    unsigned short* l, int tid) {
#pragma unroll
  for (int j = 0; j < FB_OTH * FB_D / (FD_THREADS * 8); ++j) {
    int e = (tid + j * FD_THREADS) * 8;
    fb_shortx8 v = *reinterpret_cast<const fb_shortx8*>(g + e);
    *reinterpret_cast<fb_shortx8*>((char*)l + ff_kswz(e * 2)) = v;
  }
}

extern "C" __global__ void __launch_bounds__(FD_THREADS, 1)
fa2_dvdk_bf16(const unsigned short* __restrict__ q,
              const unsigned short* __restrict__ k,
              const unsigned short* __restrict__ v,
              const unsigned short* __restrict__ dout,
              const float* __restrict__ lse,
              const float* __restrict__ delta,
              unsigned short* __restrict__ dv_part,
              unsigned short* __restrict__ dk_part,
              int B, int Hq, int Hkv, int S, float scale) {
  __shared__ unsigned short lq[FB_OTH * FB_D];    // Q q-tile (swz)
  __shared__ unsigned short ldo[FB_OTH * FB_D];   // dO q-tile (swz)
  __shared__ unsigned short lk[FD_TILE * FB_D];   // block's K kv rows (swz)
  __shared__ unsigned short lv[FD_TILE * FB_D];   // block's V kv rows (swz)

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int l31 = lane & 31;
  const int half = lane >> 5;

  int bx = blockIdx.x, by = blockIdx.y, bz = blockIdx.z;
  ff_xcd_remap(bx, by, bz);
  const int kt = bx;
  const int h = by;
  const int b = bz;
  const int hkv = h / (Hq / Hkv);
  const int64_t qbase = (((int64_t)b * Hq + h) * S) * FB_D;
  const int64_t kbase = (((int64_t)b * Hkv + hkv) * S) * FB_D;
  const int kvB = kt * FD_TILE;
  const int kv0w = kvB + wave * FB_OWN;
  const int my_kv = kv0w + l31;
  const int myrow = wave * FB_OWN + l31;  // my kv row within the block tile
  const float scale2 = scale * FB_LOG2E;

  // stage the block's K/V kv rows once (swizzled, block-cooperative)
  {
    const unsigned short* kg = k + kbase + (int64_t)kvB * FB_D;
    const unsigned short* vg = v + kbase + (int64_t)kvB * FB_D;
#pragma unroll
    for (int j = 0; j < FD_TILE * FB_D / (FD_THREADS * 8); ++j) {
      int e = (tid + j * FD_THREADS) * 8;
      fb_shortx8 kk = *reinterpret_cast<const fb_shortx8*>(kg + e);
      *reinterpret_cast<fb_shortx8*>((char*)lk + ff_kswz(e * 2)) = kk;
      fb_shortx8 vv = *reinterpret_cast<const fb_shortx8*>(vg + e);
      *reinterpret_cast<fb_shortx8*>((char*)lv + ff_kswz(e * 2)) = vv;
    }
  }

  fb_floatx16 dvacc[4], dkacc[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    dvacc[i] = (fb_floatx16)(0.f);
    dkacc[i] = (fb_floatx16)(0.f);
  }

  const int jq0 = kvB / FB_OTH;
  const int n_jq = S / FB_OTH;
  const float* lseg = lse + ((int64_t)b * Hq + h) * S;
  const float* dltg = delta + ((int64_t)b * Hq + h) * S;

  for (int jq = jq0; jq < n_jq; ++jq) {
    const int qt0 = jq * FB_OTH;
    const bool active = (qt0 + FB_OTH - 1 >= kv0w);
    const bool need_mask = (qt0 < kv0w + FB_OWN - 1);
    __syncthreads();  // previous tile readers done; first iter: K/V staged
    fd_stage64(q + qbase + (int64_t)qt0 * FB_D, lq, tid);
    fd_stage64(dout + qbase + (int64_t)qt0 * FB_D, ldo, tid);
    const float lse_reg = lseg[qt0 + lane] * FB_LOG2E;
    const float dlt_reg = dltg[qt0 + lane];
    __syncthreads();

    if (active) {
      // ST = Q.K^T and dP = dO.V^T in one pass — both D[q][kv], col = kv
      fb_floatx16 st[2], dp[2];
#pragma unroll
      for (int s2 = 0; s2 < 2; ++s2) {
        st[s2] = (fb_floatx16)(0.f);
        dp[s2] = (fb_floatx16)(0.f);
      }
#pragma unroll
      for (int s2 = 0; s2 < 2; ++s2)
#pragma unroll
        for (int ks = 0; ks < 8; ++ks) {
          fb_shortx8 kfr = fb_bfrag(lk, myrow, ks, half);
          fb_shortx8 qfr = fb_afrag(lq, s2, ks, l31, half);
          st[s2] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qfr, kfr, st[s2], 0, 0, 0);
          fb_shortx8 vfr = fb_bfrag(lv, myrow, ks, half);
          fb_shortx8 dofr = fb_afrag(ldo, s2, ks, l31, half);
          dp[s2] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dofr, vfr, dp[s2], 0, 0, 0);
        }
      // P = exp2(st*scale2 - lse2[q row]) in place (mask q < kv)
#pragma unroll
      for (int s2 = 0; s2 < 2; ++s2)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int qrow = s2 * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
          float l2 = __int_as_float(
              __builtin_amdgcn_ds_bpermute(qrow * 4, __float_as_int(lse_reg)));
          float x = st[s2][r] * scale2 - l2;
          if (need_mask && qt0 + qrow < my_kv) x = -INFINITY;
          st[s2][r] = __builtin_amdgcn_exp2f(x);
        }
      // dV phase: dVacc[d][kv] += dO^T . P
      {
        fb_shortx8 pb[4];
        fb_relayout(st, pb, half);  // B-frags [kv][q-contraction]
#pragma unroll
        for (int ds = 0; ds < 4; ++ds) {
          fb_shortx4 t[4][2];
          FB_TR_READS(ldo, t, ds);
#pragma unroll
          for (int ks = 0; ks < 4; ++ks)
            dvacc[ds] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                fb_cat(t[ks][0], t[ks][1]), pb[ks], dvacc[ds], 0, 0, 0);
        }
      }
      // dS = P * (dP - delta[q row]) in place, then dK phase
#pragma unroll
      for (int s2 = 0; s2 < 2; ++s2)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int qrow = s2 * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
          float dl = __int_as_float(
              __builtin_amdgcn_ds_bpermute(qrow * 4, __float_as_int(dlt_reg)));
          st[s2][r] = st[s2][r] * (dp[s2][r] - dl);
        }
      {
        fb_shortx8 pb[4];
        fb_relayout(st, pb, half);  // B-frags [kv][q-contraction]
        // dKacc[d][kv] += Q^T . dS
#pragma unroll
        for (int ds = 0; ds < 4; ++ds) {
          fb_shortx4 t[4][2];
          FB_TR_READS(lq, t, ds);
#pragma unroll
          for (int ks = 0; ks < 4; ++ks)
            dkacc[ds] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                fb_cat(t[ks][0], t[ks][1]), pb[ks], dkacc[ds], 0, 0, 0);
        }
      }
    }
  }

  // epilogues: dV[kv][d] = acc ; dK[kv][d] = scale * acc
  unsigned short* ogv = dv_part + qbase + (int64_t)my_kv * FB_D;
  unsigned short* ogk = dk_part + qbase + (int64_t)my_kv * FB_D;
#pragma unroll
  for (int ds = 0; ds < 4; ++ds)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int d = ds * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
      ogv[d] = f32_to_bf16(dvacc[ds][r]);
      ogk[d] = f32_to_bf16(dkacc[ds][r] * scale);
    }
}

// ---------------------------------------------------------------------------
// fa2_dk: dK[kv][d] = scale * sum_q dS Q ; same grid as fa2_dv
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(FB_THREADS, 2)
fa2_dk_bf16(const unsigned short* __restrict__ q,
            const unsigned short* __restrict__ k,
            const unsigned short* __restrict__ v,
            const unsigned short* __restrict__ dout,
            const float* __restrict__ lse,
            const float* __restrict__ delta,
            unsigned short* __restrict__ dk_part,
            int B, int Hq, int Hkv, int S, float scale) {
  __shared__ unsigned short lq[FB_OTH * FB_D];    // Q q-tile (swz)
  __shared__ unsigned short ldo[FB_OTH * FB_D];   // dO q-tile (swz)
  __shared__ float llse[FB_OTH];
  __shared__ float ldelta[FB_OTH];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int l31 = lane & 31;
  const int half = lane >> 5;

  int bx = blockIdx.x, by = blockIdx.y, bz = blockIdx.z;
  ff_xcd_remap(bx, by, bz);
  const int kt = bx;
  const int h = by;
  const int b = bz;
  const int hkv = h / (Hq / Hkv);
  const int64_t qbase = (((int64_t)b * Hq + h) * S) * FB_D;
  const int64_t kbase = (((int64_t)b * Hkv + hkv) * S) * FB_D;
  const int kvB = kt * FB_TILE;
  const int kv0w = kvB + wave * FB_OWN;
  const int my_kv = kv0w + l31;
  const float scale2 = scale * FB_LOG2E;

  // K and V fragments (B-operands) for this lane's kv row
  fb_shortx8 kf[8], vf[8];
  {
    const unsigned short* kg = k + kbase + (int64_t)my_kv * FB_D;
    const unsigned short* vg = v + kbase + (int64_t)my_kv * FB_D;
#pragma unroll
    for (int ks = 0; ks < 8; ++ks) {
      kf[ks] = *reinterpret_cast<const fb_shortx8*>(kg + ks * 16 + half * 8);
      vf[ks] = *reinterpret_cast<const fb_shortx8*>(vg + ks * 16 + half * 8);
    }
  }

  fb_floatx16 dkacc[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) dkacc[i] = (fb_floatx16)(0.f);

  const int jq0 = kvB / FB_OTH;
  const int n_jq = S / FB_OTH;
  const float* lseg = lse + ((int64_t)b * Hq + h) * S;
  const float* dltg = delta + ((int64_t)b * Hq + h) * S;

  for (int jq = jq0; jq < n_jq; ++jq) {
    const int qt0 = jq * FB_OTH;
    const bool active = (qt0 + FB_OTH - 1 >= kv0w);
    const bool need_mask = (qt0 < kv0w + FB_OWN - 1);
    __syncthreads();
    fb_stage64(q + qbase + (int64_t)qt0 * FB_D, lq, tid);
    fb_stage64(dout + qbase + (int64_t)qt0 * FB_D, ldo, tid);
    if (tid < FB_OTH) {
      llse[tid] = lseg[qt0 + tid] * FB_LOG2E;
      ldelta[tid] = dltg[qt0 + tid];
    }
    __syncthreads();

    if (active) {
      // ST = Q.K^T and dP = dO.V^T — both D[q][kv], col = kv = l31
      fb_floatx16 st[2], dp[2];
#pragma unroll
      for (int s2 = 0; s2 < 2; ++s2) {
        st[s2] = (fb_floatx16)(0.f);
        dp[s2] = (fb_floatx16)(0.f);
      }
#pragma unroll
      for (int s2 = 0; s2 < 2; ++s2)
#pragma unroll
        for (int ks = 0; ks < 8; ++ks) {
          fb_shortx8 qfr = fb_afrag(lq, s2, ks, l31, half);
          st[s2] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qfr, kf[ks], st[s2], 0, 0, 0);
          fb_shortx8 dofr = fb_afrag(ldo, s2, ks, l31, half);
          dp[s2] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dofr, vf[ks], dp[s2], 0, 0, 0);
        }
      // dS = P * (dP - delta[q row]); P = exp2(st*scale2 - lse2[q row])
#pragma unroll
      for (int s2 = 0; s2 < 2; ++s2)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int qrow = s2 * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
          float x = st[s2][r] * scale2 - llse[qrow];
          if (need_mask && qt0 + qrow < my_kv) x = -INFINITY;
          float p = __builtin_amdgcn_exp2f(x);
          st[s2][r] = p * (dp[s2][r] - ldelta[qrow]);
        }
      fb_shortx8 pb[4];
      fb_relayout(st, pb, half);  // B-frags [kv][q-contraction]
      // dKacc[d][kv] += Q^T . dS
#pragma unroll
      for (int ds = 0; ds < 4; ++ds) {
        fb_shortx4 t[4][2];
        FB_TR_READS(lq, t, ds);
#pragma unroll
        for (int ks = 0; ks < 4; ++ks)
          dkacc[ds] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              fb_cat(t[ks][0], t[ks][1]), pb[ks], dkacc[ds], 0, 0, 0);
      }
    }
  }

  // epilogue: dK[kv][d] = scale * acc
  unsigned short* og = dk_part + qbase + (int64_t)my_kv * FB_D;
#pragma unroll
  for (int ds = 0; ds < 4; ++ds)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int d = ds * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
      og[d] = f32_to_bf16(dkacc[ds][r] * scale);
    }
}
