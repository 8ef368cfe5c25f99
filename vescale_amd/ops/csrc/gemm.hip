// Hand-written CDNA4 MFMA GEMM — templated tile geometry, deep K-pipeline.
//
// C[M,N] = A[M,K] @ B[N,K]^T   (TN: both inputs row-major with K inner —
// the linear-layer forward and weight-grad shape).  bf16 in, bf16 out,
// fp32 MFMA accumulate.
//
// Design per /opt/skills/guides/cdna_hip_programming.md §5 / §5.5 (T1-T5):
//   - 512 threads = 8 waves (2M x 4N); mfma_f32_16x16x32_bf16
//   - NBUF-deep K-tile pipeline: staging issued BEFORE the phase's
//     ds_read+MFMA (T3 recipe), COUNTED s_waitcnt vmcnt at the K-tile
//     switch (T4 — never drain to 0 mid-loop)
//   - __builtin_amdgcn_global_load_lds width 16, linear LDS dest +
//     inverse-swizzled global source, swizzled ds_read (rule #21)
//   - LDS XOR swizzle (T2): 16B-chunk bits XOR row bits
//   - raw s_barrier (no implicit vmcnt(0) drain), s_setprio around MFMA (T5)
//   - XCD-aware bijective blockIdx swizzle (T1, ERRATA #11 formula)
#include "common.h"

#define NXCD 8

typedef float floatx4 __attribute__((ext_vector_type(4)));
typedef short shortx8 __attribute__((ext_vector_type(8)));

template <int ROWB, bool SWZ>
DEV int swz_off(int byte_off) {
  if (!SWZ) return byte_off;
  int row = byte_off / ROWB;
  if (ROWB == 64) {
    // 4 chunks/row: XOR chunk bits (4-5) with row bits 1-2
    return byte_off ^ (((row >> 1) & 3) << 4);
  }
  // 128B rows, 8 chunks: XOR chunk bits (4-6) with row bits 0-2 (G4 form)
  return byte_off ^ ((row & 7) << 4);
}

template <int BM, int BN, int BK, int NBUF, bool SWZ>
__device__ __forceinline__ void
gemm_tn_kernel(const unsigned short* __restrict__ A,
               const unsigned short* __restrict__ B,
               unsigned short* __restrict__ C, int M, int N, int K) {
  constexpr int THREADS = 512;
  constexpr int ROWB = BK * 2;               // bytes per LDS row
  constexpr int TILE_A = BM * BK;            // elems
  constexpr int TILE_B = BN * BK;
  constexpr int LOADS_A = TILE_A * 2 / (THREADS * 16);
  constexpr int LOADS_B = TILE_B * 2 / (THREADS * 16);
  constexpr int LOADS_PER_TILE = LOADS_A + LOADS_B;
  constexpr int MFRAG = BM / 2 / 16;         // per-wave M fragments
  constexpr int NFRAG = BN / 4 / 16;         // per-wave N fragments
  constexpr int KSTEPS = BK / 32;

  __shared__ unsigned short lds[NBUF * (TILE_A + TILE_B)];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = wave >> 2;
  const int wn = wave & 3;

  int nwg = gridDim.x;
  int orig = blockIdx.x;
  int xcd = orig % NXCD;
  int q = nwg / NXCD, r = nwg % NXCD;
  int wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + orig / NXCD;
  const int tiles_n = N / BN;
  const int bm0 = (wgid / tiles_n) * BM;
  const int bn0 = (wgid % tiles_n) * BN;

  const int KT = K / BK;

  auto stage_tile = [&](int kt, int buf) {
    const int k0 = kt * BK;
    unsigned short* base_a = lds + buf * (TILE_A + TILE_B);
    unsigned short* base_b = base_a + TILE_A;
#pragma unroll
    for (int j = 0; j < LOADS_A; ++j) {
      int o = (tid + j * THREADS) * 16;
      int so = swz_off<ROWB, SWZ>(o);
      int row = so / ROWB;
      int colb = so % ROWB;
      const unsigned short* ga = A + (int64_t)(bm0 + row) * K + k0 + colb / 2;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)ga,
          (__attribute__((address_space(3))) void*)((char*)base_a + o), 16, 0, 0);
    }
#pragma unroll
    for (int j = 0; j < LOADS_B; ++j) {
      int o = (tid + j * THREADS) * 16;
      int so = swz_off<ROWB, SWZ>(o);
      int row = so / ROWB;
      int colb = so % ROWB;
      const unsigned short* gb = B + (int64_t)(bn0 + row) * K + k0 + colb / 2;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)gb,
          (__attribute__((address_space(3))) void*)((char*)base_b + o), 16, 0, 0);
    }
  };

  floatx4 acc[MFRAG][NFRAG];
#pragma unroll
  for (int i = 0; i < MFRAG; ++i)
#pragma unroll
    for (int j = 0; j < NFRAG; ++j) acc[i][j] = (floatx4){0.f, 0.f, 0.f, 0.f};

  const int arow0 = wm * (BM / 2) + (lane & 15);
  const int brow0 = wn * (BN / 4) + (lane & 15);
  const int kch = lane >> 4;  // 0..3, 8 elems each (32 K per kstep)

  auto lda = [&](int buf, int fm, int ks) -> shortx8 {
    unsigned short* base = lds + buf * (TILE_A + TILE_B);
    int off = swz_off<ROWB, SWZ>((arow0 + fm * 16) * ROWB + (ks * 4 + kch) * 16);
    return *reinterpret_cast<shortx8*>((char*)base + off);
  };
  auto ldb = [&](int buf, int fn, int ks) -> shortx8 {
    unsigned short* base = lds + buf * (TILE_A + TILE_B) + TILE_A;
    int off = swz_off<ROWB, SWZ>((brow0 + fn * 16) * ROWB + (ks * 4 + kch) * 16);
    return *reinterpret_cast<shortx8*>((char*)base + off);
  };

#define RAW_BARRIER() asm volatile("s_barrier" ::: "memory")
  // counted vmcnt: wait until at most n LOADS remain in flight
#define VMCNT(n) asm volatile("s_waitcnt vmcnt(" #n ")" ::: "memory")
  constexpr int AHEAD = NBUF - 1;  // tiles staged ahead of compute

  // prologue: stage tiles 0..AHEAD-1, wait for tile 0
  for (int t = 0; t < AHEAD && t < KT; ++t) stage_tile(t, t % NBUF);
  {
    int inflight = (AHEAD < KT ? AHEAD : KT) * LOADS_PER_TILE;
    int want = inflight - LOADS_PER_TILE;  // all but tile 0
    // immediate-operand dispatch
    if (want >= 16) { VMCNT(16); }
    else if (want >= 12) { VMCNT(12); }
    else if (want >= 8) { VMCNT(8); }
    else if (want >= 4) { VMCNT(4); }
    else { VMCNT(0); }
  }
  RAW_BARRIER();

  for (int kt = 0; kt < KT; ++kt) {
    const int buf = kt % NBUF;
    // stage first (T3: issue loads BEFORE the ds_read+MFMA of this phase)
    if (kt + AHEAD < KT) stage_tile(kt + AHEAD, (kt + AHEAD) % NBUF);

#pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks) {
      shortx8 bfr[NFRAG];
#pragma unroll
      for (int fn = 0; fn < NFRAG; ++fn) bfr[fn] = ldb(buf, fn, ks);
      // phase A: first half of M-frags
      shortx8 afr[MFRAG / 2];
#pragma unroll
      for (int fm = 0; fm < MFRAG / 2; ++fm) afr[fm] = lda(buf, fm, ks);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int fm = 0; fm < MFRAG / 2; ++fm)
#pragma unroll
        for (int fn = 0; fn < NFRAG; ++fn)
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[fm], bfr[fn], acc[fm][fn], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      // phase B: second half
#pragma unroll
      for (int fm = 0; fm < MFRAG / 2; ++fm) afr[fm] = lda(buf, fm + MFRAG / 2, ks);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int fm = 0; fm < MFRAG / 2; ++fm)
#pragma unroll
        for (int fn = 0; fn < NFRAG; ++fn)
          acc[fm + MFRAG / 2][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[fm], bfr[fn], acc[fm + MFRAG / 2][fn], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }

    if (kt + 1 < KT) {
      // counted wait: tile kt+1 must be landed; younger tiles stay in flight
      int ahead_staged = (KT - 1 - kt < AHEAD ? KT - 1 - kt : AHEAD);
      int want = (ahead_staged - 1) * LOADS_PER_TILE;
      if (want >= 16) { VMCNT(16); }
      else if (want >= 12) { VMCNT(12); }
      else if (want >= 8) { VMCNT(8); }
      else if (want >= 4) { VMCNT(4); }
      else { VMCNT(0); }
      RAW_BARRIER();
    }
  }

  // epilogue: D mapping col=lane&15, row=4*(lane>>4)+r (guide §3, m89)
  const int crow_base = bm0 + wm * (BM / 2) + 4 * (lane >> 4);
  const int ccol_base = bn0 + wn * (BN / 4) + (lane & 15);
#pragma unroll
  for (int fm = 0; fm < MFRAG; ++fm)
#pragma unroll
    for (int fn = 0; fn < NFRAG; ++fn) {
      int col = ccol_base + fn * 16;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        int row = crow_base + fm * 16 + rr;
        C[(int64_t)row * N + col] = f32_to_bf16(acc[fm][fn][rr]);
      }
    }
#undef RAW_BARRIER
#undef VMCNT
}

// Instantiations: V0 = 256x256x32 4-buf; V1 = 128x256x64 3-buf; V2 = V0
// without LDS swizzle (A/B probe)
extern "C" __global__ void __launch_bounds__(512, 1)
gemm_tn_bf16_v0(const unsigned short* A, const unsigned short* B,
                unsigned short* C, int M, int N, int K) {
  gemm_tn_kernel<256, 256, 32, 4, true>(A, B, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512, 1)
gemm_tn_bf16_v1(const unsigned short* A, const unsigned short* B,
                unsigned short* C, int M, int N, int K) {
  gemm_tn_kernel<128, 256, 64, 3, true>(A, B, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512, 1)
gemm_tn_bf16_v2(const unsigned short* A, const unsigned short* B,
                unsigned short* C, int M, int N, int K) {
  gemm_tn_kernel<256, 256, 32, 4, false>(A, B, C, M, N, K);
}
