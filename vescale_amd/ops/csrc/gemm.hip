// Hand-written CDNA4 MFMA GEMM — templated tile geometry, deep K-pipeline.
//
// C[M,N] = A[M,K] @ B[N,K]^T   (TN: both inputs row-major with K inner —
// the linear-layer forward and weight-grad shape).  bf16 in, bf16 out,
// fp32 MFMA accumulate.
//
// Design per /opt/skills/guides/cdna_hip_programming.md §5 / §5.5 (T1-T5):
//   - 512 threads = 8 waves (2M x 4N); mfma_f32_16x16x32_bf16
//   - NBUF-deep K-tile pipeline: staging issued BEFORE the phase's
//     ds_read+MFMA (T3), COUNTED s_waitcnt vmcnt at the K-tile switch
//     (T4 — never drain to 0 mid-loop)
//   - __builtin_amdgcn_global_load_lds width 16, linear LDS dest +
//     inverse-swizzled global source, swizzled ds_read (rule #21)
//   - LDS XOR swizzle (T2); raw s_barrier; s_setprio around MFMA (T5)
//   - XCD-aware bijective blockIdx swizzle (T1, ERRATA #11 formula)
//   - ALL addressing hoisted out of the K-loop: LDS read/stage offsets are
//     per-lane constants; global sources advance by BK per staged tile
//     (the asm audit showed per-iter 64-bit MADs + bfe swizzle math
//     competing with MFMA issue)
#include "common.h"

#define NXCD 8

typedef float floatx4 __attribute__((ext_vector_type(4)));
typedef short shortx8 __attribute__((ext_vector_type(8)));

template <int ROWB, bool SWZ>
DEV int swz_off(int byte_off) {
  if (!SWZ) return byte_off;
  int row = byte_off / ROWB;
  if (ROWB == 64) {
    return byte_off ^ (((row >> 1) & 3) << 4);
  }
  return byte_off ^ ((row & 7) << 4);
}

template <int BM, int BN, int BK, int NBUF, bool SWZ, int WAVES_M = 2, int WAVES_N = 4>
__device__ __forceinline__ void
gemm_tn_kernel(const unsigned short* __restrict__ A,
               const unsigned short* __restrict__ B,
               unsigned short* __restrict__ C, int M, int N, int K) {
  constexpr int THREADS = WAVES_M * WAVES_N * 64;
  constexpr int ROWB = BK * 2;
  constexpr int TILE_A = BM * BK;
  constexpr int TILE_B = BN * BK;
  constexpr int TILE = TILE_A + TILE_B;
  constexpr int LOADS_A = TILE_A * 2 / (THREADS * 16);
  constexpr int LOADS_B = TILE_B * 2 / (THREADS * 16);
  constexpr int LOADS_PER_TILE = LOADS_A + LOADS_B;
  constexpr int MFRAG = BM / WAVES_M / 16;
  constexpr int NFRAG = BN / WAVES_N / 16;
  constexpr int KSTEPS = BK / 32;

  __shared__ unsigned short lds[NBUF * TILE];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = wave / WAVES_N;
  const int wn = wave % WAVES_N;

  int nwg = gridDim.x;
  int orig = blockIdx.x;
  int xcd = orig % NXCD;
  int q = nwg / NXCD, r = nwg % NXCD;
  int wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + orig / NXCD;
  const int tiles_n = N / BN;
  const int bm0 = (wgid / tiles_n) * BM;
  const int bn0 = (wgid % tiles_n) * BN;

  const int KT = K / BK;

  // ---- hoisted staging state: per-load LDS byte offset + global ptr ----
  int st_lds_a[LOADS_A], st_lds_b[LOADS_B];
  const unsigned short* st_ga[LOADS_A];
  const unsigned short* st_gb[LOADS_B];
#pragma unroll
  for (int j = 0; j < LOADS_A; ++j) {
    int o = (tid + j * THREADS) * 16;
    st_lds_a[j] = o;
    int so = swz_off<ROWB, SWZ>(o);
    st_ga[j] = A + (int64_t)(bm0 + so / ROWB) * K + (so % ROWB) / 2;
  }
#pragma unroll
  for (int j = 0; j < LOADS_B; ++j) {
    int o = (tid + j * THREADS) * 16;
    st_lds_b[j] = o;
    int so = swz_off<ROWB, SWZ>(o);
    st_gb[j] = B + (int64_t)(bn0 + so / ROWB) * K + (so % ROWB) / 2;
  }

  // stage the tile currently pointed at by st_* into LDS buffer `buf`,
  // then advance the global pointers by BK
  auto stage_advance = [&](int buf) {
    char* base_a = (char*)(lds + buf * TILE);
    char* base_b = (char*)(lds + buf * TILE + TILE_A);
#pragma unroll
    for (int j = 0; j < LOADS_A; ++j) {
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)st_ga[j],
          (__attribute__((address_space(3))) void*)(base_a + st_lds_a[j]), 16, 0, 0);
      st_ga[j] += BK;
    }
#pragma unroll
    for (int j = 0; j < LOADS_B; ++j) {
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)st_gb[j],
          (__attribute__((address_space(3))) void*)(base_b + st_lds_b[j]), 16, 0, 0);
      st_gb[j] += BK;
    }
  };

  floatx4 acc[MFRAG][NFRAG];
#pragma unroll
  for (int i = 0; i < MFRAG; ++i)
#pragma unroll
    for (int j = 0; j < NFRAG; ++j) acc[i][j] = (floatx4){0.f, 0.f, 0.f, 0.f};

  // ---- hoisted LDS read offsets (bytes, within the tile) ----
  const int arow0 = wm * (BM / WAVES_M) + (lane & 15);
  const int brow0 = wn * (BN / WAVES_N) + (lane & 15);
  const int kch = lane >> 4;
  int a_off[MFRAG][KSTEPS], b_off[NFRAG][KSTEPS];
#pragma unroll
  for (int fm = 0; fm < MFRAG; ++fm)
#pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks)
      a_off[fm][ks] = swz_off<ROWB, SWZ>((arow0 + fm * 16) * ROWB + (ks * 4 + kch) * 16);
#pragma unroll
  for (int fn = 0; fn < NFRAG; ++fn)
#pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks)
      b_off[fn][ks] = swz_off<ROWB, SWZ>((brow0 + fn * 16) * ROWB + (ks * 4 + kch) * 16) + TILE_A * 2;

#define RAW_BARRIER() asm volatile("s_barrier" ::: "memory")
#define VMCNT(n) asm volatile("s_waitcnt vmcnt(" #n ")" ::: "memory")
  constexpr int AHEAD = NBUF - 1;

  for (int t = 0; t < AHEAD && t < KT; ++t) stage_advance(t % NBUF);
  {
    int staged = AHEAD < KT ? AHEAD : KT;
    int want = (staged - 1) * LOADS_PER_TILE;
    if (want >= 16) { VMCNT(16); }
    else if (want >= 12) { VMCNT(12); }
    else if (want >= 8) { VMCNT(8); }
    else if (want >= 4) { VMCNT(4); }
    else { VMCNT(0); }
  }
  RAW_BARRIER();

  // main loop unrolled by NBUF so the LDS buffer base folds to a constant
  int kt = 0;
  while (kt < KT) {
#pragma unroll
    for (int ub = 0; ub < NBUF; ++ub) {
      if (kt >= KT) break;
      const char* base = (const char*)(lds + ub * TILE);
      if (kt + AHEAD < KT) stage_advance((ub + AHEAD) % NBUF);

#pragma unroll
      for (int ks = 0; ks < KSTEPS; ++ks) {
        shortx8 bfr[NFRAG];
#pragma unroll
        for (int fn = 0; fn < NFRAG; ++fn)
          bfr[fn] = *reinterpret_cast<const shortx8*>(base + b_off[fn][ks]);
        shortx8 afr[MFRAG / 2];
#pragma unroll
        for (int fm = 0; fm < MFRAG / 2; ++fm)
          afr[fm] = *reinterpret_cast<const shortx8*>(base + a_off[fm][ks]);
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int fm = 0; fm < MFRAG / 2; ++fm)
#pragma unroll
          for (int fn = 0; fn < NFRAG; ++fn)
            acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afr[fm], bfr[fn], acc[fm][fn], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
#pragma unroll
        for (int fm = 0; fm < MFRAG / 2; ++fm)
          afr[fm] = *reinterpret_cast<const shortx8*>(base + a_off[fm + MFRAG / 2][ks]);
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
        int ahead_staged = (KT - 1 - kt < AHEAD ? KT - 1 - kt : AHEAD);
        int want = (ahead_staged - 1) * LOADS_PER_TILE;
        if (want >= 16) { VMCNT(16); }
        else if (want >= 12) { VMCNT(12); }
        else if (want >= 8) { VMCNT(8); }
        else if (want >= 4) { VMCNT(4); }
        else { VMCNT(0); }
        RAW_BARRIER();
      }
      ++kt;
    }
  }

  // epilogue: D mapping col=lane&15, row=4*(lane>>4)+r (guide §3, m89)
  const int crow_base = bm0 + wm * (BM / WAVES_M) + 4 * (lane >> 4);
  const int ccol_base = bn0 + wn * (BN / WAVES_N) + (lane & 15);
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

// V3: 128x128 tile, 4 waves, 4-deep pipeline -> 64KB LDS, 2 blocks/CU
extern "C" __global__ void __launch_bounds__(256, 2)
gemm_tn_bf16_v3(const unsigned short* A, const unsigned short* B,
                unsigned short* C, int M, int N, int K) {
  gemm_tn_kernel<128, 128, 32, 4, true, 2, 2>(A, B, C, M, N, K);
}

// V4: 512x256 tile, 16 waves (8Mx2N), 3-deep pipeline: higher arithmetic
// intensity per staged byte (175 vs 131 FLOP/B) for the LLC-BW-bound regime
extern "C" __global__ void __launch_bounds__(1024, 1)
gemm_tn_bf16_v4(const unsigned short* A, const unsigned short* B,
                unsigned short* C, int M, int N, int K) {
  gemm_tn_kernel<512, 256, 32, 3, true, 8, 2>(A, B, C, M, N, K);
}
