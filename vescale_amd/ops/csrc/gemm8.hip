// 8-phase-style 256x256 TN GEMM — port of the guide's verified
// HipKittens schedule (cdna_hip_programming.md §"The 256² 8-phase
// template", m194-m204), bring-up variant:
//   C[M,N] = sum_k A[M,K] * B[N,K]   (both row-major over K, bf16 in,
//   bf16 out, fp32 accumulate) — the nn.Linear forward shape.
//
// Geometry (guide table): 256x256 C tile, BK=64 K-tiles, 8 waves
// (2M x 4N), per-wave C = 128x64 (8x4 frags of 16x16x32 MFMA), LDS =
// 128 KiB (2 dbuf x 2 half x 128x64 x {A,B}), half-tile staging with
// global_load_lds_dwordx4 (2 per thread per half-tile), st_16x32 LDS
// swizzle (byte ^= ((byte>>9)&1)<<5) applied by PRE-SWIZZLING the global
// source address (the LDS side of global_load_lds is linear — m173),
// 4 phases per K-tile: ds-read one quadrant || one half-tile prefetch ||
// 16 MFMAs under setprio(1), s_barrier fencing each phase (the m196
// lesson: the fine interleave IS the lever), bijective XCD remap (m204).
// Depth-1 prefetch (next K-tile into the other dbuf) with a full vmcnt
// drain at the tile boundary — the conservative-correct schedule; the
// counted-vmcnt deepening is a measured follow-up.
//
// Requires M%256==0, N%256==0, K%64==0 (the training GEMM shapes).
#include "common.h"

#define G8_BM 256
#define G8_BN 256
#define G8_BK 64
#define G8_THREADS 512
#define G8_MFRAG 8   // 8 x 16 rows  = wave's 128 C rows
#define G8_NFRAG 4   // 4 x 16 cols  = wave's 64 C cols

typedef float g8_floatx4 __attribute__((ext_vector_type(4)));
typedef short g8_shortx8 __attribute__((ext_vector_type(8)));

// st_16x32 swizzle on LDS byte offsets (within each 1 KiB subtile)
DEV int g8_swz(int byte) { return byte ^ (((byte >> 9) & 1) << 5); }

// one half-tile (128 rows x 64 k) prefetch: 2 global_load_lds_dwordx4 per
// thread.  The LDS destination is linear (wave base + lane*16); each lane
// loads the element that BELONGS at its linear slot under the swizzle.
DEV void g8_prefetch_half(const unsigned short* __restrict__ g, int ld,
                          unsigned short* lds_base, int tid) {
#pragma unroll
  for (int j = 0; j < 2; ++j) {
    int slot = (tid + j * G8_THREADS) * 16;  // this thread's LDS byte slot
    int e = g8_swz(slot) >> 1;               // bf16 element living there
    int row = e >> 6;                        // 64 k per row
    int col = e & 63;
    int wave = tid >> 6;
    unsigned short* lb =
        lds_base + ((wave * 64 + j * G8_THREADS) * 16) / 2;  // wave base
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)(g + (int64_t)row * ld + col),
        (__attribute__((address_space(3))) void*)lb, 16, 0, 0);
  }
}

// A/B operand fragment (16x16x32): row = frag*16 + (lane&15),
// k = kc*32 + ((lane>>4)&3)*8 .. +8, from a swizzled [128][64] half-tile
DEV g8_shortx8 g8_frag(const unsigned short* l, int frag, int kc, int lane) {
  int byte = (((frag * 16 + (lane & 15)) << 6) + kc * 32 + ((lane >> 4) & 3) * 8)
             << 1;
  return *reinterpret_cast<const g8_shortx8*>((const char*)l + g8_swz(byte));
}

template <int MODE>  // 0 = tile-boundary barriers only; 1 = per-phase
                     // barrier pairs (the guide's lockstep schedule, m196)
DEV void g8_body(const unsigned short* __restrict__ A,
                 const unsigned short* __restrict__ B,
                 unsigned short* __restrict__ C,
                 int M, int N, int K) {
  __shared__ unsigned short lA[2][2][128 * G8_BK];  // [dbuf][mhalf]
  __shared__ unsigned short lB[2][2][128 * G8_BK];  // [dbuf][nhalf]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = wave >> 2;   // A half (wave rows = wm*128 .. +128)
  const int wn = wave & 3;    // C cols = wn*64 .. +64
  const int bh = wn >> 1;     // B half this wave reads

  // bijective XCD remap (m204)
  int gx = gridDim.x, nwg = gx * (int)gridDim.y;
  int f = blockIdx.x + gx * blockIdx.y;
  {
    int q = nwg >> 3, r = nwg & 7;
    int xcd = f & 7, off = f >> 3;
    f = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + off;
  }
  const int bm = (f % gx) * G8_BM;
  const int bn = (f / gx) * G8_BN;

  const unsigned short* Ag = A + (int64_t)bm * K;
  const unsigned short* Bg = B + (int64_t)bn * K;

  g8_floatx4 acc[G8_MFRAG][G8_NFRAG];
#pragma unroll
  for (int i = 0; i < G8_MFRAG; ++i)
#pragma unroll
    for (int j = 0; j < G8_NFRAG; ++j) acc[i][j] = (g8_floatx4)(0.f);

  const int ntile = K / G8_BK;

  // prologue: K-tile 0 into dbuf 0
  g8_prefetch_half(Ag, K, lA[0][0], tid);
  g8_prefetch_half(Ag + (int64_t)128 * K, K, lA[0][1], tid);
  g8_prefetch_half(Bg, K, lB[0][0], tid);
  g8_prefetch_half(Bg + (int64_t)128 * K, K, lB[0][1], tid);
  asm volatile("s_waitcnt vmcnt(0)");
  __builtin_amdgcn_s_barrier();

  for (int kt = 0; kt < ntile; ++kt) {
    const int buf = kt & 1;
    const bool have_next = (kt + 1 < ntile);
    const unsigned short* An = Ag + (int64_t)(kt + 1) * G8_BK;
    const unsigned short* Bn = Bg + (int64_t)(kt + 1) * G8_BK;

    // 4 phases: q = (m-subhalf, kc)
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      const int kc = p & 1;
      const int mh2 = p >> 1;

      // (a) ds-reads for this phase's 16 mfmas
      g8_shortx8 af[4], bf[4];
#pragma unroll
      for (int i = 0; i < 4; ++i)
        af[i] = g8_frag(lA[buf][wm], mh2 * 4 + i, kc, lane);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        bf[j] = g8_frag(lB[buf][bh], wn * 4 - bh * 8 + j, kc, lane);

      // (b) burst ALL of K-tile kt+1's half-tiles at phase 0: they get
      // three full phases to land, so the boundary vmcnt(0) is near-free
      // (spreading them one-per-phase exposed the last one's full latency)
      if (have_next && p == 0) {
        g8_prefetch_half(An, K, lA[buf ^ 1][0], tid);
        g8_prefetch_half(An + (int64_t)128 * K, K, lA[buf ^ 1][1], tid);
        g8_prefetch_half(Bn, K, lB[buf ^ 1][0], tid);
        g8_prefetch_half(Bn + (int64_t)128 * K, K, lB[buf ^ 1][1], tid);
      }

      // (c) no explicit lgkm drain: the frag loads are plain C++ LDS
      // reads, so the compiler inserts MINIMAL counted waits and can
      // software-pipeline reads across the unrolled phases

      // (d) 16 mfmas under raised priority (T5); MODE 1 = lockstep
      if constexpr (MODE == 1) __builtin_amdgcn_s_barrier();
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[mh2 * 4 + i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[i], bf[j], acc[mh2 * 4 + i][j], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);

      if constexpr (MODE == 1) __builtin_amdgcn_s_barrier();

      // (e) K-tile boundary: all of kt+1's half-tiles must have LANDED
      // (LDS writes from vm loads become block-visible via drain+barrier)
      if (p == 3) {
        asm volatile("s_waitcnt vmcnt(0)");
        __builtin_amdgcn_s_barrier();
      }
    }
  }

  // epilogue: D-layout 16x16 (col = lane&15, row = 4*(lane>>4)+r)
  const int c0 = lane & 15;
  const int r0 = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < G8_MFRAG; ++i) {
#pragma unroll
    for (int j = 0; j < G8_NFRAG; ++j) {
      int row = bm + wm * 128 + i * 16 + r0;
      int col = bn + wn * 64 + j * 16 + c0;
      unsigned short* cg = C + (int64_t)row * N + col;
#pragma unroll
      for (int r = 0; r < 4; ++r)
        cg[(int64_t)r * N] = f32_to_bf16(acc[i][j][r]);
    }
  }
}

extern "C" __global__ void __launch_bounds__(G8_THREADS, 1)
gemm8_tn_bf16(const unsigned short* A, const unsigned short* B,
              unsigned short* C, int M, int N, int K) {
  g8_body<0>(A, B, C, M, N, K);
}
extern "C" __global__ void __launch_bounds__(G8_THREADS, 1)
gemm8_tn_bf16_lockstep(const unsigned short* A, const unsigned short* B,
                       unsigned short* C, int M, int N, int K) {
  g8_body<1>(A, B, C, M, N, K);
}
