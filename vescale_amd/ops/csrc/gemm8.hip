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

// LDS byte-offset swizzles.  SWZ=1: HK's st_16x32 (XOR bit5 with bit9,
// within each 1 KiB subtile) — dilutes the frag-read conflict 16->4-way.
// SWZ=3: 3-bit XOR (bits 4-6 ^= bits 8-10, within each 2 KiB block):
// a 16x16x32 frag read has lanes 0-15 at row stride 128 B (byte bits
// 7-10); folding row bits 1-3 into bank bits 2-4 spreads the 16 rows
// over 16 distinct 4-bank groups -> measured-floor (conflict-free)
// access for ds_read_b128, while staying involutive (stage==read math),
// 16 B-aligned, and 128 B-coalesced on the pre-swizzled global side.
template <int SWZ>
DEV int g8_swz_t(int byte) {
  if constexpr (SWZ == 3) return byte ^ (((byte >> 8) & 7) << 4);
  return byte ^ (((byte >> 9) & 1) << 5);
}
DEV int g8_swz(int byte) { return g8_swz_t<1>(byte); }

// one half-tile (128 rows x 64 k) prefetch: 2 global_load_lds_dwordx4 per
// thread.  The LDS destination is linear (wave base + lane*16); each lane
// loads the element that BELONGS at its linear slot under the swizzle.
template <int SWZ = 1>
DEV void g8_prefetch_half(const unsigned short* __restrict__ g, int ld,
                          unsigned short* lds_base, int tid) {
#pragma unroll
  for (int j = 0; j < 2; ++j) {
    int slot = (tid + j * G8_THREADS) * 16;  // this thread's LDS byte slot
    int e = g8_swz_t<SWZ>(slot) >> 1;        // bf16 element living there
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
template <int SWZ = 1>
DEV g8_shortx8 g8_frag(const unsigned short* l, int frag, int kc, int lane) {
  int byte = (((frag * 16 + (lane & 15)) << 6) + kc * 32 + ((lane >> 4) & 3) * 8)
             << 1;
  return *reinterpret_cast<const g8_shortx8*>((const char*)l + g8_swz_t<SWZ>(byte));
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

// ---------------------------------------------------------------------------
// Region-rotation schedule (NOTES_ROUND2 appendix, guide m196/m201): the
// tile-boundary vmcnt(0) drain of the burst schedule is replaced by COUNTED
// waits, with prefetch slots spread one per phase in LAST-READER order so
// every region has 4-6 phases of latency cover and the youngest slots stay
// in flight across the boundary.
//
// Region lifetimes within a K-tile's 4 phases (p = mh2*2 + kc):
//   A[half][rows  0- 63] (q0)  last read p1      (phases 0,1 use mh2=0)
//   A[half][rows 64-127] (q1)  last read p3
//   B[half][all rows]          last read p3      (every phase reads 4 bfrags)
// Issue slots during tile m (reading buf b = m&1), 2 global_load_lds each:
//   p0: B(m+1) h0 + h1 -> buf b^1   [B(m-1) died (m-1).p3; 4-phase cover]
//   p1: A(m+1) q1      -> buf b^1   [A(m-1)q1 died (m-1).p3; 5-phase cover]
//   p2: A(m+2) q0      -> buf b     [A(m)q0 died m.p1;      6-phase cover]
// Counted waits (outstanding global_load_lds instructions per wave):
//   p0: vmcnt(4)  — youngest allowed: prev p1 (A q1, 2) + prev p2 (A q0, 2)
//   p2: vmcnt(8)  — youngest: this p0 (4) + p1 (2) + prev p2 (2)
// Safety: per-phase barrier PAIRS bound wave skew so no wave can issue a
// prefetch into a region while another wave's ds_reads of it are pending —
// each phase's ds_reads complete (compiler lgkm waits) before its MFMAs,
// and the closing barrier orders that against the next phase's issues.
// Compiler fences ("" ::: "memory") pin the C++ LDS reads inside their
// phase: hoisting a read past the barrier would race the rotation.
template <int SWZ = 1>
DEV void g8_prefetch_quarter(const unsigned short* __restrict__ g, int ld,
                             unsigned short* lds_half_base, int q, int tid) {
  // quarter q = rows q*64..q*64+63 of one [128 x 64] half-tile: 8 KiB =
  // 512 threads x 16 B = ONE global_load_lds per thread (the j-loop of
  // g8_prefetch_half split out per quarter)
  int slot = (tid + q * G8_THREADS) * 16;
  int e = g8_swz_t<SWZ>(slot) >> 1;
  int row = e >> 6;
  int col = e & 63;
  int wave = tid >> 6;
  unsigned short* lb =
      lds_half_base + ((wave * 64 + q * G8_THREADS) * 16) / 2;
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void*)(g + (int64_t)row * ld + col),
      (__attribute__((address_space(3))) void*)lb, 16, 0, 0);
}

// RA (read-ahead): each phase's operand frags are ds_read during the
// PREVIOUS phase's MFMA burst into the alternate register set, so LDS
// latency overlaps the matrix pipe (the fine ds||MFMA interleave of m196).
// Safety: a region overwrite may only issue one full phase after the
// lgkmcnt(0) that drained its readers (all waves provably past it once
// they pass that phase's closing barrier); waits shift one phase early:
// vmcnt(6) at p1 (A q1), vmcnt(4) at p3 (B pair + A q0 of the next tile).
template <int SWZ, bool PRIO = true, bool U2 = false, bool RA = false>
DEV void g8_body_rot(const unsigned short* __restrict__ A,
                     const unsigned short* __restrict__ B,
                     unsigned short* __restrict__ C, int M, int N, int K) {
  __shared__ unsigned short lA[2][2][128 * G8_BK];  // [dbuf][mhalf]
  __shared__ unsigned short lB[2][2][128 * G8_BK];  // [dbuf][nhalf]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = wave >> 2;
  const int wn = wave & 3;
  const int bh = wn >> 1;

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

  // prologue: tile 0 fully staged + drained; tile 1 issued in slot order
  // (B h0, B h1, A q0, A q1) with NO drain — the loop's counted waits gate it
  g8_prefetch_half<SWZ>(Ag, K, lA[0][0], tid);
  g8_prefetch_half<SWZ>(Ag + (int64_t)128 * K, K, lA[0][1], tid);
  g8_prefetch_half<SWZ>(Bg, K, lB[0][0], tid);
  g8_prefetch_half<SWZ>(Bg + (int64_t)128 * K, K, lB[0][1], tid);
  asm volatile("s_waitcnt vmcnt(0)");
  __builtin_amdgcn_s_barrier();
  if (ntile > 1) {
    const unsigned short* A1 = Ag + G8_BK;
    const unsigned short* B1 = Bg + G8_BK;
    g8_prefetch_half<SWZ>(B1, K, lB[1][0], tid);
    g8_prefetch_half<SWZ>(B1 + (int64_t)128 * K, K, lB[1][1], tid);
    g8_prefetch_quarter<SWZ>(A1, K, lA[1][0], 0, tid);
    g8_prefetch_quarter<SWZ>(A1 + (int64_t)128 * K, K, lA[1][1], 0, tid);
    g8_prefetch_quarter<SWZ>(A1, K, lA[1][0], 1, tid);
    g8_prefetch_quarter<SWZ>(A1 + (int64_t)128 * K, K, lA[1][1], 1, tid);
  }

  g8_shortx8 afr[2][4], bfr[2][4];
  if constexpr (RA) {
    // prologue read of (tile 0, phase 0) frags into register set 0
#pragma unroll
    for (int i = 0; i < 4; ++i)
      afr[0][i] = g8_frag<SWZ>(lA[0][wm], i, 0, lane);
#pragma unroll
    for (int j = 0; j < 4; ++j)
      bfr[0][j] = g8_frag<SWZ>(lB[0][bh], wn * 4 - bh * 8 + j, 0, lane);
  }

  // U2: unroll pairs of K-tiles so `buf` and the LDS bases are
  // compile-time constants in the loop body (needs ntile even)
  for (int kt0 = 0; kt0 < ntile; kt0 += (U2 ? 2 : 1)) {
#pragma unroll
   for (int sub = 0; sub < (U2 ? 2 : 1); ++sub) {
    const int kt = kt0 + sub;
    const int buf = U2 ? sub : (kt & 1);
    const unsigned short* A1 = Ag + (int64_t)(kt + 1) * G8_BK;  // tile m+1
    const unsigned short* B1 = Bg + (int64_t)(kt + 1) * G8_BK;
    const unsigned short* A2 = Ag + (int64_t)(kt + 2) * G8_BK;  // tile m+2

#pragma unroll
    for (int p = 0; p < 4; ++p) {
      const int kc = p & 1;
      const int mh2 = p >> 1;

      if constexpr (!RA) {
        // counted boundary waits (see header comment)
        if (p == 0 && kt >= 1) asm volatile("s_waitcnt vmcnt(4)");
        if (p == 2 && kt >= 1) asm volatile("s_waitcnt vmcnt(8)");
      } else {
        // RA: waits gate the read-AHEAD issued later this phase
        if (p == 1 && kt >= 1) asm volatile("s_waitcnt vmcnt(6)");
        if (p == 3) asm volatile("s_waitcnt vmcnt(4)");
      }
      asm volatile("" ::: "memory");

      // (a) ds-reads for this phase's 16 mfmas (RA: already in regs)
      if constexpr (!RA) {
#pragma unroll
        for (int i = 0; i < 4; ++i)
          afr[p & 1][i] = g8_frag<SWZ>(lA[buf][wm], mh2 * 4 + i, kc, lane);
#pragma unroll
        for (int j = 0; j < 4; ++j)
          bfr[p & 1][j] =
              g8_frag<SWZ>(lB[buf][bh], wn * 4 - bh * 8 + j, kc, lane);
      }

      // (b) this phase's prefetch slot
      if (p == 0 && kt >= 1 && kt + 1 < ntile) {
        g8_prefetch_half<SWZ>(B1, K, lB[buf ^ 1][0], tid);
        g8_prefetch_half<SWZ>(B1 + (int64_t)128 * K, K, lB[buf ^ 1][1], tid);
      } else if (p == 1 && kt >= 1 && kt + 1 < ntile) {
        g8_prefetch_quarter<SWZ>(A1, K, lA[buf ^ 1][0], 1, tid);
        g8_prefetch_quarter<SWZ>(A1 + (int64_t)128 * K, K, lA[buf ^ 1][1], 1, tid);
      } else if (p == 2 && kt + 2 < ntile) {
        // A(m+2) q0 -> buf b (A(m) q0 died at m.p1); this slot also covers
        // A(m+1) q0, issued one tile earlier at (m-1).p2
        g8_prefetch_quarter<SWZ>(A2, K, lA[buf][0], 0, tid);
        g8_prefetch_quarter<SWZ>(A2 + (int64_t)128 * K, K, lA[buf][1], 0, tid);
      }

      // (c) lockstep: barrier pair around the MFMA burst (m196: required
      // once prefetches overwrite regions of the buffer being read)
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)");
      if constexpr (PRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[mh2 * 4 + i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[p & 1][i], bfr[p & 1][j], acc[mh2 * 4 + i][j], 0, 0, 0);
      if constexpr (PRIO) __builtin_amdgcn_s_setprio(0);

      if constexpr (RA) {
        // read NEXT phase's frags now; the scheduler interleaves these
        // ds_reads with the MFMA burst above (both are between the same
        // barrier pair and only the fence below pins them)
        const int np = (p + 1) & 3;
        const int nkt = kt + (p == 3 ? 1 : 0);
        if (nkt < ntile) {
          const int nbuf = nkt & 1;
          const int nkc = np & 1;
          const int nmh2 = np >> 1;
#pragma unroll
          for (int i = 0; i < 4; ++i)
            afr[np & 1][i] =
                g8_frag<SWZ>(lA[nbuf][wm], nmh2 * 4 + i, nkc, lane);
#pragma unroll
          for (int j = 0; j < 4; ++j)
            bfr[np & 1][j] =
                g8_frag<SWZ>(lB[nbuf][bh], wn * 4 - bh * 8 + j, nkc, lane);
        }
      }
      asm volatile("" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
   }
  }

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
gemm8_tn_bf16_rot(const unsigned short* __restrict__ A,
                  const unsigned short* __restrict__ B,
                  unsigned short* __restrict__ C, int M, int N, int K) {
  g8_body_rot<1>(A, B, C, M, N, K);
}
extern "C" __global__ void __launch_bounds__(G8_THREADS, 1)
gemm8_tn_bf16_rot3(const unsigned short* __restrict__ A,
                   const unsigned short* __restrict__ B,
                   unsigned short* __restrict__ C, int M, int N, int K) {
  g8_body_rot<3>(A, B, C, M, N, K);
}
extern "C" __global__ void __launch_bounds__(G8_THREADS, 1)
gemm8_tn_bf16_rot3np(const unsigned short* __restrict__ A,
                     const unsigned short* __restrict__ B,
                     unsigned short* __restrict__ C, int M, int N, int K) {
  g8_body_rot<3, false>(A, B, C, M, N, K);
}
extern "C" __global__ void __launch_bounds__(G8_THREADS, 1)
gemm8_tn_bf16_rot3u2(const unsigned short* __restrict__ A,
                     const unsigned short* __restrict__ B,
                     unsigned short* __restrict__ C, int M, int N, int K) {
  // requires (K/64) even — enforced by the host wrapper
  g8_body_rot<3, false, true>(A, B, C, M, N, K);
}
extern "C" __global__ void __launch_bounds__(G8_THREADS, 1)
gemm8_tn_bf16_rot3ra(const unsigned short* __restrict__ A,
                     const unsigned short* __restrict__ B,
                     unsigned short* __restrict__ C, int M, int N, int K) {
  g8_body_rot<3, false, false, true>(A, B, C, M, N, K);
}


// ---------------------------------------------------------------------------
// rot8: 8-phase / 2-K-tile iteration with a 1-slot-per-phase rotation and
// TWO counted waits per iteration (vs rot3's two per tile).
// MEASURED: correct + race-stable but SLOWER than rot3np (917-1010 TF vs
// 1090-1208 on model shapes): halving the wait COUNT doesn't pay because
// each wait here is nearly a full drain (vmcnt(2) — FIFO forces it, the
// needed slots are only 2-4 positions old).  Going deeper needs the
// needed slots pushed earlier, which the 2-dbuf region lifetimes forbid
// within the 128 KiB LDS budget.  Kept as mode 7 for A/B evidence.
//
// Steady-state slot table during tiles (T, T+1) (T = 2i), phases p0..p7
// (p0-3 compute tile T from buf0, p4-7 tile T+1 from buf1):
//   p0: A(T+1)q1 -> buf1   [A(T-1)q1 died prev p7; read p6,p7]
//   p1: B(T+1)h0 -> buf1   [B(T-1) died prev p7;   read p4..p7]
//   p2: B(T+1)h1 -> buf1
//   p3: A(T+2)q0 -> buf0   [A(T)q0 died p1;  read next p0,p1]
//   p4: A(T+2)q1 -> buf0   [A(T)q1 died p3;  read next p2,p3]
//   p5: B(T+2)h0 -> buf0   [B(T) died p3;    read next p0..p3]
//   p6: B(T+2)h1 -> buf0
//   p7: A(T+3)q0 -> buf1   [A(T+1)q0 died p5; read next p4,p5]
// Waits (FIFO vmcnt; slots issue 2 global_load_lds each):
//   p0 start: vmcnt(2) — newest allowed = prev p7's slot; guarantees
//     everything through prev p6 landed: tile T fully staged (A q0 prev
//     p3, A q1 prev p4, B prev p5/p6).
//   p4 start: vmcnt(2) — newest allowed = p3's slot; guarantees p0..p2
//     landed: tile T+1 fully staged (A q0 prev p7, A q1 p0, B p1/p2).
// Each slot thus has >= 4 phases (~64 MFMAs) of latency cover before its
// wait, with up to 5 slots in flight mid-iteration.  Requires ntile even
// and >= 4 (host falls back to rot3np otherwise).
// ---------------------------------------------------------------------------
template <int SWZ>
DEV void g8_body_rot8(const unsigned short* __restrict__ A,
                      const unsigned short* __restrict__ B,
                      unsigned short* __restrict__ C, int M, int N, int K) {
  __shared__ unsigned short lA[2][2][128 * G8_BK];
  __shared__ unsigned short lB[2][2][128 * G8_BK];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = wave >> 2;
  const int wn = wave & 3;
  const int bh = wn >> 1;

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

  // prologue: tiles 0 (buf0) and 1 (buf1) fully staged + drained — the
  // loop's waits then gate the rotation from iteration 1 on.
  {
    const unsigned short* A1 = Ag + G8_BK;
    const unsigned short* B1 = Bg + G8_BK;
    g8_prefetch_half<SWZ>(Ag, K, lA[0][0], tid);
    g8_prefetch_half<SWZ>(Ag + (int64_t)128 * K, K, lA[0][1], tid);
    g8_prefetch_half<SWZ>(Bg, K, lB[0][0], tid);
    g8_prefetch_half<SWZ>(Bg + (int64_t)128 * K, K, lB[0][1], tid);
    g8_prefetch_half<SWZ>(A1, K, lA[1][0], tid);
    g8_prefetch_half<SWZ>(A1 + (int64_t)128 * K, K, lA[1][1], tid);
    g8_prefetch_half<SWZ>(B1, K, lB[1][0], tid);
    g8_prefetch_half<SWZ>(B1 + (int64_t)128 * K, K, lB[1][1], tid);
    asm volatile("s_waitcnt vmcnt(0)");
    __builtin_amdgcn_s_barrier();
  }

  for (int T = 0; T < ntile; T += 2) {
#pragma unroll
    for (int gp = 0; gp < 8; ++gp) {
      const int buf = gp >> 2;          // tile T (buf0) then T+1 (buf1)
      const int p = gp & 3;
      const int kc = p & 1;
      const int mh2 = p >> 1;
      const int kt = T + buf;

      // counted waits: the iteration's two drain points
      if (gp == 0 && T >= 2) asm volatile("s_waitcnt vmcnt(2)");
      if (gp == 4 && T >= 2) asm volatile("s_waitcnt vmcnt(2)");
      asm volatile("" ::: "memory");

      // this phase's operand frags
      g8_shortx8 af[4], bf[4];
#pragma unroll
      for (int i = 0; i < 4; ++i)
        af[i] = g8_frag<SWZ>(lA[buf][wm], mh2 * 4 + i, kc, lane);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        bf[j] = g8_frag<SWZ>(lB[buf][bh], wn * 4 - bh * 8 + j, kc, lane);

      // slot table (see header)
      switch (gp) {
        case 0:
          // T==0: tile 1 was fully staged by the prologue
          if (T >= 2 && T + 1 < ntile) {
            g8_prefetch_quarter<SWZ>(Ag + (int64_t)(T + 1) * G8_BK, K, lA[1][0], 1, tid);
            g8_prefetch_quarter<SWZ>(Ag + (int64_t)(T + 1) * G8_BK + (int64_t)128 * K, K, lA[1][1], 1, tid);
          }
          break;
        case 1:
          if (T >= 2 && T + 1 < ntile) {
            g8_prefetch_half<SWZ>(Bg + (int64_t)(T + 1) * G8_BK, K, lB[1][0], tid);
          }
          break;
        case 2:
          if (T >= 2 && T + 1 < ntile) {
            g8_prefetch_half<SWZ>(Bg + (int64_t)(T + 1) * G8_BK + (int64_t)128 * K, K, lB[1][1], tid);
          }
          break;
        case 3:
          if (T + 2 < ntile) {
            g8_prefetch_quarter<SWZ>(Ag + (int64_t)(T + 2) * G8_BK, K, lA[0][0], 0, tid);
            g8_prefetch_quarter<SWZ>(Ag + (int64_t)(T + 2) * G8_BK + (int64_t)128 * K, K, lA[0][1], 0, tid);
          }
          break;
        case 4:
          if (T + 2 < ntile) {
            g8_prefetch_quarter<SWZ>(Ag + (int64_t)(T + 2) * G8_BK, K, lA[0][0], 1, tid);
            g8_prefetch_quarter<SWZ>(Ag + (int64_t)(T + 2) * G8_BK + (int64_t)128 * K, K, lA[0][1], 1, tid);
          }
          break;
        case 5:
          if (T + 2 < ntile)
            g8_prefetch_half<SWZ>(Bg + (int64_t)(T + 2) * G8_BK, K, lB[0][0], tid);
          break;
        case 6:
          if (T + 2 < ntile)
            g8_prefetch_half<SWZ>(Bg + (int64_t)(T + 2) * G8_BK + (int64_t)128 * K, K, lB[0][1], tid);
          break;
        case 7:
          if (T + 3 < ntile) {
            g8_prefetch_quarter<SWZ>(Ag + (int64_t)(T + 3) * G8_BK, K, lA[1][0], 0, tid);
            g8_prefetch_quarter<SWZ>(Ag + (int64_t)(T + 3) * G8_BK + (int64_t)128 * K, K, lA[1][1], 0, tid);
          }
          break;
      }

      // lockstep barrier pair around the MFMA burst
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)");
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[mh2 * 4 + i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[i], bf[j], acc[mh2 * 4 + i][j], 0, 0, 0);
      asm volatile("" ::: "memory");
      __builtin_amdgcn_s_barrier();
      (void)kt;
    }
  }

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
gemm8_tn_bf16_rot8(const unsigned short* __restrict__ A,
                   const unsigned short* __restrict__ B,
                   unsigned short* __restrict__ C, int M, int N, int K) {
  g8_body_rot8<3>(A, B, C, M, N, K);
}


// ---------------------------------------------------------------------------
// rot9: the deep 8-phase schedule UNLOCKED by triple-buffering B.
// MEASURED: correct + race-stable, 893-979 TF vs rot3np's 1027-1137 on
// the same box — even with the guide's exact wait discipline (vmcnt(6)
// at phases 0/4, 3 slots in flight, every producer 4-7 phases ahead) the
// schedule macro-structure is NOT the remaining lever.  Together with
// rot8 this isolates the rot3np->guide gap to instruction-level
// ds_read/MFMA interleave (m196's hand-ordered stream), not waits,
// depth, or buffering.  Kept as mode 8 for A/B.
//
// rot8's lesson: with 2 dbufs for both operands, FIFO vmcnt forces
// near-full drains (consumers sit 2-4 slots behind producers).  B is the
// constraint — its regions only die at a tile's LAST phase.  Giving B a
// third buffer (LDS 64 KiB A + 96 KiB B = 160 KiB, the exact CU capacity)
// makes every B slot ancient by its first read, and the A rotation then
// supports the guide template's exact wait discipline: vmcnt(6) at
// phases 0 and 4 only — 3 slots in flight across each wait, every slot
// issued 4-7 phases before its consumer.
//
// Steady slot table, iteration (T, T+1), phases gp0..gp7
// (gp0-3 compute tile T: A from lA[T&1], B from lB[T%3]):
//   gp0: A(T+1)q1   [over A(T-1)q1, died prev gp7]   read gp6,7
//   gp1: B(T+2)h0   [over B(T-1),   died prev gp7]   read next gp0-3
//   gp2: B(T+2)h1
//   gp3: A(T+2)q0   [over A(T)q0,   died gp1]        read next gp0,1
//   gp4: A(T+2)q1   [over A(T)q1,   died gp3]        read next gp2,3
//   gp5: B(T+3)h0   [over B(T),     died gp3]        read next gp4-7
//   gp6: B(T+3)h1
//   gp7: A(T+3)q0   [over A(T+1)q0, died gp5]        read next gp4,5
// Waits (2 global_load_lds per slot, FIFO):
//   vmcnt(6)@gp0: newest 3 slots (prev gp5,6,7) may be outstanding ->
//     prev gp3 (A(T)q0) and prev gp4 (A(T)q1) are LANDED; B(T) was staged
//     a full iteration earlier.
//   vmcnt(6)@gp4: newest 3 (gp1,2,3) outstanding -> gp0 (A(T+1)q1) and
//     prev gp7 (A(T+1)q0) landed; B(T+1) ancient.
// Requires even ntile >= 2 (host mode guard).
// ---------------------------------------------------------------------------
template <int SWZ>
DEV void g8_body_rot9(const unsigned short* __restrict__ A,
                      const unsigned short* __restrict__ B,
                      unsigned short* __restrict__ C, int M, int N, int K) {
  __shared__ unsigned short lA[2][2][128 * G8_BK];
  __shared__ unsigned short lB[3][2][128 * G8_BK];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = wave >> 2;
  const int wn = wave & 3;
  const int bh = wn >> 1;

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

  // prologue: tiles 0,1 fully staged (A bufs 0,1 + B bufs 0,1), drained
  {
    const unsigned short* A1 = Ag + G8_BK;
    const unsigned short* B1 = Bg + G8_BK;
    g8_prefetch_half<SWZ>(Ag, K, lA[0][0], tid);
    g8_prefetch_half<SWZ>(Ag + (int64_t)128 * K, K, lA[0][1], tid);
    g8_prefetch_half<SWZ>(Bg, K, lB[0][0], tid);
    g8_prefetch_half<SWZ>(Bg + (int64_t)128 * K, K, lB[0][1], tid);
    g8_prefetch_half<SWZ>(A1, K, lA[1][0], tid);
    g8_prefetch_half<SWZ>(A1 + (int64_t)128 * K, K, lA[1][1], tid);
    g8_prefetch_half<SWZ>(B1, K, lB[1][0], tid);
    g8_prefetch_half<SWZ>(B1 + (int64_t)128 * K, K, lB[1][1], tid);
    asm volatile("s_waitcnt vmcnt(0)");
    __builtin_amdgcn_s_barrier();
  }

  int b3 = 0;  // lB buffer of tile T (cycles 0,2,1,0,...: +2 mod 3)
  for (int T = 0; T < ntile; T += 2) {
    const int bT = b3;                    // B buf of tile T
    const int bT1 = b3 + 1 == 3 ? 0 : b3 + 1;   // tile T+1
    const int bT2 = bT1 + 1 == 3 ? 0 : bT1 + 1; // tiles T+2 / T+3 staging
#pragma unroll
    for (int gp = 0; gp < 8; ++gp) {
      const int abuf = gp >> 2;  // lA index: tile T -> 0/1 by parity of T
      const int p = gp & 3;
      const int kc = p & 1;
      const int mh2 = p >> 1;
      const unsigned short* lAband = lA[(T + abuf) & 1][wm];
      const unsigned short* lBband = lB[gp < 4 ? bT : bT1][bh];

      if ((gp == 0 || gp == 4) && T >= 2) asm volatile("s_waitcnt vmcnt(6)");
      asm volatile("" ::: "memory");

      g8_shortx8 af[4], bf[4];
#pragma unroll
      for (int i = 0; i < 4; ++i)
        af[i] = g8_frag<SWZ>(lAband, mh2 * 4 + i, kc, lane);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        bf[j] = g8_frag<SWZ>(lBband, wn * 4 - bh * 8 + j, kc, lane);

      switch (gp) {
        case 0:  // A(T+1)q1 (prologue covered T==0)
          if (T >= 2 && T + 1 < ntile) {
            g8_prefetch_quarter<SWZ>(Ag + (int64_t)(T + 1) * G8_BK, K, lA[(T + 1) & 1][0], 1, tid);
            g8_prefetch_quarter<SWZ>(Ag + (int64_t)(T + 1) * G8_BK + (int64_t)128 * K, K, lA[(T + 1) & 1][1], 1, tid);
          }
          break;
        case 1:
          if (T + 2 < ntile)
            g8_prefetch_half<SWZ>(Bg + (int64_t)(T + 2) * G8_BK, K, lB[bT2][0], tid);
          break;
        case 2:
          if (T + 2 < ntile)
            g8_prefetch_half<SWZ>(Bg + (int64_t)(T + 2) * G8_BK + (int64_t)128 * K, K, lB[bT2][1], tid);
          break;
        case 3:
          if (T + 2 < ntile) {
            g8_prefetch_quarter<SWZ>(Ag + (int64_t)(T + 2) * G8_BK, K, lA[T & 1][0], 0, tid);
            g8_prefetch_quarter<SWZ>(Ag + (int64_t)(T + 2) * G8_BK + (int64_t)128 * K, K, lA[T & 1][1], 0, tid);
          }
          break;
        case 4:
          if (T + 2 < ntile) {
            g8_prefetch_quarter<SWZ>(Ag + (int64_t)(T + 2) * G8_BK, K, lA[T & 1][0], 1, tid);
            g8_prefetch_quarter<SWZ>(Ag + (int64_t)(T + 2) * G8_BK + (int64_t)128 * K, K, lA[T & 1][1], 1, tid);
          }
          break;
        case 5:
          if (T + 3 < ntile)
            g8_prefetch_half<SWZ>(Bg + (int64_t)(T + 3) * G8_BK, K, lB[bT][0], tid);
          break;
        case 6:
          if (T + 3 < ntile)
            g8_prefetch_half<SWZ>(Bg + (int64_t)(T + 3) * G8_BK + (int64_t)128 * K, K, lB[bT][1], tid);
          break;
        case 7:
          if (T + 3 < ntile) {
            g8_prefetch_quarter<SWZ>(Ag + (int64_t)(T + 3) * G8_BK, K, lA[(T + 1) & 1][0], 0, tid);
            g8_prefetch_quarter<SWZ>(Ag + (int64_t)(T + 3) * G8_BK + (int64_t)128 * K, K, lA[(T + 1) & 1][1], 0, tid);
          }
          break;
      }

      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)");
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[mh2 * 4 + i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[i], bf[j], acc[mh2 * 4 + i][j], 0, 0, 0);
      asm volatile("" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
    b3 = bT2;  // advance by 2 mod 3
  }

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
gemm8_tn_bf16_rot9(const unsigned short* __restrict__ A,
                   const unsigned short* __restrict__ B,
                   unsigned short* __restrict__ C, int M, int N, int K) {
  g8_body_rot9<3>(A, B, C, M, N, K);
}
