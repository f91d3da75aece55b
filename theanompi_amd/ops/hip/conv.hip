// Implicit-GEMM convolution kernels for MI355X (gfx950), bf16 NHWC.
//
// One gather-GEMM kernel covers conv fwd, conv dgrad (per stride class)
// and FC fwd/dgrad (R=S=1): replaces cuDNN conv fwd/bgrad + cuBLAS FC
// GEMMs of the reference.  A TN split-K
// variant produces wgrad (N3).
//
// GEMM view (fwd):  Y[M=N*P*Q, Kgp] = gatherA(X)[M, Kdf] @ W[Kdf, Kgp]
//   - A rows are output pixels; the flat reduction dim (r,s,c) is read
//     through a 16-element-piece offset TABLE (host-built; see
//     models/layers.py ConvPlan).  Inputs are physically halo-padded so
//     the gather needs no bounds checks.
//   - W is stored out-channel-major [Kgp][Kdf] so both LDS tiles are
//     k-contiguous per row and MFMA fragments load as 16-B ds_reads.
//   - MFMA v_mfma_f32_16x16x32_bf16, 128x128 (or 128x64) block tile,
//     BK=64, 4 waves, double-buffered LDS with stage-ahead (the
//     2-phase schedule).
//
// Numerics: bf16 inputs, fp32 MFMA accumulation, bf16 outputs (wgrad:
// fp32 atomic accumulation).

#include "common.hpp"
#include <cstdlib>

// ---------------------------------------------------------------------------
// conv_gather_gemm: fwd / dgrad / linear
// ---------------------------------------------------------------------------

struct GatherGemmArgs {
  const __bf16* __restrict__ X;     // gather source (halo-padded)
  const __bf16* __restrict__ W;     // [G][BNtot][Kdf]
  const float* __restrict__ bias;   // [G*BNtot] or nullptr
  const int* __restrict__ table;    // [Kdf/16] element offsets
  __bf16* __restrict__ Y;
  int M, P, Q;                      // M = N*P*Q output pixels
  long sN, sP, sQ, outOff;          // output strides/base (elements)
  long inNS, inPS, inQS;            // input pixel strides (elements)
  int groupInOff;                   // input channel offset per group
  int Kdf;                          // flat reduction size (mult of 64)
  int BNtot;                        // output channels per group (Kgp)
  int relu;
  int splitK;                       // >1: K-slices atomic-add into Y32
  float* __restrict__ Y32;          // fp32 workspace for split-K
  long xTotal;                      // X elements (gather tail clamp)
  // optional fused BN statistics: bucketed per-channel sum / sum-sq of
  // the bf16 output, [STATS_NB][2][statsC] fp32 (atomics per bucket).
  // Collected only on the LDS-staged epilogue with splitK == 1.
  float* __restrict__ stats;
  int statsC;
};

constexpr int STATS_NB = 32;      // stat atomic buckets (contention)

template <int BM, int BN, int LAYOUT, int DEPTH, int BK = 64,
          int NFAST = 0>
__global__ __launch_bounds__(BM * 2) void conv_gather_gemm(GatherGemmArgs a) {
  constexpr int WAVES = (BM == 256) ? 8 : 4;     // block = WAVES*64 thr
  constexpr int WAVES_N = (BN == 128) ? 2 : 1;
  constexpr int WAVES_M = WAVES / WAVES_N;
  constexpr int TM = BM / WAVES_M;       // 64 or 32
  constexpr int FM = TM / 16;            // 4 or 2
  constexpr int FN = 64 / 16;            // wave tile N = 64 always
  constexpr int PPR = BK / 8;            // 16-B pieces per tile row
  constexpr int AI = BM * BK / (512 * WAVES);  // glds per wave (A tile)
  constexpr int BI = BN * BK / (512 * WAVES);
  constexpr int TPW = AI + BI;           // glds per wave per tile

  // lane-linear tiles (glds writes base + lane*16B); content is
  // XOR-swizzled by source-piece permutation, undone at the frag reads.
  __shared__ __align__(16) __bf16 As[DEPTH][BM][BK];
  __shared__ __align__(16) __bf16 Bs[DEPTH][BN][BK];
  // gather table staged to LDS once: its reads then ride the lgkm
  // counter, so they never force a vmcnt drain of in-flight glds
  // (cdna guide §5 "three .s-level traps" (b)).  Dynamically sized
  // (launcher passes Kdf/16 ints): most layers need <1 KB, and the old
  // fixed 8 KB was the difference between 2 and 3 blocks/CU on the
  // slimmer tile variants.
  extern __shared__ int Tab[];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  // XCD-aware tile remap.  Default: give each XCD a contiguous chunk
  // of m-tiles so neighbouring tiles (overlapping conv receptive
  // fields) share L2.  NFAST (launcher requires gridDim.x % 8 == 0
  // and > 1 n-tile): an XCD walks ALL n-tiles of one m-tile before
  // advancing, so a wide-output layer's A tile is read from DRAM once
  // instead of once per n-tile (ResNet's 256-out 1x1s re-read X 4x).
  int mTile, nTile;
  if (NFAST) {
    long lid = (long)blockIdx.y * gridDim.x + blockIdx.x;
    const int nT = gridDim.y;
    const int xcd = (int)(lid & 7);
    long idx = lid >> 3;
    nTile = (int)(idx % nT);
    mTile = xcd * ((int)gridDim.x >> 3) + (int)(idx / nT);
  } else {
    mTile = blockIdx.x;
    const int nwg = gridDim.x;
    if (nwg >= 16) {
      const int q = nwg >> 3, r = nwg & 7;
      const int xcd = mTile & 7, idx = mTile >> 3;
      mTile = (xcd < r) ? xcd * (q + 1) + idx
                        : r * (q + 1) + (xcd - r) * q + idx;
    }
    nTile = blockIdx.y;
  }
  const int g = blockIdx.z / a.splitK;
  const int slice = blockIdx.z % a.splitK;

  const __bf16* Wg = a.W + (long)g * a.BNtot * a.Kdf;
  const long gInOff = (long)g * a.groupInOff;
  const int bn0 = nTile * BN;

  {  // table -> LDS (bounded by binding: Kdf/16 <= 2048)
    const int n = a.Kdf >> 4;
    for (int i = tid; i < n; i += WAVES * 64) Tab[i] = a.table[i];
  }

  // --- per-lane glds descriptors (fixed across K) --------------------------
  long aRowBase[AI];
  int aK8x8[AI];
#pragma unroll
  for (int i = 0; i < AI; ++i) {
    int piece = (wave * AI + i) * 64 + lane;
    int row = piece / PPR;
    int k8 = (piece % PPR) ^ (row & (PPR - 1));  // source-piece swizzle
    int m = mTile * BM + row;
    if (m >= a.M) m = a.M - 1;                 // clamped rows discarded
    int q = m % a.Q;
    int tmp = m / a.Q;
    int p = tmp % a.P;
    int n = tmp / a.P;
    aRowBase[i] = (long)n * a.inNS + (long)p * a.inPS + (long)q * a.inQS
                  + gInOff;
    aK8x8[i] = k8 * 8;
  }
  const __bf16* bSrc[BI];
#pragma unroll
  for (int i = 0; i < BI; ++i) {
    int piece = (wave * BI + i) * 64 + lane;
    int row = piece / PPR;
    int k8 = (piece % PPR) ^ (row & (PPR - 1));
    int n = bn0 + row;
    if (n >= a.BNtot) n = a.BNtot - 1;         // discarded at C-write
    bSrc[i] = Wg + (long)n * a.Kdf + k8 * 8;
  }
  __syncthreads();                             // table visible

  auto stage = [&](int buf, int k0) {
#pragma unroll
    for (int i = 0; i < AI; ++i) {
      int off = Tab[(k0 + aK8x8[i]) >> 4] + (aK8x8[i] & 15);
      long ga = aRowBase[i] + off;
      // span padding may point past the last pixel of the last image
      // (stem layers); the values are multiplied by zero weights, the
      // ADDRESS must stay in bounds.
      if (ga + 8 > a.xTotal) ga = a.xTotal - 8;
      glds16(a.X + ga,
             &As[buf][0][0] + (wave * AI + i) * 512);
    }
#pragma unroll
    for (int i = 0; i < BI; ++i) {
      glds16(bSrc[i] + k0,
             &Bs[buf][0][0] + (wave * BI + i) * 512);
    }
  };

  // --- main loop ------------------------------------------------------------
  const int wm = (WAVES_N == 2) ? (wave >> 1) : wave;
  const int wn = (WAVES_N == 2) ? (wave & 1) : 0;
  const int m0 = wm * TM;
  const int n0 = wn * 64;
  static_assert(WAVES_M * TM == BM && WAVES_N * 64 == BN, "tiling");
  const int lrow = lane & 15;

  f32x4 acc[FM][FN];
#pragma unroll
  for (int i = 0; i < FM; ++i)
#pragma unroll
    for (int j = 0; j < FN; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int nt = a.Kdf / BK;
  const int nti = (nt - slice + a.splitK - 1) / a.splitK;  // my K-tiles
  // prologue: fill the ring DEPTH-1 deep
#pragma unroll
  for (int d = 0; d < DEPTH - 1; ++d)
    if (d < nti) stage(d % DEPTH, (slice + (long)d * a.splitK) * BK);

  for (int ti = 0; ti < nti; ++ti) {
    const int tnext = ti + DEPTH - 1;
    if (tnext < nti)
      stage(tnext % DEPTH, (slice + (long)tnext * a.splitK) * BK);
    // wait MY glds of tile ti: outstanding newer-tile instrs stay in
    // flight (counted wait; never vmcnt(0) in the loop)
    const int newer = (nti - 1 - ti) < (DEPTH - 1) ? (nti - 1 - ti)
                                                   : (DEPTH - 1);
    if (newer == 2)
      asm volatile("s_waitcnt vmcnt(%0)" ::"i"(2 * TPW) : "memory");
    else if (newer == 1)
      asm volatile("s_waitcnt vmcnt(%0)" ::"i"(TPW) : "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();              // raw: no vmcnt drain
    const int buf = ti % DEPTH;
#pragma unroll
    for (int ks = 0; ks < BK; ks += 32) {
      bf16x8 af[FM], bf[FN];
#pragma unroll
      for (int i = 0; i < FM; ++i) {
        int row = m0 + i * 16 + lrow;
        af[i] = frag_load_swz<LAYOUT, BK>(&As[buf][row][0], row, ks,
                                          lane);
      }
#pragma unroll
      for (int j = 0; j < FN; ++j) {
        int row = n0 + j * 16 + lrow;
        bf[j] = frag_load_swz<LAYOUT, BK>(&Bs[buf][row][0], row, ks,
                                          lane);
      }
#pragma unroll
      for (int i = 0; i < FM; ++i)
#pragma unroll
        for (int j = 0; j < FN; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[i], bf[j], acc[i][j], 0, 0, 0);
    }
    __builtin_amdgcn_s_barrier();              // readers done with buf
  }

  // --- epilogue ------------------------------------------------------------
  const int cOut0 = g * a.BNtot + bn0;
  if (a.Y32) {
    // split-K: fp32 atomic partials (bias/relu in the finalize kernel)
#pragma unroll
    for (int i = 0; i < FM; ++i) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        int rowl = m0 + i * 16 + (lane >> 4) * 4 + reg;
        int m = mTile * BM + rowl;
        if (m >= a.M) continue;
        int q = m % a.Q;
        int tmp = m / a.Q;
        int p = tmp % a.P;
        int n = tmp / a.P;
        long obase = a.outOff + (long)n * a.sN + (long)p * a.sP
                     + (long)q * a.sQ + cOut0;
#pragma unroll
        for (int j = 0; j < FN; ++j) {
          int nl = n0 + j * 16 + (lane & 15);
          if (bn0 + nl >= a.BNtot) continue;
          atomicAdd(&a.Y32[obase + nl], acc[i][j][reg]);
        }
      }
    }
    return;
  }
  // direct path: stage the bf16 tile through LDS (reusing the staging
  // buffers) so global stores are full coalesced rows — the natural
  // per-lane layout writes 32-B scattered runs, which made low-K convs
  // store-bound.  (Falls back to scattered stores when the staging
  // buffers are smaller than the Y tile, e.g. the BK=32 variant.)
  constexpr bool LDSEP = (2 * BK) >= BN;
  if constexpr (!LDSEP) {
#pragma unroll
    for (int i = 0; i < FM; ++i) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        int rowl = m0 + i * 16 + (lane >> 4) * 4 + reg;
        int m = mTile * BM + rowl;
        if (m >= a.M) continue;
        int q = m % a.Q;
        int tmp = m / a.Q;
        int p = tmp % a.P;
        int n = tmp / a.P;
        long obase = a.outOff + (long)n * a.sN + (long)p * a.sP
                     + (long)q * a.sQ + cOut0;
#pragma unroll
        for (int j = 0; j < FN; ++j) {
          int nl = n0 + j * 16 + (lane & 15);
          if (bn0 + nl >= a.BNtot) continue;
          float v = acc[i][j][reg];
          if (a.bias) v += a.bias[cOut0 + nl];
          if (a.relu) v = v > 0.f ? v : 0.f;
          a.Y[obase + nl] = (__bf16)v;
        }
      }
    }
    return;
  }
  __syncthreads();
  __bf16* Yt = &As[0][0][0];               // [BM][BN] view, 16/32 KiB
#pragma unroll
  for (int i = 0; i < FM; ++i) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      int rowl = m0 + i * 16 + (lane >> 4) * 4 + reg;
#pragma unroll
      for (int j = 0; j < FN; ++j) {
        int nl = n0 + j * 16 + (lane & 15);
        float v = acc[i][j][reg];
        if (a.bias) v += a.bias[cOut0 + nl];
        if (a.relu) v = v > 0.f ? v : 0.f;
        Yt[rowl * BN + nl] = (__bf16)v;
      }
    }
  }
  __syncthreads();
  // each lane streams 16-B pieces of consecutive rows out to global.
  // With the piece stride (THREADS is a multiple of BN/8) every thread
  // keeps ONE fixed 8-channel group, so fused BN statistics ride along
  // as per-thread partials.
  constexpr int PIECES = (BM * BN) / 8;    // 8 bf16 per piece
  constexpr int THREADS = BM * 2;
  static_assert(THREADS % (BN / 8) == 0, "fixed c8 per thread");
  const int nvalid = a.BNtot - bn0;        // columns that exist (< BN ok)
  float s1l[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  float s2l[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  for (int pc = tid; pc < PIECES; pc += THREADS) {
    int rowl = pc / (BN / 8);
    int c8 = (pc % (BN / 8)) * 8;
    int m = mTile * BM + rowl;
    if (m >= a.M || c8 >= nvalid) continue;
    int q = m % a.Q;
    int tmp = m / a.Q;
    int p = tmp % a.P;
    int n = tmp / a.P;
    long obase = a.outOff + (long)n * a.sN + (long)p * a.sP
                 + (long)q * a.sQ + cOut0 + c8;
    int ne = nvalid - c8 < 8 ? nvalid - c8 : 8;
    if (ne == 8) {
      *(pack16*)(a.Y + obase) = *(const pack16*)(Yt + rowl * BN + c8);
    } else {
      for (int e = 0; e < ne; ++e)
        a.Y[obase + e] = Yt[rowl * BN + c8 + e];
    }
    if (a.stats) {
      for (int e = 0; e < ne; ++e) {
        float v = (float)Yt[rowl * BN + c8 + e];
        s1l[e] += v;
        s2l[e] += v * v;
      }
    }
  }
  if (a.stats) {
    // cross-thread reduce (threads sharing a c8 group) through Bs,
    // which is idle after the MFMA loop; exact fit at BN=64.
    __syncthreads();
    float* S = (float*)&Bs[0][0][0];      // [THREADS][8] x2
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      S[tid * 8 + e] = s1l[e];
      S[(THREADS + tid) * 8 + e] = s2l[e];
    }
    __syncthreads();
    constexpr int GROUPS = BN / 8;
    constexpr int PER = THREADS / GROUPS;
    if (tid < GROUPS * 8) {
      int c8i = tid >> 3, e = tid & 7;
      int c = c8i * 8 + e;
      if (c < nvalid) {
        float t1 = 0.f, t2 = 0.f;
#pragma unroll
        for (int r = 0; r < PER; ++r) {
          int src = c8i + r * GROUPS;
          t1 += S[src * 8 + e];
          t2 += S[(THREADS + src) * 8 + e];
        }
        int bucket = blockIdx.x & (STATS_NB - 1);
        long gc = cOut0 + c;
        atomicAdd(&a.stats[((long)bucket * 2) * a.statsC + gc], t1);
        atomicAdd(&a.stats[((long)bucket * 2 + 1) * a.statsC + gc], t2);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// 8-phase deep-pipelined gather-GEMM (guide T3+T4+T5): 256x128 tile,
// 8 waves, 3-tile glds ring, counted vmcnt(6) once per K-tile (never 0
// in the loop), per-phase {ds_read || glds issue || MFMA cluster} with
// raw barriers and setprio around the MFMA cluster.
// ---------------------------------------------------------------------------

template <int LAYOUT>
__global__ __launch_bounds__(512) void conv_gather_gemm_8p(
    GatherGemmArgs a) {
  constexpr int BM = 256, BN = 128, BK = 64, WAVES = 8;
  constexpr int AI = BM / (8 * WAVES);   // 4 glds per wave (A)
  constexpr int BI = BN / (8 * WAVES);   // 2 glds per wave (B)
  constexpr int TPW = AI + BI;           // 6

  __shared__ __align__(16) __bf16 As[3][BM][BK];
  __shared__ __align__(16) __bf16 Bs[3][BN][BK];
  extern __shared__ int Tab[];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  int mTile = blockIdx.x;
  {
    const int nwg = gridDim.x;
    if (nwg >= 16) {
      const int q = nwg >> 3, r = nwg & 7;
      const int xcd = mTile & 7, idx = mTile >> 3;
      mTile = (xcd < r) ? xcd * (q + 1) + idx
                        : r * (q + 1) + (xcd - r) * q + idx;
    }
  }
  const int nTile = blockIdx.y, g = blockIdx.z;

  const __bf16* Wg = a.W + (long)g * a.BNtot * a.Kdf;
  const long gInOff = (long)g * a.groupInOff;
  const int bn0 = nTile * BN;

  {
    const int n = a.Kdf >> 4;
    for (int i = tid; i < n; i += 512) Tab[i] = a.table[i];
  }

  long aRowBase[AI];
  int aK8x8[AI];
#pragma unroll
  for (int i = 0; i < AI; ++i) {
    int piece = (wave * AI + i) * 64 + lane;
    int row = piece >> 3;
    int k8 = (piece & 7) ^ (row & 7);
    int m = mTile * BM + row;
    if (m >= a.M) m = a.M - 1;
    int q = m % a.Q;
    int tmp = m / a.Q;
    int p = tmp % a.P;
    int n = tmp / a.P;
    aRowBase[i] = (long)n * a.inNS + (long)p * a.inPS + (long)q * a.inQS
                  + gInOff;
    aK8x8[i] = k8 * 8;
  }
  const __bf16* bSrc[BI];
#pragma unroll
  for (int i = 0; i < BI; ++i) {
    int piece = (wave * BI + i) * 64 + lane;
    int row = piece >> 3;
    int k8 = (piece & 7) ^ (row & 7);
    int n = bn0 + row;
    if (n >= a.BNtot) n = a.BNtot - 1;
    bSrc[i] = Wg + (long)n * a.Kdf + k8 * 8;
  }
  __syncthreads();

  // glds issue for one (tile, sub-step) pair: sub 0..2 issues 2 glds
  // each (A0+A1, A2+A3, B0+B1)
  auto stage2 = [&](int buf, int k0, int sub) {
    if (sub < 2) {
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        int ii = sub * 2 + i;
        int off = Tab[(k0 + aK8x8[ii]) >> 4] + (aK8x8[ii] & 15);
        long ga = aRowBase[ii] + off;
        if (ga + 8 > a.xTotal) ga = a.xTotal - 8;
        glds16(a.X + ga, &As[buf][(wave * AI + ii) * 8][0]);
      }
    } else {
#pragma unroll
      for (int i = 0; i < BI; ++i) {
        glds16(bSrc[i] + k0, &Bs[buf][(wave * BI + i) * 8][0]);
      }
    }
  };

  const int wm = wave >> 1;          // 4 x 2 wave grid
  const int wn = wave & 1;
  const int m0 = wm * 64;
  const int n0 = wn * 64;
  const int lrow = lane & 15;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int nt = a.Kdf / BK;
  // prologue: tiles 0 and 1 in flight
  for (int sub = 0; sub < 3; ++sub) stage2(0, 0, sub);
  if (nt > 1)
    for (int sub = 0; sub < 3; ++sub) stage2(1, BK, sub);
  if (nt > 1)
    asm volatile("s_waitcnt vmcnt(%0)" ::"i"(TPW) : "memory");
  else
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (int t = 0; t < nt; ++t) {
    const int buf = t % 3;
    const int nbuf = (t + 2) % 3;
    const bool more = (t + 2) < nt;

    bf16x8 bf[4];                    // B frags for current ks
#pragma unroll
    for (int ph = 0; ph < 4; ++ph) {
      const int ks = (ph >> 1) * 32;
      const int f0 = (ph & 1) * 2;
      bf16x8 af[2];
      if ((ph & 1) == 0) {
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          int row = n0 + j * 16 + lrow;
          bf[j] = frag_load_swz<LAYOUT>(&Bs[buf][row][0], row, ks, lane);
        }
      }
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        int row = m0 + (f0 + i) * 16 + lrow;
        af[i] = frag_load_swz<LAYOUT>(&As[buf][row][0], row, ks, lane);
      }
      if (more && ph < 3) stage2(nbuf, (t + 2) * BK, ph);
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_sched_barrier(0);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[f0 + i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[i], bf[j], acc[f0 + i][j], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      if (ph == 3) {
        // wait for tile t+1 here, one barrier ahead of its first read
        if (more)
          asm volatile("s_waitcnt vmcnt(%0)" ::"i"(TPW) : "memory");
        else if (t + 1 < nt)
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      __builtin_amdgcn_s_barrier();
    }
  }

  const int cOut0 = g * a.BNtot + bn0;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      int rowl = m0 + i * 16 + (lane >> 4) * 4 + reg;
      int m = mTile * BM + rowl;
      if (m >= a.M) continue;
      int q = m % a.Q;
      int tmp = m / a.Q;
      int p = tmp % a.P;
      int n = tmp / a.P;
      long obase = a.outOff + (long)n * a.sN + (long)p * a.sP
                   + (long)q * a.sQ + cOut0;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int nl = n0 + j * 16 + (lane & 15);
        if (bn0 + nl >= a.BNtot) continue;
        float v = acc[i][j][reg];
        if (a.bias) v += a.bias[cOut0 + nl];
        if (a.relu) v = v > 0.f ? v : 0.f;
        a.Y[obase + nl] = (__bf16)v;
      }
    }
  }
}

// split-K finalize: y = bf16(relu(Y32 + bias))
__global__ void splitk_finalize_k(float* __restrict__ y32,
                                  const float* __restrict__ bias,
                                  __bf16* __restrict__ y, long total,
                                  int C, int relu) {
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    float v = y32[i];
    y32[i] = 0.f;     // leave the cached workspace zeroed for reuse
    if (bias) v += bias[i % C];
    if (relu && v < 0.f) v = 0.f;
    y[i] = (__bf16)v;
  }
}

// ---------------------------------------------------------------------------
// conv_wgrad: TN split-K GEMM, fp32 atomic accumulation
//   DW[g][kout][fd] += sum_pix DY[pix][g*Kgp+kout] * gatherA(X)[pix][fd]
// ---------------------------------------------------------------------------

struct WgradArgs {
  const __bf16* __restrict__ X;     // halo-padded input
  const __bf16* __restrict__ DY;    // [M][Kphys]
  const int* __restrict__ table;    // [Kdf/16]
  const float* __restrict__ kdfMask;  // [Kdf] 1/0: zero padded slots
  float* __restrict__ DW;           // [G][Kgp][Kdf]
  int M, P, Q;
  long inNS, inPS, inQS;
  int groupInOff;
  int Kdf, Kgp, Kphys;
  int splitK;
  long xTotal;                      // X elements (gather tail clamp)
  // optional fused bias grad: db[kout] += sum_pix dy — the fTile==0
  // blocks re-read their staged dy tiles from LDS (the separate
  // bias_wgrad kernel was ~51 launches/step on GoogLeNet)
  float* __restrict__ db;
  // split-K workspace [splitK][G*Kgp*Kdf]: slices write their partial
  // tiles PLAIN (each block owns a disjoint region) and a segsum pass
  // reduces into DW — split-K atomics serialized up to ~455 writers
  // per dW cache line on the VGG shapes.  nullptr = atomic fallback.
  float* __restrict__ WS;
};

template <int LAYOUT, int BNF, int BMK = 64, int DEPTH = 2,
          int BKP = 64>
__global__ __launch_bounds__(256) void conv_wgrad_k(WgradArgs a) {
  // TN GEMM over pixels.  Both operands stage LINEARLY ([pix][64ch]
  // subtiles, 16-B glds pieces, content-swizzled) and the k-major MFMA
  // fragments come from ds_read_b64_tr_b16 (hardware transpose).
  // Counted-vmcnt DEPTH-deep ring (like the fwd gather), LDS-resident
  // gather table.  BMK (kout tile) 64 or 128: the wider tile does 2x
  // the MFMA work per staged dy byte for wide-Kgp layers; BKP (pixel
  // chunk) 64 or 128 — the swizzles are 32-row periodic, so the wider
  // chunk halves barrier/wait overhead per pixel (all autotuned).
  constexpr int XI = BKP * BNF / 2048;   // x-tile glds per wave
  constexpr int DI = BKP * BMK / 2048;   // dy-tile glds per wave
  constexpr int TPW = XI + DI;

  __shared__ __align__(16) __bf16 DYt[DEPTH][BKP][BMK];   // [pix][kout]
  __shared__ __align__(16) __bf16 XT[DEPTH][BKP][BNF];    // [pix][fd]
  extern __shared__ int Tab[];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int kTile = blockIdx.x, fTile = blockIdx.y;
  const int g = blockIdx.z / a.splitK;
  const int slice = blockIdx.z % a.splitK;

  const int kout0 = kTile * BMK;
  const int fd0 = fTile * BNF;
  const long gInOff = (long)g * a.groupInOff;
  const long dyTotal = (long)a.M * a.Kphys;

  {
    const int n = a.Kdf >> 4;
    for (int i = tid; i < n; i += 256) Tab[i] = a.table[i];
  }
  __syncthreads();

  // piece descriptors.  dy tile: piece = (wave*DI + i)*64 + lane over
  // [BKP][64]; x tile: piece over [BKP][BNF], row = pix.
  auto stage = [&](int buf, int pix0) {
    const bool interior = (pix0 + BKP <= a.M);
    // ---- x gather pieces ----
#pragma unroll
    for (int i = 0; i < XI; ++i) {
      int piece = (wave * XI + i) * 64 + lane;
      int row = piece / (BNF / 8);
      int k8i = piece % (BNF / 8);
      // slot k8i*8 holds logical k = slot ^ s(row); s matches
      // frag_load_tr_swz (row-octave XOR + group-phase XOR)
      int k8 = (k8i * 8) ^ ((row & 7) * 8) ^ (((row >> 3) & 3)
                                              * (BNF / 4));
      int pa = pix0 + row;
      int pc = pa < a.M ? pa : a.M - 1;
      int q = pc % a.Q;
      int tmp = pc / a.Q;
      int p = tmp % a.P;
      int n = tmp / a.P;
      long base = (long)n * a.inNS + (long)p * a.inPS
                  + (long)q * a.inQS + gInOff;
      int ti2 = (fd0 + k8) >> 4;
      int tmax = (a.Kdf >> 4) - 1;
      if (ti2 > tmax) ti2 = tmax;        // BNF tail: masked out below
      int off = Tab[ti2] + (k8 & 15);
      if (base + off + 8 > a.xTotal) base = a.xTotal - 8 - off;
      glds16(a.X + base + off,
             &XT[buf][0][0] + (long)(wave * XI + i) * 512);
    }
    // ---- dy pieces ----
#pragma unroll
    for (int i = 0; i < DI; ++i) {
      int piece = (wave * DI + i) * 64 + lane;
      int row = piece / (BMK / 8);
      int k8i = piece % (BMK / 8);
      int k8 = ((k8i * 8) ^ ((row & 7) * 8)
                ^ (((row >> 3) & 3) * (BMK / 4)));
      int pa = pix0 + row;
      int pc = pa < a.M ? pa : a.M - 1;
      long dyaddr = (long)pc * a.Kphys + g * a.Kgp + kout0 + k8;
      if (dyaddr + 8 > dyTotal) dyaddr = dyTotal - 8;
      if (interior || pa < a.M) {
        glds16(a.DY + dyaddr, &DYt[buf][0][0] + (wave * DI + i) * 512);
      } else {
        // tail: rows past M must be EXACT zeros (summed dimension)
        pack16 v;
        v.u[0] = v.u[1] = v.u[2] = v.u[3] = 0;
        *(pack16*)&DYt[buf][row][k8] = v;
      }
    }
  };
  const int wm = wave >> 1, wn = wave & 1;  // 2x2 waves
  constexpr int MW = BMK / 2 / 16;          // m-frags per wave (2 or 4)
  constexpr int FN = BNF / 2 / 16;          // 32(BNF=64) or 64(128)/16
  const int m0 = wm * (BMK / 2), n0 = wn * (BNF / 2);

  f32x4 acc[MW][FN];
#pragma unroll
  for (int i = 0; i < MW; ++i)
#pragma unroll
    for (int j = 0; j < FN; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  // fused bias grad: thread j < BMK owns kout column j of this block's
  // dy tiles (only fTile==0 blocks participate; split-K slices cover
  // disjoint pixel sets, atomics combine)
  const bool doBias = (a.db != nullptr) && (fTile == 0);
  float dbAcc = 0.f;

  const int nchunks = (a.M + BKP - 1) / BKP;
  const int nci = (nchunks - slice + a.splitK - 1) / a.splitK;
  if (slice < nchunks) {
#pragma unroll
    for (int d = 0; d < DEPTH - 1; ++d)
      if (d < nci) stage(d % DEPTH, (slice + (long)d * a.splitK) * BKP);
    for (int ci = 0; ci < nci; ++ci) {
      const int tnext = ci + DEPTH - 1;
      if (tnext < nci)
        stage(tnext % DEPTH,
              (slice + (long)tnext * a.splitK) * BKP);
      const int newer = (nci - 1 - ci) < (DEPTH - 1) ? (nci - 1 - ci)
                                                     : (DEPTH - 1);
      if (newer == 2)
        asm volatile("s_waitcnt vmcnt(%0) lgkmcnt(0)" ::"i"(2 * TPW)
                     : "memory");
      else if (newer == 1)
        asm volatile("s_waitcnt vmcnt(%0) lgkmcnt(0)" ::"i"(TPW)
                     : "memory");
      else
        asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
      const int buf = ci % DEPTH;
      if (doBias && tid < BMK) {
        // logical col j of row r sits at the staging-swizzled slot
        const int j = tid;
        const int jg = j & ~7, jo = j & 7;
        for (int r2 = 0; r2 < BKP; ++r2) {
          int slot = (jg ^ ((r2 & 7) * 8) ^ (((r2 >> 3) & 3)
                                            * (BMK / 4))) + jo;
          dbAcc += (float)DYt[buf][r2][slot];
        }
      }
#pragma unroll
      for (int ks = 0; ks < BKP; ks += 32) {
        bf16x8 af[MW], bfr[FN];
#pragma unroll
        for (int i = 0; i < MW; ++i)
          af[i] = frag_load_tr_swz<BMK>(&DYt[buf][0][0], ks,
                                        m0 + i * 16, lane);
#pragma unroll
        for (int j = 0; j < FN; ++j)
          bfr[j] = frag_load_tr_swz<BNF>(&XT[buf][0][0], ks,
                                         n0 + j * 16, lane);
#pragma unroll
        for (int i = 0; i < MW; ++i)
#pragma unroll
          for (int j = 0; j < FN; ++j)
            acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                af[i], bfr[j], acc[i][j], 0, 0, 0);
      }
      __builtin_amdgcn_s_barrier();
    }
  }

  if (doBias && tid < BMK) {
    int kout = kout0 + tid;
    if (kout < a.Kgp) atomicAdd(&a.db[g * a.Kgp + kout], dbAcc);
  }

  // epilogue: fp32 accumulate.  splitK == 1: ONE writer per dW element
  // -> plain stores (the FC wgrads were atomics-bound).  splitK > 1
  // with WS: each slice stores plain into its own workspace region and
  // segsum reduces afterwards.  Otherwise: atomics.
  const bool plain = (a.splitK == 1);
  float* out = a.DW;
  if (!plain && a.WS)
    out = a.WS + (long)slice * a.Kgp * a.Kdf * gridDim.z / a.splitK;
#pragma unroll
  for (int i = 0; i < MW; ++i) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      int kout = kout0 + m0 + i * 16 + (lane >> 4) * 4 + reg;
      if (kout >= a.Kgp) continue;
      long base = ((long)g * a.Kgp + kout) * a.Kdf + fd0;
#pragma unroll
      for (int j = 0; j < FN; ++j) {
        int fd = n0 + j * 16 + (lane & 15);
        if (fd0 + fd >= a.Kdf) continue;
        // fold the pad-slot mask in here (saves a full dw pass)
        float v = acc[i][j][reg] * a.kdfMask[fd0 + fd];
        if (plain || a.WS)
          out[base + fd] = v;
        else
          atomicAdd(&a.DW[base + fd], v);
      }
    }
  }
}

// ---------------------------------------------------------------------------

// block = CLW channel-group lanes x (256/CLW) row-lanes; LDS-reduce
// the row lanes so each block issues at most 256 atomics (the naive
// version serialized on ~1M atomics over a few hundred addresses).
// CLW adapts to the layer width: the fixed 32-wide split idled 1/2 to
// 3/4 of every block on narrow layers (C 64-128 — most of GoogLeNet),
// dividing effective load bandwidth by the idle fraction.
template <int CLW>
__global__ void bias_wgrad_k(const __bf16* __restrict__ dy,
                             float* __restrict__ db, long rows, int C) {
  constexpr int RLN = 256 / CLW;
  const int c8n = C / 8;
  const int cl = threadIdx.x & (CLW - 1);   // channel-group lane
  const int rl = threadIdx.x / CLW;         // row lane (0..RLN-1)
  const int c8 = blockIdx.x * CLW + cl;
  const long sl = (long)blockIdx.y * RLN + rl;
  const long nsl = (long)gridDim.y * RLN;
  float s[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  if (c8 < c8n) {
    const __bf16* base = dy + c8 * 8;
    // 4 consecutive rows per iteration: 4 independent 16-B loads in
    // flight per thread (the 1-deep version was latency-bound)
    long r4 = sl * 4;
    for (; r4 + 4 <= rows; r4 += nsl * 4) {
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        bf16x8 v = *(const bf16x8*)(base + (r4 + u) * C);
#pragma unroll
        for (int j = 0; j < 8; ++j) s[j] += (float)v[j];
      }
    }
    // tail rows (rows % 4): once, by the first row-lane of y-block 0
    if (blockIdx.y == 0 && rl == 0) {
      for (long r = rows & ~3L; r < rows; ++r) {
        bf16x8 v = *(const bf16x8*)(base + r * C);
#pragma unroll
        for (int j = 0; j < 8; ++j) s[j] += (float)v[j];
      }
    }
  }
  __shared__ float red[RLN][CLW][8];        // 16 KB for any CLW
#pragma unroll
  for (int j = 0; j < 8; ++j) red[rl][cl][j] = s[j];
  __syncthreads();
  // final reduce spread over (channel-group, j) pairs: CLW*8 threads
  // each sum RLN values — the rl==0-only form ran 8 threads x 256
  // serial adds at CLW=8
  int t = threadIdx.x;
  if (t < CLW * 8) {
    int cl2 = t / 8, j = t & 7;
    int c82 = blockIdx.x * CLW + cl2;
    if (c82 < c8n) {
      float a = 0.f;
#pragma unroll
      for (int r = 0; r < RLN; ++r) a += red[r][cl2][j];
      atomicAdd(&db[c82 * 8 + j], a);
    }
  }
}


// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------

extern "C" {

hipError_t launch_splitk_finalize(float* y32, const float* bias,
                                  void* y, long total, int C, int relu,
                                  hipStream_t stream) {
  long g = (total + 255) / 256;
  if (g > 2048) g = 2048;
  hipLaunchKernelGGL(splitk_finalize_k, dim3((int)g), dim3(256), 0,
                     stream, y32, bias, (__bf16*)y, total, C, relu);
  return hipGetLastError();
}

// Variant table for the per-shape autotuner (tools/autotune_conv.py;
// table shipped as theanompi_amd/ops/conv_tuned.json, threaded through
// the bindings as the `variant` argument):
//   0  auto (BN by width heuristic + env overrides — round-1 behavior)
//   1  128x128 D2 BK64          5  128x128 D2 BK32
//   2  128x 64 D2 BK64          6  128x 64 D2 BK32
//   3  256x128 D2 BK64 (8 wv)   7  128x128 D3 BK32
//   4  256x128 8-phase          8  128x 64 D3 BK32
//                               9  128x 64 D3 BK64
//                              10  128x128 D3 BK64
// Slimmer tiles (BN=64 / BK=32 / dynamic gather table) trade MFMA work
// per staged byte for occupancy — the round-1 PMC data shows the
// default tile L2-LATENCY bound (WAIT_ANY ~50%) on mid-size shapes, so
// more blocks/CU is the lever, measured per shape, not guessed.
hipError_t launch_conv_gather_gemm(GatherGemmArgs* args, int G,
                                   int layout, hipStream_t stream,
                                   int variant) {
  GatherGemmArgs a = *args;
  if (a.Kdf > 2048 * 16) return hipErrorInvalidValue;  // LDS table cap
  const int tabBytes = ((a.Kdf >> 4) * 4 + 15) & ~15;
  // fused BN stats only exist on the default LDS-staged epilogue with
  // splitK == 1; clear the field (and tell the caller) on every other
  // route so the host falls back to the separate bn_sums kernel
  if (a.splitK > 1) { a.stats = nullptr; args->stats = nullptr; }

  static int bigTile = -1;
  if (bigTile < 0) {
    const char* e = getenv("THEANOMPI_GEMM_BIG");
    bigTile = e ? atoi(e) : 0;
  }
  static int bk32 = -1;
  if (bk32 < 0) {
    const char* e = getenv("THEANOMPI_GEMM_BK32");
    bk32 = e ? atoi(e) : 0;
  }
  static int forceVar = -1;
  if (forceVar < 0) {
    const char* e = getenv("THEANOMPI_GEMM_VARIANT");
    forceVar = e ? atoi(e) : 0;
  }
  if (forceVar) variant = forceVar;

  // map the legacy env switches onto variants
  if (variant == 0) {
    if (bigTile && a.splitK == 1 && a.M >= 4096 && a.BNtot >= 128
        && a.Kdf >= 128)
      variant = (bigTile == 2 && a.Kdf >= 3 * 64) ? 4 : 3;
    else if (bk32)
      variant = a.BNtot > 64 ? 5 : 6;
    else
      variant = a.BNtot > 64 ? 1 : 2;
  }
  // feasibility clamps
  if (variant == 4 && a.Kdf < 3 * 64) variant = 3;
  if ((variant == 3 || variant == 4) && a.splitK > 1) {
    // big tiles carry no split-K path; drop the split (caller's
    // finalize is skipped because args->splitK is rewritten)
    a.splitK = 1;
    args->splitK = 1;
    a.Y32 = nullptr;
    args->Y32 = nullptr;
  }
  // fused stats ride only the LDS-staged bf16 epilogue (2*BK >= BN)
  const bool ldsep = (variant == 1 || variant == 3 || variant == 10)
                     ? true
                     : (variant == 2 || variant == 6 || variant == 8
                        || variant == 9);
  if (!ldsep || variant == 4) { a.stats = nullptr; args->stats = nullptr; }

  const int BM = (variant == 3 || variant == 4) ? 256 : 128;
  const int BN = (variant == 2 || variant == 6 || variant == 8
                  || variant == 9 || variant == 12
                  || variant == 13) ? 64 : 128;
  dim3 grid(cdiv(a.M, BM), cdiv(a.BNtot, BN), G * a.splitK);
  dim3 block(BM * 2);

#define GG(BM_, BN_, D_, BK_)                                              \
  do {                                                                     \
    if (layout == FRAG_CONTIG8)                                            \
      hipLaunchKernelGGL((conv_gather_gemm<BM_, BN_, FRAG_CONTIG8, D_,     \
                                           BK_>),                          \
                         grid, block, tabBytes, stream, a);                \
    else                                                                   \
      hipLaunchKernelGGL((conv_gather_gemm<BM_, BN_, FRAG_HALVES, D_,      \
                                           BK_>),                          \
                         grid, block, tabBytes, stream, a);                \
  } while (0)
  // NFAST variants (11-13) need gx % 8 == 0 and > 1 n-tile; fall back
  // to their m-fast twins otherwise
  if (variant >= 11 && variant <= 13
      && !(grid.x % 8 == 0 && grid.y > 1)) {
    variant = (variant == 11) ? 1 : (variant == 12) ? 6 : 2;
  }
#define GGN(BM_, BN_, D_, BK_)                                             \
  do {                                                                     \
    if (layout == FRAG_CONTIG8)                                            \
      hipLaunchKernelGGL((conv_gather_gemm<BM_, BN_, FRAG_CONTIG8, D_,     \
                                           BK_, 1>),                       \
                         grid, block, tabBytes, stream, a);                \
    else                                                                   \
      hipLaunchKernelGGL((conv_gather_gemm<BM_, BN_, FRAG_HALVES, D_,      \
                                           BK_, 1>),                       \
                         grid, block, tabBytes, stream, a);                \
  } while (0)
  switch (variant) {
    case 3: GG(256, 128, 2, 64); break;
    case 4:
      if (layout == FRAG_CONTIG8)
        hipLaunchKernelGGL((conv_gather_gemm_8p<FRAG_CONTIG8>), grid,
                           block, tabBytes, stream, a);
      else
        hipLaunchKernelGGL((conv_gather_gemm_8p<FRAG_HALVES>), grid,
                           block, tabBytes, stream, a);
      break;
    case 2: GG(128, 64, 2, 64); break;
    case 5: GG(128, 128, 2, 32); break;
    case 6: GG(128, 64, 2, 32); break;
    case 7: GG(128, 128, 3, 32); break;
    case 8: GG(128, 64, 3, 32); break;
    case 9: GG(128, 64, 3, 64); break;
    case 10: GG(128, 128, 3, 64); break;
    case 11: GGN(128, 128, 2, 64); break;
    case 12: GGN(128, 64, 2, 32); break;
    case 13: GGN(128, 64, 2, 64); break;
    case 1:
    default: GG(128, 128, 2, 64); break;
  }
#undef GGN
#undef GG
  return hipGetLastError();
}

// wgrad variants (autotuner): 0 auto (BNF heuristic + env), 1 BNF=64,
// 2 BNF=128, 3 BNF=64/BMK=128, 4 BNF=128/BMK=128, and 3-deep-ring
// versions 5 (64/64/D3), 6 (128/64/D3), 7 (64/128/D3).  The gather
// table is dynamically sized here too: at BNF=128 that is the
// difference between 2 and 3 blocks/CU.
hipError_t launch_conv_wgrad(const WgradArgs* args, int G, int layout,
                             hipStream_t stream, int variant) {
  WgradArgs a = *args;
  if (a.Kdf > 2048 * 16) return hipErrorInvalidValue;
  const int tabBytes = ((a.Kdf >> 4) * 4 + 15) & ~15;
  static const int bnf_env = [] {
    const char* e = getenv("THEANOMPI_WGRAD_BNF");
    return e ? atoi(e) : 0;
  }();
  // 64-column tiles (occupancy 4) measured faster than the old
  // dense-Kdf 128-tile heuristic (occupancy 2) once the reduction
  // epilogues were parallelized; THEANOMPI_WGRAD_BNF=128 restores it.
  int BNF = 64;
  if (bnf_env == 128 && a.Kdf % 128 == 0) BNF = 128;
  if (variant == 1 || variant == 3 || variant == 5 || variant == 7
      || variant == 8)
    BNF = 64;
  if ((variant == 2 || variant == 4 || variant == 6)
      && a.Kdf % 128 == 0)
    BNF = 128;
  int BMK = (variant == 3 || variant == 4 || variant == 7) ? 128 : 64;
  int D = (variant >= 5 && variant <= 7) ? 3 : 2;
  int BKP = (variant == 8) ? 128 : 64;
  dim3 grid(cdiv(a.Kgp, BMK), cdiv(a.Kdf, BNF), G * a.splitK);
  dim3 block(256);
#define WG(BNF_, BMK_, D_, BKP_)                                           \
  do {                                                                     \
    if (layout == FRAG_CONTIG8)                                            \
      hipLaunchKernelGGL((conv_wgrad_k<FRAG_CONTIG8, BNF_, BMK_, D_,       \
                                       BKP_>),                             \
                         grid, block, tabBytes, stream, a);                \
    else                                                                   \
      hipLaunchKernelGGL((conv_wgrad_k<FRAG_HALVES, BNF_, BMK_, D_,        \
                                       BKP_>),                             \
                         grid, block, tabBytes, stream, a);                \
  } while (0)
  if (BKP == 128) {
    WG(64, 64, 2, 128);
  } else if (D == 3) {
    if (BMK == 128) WG(64, 128, 3, 64);
    else if (BNF == 128) WG(128, 64, 3, 64);
    else WG(64, 64, 3, 64);
  } else if (BMK == 128) {
    if (BNF == 128) WG(128, 128, 2, 64);
    else WG(64, 128, 2, 64);
  } else if (BNF == 128) {
    WG(128, 64, 2, 64);
  } else {
    WG(64, 64, 2, 64);
  }
#undef WG
  return hipGetLastError();
}

hipError_t launch_bias_wgrad(const void* dy, float* db, long rows, int C,
                             hipStream_t stream) {
  int c8n = C / 8;
  int clw = c8n >= 32 ? 32 : (c8n >= 16 ? 16 : 8);
  int gx = cdiv(c8n, clw);
  int rln = 256 / clw;
  long gy = (rows + rln * 4 - 1) / (rln * 4);
  long target = 1024 / gx;                 // bounded atomics per channel
  if (target > 128) target = 128;
  if (gy > target) gy = target;
  if (gy < 1) gy = 1;
  dim3 grid(gx, (unsigned)gy);
  if (clw == 32)
    hipLaunchKernelGGL(bias_wgrad_k<32>, grid, dim3(256), 0, stream,
                       (const __bf16*)dy, db, rows, C);
  else if (clw == 16)
    hipLaunchKernelGGL(bias_wgrad_k<16>, grid, dim3(256), 0, stream,
                       (const __bf16*)dy, db, rows, C);
  else
    hipLaunchKernelGGL(bias_wgrad_k<8>, grid, dim3(256), 0, stream,
                       (const __bf16*)dy, db, rows, C);
  return hipGetLastError();
}


}  // extern "C"
