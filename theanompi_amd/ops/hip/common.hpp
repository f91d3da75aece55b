// Common definitions for the theanompi_amd CDNA4 (gfx950) kernel library.
// Hand-written HIP targeting MI355X: wave64, MFMA bf16 matrix cores,
// 160 KiB LDS per CU.  No CUDA-compat paths, no hipify, no library math.
#pragma once

#include <hip/hip_runtime.h>

#define DEV __device__ __forceinline__

typedef __attribute__((ext_vector_type(2))) float f32x2;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) short short4v;
typedef __attribute__((ext_vector_type(8))) short short8v;

// 16-byte and 8-byte global load/store carriers with explicit (low)
// alignment so pixel bases aligned only to the channel pad (8 B when
// C_phys == 4) are legal.
struct __align__(8) pack16 { unsigned int u[4]; };
struct __align__(4) pack8  { unsigned int u[2]; };

static inline int cdiv(int a, int b) { return (a + b - 1) / b; }

DEV float bf2f(__bf16 v) { return (float)v; }
DEV __bf16 f2bf(float v) { return (__bf16)v; }

// ---------------------------------------------------------------------------
// MFMA fragment addressing (v_mfma_f32_16x16x32_bf16)
//
// C/D layout (fixed, dtype-independent on gfx950): lane l holds 4 fp32,
//   col = l & 15, row = (l >> 4) * 4 + reg.
// A/B input layout: lane l holds 8 bf16 along K at row/col (l & 15).
// The K offsets of those 8 registers are probed at runtime (bindings.cpp
// mfma_probe): either 8 contiguous (koff = (l>>4)*8 + j) or two halves
// of 4 (koff = (l>>4)*4 + j%4 + (j/4)*16).  Kernels take LAYOUT as a
// template parameter and the host picks the instantiation once.
// ---------------------------------------------------------------------------

#define FRAG_CONTIG8 0
#define FRAG_HALVES  1

// Load one A/B fragment (8 bf16 along K) for lane `lane` from an LDS row
// starting at `row_ptr` (element pointer to k-offset 0 of the 32-wide
// K-step).  Rows are padded so 16-B alignment holds for layout 0 and 8-B
// for layout 1.
template <int LAYOUT>
DEV bf16x8 frag_load(const __bf16* row_ptr, int lane) {
  if constexpr (LAYOUT == FRAG_CONTIG8) {
    return *(const bf16x8*)(row_ptr + (lane >> 4) * 8);
  } else {
    bf16x4 lo = *(const bf16x4*)(row_ptr + (lane >> 4) * 4);
    bf16x4 hi = *(const bf16x4*)(row_ptr + (lane >> 4) * 4 + 16);
    bf16x8 r;
    r[0] = lo[0]; r[1] = lo[1]; r[2] = lo[2]; r[3] = lo[3];
    r[4] = hi[0]; r[5] = hi[1]; r[6] = hi[2]; r[7] = hi[3];
    return r;
  }
}

// Fragment load with the content XOR-swizzle used by the glds-staged
// GEMM tiles: LDS slot (row, p8) holds the source 8-element piece
// p8 ^ (row & 7), so reads XOR the same pattern back (guide T2/rule 21:
// swizzle on the SOURCE address + the read, LDS destination linear).
template <int LAYOUT, int BK = 64>
DEV bf16x8 frag_load_swz(const __bf16* row_start, int row, int ks,
                         int lane) {
  // swizzle span = the row's BK elements (BK/8 eight-element slots)
  const int x8 = (row & (BK / 8 - 1)) * 8;
  if constexpr (LAYOUT == FRAG_CONTIG8) {
    int koff = ks + (lane >> 4) * 8;
    return *(const bf16x8*)(row_start + (koff ^ x8));
  } else {
    int k1 = (ks + (lane >> 4) * 4) ^ x8;
    int k2 = (ks + (lane >> 4) * 4 + 16) ^ x8;
    bf16x4 lo = *(const bf16x4*)(row_start + k1);
    bf16x4 hi = *(const bf16x4*)(row_start + k2);
    bf16x8 r;
    r[0] = lo[0]; r[1] = lo[1]; r[2] = lo[2]; r[3] = lo[3];
    r[4] = hi[0]; r[5] = hi[1]; r[6] = hi[2]; r[7] = hi[3];
    return r;
  }
}

// Transpose-read fragment load for the wgrad TN tiles.
//
// ds_read_b64_tr_b16 semantics on gfx950 (empirically probed, see
// tests/gpu tr probe): within each 16-lane group, lane l's element j
// comes from lds[align8B(addr of lane 16g+4j+((l&15)>>2)) + (l&3)].
// With a LINEAR [pix][64ch] row-major LDS image whose 8-element slots
// are content-swizzled by slot^ (pix&7) (same convention as the glds
// GEMM tiles), lane s passes
//   A_s = pix_s*64 + ((ch0 + 4*(s&3)) ^ ((pix_s&7)*8)),
//   pix_s = ks + 8*(s>>4) + ((s&15)>>2)
// and two reads (pix_s, pix_s+4) give lane l the 8 k-major elements
// (ch = ch0 + (l&15), pix = ks + (l>>4)*8 + 0..7) of an MFMA fragment
// in FRAG_CONTIG8 order.
typedef __attribute__((address_space(3))) bf16x4 lds_bf16x4;

template <int LDW>
DEV bf16x8 frag_load_tr_swz(const __bf16* tile, int ks, int ch0,
                            int lane) {
  const int pix1 = ks + 8 * (lane >> 4) + ((lane & 15) >> 2);
  const int pix2 = pix1 + 4;
  const int co = ch0 + 4 * (lane & 3);
  // swizzle: (pix&7) term decorrelates rows; the (pix>>3) term
  // decorrelates the four 16-lane groups, whose pixel octaves share a
  // bank phase (without it the tr-reads ran 4-8 lanes/bank — measured
  // as SQ_LDS_BANK_CONFLICT = 86% of wgrad's SQ_BUSY; with it the
  // model hits the ideal 2).  Must match the wgrad staging map.
  const int s1 = ((pix1 & 7) * 8) ^ (((pix1 >> 3) & 3) * (LDW / 4));
  const int s2 = ((pix2 & 7) * 8) ^ (((pix2 >> 3) & 3) * (LDW / 4));
  const __bf16* a1 = tile + pix1 * LDW + (co ^ s1);
  const __bf16* a2 = tile + pix2 * LDW + (co ^ s2);
  bf16x4 lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16((lds_bf16x4*)a1);
  bf16x4 hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16((lds_bf16x4*)a2);
  bf16x8 r;
  r[0] = lo[0]; r[1] = lo[1]; r[2] = lo[2]; r[3] = lo[3];
  r[4] = hi[0]; r[5] = hi[1]; r[6] = hi[2]; r[7] = hi[3];
  return r;
}

// async global->LDS copy, 16 B per lane; LDS destination is
// wave-uniform base + lane*16 (hardware adds the lane offset).
typedef __attribute__((address_space(1))) const unsigned int as1_uint;
typedef __attribute__((address_space(3))) unsigned int as3_uint;
DEV void glds16(const void* gsrc, void* lds_base) {
  __builtin_amdgcn_global_load_lds((as1_uint*)gsrc, (as3_uint*)lds_base,
                                   16, 0, 0);
}

#define HIP_KERNEL_ASSERT_OK(expr)                                          \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    if (_e != hipSuccess) return _e;                                        \
  } while (0)
