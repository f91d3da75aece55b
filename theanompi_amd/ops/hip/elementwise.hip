// Elementwise / data-movement kernels:
// halo padding, relu fwd/bwd, residual add, dropout (counter-based RNG),
// repack gather, bf16<->fp32 pack.  All memory-bound: vectorized 8-16 B
// per lane, grid-stride.

#include "common.hpp"

// ---- halo pad -------------------------------------------------------------
// x [N,H,W,C] -> y [N, H+t+b, W+l+r, C], borders = fill. C % 4 == 0 so
// 4-element (8 B) chunks never straddle a pixel.

__global__ void halo_pad_k(const __bf16* __restrict__ x,
                           __bf16* __restrict__ y, int N, int H, int W,
                           int C, int t, int b, int l, int r, float fill,
                           long srcC, long total4) {
  // srcC: pixel stride of the source (== C, or wider when x is a
  // channel slice of a merged-branch buffer)
  const int Hp = H + t + b, Wp = W + l + r;
  const __bf16 f = (__bf16)fill;
  for (long i4 = blockIdx.x * blockDim.x + threadIdx.x; i4 < total4;
       i4 += (long)gridDim.x * blockDim.x) {
    long i = i4 * 4;
    int c = (int)(i % C);
    long pix = i / C;
    int wp = (int)(pix % Wp);
    long tmp = pix / Wp;
    int hp = (int)(tmp % Hp);
    int n = (int)(tmp / Hp);
    int h = hp - t, w = wp - l;
    pack8 v;
    if (h >= 0 && h < H && w >= 0 && w < W) {
      v = *(const pack8*)(x + (((long)n * H + h) * W + w) * srcC + c);
    } else {
      __bf16* e = (__bf16*)&v;
      e[0] = f; e[1] = f; e[2] = f; e[3] = f;
    }
    *(pack8*)(y + i) = v;
  }
}

// ---- relu / add -----------------------------------------------------------

__global__ void relu_fwd_k(const __bf16* __restrict__ x,
                           __bf16* __restrict__ y, long n4) {
  const __bf16 z = (__bf16)0.f;
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += (long)gridDim.x * blockDim.x) {
    bf16x4 v = *(const bf16x4*)(x + i * 4);
#pragma unroll
    for (int j = 0; j < 4; ++j) v[j] = v[j] > z ? v[j] : z;
    *(bf16x4*)(y + i * 4) = v;
  }
}

__global__ void relu_bwd_k(const __bf16* __restrict__ dy,
                           const __bf16* __restrict__ y,
                           __bf16* __restrict__ dx, long n4) {
  const __bf16 z = (__bf16)0.f;
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += (long)gridDim.x * blockDim.x) {
    bf16x4 g = *(const bf16x4*)(dy + i * 4);
    bf16x4 v = *(const bf16x4*)(y + i * 4);
#pragma unroll
    for (int j = 0; j < 4; ++j) g[j] = v[j] > z ? g[j] : z;
    *(bf16x4*)(dx + i * 4) = g;
  }
}

// 16-B, 2-deep-unrolled variant: the 8-B kernels above are
// LATENCY-bound on big tensors (measured ~2.6x off roofline on the
// ResNet-50 residual adds) — four independent 16-B loads in flight
// per thread instead of two dependent 8-B ones.
__global__ void relu_bwd_k16(const __bf16* __restrict__ dy,
                             const __bf16* __restrict__ y,
                             __bf16* __restrict__ dx, long n8) {
  const __bf16 z = (__bf16)0.f;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += 2 * stride) {
    long i2 = i + stride;
    bf16x8 g0 = *(const bf16x8*)(dy + i * 8);
    bf16x8 v0 = *(const bf16x8*)(y + i * 8);
    bf16x8 g1, v1;
    if (i2 < n8) {
      g1 = *(const bf16x8*)(dy + i2 * 8);
      v1 = *(const bf16x8*)(y + i2 * 8);
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) g0[j] = v0[j] > z ? g0[j] : z;
    *(bf16x8*)(dx + i * 8) = g0;
    if (i2 < n8) {
#pragma unroll
      for (int j = 0; j < 8; ++j) g1[j] = v1[j] > z ? g1[j] : z;
      *(bf16x8*)(dx + i2 * 8) = g1;
    }
  }
}

// relu backward with DUAL output: the contiguous masked dy (consumed
// by wgrad/bias) and the same values written into the interior of a
// halo-padded buffer whose borders stay zero (cached per layer) — the
// single-class dgrad then needs no separate halo_pad launch + pass.
// dy/y may be channel slices (uniform per-pixel stride ld*).
__global__ void relu_bwd_dual_k(const __bf16* __restrict__ dy,
                                const __bf16* __restrict__ y,
                                __bf16* __restrict__ dx,
                                __bf16* __restrict__ dxp, long n8,
                                int C, int H, int W, int Hp2, int Wp2,
                                int ht, int hl, long ldDy, long ldY) {
  const __bf16 z = (__bf16)0.f;
  for (long i8 = blockIdx.x * blockDim.x + threadIdx.x; i8 < n8;
       i8 += (long)gridDim.x * blockDim.x) {
    long i = i8 * 8;
    int c = (int)(i % C);
    long pix = i / C;
    int w = (int)(pix % W);
    long tmp = pix / W;
    int h = (int)(tmp % H);
    int n = (int)(tmp / H);
    bf16x8 g = *(const bf16x8*)(dy + pix * ldDy + c);
    bf16x8 v = *(const bf16x8*)(y + pix * ldY + c);
#pragma unroll
    for (int j = 0; j < 8; ++j) g[j] = v[j] > z ? g[j] : z;
    *(bf16x8*)(dx + i) = g;
    long po = ((((long)n * Hp2) + h + ht) * Wp2 + w + hl) * C + c;
    *(bf16x8*)(dxp + po) = g;
  }
}

__global__ void add_relu_k16(const __bf16* __restrict__ p,
                             const __bf16* __restrict__ q,
                             __bf16* __restrict__ y, long n8, int relu) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += 2 * stride) {
    long i2 = i + stride;
    bf16x8 u0 = *(const bf16x8*)(p + i * 8);
    bf16x8 v0 = *(const bf16x8*)(q + i * 8);
    bf16x8 u1, v1;
    if (i2 < n8) {
      u1 = *(const bf16x8*)(p + i2 * 8);
      v1 = *(const bf16x8*)(q + i2 * 8);
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float s = (float)u0[j] + (float)v0[j];
      if (relu && s < 0.f) s = 0.f;
      u0[j] = (__bf16)s;
    }
    *(bf16x8*)(y + i * 8) = u0;
    if (i2 < n8) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float s = (float)u1[j] + (float)v1[j];
        if (relu && s < 0.f) s = 0.f;
        u1[j] = (__bf16)s;
      }
      *(bf16x8*)(y + i2 * 8) = u1;
    }
  }
}

// Strided-source variant: dy and/or y rows of length C sit ldDy/ldY
// elements apart — channel slices of a concat buffer (the inception
// split in backward; the concat-written y in forward).  The masked
// copy this kernel performs anyway subsumes the .contiguous() the
// slices would otherwise need.
__global__ void relu_bwd_s_k(const __bf16* __restrict__ dy,
                             const __bf16* __restrict__ y,
                             __bf16* __restrict__ dx, long n4, int C4,
                             long ldDy, long ldY) {
  const __bf16 z = (__bf16)0.f;
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += (long)gridDim.x * blockDim.x) {
    long row = i / C4;
    long c = (i - row * C4) * 4;
    bf16x4 g = *(const bf16x4*)(dy + row * ldDy + c);
    bf16x4 v = *(const bf16x4*)(y + row * ldY + c);
#pragma unroll
    for (int j = 0; j < 4; ++j) g[j] = v[j] > z ? g[j] : z;
    *(bf16x4*)(dx + i * 4) = g;
  }
}

__global__ void add_relu_k(const __bf16* __restrict__ p,
                           const __bf16* __restrict__ q,
                           __bf16* __restrict__ y, long n4, int relu) {
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += (long)gridDim.x * blockDim.x) {
    bf16x4 u = *(const bf16x4*)(p + i * 4);
    bf16x4 v = *(const bf16x4*)(q + i * 4);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float s = (float)u[j] + (float)v[j];
      if (relu && s < 0.f) s = 0.f;
      u[j] = (__bf16)s;
    }
    *(bf16x4*)(y + i * 4) = u;
  }
}

// ---- dropout --------------------------------------------------------------
// Counter-based RNG (xorshift-mixed, Philox-style statelessness): mask
// regenerable from (seed, index) but we store it for the backward pass.

DEV unsigned int mix32(unsigned long long x) {
  x ^= x >> 33; x *= 0xff51afd7ed558ccdULL;
  x ^= x >> 33; x *= 0xc4ceb9fe1a85ec53ULL;
  x ^= x >> 33;
  return (unsigned int)x;
}

__global__ void dropout_fwd_k(const __bf16* __restrict__ x,
                              __bf16* __restrict__ y,
                              unsigned char* __restrict__ mask, long n,
                              float p,
                              const unsigned long long* __restrict__ seedp) {
  // seed comes from DEVICE memory so a hipGraph replay (which freezes
  // kernel args) still draws fresh masks: the companion bump kernel
  // advances the counter inside the graph.
  const float scale = 1.f / (1.f - p);
  const unsigned int thr = (unsigned int)(p * 4294967296.f);
  const unsigned long long seed = *seedp;
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    unsigned int u = mix32(seed * 0x9E3779B97F4A7C15ULL + (unsigned long long)i);
    bool keep = u >= thr;
    mask[i] = keep;
    y[i] = keep ? (__bf16)((float)x[i] * scale) : (__bf16)0.f;
  }
}

__global__ void seed_bump_k(unsigned long long* seed) {
  if (threadIdx.x == 0 && blockIdx.x == 0) *seed += 1;
}

__global__ void dropout_bwd_k(const __bf16* __restrict__ dy,
                              const unsigned char* __restrict__ mask,
                              __bf16* __restrict__ dx, long n, float p) {
  const float scale = 1.f / (1.f - p);
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x)
    dx[i] = mask[i] ? (__bf16)((float)dy[i] * scale) : (__bf16)0.f;
}

// ---- repack gather (dgrad weight copies, transposes) ----------------------

__global__ void gather_bf16_k(const __bf16* __restrict__ src,
                              const int* __restrict__ map,
                              __bf16* __restrict__ dst, long n) {
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    int m = map[i];
    dst[i] = m < 0 ? (__bf16)0.f : src[m];
  }
}

// ---- fused loader finish: uint8 NHWC -> bf16, x/128 - 1, zero pad chans
// uint8 NHWC -> bf16 on device.  Default normalisation (255-level
// images): x/128 - 1.  With a mean image (fp32, one per-pixel value
// per (h, w, c) of the crop, the reference's subtracted dataset mean
// [pub data/proc_load_mpi.py]): (x - mean) / 128 — the subtraction
// happens HERE so the host still ships uint8.
__global__ void u8_finish_k(const unsigned char* __restrict__ src,
                            __bf16* __restrict__ dst,
                            const float* __restrict__ meanimg,
                            long total4, long perImg, int C, int nLogC) {
  for (long i4 = blockIdx.x * (long)blockDim.x + threadIdx.x; i4 < total4;
       i4 += (long)gridDim.x * blockDim.x) {
    long i = i4 * 4;
    int c = (int)(i % C);
    unsigned int u = *(const unsigned int*)(src + i);
    f32x4 mi = {128.f, 128.f, 128.f, 128.f};
    if (meanimg) mi = *(const f32x4*)(meanimg + (i % perImg));
    bf16x4 o;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float v = ((float)((u >> (8 * j)) & 0xff) - mi[j]) * (1.f / 128.f);
      o[j] = (c + j < nLogC) ? (__bf16)v : (__bf16)0.f;
    }
    *(bf16x4*)(dst + i) = o;
  }
}

// ---- LDS-tiled transpose (FC / 1x1-conv dgrad weight repack) --------------
// dst[c][k] = src[k][c] for k < rows, c < cols; strides in elements.
// 64(col)x32(row) tiles; 8-element vector loads, column gathers from LDS.

__device__ __forceinline__ void transpose_tile_body(
    const __bf16* __restrict__ src, __bf16* __restrict__ dst, int rows,
    int cols, long sstride, long dstride, int r0, int c0, int t,
    __bf16 (*tile)[72]);

__global__ void transpose_bf16_k(const __bf16* __restrict__ src,
                                 __bf16* __restrict__ dst, int rows,
                                 int cols, long sstride, long dstride) {
  __shared__ __bf16 tile[32][72];    // +8 pad kills column-read conflicts
  transpose_tile_body(src, dst, rows, cols, sstride, dstride,
                      blockIdx.y * 32, blockIdx.x * 64, threadIdx.x,
                      tile);
}

// one launch covers every dgrad-repack transpose of the step (the
// per-layer launches were 46/step on GoogLeNet — pure launch latency)
struct TransSeg {
  const __bf16* src;
  __bf16* dst;
  int rows, cols;
  long sstride, dstride;
};

__global__ void transpose_multi_k(const TransSeg* __restrict__ segs) {
  __shared__ __bf16 tile[32][72];
  const TransSeg s = segs[blockIdx.z];
  const int r0 = blockIdx.y * 32;
  const int c0 = blockIdx.x * 64;
  if (r0 >= s.rows || c0 >= s.cols) return;
  transpose_tile_body(s.src, s.dst, s.rows, s.cols, s.sstride,
                      s.dstride, r0, c0, threadIdx.x, tile);
}

__device__ __forceinline__ void transpose_tile_body(
    const __bf16* __restrict__ src, __bf16* __restrict__ dst, int rows,
    int cols, long sstride, long dstride, int r0, int c0, int t,
    __bf16 (*tile)[72]) {
  {
    const int row = t >> 3;          // 0..31
    const int cp = (t & 7) * 8;      // 0..56
    int r = r0 + row;
    if (r < rows) {
      int c = c0 + cp;
      // clamp the 8-wide load inside the row
      if (c + 8 <= cols) {
        *(bf16x8*)&tile[row][cp] =
            *(const bf16x8*)(src + (long)r * sstride + c);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          tile[row][cp + j] = (c + j < cols)
                                  ? src[(long)r * sstride + c + j]
                                  : (__bf16)0.f;
      }
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) tile[row][cp + j] = (__bf16)0.f;
    }
  }
  __syncthreads();
  {
    const int col = t >> 2;          // 0..63 (source col = dst row)
    const int rp = (t & 3) * 8;      // 0..24
    int c = c0 + col;
    if (c >= cols) return;
    bf16x8 v;
#pragma unroll
    for (int j = 0; j < 8; ++j) v[j] = tile[rp + j][col];
    long r = r0 + rp;
    if (r + 8 <= rows) {
      *(bf16x8*)(dst + (long)c * dstride + r) = v;
    } else {
      for (int j = 0; j < 8 && r + j < rows; ++j)
        dst[(long)c * dstride + r + j] = v[j];
    }
  }
}

// ---- pack / unpack (N13: bf16 wire format) --------------------------------

__global__ void pack_bf16_k(const float* __restrict__ src,
                            __bf16* __restrict__ dst, long n4) {
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += (long)gridDim.x * blockDim.x) {
    f32x4 v = *(const f32x4*)(src + i * 4);
    bf16x4 o;
#pragma unroll
    for (int j = 0; j < 4; ++j) o[j] = (__bf16)v[j];
    *(bf16x4*)(dst + i * 4) = o;
  }
}

__global__ void unpack_bf16_k(const __bf16* __restrict__ src,
                              float* __restrict__ dst, long n4) {
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += (long)gridDim.x * blockDim.x) {
    bf16x4 v = *(const bf16x4*)(src + i * 4);
    f32x4 o;
#pragma unroll
    for (int j = 0; j < 4; ++j) o[j] = (float)v[j];
    *(f32x4*)(dst + i * 4) = o;
  }
}

// ---- N14: segment sum for the asa_xgmi exchange ---------------------------
// Rebuild of the reference's hand-written ASA segment-sum CUDA kernel
// (exchanger_strategy.py [pub]): after the all-to-all
// phase each rank holds `w` remote copies of its 1/world bucket shard;
// out[i] = sum_j in[j*seg + i], accumulated in fp32 even on a bf16 wire.
// seg is a multiple of 4 (the strategy pads buckets to 4*world elems).

template <typename T, typename V4>
__global__ void segsum_k(const T* __restrict__ in, T* __restrict__ out,
                         long seg4, int w) {
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < seg4;
       i += (long)gridDim.x * blockDim.x) {
    float acc[4] = {0.f, 0.f, 0.f, 0.f};
    for (int j = 0; j < w; ++j) {
      V4 v = *(const V4*)(in + (j * seg4 + i) * 4);
#pragma unroll
      for (int k = 0; k < 4; ++k) acc[k] += (float)v[k];
    }
    V4 o;
#pragma unroll
    for (int k = 0; k < 4; ++k) o[k] = (T)acc[k];
    *(V4*)(out + i * 4) = o;
  }
}

// ---------------------------------------------------------------------------

static inline int ew_grid(long work, int block = 256) {
  long g = (work + block - 1) / block;
  if (g > 2048) g = 2048;                 // grid-stride the rest
  if (g < 1) g = 1;
  return (int)g;
}

extern "C" {

hipError_t launch_halo_pad(const void* x, void* y, int N, int H, int W,
                           int C, int t, int b, int l, int r, float fill,
                           long srcC, hipStream_t stream) {
  long total4 = (long)N * (H + t + b) * (W + l + r) * C / 4;
  hipLaunchKernelGGL(halo_pad_k, dim3(ew_grid(total4)), dim3(256), 0,
                     stream, (const __bf16*)x, (__bf16*)y, N, H, W, C, t,
                     b, l, r, fill, srcC, total4);
  return hipGetLastError();
}

hipError_t launch_relu_fwd(const void* x, void* y, long n,
                           hipStream_t stream) {
  hipLaunchKernelGGL(relu_fwd_k, dim3(ew_grid(n / 4)), dim3(256), 0,
                     stream, (const __bf16*)x, (__bf16*)y, n / 4);
  return hipGetLastError();
}

hipError_t launch_relu_bwd(const void* dy, const void* y, void* dx, long n,
                           hipStream_t stream) {
  if (n % 8 == 0) {
    hipLaunchKernelGGL(relu_bwd_k16, dim3(ew_grid(n / 16)), dim3(256),
                       0, stream, (const __bf16*)dy, (const __bf16*)y,
                       (__bf16*)dx, n / 8);
    return hipGetLastError();
  }
  hipLaunchKernelGGL(relu_bwd_k, dim3(ew_grid(n / 4)), dim3(256), 0,
                     stream, (const __bf16*)dy, (const __bf16*)y,
                     (__bf16*)dx, n / 4);
  return hipGetLastError();
}

hipError_t launch_relu_bwd_dual(const void* dy, const void* y, void* dx,
                                void* dxp, long n, int C, int H, int W,
                                int Hp2, int Wp2, int ht, int hl,
                                long ldDy, long ldY,
                                hipStream_t stream) {
  hipLaunchKernelGGL(relu_bwd_dual_k, dim3(ew_grid(n / 8)), dim3(256),
                     0, stream, (const __bf16*)dy, (const __bf16*)y,
                     (__bf16*)dx, (__bf16*)dxp, n / 8, C, H, W, Hp2,
                     Wp2, ht, hl, ldDy, ldY);
  return hipGetLastError();
}

hipError_t launch_relu_bwd_s(const void* dy, const void* y, void* dx,
                             long n, int C, long ldDy, long ldY,
                             hipStream_t stream) {
  hipLaunchKernelGGL(relu_bwd_s_k, dim3(ew_grid(n / 4)), dim3(256), 0,
                     stream, (const __bf16*)dy, (const __bf16*)y,
                     (__bf16*)dx, n / 4, C / 4, ldDy, ldY);
  return hipGetLastError();
}

hipError_t launch_add_relu(const void* p, const void* q, void* y, long n,
                           int relu, hipStream_t stream) {
  if (n % 8 == 0) {
    hipLaunchKernelGGL(add_relu_k16, dim3(ew_grid(n / 16)), dim3(256),
                       0, stream, (const __bf16*)p, (const __bf16*)q,
                       (__bf16*)y, n / 8, relu);
    return hipGetLastError();
  }
  hipLaunchKernelGGL(add_relu_k, dim3(ew_grid(n / 4)), dim3(256), 0,
                     stream, (const __bf16*)p, (const __bf16*)q,
                     (__bf16*)y, n / 4, relu);
  return hipGetLastError();
}

hipError_t launch_dropout_fwd(const void* x, void* y, unsigned char* mask,
                              long n, float p,
                              const unsigned long long* seedp,
                              hipStream_t stream) {
  hipLaunchKernelGGL(seed_bump_k, dim3(1), dim3(64), 0, stream,
                     (unsigned long long*)seedp);
  hipLaunchKernelGGL(dropout_fwd_k, dim3(ew_grid(n)), dim3(256), 0, stream,
                     (const __bf16*)x, (__bf16*)y, mask, n, p, seedp);
  return hipGetLastError();
}

hipError_t launch_dropout_bwd(const void* dy, const unsigned char* mask,
                              void* dx, long n, float p,
                              hipStream_t stream) {
  hipLaunchKernelGGL(dropout_bwd_k, dim3(ew_grid(n)), dim3(256), 0, stream,
                     (const __bf16*)dy, mask, (__bf16*)dx, n, p);
  return hipGetLastError();
}

hipError_t launch_u8_finish(const unsigned char* src, void* dst,
                            const float* meanimg, long total, long perImg,
                            int C, int nLogC, hipStream_t stream) {
  long t4 = total / 4;
  hipLaunchKernelGGL(u8_finish_k, dim3(ew_grid(t4)), dim3(256), 0, stream,
                     src, (__bf16*)dst, meanimg, t4, perImg, C, nLogC);
  return hipGetLastError();
}

hipError_t launch_transpose_multi(const void* segs_dev, int nseg,
                                  int max_rows, int max_cols,
                                  hipStream_t stream) {
  dim3 grid((max_cols + 63) / 64, (max_rows + 31) / 32, nseg);
  hipLaunchKernelGGL(transpose_multi_k, grid, dim3(256), 0, stream,
                     (const TransSeg*)segs_dev);
  return hipGetLastError();
}

hipError_t launch_transpose_bf16(const void* src, void* dst, int rows,
                                 int cols, long sstride, long dstride,
                                 hipStream_t stream) {
  dim3 grid(cdiv(cols, 64), cdiv(rows, 32));
  hipLaunchKernelGGL(transpose_bf16_k, grid, dim3(256), 0, stream,
                     (const __bf16*)src, (__bf16*)dst, rows, cols,
                     sstride, dstride);
  return hipGetLastError();
}

hipError_t launch_gather_bf16(const void* src, const int* map, void* dst,
                              long n, hipStream_t stream) {
  hipLaunchKernelGGL(gather_bf16_k, dim3(ew_grid(n)), dim3(256), 0, stream,
                     (const __bf16*)src, map, (__bf16*)dst, n);
  return hipGetLastError();
}

hipError_t launch_pack_bf16(const float* src, void* dst, long n,
                            hipStream_t stream) {
  hipLaunchKernelGGL(pack_bf16_k, dim3(ew_grid(n / 4)), dim3(256), 0,
                     stream, src, (__bf16*)dst, n / 4);
  return hipGetLastError();
}

hipError_t launch_segsum(const void* in, void* out, long seg, int w,
                         int is_bf16, hipStream_t stream) {
  long seg4 = seg / 4;
  if (is_bf16) {
    hipLaunchKernelGGL((segsum_k<__bf16, bf16x4>), dim3(ew_grid(seg4)),
                       dim3(256), 0, stream, (const __bf16*)in,
                       (__bf16*)out, seg4, w);
  } else {
    hipLaunchKernelGGL((segsum_k<float, f32x4>), dim3(ew_grid(seg4)),
                       dim3(256), 0, stream, (const float*)in,
                       (float*)out, seg4, w);
  }
  return hipGetLastError();
}

hipError_t launch_unpack_bf16(const void* src, float* dst, long n,
                              hipStream_t stream) {
  hipLaunchKernelGGL(unpack_bf16_k, dim3(ew_grid(n / 4)), dim3(256), 0,
                     stream, (const __bf16*)src, dst, n / 4);
  return hipGetLastError();
}

}  // extern "C"
