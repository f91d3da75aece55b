// Pooling kernels, NHWC bf16.  Max pool saves a
// per-element window argmax (u8) so the backward is a pure gather (no
// atomics); avg pool backward is analytic.  Inputs arrive halo-padded
// (max: -inf fill, avg: 0 fill + count includes padding).

#include "common.hpp"

typedef __attribute__((ext_vector_type(8))) unsigned char uchar8;

// thread handles 4 channels of one output pixel.
__global__ void maxpool_fwd_k(const __bf16* __restrict__ xp,
                              __bf16* __restrict__ y,
                              unsigned char* __restrict__ am, int N,
                              int Hp, int Wp, int C, int P, int Q, int kh,
                              int kw, int sh, int sw, long total4) {
  for (long i4 = blockIdx.x * blockDim.x + threadIdx.x; i4 < total4;
       i4 += (long)gridDim.x * blockDim.x) {
    long i = i4 * 4;
    int c = (int)(i % C);
    long pix = i / C;
    int q = (int)(pix % Q);
    long tmp = pix / Q;
    int p = (int)(tmp % P);
    int n = (int)(tmp / P);
    const __bf16* base = xp + (((long)n * Hp + p * sh) * Wp + q * sw) * C + c;
    float best[4] = {-3.0e38f, -3.0e38f, -3.0e38f, -3.0e38f};
    int arg[4] = {0, 0, 0, 0};
    for (int r = 0; r < kh; ++r)
      for (int s = 0; s < kw; ++s) {
        bf16x4 v = *(const bf16x4*)(base + ((long)r * Wp + s) * C);
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          float f = (float)v[j];
          if (f > best[j]) { best[j] = f; arg[j] = r * kw + s; }
        }
      }
    bf16x4 o;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      o[j] = (__bf16)best[j];
      am[i + j] = (unsigned char)arg[j];
    }
    *(bf16x4*)(y + i) = o;
  }
}

// 8-channel (16-B) variant: the 4-wide kernel above issues kh*kw
// dependent 8-B loads per 4 outputs and measured latency-bound on the
// inception pools (~8% of the GoogLeNet step); 16-B transactions halve
// both the loads and the thread count.  C % 8 == 0 (always, padded).
__global__ void maxpool_fwd_k8(const __bf16* __restrict__ xp,
                               __bf16* __restrict__ y,
                               unsigned char* __restrict__ am, int N,
                               int Hp, int Wp, int C, int P, int Q,
                               int kh, int kw, int sh, int sw,
                               long total8) {
  for (long i8 = blockIdx.x * blockDim.x + threadIdx.x; i8 < total8;
       i8 += (long)gridDim.x * blockDim.x) {
    long i = i8 * 8;
    int c = (int)(i % C);
    long pix = i / C;
    int q = (int)(pix % Q);
    long tmp = pix / Q;
    int p = (int)(tmp % P);
    int n = (int)(tmp / P);
    const __bf16* base = xp
        + (((long)n * Hp + p * sh) * Wp + q * sw) * C + c;
    float best[8];
    int arg[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) { best[j] = -3.0e38f; arg[j] = 0; }
    for (int r = 0; r < kh; ++r)
      for (int s = 0; s < kw; ++s) {
        bf16x8 v = *(const bf16x8*)(base + ((long)r * Wp + s) * C);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = (float)v[j];
          if (f > best[j]) { best[j] = f; arg[j] = r * kw + s; }
        }
      }
    bf16x8 o;
    uchar8 a8;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      o[j] = (__bf16)best[j];
      a8[j] = (unsigned char)arg[j];
    }
    *(bf16x8*)(y + i) = o;
    *(uchar8*)(am + i) = a8;
  }
}

__global__ void maxpool_bwd_k8(const __bf16* __restrict__ dy,
                               const unsigned char* __restrict__ am,
                               __bf16* __restrict__ dx, int N, int H,
                               int W, int C, int P, int Q, int kh,
                               int kw, int sh, int sw, int ph, int pw,
                               long total8) {
  for (long i8 = blockIdx.x * (long)blockDim.x + threadIdx.x;
       i8 < total8; i8 += (long)gridDim.x * blockDim.x) {
    long i = i8 * 8;
    int c = (int)(i % C);
    long pix = i / C;
    int w = (int)(pix % W);
    long tmp = pix / W;
    int h = (int)(tmp % H);
    int n = (int)(tmp / H);
    int hp = h + ph, wp = w + pw;
    float acc[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
    int pLo = (hp - kh + sh) / sh; if (pLo < 0) pLo = 0;
    int pHi = hp / sh; if (pHi > P - 1) pHi = P - 1;
    int qLo = (wp - kw + sw) / sw; if (qLo < 0) qLo = 0;
    int qHi = wp / sw; if (qHi > Q - 1) qHi = Q - 1;
    for (int p = pLo; p <= pHi; ++p)
      for (int q = qLo; q <= qHi; ++q) {
        int r = hp - p * sh, s2 = wp - q * sw;
        if (r < 0 || r >= kh || s2 < 0 || s2 >= kw) continue;
        long oi = (((long)n * P + p) * Q + q) * C + c;
        unsigned char code = (unsigned char)(r * kw + s2);
        bf16x8 g = *(const bf16x8*)(dy + oi);
        uchar8 m8 = *(const uchar8*)(am + oi);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          if (m8[j] == code) acc[j] += (float)g[j];
      }
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = (__bf16)acc[j];
    *(bf16x8*)(dx + i) = o;
  }
}

// thread handles 4 channels of one INPUT pixel; sums dy over the few
// windows that contain it and selected it.
__global__ void maxpool_bwd_k(const __bf16* __restrict__ dy,
                              const unsigned char* __restrict__ am,
                              __bf16* __restrict__ dx, int N, int H,
                              int W, int C, int P, int Q, int kh, int kw,
                              int sh, int sw, int ph, int pw,
                              long total4) {
  for (long i4 = blockIdx.x * (long)blockDim.x + threadIdx.x; i4 < total4;
       i4 += (long)gridDim.x * blockDim.x) {
    long i = i4 * 4;
    int c = (int)(i % C);
    long pix = i / C;
    int w = (int)(pix % W);
    long tmp = pix / W;
    int h = (int)(tmp % H);
    int n = (int)(tmp / H);
    int hp = h + ph, wp = w + pw;      // position in padded coords
    float acc[4] = {0.f, 0.f, 0.f, 0.f};
    int pLo = (hp - kh + sh) / sh; if (pLo < 0) pLo = 0;
    int pHi = hp / sh; if (pHi > P - 1) pHi = P - 1;
    int qLo = (wp - kw + sw) / sw; if (qLo < 0) qLo = 0;
    int qHi = wp / sw; if (qHi > Q - 1) qHi = Q - 1;
    for (int p = pLo; p <= pHi; ++p)
      for (int q = qLo; q <= qHi; ++q) {
        int r = hp - p * sh, s2 = wp - q * sw;
        if (r < 0 || r >= kh || s2 < 0 || s2 >= kw) continue;
        long oi = (((long)n * P + p) * Q + q) * C + c;
        unsigned char code = (unsigned char)(r * kw + s2);
        bf16x4 g = *(const bf16x4*)(dy + oi);
        const unsigned int m4 = *(const unsigned int*)(am + oi);
#pragma unroll
        for (int j = 0; j < 4; ++j)
          if (((m4 >> (8 * j)) & 0xff) == code) acc[j] += (float)g[j];
      }
    bf16x4 o;
#pragma unroll
    for (int j = 0; j < 4; ++j) o[j] = (__bf16)acc[j];
    *(bf16x4*)(dx + i) = o;
  }
}

__global__ void avgpool_fwd_k(const __bf16* __restrict__ xp,
                              __bf16* __restrict__ y, int N, int Hp,
                              int Wp, int C, int P, int Q, int kh, int kw,
                              int sh, int sw, long total4) {
  const float inv = 1.f / (kh * kw);
  for (long i4 = blockIdx.x * blockDim.x + threadIdx.x; i4 < total4;
       i4 += (long)gridDim.x * blockDim.x) {
    long i = i4 * 4;
    int c = (int)(i % C);
    long pix = i / C;
    int q = (int)(pix % Q);
    long tmp = pix / Q;
    int p = (int)(tmp % P);
    int n = (int)(tmp / P);
    const __bf16* base = xp + (((long)n * Hp + p * sh) * Wp + q * sw) * C + c;
    float acc[4] = {0.f, 0.f, 0.f, 0.f};
    for (int r = 0; r < kh; ++r)
      for (int s = 0; s < kw; ++s) {
        bf16x4 v = *(const bf16x4*)(base + ((long)r * Wp + s) * C);
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[j] += (float)v[j];
      }
    bf16x4 o;
#pragma unroll
    for (int j = 0; j < 4; ++j) o[j] = (__bf16)(acc[j] * inv);
    *(bf16x4*)(y + i) = o;
  }
}


// 8-channel avgpool variants (16-B transactions; C % 8 == 0)
__global__ void avgpool_fwd_k8(const __bf16* __restrict__ xp,
                               __bf16* __restrict__ y, int N, int Hp,
                               int Wp, int C, int P, int Q, int kh,
                               int kw, int sh, int sw, long total8) {
  const float inv = 1.f / (kh * kw);
  for (long i8 = blockIdx.x * blockDim.x + threadIdx.x; i8 < total8;
       i8 += (long)gridDim.x * blockDim.x) {
    long i = i8 * 8;
    int c = (int)(i % C);
    long pix = i / C;
    int q = (int)(pix % Q);
    long tmp = pix / Q;
    int p = (int)(tmp % P);
    int n = (int)(tmp / P);
    const __bf16* base = xp
        + (((long)n * Hp + p * sh) * Wp + q * sw) * C + c;
    float acc[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
    for (int r = 0; r < kh; ++r)
      for (int s = 0; s < kw; ++s) {
        bf16x8 v = *(const bf16x8*)(base + ((long)r * Wp + s) * C);
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[j] += (float)v[j];
      }
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = (__bf16)(acc[j] * inv);
    *(bf16x8*)(y + i) = o;
  }
}

__global__ void avgpool_bwd_k8(const __bf16* __restrict__ dy,
                               __bf16* __restrict__ dx, int N, int H,
                               int W, int C, int P, int Q, int kh,
                               int kw, int sh, int sw, int ph, int pw,
                               long total8) {
  const float inv = 1.f / (kh * kw);
  for (long i8 = blockIdx.x * blockDim.x + threadIdx.x; i8 < total8;
       i8 += (long)gridDim.x * blockDim.x) {
    long i = i8 * 8;
    int c = (int)(i % C);
    long pix = i / C;
    int w = (int)(pix % W);
    long tmp = pix / W;
    int h = (int)(tmp % H);
    int n = (int)(tmp / H);
    int hp = h + ph, wp = w + pw;
    float acc[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
    int pLo = (hp - kh + sh) / sh; if (pLo < 0) pLo = 0;
    int pHi = hp / sh; if (pHi > P - 1) pHi = P - 1;
    int qLo = (wp - kw + sw) / sw; if (qLo < 0) qLo = 0;
    int qHi = wp / sw; if (qHi > Q - 1) qHi = Q - 1;
    for (int p = pLo; p <= pHi; ++p)
      for (int q = qLo; q <= qHi; ++q) {
        int r = hp - p * sh, s = wp - q * sw;
        if (r < 0 || r >= kh || s < 0 || s >= kw) continue;
        bf16x8 g = *(const bf16x8*)(dy
            + (((long)n * P + p) * Q + q) * C + c);
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[j] += (float)g[j];
      }
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = (__bf16)(acc[j] * inv);
    *(bf16x8*)(dx + i) = o;
  }
}

__global__ void avgpool_bwd_k(const __bf16* __restrict__ dy,
                              __bf16* __restrict__ dx, int N, int H,
                              int W, int C, int P, int Q, int kh, int kw,
                              int sh, int sw, int ph, int pw, long total) {
  const float inv = 1.f / (kh * kw);
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    long pix = i / C;
    int w = (int)(pix % W);
    long tmp = pix / W;
    int h = (int)(tmp % H);
    int n = (int)(tmp / H);
    int hp = h + ph, wp = w + pw;
    float acc = 0.f;
    int pLo = (hp - kh + sh) / sh; if (pLo < 0) pLo = 0;
    int pHi = hp / sh; if (pHi > P - 1) pHi = P - 1;
    int qLo = (wp - kw + sw) / sw; if (qLo < 0) qLo = 0;
    int qHi = wp / sw; if (qHi > Q - 1) qHi = Q - 1;
    for (int p = pLo; p <= pHi; ++p)
      for (int q = qLo; q <= qHi; ++q) {
        int r = hp - p * sh, s = wp - q * sw;
        if (r < 0 || r >= kh || s < 0 || s >= kw) continue;
        acc += (float)dy[(((long)n * P + p) * Q + q) * C + c];
      }
    dx[i] = (__bf16)(acc * inv);
  }
}

static inline int pool_grid(long work) {
  long g = (work + 255) / 256;
  if (g > 4096) g = 4096;
  if (g < 1) g = 1;
  return (int)g;
}

extern "C" {

hipError_t launch_maxpool_fwd(const void* xp, void* y, unsigned char* am,
                              int N, int Hp, int Wp, int C, int P, int Q,
                              int kh, int kw, int sh, int sw,
                              hipStream_t stream) {
  if (C % 8 == 0) {
    long total8 = (long)N * P * Q * C / 8;
    hipLaunchKernelGGL(maxpool_fwd_k8, dim3(pool_grid(total8)),
                       dim3(256), 0, stream, (const __bf16*)xp,
                       (__bf16*)y, am, N, Hp, Wp, C, P, Q, kh, kw, sh,
                       sw, total8);
    return hipGetLastError();
  }
  long total4 = (long)N * P * Q * C / 4;
  hipLaunchKernelGGL(maxpool_fwd_k, dim3(pool_grid(total4)), dim3(256), 0,
                     stream, (const __bf16*)xp, (__bf16*)y, am, N, Hp, Wp,
                     C, P, Q, kh, kw, sh, sw, total4);
  return hipGetLastError();
}

hipError_t launch_maxpool_bwd(const void* dy, const unsigned char* am,
                              void* dx, int N, int H, int W, int C, int P,
                              int Q, int kh, int kw, int sh, int sw,
                              int ph, int pw, hipStream_t stream) {
  if (C % 8 == 0) {
    long total8 = (long)N * H * W * C / 8;
    hipLaunchKernelGGL(maxpool_bwd_k8, dim3(pool_grid(total8)),
                       dim3(256), 0, stream, (const __bf16*)dy, am,
                       (__bf16*)dx, N, H, W, C, P, Q, kh, kw, sh, sw,
                       ph, pw, total8);
    return hipGetLastError();
  }
  long total4 = (long)N * H * W * C / 4;
  hipLaunchKernelGGL(maxpool_bwd_k, dim3(pool_grid(total4)), dim3(256), 0,
                     stream, (const __bf16*)dy, am, (__bf16*)dx, N, H, W,
                     C, P, Q, kh, kw, sh, sw, ph, pw, total4);
  return hipGetLastError();
}

hipError_t launch_avgpool_fwd(const void* xp, void* y, int N, int Hp,
                              int Wp, int C, int P, int Q, int kh, int kw,
                              int sh, int sw, hipStream_t stream) {
  if (C % 8 == 0) {
    long total8 = (long)N * P * Q * C / 8;
    hipLaunchKernelGGL(avgpool_fwd_k8, dim3(pool_grid(total8)),
                       dim3(256), 0, stream, (const __bf16*)xp,
                       (__bf16*)y, N, Hp, Wp, C, P, Q, kh, kw, sh, sw,
                       total8);
    return hipGetLastError();
  }
  long total4 = (long)N * P * Q * C / 4;
  hipLaunchKernelGGL(avgpool_fwd_k, dim3(pool_grid(total4)), dim3(256), 0,
                     stream, (const __bf16*)xp, (__bf16*)y, N, Hp, Wp, C,
                     P, Q, kh, kw, sh, sw, total4);
  return hipGetLastError();
}

hipError_t launch_avgpool_bwd(const void* dy, void* dx, int N, int H,
                              int W, int C, int P, int Q, int kh, int kw,
                              int sh, int sw, int ph, int pw,
                              hipStream_t stream) {
  if (C % 8 == 0) {
    long total8 = (long)N * H * W * C / 8;
    hipLaunchKernelGGL(avgpool_bwd_k8, dim3(pool_grid(total8)),
                       dim3(256), 0, stream, (const __bf16*)dy,
                       (__bf16*)dx, N, H, W, C, P, Q, kh, kw, sh, sw,
                       ph, pw, total8);
    return hipGetLastError();
  }
  long total = (long)N * H * W * C;
  hipLaunchKernelGGL(avgpool_bwd_k, dim3(pool_grid(total)), dim3(256), 0,
                     stream, (const __bf16*)dy, (__bf16*)dx, N, H, W, C,
                     P, Q, kh, kw, sh, sw, ph, pw, total);
  return hipGetLastError();
}

}  // extern "C"
