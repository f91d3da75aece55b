// Normalisation kernels, NHWC bf16, fp32 statistics (SURVEY.md §2.6
// N6 LRN cross-map, N7 BatchNorm).

#include "common.hpp"

// ---------------------------------------------------------------------------
// LRN (cross-map):  scale = k + alpha/size * sum_{|j-c|<=size/2} x_j^2
//                   y = x * scale^(-beta)
// ---------------------------------------------------------------------------

__global__ void lrn_fwd_k(const __bf16* __restrict__ x,
                          __bf16* __restrict__ y,
                          __bf16* __restrict__ scale, long rows, int C,
                          int size, float alpha, float beta, float k) {
  // one thread per (row, 4-channel group); the +-4 window around the
  // group is covered by three aligned 8-B loads (size <= 9).
  const int half = size / 2;
  const float a = alpha / size;
  const int C4 = C / 4;
  const long total = rows * C4;
  for (long i4 = blockIdx.x * (long)blockDim.x + threadIdx.x; i4 < total;
       i4 += (long)gridDim.x * blockDim.x) {
    int c4 = (int)(i4 % C4);
    long row = i4 / C4;
    const __bf16* xr = x + row * C;
    float v[12];
#pragma unroll
    for (int b = 0; b < 3; ++b) {
      int cb = c4 + b - 1;
      if (cb >= 0 && cb < C4) {
        bf16x4 t = *(const bf16x4*)(xr + cb * 4);
#pragma unroll
        for (int j = 0; j < 4; ++j) v[b * 4 + j] = (float)t[j];
      } else {
#pragma unroll
        for (int j = 0; j < 4; ++j) v[b * 4 + j] = 0.f;
      }
    }
    long o = row * C + c4 * 4;
    bf16x4 out;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int c = c4 * 4 + j;
      int lo = c - half < 0 ? -c4 * 4 : j - half;        // window rel j
      int hi = c + half >= C ? C - c4 * 4 - 1 : j + half;
      float s = 0.f;
      for (int w = lo; w <= hi; ++w) {
        float f = v[4 + w];
        s += f * f;
      }
      float sc = k + a * s;
      float inv = 1.f / sc;                // saved: bwd needs 1/scale
      scale[o + j] = (__bf16)inv;          // bf16 save halves the write
      // inv^beta; beta = 0.75 fast path: sqrt(inv)*sqrt(sqrt(inv))
      float pw = (beta == 0.75f)
                     ? __fsqrt_rn(inv) * __fsqrt_rn(__fsqrt_rn(inv))
                     : __powf(inv, beta);
      out[j] = (__bf16)(v[4 + j] * pw);
    }
    *(bf16x4*)(y + o) = out;
  }
}

// dx_c = dy_c*scale_c^-b - (2*a*b/n) * x_c * sum_{j in win(c)} dy_j*y_j/scale_j
// x is reconstructed from the saved output (x = y / scale^beta), so
// the backward reads dy + y + bf16 scale only — no x pass.
__global__ void lrn_bwd_k(const __bf16* __restrict__ dy,
                          const __bf16* __restrict__ y,
                          const __bf16* __restrict__ scale,
                          __bf16* __restrict__ dx, long rows, int C,
                          int size, float alpha, float beta, float k) {
  const int half = size / 2;
  const float coef = 2.f * alpha * beta / size;
  const int C4 = C / 4;
  const long total = rows * C4;
  for (long i4 = blockIdx.x * (long)blockDim.x + threadIdx.x; i4 < total;
       i4 += (long)gridDim.x * blockDim.x) {
    int c4 = (int)(i4 % C4);
    long row = i4 / C4;
    long b0 = row * C;
    // t[w] = dy*y/scale over the 12-channel neighbourhood
    float t[12], yv[4], dyv[4];
#pragma unroll
    for (int b = 0; b < 3; ++b) {
      int cb = c4 + b - 1;
      if (cb >= 0 && cb < C4) {
        bf16x4 d = *(const bf16x4*)(dy + b0 + cb * 4);
        bf16x4 yy = *(const bf16x4*)(y + b0 + cb * 4);
        bf16x4 sc = *(const bf16x4*)(scale + b0 + cb * 4); // = 1/scale
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          t[b * 4 + j] = (float)d[j] * (float)yy[j] * (float)sc[j];
          if (b == 1) {
            dyv[j] = (float)d[j];
            yv[j] = (float)yy[j];
          }
        }
      } else {
#pragma unroll
        for (int j = 0; j < 4; ++j) t[b * 4 + j] = 0.f;
      }
    }
    bf16x4 sccb = *(const bf16x4*)(scale + b0 + c4 * 4);
    f32x4 scc;
#pragma unroll
    for (int j = 0; j < 4; ++j) scc[j] = (float)sccb[j];
    long o = b0 + c4 * 4;
    bf16x4 out;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int c = c4 * 4 + j;
      int lo = c - half < 0 ? -c4 * 4 : j - half;
      int hi = c + half >= C ? C - c4 * 4 - 1 : j + half;
      float s = 0.f;
      for (int w = lo; w <= hi; ++w) s += t[4 + w];
      float inv = scc[j];                  // 1/scale
      float pw = (beta == 0.75f)
                     ? __fsqrt_rn(inv) * __fsqrt_rn(__fsqrt_rn(inv))
                     : __powf(inv, beta);
      float xv = yv[j] * __frcp_rn(pw);    // x = y / scale^beta
      float d = dyv[j] * pw - coef * xv * s;
      out[j] = (__bf16)d;
    }
    *(bf16x4*)(dx + o) = out;
  }
}

// 16-channels-per-thread variants: the 4-wide kernels above are
// LATENCY-bound (measured 3.5x off the bandwidth roofline at 15.8% of
// the AlexNet step, profiles/alexnet r2): each thread issued 3 tiny
// dependent loads for 4 outputs.  Here a thread owns 16 channels of
// one pixel — six independent 8-B loads (1.5x read amplification from
// the +-half halo), register-resident squares, and an incremental
// sliding-window sum (2 adds per output).  C % 16 == 0 and half <= 4
// (the reference models use size 5); launcher falls back otherwise.

__global__ void lrn_fwd16_k(const __bf16* __restrict__ x,
                            __bf16* __restrict__ y,
                            __bf16* __restrict__ scale, long rows, int C,
                            int size, float alpha, float beta, float k) {
  const int half = size / 2;               // <= 4
  const float a = alpha / size;
  const int C16 = C / 16;
  const int C8 = C / 8;
  const long total = rows * C16;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int c16 = (int)(i % C16);
    long row = i / C16;
    const __bf16* xr = x + row * C + c16 * 16;
    // neighbourhood [c16*16-8, c16*16+24) via four 16-B loads
    float v[32], sq[32];
#pragma unroll
    for (int b8 = 0; b8 < 4; ++b8) {
      int cb = c16 * 2 + b8 - 1;           // absolute 8-channel block
      if (cb >= 0 && cb < C8) {
        bf16x8 t = *(const bf16x8*)(xr + (b8 - 1) * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) v[b8 * 8 + j] = (float)t[j];
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) v[b8 * 8 + j] = 0.f;
      }
    }
#pragma unroll
    for (int j = 0; j < 32; ++j) sq[j] = v[j] * v[j];
    bf16x8 out0, out1, sc0, sc1;
    float s = 0.f;
    for (int w = -half; w <= half; ++w) s += sq[8 + w];
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      float sc = k + a * s;
      float inv = 1.f / sc;                // saved: bwd needs 1/scale
      float pw = (beta == 0.75f)
                     ? __fsqrt_rn(inv) * __fsqrt_rn(__fsqrt_rn(inv))
                     : __powf(inv, beta);
      float o = v[8 + j] * pw;
      if (j < 8) { sc0[j] = (__bf16)inv; out0[j] = (__bf16)o; }
      else { sc1[j - 8] = (__bf16)inv; out1[j - 8] = (__bf16)o; }
      if (j < 15) s += sq[9 + j + half] - sq[8 + j - half];
    }
    long o = row * C + c16 * 16;
    *(bf16x8*)(y + o) = out0;
    *(bf16x8*)(y + o + 8) = out1;
    *(bf16x8*)(scale + o) = sc0;
    *(bf16x8*)(scale + o + 8) = sc1;
  }
}

__global__ void lrn_bwd16_k(const __bf16* __restrict__ dy,
                            const __bf16* __restrict__ y,
                            const __bf16* __restrict__ scale,
                            __bf16* __restrict__ dx, long rows, int C,
                            int size, float alpha, float beta, float k) {
  const int half = size / 2;
  const float coef = 2.f * alpha * beta / size;
  const int C16 = C / 16;
  const int C4 = C / 4;
  const long total = rows * C16;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int c16 = (int)(i % C16);
    long row = i / C16;
    long base = row * C + c16 * 16;
    // neighbourhood [c16*16-8, +24) via four 16-B loads per tensor
    float t[32], yv[16], dyv[16], scv[16];
#pragma unroll
    for (int b8 = 0; b8 < 4; ++b8) {
      int cb = c16 * 2 + b8 - 1;           // absolute 8-channel block
      if (cb >= 0 && cb < C / 8) {
        long ba = base + (b8 - 1) * 8;
        bf16x8 d = *(const bf16x8*)(dy + ba);
        bf16x8 yy = *(const bf16x8*)(y + ba);
        bf16x8 sc = *(const bf16x8*)(scale + ba);     // = 1/scale
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          t[b8 * 8 + j] = (float)d[j] * (float)yy[j] * (float)sc[j];
          int cj = b8 * 8 + j - 8;
          if (cj >= 0 && cj < 16) {
            dyv[cj] = (float)d[j];
            yv[cj] = (float)yy[j];
            scv[cj] = (float)sc[j];
          }
        }
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) t[b8 * 8 + j] = 0.f;
      }
    }
    bf16x8 out0, out1;
    float s = 0.f;
    for (int w = -half; w <= half; ++w) s += t[8 + w];
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      float inv = scv[j];                  // 1/scale
      float pw = (beta == 0.75f)
                     ? __fsqrt_rn(inv) * __fsqrt_rn(__fsqrt_rn(inv))
                     : __powf(inv, beta);
      float xv = yv[j] * __frcp_rn(pw);    // x = y / scale^beta
      float d = dyv[j] * pw - coef * xv * s;
      if (j < 8) out0[j] = (__bf16)d;
      else out1[j - 8] = (__bf16)d;
      if (j < 15) s += t[9 + j + half] - t[8 + j - half];
    }
    *(bf16x8*)(dx + base) = out0;
    *(bf16x8*)(dx + base + 8) = out1;
  }
}

// ---------------------------------------------------------------------------
// BatchNorm (training): per-channel mean/var over N*H*W rows.
// Pass 1: partial sums (fp32 atomics into [C] buffers).
// Pass 2 (host-side tiny kernel): finalize mean/invstd + running stats.
// Pass 3: normalize (+ optional fused ReLU).
// ---------------------------------------------------------------------------

// 32 channel-groups x 8 row-lanes per block, 4-row ILP, LDS-reduced
// (same shape as bias_wgrad_k; the naive per-channel-thread version was
// the ResNet-50 bottleneck).
template <int CLW>     // channel-lane width adapts to narrow layers
__global__ void bn_sums_k(const __bf16* __restrict__ x,
                          float* __restrict__ s1, float* __restrict__ s2,
                          long rows, int C) {
  constexpr int RLN = 256 / CLW;
  const int c8n = C / 8;
  const int cl = threadIdx.x & (CLW - 1);
  const int rl = threadIdx.x / CLW;
  const int c8 = blockIdx.x * CLW + cl;
  const long sl = (long)blockIdx.y * RLN + rl;
  const long nsl = (long)gridDim.y * RLN;
  float a[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  float b[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  if (c8 < c8n) {
    const __bf16* base = x + c8 * 8;
    long r4 = sl * 4;
    for (; r4 + 4 <= rows; r4 += nsl * 4) {
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        bf16x8 v = *(const bf16x8*)(base + (r4 + u) * C);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = (float)v[j];
          a[j] += f;
          b[j] += f * f;
        }
      }
    }
    if (blockIdx.y == 0 && rl == 0) {
      for (long r = rows & ~3L; r < rows; ++r) {
        bf16x8 v = *(const bf16x8*)(base + r * C);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = (float)v[j];
          a[j] += f;
          b[j] += f * f;
        }
      }
    }
  }
  __shared__ float red[2][RLN][CLW][8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    red[0][rl][cl][j] = a[j];
    red[1][rl][cl][j] = b[j];
  }
  __syncthreads();
  int t = threadIdx.x;          // reduce spread over (channel, j) pairs
  if (t < CLW * 8) {
    int cl2 = t / 8, j = t & 7;
    int c82 = blockIdx.x * CLW + cl2;
    if (c82 < c8n) {
      float sa = 0.f, sb = 0.f;
#pragma unroll
      for (int r = 0; r < RLN; ++r) {
        sa += red[0][r][cl2][j];
        sb += red[1][r][cl2][j];
      }
      atomicAdd(&s1[c82 * 8 + j], sa);
      atomicAdd(&s2[c82 * 8 + j], sb);
    }
  }
}

// Reduce the conv epilogue's bucketed statistics ([NB][2][C]) into
// s1/s2 and ZERO the buckets behind itself (steady state needs no
// fill launch, same trick as the split-K workspace).
__global__ void bn_bucket_reduce_k(float* __restrict__ buckets,
                                   float* __restrict__ s1,
                                   float* __restrict__ s2, int C,
                                   int NB) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float a = 0.f, b = 0.f;
  for (int r = 0; r < NB; ++r) {
    long i1 = ((long)r * 2) * C + c;
    long i2 = ((long)r * 2 + 1) * C + c;
    a += buckets[i1];
    b += buckets[i2];
    buckets[i1] = 0.f;
    buckets[i2] = 0.f;
  }
  s1[c] = a;
  s2[c] = b;
}

__global__ void bn_finalize_k(float* __restrict__ s1,
                              float* __restrict__ s2,
                              float* __restrict__ mean,
                              float* __restrict__ invstd,
                              float* __restrict__ rmean,
                              float* __restrict__ rvar, long rows, int C,
                              float momentum, float eps) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float m = s1[c] / rows;
  float v = s2[c] / rows - m * m;
  if (v < 0.f) v = 0.f;
  mean[c] = m;
  invstd[c] = rsqrtf(v + eps);
  // running stats (unbiased var like torch)
  float unb = rows > 1 ? v * rows / (rows - 1) : v;
  rmean[c] = (1.f - momentum) * rmean[c] + momentum * m;
  rvar[c] = (1.f - momentum) * rvar[c] + momentum * unb;
  // leave the accumulators zeroed for the next step: the caller keeps
  // s1/s2 in a per-(device,C) cache, so bn_fwd needs no per-call fill
  // (those were 106 fill launches/step on ResNet-50)
  s1[c] = 0.f;
  s2[c] = 0.f;
}

__global__ void bn_apply_k(const __bf16* __restrict__ x,
                           __bf16* __restrict__ y,
                           const float* __restrict__ mean,
                           const float* __restrict__ invstd,
                           const float* __restrict__ gamma,
                           const float* __restrict__ beta, long rows,
                           int C, int relu,
                           const __bf16* __restrict__ res) {
  // res (optional): fused residual join — y = relu(bn(x) + res), one
  // pass instead of bn_apply + add_relu (the ResNet residual adds
  // were ~8% of the step)
  const long t4 = rows * C / 4;
  for (long i4 = blockIdx.x * (long)blockDim.x + threadIdx.x; i4 < t4;
       i4 += (long)gridDim.x * blockDim.x) {
    long i = i4 * 4;
    int c = (int)(i % C);
    bf16x4 xv = *(const bf16x4*)(x + i);
    bf16x4 rv;
    if (res) rv = *(const bf16x4*)(res + i);
    f32x4 mv = *(const f32x4*)(mean + c);
    f32x4 iv = *(const f32x4*)(invstd + c);
    f32x4 gv = *(const f32x4*)(gamma + c);
    f32x4 bv = *(const f32x4*)(beta + c);
    bf16x4 o;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float v = ((float)xv[j] - mv[j]) * iv[j] * gv[j] + bv[j];
      if (res) v += (float)rv[j];
      if (relu && v < 0.f) v = 0.f;
      o[j] = (__bf16)v;
    }
    *(bf16x4*)(y + i) = o;
  }
}

// 16-B, 2-deep-unrolled normalize (the 8-B kernel above is
// latency-bound on the big early-layer tensors; C % 8 == 0 always
// holds for the channel-padded layouts)
__global__ void bn_apply_k16(const __bf16* __restrict__ x,
                             __bf16* __restrict__ y,
                             const float* __restrict__ mean,
                             const float* __restrict__ invstd,
                             const float* __restrict__ gamma,
                             const float* __restrict__ beta, long rows,
                             int C, int relu,
                             const __bf16* __restrict__ res) {
  const long t8 = rows * C / 8;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i8 = blockIdx.x * (long)blockDim.x + threadIdx.x; i8 < t8;
       i8 += 2 * stride) {
    long j8 = i8 + stride;
    long i = i8 * 8;
    long j = j8 < t8 ? j8 * 8 : i;
    bf16x8 xv0 = *(const bf16x8*)(x + i);
    bf16x8 xv1 = *(const bf16x8*)(x + j);
    bf16x8 rv0, rv1;
    if (res) {
      rv0 = *(const bf16x8*)(res + i);
      rv1 = *(const bf16x8*)(res + j);
    }
    int c0 = (int)(i % C), c1 = (int)(j % C);
    bf16x8 o0, o1;
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      int c = h ? c1 : c0;
      const bf16x8& xv = h ? xv1 : xv0;
      const bf16x8& rvv = h ? rv1 : rv0;
      bf16x8& o = h ? o1 : o0;
#pragma unroll
      for (int q = 0; q < 2; ++q) {
        f32x4 mv = *(const f32x4*)(mean + c + q * 4);
        f32x4 iv = *(const f32x4*)(invstd + c + q * 4);
        f32x4 gv = *(const f32x4*)(gamma + c + q * 4);
        f32x4 bv = *(const f32x4*)(beta + c + q * 4);
#pragma unroll
        for (int k2 = 0; k2 < 4; ++k2) {
          float v = ((float)xv[q * 4 + k2] - mv[k2]) * iv[k2] * gv[k2]
                    + bv[k2];
          if (res) v += (float)rvv[q * 4 + k2];
          if (relu && v < 0.f) v = 0.f;
          o[q * 4 + k2] = (__bf16)v;
        }
      }
    }
    *(bf16x8*)(y + i) = o0;
    if (j8 < t8) *(bf16x8*)(y + j8 * 8) = o1;
  }
}

// Backward prep: per-channel 1/gamma plus a single "all channels safe"
// flag.  When every |gamma| is comfortably away from 0 the backward
// kernels reconstruct xhat from the SAVED bf16 OUTPUT
// (xhat = (y - beta) / gamma; relu-masked lanes contribute 0 anyway)
// and skip the x read entirely — one fewer full-tensor pass in each of
// bn_bwd_sums / bn_bwd_apply, which are memory-bound (ResNet-50: BN is
// ~31% of the step).  A tiny |gamma| would amplify y's bf16 rounding,
// so that case keeps the exact x path (uniform per-layer branch).
// allok is a persistent per-device "bad token" cell (no per-call
// reset fill — that was one FillFunctor launch per BN backward, 53 a
// step on ResNet-50): the y-path is usable iff *allok != token.
__global__ void bn_gamma_prep_k(const float* __restrict__ gamma,
                                float* __restrict__ rgamma,
                                int* __restrict__ allok,
                                float* __restrict__ sdy,
                                float* __restrict__ sdyx, int C,
                                int token, int forceX) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  sdy[c] = 0.f;                 // accumulators for the sums pass
  sdyx[c] = 0.f;
  float g = gamma[c];
  bool ok = fabsf(g) > 1e-2f;
  rgamma[c] = ok ? 1.f / g : 0.f;
  if (!ok || forceX) atomicExch(allok, token);
}

// backward pass 1: per-channel  sum(dy)  and  sum(dy * xhat)
// (dy masked by fused relu via y>0 when relu=1).  Templated on the
// xhat source and launched BOTH ways; the instantiation that does not
// match the device-side allok flag exits immediately.  A single kernel
// carrying both loop bodies spilled 68 B/lane of scratch and ran 31%
// slower than the x path it was meant to beat.
template <int USEY, int RELU, int CLW>
__global__ void bn_bwd_sums_k(const __bf16* __restrict__ dy,
                              const __bf16* __restrict__ x,
                              const __bf16* __restrict__ y,
                              const float* __restrict__ mean,
                              const float* __restrict__ invstd,
                              const float* __restrict__ beta,
                              const float* __restrict__ rgamma,
                              const int* __restrict__ allok,
                              float* __restrict__ sdy,
                              float* __restrict__ sdyx, long rows,
                              int C, int token) {
  if ((*allok != token) != (USEY != 0)) return;  // other path owns this
  constexpr int RLN = 256 / CLW;
  const int c8n = C / 8;
  const int cl = threadIdx.x & (CLW - 1);
  const int rl = threadIdx.x / CLW;
  const int c8 = blockIdx.x * CLW + cl;
  const long sl = (long)blockIdx.y * RLN + rl;
  const long nsl = (long)gridDim.y * RLN;
  float a[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  float b[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  float m[8], is[8];
  if (c8 < c8n) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      // USEY: m := beta, is := 1/gamma  (xhat = (y - beta) / gamma)
      m[j] = USEY ? beta[c8 * 8 + j] : mean[c8 * 8 + j];
      is[j] = USEY ? rgamma[c8 * 8 + j] : invstd[c8 * 8 + j];
    }
    const long co = c8 * 8;
    auto accrow = [&](long r) {
      bf16x8 g8 = *(const bf16x8*)(dy + r * C + co);
      bf16x8 s8;            // xhat source: saved output or input
      if (USEY) s8 = *(const bf16x8*)(y + r * C + co);
      else      s8 = *(const bf16x8*)(x + r * C + co);
      bf16x8 y8;
      if (!USEY && RELU) y8 = *(const bf16x8*)(y + r * C + co);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float g = (float)g8[j];
        if (RELU) {
          float yv = USEY ? (float)s8[j] : (float)y8[j];
          g *= (float)(yv > 0.f);       // VALU mask, no select chain
        }
        float xh = ((float)s8[j] - m[j]) * is[j];
        a[j] += g;
        b[j] += g * xh;
      }
    };
    long r4 = sl * 4;
    for (; r4 + 4 <= rows; r4 += nsl * 4) {
#pragma unroll
      for (int u = 0; u < 4; ++u) accrow(r4 + u);
    }
    if (blockIdx.y == 0 && rl == 0)
      for (long r = rows & ~3L; r < rows; ++r) accrow(r);
  }
  __shared__ float red[2][RLN][CLW][8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    red[0][rl][cl][j] = a[j];
    red[1][rl][cl][j] = b[j];
  }
  __syncthreads();
  int t = threadIdx.x;          // reduce spread over (channel, j) pairs
  if (t < CLW * 8) {
    int cl2 = t / 8, j = t & 7;
    int c82 = blockIdx.x * CLW + cl2;
    if (c82 < c8n) {
      float sa = 0.f, sb = 0.f;
#pragma unroll
      for (int r = 0; r < RLN; ++r) {
        sa += red[0][r][cl2][j];
        sb += red[1][r][cl2][j];
      }
      atomicAdd(&sdy[c82 * 8 + j], sa);
      atomicAdd(&sdyx[c82 * 8 + j], sb);
    }
  }
}

__global__ void bn_bwd_apply_k(const __bf16* __restrict__ dy,
                               const __bf16* __restrict__ x,
                               const __bf16* __restrict__ y,
                               const float* __restrict__ mean,
                               const float* __restrict__ invstd,
                               const float* __restrict__ gamma,
                               const float* __restrict__ beta,
                               const float* __restrict__ rgamma,
                               const int* __restrict__ allok,
                               const float* __restrict__ sdy,
                               const float* __restrict__ sdyx,
                               __bf16* __restrict__ dx, long rows, int C,
                               int relu, int token,
                               __bf16* __restrict__ dres) {
  // dres (optional, relu path only): the relu-masked dy — the
  // gradient of a residual fused into this BN's apply
  const float invM = 1.f / rows;
  const long t4 = rows * C / 4;
  // xhat from y — skip the x read.  Only valid without fused relu:
  // the mean-subtraction terms need the true xhat at MASKED positions
  // too, and y=0 there cannot reconstruct it.  (bn_bwd_sums_k has no
  // such restriction: its xhat is always multiplied by the masked g.)
  const bool useY = (*allok != token) && !relu;
  if (useY) {           // non-relu only: dy + y loads, no x traffic
    for (long i4 = blockIdx.x * (long)blockDim.x + threadIdx.x; i4 < t4;
         i4 += (long)gridDim.x * blockDim.x) {
      long i = i4 * 4;
      int c = (int)(i % C);
      bf16x4 gv4 = *(const bf16x4*)(dy + i);
      bf16x4 yv4 = *(const bf16x4*)(y + i);
      f32x4 bv = *(const f32x4*)(beta + c);
      f32x4 rg = *(const f32x4*)(rgamma + c);
      f32x4 ist = *(const f32x4*)(invstd + c);
      f32x4 ga = *(const f32x4*)(gamma + c);
      f32x4 s1 = *(const f32x4*)(sdy + c);
      f32x4 s2 = *(const f32x4*)(sdyx + c);
      bf16x4 o;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float g = (float)gv4[j];
        float xh = ((float)yv4[j] - bv[j]) * rg[j];
        o[j] = (__bf16)(ist[j] * ga[j]
                        * (g - s1[j] * invM - xh * s2[j] * invM));
      }
      *(bf16x4*)(dx + i) = o;
    }
    return;
  }
  for (long i4 = blockIdx.x * (long)blockDim.x + threadIdx.x; i4 < t4;
       i4 += (long)gridDim.x * blockDim.x) {
    long i = i4 * 4;
    int c = (int)(i % C);
    bf16x4 gv4 = *(const bf16x4*)(dy + i);
    bf16x4 xv4 = *(const bf16x4*)(x + i);
    bf16x4 yv4;
    if (relu) yv4 = *(const bf16x4*)(y + i);
    f32x4 mv = *(const f32x4*)(mean + c);
    f32x4 iv = *(const f32x4*)(invstd + c);
    f32x4 ga = *(const f32x4*)(gamma + c);
    f32x4 s1 = *(const f32x4*)(sdy + c);
    f32x4 s2 = *(const f32x4*)(sdyx + c);
    bf16x4 o, dr;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float g = (float)gv4[j];
      if (relu && (float)yv4[j] <= 0.f) g = 0.f;
      if (dres) dr[j] = (__bf16)g;
      float xh = ((float)xv4[j] - mv[j]) * iv[j];
      o[j] = (__bf16)(iv[j] * ga[j]
                      * (g - s1[j] * invM - xh * s2[j] * invM));
    }
    *(bf16x4*)(dx + i) = o;
    if (dres) *(bf16x4*)(dres + i) = dr;
  }
}

// 16-B variant (C % 8 == 0): same math, half the memory transactions
__global__ void bn_bwd_apply_k16(const __bf16* __restrict__ dy,
                                 const __bf16* __restrict__ x,
                                 const __bf16* __restrict__ y,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ invstd,
                                 const float* __restrict__ gamma,
                                 const float* __restrict__ beta,
                                 const float* __restrict__ rgamma,
                                 const int* __restrict__ allok,
                                 const float* __restrict__ sdy,
                                 const float* __restrict__ sdyx,
                                 __bf16* __restrict__ dx, long rows,
                                 int C, int relu, int token,
                                 __bf16* __restrict__ dres) {
  const float invM = 1.f / rows;
  const long t8 = rows * C / 8;
  const bool useY = (*allok != token) && !relu;
  if (useY) {
    for (long i8 = blockIdx.x * (long)blockDim.x + threadIdx.x; i8 < t8;
         i8 += (long)gridDim.x * blockDim.x) {
      long i = i8 * 8;
      int c = (int)(i % C);
      bf16x8 gv8 = *(const bf16x8*)(dy + i);
      bf16x8 yv8 = *(const bf16x8*)(y + i);
      bf16x8 o;
#pragma unroll
      for (int q = 0; q < 2; ++q) {
        f32x4 bv = *(const f32x4*)(beta + c + q * 4);
        f32x4 rg = *(const f32x4*)(rgamma + c + q * 4);
        f32x4 ist = *(const f32x4*)(invstd + c + q * 4);
        f32x4 ga = *(const f32x4*)(gamma + c + q * 4);
        f32x4 s1 = *(const f32x4*)(sdy + c + q * 4);
        f32x4 s2 = *(const f32x4*)(sdyx + c + q * 4);
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          float g = (float)gv8[q * 4 + j];
          float xh = ((float)yv8[q * 4 + j] - bv[j]) * rg[j];
          o[q * 4 + j] = (__bf16)(ist[j] * ga[j]
                                  * (g - s1[j] * invM
                                     - xh * s2[j] * invM));
        }
      }
      *(bf16x8*)(dx + i) = o;
    }
    return;
  }
  for (long i8 = blockIdx.x * (long)blockDim.x + threadIdx.x; i8 < t8;
       i8 += (long)gridDim.x * blockDim.x) {
    long i = i8 * 8;
    int c = (int)(i % C);
    bf16x8 gv8 = *(const bf16x8*)(dy + i);
    bf16x8 xv8 = *(const bf16x8*)(x + i);
    bf16x8 yv8;
    if (relu) yv8 = *(const bf16x8*)(y + i);
    bf16x8 o, dr;
#pragma unroll
    for (int q = 0; q < 2; ++q) {
      f32x4 mv = *(const f32x4*)(mean + c + q * 4);
      f32x4 iv = *(const f32x4*)(invstd + c + q * 4);
      f32x4 ga = *(const f32x4*)(gamma + c + q * 4);
      f32x4 s1 = *(const f32x4*)(sdy + c + q * 4);
      f32x4 s2 = *(const f32x4*)(sdyx + c + q * 4);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float g = (float)gv8[q * 4 + j];
        if (relu && (float)yv8[q * 4 + j] <= 0.f) g = 0.f;
        if (dres) dr[q * 4 + j] = (__bf16)g;
        float xh = ((float)xv8[q * 4 + j] - mv[j]) * iv[j];
        o[q * 4 + j] = (__bf16)(iv[j] * ga[j]
                                * (g - s1[j] * invM
                                   - xh * s2[j] * invM));
      }
    }
    *(bf16x8*)(dx + i) = o;
    if (dres) *(bf16x8*)(dres + i) = dr;
  }
}

static inline int ngrid(long work) {
  long g = (work + 255) / 256;
  if (g > 4096) g = 4096;
  if (g < 1) g = 1;
  return (int)g;
}

__global__ void bn_eval_stats_k(const float* __restrict__ rmean,
                                const float* __restrict__ rvar,
                                float* __restrict__ mean,
                                float* __restrict__ invstd, int C,
                                float eps) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  mean[c] = rmean[c];
  invstd[c] = rsqrtf(rvar[c] + eps);
}

extern "C" {

hipError_t launch_lrn_fwd(const void* x, void* y, void* scale, long rows,
                          int C, int size, float alpha, float beta,
                          float k, hipStream_t stream) {
  if (C % 16 == 0 && size / 2 <= 4) {
    hipLaunchKernelGGL(lrn_fwd16_k, dim3(ngrid(rows * C / 16)),
                       dim3(256), 0, stream, (const __bf16*)x,
                       (__bf16*)y, (__bf16*)scale, rows, C, size, alpha,
                       beta, k);
    return hipGetLastError();
  }
  hipLaunchKernelGGL(lrn_fwd_k, dim3(ngrid(rows * C / 4)), dim3(256), 0,
                     stream, (const __bf16*)x, (__bf16*)y,
                     (__bf16*)scale, rows, C, size, alpha, beta, k);
  return hipGetLastError();
}

hipError_t launch_lrn_bwd(const void* dy, const void* y,
                          const void* scale, void* dx, long rows, int C,
                          int size, float alpha, float beta, float k,
                          hipStream_t stream) {
  if (C % 16 == 0 && size / 2 <= 4) {
    hipLaunchKernelGGL(lrn_bwd16_k, dim3(ngrid(rows * C / 16)),
                       dim3(256), 0, stream, (const __bf16*)dy,
                       (const __bf16*)y, (const __bf16*)scale,
                       (__bf16*)dx, rows, C, size, alpha, beta, k);
    return hipGetLastError();
  }
  hipLaunchKernelGGL(lrn_bwd_k, dim3(ngrid(rows * C / 4)), dim3(256), 0,
                     stream, (const __bf16*)dy, (const __bf16*)y,
                     (const __bf16*)scale, (__bf16*)dx, rows, C, size,
                     alpha, beta, k);
  return hipGetLastError();
}

hipError_t launch_bn_fwd_train(const void* x, void* y, float* s1,
                               float* s2, float* mean, float* invstd,
                               float* rmean, float* rvar,
                               const float* gamma, const float* beta,
                               float* prestats, int nbuckets,
                               long rows, int C, float momentum,
                               float eps, int relu, const void* res,
                               hipStream_t stream) {
  if (prestats) {
    hipLaunchKernelGGL(bn_bucket_reduce_k, dim3(cdiv(C, 256)), dim3(256),
                       0, stream, prestats, s1, s2, C, nbuckets);
  } else {
    int c8n = C / 8;
    int clw = c8n >= 32 ? 32 : (c8n >= 16 ? 16 : 8);
    int gx = cdiv(c8n, clw);
    int rln = 256 / clw;
    long gy = (rows + rln * 4 - 1) / (rln * 4);
    long cap = 1024 / gx; if (cap > 128) cap = 128;
    if (gy > cap) gy = cap;
    if (gy < 1) gy = 1;
    dim3 g(gx, (unsigned)gy);
    if (clw == 32)
      hipLaunchKernelGGL(bn_sums_k<32>, g, dim3(256), 0, stream,
                         (const __bf16*)x, s1, s2, rows, C);
    else if (clw == 16)
      hipLaunchKernelGGL(bn_sums_k<16>, g, dim3(256), 0, stream,
                         (const __bf16*)x, s1, s2, rows, C);
    else
      hipLaunchKernelGGL(bn_sums_k<8>, g, dim3(256), 0, stream,
                         (const __bf16*)x, s1, s2, rows, C);
  }
  hipLaunchKernelGGL(bn_finalize_k, dim3(cdiv(C, 256)), dim3(256), 0,
                     stream, s1, s2, mean, invstd, rmean, rvar, rows, C,
                     momentum, eps);
  if (C % 8 == 0) {
    hipLaunchKernelGGL(bn_apply_k16, dim3(ngrid(rows * C / 16)),
                       dim3(256), 0, stream, (const __bf16*)x,
                       (__bf16*)y, mean, invstd, gamma, beta, rows, C,
                       relu, (const __bf16*)res);
    return hipGetLastError();
  }
  hipLaunchKernelGGL(bn_apply_k, dim3(ngrid(rows * C / 4)), dim3(256), 0,
                     stream, (const __bf16*)x, (__bf16*)y, mean, invstd,
                     gamma, beta, rows, C, relu, (const __bf16*)res);
  return hipGetLastError();
}

hipError_t launch_bn_fwd_eval(const void* x, void* y, float* mean,
                               float* invstd, const float* rmean,
                               const float* rvar, const float* gamma,
                               const float* beta, long rows, int C,
                               float eps, int relu, const void* res,
                               hipStream_t stream) {
  hipLaunchKernelGGL(bn_eval_stats_k, dim3(cdiv(C, 256)), dim3(256), 0,
                     stream, rmean, rvar, mean, invstd, C, eps);
  hipLaunchKernelGGL(bn_apply_k, dim3(ngrid(rows * C)), dim3(256), 0,
                     stream, (const __bf16*)x, (__bf16*)y, mean, invstd,
                     gamma, beta, rows, C, relu,
                     (const __bf16*)res);
  return hipGetLastError();
}

hipError_t launch_bn_bwd(const void* dy, const void* x, const void* y,
                         const float* mean, const float* invstd,
                         const float* gamma, const float* beta,
                         float* rgamma, int* allok, float* sdy,
                         float* sdyx, void* dx, long rows, int C,
                         int relu, int token, int forceX, void* dres,
                         hipStream_t stream) {
  hipLaunchKernelGGL(bn_gamma_prep_k, dim3(cdiv(C, 256)), dim3(256), 0,
                     stream, gamma, rgamma, allok, sdy, sdyx, C, token,
                     forceX);
  {
    int c8n = C / 8;
    int clw = c8n >= 32 ? 32 : (c8n >= 16 ? 16 : 8);
    int gx = cdiv(c8n, clw);
    int rln = 256 / clw;
    long gy = (rows + rln * 4 - 1) / (rln * 4);
    // measured: 512 blocks only pay off for the relu y-path layers
    // (smaller C, long rows); elsewhere the fixed LDS-reduce + atomic
    // cost of extra blocks outweighs the parallelism.
    long cap = 1024 / gx;
    long lim = relu ? 512 : 128;
    if (cap > lim) cap = lim;
    if (gy > cap) gy = cap;
    if (gy < 1) gy = 1;
    dim3 g(gx, (unsigned)gy);
    auto launch_pair = [&](auto k1, auto k0) {
      hipLaunchKernelGGL(k1, g, dim3(256), 0, stream, (const __bf16*)dy,
                         (const __bf16*)x, (const __bf16*)y, mean,
                         invstd, beta, rgamma, allok, sdy, sdyx, rows,
                         C, token);
      hipLaunchKernelGGL(k0, g, dim3(256), 0, stream, (const __bf16*)dy,
                         (const __bf16*)x, (const __bf16*)y, mean,
                         invstd, beta, rgamma, allok, sdy, sdyx, rows,
                         C, token);
    };
    auto pick = [&](auto tag) {
      constexpr int W = decltype(tag)::value;
      if (relu) launch_pair(bn_bwd_sums_k<1, 1, W>,
                            bn_bwd_sums_k<0, 1, W>);
      else      launch_pair(bn_bwd_sums_k<1, 0, W>,
                            bn_bwd_sums_k<0, 0, W>);
    };
    if (clw == 32)      pick(std::integral_constant<int, 32>{});
    else if (clw == 16) pick(std::integral_constant<int, 16>{});
    else                pick(std::integral_constant<int, 8>{});
  }
  if (C % 8 == 0) {
    hipLaunchKernelGGL(bn_bwd_apply_k16, dim3(ngrid(rows * C / 8)),
                       dim3(256), 0, stream, (const __bf16*)dy,
                       (const __bf16*)x, (const __bf16*)y, mean, invstd,
                       gamma, beta, rgamma, allok, sdy, sdyx,
                       (__bf16*)dx, rows, C, relu, token,
                       (__bf16*)dres);
  } else {
    hipLaunchKernelGGL(bn_bwd_apply_k, dim3(ngrid(rows * C / 4)),
                       dim3(256), 0, stream, (const __bf16*)dy,
                       (const __bf16*)x, (const __bf16*)y, mean, invstd,
                       gamma, beta, rgamma, allok, sdy, sdyx,
                       (__bf16*)dx, rows, C, relu, token,
                       (__bf16*)dres);
  }
  return hipGetLastError();
}

}  // extern "C"
