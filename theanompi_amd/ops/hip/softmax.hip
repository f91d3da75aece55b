// Fused softmax + cross-entropy and the fused
// SGD+momentum update (N12), plus the MFMA fragment-layout probe.

#include "common.hpp"

// ---------------------------------------------------------------------------
// softmax-xent fwd: one 256-thread block per row.  logits bf16 [B][Cp],
// labels int64, columns >= nCls masked to -inf.  Outputs: probs bf16,
// per-row loss f32.
// ---------------------------------------------------------------------------

__global__ void softmax_xent_fwd_k(const __bf16* __restrict__ logits,
                                   const long* __restrict__ labels,
                                   __bf16* __restrict__ probs,
                                   float* __restrict__ loss, int Cp,
                                   int nCls,
                                   float* __restrict__ stats) {
  // stats (optional, 2 fp32): fused batch reductions — stats[0] +=
  // row loss, stats[1] += (argmax != label).  Replaces the torch
  // loss.mean()/argmax/ne/mean glue kernels in the train step
  // (hot path must be hand-written).
  const int b = blockIdx.x;
  const int tid = threadIdx.x;
  const __bf16* row = logits + (long)b * Cp;
  __shared__ float red[8];
  __shared__ int redi[8];

  // 1) row max (+ argmax for the fused top-1 error)
  float mx = -3.0e38f;
  int am = 0;
  for (int c = tid; c < nCls; c += blockDim.x) {
    float v = (float)row[c];
    if (v > mx) { mx = v; am = c; }
  }
  for (int off = 32; off > 0; off >>= 1) {
    float o = __shfl_down(mx, off, 64);
    int oi = __shfl_down(am, off, 64);
    if (o > mx || (o == mx && oi < am)) { mx = o; am = oi; }
  }
  if ((tid & 63) == 0) { red[tid >> 6] = mx; redi[tid >> 6] = am; }
  __syncthreads();
  if (tid < 8) {
    const bool live = tid < (blockDim.x >> 6);
    float v = live ? red[tid] : -3.0e38f;
    int vi = live ? redi[tid] : 0;
    for (int off = 4; off > 0; off >>= 1) {
      float o = __shfl_down(v, off, 64);
      int oi = __shfl_down(vi, off, 64);
      if (o > v || (o == v && oi < vi)) { v = o; vi = oi; }
    }
    if (tid == 0) { red[0] = v; redi[0] = vi; }
  }
  __syncthreads();
  mx = red[0];
  const int amx = redi[0];
  __syncthreads();

  // 2) exp-sum
  float sum = 0.f;
  for (int c = tid; c < nCls; c += blockDim.x)
    sum += __expf((float)row[c] - mx);
  for (int off = 32; off > 0; off >>= 1) sum += __shfl_down(sum, off, 64);
  if ((tid & 63) == 0) red[tid >> 6] = sum;
  __syncthreads();
  if (tid < 8) {
    float v = (tid < (blockDim.x >> 6)) ? red[tid] : 0.f;
    for (int off = 4; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
    if (tid == 0) red[0] = v;
  }
  __syncthreads();
  sum = red[0];
  const float inv = 1.f / sum;

  // 3) probs + loss
  __bf16* prow = probs + (long)b * Cp;
  for (int c = tid; c < Cp; c += blockDim.x)
    prow[c] = c < nCls
                  ? (__bf16)(__expf((float)row[c] - mx) * inv)
                  : (__bf16)0.f;
  if (tid == 0) {
    int lab = (int)labels[b];
    float l = __logf(sum) - ((float)row[lab] - mx);
    loss[b] = l;
    if (stats) {
      atomicAdd(&stats[0], l);
      if (amx != lab) atomicAdd(&stats[1], 1.f);
    }
  }
}

__global__ void softmax_xent_bwd_k(const __bf16* __restrict__ probs,
                                   const long* __restrict__ labels,
                                   __bf16* __restrict__ dlogits, int Cp,
                                   int nCls, float scale) {
  const int b = blockIdx.x;
  const int lab = (int)labels[b];
  const __bf16* prow = probs + (long)b * Cp;
  __bf16* drow = dlogits + (long)b * Cp;
  for (int c = threadIdx.x; c < Cp; c += blockDim.x) {
    float v = c < nCls ? (float)prow[c] : 0.f;
    if (c == lab) v -= 1.f;
    drow[c] = (__bf16)(v * scale);
  }
}

// ---------------------------------------------------------------------------
// fused SGD + momentum (flat fp32 master + fp32 grad + fp32 vel; optional
// bf16 mirror write).  phases: 0 = full, 1 = cdd phase1, 2 = cdd phase2.
// ---------------------------------------------------------------------------

__global__ void sgd_step_k(float* __restrict__ w,
                           const float* __restrict__ g,
                           float* __restrict__ v,
                           __bf16* __restrict__ out, long n4, float lr,
                           float mu, float wd, int phase, long n) {
  for (long i4 = blockIdx.x * blockDim.x + threadIdx.x; i4 < n4;
       i4 += (long)gridDim.x * blockDim.x) {
    long i = i4 * 4;
    f32x4 wv = *(const f32x4*)(w + i);
    f32x4 gv = *(const f32x4*)(g + i);
    f32x4 vv = *(const f32x4*)(v + i);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      if (phase == 2) {
        wv[j] += gv[j];                 // gv = exchanged averaged vels
      } else {
        float gr = gv[j] + wd * wv[j];
        vv[j] = mu * vv[j] - lr * gr;
        if (phase == 0) wv[j] += vv[j];
      }
    }
    if (phase != 2) *(f32x4*)(v + i) = vv;
    if (phase != 1) {
      *(f32x4*)(w + i) = wv;
      if (out) {
        bf16x4 o;
#pragma unroll
        for (int j = 0; j < 4; ++j) o[j] = (__bf16)wv[j];
        *(bf16x4*)(out + i) = o;
      }
    }
  }
}

// tail (n % 4) handled scalar
__global__ void sgd_step_tail_k(float* w, const float* g, float* v,
                                __bf16* out, long start, long n, float lr,
                                float mu, float wd, int phase) {
  long i = start + threadIdx.x;
  if (i >= n) return;
  float wv = w[i], gv = g[i], vv = v[i];
  if (phase == 2) {
    wv += gv;
  } else {
    vv = mu * vv - lr * (gv + wd * wv);
    if (phase == 0) wv += vv;
  }
  if (phase != 2) v[i] = vv;
  if (phase != 1) {
    w[i] = wv;
    if (out) out[i] = (__bf16)wv;
  }
}

// ---------------------------------------------------------------------------
// MFMA fragment-layout probe: one wave computes D = A(16x32) @ B(32x16)
// with the given LAYOUT; host compares against a CPU reference to pick
// the layout used by every GEMM kernel (see common.hpp).
// ---------------------------------------------------------------------------

template <int LAYOUT>
__global__ void mfma_probe_k(const __bf16* __restrict__ A,  // [16][32]
                             const __bf16* __restrict__ B,  // [16][32] (n-major!)
                             float* __restrict__ D) {       // [16][16]
  int lane = threadIdx.x;
  if (lane >= 64) return;
  bf16x8 a = frag_load<LAYOUT>(A + (lane & 15) * 32, lane);
  bf16x8 b = frag_load<LAYOUT>(B + (lane & 15) * 32, lane);
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r)
    D[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = acc[r];
}

// ds_read_b64_tr_b16 semantics probe: LDS filled with buf[i] = i (two
// halves so bf16 stays exact); each lane reads at a host-given element
// offset and dumps its 4 returned values.
typedef __attribute__((address_space(3))) bf16x4 lds_bf16x4;

__global__ void tr_probe_k(const int* __restrict__ offs,
                           float* __restrict__ out, int fill_mode) {
  __shared__ __align__(16) __bf16 buf[512];
  int t = threadIdx.x;
  for (int i = t; i < 512; i += 64)
    buf[i] = (__bf16)(float)(fill_mode == 0 ? (i & 0xff) : (i >> 8));
  __syncthreads();
  if (t < 64) {
    bf16x4 v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
        (lds_bf16x4*)&buf[offs[t]]);
#pragma unroll
    for (int j = 0; j < 4; ++j) out[t * 4 + j] = (float)v[j];
  }
}

extern "C" {

hipError_t launch_tr_probe(const int* offs, float* out, int fill_mode,
                           hipStream_t stream) {
  hipLaunchKernelGGL(tr_probe_k, dim3(1), dim3(64), 0, stream, offs, out,
                     fill_mode);
  return hipGetLastError();
}

hipError_t launch_softmax_xent_fwd(const void* logits, const long* labels,
                                   void* probs, float* loss, int B,
                                   int Cp, int nCls, float* stats,
                                   hipStream_t stream) {
  hipLaunchKernelGGL(softmax_xent_fwd_k, dim3(B), dim3(256), 0, stream,
                     (const __bf16*)logits, labels, (__bf16*)probs, loss,
                     Cp, nCls, stats);
  return hipGetLastError();
}

hipError_t launch_softmax_xent_bwd(const void* probs, const long* labels,
                                   void* dlogits, int B, int Cp, int nCls,
                                   float scale, hipStream_t stream) {
  hipLaunchKernelGGL(softmax_xent_bwd_k, dim3(B), dim3(256), 0, stream,
                     (const __bf16*)probs, labels, (__bf16*)dlogits, Cp,
                     nCls, scale);
  return hipGetLastError();
}

hipError_t launch_sgd_step(float* w, const float* g, float* v, void* out,
                           long n, float lr, float mu, float wd,
                           int phase, hipStream_t stream) {
  long n4 = n / 4;
  if (n4 > 0) {
    long grid = (n4 + 255) / 256;
    if (grid > 2048) grid = 2048;
    hipLaunchKernelGGL(sgd_step_k, dim3((int)grid), dim3(256), 0, stream,
                       w, g, v, (__bf16*)out, n4, lr, mu, wd, phase, n);
  }
  if (n % 4) {
    hipLaunchKernelGGL(sgd_step_tail_k, dim3(1), dim3(4), 0, stream, w, g,
                       v, (__bf16*)out, n4 * 4, n, lr, mu, wd, phase);
  }
  return hipGetLastError();
}

hipError_t launch_mfma_probe(const void* A, const void* B, float* D,
                             int layout, hipStream_t stream) {
  if (layout == FRAG_CONTIG8)
    hipLaunchKernelGGL(mfma_probe_k<FRAG_CONTIG8>, dim3(1), dim3(64), 0,
                       stream, (const __bf16*)A, (const __bf16*)B, D);
  else
    hipLaunchKernelGGL(mfma_probe_k<FRAG_HALVES>, dim3(1), dim3(64), 0,
                       stream, (const __bf16*)A, (const __bf16*)B, D);
  return hipGetLastError();
}

}  // extern "C"

// ---------------------------------------------------------------------------
// batched variants: one launch covers every parameter tensor (the
// per-tensor launches were 11% of the AlexNet step — see
// profiles/alexnet_r01_baseline.md).
// ---------------------------------------------------------------------------

struct SgdSeg {
  float* w;
  const float* g;
  float* v;
  __bf16* out;
  long n4;        // vector chunks
  long ntail;     // total n (scalar tail handled by last thread)
  float lr, wd;
};

__global__ void sgd_multi_k(const SgdSeg* __restrict__ segs, float mu,
                            int phase) {
  const SgdSeg s = segs[blockIdx.y];
  const long stride = (long)gridDim.x * blockDim.x;
  // 2-deep unroll: six independent 16-B loads in flight per thread
  // (the 1-deep version measured ~65% of the bandwidth roofline)
  for (long i4 = blockIdx.x * (long)blockDim.x + threadIdx.x; i4 < s.n4;
       i4 += 2 * stride) {
    const long k4 = i4 + stride;
    const bool two = k4 < s.n4;
    long i = i4 * 4;
    long k = two ? k4 * 4 : i;
    f32x4 wv = *(const f32x4*)(s.w + i);
    f32x4 gv = *(const f32x4*)(s.g + i);
    f32x4 vv = *(const f32x4*)(s.v + i);
    f32x4 wv2 = *(const f32x4*)(s.w + k);
    f32x4 gv2 = *(const f32x4*)(s.g + k);
    f32x4 vv2 = *(const f32x4*)(s.v + k);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      if (phase == 2) {
        wv[j] += gv[j];
        wv2[j] += gv2[j];
      } else {
        float gr = gv[j] + s.wd * wv[j];
        vv[j] = mu * vv[j] - s.lr * gr;
        float gr2 = gv2[j] + s.wd * wv2[j];
        vv2[j] = mu * vv2[j] - s.lr * gr2;
        if (phase == 0) { wv[j] += vv[j]; wv2[j] += vv2[j]; }
      }
    }
    if (phase != 2) {
      *(f32x4*)(s.v + i) = vv;
      if (two) *(f32x4*)(s.v + k) = vv2;
    }
    if (phase != 1) {
      *(f32x4*)(s.w + i) = wv;
      if (two) *(f32x4*)(s.w + k) = wv2;
      if (s.out) {
        bf16x4 o, o2;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          o[j] = (__bf16)wv[j];
          o2[j] = (__bf16)wv2[j];
        }
        *(bf16x4*)(s.out + i) = o;
        if (two) *(bf16x4*)(s.out + k) = o2;
      }
    }
  }
  // scalar tail
  if (blockIdx.x == 0 && threadIdx.x < (s.ntail - s.n4 * 4)) {
    long i = s.n4 * 4 + threadIdx.x;
    float wv = s.w[i], gv = s.g[i], vv = s.v[i];
    if (phase == 2) {
      wv += gv;
    } else {
      vv = mu * vv - s.lr * (gv + s.wd * wv);
      if (phase == 0) wv += vv;
    }
    if (phase != 2) s.v[i] = vv;
    if (phase != 1) {
      s.w[i] = wv;
      if (s.out) s.out[i] = (__bf16)wv;
    }
  }
}

struct GatherSeg {
  const __bf16* src;
  const int* map;
  __bf16* dst;
  long n;
};

__global__ void gather_multi_k(const GatherSeg* __restrict__ segs) {
  const GatherSeg s = segs[blockIdx.y];
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < s.n;
       i += (long)gridDim.x * blockDim.x) {
    int m = s.map[i];
    s.dst[i] = m < 0 ? (__bf16)0.f : s.src[m];
  }
}

extern "C" {

hipError_t launch_sgd_multi(const void* segs_dev, int nseg, long max_n4,
                            float mu, int phase, hipStream_t stream) {
  long gx = (max_n4 + 255) / 256;
  if (gx > 1024) gx = 1024;    // big FC segments dominate; fill the chip
  if (gx < 1) gx = 1;
  dim3 grid((unsigned)gx, nseg);
  hipLaunchKernelGGL(sgd_multi_k, grid, dim3(256), 0, stream,
                     (const SgdSeg*)segs_dev, mu, phase);
  return hipGetLastError();
}

hipError_t launch_gather_multi(const void* segs_dev, int nseg, long max_n,
                               hipStream_t stream) {
  long gx = (max_n + 255) / 256;
  if (gx > 512) gx = 512;
  if (gx < 1) gx = 1;
  dim3 grid((unsigned)gx, nseg);
  hipLaunchKernelGGL(gather_multi_k, grid, dim3(256), 0, stream,
                     (const GatherSeg*)segs_dev);
  return hipGetLastError();
}

}  // extern "C"
