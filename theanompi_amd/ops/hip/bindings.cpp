// Python bindings for the theanompi_amd HIP kernel library.
// Pure glue: tensor checks, dimension plumbing, stream lookup.  All math
// lives in the .hip files (hand-written CDNA4 kernels).

#include <torch/extension.h>

#include <ATen/hip/HIPContext.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <atomic>
#include <cstdlib>
#include <cstring>
#include <mutex>
#include <unordered_map>
#include <vector>

// ---- extern "C" launchers from the .hip TUs --------------------------------

struct GatherGemmArgs {
  const void* X;
  const void* W;
  const float* bias;
  const int* table;
  void* Y;
  int M, P, Q;
  long sN, sP, sQ, outOff;
  long inNS, inPS, inQS;
  int groupInOff;
  int Kdf;
  int BNtot;
  int relu;
  int splitK;
  float* Y32;
  long xTotal;
  float* stats;      // fused BN statistics buckets (or nullptr)
  int statsC;
};

struct WgradArgs {
  const void* X;
  const void* DY;
  const int* table;
  const float* kdfMask;
  float* DW;
  int M, P, Q;
  long inNS, inPS, inQS;
  int groupInOff;
  int Kdf, Kgp, Kphys;
  int splitK;
  long xTotal;
  float* db;          // optional fused bias grad
  float* WS;          // split-K plain-store workspace (or nullptr)
};

extern "C" {
hipError_t launch_conv_gather_gemm(GatherGemmArgs*, int G, int layout,
                                   hipStream_t, int variant);
hipError_t launch_conv_wgrad(const WgradArgs*, int G, int layout,
                             hipStream_t, int variant);
hipError_t launch_bias_wgrad(const void*, float*, long, int, hipStream_t);
hipError_t launch_halo_pad(const void*, void*, int, int, int, int, int, int,
                           int, int, float, long, hipStream_t);
hipError_t launch_relu_fwd(const void*, void*, long, hipStream_t);
hipError_t launch_relu_bwd_s(const void*, const void*, void*, long, int,
                             long, long, hipStream_t);
hipError_t launch_relu_bwd(const void*, const void*, void*, long,
                           hipStream_t);
hipError_t launch_relu_bwd_dual(const void*, const void*, void*, void*,
                                long, int, int, int, int, int, int,
                                int, long, long, hipStream_t);
hipError_t launch_add_relu(const void*, const void*, void*, long, int,
                           hipStream_t);
hipError_t launch_dropout_fwd(const void*, void*, unsigned char*, long,
                              float, const unsigned long long*,
                              hipStream_t);
hipError_t launch_dropout_bwd(const void*, const unsigned char*, void*, long,
                              float, hipStream_t);
hipError_t launch_gather_bf16(const void*, const int*, void*, long,
                              hipStream_t);
hipError_t launch_transpose_bf16(const void*, void*, int, int, long, long,
                                 hipStream_t);
hipError_t launch_u8_finish(const unsigned char*, void*, const float*,
                            long, long, int, int,
                            hipStream_t);
hipError_t launch_pack_bf16(const float*, void*, long, hipStream_t);
hipError_t launch_unpack_bf16(const void*, float*, long, hipStream_t);
hipError_t launch_segsum(const void*, void*, long, int, int, hipStream_t);
hipError_t launch_maxpool_fwd(const void*, void*, unsigned char*, int, int,
                              int, int, int, int, int, int, int, int,
                              hipStream_t);
hipError_t launch_maxpool_bwd(const void*, const unsigned char*, void*, int,
                              int, int, int, int, int, int, int, int, int,
                              int, int, hipStream_t);
hipError_t launch_avgpool_fwd(const void*, void*, int, int, int, int, int,
                              int, int, int, int, int, hipStream_t);
hipError_t launch_avgpool_bwd(const void*, void*, int, int, int, int, int,
                              int, int, int, int, int, int, int,
                              hipStream_t);
hipError_t launch_lrn_fwd(const void*, void*, void*, long, int, int, float,
                          float, float, hipStream_t);
hipError_t launch_lrn_bwd(const void*, const void*, const void*, void*,
                          long, int, int, float, float, float,
                          hipStream_t);
hipError_t launch_bn_fwd_train(const void*, void*, float*, float*,
                               float*, float*, float*, float*,
                               const float*, const float*, float*, int,
                               long, int, float, float, int,
                               const void*, hipStream_t);
hipError_t launch_bn_fwd_eval(const void*, void*, float*, float*,
                              const float*, const float*, const float*,
                              const float*, long, int, float, int,
                              const void*, hipStream_t);
hipError_t launch_bn_bwd(const void*, const void*, const void*, const float*,
                         const float*, const float*, const float*, float*,
                         int*, float*, float*, void*, long, int, int,
                         int, int, void*, hipStream_t);
hipError_t launch_softmax_xent_fwd(const void*, const long*, void*, float*,
                                   int, int, int, float*, hipStream_t);
hipError_t launch_softmax_xent_bwd(const void*, const long*, void*, int, int,
                                   int, float, hipStream_t);
hipError_t launch_sgd_step(float*, const float*, float*, void*, long, float,
                           float, float, int, hipStream_t);
hipError_t launch_mfma_probe(const void*, const void*, float*, int,
                             hipStream_t);
hipError_t launch_sgd_multi(const void*, int, long, float, int,
                            hipStream_t);
hipError_t launch_gather_multi(const void*, int, long, hipStream_t);
hipError_t launch_transpose_multi(const void*, int, int, int,
                                  hipStream_t);
hipError_t launch_tr_probe(const int*, float*, int, hipStream_t);
hipError_t launch_splitk_finalize(float*, const float*, void*, long,
                                  int, int, hipStream_t);
}

// pick a K-split so small-M GEMMs (FC layers at batch<=256) still fill
// the 256-CU chip; partials go through a zeroed fp32 workspace.
static int pick_splitk(long baseBlocks, long nt) {
  static long thresh = -1;
  if (thresh < 0) {
    const char* e = getenv("THEANOMPI_SPLITK_T");
    // swept twice on hardware: 128 best mid-round; after the
    // launch-hygiene work 192 measured +1.8% on AlexNet (FC split-K),
    // neutral on GoogLeNet
    thresh = e ? atol(e) : 192;
  }
  if (baseBlocks >= thresh || nt < 4) return 1;
  long s = 512 / (baseBlocks ? baseBlocks : 1);
  if (s > nt) s = nt;
  if (s < 1) s = 1;
  return (int)s;
}

struct SgdSegH {
  float* w;
  const float* g;
  float* v;
  void* out;
  long n4;
  long ntail;
  float lr, wd;
};

struct GatherSegH {
  const void* src;
  const int* map;
  void* dst;
  long n;
};

struct TransSegH {
  const void* src;
  void* dst;
  int rows, cols;
  long sstride, dstride;
};

// ---- helpers ---------------------------------------------------------------

#define CHECK_HIP(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    TORCH_CHECK(_e == hipSuccess, "HIP kernel launch failed: ",             \
                hipGetErrorString(_e));                                     \
  } while (0)

static hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

static void check_bf16(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
}

static void check_f32(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.is_contiguous()
                  && t.scalar_type() == torch::kFloat32,
              name, " must be contiguous f32 GPU tensor");
}

// ---- MFMA layout probe ------------------------------------------------------

static int g_layout = -1;

static float bf16_quant(float v) {
  unsigned int u;
  std::memcpy(&u, &v, 4);
  unsigned int lsb = (u >> 16) & 1;
  u += 0x7fff + lsb;
  u &= 0xffff0000u;
  float r;
  std::memcpy(&r, &u, 4);
  return r;
}

static int detect_layout() {
  if (g_layout >= 0) return g_layout;
  auto optsF = torch::TensorOptions()
                   .dtype(torch::kFloat32)
                   .device(torch::kCUDA);
  torch::manual_seed(0);
  auto Af = torch::randn({16, 32}, optsF.device(torch::kCPU));
  auto Bf = torch::randn({16, 32}, optsF.device(torch::kCPU));
  // quantize to bf16 grid so the host reference is exact-ish
  float* ap = Af.data_ptr<float>();
  float* bp = Bf.data_ptr<float>();
  for (int i = 0; i < 512; ++i) {
    ap[i] = bf16_quant(ap[i]);
    bp[i] = bf16_quant(bp[i]);
  }
  // reference: D[m][n] = sum_k A[m][k] * B[n][k]   (B stored n-major)
  float ref[16][16];
  for (int m = 0; m < 16; ++m)
    for (int n = 0; n < 16; ++n) {
      float s = 0.f;
      for (int k = 0; k < 32; ++k) s += ap[m * 32 + k] * bp[n * 32 + k];
      ref[m][n] = s;
    }
  auto A = Af.to(torch::kBFloat16).to(torch::kCUDA);
  auto B = Bf.to(torch::kBFloat16).to(torch::kCUDA);
  auto D = torch::zeros({16, 16}, optsF);
  for (int layout = 0; layout < 2; ++layout) {
    CHECK_HIP(launch_mfma_probe(A.data_ptr(), B.data_ptr(),
                                D.data_ptr<float>(), layout, cur_stream()));
    auto Dh = D.cpu();
    const float* dp = Dh.data_ptr<float>();
    double err = 0.0;
    for (int i = 0; i < 256; ++i)
      err += std::abs(dp[i] - ((const float*)ref)[i]);
    if (err / 256.0 < 1e-2) {
      g_layout = layout;
      return g_layout;
    }
  }
  TORCH_CHECK(false,
              "mfma_probe: neither fragment layout matched the reference — "
              "A/B operand mapping for v_mfma_f32_16x16x32_bf16 differs "
              "from both candidates");
  return -1;
}

// ---- conv ops ---------------------------------------------------------------

static torch::Tensor halo_pad(torch::Tensor x, long t, long b, long l,
                              long r, double fill) {
  // x may be a CHANNEL SLICE of a wider NHWC buffer (merged-branch
  // outputs): inner dim contiguous, pixels a uniform srcC apart
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(x.dim() == 4, "x must be NHWC");
  auto N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  TORCH_CHECK(C % 4 == 0, "C must be a multiple of 4");
  long srcC = C;
  if (!x.is_contiguous()) {
    TORCH_CHECK(x.stride(3) == 1 && x.stride(2) >= C
                && x.stride(1) == x.stride(2) * W
                && x.stride(0) == x.stride(1) * H,
                "halo_pad: need a channel-slice layout");
    srcC = x.stride(2);
  }
  auto y = torch::empty({N, H + t + b, W + l + r, C}, x.options());
  CHECK_HIP(launch_halo_pad(x.data_ptr(), y.data_ptr(), N, H, W, C, t, b,
                            l, r, (float)fill, srcC, cur_stream()));
  return y;
}

static torch::Tensor halo_pad_sym(torch::Tensor x, long ph, long pw,
                                  double fill) {
  return halo_pad(x, ph, ph, pw, pw, fill);
}

static long row_ld(const torch::Tensor& t);

constexpr int STATS_NB = 32;   // must match conv.hip

// Split-K fp32 workspace, cached per device and re-zeroed by the
// finalize kernel as it drains it, so the steady state needs no fill
// launch at all (fills were ~14% of the GoogLeNet step).
static float* splitk_ws(long numel, const torch::TensorOptions& opts) {
  static torch::Tensor ws[16];
  int d = opts.device().index() < 0 ? 0 : opts.device().index();
  if (!ws[d].defined() || ws[d].numel() < numel)
    ws[d] = torch::zeros({numel}, opts.dtype(torch::kFloat32));
  return ws[d].data_ptr<float>();
}

static bool conv_fwd(torch::Tensor xp, torch::Tensor w, torch::Tensor bias,
                     torch::Tensor table, torch::Tensor y, long rowStride,
                     long colStride, long sh, long sw, long groupInOff,
                     bool relu,
                     c10::optional<torch::Tensor> stats = c10::nullopt,
                     long variant = 0) {
  check_bf16(xp, "xp");
  check_bf16(w, "w");
  // y may be a channel slice of a concat buffer (concat-writing conv:
  // last dim contiguous, uniform row stride) — strided epilogue writes.
  TORCH_CHECK(y.scalar_type() == torch::kBFloat16 && y.is_cuda());
  const bool yContig = y.is_contiguous();
  TORCH_CHECK(yContig || row_ld(y) > 0, "conv_fwd: unsupported y layout");
  if (bias.numel()) check_f32(bias, "bias");
  TORCH_CHECK(table.scalar_type() == torch::kInt32 && table.is_cuda());
  TORCH_CHECK(w.dim() == 3 && xp.dim() == 4 && y.dim() == 4);
  long N = y.size(0), P = y.size(1), Q = y.size(2), Kphys = y.size(3);
  long G = w.size(0), Kgp = w.size(1), Kdf = w.size(2);
  TORCH_CHECK(Kphys == G * Kgp, "output channels mismatch");
  TORCH_CHECK(Kdf % 64 == 0, "Kdf must be padded to 64");
  long Hp = xp.size(1), Wp = xp.size(2), Cphys = xp.size(3);
  TORCH_CHECK(rowStride == Wp * Cphys && colStride == Cphys,
              "stride metadata mismatch");
  GatherGemmArgs a;
  a.X = xp.data_ptr();
  a.W = w.data_ptr();
  a.bias = bias.numel() ? bias.data_ptr<float>() : nullptr;
  a.table = table.data_ptr<int>();
  a.Y = y.data_ptr();
  a.M = (int)(N * P * Q);
  a.P = (int)P;
  a.Q = (int)Q;
  a.sN = yContig ? P * Q * Kphys : y.stride(0);
  a.sP = yContig ? Q * Kphys : y.stride(1);
  a.sQ = yContig ? Kphys : y.stride(2);
  a.outOff = 0;
  a.inNS = Hp * Wp * Cphys;
  a.inPS = sh * rowStride;
  a.inQS = sw * colStride;
  a.groupInOff = (int)groupInOff;
  a.Kdf = (int)Kdf;
  a.BNtot = (int)Kgp;
  a.relu = relu ? 1 : 0;
  a.xTotal = xp.numel();
  long BN = Kgp > 64 ? 128 : 64;
  long baseBlocks = ((a.M + 127) / 128) * ((Kgp + BN - 1) / BN) * G;
  // split-K indexes its fp32 workspace with the output strides, so it
  // requires a contiguous y
  a.splitK = yContig ? pick_splitk(baseBlocks, Kdf / 64) : 1;
  a.Y32 = nullptr;
  if (a.splitK > 1)
    a.Y32 = splitk_ws(y.numel(), y.options());
  a.stats = nullptr;
  a.statsC = (int)(G * Kgp);
  if (stats.has_value()) {
    auto& st = stats.value();
    TORCH_CHECK(st.scalar_type() == torch::kFloat32 && st.is_cuda()
                && st.is_contiguous()
                && st.numel() == (long)STATS_NB * 2 * G * Kgp,
                "stats must be fp32 [", STATS_NB, ", 2, C]");
    a.stats = st.data_ptr<float>();
  }
  CHECK_HIP(launch_conv_gather_gemm(&a, (int)G, detect_layout(),
                                    cur_stream(), (int)variant));
  if (a.splitK > 1) {
    CHECK_HIP(launch_splitk_finalize(a.Y32, a.bias, y.data_ptr(),
                                     y.numel(), (int)(G * Kgp),
                                     a.relu, cur_stream()));
  }
  return a.stats != nullptr;   // whether fused stats were collected
}

static void conv_dgrad(torch::Tensor dyp, torch::Tensor wdg,
                       torch::Tensor table, torch::Tensor dx,
                       long rowStride, long colStride, long Hi, long Wj,
                       long clsI, long clsJ, long sh, long sw,
                       long groupInOff, long Cgp, long variant = 0) {
  check_bf16(dyp, "dyp");
  check_bf16(wdg, "wdg");
  check_bf16(dx, "dx");
  TORCH_CHECK(wdg.dim() == 3 && dyp.dim() == 4 && dx.dim() == 4);
  long N = dx.size(0), H = dx.size(1), W = dx.size(2), Cphys = dx.size(3);
  long G = wdg.size(0), BNtot = wdg.size(1), Kdf2 = wdg.size(2);
  TORCH_CHECK(BNtot == Cgp, "dgrad output-channel mismatch");
  long Hp2 = dyp.size(1), Qp = dyp.size(2), Kphys = dyp.size(3);
  TORCH_CHECK(rowStride == Qp * Kphys && colStride == Kphys);
  TORCH_CHECK(Kdf2 % 64 == 0);
  static torch::Tensor dummy_bias;  // dgrad has no bias
  GatherGemmArgs a;
  a.X = dyp.data_ptr();
  a.W = wdg.data_ptr();
  a.bias = nullptr;
  a.table = table.data_ptr<int>();
  a.Y = dx.data_ptr();
  a.M = (int)(N * Hi * Wj);
  a.P = (int)Hi;
  a.Q = (int)Wj;
  a.sN = H * W * Cphys;
  a.sP = sh * W * Cphys;
  a.sQ = sw * Cphys;
  a.outOff = (clsI * W + clsJ) * Cphys;
  a.inNS = Hp2 * Qp * Kphys;
  a.inPS = rowStride;   // dy p-step is 1 row of the padded dy
  a.inQS = colStride;
  a.groupInOff = (int)groupInOff;
  a.Kdf = (int)Kdf2;
  a.BNtot = (int)BNtot;
  a.relu = 0;
  a.xTotal = dyp.numel();
  a.splitK = 1;
  a.Y32 = nullptr;
  a.stats = nullptr;
  a.statsC = 0;
  if (sh == 1 && sw == 1) {      // one class = dense dx: split-K legal
    long BN = BNtot > 64 ? 128 : 64;
    long baseBlocks = ((a.M + 127) / 128) * ((BNtot + BN - 1) / BN) * G;
    a.splitK = pick_splitk(baseBlocks, Kdf2 / 64);
    if (a.splitK > 1)
      a.Y32 = splitk_ws(dx.numel(), dx.options());
  }
  CHECK_HIP(launch_conv_gather_gemm(&a, (int)G, detect_layout(),
                                    cur_stream(), (int)variant));
  if (a.splitK > 1) {
    CHECK_HIP(launch_splitk_finalize(a.Y32, nullptr, dx.data_ptr(),
                                     dx.numel(), (int)Cgp, 0,
                                     cur_stream()));
  }
}

static void conv_wgrad(torch::Tensor xp, torch::Tensor dy,
                       torch::Tensor table, torch::Tensor dw,
                       torch::Tensor kdfMask,
                       long rowStride, long colStride, long sh, long sw,
                       long groupInOff, long variant = 0,
                       long split_target = 0,
                       c10::optional<torch::Tensor> db = c10::nullopt) {
  check_bf16(xp, "xp");
  check_bf16(dy, "dy");
  check_f32(dw, "dw");
  check_f32(kdfMask, "kdfMask");
  long N = dy.size(0), P = dy.size(1), Q = dy.size(2), Kphys = dy.size(3);
  long G = dw.size(0), Kgp = dw.size(1), Kdf = dw.size(2);
  TORCH_CHECK(Kphys == G * Kgp);
  long Hp = xp.size(1), Wp = xp.size(2), Cphys = xp.size(3);
  TORCH_CHECK(rowStride == Wp * Cphys && colStride == Cphys);
  WgradArgs a;
  a.X = xp.data_ptr();
  a.DY = dy.data_ptr();
  a.table = table.data_ptr<int>();
  a.kdfMask = kdfMask.data_ptr<float>();
  a.DW = dw.data_ptr<float>();
  a.M = (int)(N * P * Q);
  a.P = (int)P;
  a.Q = (int)Q;
  a.inNS = Hp * Wp * Cphys;
  a.inPS = sh * rowStride;
  a.inQS = sw * colStride;
  a.groupInOff = (int)groupInOff;
  a.Kdf = (int)Kdf;
  a.Kgp = (int)Kgp;
  a.Kphys = (int)Kphys;
  a.xTotal = xp.numel();
  a.db = nullptr;
  if (db.has_value()) {
    auto& dbt = db.value();
    check_f32(dbt, "db");
    TORCH_CHECK(dbt.numel() == G * Kgp, "db must be [G*Kgp]");
    a.db = dbt.data_ptr<float>();
  }
  long base_blocks = ((Kgp + 63) / 64) * ((Kdf + 63) / 64) * G;
  static const long wgrad_env = [] {
    const char* e = getenv("THEANOMPI_WGRAD_T");
    return e ? atol(e) : 0L;             // total-block target override
  }();
  // grouped convs shrink the per-group GEMM (few base blocks) and
  // measured ~1.5% faster end-to-end with a 2048-block target;
  // ungrouped layers measured best at 1024 (swept on hardware).
  long wgrad_target = wgrad_env ? wgrad_env
                      : split_target ? split_target
                      : (G > 1 ? 2048 : 1024);
  long splitK = wgrad_target / (base_blocks ? base_blocks : 1);
  long maxSplit = (a.M + 63) / 64;
  if (splitK < 1) splitK = 1;
  if (splitK > maxSplit) splitK = maxSplit;
  a.splitK = (int)splitK;
  a.WS = nullptr;
  torch::Tensor ws;
  // measured on hardware (interleaved A/B, both reps): the atomic
  // split-K epilogue BEATS the plain-workspace+segsum two-stage by
  // ~2% end-to-end on VGG/ResNet — MI355X fp32 HBM atomics absorb
  // even ~455 writers per line, while the workspace pays an extra
  // dW-sized round-trip per slice.  Kept opt-in for other silicon.
  static const bool use_ws = [] {
    const char* e = getenv("THEANOMPI_WGRAD_WS");
    return e && e[0] == '1';
  }();
  if (a.splitK > 1 && use_ws) {
    // fully overwritten every call, so reused unzeroed
    static std::mutex ws_mu;
    static std::unordered_map<int, torch::Tensor> ws_cache;
    long need = (long)a.splitK * G * Kgp * Kdf;
    {
      std::lock_guard<std::mutex> lk(ws_mu);
      auto& slot = ws_cache[(int)dw.device().index()];
      if (!slot.defined() || slot.numel() < need)
        slot = torch::empty({need}, dw.options());
      ws = slot;
    }
    a.WS = ws.data_ptr<float>();
  }
  CHECK_HIP(launch_conv_wgrad(&a, (int)G, detect_layout(), cur_stream(),
                              (int)variant));
  if (a.WS) {
    CHECK_HIP(launch_segsum(a.WS, dw.data_ptr<float>(),
                            (long)G * Kgp * Kdf, a.splitK, 0,
                            cur_stream()));
  }
}

static void bias_wgrad(torch::Tensor dy, torch::Tensor db) {
  check_bf16(dy, "dy");
  check_f32(db, "db");
  long C = dy.size(-1);
  long rows = dy.numel() / C;
  TORCH_CHECK(db.numel() == C);
  CHECK_HIP(launch_bias_wgrad(dy.data_ptr(), db.data_ptr<float>(), rows,
                              (int)C, cur_stream()));
}

// ---- stateless ops ----------------------------------------------------------

static void relu_fwd(torch::Tensor x, torch::Tensor y) {
  check_bf16(x, "x");
  CHECK_HIP(launch_relu_fwd(x.data_ptr(), y.data_ptr(), x.numel(),
                            cur_stream()));
}

// Row stride of a 4D view whose last dim is contiguous and whose rows
// sit a uniform stride apart (contiguous tensors return C; anything
// else returns -1).
static long row_ld(const torch::Tensor& t) {
  if (t.is_contiguous()) return t.size(-1);
  if (t.dim() != 4 || t.stride(3) != 1) return -1;
  long ld = t.stride(2);
  if (ld < t.size(3)) return -1;
  if (t.stride(1) != ld * t.size(2)) return -1;
  if (t.stride(0) != t.stride(1) * t.size(1)) return -1;
  return ld;
}

// Accepts contiguous dy/y OR channel slices of a concat buffer
// (last-dim contiguous, uniform row stride) — the inception backward
// hands slices straight in, no .contiguous() copy.
static void relu_bwd(torch::Tensor dy, torch::Tensor y, torch::Tensor dx) {
  TORCH_CHECK(dy.scalar_type() == torch::kBFloat16);
  if (dy.is_contiguous() && y.is_contiguous()) {
    CHECK_HIP(launch_relu_bwd(dy.data_ptr(), y.data_ptr(), dx.data_ptr(),
                              dx.numel(), cur_stream()));
    return;
  }
  long C = dy.size(-1);
  long ldDy = row_ld(dy), ldY = row_ld(y);
  TORCH_CHECK(ldDy > 0 && ldY > 0, "relu_bwd: unsupported layout");
  long rows = dx.numel() / C;
  CHECK_HIP(launch_relu_bwd_s(dy.data_ptr(), y.data_ptr(), dx.data_ptr(),
                              rows * C, (int)C, ldDy, ldY,
                              cur_stream()));
}

// dual-output relu backward: dx contiguous [N,H,W,C] for wgrad/bias,
// AND the same masked values into the interior of a halo-padded
// buffer dxp [N,H+2ht',W+2hl',C] (borders stay zero; cached by the
// layer) — the single-class dgrad then skips its halo_pad launch.
static void relu_bwd_dual(torch::Tensor dy, torch::Tensor y,
                          torch::Tensor dx, torch::Tensor dxp, long ht,
                          long hl) {
  TORCH_CHECK(dy.scalar_type() == torch::kBFloat16);
  check_bf16(dx, "dx");
  check_bf16(dxp, "dxp");
  long N = dx.size(0), H = dx.size(1), W = dx.size(2), C = dx.size(3);
  TORCH_CHECK(dxp.size(0) == N && dxp.size(3) == C);
  long Hp2 = dxp.size(1), Wp2 = dxp.size(2);
  TORCH_CHECK(C % 8 == 0, "relu_bwd_dual: C % 8");
  long ldDy = dy.is_contiguous() ? C : row_ld(dy);
  long ldY = y.is_contiguous() ? C : row_ld(y);
  TORCH_CHECK(ldDy > 0 && ldY > 0, "relu_bwd_dual: unsupported layout");
  CHECK_HIP(launch_relu_bwd_dual(dy.data_ptr(), y.data_ptr(),
                                 dx.data_ptr(), dxp.data_ptr(),
                                 dx.numel(), (int)C, (int)H, (int)W,
                                 (int)Hp2, (int)Wp2, (int)ht, (int)hl,
                                 ldDy, ldY, cur_stream()));
}

static void add_relu(torch::Tensor a, torch::Tensor b, torch::Tensor y,
                     bool relu) {
  check_bf16(a, "a");
  CHECK_HIP(launch_add_relu(a.data_ptr(), b.data_ptr(), y.data_ptr(),
                            a.numel(), relu ? 1 : 0, cur_stream()));
}

static void dropout_fwd(torch::Tensor x, torch::Tensor y, torch::Tensor mask,
                        double p, torch::Tensor seed) {
  check_bf16(x, "x");
  TORCH_CHECK(mask.scalar_type() == torch::kUInt8);
  TORCH_CHECK(seed.scalar_type() == torch::kInt64 && seed.is_cuda());
  CHECK_HIP(launch_dropout_fwd(
      x.data_ptr(), y.data_ptr(), mask.data_ptr<unsigned char>(),
      x.numel(), (float)p,
      (const unsigned long long*)seed.data_ptr(), cur_stream()));
}

static void dropout_bwd(torch::Tensor dy, torch::Tensor mask,
                        torch::Tensor dx, double p) {
  check_bf16(dy, "dy");
  CHECK_HIP(launch_dropout_bwd(dy.data_ptr(),
                               mask.data_ptr<unsigned char>(),
                               dx.data_ptr(), dy.numel(), (float)p,
                               cur_stream()));
}

static void gather_bf16(torch::Tensor src, torch::Tensor map,
                        torch::Tensor dst) {
  check_bf16(src, "src");
  check_bf16(dst, "dst");
  TORCH_CHECK(map.scalar_type() == torch::kInt32);
  TORCH_CHECK(map.numel() == dst.numel());
  CHECK_HIP(launch_gather_bf16(src.data_ptr(), map.data_ptr<int>(),
                               dst.data_ptr(), dst.numel(), cur_stream()));
}

static torch::Tensor u8_finish(torch::Tensor src, long nLogC,
                               c10::optional<torch::Tensor> meanimg) {
  TORCH_CHECK(src.scalar_type() == torch::kUInt8 && src.is_cuda()
              && src.is_contiguous());
  auto dst = torch::empty(src.sizes(),
                          src.options().dtype(torch::kBFloat16));
  long C = src.size(-1);
  long perImg = src.numel() / src.size(0);
  const float* mp = nullptr;
  if (meanimg.has_value()) {
    auto& m = meanimg.value();
    TORCH_CHECK(m.scalar_type() == torch::kFloat32 && m.is_cuda()
                && m.is_contiguous() && m.numel() == perImg,
                "mean image must be fp32 [crop, crop, C] on device");
    mp = m.data_ptr<float>();
  }
  CHECK_HIP(launch_u8_finish(src.data_ptr<unsigned char>(),
                             dst.data_ptr(), mp, src.numel(), perImg,
                             (int)C, (int)nLogC, cur_stream()));
  return dst;
}

static void transpose_bf16(torch::Tensor src, torch::Tensor dst,
                           long rows, long cols, long sstride,
                           long dstride) {
  check_bf16(src, "src");
  check_bf16(dst, "dst");
  CHECK_HIP(launch_transpose_bf16(src.data_ptr(), dst.data_ptr(),
                                  (int)rows, (int)cols, sstride, dstride,
                                  cur_stream()));
}

static void pack_bf16(torch::Tensor src, torch::Tensor dst) {
  check_f32(src, "src");
  check_bf16(dst, "dst");
  TORCH_CHECK(src.numel() == dst.numel() && src.numel() % 4 == 0);
  CHECK_HIP(launch_pack_bf16(src.data_ptr<float>(), dst.data_ptr(),
                             src.numel(), cur_stream()));
}

static void unpack_bf16(torch::Tensor src, torch::Tensor dst) {
  check_bf16(src, "src");
  check_f32(dst, "dst");
  TORCH_CHECK(src.numel() == dst.numel() && src.numel() % 4 == 0);
  CHECK_HIP(launch_unpack_bf16(src.data_ptr(), dst.data_ptr<float>(),
                               src.numel(), cur_stream()));
}

static void segsum(torch::Tensor in, torch::Tensor out, long w) {
  // in: [w*seg] contiguous, out: [seg]; fp32 accumulation (N14)
  TORCH_CHECK(in.dtype() == out.dtype(), "segsum: dtype mismatch");
  TORCH_CHECK(in.is_contiguous() && out.is_contiguous());
  long seg = out.numel();
  TORCH_CHECK(in.numel() == w * seg, "segsum: in != w*out");
  TORCH_CHECK(seg % 4 == 0, "segsum: seg must be a multiple of 4");
  int bf16 = in.dtype() == torch::kBFloat16;
  TORCH_CHECK(bf16 || in.dtype() == torch::kFloat32);
  CHECK_HIP(launch_segsum(in.data_ptr(), out.data_ptr(), seg, (int)w,
                          bf16, cur_stream()));
}

// ---- pooling ----------------------------------------------------------------

static void maxpool_fwd(torch::Tensor xp, torch::Tensor y,
                        torch::Tensor am, long kh, long kw, long sh,
                        long sw) {
  check_bf16(xp, "xp");
  check_bf16(y, "y");
  long N = xp.size(0), Hp = xp.size(1), Wp = xp.size(2), C = xp.size(3);
  long P = y.size(1), Q = y.size(2);
  CHECK_HIP(launch_maxpool_fwd(xp.data_ptr(), y.data_ptr(),
                               am.data_ptr<unsigned char>(), N, Hp, Wp, C,
                               P, Q, kh, kw, sh, sw, cur_stream()));
}

static void maxpool_bwd(torch::Tensor dy, torch::Tensor am,
                        torch::Tensor dx, long kh, long kw, long sh,
                        long sw, long ph, long pw) {
  check_bf16(dy, "dy");
  long N = dx.size(0), H = dx.size(1), W = dx.size(2), C = dx.size(3);
  long P = dy.size(1), Q = dy.size(2);
  CHECK_HIP(launch_maxpool_bwd(dy.data_ptr(),
                               am.data_ptr<unsigned char>(), dx.data_ptr(),
                               N, H, W, C, P, Q, kh, kw, sh, sw, ph, pw,
                               cur_stream()));
}

static void avgpool_fwd(torch::Tensor xp, torch::Tensor y, long kh, long kw,
                        long sh, long sw) {
  check_bf16(xp, "xp");
  long N = xp.size(0), Hp = xp.size(1), Wp = xp.size(2), C = xp.size(3);
  long P = y.size(1), Q = y.size(2);
  CHECK_HIP(launch_avgpool_fwd(xp.data_ptr(), y.data_ptr(), N, Hp, Wp, C,
                               P, Q, kh, kw, sh, sw, cur_stream()));
}

static void avgpool_bwd(torch::Tensor dy, torch::Tensor dx, long kh,
                        long kw, long sh, long sw, long ph, long pw) {
  check_bf16(dy, "dy");
  long N = dx.size(0), H = dx.size(1), W = dx.size(2), C = dx.size(3);
  long P = dy.size(1), Q = dy.size(2);
  CHECK_HIP(launch_avgpool_bwd(dy.data_ptr(), dx.data_ptr(), N, H, W, C, P,
                               Q, kh, kw, sh, sw, ph, pw, cur_stream()));
}

// ---- normalisation ----------------------------------------------------------

static void lrn_fwd(torch::Tensor x, torch::Tensor y, torch::Tensor scale,
                    long size, double alpha, double beta, double k) {
  check_bf16(x, "x");
  check_bf16(scale, "scale");          // 1/denom saved in bf16
  long C = x.size(-1);
  long rows = x.numel() / C;
  CHECK_HIP(launch_lrn_fwd(x.data_ptr(), y.data_ptr(), scale.data_ptr(),
                           rows, (int)C, (int)size, (float)alpha,
                           (float)beta, (float)k, cur_stream()));
}

static void lrn_bwd(torch::Tensor dy, torch::Tensor y,
                    torch::Tensor scale, torch::Tensor dx, long size,
                    double alpha, double beta, double k) {
  check_bf16(dy, "dy");
  check_bf16(scale, "scale");
  long C = y.size(-1);
  long rows = y.numel() / C;
  CHECK_HIP(launch_lrn_bwd(dy.data_ptr(), y.data_ptr(), scale.data_ptr(),
                           dx.data_ptr(), rows, (int)C, (int)size,
                           (float)alpha, (float)beta, (float)k,
                           cur_stream()));
}

static void bn_fwd(torch::Tensor x, torch::Tensor gamma, torch::Tensor beta,
                   torch::Tensor y, torch::Tensor save_mean,
                   torch::Tensor save_invstd, torch::Tensor rmean,
                   torch::Tensor rvar, double momentum, double eps,
                   bool training, bool relu,
                   c10::optional<torch::Tensor> prestats = c10::nullopt,
                   c10::optional<torch::Tensor> res = c10::nullopt) {
  check_bf16(x, "x");
  check_f32(gamma, "gamma");
  long C = x.size(-1);
  long rows = x.numel() / C;
  const void* resp = nullptr;
  if (res.has_value()) {
    check_bf16(res.value(), "res");
    TORCH_CHECK(res.value().numel() == x.numel());
    resp = res.value().data_ptr();
  }
  if (training) {
    // per-(device,C) cached accumulators, re-zeroed by bn_finalize_k
    // after each consume — no per-call fill launches
    static std::mutex bn_mu;
    static std::unordered_map<long, std::pair<torch::Tensor,
                                              torch::Tensor>> bn_accs;
    torch::Tensor s1, s2;
    {
      long key = ((long)x.device().index() << 32) | C;
      std::lock_guard<std::mutex> lk(bn_mu);
      auto it = bn_accs.find(key);
      if (it == bn_accs.end()) {
        it = bn_accs
                 .emplace(key,
                          std::make_pair(torch::zeros_like(save_mean),
                                         torch::zeros_like(save_mean)))
                 .first;
      }
      s1 = it->second.first;
      s2 = it->second.second;
    }
    float* pre = nullptr;
    if (prestats.has_value()) {
      auto& st = prestats.value();
      TORCH_CHECK(st.numel() == (long)STATS_NB * 2 * C,
                  "prestats must be [", STATS_NB, ", 2, C]");
      pre = st.data_ptr<float>();
    }
    CHECK_HIP(launch_bn_fwd_train(
        x.data_ptr(), y.data_ptr(), s1.data_ptr<float>(),
        s2.data_ptr<float>(), save_mean.data_ptr<float>(),
        save_invstd.data_ptr<float>(), rmean.data_ptr<float>(),
        rvar.data_ptr<float>(), gamma.data_ptr<float>(),
        beta.data_ptr<float>(), pre, STATS_NB, rows, (int)C,
        (float)momentum, (float)eps,
        relu ? 1 : 0, resp, cur_stream()));
  } else {
    CHECK_HIP(launch_bn_fwd_eval(
        x.data_ptr(), y.data_ptr(), save_mean.data_ptr<float>(),
        save_invstd.data_ptr<float>(), rmean.data_ptr<float>(),
        rvar.data_ptr<float>(), gamma.data_ptr<float>(),
        beta.data_ptr<float>(), rows, (int)C, (float)eps, relu ? 1 : 0,
        resp, cur_stream()));
  }
}

static void bn_bwd(torch::Tensor dy, torch::Tensor x, torch::Tensor gamma,
                   torch::Tensor beta, torch::Tensor save_mean,
                   torch::Tensor save_invstd, torch::Tensor y,
                   torch::Tensor dx, torch::Tensor dgamma,
                   torch::Tensor dbeta, bool relu,
                   c10::optional<torch::Tensor> dres = c10::nullopt,
                   bool has_res = false) {
  check_bf16(dy, "dy");
  long C = x.size(-1);
  long rows = x.numel() / C;
  // the per-channel accumulators ARE the grad outputs: sdy == dbeta
  // (sum dy), sdyx == dgamma (sum dy*xhat); prep zeroes them, the sums
  // pass accumulates, nothing to copy afterwards
  auto& sdy = dbeta;
  auto& sdyx = dgamma;
  auto rgamma = torch::empty_like(gamma);
  // THEANOMPI_BN_YPATH=0 forces the exact x path (A/B knob).  allok
  // is a PERSISTENT per-device cell holding the token of the last
  // call whose y-path was unusable — no per-call reset fill (that was
  // one FillFunctor launch per BN backward, 53/step on ResNet-50).
  static const bool ypath_on = [] {
    const char* e = getenv("THEANOMPI_BN_YPATH");
    return !(e && e[0] == '0');
  }();
  static std::mutex allok_mu;
  static std::unordered_map<int, torch::Tensor> allok_cells;
  torch::Tensor allok;
  {
    std::lock_guard<std::mutex> lk(allok_mu);
    auto it = allok_cells.find((int)gamma.device().index());
    if (it == allok_cells.end()) {
      it = allok_cells
               .emplace((int)gamma.device().index(),
                        torch::zeros({1}, torch::dtype(torch::kInt32)
                                              .device(gamma.device())))
               .first;
    }
    allok = it->second;
  }
  static std::atomic<int> token_ctr{1};
  int token = token_ctr.fetch_add(1);
  if (token <= 0) { token_ctr.store(2); token = 1; }  // wrap safety
  CHECK_HIP(launch_bn_bwd(dy.data_ptr(), x.data_ptr(), y.data_ptr(),
                          save_mean.data_ptr<float>(),
                          save_invstd.data_ptr<float>(),
                          gamma.data_ptr<float>(),
                          beta.data_ptr<float>(),
                          rgamma.data_ptr<float>(),
                          allok.data_ptr<int>(), sdy.data_ptr<float>(),
                          sdyx.data_ptr<float>(), dx.data_ptr(), rows,
                          (int)C, relu ? 1 : 0, token,
                          // a fused residual means the saved y is NOT
                          // bn(x): the xhat-from-y reconstruction is
                          // invalid, force the exact x path
                          (!ypath_on || has_res) ? 1 : 0,
                          dres.has_value() ? dres.value().data_ptr()
                                           : nullptr,
                          cur_stream()));
}

// ---- softmax / sgd ----------------------------------------------------------

static void softmax_xent_fwd(torch::Tensor logits, torch::Tensor labels,
                             torch::Tensor probs, torch::Tensor loss,
                             long nCls,
                             c10::optional<torch::Tensor> stats
                             = c10::nullopt) {
  check_bf16(logits, "logits");
  TORCH_CHECK(labels.scalar_type() == torch::kInt64);
  long B = logits.size(0), Cp = logits.size(1);
  float* sp = nullptr;
  if (stats.has_value()) {
    auto& st = stats.value();
    TORCH_CHECK(st.scalar_type() == torch::kFloat32 && st.is_cuda()
                && st.numel() == 2, "stats must be fp32 [2]");
    sp = st.data_ptr<float>();
  }
  CHECK_HIP(launch_softmax_xent_fwd(logits.data_ptr(),
                                    labels.data_ptr<long>(),
                                    probs.data_ptr(),
                                    loss.data_ptr<float>(), (int)B,
                                    (int)Cp, (int)nCls, sp,
                                    cur_stream()));
}

static void softmax_xent_bwd(torch::Tensor probs, torch::Tensor labels,
                             torch::Tensor dlogits, double scale,
                             long nCls) {
  check_bf16(probs, "probs");
  long B = probs.size(0), Cp = probs.size(1);
  CHECK_HIP(launch_softmax_xent_bwd(probs.data_ptr(),
                                    labels.data_ptr<long>(),
                                    dlogits.data_ptr(), (int)B, (int)Cp,
                                    (int)nCls, (float)scale,
                                    cur_stream()));
}

static void sgd_step(torch::Tensor w, torch::Tensor g, torch::Tensor v,
                     torch::Tensor out, double lr, double mu, double wd,
                     long phase) {
  check_f32(w, "w");
  check_f32(g, "g");
  check_f32(v, "v");
  void* outp = out.numel() ? out.data_ptr() : nullptr;
  if (outp) TORCH_CHECK(out.numel() == w.numel());
  CHECK_HIP(launch_sgd_step(w.data_ptr<float>(), g.data_ptr<float>(),
                            v.data_ptr<float>(), outp, w.numel(),
                            (float)lr, (float)mu, (float)wd, (int)phase,
                            cur_stream()));
}

static long mfma_layout() { return detect_layout(); }

static torch::Tensor tr_probe(torch::Tensor offs, long fill_mode) {
  TORCH_CHECK(offs.scalar_type() == torch::kInt32 && offs.numel() == 64);
  auto out = torch::zeros({64, 4}, torch::TensorOptions()
                                       .dtype(torch::kFloat32)
                                       .device(offs.device()));
  CHECK_HIP(launch_tr_probe(offs.data_ptr<int>(), out.data_ptr<float>(),
                            (int)fill_mode, cur_stream()));
  return out;
}

// one launch for the whole parameter set (per optimizer phase)
static void sgd_step_multi(std::vector<torch::Tensor> ws,
                           std::vector<torch::Tensor> gs,
                           std::vector<torch::Tensor> vs,
                           std::vector<torch::Tensor> outs,
                           std::vector<double> lrs,
                           std::vector<double> wds, double mu,
                           long phase) {
  int n = (int)ws.size();
  TORCH_CHECK(n > 0);
  std::vector<SgdSegH> segs(n);
  long max_n4 = 1;
  for (int i = 0; i < n; ++i) {
    check_f32(ws[i], "w");
    long nn = ws[i].numel();
    segs[i] = {ws[i].data_ptr<float>(), gs[i].data_ptr<float>(),
               vs[i].data_ptr<float>(),
               outs[i].numel() ? outs[i].data_ptr() : nullptr,
               nn / 4, nn, (float)lrs[i], (float)wds[i]};
    if (nn / 4 > max_n4) max_n4 = nn / 4;
  }
  auto buf = torch::empty({(long)(n * sizeof(SgdSegH))},
                          torch::TensorOptions()
                              .dtype(torch::kUInt8)
                              .device(ws[0].device()));
  hipMemcpyAsync(buf.data_ptr(), segs.data(), n * sizeof(SgdSegH),
                 hipMemcpyHostToDevice, cur_stream());
  CHECK_HIP(launch_sgd_multi(buf.data_ptr(), n, max_n4, (float)mu,
                             (int)phase, cur_stream()));
  // segs is host memory; hipMemcpyAsync from pageable memory completes
  // synchronously w.r.t. the host, so letting it go out of scope is safe.
}

// cached variant for graph capture / persistent grads: descriptor
// built once (device tensor), replayed with one launch.
static torch::Tensor sgd_multi_make(std::vector<torch::Tensor> ws,
                                    std::vector<torch::Tensor> gs,
                                    std::vector<torch::Tensor> vs,
                                    std::vector<torch::Tensor> outs,
                                    std::vector<double> lrs,
                                    std::vector<double> wds) {
  int n = (int)ws.size();
  std::vector<SgdSegH> segs(n);
  for (int i = 0; i < n; ++i) {
    long nn = ws[i].numel();
    segs[i] = {ws[i].data_ptr<float>(), gs[i].data_ptr<float>(),
               vs[i].data_ptr<float>(),
               outs[i].numel() ? outs[i].data_ptr() : nullptr,
               nn / 4, nn, (float)lrs[i], (float)wds[i]};
  }
  auto buf = torch::empty({(long)(n * sizeof(SgdSegH))},
                          torch::TensorOptions()
                              .dtype(torch::kUInt8)
                              .device(ws[0].device()));
  hipMemcpyAsync(buf.data_ptr(), segs.data(), n * sizeof(SgdSegH),
                 hipMemcpyHostToDevice, cur_stream());
  hipStreamSynchronize(cur_stream());
  return buf;
}

static void sgd_multi_run(torch::Tensor desc, long nseg, long max_n4,
                          double mu, long phase) {
  CHECK_HIP(launch_sgd_multi(desc.data_ptr(), (int)nseg, max_n4,
                             (float)mu, (int)phase, cur_stream()));
}

// static repack set: build the descriptor buffer once...
static torch::Tensor gather_multi_make(std::vector<torch::Tensor> srcs,
                                       std::vector<torch::Tensor> maps,
                                       std::vector<torch::Tensor> dsts) {
  int n = (int)srcs.size();
  std::vector<GatherSegH> segs(n);
  for (int i = 0; i < n; ++i) {
    segs[i] = {srcs[i].data_ptr(), maps[i].data_ptr<int>(),
               dsts[i].data_ptr(), dsts[i].numel()};
  }
  auto buf = torch::empty({(long)(n * sizeof(GatherSegH))},
                          torch::TensorOptions()
                              .dtype(torch::kUInt8)
                              .device(srcs[0].device()));
  hipMemcpyAsync(buf.data_ptr(), segs.data(), n * sizeof(GatherSegH),
                 hipMemcpyHostToDevice, cur_stream());
  hipStreamSynchronize(cur_stream());
  return buf;
}

// ...then replay with one kernel launch per refresh
static void gather_multi_run(torch::Tensor desc, long nseg, long max_n) {
  CHECK_HIP(launch_gather_multi(desc.data_ptr(), (int)nseg, max_n,
                                cur_stream()));
}

static std::vector<long> transpose_multi_make(
    std::vector<torch::Tensor> srcs, std::vector<torch::Tensor> dsts,
    std::vector<long> rows, std::vector<long> cols,
    std::vector<long> sstrides, std::vector<long> dstrides,
    torch::Tensor buf) {
  // buf: caller-held uint8 tensor of n*sizeof(TransSegH); returns
  // [max_rows, max_cols] for the run call
  int n = (int)srcs.size();
  std::vector<TransSegH> segs(n);
  long mr = 1, mc = 1;
  for (int i = 0; i < n; ++i) {
    segs[i] = {srcs[i].data_ptr(), dsts[i].data_ptr(), (int)rows[i],
               (int)cols[i], sstrides[i], dstrides[i]};
    if (rows[i] > mr) mr = rows[i];
    if (cols[i] > mc) mc = cols[i];
  }
  TORCH_CHECK(buf.numel() >= (long)(n * sizeof(TransSegH)));
  hipMemcpyAsync(buf.data_ptr(), segs.data(), n * sizeof(TransSegH),
                 hipMemcpyHostToDevice, cur_stream());
  hipStreamSynchronize(cur_stream());
  return {mr, mc};
}

static void transpose_multi_run(torch::Tensor desc, long nseg,
                                long max_rows, long max_cols) {
  CHECK_HIP(launch_transpose_multi(desc.data_ptr(), (int)nseg,
                                   (int)max_rows, (int)max_cols,
                                   cur_stream()));
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("halo_pad", &halo_pad_sym);
  m.def("halo_pad_lrtb", &halo_pad);
  m.def("conv_fwd", &conv_fwd, py::arg("xp"), py::arg("w"),
        py::arg("bias"), py::arg("table"), py::arg("y"),
        py::arg("rowStride"), py::arg("colStride"), py::arg("sh"),
        py::arg("sw"), py::arg("groupInOff"), py::arg("relu"),
        py::arg("stats") = py::none(), py::arg("variant") = 0);
  m.def("conv_dgrad", &conv_dgrad, py::arg("dyp"), py::arg("wdg"),
        py::arg("table"), py::arg("dx"), py::arg("rowStride"),
        py::arg("colStride"), py::arg("Hi"), py::arg("Wj"),
        py::arg("clsI"), py::arg("clsJ"), py::arg("sh"), py::arg("sw"),
        py::arg("groupInOff"), py::arg("Cgp"), py::arg("variant") = 0);
  m.def("conv_wgrad", &conv_wgrad, py::arg("xp"), py::arg("dy"),
        py::arg("table"), py::arg("dw"), py::arg("kdfMask"),
        py::arg("rowStride"), py::arg("colStride"), py::arg("sh"),
        py::arg("sw"), py::arg("groupInOff"), py::arg("variant") = 0,
        py::arg("split_target") = 0, py::arg("db") = py::none());
  m.def("bias_wgrad", &bias_wgrad);
  m.def("relu_fwd", &relu_fwd);
  m.def("relu_bwd", &relu_bwd);
  m.def("relu_bwd_dual", &relu_bwd_dual);
  m.def("add_relu", &add_relu);
  m.def("dropout_fwd", &dropout_fwd);
  m.def("dropout_bwd", &dropout_bwd);
  m.def("gather_bf16", &gather_bf16);
  m.def("transpose_bf16", &transpose_bf16);
  m.def("u8_finish", &u8_finish, py::arg("src"), py::arg("nLogC"),
        py::arg("meanimg") = py::none());
  m.def("pack_bf16", &pack_bf16);
  m.def("unpack_bf16", &unpack_bf16);
  m.def("segsum", &segsum);
  m.def("maxpool_fwd", &maxpool_fwd);
  m.def("maxpool_bwd", &maxpool_bwd);
  m.def("avgpool_fwd", &avgpool_fwd);
  m.def("avgpool_bwd", &avgpool_bwd);
  m.def("lrn_fwd", &lrn_fwd);
  m.def("lrn_bwd", &lrn_bwd);
  m.def("bn_fwd", &bn_fwd, py::arg("x"), py::arg("gamma"),
        py::arg("beta"), py::arg("y"), py::arg("save_mean"),
        py::arg("save_invstd"), py::arg("rmean"), py::arg("rvar"),
        py::arg("momentum"), py::arg("eps"), py::arg("training"),
        py::arg("relu"), py::arg("prestats") = py::none(),
        py::arg("res") = py::none());
  m.def("bn_bwd", &bn_bwd, py::arg("dy"), py::arg("x"),
        py::arg("gamma"), py::arg("beta"), py::arg("save_mean"),
        py::arg("save_invstd"), py::arg("y"), py::arg("dx"),
        py::arg("dgamma"), py::arg("dbeta"), py::arg("relu"),
        py::arg("dres") = py::none(), py::arg("has_res") = false);
  m.def("softmax_xent_fwd", &softmax_xent_fwd, py::arg("logits"),
        py::arg("labels"), py::arg("probs"), py::arg("loss"),
        py::arg("nCls"), py::arg("stats") = py::none());
  m.def("softmax_xent_bwd", &softmax_xent_bwd);
  m.def("sgd_step", &sgd_step);
  m.def("mfma_layout", &mfma_layout);
  m.def("tr_probe", &tr_probe);
  m.def("sgd_step_multi", &sgd_step_multi);
  m.def("sgd_multi_make", &sgd_multi_make);
  m.def("sgd_multi_run", &sgd_multi_run);
  m.def("gather_multi_make", &gather_multi_make);
  m.def("gather_multi_run", &gather_multi_run);
  m.def("transpose_multi_make", &transpose_multi_make);
  m.def("transpose_multi_run", &transpose_multi_run);
}
