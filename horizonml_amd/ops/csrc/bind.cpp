// Torch bindings for the horizonml_amd gfx950 kernel set.
// Compiled host-side (g++); all device code lives in *.hip translation units
// linked as extra objects (no hipify, no CUDA compat).
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <c10/hip/HIPCachingAllocator.h>

#include <memory>
#include <vector>

struct ConvP {
  int Nb, H, W, C, K;
  int Ho, Wo, R, S;
  int str, pad;
  int M, Kd;
};

extern "C" {
int conv_fwd_splitk(ConvP);
int conv_dgrad_splitk(ConvP);
void launch_conv_fwd(const void*, const void*, void*, float*, ConvP,
                     hipStream_t);
void launch_conv_fwd_splitk(const void*, const void*, float*, ConvP, int,
                            hipStream_t);
void launch_conv_dgrad(const void*, const void*, void*, float*, int, ConvP,
                       int, hipStream_t);
void launch_gemm_bf16(const void*, const void*, void*, int, int, int,
                      hipStream_t);
int wgrad_msplit(ConvP);
void launch_wgrad(const void*, const void*, float*, float*, ConvP,
                  hipStream_t);
void launch_wgrad_batched_t(const void*, int, int, int, hipStream_t);
void launch_wgrad_reduce_batched(const void*, int, hipStream_t);
void launch_bn_apply(const void*, const void*, void*, const float*,
                     const float*, const float*, float*, float*, float*,
                     float*, long, int, float, float, int, int, hipStream_t);
void launch_bn_apply_f32(const float*, const void*, void*, void*,
                         const float*, const float*, const float*, float*,
                         float*, float*, float*, long, int, float, float,
                         int, int, int, hipStream_t);
void launch_stats_reduce(const float*, float*, long, int, int, hipStream_t);
void launch_cast_f32_bf16(const float*, void*, long, int, int, hipStream_t);
void set_kernels_deterministic(int);
void launch_stats_bf16_det(const void*, float*, long, int, hipStream_t);
void launch_cast_bnact(const float*, void*, long, int, int, int, const void*,
                       const void*, const float*, const float*, const float*,
                       const float*, float*, float*, int, hipStream_t);
void launch_bnact_bwd_reduce(const void*, const void*, const void*,
                             const float*, const float*, const float*,
                             const float*, float*, float*, long, int, int,
                             hipStream_t);
void launch_bn_bwd_apply(const void*, const void*, const void*, const float*,
                         const float*, const float*, const float*,
                         const float*, const float*, void*, void*, long, int,
                         int, hipStream_t);
void launch_dw_fwd(const void*, const void*, void*, float*, int, int, int,
                   int, int, int, int, int, int, int, hipStream_t);
void launch_dw_dgrad(const void*, const void*, void*, int, int, int, int,
                     int, int, int, int, int, int, hipStream_t);
void launch_dw_wgrad(const void*, const void*, float*, int, int, int, int,
                     int, int, int, int, int, int, hipStream_t);
void launch_maxpool_fwd(const void*, void*, unsigned char*, int, int, int,
                        int, int, int, hipStream_t);
void launch_maxpool_bwd(const void*, const unsigned char*, void*, int, int,
                        int, int, int, int, hipStream_t);
void launch_avgpool_fwd(const void*, void*, int, int, int, hipStream_t);
void launch_avgpool_bwd(const void*, void*, int, int, int, hipStream_t);
void launch_linear_fwd(const void*, const void*, const float*, float*, int,
                       int, int, hipStream_t);
void launch_linear_bwd(const float*, const void*, const void*, void*, float*,
                       float*, int, int, int, int, hipStream_t);
void launch_ce_fwd_bwd(const float*, const long*, float*, float*, int, int,
                       hipStream_t);
void launch_adam_step(float*, float*, float*, float*, void*, const float*,
                      long, float, float, float, float, float, int, float*,
                      long, const void*, float, float*, float*, float*,
                      hipStream_t);
void launch_sgd_step(float*, float*, float*, void*, long, float, float, float,
                     int, const void*, float, float*, float*, float*,
                     hipStream_t);
void launch_permute_krsc_rsck(const void*, void*, const int*, int, int,
                              hipStream_t);
void launch_grad_divergence(const float*, float*, float*, float*, long, int,
                            hipStream_t);
void launch_normalize_u8(const void*, void*, long, int, long, float, float,
                         hipStream_t);
}

namespace {

using torch::Tensor;

hipStream_t cur_stream() { return c10::hip::getCurrentHIPStream().stream(); }

// Deterministic mode: fixed-order reductions replace every atomic-order
// dependence (BN batch stats, backward channel sums, CE loss scalar).
// Measured: bitwise run-to-run reproducible at ~9x step time (bs=64) —
// the single-block serial reductions dominate at small batches; a
// reproducibility/debugging tool.  Weight gradients are already
// deterministic (atomic-free slab reduction in a fixed z order).
bool g_deterministic = false;
void set_deterministic(bool on) {
  g_deterministic = on;
  set_kernels_deterministic(on ? 1 : 0);
}
bool deterministic_enabled() { return g_deterministic; }

// ---- deferred (batched) wgrad ---------------------------------------------
// The weight-gradient GEMMs are off the backward's critical chain (nothing
// reads dW until the optimizer step), and each conv's own wgrad underfills
// the 256 CUs at CIFAR shapes.  In direct-grad mode with defer enabled
// (FlatParamManager), conv backward only records the task; ``flush_wgrad``
// (called by the fused optimizer / before the DP all-reduce) then runs ALL
// pending weight gradients as one batched kernel + one batched slab-reduce —
// 2 launches instead of ~40, full chip occupancy, no atomics.  Task tables
// travel as kernel arguments by value, so the path is hipGraph-capture-safe
// (no host staging buffers whose contents could change between replays).
constexpr int WG_MAX_TASKS = 20;

struct MagicP {
  unsigned m[4];
  int s[4];
  unsigned d[4];
};

static void magic_u32_h(unsigned d, unsigned* m, int* sh) {
  if (d <= 1) { *m = 0; *sh = 0; return; }
  int p = 0;
  while ((1ull << p) < d) p++;
  *m = (unsigned)((((unsigned long long)((1ull << p) - d) << 32) / d) + 1);
  *sh = p;
}

static MagicP make_magic_mode1(const ConvP& p) {
  MagicP mg{};
  unsigned d[4] = {(unsigned)(p.Ho * p.Wo), (unsigned)p.Wo,
                   (unsigned)(p.S * p.C), (unsigned)p.C};
  for (int i = 0; i < 4; i++) {
    mg.d[i] = d[i] ? d[i] : 1;
    magic_u32_h(mg.d[i], &mg.m[i], &mg.s[i]);
  }
  return mg;
}

struct WgradTask {
  const void* X;
  const void* Dz;
  float* out;
  ConvP p;
  MagicP mg;
  int Ntot, mchunk, msplit, tx, ty;
  int base;
  int vec;
};

struct WgradBatchArgs {
  int n;
  WgradTask t[WG_MAX_TASKS];
};

struct WredTask {
  const float* ws;
  float* dW;
  long n;
  int msplit;
  int base;
};

struct WredBatchArgs {
  int n;
  WredTask t[WG_MAX_TASKS];
};

struct PendingWgrad {
  torch::Tensor x, dconv, dw;
  ConvP p;
};

bool g_wgrad_defer = false;
std::vector<PendingWgrad> g_pending;
torch::Tensor g_wgrad_ws;  // persistent slab workspace (grows on demand)

void set_wgrad_defer(bool on) { g_wgrad_defer = on; }
bool wgrad_defer_enabled() { return g_wgrad_defer; }
int64_t wgrad_pending() { return (int64_t)g_pending.size(); }

static inline int cdiv_i(long a, long b) { return (int)((a + b - 1) / b); }

void flush_one_group(int lo, int hi);

// Tasks writing the SAME grad tensor (gradient accumulation: several
// backwards before one flush) must not RMW concurrently — partition the
// pending list into maximal runs of unique dW targets and flush each run
// as its own batched launch pair (runs are stream-ordered).
void flush_wgrad() {
  if (g_pending.empty()) return;
  const int total = (int)g_pending.size();
  int lo = 0;
  while (lo < total) {
    std::vector<const void*> seen;
    int hi = lo;
    for (; hi < total; hi++) {
      const void* ptr = g_pending[hi].dw.data_ptr();
      bool dup = false;
      for (const void* q : seen)
        if (q == ptr) { dup = true; break; }
      if (dup) break;
      seen.push_back(ptr);
    }
    flush_one_group(lo, hi);
    lo = hi;
  }
  g_pending.clear();
}

// Flush only the pending wgrads whose dW target lies inside the flat
// gradient range [lo_elem, hi_elem) of `flat` (the FlatParamManager grad
// buffer) — the per-bucket flush of the overlapped DP all-reduce
// (parallel/flat_reducer.py).  Entries outside the range stay pending for
// their own bucket.  Reuses flush_wgrad's dup-partitioned batched launch
// by swapping the matching run into g_pending.
void flush_wgrad_range(torch::Tensor flat, int64_t lo_elem, int64_t hi_elem) {
  if (g_pending.empty()) return;
  const char* base = (const char*)flat.data_ptr();
  const char* plo = base + lo_elem * flat.element_size();
  const char* phi = base + hi_elem * flat.element_size();
  std::vector<PendingWgrad> take, keep;
  for (auto& p : g_pending) {
    const char* q = (const char*)p.dw.data_ptr();
    if (q >= plo && q < phi)
      take.push_back(std::move(p));
    else
      keep.push_back(std::move(p));
  }
  g_pending = std::move(take);
  flush_wgrad();  // clears g_pending
  g_pending = std::move(keep);
}

void flush_one_group(int group_lo, int group_hi) {
  auto st = cur_stream();
  const int n = group_hi - group_lo;
  const PendingWgrad* pend = g_pending.data() + group_lo;
  // split choice: mchunk ≈ 512 rows keeps per-block m-loops short while the
  // batched grid (all tasks together) fills the chip
  std::vector<int> msplit(n), mchunk(n);
  std::vector<long> ws_off(n, 0);
  long ws_total = 0;
  for (int i = 0; i < n; i++) {
    const ConvP& p = pend[i].p;
    // large-M tasks take 2048-row chunks: the batched grid pools every
    // task's blocks so per-task fill matters less than slab volume — at
    // mchunk 512 a ResNet50 layer1 conv makes 196 f32 slabs whose
    // write+reduce traffic dominated wgrad_reduce (r02 traces)
    const int mc_base = p.M >= 32768 ? 2048 : 512;
    int ms = std::max(1, cdiv_i(p.M, mc_base));
    int mc = cdiv_i(cdiv_i(p.M, ms), 32) * 32;
    ms = cdiv_i(p.M, mc);
    msplit[i] = ms;
    mchunk[i] = mc;
    if (ms > 1) {
      ws_off[i] = ws_total;
      ws_total += (long)ms * p.K * p.Kd;
    }
  }
  if (ws_total > 0 &&
      (!g_wgrad_ws.defined() || g_wgrad_ws.numel() < ws_total))
    g_wgrad_ws = at::empty({ws_total},
                           pend[0].x.options().dtype(torch::kFloat32));
  float* ws_base = ws_total > 0 ? g_wgrad_ws.data_ptr<float>() : nullptr;

  // four tile classes (tk3, tko ∈ {64,128}): 128-wide k3 for big-Kd
  // large-M tasks halves the Dz re-read; 128-wide ko for K>=128 tasks
  // halves the X re-read — separate launches keep per-class LDS/occupancy
  auto tk3_of = [](const ConvP& p) {
    static int kd_min = [] {
      const char* e = getenv("HZ_WG_TK3_KD");
      return e ? atoi(e) : 256;  // swept on MI355X (r50 224 + bs1024)
    }();
    static int m_min = [] {
      const char* e = getenv("HZ_WG_TK3_M");
      // swept on MI355X: M=2048 admits CIFAR-bs64 layer1 (M=4096) and
      // costs ~2% at the parity point; 5000 keeps r50@224 at 9.39 ms
      return e ? atoi(e) : 5000;
    }();
    return (p.Kd >= kd_min && p.M >= m_min) ? 128 : 64;
  };
  auto tko_of = [](const ConvP& p) {
    static int k_min = [] {
      const char* e = getenv("HZ_WG_TKO_K");
      return e ? atoi(e) : 128;
    }();
    return p.K >= k_min ? 128 : 64;
  };
  for (int cls = 0; cls < 4; cls++) {
    const int tk3 = (cls & 1) ? 128 : 64;
    const int tko = (cls & 2) ? 128 : 64;
    std::vector<int> idx;
    for (int i = 0; i < n; i++)
      if (tk3_of(pend[i].p) == tk3 && tko_of(pend[i].p) == tko)
        idx.push_back(i);
    for (size_t lo = 0; lo < idx.size(); lo += WG_MAX_TASKS) {
      WgradBatchArgs a{};
      a.n = (int)std::min((size_t)WG_MAX_TASKS, idx.size() - lo);
      int blocks = 0;
      for (int j = 0; j < a.n; j++) {
        const int i = idx[lo + j];
        const PendingWgrad& pw = pend[i];
        WgradTask& t = a.t[j];
        t.X = pw.x.data_ptr();
        t.Dz = pw.dconv.data_ptr();
        t.out = msplit[i] > 1 ? ws_base + ws_off[i]
                              : pw.dw.data_ptr<float>();
        t.p = pw.p;
        t.mg = make_magic_mode1(pw.p);
        t.Ntot = pw.p.K;
        t.mchunk = mchunk[i];
        t.msplit = msplit[i];
        t.tx = cdiv_i(pw.p.Kd, tk3);
        t.ty = cdiv_i(pw.p.K, tko);
        t.base = blocks;
        t.vec = (pw.p.C % 8) == 0;
        blocks += t.tx * t.ty * t.msplit;
      }
      launch_wgrad_batched_t(&a, blocks, tk3, tko, st);
    }
  }

  // batched slab reduce for the split tasks
  WredBatchArgs r{};
  r.n = 0;
  int rblocks = 0;
  auto flush_red = [&]() {
    if (r.n > 0) launch_wgrad_reduce_batched(&r, rblocks, st);
    r.n = 0;
    rblocks = 0;
  };
  for (int i = 0; i < n; i++) {
    if (msplit[i] <= 1) continue;
    if (r.n == WG_MAX_TASKS) flush_red();
    WredTask& t = r.t[r.n++];
    t.ws = ws_base + ws_off[i];
    t.dW = pend[i].dw.data_ptr<float>();
    t.n = (long)pend[i].p.K * pend[i].p.Kd;
    t.msplit = msplit[i];
    t.base = rblocks;
    rblocks += cdiv_i(t.n, 1024);
  }
  flush_red();
}

void check_cl(const Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.dim() == 4, name, " must be 4-D");
  bool cl = t.is_contiguous(at::MemoryFormat::ChannelsLast);
  bool hw1 = t.size(2) == 1 && t.size(3) == 1 && t.is_contiguous();
  TORCH_CHECK(cl || hw1, name, " must be channels_last contiguous");
}

void check_f32(const Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kFloat32 &&
                  t.is_contiguous(),
              name, " must be contiguous f32 on GPU");
}

Tensor empty_cl_bf16(int64_t n, int64_t c, int64_t h, int64_t w,
                     const Tensor& like) {
  return at::empty({n, c, h, w},
                   like.options().dtype(torch::kBFloat16),
                   at::MemoryFormat::ChannelsLast);
}

ConvP make_convp(const Tensor& x, int K, int R, int S, int stride, int pad) {
  ConvP p{};
  p.Nb = (int)x.size(0);
  p.C = (int)x.size(1);
  p.H = (int)x.size(2);
  p.W = (int)x.size(3);
  p.K = K;
  p.R = R;
  p.S = S;
  p.str = stride;
  p.pad = pad;
  p.Ho = (p.H + 2 * pad - R) / stride + 1;
  p.Wo = (p.W + 2 * pad - S) / stride + 1;
  p.M = p.Nb * p.Ho * p.Wo;
  p.Kd = R * S * p.C;
  return p;
}

// ---------------------------------------------------------------- conv+BN --
std::vector<Tensor> conv_bn_act_fwd(
    Tensor x, Tensor w, Tensor gamma, Tensor beta, Tensor running_mean,
    Tensor running_var, int64_t stride, int64_t pad, double momentum,
    double eps, bool training, int64_t act,
    c10::optional<Tensor> residual, c10::optional<Tensor> stats_buf) {
  check_cl(x, "x");
  TORCH_CHECK(w.is_cuda() && w.scalar_type() == torch::kBFloat16 &&
                  w.is_contiguous() && w.dim() == 4,
              "w must be contiguous bf16 KRSC");
  check_f32(gamma, "gamma");
  check_f32(beta, "beta");
  int K = (int)w.size(0), R = (int)w.size(1), S = (int)w.size(2);
  TORCH_CHECK(w.size(3) == x.size(1), "w in-channels mismatch");
  ConvP p = make_convp(x, K, R, S, (int)stride, (int)pad);

  Tensor convout = empty_cl_bf16(p.Nb, K, p.Ho, p.Wo, x);
  Tensor y = empty_cl_bf16(p.Nb, K, p.Ho, p.Wo, x);
  auto fopt = x.options().dtype(torch::kFloat32);
  Tensor stats, smean, sinvstd;
  float* stats_ptr = nullptr;
  if (training) {
    // stats_buf (manager arena) arrives pre-zeroed by the fused optimizer
    stats = stats_buf.has_value() ? *stats_buf : at::zeros({2L * K}, fopt);
    smean = at::empty({K}, fopt);
    sinvstd = at::empty({K}, fopt);
    stats_ptr = stats.data_ptr<float>();
  } else {
    smean = at::empty({0}, fopt);
    sinvstd = at::empty({0}, fopt);
  }
  const void* res_ptr = nullptr;
  if (residual.has_value()) {
    check_cl(*residual, "residual");
    res_ptr = residual->data_ptr();
  }
  auto st = cur_stream();
  int splitk = conv_fwd_splitk(p);
  if (splitk > 1) {
    int kchunk = ((p.Kd + splitk - 1) / splitk + 31) / 32 * 32;
    splitk = (p.Kd + kchunk - 1) / kchunk;
    // latency path: per-split f32 slabs -> stats reduce -> BN apply (+cast);
    // slabs are fully written by the conv grid, so no zeroing needed
    Tensor ws = at::empty({(long)splitk * p.M * K}, fopt);
    launch_conv_fwd_splitk(x.data_ptr(), w.data_ptr(), ws.data_ptr<float>(),
                           p, splitk, st);
    if (training)
      launch_stats_reduce(ws.data_ptr<float>(), stats_ptr, (long)p.M, K,
                          splitk, st);
    launch_bn_apply_f32(ws.data_ptr<float>(), res_ptr, y.data_ptr(),
                        convout.data_ptr(), stats_ptr,
                        gamma.data_ptr<float>(), beta.data_ptr<float>(),
                        running_mean.data_ptr<float>(),
                        running_var.data_ptr<float>(),
                        training ? smean.data_ptr<float>() : nullptr,
                        training ? sinvstd.data_ptr<float>() : nullptr,
                        (long)p.M, K, (float)momentum, (float)eps,
                        training ? 1 : 0, (int)act, splitk, st);
  } else {
    launch_conv_fwd(x.data_ptr(), w.data_ptr(), convout.data_ptr(),
                    g_deterministic ? nullptr : stats_ptr, p, st);
    if (g_deterministic && training)
      launch_stats_bf16_det(convout.data_ptr(), stats_ptr, (long)p.M, K,
                            st);
    launch_bn_apply(convout.data_ptr(), res_ptr, y.data_ptr(), stats_ptr,
                    gamma.data_ptr<float>(), beta.data_ptr<float>(),
                    running_mean.data_ptr<float>(),
                    running_var.data_ptr<float>(),
                    training ? smean.data_ptr<float>() : nullptr,
                    training ? sinvstd.data_ptr<float>() : nullptr,
                    (long)p.M, K, (float)momentum, (float)eps,
                    training ? 1 : 0, (int)act, st);
  }
  return {y, convout, smean, sinvstd};
}

std::vector<Tensor> conv_bn_act_bwd(
    Tensor dy, Tensor y, Tensor x, Tensor w, Tensor w_rsck, Tensor convout,
    Tensor gamma, Tensor beta, Tensor save_mean, Tensor save_invstd,
    int64_t stride, int64_t pad, int64_t act, bool need_dx, bool has_res,
    c10::optional<Tensor> dw_out, c10::optional<Tensor> dgamma_out,
    c10::optional<Tensor> dbeta_out, c10::optional<Tensor> dx_accum,
    c10::optional<std::vector<Tensor>> fuse_up, int64_t up_mask_mode,
    bool sums_ready, c10::optional<Tensor> sum_dz_in,
    c10::optional<Tensor> sum_dzx_in) {
  // fuse_up = [x_up, y_up, smean_up, sinvstd_up, gamma_up, beta_up,
  //            sum_dz_up, sum_dzx_up]: while producing dx (= the UPSTREAM
  //            conv's dy), also complete that conv's BN-backward channel
  //            sums — fused into the slab-sum cast when dgrad is split-K,
  //            else run as the standalone reduce right after dgrad.  The
  //            upstream conv's own backward then passes sums_ready=true
  //            (with sum_dz_in/sum_dzx_in in non-direct-grad mode) and
  //            skips its reduce pass.
  // Direct-grad mode: when dw_out/dgamma_out/dbeta_out are given they are
  // PRE-ZEROED flat .grad views — the kernels accumulate straight into
  // them, skipping autograd's per-parameter accumulate pass.
  check_cl(dy, "dy");
  check_cl(y, "y");
  check_cl(x, "x");
  check_cl(convout, "convout");
  int K = (int)w.size(0), R = (int)w.size(1), S = (int)w.size(2);
  ConvP p = make_convp(x, K, R, S, (int)stride, (int)pad);
  auto fopt = x.options().dtype(torch::kFloat32);
  auto st = cur_stream();

  bool direct = dw_out.has_value();
  // ReLU-mask source: residual epilogue needs y; otherwise the mask is
  // recovered from convout via BN algebra and y is never read
  int mask_mode = act == 1 ? (has_res ? 1 : 2)
                 : act == 2 ? (has_res ? 3 : 4) : 0;
  Tensor sum_dz, sum_dzx;
  if (sums_ready) {  // computed by the downstream conv's dx producer
    sum_dz = direct ? *dbeta_out : *sum_dz_in;
    sum_dzx = direct ? *dgamma_out : *sum_dzx_in;
  } else {
    sum_dz = direct ? *dbeta_out : at::zeros({K}, fopt);
    sum_dzx = direct ? *dgamma_out : at::zeros({K}, fopt);
    launch_bnact_bwd_reduce(dy.data_ptr(), y.data_ptr(), convout.data_ptr(),
                            save_mean.data_ptr<float>(),
                            save_invstd.data_ptr<float>(),
                            gamma.data_ptr<float>(), beta.data_ptr<float>(),
                            sum_dz.data_ptr<float>(),
                            sum_dzx.data_ptr<float>(),
                            (long)p.M, K, mask_mode, st);
  }

  Tensor dconv = empty_cl_bf16(p.Nb, K, p.Ho, p.Wo, x);
  Tensor dres;
  void* dres_ptr = nullptr;
  if (has_res) {
    dres = empty_cl_bf16(p.Nb, K, p.Ho, p.Wo, x);
    dres_ptr = dres.data_ptr();
  }
  launch_bn_bwd_apply(dy.data_ptr(), y.data_ptr(), convout.data_ptr(),
                      save_mean.data_ptr<float>(),
                      save_invstd.data_ptr<float>(), gamma.data_ptr<float>(),
                      beta.data_ptr<float>(),
                      sum_dz.data_ptr<float>(), sum_dzx.data_ptr<float>(),
                      dconv.data_ptr(), dres_ptr, (long)p.M, K, mask_mode,
                      st);

  Tensor dw = direct ? *dw_out
                     : at::zeros({(int64_t)K, R, S, (int64_t)p.C}, fopt);
  if (direct && g_wgrad_defer) {
    g_pending.push_back({x, dconv, dw, p});  // runs batched at flush_wgrad
  } else {
    int msplit = wgrad_msplit(p);
    Tensor ws;
    float* ws_ptr = nullptr;
    if (msplit > 1) {
      ws = at::empty({(long)msplit * K * p.Kd}, fopt);
      ws_ptr = ws.data_ptr<float>();
    }
    launch_wgrad(x.data_ptr(), dconv.data_ptr(), dw.data_ptr<float>(), ws_ptr,
                 p, st);
  }

  TORCH_CHECK(need_dx || !fuse_up.has_value(),
              "fuse_up requires need_dx (the fused reduce rides the dx "
              "production)");
  Tensor dx;
  if (need_dx) {
    TORCH_CHECK(w_rsck.is_contiguous() &&
                    w_rsck.scalar_type() == torch::kBFloat16,
                "w_rsck must be contiguous bf16");
    // Fused residual-junction add: with dx_accum given, dgrad ACCUMULATES
    // into it (dx_total = dx + residual-path grad without a separate
    // elementwise add kernel).
    int accum = dx_accum.has_value() ? 1 : 0;
    dx = accum ? *dx_accum : empty_cl_bf16(p.Nb, p.C, p.H, p.W, x);
    ConvP pd = p;
    pd.M = p.Nb * p.H * p.W;
    pd.Kd = R * S * K;
    int splitk = conv_dgrad_splitk(p);
    const bool fuse = fuse_up.has_value();
    if (splitk > 1) {
      int kchunk = ((pd.Kd + splitk - 1) / splitk + 31) / 32 * 32;
      splitk = (pd.Kd + kchunk - 1) / kchunk;
      Tensor wsd = at::empty({(long)splitk * pd.M * p.C}, fopt);
      launch_conv_dgrad(dconv.data_ptr(), w_rsck.data_ptr(), nullptr,
                        wsd.data_ptr<float>(), splitk, pd, 0, st);
      if (fuse) {
        auto& f = *fuse_up;
        launch_cast_bnact(wsd.data_ptr<float>(), dx.data_ptr(), (long)pd.M,
                          p.C, splitk, accum, f[0].data_ptr(),
                          f[1].defined() ? f[1].data_ptr() : f[0].data_ptr(),
                          f[2].data_ptr<float>(), f[3].data_ptr<float>(),
                          f[4].data_ptr<float>(), f[5].data_ptr<float>(),
                          f[6].data_ptr<float>(), f[7].data_ptr<float>(),
                          (int)up_mask_mode, st);
      } else {
        launch_cast_f32_bf16(wsd.data_ptr<float>(), dx.data_ptr(),
                             (long)pd.M * p.C, splitk, accum, st);
      }
    } else {
      launch_conv_dgrad(dconv.data_ptr(), w_rsck.data_ptr(), dx.data_ptr(),
                        nullptr, 1, pd, accum, st);
      if (fuse) {
        auto& f = *fuse_up;
        launch_bnact_bwd_reduce(dx.data_ptr(), f[1].defined()
                                    ? f[1].data_ptr() : f[0].data_ptr(),
                                f[0].data_ptr(), f[2].data_ptr<float>(),
                                f[3].data_ptr<float>(),
                                f[4].data_ptr<float>(),
                                f[5].data_ptr<float>(),
                                f[6].data_ptr<float>(),
                                f[7].data_ptr<float>(), (long)pd.M, p.C,
                                (int)up_mask_mode, st);
      }
    }
  }
  // dgamma = Σ dz·xhat, dbeta = Σ dz
  return {dx, dw, sum_dzx, sum_dz, dres};
}

// --------------------------------------------------------- depthwise conv --
std::vector<Tensor> dw_conv_bn_fwd(
    Tensor x, Tensor w, Tensor gamma, Tensor beta, Tensor running_mean,
    Tensor running_var, int64_t stride, int64_t pad, double momentum,
    double eps, bool training, int64_t act, c10::optional<Tensor> residual) {
  check_cl(x, "x");
  TORCH_CHECK(w.is_cuda() && w.scalar_type() == torch::kBFloat16 &&
                  w.is_contiguous() && w.dim() == 3,
              "w must be contiguous bf16 [R,S,C]");
  int Nb = (int)x.size(0), C = (int)x.size(1), H = (int)x.size(2),
      W = (int)x.size(3);
  int R = (int)w.size(0), S = (int)w.size(1);
  TORCH_CHECK((int)w.size(2) == C, "depthwise channel mismatch");
  int Ho = (H + 2 * (int)pad - R) / (int)stride + 1;
  int Wo = (W + 2 * (int)pad - S) / (int)stride + 1;
  long M = (long)Nb * Ho * Wo;
  auto fopt = x.options().dtype(torch::kFloat32);
  Tensor convout = empty_cl_bf16(Nb, C, Ho, Wo, x);
  Tensor y = empty_cl_bf16(Nb, C, Ho, Wo, x);
  Tensor stats, smean, sinvstd;
  float* stats_ptr = nullptr;
  if (training) {
    stats = at::zeros({2L * C}, fopt);
    smean = at::empty({C}, fopt);
    sinvstd = at::empty({C}, fopt);
    stats_ptr = stats.data_ptr<float>();
  } else {
    smean = at::empty({0}, fopt);
    sinvstd = at::empty({0}, fopt);
  }
  const void* res_ptr = nullptr;
  if (residual.has_value()) {
    check_cl(*residual, "residual");
    res_ptr = residual->data_ptr();
  }
  auto st = cur_stream();
  launch_dw_fwd(x.data_ptr(), w.data_ptr(), convout.data_ptr(),
                g_deterministic ? nullptr : stats_ptr, Nb, H, W, C, Ho, Wo,
                R, S, (int)stride, (int)pad, st);
  if (g_deterministic && training)
    launch_stats_bf16_det(convout.data_ptr(), stats_ptr, M, C, st);
  launch_bn_apply(convout.data_ptr(), res_ptr, y.data_ptr(), stats_ptr,
                  gamma.data_ptr<float>(), beta.data_ptr<float>(),
                  running_mean.data_ptr<float>(),
                  running_var.data_ptr<float>(),
                  training ? smean.data_ptr<float>() : nullptr,
                  training ? sinvstd.data_ptr<float>() : nullptr, M, C,
                  (float)momentum, (float)eps, training ? 1 : 0, (int)act,
                  st);
  return {y, convout, smean, sinvstd};
}

std::vector<Tensor> dw_conv_bn_bwd(
    Tensor dy, Tensor y, Tensor x, Tensor w, Tensor convout, Tensor gamma,
    Tensor beta, Tensor save_mean, Tensor save_invstd, int64_t stride,
    int64_t pad, int64_t act, bool need_dx, bool has_res) {
  check_cl(dy, "dy");
  check_cl(x, "x");
  int Nb = (int)x.size(0), C = (int)x.size(1), H = (int)x.size(2),
      W = (int)x.size(3);
  int R = (int)w.size(0), S = (int)w.size(1);
  int Ho = (int)dy.size(2), Wo = (int)dy.size(3);
  long M = (long)Nb * Ho * Wo;
  auto fopt = x.options().dtype(torch::kFloat32);
  auto st = cur_stream();
  int mask_mode = act == 1 ? (has_res ? 1 : 2)
                 : act == 2 ? (has_res ? 3 : 4) : 0;
  Tensor sum_dz = at::zeros({C}, fopt);
  Tensor sum_dzx = at::zeros({C}, fopt);
  launch_bnact_bwd_reduce(dy.data_ptr(), y.data_ptr(), convout.data_ptr(),
                          save_mean.data_ptr<float>(),
                          save_invstd.data_ptr<float>(),
                          gamma.data_ptr<float>(), beta.data_ptr<float>(),
                          sum_dz.data_ptr<float>(), sum_dzx.data_ptr<float>(),
                          M, C, mask_mode, st);
  Tensor dconv = empty_cl_bf16(Nb, C, Ho, Wo, x);
  Tensor dres;
  void* dres_ptr = nullptr;
  if (has_res) {
    dres = empty_cl_bf16(Nb, C, Ho, Wo, x);
    dres_ptr = dres.data_ptr();
  }
  launch_bn_bwd_apply(dy.data_ptr(), y.data_ptr(), convout.data_ptr(),
                      save_mean.data_ptr<float>(),
                      save_invstd.data_ptr<float>(), gamma.data_ptr<float>(),
                      beta.data_ptr<float>(), sum_dz.data_ptr<float>(),
                      sum_dzx.data_ptr<float>(), dconv.data_ptr(), dres_ptr,
                      M, C, mask_mode, st);
  Tensor dw = at::zeros({(int64_t)R, (int64_t)S, (int64_t)C}, fopt);
  launch_dw_wgrad(x.data_ptr(), dconv.data_ptr(), dw.data_ptr<float>(), Nb,
                  H, W, C, Ho, Wo, R, S, (int)stride, (int)pad, st);
  Tensor dx;
  if (need_dx) {
    dx = empty_cl_bf16(Nb, C, H, W, x);
    launch_dw_dgrad(dconv.data_ptr(), w.data_ptr(), dx.data_ptr(), Nb, H, W,
                    C, Ho, Wo, R, S, (int)stride, (int)pad, st);
  }
  return {dx, dw, sum_dzx, sum_dz, dres};
}

// ----------------------------------------------------------------- pools --
std::vector<Tensor> maxpool_fwd(Tensor x) {
  check_cl(x, "x");
  int Nb = (int)x.size(0), C = (int)x.size(1), H = (int)x.size(2),
      W = (int)x.size(3);
  int Hp = (H + 2 - 3) / 2 + 1, Wp = (W + 2 - 3) / 2 + 1;
  Tensor y = empty_cl_bf16(Nb, C, Hp, Wp, x);
  Tensor idx = at::empty({Nb, Hp, Wp, C}, x.options().dtype(torch::kUInt8));
  launch_maxpool_fwd(x.data_ptr(), y.data_ptr(), idx.data_ptr<unsigned char>(),
                     Nb, H, W, C, Hp, Wp, cur_stream());
  return {y, idx};
}

Tensor maxpool_bwd(Tensor dy, Tensor idx, int64_t H, int64_t W) {
  check_cl(dy, "dy");
  int Nb = (int)dy.size(0), C = (int)dy.size(1), Hp = (int)dy.size(2),
      Wp = (int)dy.size(3);
  Tensor dx = empty_cl_bf16(Nb, C, H, W, dy);
  launch_maxpool_bwd(dy.data_ptr(), idx.data_ptr<unsigned char>(),
                     dx.data_ptr(), Nb, (int)H, (int)W, C, Hp, Wp,
                     cur_stream());
  return dx;
}

Tensor avgpool_fwd(Tensor x) {
  check_cl(x, "x");
  int Nb = (int)x.size(0), C = (int)x.size(1);
  int HW = (int)(x.size(2) * x.size(3));
  Tensor y = at::empty({Nb, C}, x.options());
  launch_avgpool_fwd(x.data_ptr(), y.data_ptr(), Nb, HW, C, cur_stream());
  return y;
}

Tensor avgpool_bwd(Tensor dy, int64_t H, int64_t W) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() &&
              dy.scalar_type() == torch::kBFloat16);
  int Nb = (int)dy.size(0), C = (int)dy.size(1);
  Tensor dx = empty_cl_bf16(Nb, C, H, W, dy);
  launch_avgpool_bwd(dy.data_ptr(), dx.data_ptr(), Nb, (int)(H * W), C,
                     cur_stream());
  return dx;
}

// -------------------------------------------------------------- classifier --
Tensor linear_fwd(Tensor x, Tensor w, c10::optional<Tensor> bias) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() &&
              x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(w.is_contiguous() && w.scalar_type() == torch::kBFloat16);
  int B = (int)x.size(0), In = (int)x.size(1), Out = (int)w.size(0);
  Tensor y = at::empty({B, Out}, x.options().dtype(torch::kFloat32));
  const float* bptr = nullptr;
  if (bias.has_value()) {
    check_f32(*bias, "bias");
    bptr = bias->data_ptr<float>();
  }
  launch_linear_fwd(x.data_ptr(), w.data_ptr(), bptr, y.data_ptr<float>(), B,
                    In, Out, cur_stream());
  return y;
}

std::vector<Tensor> linear_bwd(Tensor dy, Tensor x, Tensor w, bool need_dx,
                               bool need_db, c10::optional<Tensor> dw_out,
                               c10::optional<Tensor> db_out) {
  check_f32(dy, "dy");
  int B = (int)x.size(0), In = (int)x.size(1), Out = (int)w.size(0);
  auto fopt = x.options().dtype(torch::kFloat32);
  bool direct = dw_out.has_value();
  Tensor dx, dw = direct ? *dw_out : at::empty({Out, In}, fopt);
  Tensor db = direct ? (db_out.has_value() ? *db_out : Tensor())
                     : (need_db ? at::empty({Out}, fopt) : Tensor());
  void* dx_ptr = nullptr;
  if (need_dx) {
    dx = at::empty({B, In}, x.options());
    dx_ptr = dx.data_ptr();
  }
  launch_linear_bwd(dy.data_ptr<float>(), x.data_ptr(), w.data_ptr(), dx_ptr,
                    dw.data_ptr<float>(),
                    db.defined() ? db.data_ptr<float>() : nullptr, B, In, Out,
                    direct ? 1 : 0, cur_stream());
  return {dx, dw, db};
}

std::vector<Tensor> cross_entropy_fwd_bwd(Tensor logits, Tensor target) {
  check_f32(logits, "logits");
  TORCH_CHECK(target.scalar_type() == torch::kLong && target.is_contiguous());
  int B = (int)logits.size(0), NC = (int)logits.size(1);
  Tensor loss = at::zeros({}, logits.options());
  Tensor dlogits = at::empty({B, NC}, logits.options());
  launch_ce_fwd_bwd(logits.data_ptr<float>(), target.data_ptr<long>(),
                    loss.data_ptr<float>(), dlogits.data_ptr<float>(), B, NC,
                    cur_stream());
  return {loss, dlogits};
}

// ------------------------------------------------------------------ gemm ---
Tensor gemm_bf16(Tensor a, Tensor b) {
  // C[M,N] = A[M,K] · B[N,K]^T
  TORCH_CHECK(a.is_cuda() && a.is_contiguous() &&
              a.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(b.is_contiguous() && b.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(a.size(1) == b.size(1), "K mismatch");
  int M = (int)a.size(0), K = (int)a.size(1), N = (int)b.size(0);
  Tensor c = at::empty({M, N}, a.options());
  launch_gemm_bf16(a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K,
                   cur_stream());
  return c;
}

// Debug/test entry: standalone weight-gradient (immediate or deferred).
Tensor wgrad_only(Tensor x, Tensor dz, int64_t K, int64_t R, int64_t S,
                  int64_t stride, int64_t pad, bool defer,
                  c10::optional<Tensor> dw_out) {
  check_cl(x, "x");
  check_cl(dz, "dz");
  ConvP p = make_convp(x, (int)K, (int)R, (int)S, (int)stride, (int)pad);
  Tensor dw = dw_out.has_value()
                  ? *dw_out
                  : at::zeros({K, R, S, (int64_t)p.C},
                              x.options().dtype(torch::kFloat32));
  if (defer) {
    g_pending.push_back({x, dz, dw, p});
  } else {
    int msplit = wgrad_msplit(p);
    Tensor ws;
    float* ws_ptr = nullptr;
    if (msplit > 1) {
      ws = at::empty({(long)msplit * p.K * p.Kd},
                     x.options().dtype(torch::kFloat32));
      ws_ptr = ws.data_ptr<float>();
    }
    launch_wgrad(x.data_ptr(), dz.data_ptr(), dw.data_ptr<float>(), ws_ptr,
                 p, cur_stream());
  }
  return dw;
}

// ------------------------------------------------------------- optimizers --
// probe_prev/probe_sumsq/probe_out (all-or-none): fused grad-divergence
// probe — Σ(g−prev)² accumulated in-kernel, sqrt appended to probe_out.
void adam_step(Tensor master, Tensor grad, Tensor m, Tensor v,
               c10::optional<Tensor> shadow, Tensor step_t, double lr,
               double b1, double b2, double eps, double wd, bool zero_grad,
               c10::optional<Tensor> extra_zero,
               c10::optional<Tensor> grad_bf16, double grad_scale,
               c10::optional<Tensor> probe_prev,
               c10::optional<Tensor> probe_sumsq,
               c10::optional<Tensor> probe_out) {
  check_f32(master, "master");
  TORCH_CHECK(probe_prev.has_value() == probe_sumsq.has_value()
                  && probe_prev.has_value() == probe_out.has_value(),
              "probe tensors must be passed together");
  launch_adam_step(master.data_ptr<float>(), grad.data_ptr<float>(),
                   m.data_ptr<float>(), v.data_ptr<float>(),
                   shadow.has_value() ? shadow->data_ptr() : nullptr,
                   step_t.data_ptr<float>(), master.numel(), (float)lr,
                   (float)b1, (float)b2, (float)eps, (float)wd,
                   zero_grad ? 1 : 0,
                   extra_zero.has_value() ? extra_zero->data_ptr<float>()
                                          : nullptr,
                   extra_zero.has_value() ? extra_zero->numel() : 0,
                   grad_bf16.has_value() ? grad_bf16->data_ptr() : nullptr,
                   (float)grad_scale,
                   probe_prev.has_value() ? probe_prev->data_ptr<float>()
                                          : nullptr,
                   probe_sumsq.has_value() ? probe_sumsq->data_ptr<float>()
                                           : nullptr,
                   probe_out.has_value() ? probe_out->data_ptr<float>()
                                         : nullptr,
                   cur_stream());
}

void sgd_step(Tensor master, Tensor grad, c10::optional<Tensor> mom,
              c10::optional<Tensor> shadow, double lr, double mu, double wd,
              bool zero_grad, c10::optional<Tensor> grad_bf16,
              double grad_scale, c10::optional<Tensor> probe_prev,
              c10::optional<Tensor> probe_sumsq,
              c10::optional<Tensor> probe_out) {
  check_f32(master, "master");
  TORCH_CHECK(probe_prev.has_value() == probe_sumsq.has_value()
                  && probe_prev.has_value() == probe_out.has_value(),
              "probe tensors must be passed together");
  launch_sgd_step(master.data_ptr<float>(), grad.data_ptr<float>(),
                  mom.has_value() ? mom->data_ptr<float>() : nullptr,
                  shadow.has_value() ? shadow->data_ptr() : nullptr,
                  master.numel(), (float)lr, (float)mu, (float)wd,
                  zero_grad ? 1 : 0,
                  grad_bf16.has_value() ? grad_bf16->data_ptr() : nullptr,
                  (float)grad_scale,
                  probe_prev.has_value() ? probe_prev->data_ptr<float>()
                                         : nullptr,
                  probe_sumsq.has_value() ? probe_sumsq->data_ptr<float>()
                                          : nullptr,
                  probe_out.has_value() ? probe_out->data_ptr<float>()
                                        : nullptr,
                  cur_stream());
}

void permute_krsc_rsck(Tensor src, Tensor dst, Tensor meta,
                       int64_t max_elem) {
  TORCH_CHECK(meta.scalar_type() == torch::kInt32 && meta.is_cuda());
  launch_permute_krsc_rsck(src.data_ptr(), dst.data_ptr(),
                           meta.data_ptr<int>(), (int)meta.size(0),
                           (int)max_elem, cur_stream());
}

void grad_divergence(Tensor g, Tensor prev, Tensor sumsq, Tensor out,
                     bool skip_first) {
  launch_grad_divergence(g.data_ptr<float>(), prev.data_ptr<float>(),
                         sumsq.data_ptr<float>(), out.data_ptr<float>(),
                         g.numel(), skip_first ? 1 : 0, cur_stream());
}

// uint8 NCHW batch -> normalized bf16 channels_last, one kernel (the
// engines' H2D preprocessing; reference transform (0.5,0.5,0.5) mean/std,
// data_parallel_train.py:44-47).
// Debug/tuning entry: launch the BN-backward reduce directly (variant
// pinned by HZ_BN_V8) for kernel-isolation timing sweeps.
void bn_reduce_bench(Tensor dy, Tensor y, Tensor x, Tensor mean,
                     Tensor invstd, Tensor gamma, Tensor beta, Tensor sdz,
                     Tensor sdzx, int64_t M, int64_t C, int64_t mask) {
  launch_bnact_bwd_reduce(dy.data_ptr(), y.data_ptr(), x.data_ptr(),
                          mean.data_ptr<float>(), invstd.data_ptr<float>(),
                          gamma.data_ptr<float>(), beta.data_ptr<float>(),
                          sdz.data_ptr<float>(), sdzx.data_ptr<float>(), M,
                          (int)C, (int)mask, cur_stream());
}

Tensor normalize_u8(Tensor x, double mean, double std) {
  TORCH_CHECK(x.dim() == 4 && x.dtype() == torch::kUInt8 && x.is_cuda()
                  && x.is_contiguous(),
              "normalize_u8 expects contiguous NCHW uint8 on GPU");
  const long N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  auto y = at::empty({N, C, H, W},
                     x.options().dtype(torch::kBFloat16),
                     at::MemoryFormat::ChannelsLast);
  launch_normalize_u8(x.data_ptr(), y.data_ptr(), N * C * H * W, (int)C,
                      H * W, (float)mean, (float)std, cur_stream());
  return y;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("conv_bn_act_fwd", &conv_bn_act_fwd);
  m.def("conv_bn_act_bwd", &conv_bn_act_bwd);
  m.def("dw_conv_bn_fwd", &dw_conv_bn_fwd);
  m.def("dw_conv_bn_bwd", &dw_conv_bn_bwd);
  m.def("maxpool_fwd", &maxpool_fwd);
  m.def("maxpool_bwd", &maxpool_bwd);
  m.def("avgpool_fwd", &avgpool_fwd);
  m.def("avgpool_bwd", &avgpool_bwd);
  m.def("linear_fwd", &linear_fwd);
  m.def("linear_bwd", &linear_bwd);
  m.def("cross_entropy_fwd_bwd", &cross_entropy_fwd_bwd);
  m.def("gemm_bf16", &gemm_bf16);
  m.def("adam_step", &adam_step);
  m.def("sgd_step", &sgd_step);
  m.def("permute_krsc_rsck", &permute_krsc_rsck);
  m.def("grad_divergence", &grad_divergence);
  m.def("normalize_u8", &normalize_u8);
  m.def("bn_reduce_bench", &bn_reduce_bench);
  m.def("set_deterministic", &set_deterministic);
  m.def("deterministic_enabled", &deterministic_enabled);
  m.def("set_wgrad_defer", &set_wgrad_defer);
  m.def("wgrad_defer_enabled", &wgrad_defer_enabled);
  m.def("wgrad_pending", &wgrad_pending);
  m.def("flush_wgrad", &flush_wgrad);
  m.def("flush_wgrad_range", &flush_wgrad_range);
  m.def("wgrad_only", &wgrad_only);
}
