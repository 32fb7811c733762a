// PyTorch bindings for the gfx950 kernel library (_dmnist_hip).
// Every op validates dtype/contiguity and fails loudly — no silent
// fallbacks (the GPU path must run these kernels, _C.py policy).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

#include "kernels.h"

namespace {

#define CHECK_BF16(t) TORCH_CHECK((t).scalar_type() == at::kBFloat16, #t " must be bf16")
#define CHECK_F32(t) TORCH_CHECK((t).scalar_type() == at::kFloat, #t " must be fp32")
#define CHECK_CONTIG(t) TORCH_CHECK((t).is_contiguous(), #t " must be contiguous")
#define CHECK_CUDA(t) TORCH_CHECK((t).is_cuda(), #t " must be on GPU")

const unsigned short* bf16_ptr(const torch::Tensor& t) {
  return reinterpret_cast<const unsigned short*>(t.data_ptr());
}
unsigned short* bf16_mut(torch::Tensor& t) {
  return reinterpret_cast<unsigned short*>(t.data_ptr());
}

hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

int cdiv(int a, int b) { return (a + b - 1) / b; }

int pick_splitk(int mtiles, int ntiles, int ksteps) {
  int tiles = mtiles * ntiles;
  if (tiles >= 256 || ksteps <= 1) return 1;
  int want = std::min(ksteps, std::max(1, 512 / tiles));
  return want;
}

// glds + tr16 dW staging (dw_tr.hip); DMNIST_DW_TR=0 reverts to the
// scatter-staged gemm_tile path for A/B comparison
int dw_tr_level() {
  static int v = [] {
    const char* e = getenv("DMNIST_DW_TR");
    return e ? atoi(e) : 1;
  }();
  return v;
}
bool dw_tr_enabled() { return dw_tr_level() != 0; }

// --------------------------------------------------------------------------
torch::Tensor linear_act_fwd_impl(torch::Tensor x, torch::Tensor w,
                                  torch::Tensor b, bool relu, double p_keep,
                                  int64_t seed, int64_t offset,
                                  const long* offset_dev,
                                  const c10::optional<torch::Tensor>& wT) {
  CHECK_CUDA(x); CHECK_BF16(x); CHECK_CONTIG(x);
  CHECK_BF16(w); CHECK_CONTIG(w);
  CHECK_F32(b); CHECK_CONTIG(b);
  int M = x.size(0), K = x.size(1), N = w.size(1);
  TORCH_CHECK(w.size(0) == K, "w/x shape mismatch");
  auto y = torch::empty({M, N}, x.options());
  GemmParams p{};
  p.A = bf16_ptr(x);
  p.bias = b.data_ptr<float>();
  p.C = y.data_ptr(); p.amax = nullptr;
  p.M = M; p.N = N; p.K = K;
  p.lda = K; p.ldc = N;
  p.splitk = 1;
  p.p_keep = (float)p_keep; p.seed = (uint64_t)seed; p.offset = (uint64_t)offset;
  p.offset_dev = offset_dev;
  bool drop = p_keep < 1.0;
  TORCH_CHECK(!drop || relu, "dropout path requires relu epilogue");
  // 128x128 tiles amortize staging best, but a 128^2 grid below ~2 blocks/CU
  // is occupancy-starved (fc1 fwd @8192: 256 blocks = 1/CU, MFMA 6.9%);
  // drop to 64x64 when the 128-grid can't fill the chip twice
  int g128 = cdiv(M, 128) * cdiv(N, 128);
  bool big = g128 >= 512;
  if (const char* e = getenv("DMNIST_FC_TILE")) big = atoi(e) >= 128;
  auto s = cur_stream();
  bool bt = wT.has_value() && wT->defined();
  // split-K forward for grid-starved shapes (fc1 fwd @B=1024: 128 blocks
  // = half the chip idle): each k-slice writes its own fp32 [M][N] plane
  // (deterministic — no atomics), a fused epilogue sums planes + applies
  // bias/relu/philox-dropout.  DMNIST_FC_NOSPLIT=1 disables.
  int tiles64 = cdiv(M, 64) * cdiv(N, 64);
  int ksteps = cdiv(K, 64);
  // only deep-K shapes: short k-chains (fc2: ksteps=8) lose the plane +
  // epilogue overhead (measured 9.7 -> 16.4 us)
  if (!big && tiles64 < 256 && ksteps >= 16 &&
      !getenv("DMNIST_FC_NOSPLIT")) {
    int sk = std::min(ksteps, std::max(2, 512 / tiles64));
    auto acc = torch::empty({(int64_t)sk, (int64_t)M, (int64_t)N},
                            x.options().dtype(at::kFloat));
    p.C = acc.data_ptr();
    p.splitk = sk;
    if (bt) {
      CHECK_BF16((*wT)); CHECK_CONTIG((*wT));
      TORCH_CHECK(wT->size(0) == N && wT->size(1) == K, "wT shape mismatch");
      p.B = bf16_ptr(*wT); p.ldb = K;
      gemm_fwd_slices_64_bt(p, s);
    } else {
      p.B = bf16_ptr(w); p.ldb = N;
      gemm_fwd_slices_64(p, s);
    }
    launch_fwd_epilogue(acc.data_ptr<float>(), b.data_ptr<float>(),
                        bf16_mut(y), M, N, sk, relu ? 1 : 0, (float)p_keep,
                        (uint64_t)seed, (uint64_t)offset, offset_dev, s);
    return y;
  }
  if (bt) {
    // pre-transposed weight copy [N][K] -> vector B staging (B_NMAJ)
    CHECK_BF16((*wT)); CHECK_CONTIG((*wT));
    TORCH_CHECK(wT->size(0) == N && wT->size(1) == K, "wT shape mismatch");
    p.B = bf16_ptr(*wT); p.ldb = K;
    if (drop) (big ? gemm_fwd_drop_128_bt : gemm_fwd_drop_64_bt)(p, s);
    else if (relu) (big ? gemm_fwd_relu_128_bt : gemm_fwd_relu_64_bt)(p, s);
    else (big ? gemm_fwd_bias_128_bt : gemm_fwd_bias_64_bt)(p, s);
  } else {
    p.B = bf16_ptr(w); p.ldb = N;
    if (drop) (big ? gemm_fwd_drop_128 : gemm_fwd_drop_64)(p, s);
    else if (relu) (big ? gemm_fwd_relu_128 : gemm_fwd_relu_64)(p, s);
    else (big ? gemm_fwd_bias_128 : gemm_fwd_bias_64)(p, s);
  }
  return y;
}

torch::Tensor linear_act_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor b,
                             bool relu, double p_keep, int64_t seed,
                             int64_t offset,
                             c10::optional<torch::Tensor> wT) {
  return linear_act_fwd_impl(x, w, b, relu, p_keep, seed, offset, nullptr, wT);
}

// hipGraph-capturable variant: dropout RNG offset read from device memory
torch::Tensor linear_act_fwd_dev(torch::Tensor x, torch::Tensor w,
                                 torch::Tensor b, bool relu, double p_keep,
                                 int64_t seed, torch::Tensor offset_dev,
                                 c10::optional<torch::Tensor> wT) {
  TORCH_CHECK(offset_dev.scalar_type() == at::kLong && offset_dev.is_cuda(),
              "offset_dev must be a cuda int64 tensor");
  return linear_act_fwd_impl(x, w, b, relu, p_keep, seed, 0,
                             offset_dev.data_ptr<long>(), wT);
}

// core backward; when dw/db are passed they are accumulated into
// (pre-zeroed fp32 bucket views — the direct-grad path); else allocated.
std::vector<torch::Tensor> linear_act_bwd_impl(torch::Tensor dy, torch::Tensor x,
                                               torch::Tensor w, torch::Tensor y,
                                               bool relu, double p_keep,
                                               bool need_dx,
                                               torch::Tensor dw, torch::Tensor db) {
  CHECK_CUDA(dy); CHECK_BF16(dy); CHECK_CONTIG(dy);
  CHECK_BF16(x); CHECK_CONTIG(x);
  CHECK_BF16(w); CHECK_CONTIG(w);
  int B = x.size(0), K = x.size(1), N = w.size(1);
  auto s = cur_stream();
  if (!db.defined())
    db = torch::zeros({N}, x.options().dtype(at::kFloat));
  CHECK_F32(db); CHECK_CONTIG(db);
  torch::Tensor dyeff;
  bool mask = relu || p_keep < 1.0;
  if (mask) {
    dyeff = torch::empty_like(dy);
    launch_relu_drop_bwd(bf16_ptr(dy), bf16_ptr(y), bf16_mut(dyeff),
                         db.data_ptr<float>(), B, N,
                         (float)(1.0 / p_keep), 1, s);
  } else {
    dyeff = dy;
    launch_relu_drop_bwd(bf16_ptr(dy), bf16_ptr(y), bf16_mut(dyeff),
                         db.data_ptr<float>(), B, N, 1.0f, 0, s);
  }
  // dW[K,N] = x^T @ dyeff  (fp32 split-K atomics)
  if (!dw.defined())
    dw = torch::zeros({K, N}, x.options().dtype(at::kFloat));
  CHECK_F32(dw); CHECK_CONTIG(dw);
  {
    GemmParams p{};
    p.A = bf16_ptr(x); p.B = bf16_ptr(dyeff);
    p.C = dw.data_ptr();
    p.M = K; p.N = N; p.K = B;
    p.lda = K;  // A_T: A[m][k'] = x[k'*lda + m]
    p.ldb = N; p.ldc = N;
    if (dw_tr_enabled() && K % 8 == 0 && N % 64 == 0) {
      int bn = (N % 128 == 0) ? 128 : 64;
      p.splitk = pick_splitk(cdiv(K, 128), cdiv(N, bn), cdiv(B, 32));
      (bn == 128 ? gemm_dw_tr_128 : gemm_dw_tr_64)(p, s);
    } else {
      bool big = cdiv(K, 128) * cdiv(N, 128) >= 128;
      int bm = big ? 128 : 64;
      p.splitk = pick_splitk(cdiv(K, bm), cdiv(N, bm), cdiv(B, 32));
      (big ? gemm_dw_128 : gemm_dw_64)(p, s);
    }
  }
  torch::Tensor dx;
  if (need_dx) {
    dx = torch::empty({B, K}, x.options());
    GemmParams p{};
    p.A = bf16_ptr(dyeff); p.B = bf16_ptr(w);
    p.C = dx.data_ptr();
    p.M = B; p.N = K; p.K = N;
    p.lda = N; p.ldb = N; p.ldc = K;  // B_NMAJ: Bs[n'][k'] = w[n'*N + k']
    p.splitk = 1;
    bool big = cdiv(B, 128) * cdiv(K, 128) >= 128;
    (big ? gemm_dx_128 : gemm_dx_64)(p, s);
  } else {
    dx = torch::empty({0}, x.options());
  }
  return {dx, dw, db};
}

std::vector<torch::Tensor> linear_act_bwd(torch::Tensor dy, torch::Tensor x,
                                          torch::Tensor w, torch::Tensor y,
                                          bool relu, double p_keep,
                                          bool need_dx) {
  return linear_act_bwd_impl(dy, x, w, y, relu, p_keep, need_dx,
                             torch::Tensor(), torch::Tensor());
}

torch::Tensor linear_act_bwd_into(torch::Tensor dy, torch::Tensor x,
                                  torch::Tensor w, torch::Tensor y,
                                  bool relu, double p_keep, bool need_dx,
                                  torch::Tensor dw_out, torch::Tensor db_out) {
  auto r = linear_act_bwd_impl(dy, x, w, y, relu, p_keep, need_dx,
                               dw_out.view({x.size(1), w.size(1)}), db_out);
  return r[0];
}

// --------------------------------------------------------------------------
std::vector<torch::Tensor> conv_pool_fwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor b,
                                         c10::optional<torch::Tensor> wT) {
  CHECK_CUDA(x); CHECK_BF16(x); CHECK_CONTIG(x);
  CHECK_BF16(w); CHECK_CONTIG(w);
  CHECK_F32(b); CHECK_CONTIG(b);
  int NB = x.size(0), H = x.size(1), W = x.size(2), Cin = x.size(3);
  TORCH_CHECK(w.dim() == 4 && w.size(0) == 5 && w.size(1) == 5 &&
              w.size(2) == Cin, "conv weight must be [5,5,Cin,Cout]");
  int Cout = w.size(3);
  int Ho = H / 2, Wo = W / 2;
  auto y = torch::empty({NB, Ho, Wo, Cout}, x.options());
  auto amax = torch::empty({NB, Ho, Wo, Cout}, x.options().dtype(at::kByte));
  GemmParams p{};
  p.A = bf16_ptr(x); p.B = bf16_ptr(w);
  p.bias = b.data_ptr<float>();
  p.C = y.data_ptr(); p.amax = amax.data_ptr<uint8_t>();
  p.M = NB * Ho * Wo * 4; p.N = Cout; p.K = 25 * Cin;
  p.ldb = Cout; p.ldc = Cout;
  p.splitk = 1;
  p.CB = NB; p.CH = H; p.CW = W; p.CHo = Ho; p.CWo = Wo;
  p.Cin = Cin; p.Cout = Cout;
  auto s = cur_stream();
  if (Cin == 1) {
    // direct VALU kernel: K=25 is too small for MFMA to win here
    launch_conv1_direct_fwd(bf16_ptr(x), bf16_ptr(w), b.data_ptr<float>(),
                            bf16_mut(y), amax.data_ptr<uint8_t>(), NB, H, W,
                            Cout, s);
  } else if (conv_slab_supported(H, W, Cin, Cout)) {
    // per-image LDS slab: activations cross the fabric once (not 25x)
    launch_conv_fwd_slab(bf16_ptr(x), bf16_ptr(w), b.data_ptr<float>(),
                         bf16_mut(y), amax.data_ptr<uint8_t>(), NB, H, W,
                         Cin, Cout, s);
  } else {
    TORCH_CHECK(Cin % 8 == 0, "conv requires Cin==1 or Cin%8==0");
    if (wT.has_value() && wT->defined()) {
      CHECK_BF16((*wT)); CHECK_CONTIG((*wT));
      p.B = bf16_ptr(*wT); p.ldb = 25 * Cin;
      conv_fwd_pool_bt(p, s);
    } else {
      conv_fwd_pool(p, s);
    }
  }
  return {y, amax};
}

std::vector<torch::Tensor> conv_pool_bwd_impl(torch::Tensor dy, torch::Tensor x,
                                              torch::Tensor w, torch::Tensor y,
                                              torch::Tensor amax, bool need_dx,
                                              torch::Tensor dw, torch::Tensor db) {
  CHECK_CUDA(dy); CHECK_BF16(dy); CHECK_CONTIG(dy);
  CHECK_BF16(x); CHECK_CONTIG(x); CHECK_BF16(w); CHECK_CONTIG(w);
  int NB = x.size(0), H = x.size(1), W = x.size(2), Cin = x.size(3);
  int Cout = w.size(3);
  int Ho = H / 2, Wo = W / 2;
  auto s = cur_stream();
  // 1) scatter pooled grad (relu-masked) back to conv-output positions
  auto dact = torch::empty({NB, H, W, Cout}, x.options());
  if (!db.defined())
    db = torch::zeros({Cout}, x.options().dtype(at::kFloat));
  CHECK_F32(db); CHECK_CONTIG(db);
  launch_pool_bwd_scatter(bf16_ptr(dy), bf16_ptr(y),
                          amax.data_ptr<uint8_t>(), bf16_mut(dact),
                          db.data_ptr<float>(), NB * Ho * Wo, Cout, H, W, Wo,
                          s);
  // 2) dW[(khkw,ci), co] = sum_pixels x_shift * dact  (im2col^T GEMM)
  if (!dw.defined())
    dw = torch::zeros({5, 5, Cin, Cout}, x.options().dtype(at::kFloat));
  CHECK_F32(dw); CHECK_CONTIG(dw);
  {
    GemmParams p{};
    p.A = bf16_ptr(x); p.B = bf16_ptr(dact);
    p.C = dw.data_ptr();
    p.M = 25 * Cin; p.N = Cout; p.K = NB * H * W;
    p.ldb = Cout; p.ldc = Cout;
    p.CB = NB; p.CH = H; p.CW = W; p.Cin = Cin; p.Cout = Cout;
    if (Cin == 1 && H <= 28 && W <= 28 && Cout <= 32 &&
        !getenv("DMNIST_DW1_SLAB")) {
      launch_conv1_dw_direct(bf16_ptr(x), bf16_ptr(dact),
                             dw.data_ptr<float>(), NB, H, W, Cout, s);
    } else if (Cin == 1 && conv1_slab_supported(H, W, Cin, Cout)) {
      launch_conv1_dw_slab(bf16_ptr(x), bf16_ptr(dact), dw.data_ptr<float>(),
                           NB, H, W, Cout, s);
    } else if (Cin == 1) {
      // single 32x32 tile: target ~2048 blocks so the latency-bound gather
      // k-chain is short and the chip stays full
      p.splitk = std::min(cdiv(p.K, 64), 2048);
      conv1_dw_gemm(p, s);
    } else if (conv_slab_supported(H, W, Cin, Cout) &&
               ((dw_tr_level() == 0 && NB >= 2048) || getenv("DMNIST_DW_G"))) {
      // per-image-group slab dW: wins when the flush atomics amortize over
      // >=4 images/block; below that the implicit-GEMM form is faster
      launch_conv_dw_slab(bf16_ptr(x), bf16_ptr(dact), dw.data_ptr<float>(),
                          NB, H, W, Cin, Cout, s);
    } else {
      int tiles = cdiv(p.M, 128) * cdiv(p.N, 64);
      // total blocks target (DMNIST_DW_BLOCKS sweeps it): at BK=32 the
      // small-batch optimum is 1024 blocks (63 vs 76 us @1024), large
      // batches want 2048 (tools/dwsweep.py)
      int want = p.K <= 262144 ? 1024 : 2048;
      if (const char* e = getenv("DMNIST_DW_BLOCKS")) want = atoi(e);
      p.splitk = std::min(cdiv(p.K, 64), std::max(1, want / tiles));
      if (dw_tr_enabled() && Cin % 8 == 0 && Cout % 64 == 0)
        conv_dw_tr(p, s);
      else
        conv_dw_gemm(p, s);
    }
  }
  torch::Tensor dx;
  if (need_dx && conv_slab_supported(H, W, Cin, Cout)) {
    dx = torch::empty({NB, H, W, Cin}, x.options());
    launch_conv_dx_slab(bf16_ptr(dact), bf16_ptr(w), bf16_mut(dx), NB, H, W,
                        Cin, Cout, s);
  } else if (need_dx) {
    TORCH_CHECK(Cout % 32 == 0, "conv_dx requires Cout%32==0");
    dx = torch::empty({NB, H, W, Cin}, x.options());
    GemmParams p{};
    p.A = bf16_ptr(dact); p.B = bf16_ptr(w);
    p.C = dx.data_ptr();
    p.M = NB * H * W; p.N = Cin; p.K = 25 * Cout;
    p.ldc = Cin;
    p.CB = NB; p.CH = H; p.CW = W; p.Cin = Cin; p.Cout = Cout;
    p.splitk = 1;
    conv_dx_gemm(p, s);
  } else {
    dx = torch::empty({0}, x.options());
  }
  return {dx, dw, db};
}

std::vector<torch::Tensor> conv_pool_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor w, torch::Tensor y,
                                         torch::Tensor amax, bool need_dx) {
  return conv_pool_bwd_impl(dy, x, w, y, amax, need_dx, torch::Tensor(),
                            torch::Tensor());
}

torch::Tensor conv_pool_bwd_into(torch::Tensor dy, torch::Tensor x,
                                 torch::Tensor w, torch::Tensor y,
                                 torch::Tensor amax, bool need_dx,
                                 torch::Tensor dw_out, torch::Tensor db_out) {
  auto r = conv_pool_bwd_impl(dy, x, w, y, amax, need_dx,
                              dw_out.view_as(w), db_out);
  return r[0];
}

// --------------------------------------------------------------------------
// Fine-grained backward pieces for the hand-scheduled fused step
// (engine/fused_step.py): the composite linear_act_bwd/conv_pool_bwd above
// stay for the autograd path; these let dW run on a side stream while the
// dX chain proceeds.
// --------------------------------------------------------------------------
torch::Tensor mask_db(torch::Tensor dy, torch::Tensor y, bool relu,
                      double p_keep, torch::Tensor db_out) {
  CHECK_CUDA(dy); CHECK_BF16(dy); CHECK_CONTIG(dy);
  CHECK_F32(db_out); CHECK_CONTIG(db_out);
  int B = dy.size(0), N = dy.size(1);
  auto s = cur_stream();
  bool mask = relu || p_keep < 1.0;
  torch::Tensor dyeff = mask ? torch::empty_like(dy) : dy;
  launch_relu_drop_bwd(bf16_ptr(dy), mask ? bf16_ptr(y) : bf16_ptr(dy),
                       bf16_mut(dyeff), db_out.data_ptr<float>(), B, N,
                       (float)(1.0 / p_keep), mask ? 1 : 0, s);
  return dyeff;
}

void linear_dw_into(torch::Tensor x, torch::Tensor dyeff,
                    torch::Tensor dw_out) {
  CHECK_BF16(x); CHECK_CONTIG(x); CHECK_BF16(dyeff); CHECK_CONTIG(dyeff);
  CHECK_F32(dw_out);
  int B = x.size(0), K = x.size(1), N = dyeff.size(1);
  GemmParams p{};
  p.A = bf16_ptr(x); p.B = bf16_ptr(dyeff);
  p.C = dw_out.data_ptr();
  p.M = K; p.N = N; p.K = B;
  p.lda = K; p.ldb = N; p.ldc = N;
  if (dw_tr_enabled() && K % 8 == 0 && N % 64 == 0) {
    int bn = (N % 128 == 0) ? 128 : 64;
    p.splitk = pick_splitk(cdiv(K, 128), cdiv(N, bn), cdiv(B, 32));
    (bn == 128 ? gemm_dw_tr_128 : gemm_dw_tr_64)(p, cur_stream());
    return;
  }
  bool big = cdiv(K, 128) * cdiv(N, 128) >= 128;
  int bm = big ? 128 : 64;
  p.splitk = pick_splitk(cdiv(K, bm), cdiv(N, bm), cdiv(B, 32));
  (big ? gemm_dw_128 : gemm_dw_64)(p, cur_stream());
}

torch::Tensor linear_dx(torch::Tensor dyeff, torch::Tensor w) {
  CHECK_BF16(dyeff); CHECK_CONTIG(dyeff); CHECK_BF16(w); CHECK_CONTIG(w);
  int B = dyeff.size(0), N = dyeff.size(1), K = w.size(0);
  auto dx = torch::empty({B, K}, dyeff.options());
  GemmParams p{};
  p.A = bf16_ptr(dyeff); p.B = bf16_ptr(w);
  p.C = dx.data_ptr();
  p.M = B; p.N = K; p.K = N;
  p.lda = N; p.ldb = N; p.ldc = K;
  p.splitk = 1;
  bool big = cdiv(B, 128) * cdiv(K, 128) >= 128;
  (big ? gemm_dx_128 : gemm_dx_64)(p, cur_stream());
  return dx;
}

torch::Tensor linear_dx_unpool(torch::Tensor dyeff, torch::Tensor w,
                               torch::Tensor amax, torch::Tensor db_out,
                               int64_t Ho, int64_t Wo, int64_t C) {
  // dx = dyeff @ W^T fused with maxpool2x2-backward: the dX rows are pooled
  // [Ho][Wo][C] features; the epilogue scatters each value to its argmax
  // position of the 2x2 window (amax byte: 0..3 live, 7 dead — liveness
  // encoded at forward time so no y read is needed here) and accumulates
  // the conv bias grad — dact [B][2Ho][2Wo][C] comes straight out of the
  // GEMM (replaces linear_dx + pool_scatter on the backward critical path).
  CHECK_BF16(dyeff); CHECK_CONTIG(dyeff); CHECK_BF16(w); CHECK_CONTIG(w);
  int B = dyeff.size(0), N = dyeff.size(1), K = w.size(0);
  TORCH_CHECK(K == Ho * Wo * C, "linear_dx_unpool: K != Ho*Wo*C");
  auto dact = torch::empty({B, 2 * (int)Ho, 2 * (int)Wo, (int)C},
                           dyeff.options());
  GemmParams p{};
  p.A = bf16_ptr(dyeff); p.B = bf16_ptr(w);
  p.C = dact.data_ptr();
  p.M = B; p.N = K; p.K = N;
  p.lda = N; p.ldb = N; p.ldc = K;
  p.splitk = 1;
  p.CHo = Ho; p.CWo = Wo; p.Cout = C;
  p.amax = amax.data_ptr<uint8_t>();
  p.db = db_out.defined() ? db_out.data_ptr<float>() : nullptr;
  // 128-tile under ~2 blocks/CU is grid-starved (PMC: 57% parked at
  // B=1024's 200-block grid); 64-tiles quadruple the grid
  bool big = cdiv(B, 128) * cdiv(K, 128) >= 512;
  (big ? gemm_dx_unpool_128 : gemm_dx_unpool_64)(p, cur_stream());
  return dact;
}

torch::Tensor linear_dx_mask(torch::Tensor dyeff, torch::Tensor w,
                             torch::Tensor actm, torch::Tensor db_out,
                             double p_keep) {
  // dx = dyeff @ W^T with the downstream relu+dropout mask folded into the
  // epilogue (mask recovered from sign(actm): kept positions are scaled by
  // 1/p_keep, dropped/clipped are 0) + bias-grad column sums into db_out —
  // replaces linear_dx + the standalone mask_db pass (fc2-dX -> dyeff1).
  CHECK_BF16(dyeff); CHECK_CONTIG(dyeff); CHECK_BF16(w); CHECK_CONTIG(w);
  CHECK_BF16(actm); CHECK_CONTIG(actm);
  int B = dyeff.size(0), N = dyeff.size(1), K = w.size(0);
  TORCH_CHECK(actm.size(0) == B && actm.size(1) == K, "actm shape mismatch");
  auto dx = torch::empty({B, K}, dyeff.options());
  GemmParams p{};
  p.A = bf16_ptr(dyeff); p.B = bf16_ptr(w);
  p.C = dx.data_ptr();
  p.M = B; p.N = K; p.K = N;
  p.lda = N; p.ldb = N; p.ldc = K;
  p.splitk = 1;
  p.p_keep = (float)(1.0 / p_keep);  // kernel multiplies by the inverse
  p.actm = bf16_ptr(actm);
  p.db = db_out.defined() ? db_out.data_ptr<float>() : nullptr;
  bool big = cdiv(B, 128) * cdiv(K, 128) >= 128;
  (big ? gemm_dx_mask_128 : gemm_dx_mask_64)(p, cur_stream());
  return dx;
}

torch::Tensor pool_scatter(torch::Tensor dy, torch::Tensor y,
                           torch::Tensor amax, torch::Tensor db_out,
                           int64_t H, int64_t W) {
  CHECK_BF16(dy); CHECK_CONTIG(dy);
  int NB = dy.size(0), Ho = dy.size(1), Wo = dy.size(2), C = dy.size(3);
  auto dact = torch::empty({NB, (int)H, (int)W, C}, dy.options());
  launch_pool_bwd_scatter(bf16_ptr(dy), bf16_ptr(y),
                          amax.data_ptr<uint8_t>(), bf16_mut(dact),
                          db_out.data_ptr<float>(), NB * Ho * Wo, C, H, W,
                          Wo, cur_stream());
  return dact;
}

void conv_dw_into(torch::Tensor x, torch::Tensor dact, torch::Tensor dw_out) {
  CHECK_BF16(x); CHECK_CONTIG(x); CHECK_BF16(dact); CHECK_CONTIG(dact);
  int NB = x.size(0), H = x.size(1), W = x.size(2), Cin = x.size(3);
  int Cout = dact.size(3);
  auto s = cur_stream();
  GemmParams p{};
  p.A = bf16_ptr(x); p.B = bf16_ptr(dact);
  p.C = dw_out.data_ptr();
  p.M = 25 * Cin; p.N = Cout; p.K = NB * H * W;
  p.ldb = Cout; p.ldc = Cout;
  p.CB = NB; p.CH = H; p.CW = W; p.Cin = Cin; p.Cout = Cout;
  if (Cin == 1 && H <= 28 && W <= 28 && Cout <= 32 &&
      !getenv("DMNIST_DW1_SLAB")) {
    launch_conv1_dw_direct(bf16_ptr(x), bf16_ptr(dact),
                           dw_out.data_ptr<float>(), NB, H, W, Cout, s);
  } else if (Cin == 1 && conv1_slab_supported(H, W, Cin, Cout)) {
    launch_conv1_dw_slab(bf16_ptr(x), bf16_ptr(dact),
                         dw_out.data_ptr<float>(), NB, H, W, Cout, s);
  } else if (Cin == 1) {
    p.splitk = std::min(cdiv(p.K, 64), 2048);
    conv1_dw_gemm(p, s);
  } else if (conv_slab_supported(H, W, Cin, Cout) &&
             ((dw_tr_level() == 0 && NB >= 2048) || getenv("DMNIST_DW_G"))) {
    launch_conv_dw_slab(bf16_ptr(x), bf16_ptr(dact), dw_out.data_ptr<float>(),
                        NB, H, W, Cin, Cout, s);
  } else {
    int tiles = cdiv(p.M, 128) * cdiv(p.N, 64);
    // total blocks target (DMNIST_DW_BLOCKS sweeps it): at BK=32 the
    // small-batch optimum is 1024 blocks (63 vs 76 us @1024), large
    // batches want 2048 (tools/dwsweep.py)
    int want = p.K <= 262144 ? 1024 : 2048;
    if (const char* e = getenv("DMNIST_DW_BLOCKS")) want = atoi(e);
    p.splitk = std::min(cdiv(p.K, 64), std::max(1, want / tiles));
    if (getenv("DMNIST_DW_SPREAD")) {
      // contention probe: atomics land in a throwaway 16-plane scratch
      static torch::Tensor scratch;
      int64_t need = 16LL * p.M * p.N;
      if (!scratch.defined() || scratch.numel() < need)
        scratch = torch::zeros({need}, dact.options().dtype(at::kFloat));
      p.C = scratch.data_ptr();
      p.offset = 1;
    }
    if (dw_tr_enabled() && Cin % 8 == 0 && Cout % 64 == 0)
      conv_dw_tr(p, s);
    else
      conv_dw_gemm(p, s);
  }
}

void conv1_dw_pooled(torch::Tensor x, torch::Tensor dyp, torch::Tensor am,
                     torch::Tensor dw_out, torch::Tensor db_out) {
  // conv1 dW+db straight from the pooled gradient (pool backward fused in
  // the consumer; dact1 never materialized; liveness in the amax byte) —
  // see conv1_dw_pooled_kernel.
  CHECK_BF16(x); CHECK_CONTIG(x);
  CHECK_BF16(dyp); CHECK_CONTIG(dyp);
  int NB = x.size(0), H = x.size(1), W = x.size(2);
  int Cout = dyp.size(3);
  TORCH_CHECK(x.size(3) == 1 && H == 28 && W == 28 && Cout == 32 &&
                  dyp.size(1) == H / 2 && dyp.size(2) == W / 2,
              "conv1_dw_pooled: LeNet conv1 shapes only");
  launch_conv1_dw_pooled(bf16_ptr(x), bf16_ptr(dyp),
                         am.data_ptr<uint8_t>(), dw_out.data_ptr<float>(),
                         db_out.defined() ? db_out.data_ptr<float>() : nullptr,
                         NB, H, W, Cout, cur_stream());
}

torch::Tensor conv_dx(torch::Tensor dact, torch::Tensor w, int64_t Cin) {
  CHECK_BF16(dact); CHECK_CONTIG(dact); CHECK_BF16(w); CHECK_CONTIG(w);
  int NB = dact.size(0), H = dact.size(1), W = dact.size(2),
      Cout = dact.size(3);
  auto s = cur_stream();
  auto dx = torch::empty({NB, H, W, (int)Cin}, dact.options());
  if (conv_slab_supported(H, W, Cin, Cout)) {
    launch_conv_dx_slab(bf16_ptr(dact), bf16_ptr(w), bf16_mut(dx), NB, H, W,
                        Cin, Cout, s);
  } else {
    TORCH_CHECK(Cout % 32 == 0, "conv_dx requires Cout%32==0");
    GemmParams p{};
    p.A = bf16_ptr(dact); p.B = bf16_ptr(w);
    p.C = dx.data_ptr();
    p.M = NB * H * W; p.N = Cin; p.K = 25 * Cout;
    p.ldc = Cin;
    p.CB = NB; p.CH = H; p.CW = W; p.Cin = Cin; p.Cout = Cout;
    p.splitk = 1;
    conv_dx_gemm(p, s);
  }
  return dx;
}

// --------------------------------------------------------------------------
std::vector<torch::Tensor> softmax_xent_fwd(torch::Tensor logits,
                                            torch::Tensor labels,
                                            c10::optional<torch::Tensor> db_out,
                                            double inv_n) {
  // db_out (optional): fc2 bias grad accumulated in the same pass (column
  // sums of dlogits) — removes the standalone mask_db launch.
  // inv_n: scale on the correct-count output (pass 1/B to get the MEAN
  // accuracy straight out of the kernel; default 1.0 = raw count)
  CHECK_CUDA(logits); CHECK_BF16(logits); CHECK_CONTIG(logits);
  TORCH_CHECK(labels.scalar_type() == at::kLong, "labels must be int64");
  CHECK_CONTIG(labels);
  int B = logits.size(0), C = logits.size(1);
  TORCH_CHECK(C <= 16, "softmax_xent kernel supports C<=16");
  auto dl = torch::empty_like(logits);
  auto out = torch::zeros({2}, logits.options().dtype(at::kFloat));
  float* dbp = nullptr;
  if (db_out.has_value() && db_out->defined())
    dbp = db_out->data_ptr<float>();
  launch_softmax_xent(bf16_ptr(logits), labels.data_ptr<long>(),
                      bf16_mut(dl), out.data_ptr<float>(), B, C, dbp,
                      (float)inv_n, cur_stream());
  auto loss = out.select(0, 0);
  auto correct = out.select(0, 1);
  return {loss, correct, dl};
}

void sgd_step(torch::Tensor master, torch::Tensor grad, torch::Tensor shadow,
              bool has_shadow, double lr, double scale, double dc_keep,
              int64_t seed, int64_t offset,
              c10::optional<torch::Tensor> momentum, double mu) {
  CHECK_CUDA(master); CHECK_F32(master); CHECK_CONTIG(master);
  CHECK_F32(grad); CHECK_CONTIG(grad);
  float* mom = nullptr;
  if (momentum.has_value() && momentum->defined()) {
    CHECK_F32((*momentum)); CHECK_CONTIG((*momentum));
    mom = momentum->data_ptr<float>();
  }
  launch_sgd_step(master.data_ptr<float>(), grad.data_ptr<float>(),
                  has_shadow ? bf16_mut(shadow) : nullptr,
                  has_shadow ? 1 : 0, master.numel(),
                  (float)(lr * scale), (float)dc_keep, (uint64_t)seed,
                  (uint64_t)offset, mom, (float)mu, cur_stream());
}

void sgd_step_dev(torch::Tensor master, torch::Tensor grad,
                  torch::Tensor shadow, bool has_shadow,
                  torch::Tensor lr_scale_dev, double dc_keep, int64_t seed,
                  torch::Tensor offset_dev,
                  c10::optional<torch::Tensor> momentum, double mu,
                  bool zero_grad) {
  CHECK_CUDA(master); CHECK_F32(master); CHECK_CONTIG(master);
  CHECK_F32(grad); CHECK_CONTIG(grad);
  CHECK_F32(lr_scale_dev);
  float* mom = nullptr;
  if (momentum.has_value() && momentum->defined()) {
    CHECK_F32((*momentum)); CHECK_CONTIG((*momentum));
    mom = momentum->data_ptr<float>();
  }
  launch_sgd_step_dev(master.data_ptr<float>(), grad.data_ptr<float>(),
                      has_shadow ? bf16_mut(shadow) : nullptr,
                      has_shadow ? 1 : 0, master.numel(),
                      lr_scale_dev.data_ptr<float>(), (float)dc_keep,
                      (uint64_t)seed, offset_dev.data_ptr<long>(),
                      mom, (float)mu, zero_grad ? 1 : 0, cur_stream());
}

void transpose_bf16_batch(std::vector<torch::Tensor> srcs,
                          std::vector<torch::Tensor> dsts) {
  TORCH_CHECK(srcs.size() == dsts.size() && srcs.size() <= 4,
              "up to 4 transpose pairs");
  TransposeDesc d[4];
  for (size_t i = 0; i < srcs.size(); ++i) {
    CHECK_BF16(srcs[i]); CHECK_CONTIG(srcs[i]);
    CHECK_BF16(dsts[i]); CHECK_CONTIG(dsts[i]);
    d[i] = {bf16_ptr(srcs[i]), bf16_mut(dsts[i]),
            (int)srcs[i].size(0), (int)srcs[i].size(1)};
  }
  launch_transpose_bf16_batch(d, (int)srcs.size(), cur_stream());
}

void transpose_bf16_batch_adv(std::vector<torch::Tensor> srcs,
                              std::vector<torch::Tensor> dsts,
                              torch::Tensor step_dev,
                              torch::Tensor lr_scale_dev, double lr0,
                              double decay, int64_t decay_steps,
                              double inv_contrib) {
  // graph-tail variant: the batched transposes + the device step/LR
  // advance in ONE dispatch (step_advance folded into the last launch)
  TORCH_CHECK(srcs.size() == dsts.size() && srcs.size() <= 4,
              "up to 4 transpose pairs");
  TransposeDesc d[4];
  for (size_t i = 0; i < srcs.size(); ++i) {
    CHECK_BF16(srcs[i]); CHECK_CONTIG(srcs[i]);
    CHECK_BF16(dsts[i]); CHECK_CONTIG(dsts[i]);
    d[i] = {bf16_ptr(srcs[i]), bf16_mut(dsts[i]),
            (int)srcs[i].size(0), (int)srcs[i].size(1)};
  }
  launch_transpose_bf16_batch_adv(
      d, (int)srcs.size(), step_dev.data_ptr<long>(),
      lr_scale_dev.data_ptr<float>(), (float)lr0, (float)decay,
      (int)decay_steps, (float)inv_contrib, cur_stream());
}

void transpose_bf16(torch::Tensor src, torch::Tensor dst) {
  CHECK_CUDA(src); CHECK_BF16(src); CHECK_CONTIG(src);
  CHECK_BF16(dst); CHECK_CONTIG(dst);
  TORCH_CHECK(src.dim() == 2 && dst.size(0) == src.size(1) &&
              dst.size(1) == src.size(0), "transpose shape mismatch");
  launch_transpose_bf16(bf16_ptr(src), bf16_mut(dst), src.size(0),
                        src.size(1), cur_stream());
}

void step_advance(torch::Tensor step_dev, torch::Tensor lr_scale_dev,
                  double lr0, double decay, int64_t decay_steps,
                  double inv_contrib) {
  launch_step_advance(step_dev.data_ptr<long>(),
                      lr_scale_dev.data_ptr<float>(), (float)lr0,
                      (float)decay, (int)decay_steps, (float)inv_contrib,
                      cur_stream());
}

void grad_mask(torch::Tensor g, double keep, int64_t seed, int64_t step,
               int64_t rank, int64_t base,
               c10::optional<torch::Tensor> step_dev) {
  CHECK_CUDA(g); CHECK_F32(g); CHECK_CONTIG(g);
  TORCH_CHECK(base % 4 == 0,
              "grad_mask: slice base must be 16B-aligned so slice-wise "
              "masks compose to the whole-buffer mask");
  const long* sd = nullptr;
  if (step_dev.has_value() && step_dev->defined())
    sd = step_dev->data_ptr<long>();
  launch_grad_mask(g.data_ptr<float>(), g.numel(), base, (float)keep,
                   (uint64_t)seed, (uint64_t)step, (uint64_t)rank, sd,
                   cur_stream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("linear_act_fwd", &linear_act_fwd, "fused linear+bias+relu+dropout",
        py::arg("x"), py::arg("w"), py::arg("b"), py::arg("relu"),
        py::arg("p_keep"), py::arg("seed"), py::arg("offset"),
        py::arg("wT") = c10::nullopt);
  m.def("transpose_bf16", &transpose_bf16, "bf16 2-D transpose (wT refresh)");
  m.def("transpose_bf16_batch_adv", &transpose_bf16_batch_adv,
        "batched transposes + fused device step/LR advance (graph tail)");
  m.def("transpose_bf16_batch", &transpose_bf16_batch,
        "up to 4 transposes in one launch");
  m.def("mask_db", &mask_db, "relu/dropout grad mask + bias-grad column sums");
  m.def("linear_dw_into", &linear_dw_into, "dW = x^T dyeff into bucket view");
  m.def("linear_dx", &linear_dx, "dx = dyeff @ W^T");
  m.def("linear_dx_mask", &linear_dx_mask,
        "dx GEMM with fused relu/dropout mask + bias-grad column sums",
        py::arg("dyeff"), py::arg("w"), py::arg("actm"), py::arg("db_out"),
        py::arg("p_keep"));
  m.def("linear_dx_unpool", &linear_dx_unpool,
        "dx GEMM fused with maxpool2x2 backward scatter + conv db",
        py::arg("dyeff"), py::arg("w"), py::arg("amax"),
        py::arg("db_out"), py::arg("Ho"), py::arg("Wo"), py::arg("C"));
  m.def("pool_scatter", &pool_scatter, "maxpool bwd scatter + conv db");
  m.def("conv_dw_into", &conv_dw_into, "conv dW into bucket view");
  m.def("conv1_dw_pooled", &conv1_dw_pooled,
        "conv1 dW+db from the pooled gradient (pool bwd fused, no dact1)",
        py::arg("x"), py::arg("dyp"), py::arg("am"),
        py::arg("dw_out"), py::arg("db_out"));
  m.def("conv_dx", &conv_dx, "conv dX");
  m.def("linear_act_bwd", &linear_act_bwd, "linear backward (dx, dw, db)");
  m.def("linear_act_bwd_into", &linear_act_bwd_into,
        "linear backward accumulating dw/db into bucket views");
  m.def("conv_pool_fwd", &conv_pool_fwd, "fused conv5x5+bias+relu+maxpool",
        py::arg("x"), py::arg("w"), py::arg("b"), py::arg("wT") = c10::nullopt);
  m.def("conv_pool_bwd", &conv_pool_bwd, "conv+pool backward (dx, dw, db)");
  m.def("conv_pool_bwd_into", &conv_pool_bwd_into,
        "conv backward accumulating dw/db into bucket views");
  m.def("softmax_xent_fwd", &softmax_xent_fwd, "fused softmax-CE (+grad, +optional fc2 db, +optional acc-mean scale)",
        py::arg("logits"), py::arg("labels"),
        py::arg("db_out") = c10::nullopt, py::arg("inv_n") = 1.0);
  m.def("sgd_step", &sgd_step, "fused flat SGD(+momentum) apply",
        py::arg("master"), py::arg("grad"), py::arg("shadow"),
        py::arg("has_shadow"), py::arg("lr"), py::arg("scale"),
        py::arg("dc_keep"), py::arg("seed"), py::arg("offset"),
        py::arg("momentum") = c10::nullopt, py::arg("mu") = 0.0);
  m.def("sgd_step_dev", &sgd_step_dev,
        "SGD apply with device-side lr/offset (hipGraph-capturable); "
        "zero_grad clears the bucket in the same pass",
        py::arg("master"), py::arg("grad"), py::arg("shadow"),
        py::arg("has_shadow"), py::arg("lr_scale_dev"), py::arg("dc_keep"),
        py::arg("seed"), py::arg("offset_dev"),
        py::arg("momentum") = c10::nullopt, py::arg("mu") = 0.0,
        py::arg("zero_grad") = false);
  m.def("step_advance", &step_advance,
        "device-side staircase LR + step increment (inside the graph)");
  m.def("grad_mask", &grad_mask,
        "per-rank pre-aggregation drop-connect mask (in-place, philox; "
        "reference distributed_train.py:194-203)",
        py::arg("g"), py::arg("keep"), py::arg("seed"), py::arg("step"),
        py::arg("rank"), py::arg("base") = 0,
        py::arg("step_dev") = c10::nullopt);
  m.def("linear_act_fwd_dev", &linear_act_fwd_dev,
        "linear fwd with device-side dropout offset",
        py::arg("x"), py::arg("w"), py::arg("b"), py::arg("relu"),
        py::arg("p_keep"), py::arg("seed"), py::arg("offset_dev"),
        py::arg("wT") = c10::nullopt);
}
