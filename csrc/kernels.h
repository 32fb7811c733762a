// Host-visible API of the HIP kernel library (implemented in *.hip).
#pragma once
#include <hip/hip_runtime.h>
#include <stdint.h>

struct GemmParams {
  const unsigned short* A;
  const unsigned short* B;
  const float* bias;
  void* C;
  uint8_t* amax;
  int M, N, K;
  int lda, ldb, ldc;
  int splitk;
  int CB, CH, CW, CHo, CWo, Cin, Cout;
  float p_keep;
  uint64_t seed, offset;
  const long* offset_dev;  // when set, RNG offset is read from device memory
                           // (hipGraph replay: host args are frozen)
  float* db;  // EPI_UNPOOL / EPI_MASK_DB: bias-grad accumulator
  const unsigned short* actm;  // EPI_MASK_DB: saved activation (mask source)
};

// gemm_tile.hip — implicit-GEMM MFMA entry points
void gemm_fwd_bias_128(const GemmParams&, hipStream_t);
void gemm_fwd_bias_64(const GemmParams&, hipStream_t);
void gemm_fwd_relu_128(const GemmParams&, hipStream_t);
void gemm_fwd_relu_64(const GemmParams&, hipStream_t);
void gemm_fwd_drop_128(const GemmParams&, hipStream_t);
void gemm_fwd_drop_64(const GemmParams&, hipStream_t);
void gemm_dx_128(const GemmParams&, hipStream_t);
void gemm_dx_64(const GemmParams&, hipStream_t);
void gemm_dx_unpool_128(const GemmParams&, hipStream_t);
void gemm_dx_unpool_64(const GemmParams&, hipStream_t);
void gemm_dx_mask_128(const GemmParams&, hipStream_t);
void gemm_dx_mask_64(const GemmParams&, hipStream_t);
void gemm_dw_128(const GemmParams&, hipStream_t);
void gemm_dw_64(const GemmParams&, hipStream_t);
void conv_fwd_pool(const GemmParams&, hipStream_t);
void conv1_fwd_pool(const GemmParams&, hipStream_t);
void conv_dx_gemm(const GemmParams&, hipStream_t);
void conv_dw_gemm(const GemmParams&, hipStream_t);
// dw_tr.hip — glds + ds_read_b64_tr_b16 k-major dW GEMMs (no staging scatter)
void conv_dw_tr(const GemmParams&, hipStream_t);    // 5x5-SAME gather A
void gemm_dw_tr_128(const GemmParams&, hipStream_t);  // plain k-major A, BN=128
void gemm_dw_tr_64(const GemmParams&, hipStream_t);   // plain k-major A, BN=64

// ops_misc.hip — host wrappers
void launch_relu_drop_bwd(const unsigned short* dy, const unsigned short* y,
                          unsigned short* dyeff, float* db, int B, int N,
                          float inv_keep, int apply_mask, hipStream_t);
void launch_pool_bwd_scatter(const unsigned short* dy, const unsigned short* y,
                             const uint8_t* amax, unsigned short* dact,
                             float* db, int Mpool, int C, int H, int W, int Wo,
                             hipStream_t);
void launch_softmax_xent(const unsigned short* logits, const long* labels,
                         unsigned short* dlogits, float* out, int B, int C,
                         float* db, float inv_n, hipStream_t);
void launch_sgd_step(float* master, float* grad, unsigned short* shadow,
                     int has_shadow, long n, float lr_scale, float dc_keep,
                     uint64_t seed, uint64_t offset, float* momentum, float mu,
                     hipStream_t);
// graph-capturable variants: lr_scale / RNG offset read from device memory
void launch_sgd_step_dev(float* master, float* grad,
                         unsigned short* shadow, int has_shadow, long n,
                         const float* lr_scale_dev, float dc_keep,
                         uint64_t seed, const long* offset_dev,
                         float* momentum, float mu, int zero_grad,
                         hipStream_t);
void launch_conv1_dw_pooled(const unsigned short* x, const unsigned short* dyp,
                            const uint8_t* am, float* dw, float* db, int NB,
                            int H, int W, int Cout, hipStream_t s);
void launch_grad_mask(float* g, long n, long base, float keep, uint64_t seed,
                      uint64_t step, uint64_t rank, const long* step_dev,
                      hipStream_t s);
void launch_step_advance(long* step_dev, float* lr_scale_dev, float lr0,
                         float decay, int decay_steps, float inv_contrib,
                         hipStream_t);
void launch_transpose_bf16(const unsigned short* src, unsigned short* dst,
                           int R, int C, hipStream_t);
struct TransposeDesc {
  const unsigned short* src;
  unsigned short* dst;
  int R, C;
};
struct TransposeBatchArgs {
  TransposeDesc d[4];
  int tile0[4];
  int n, total;
  // optional fused step/LR advance (graph tail: saves one dispatch)
  int do_advance;
  long* step_dev;
  float* lr_scale_dev;
  float lr0, decay, inv_contrib;
  int decay_steps;
};
void launch_transpose_bf16_batch(const TransposeDesc* descs, int n,
                                 hipStream_t);
void launch_transpose_bf16_batch_adv(const TransposeDesc* descs, int n,
                                     long* step_dev, float* lr_scale_dev,
                                     float lr0, float decay, int decay_steps,
                                     float inv_contrib, hipStream_t);
void launch_conv1_direct_fwd(const unsigned short* x, const unsigned short* w,
                             const float* bias, unsigned short* y,
                             uint8_t* amax, int NB, int H, int W, int Cout,
                             hipStream_t);
// B-transposed (pre-transposed weight) forward GEMM entries
void gemm_fwd_bias_128_bt(const GemmParams&, hipStream_t);
void gemm_fwd_bias_64_bt(const GemmParams&, hipStream_t);
void gemm_fwd_relu_128_bt(const GemmParams&, hipStream_t);
void gemm_fwd_relu_64_bt(const GemmParams&, hipStream_t);
void gemm_fwd_drop_128_bt(const GemmParams&, hipStream_t);
void gemm_fwd_drop_64_bt(const GemmParams&, hipStream_t);
void conv_fwd_pool_bt(const GemmParams&, hipStream_t);
void conv1_dw_gemm(const GemmParams&, hipStream_t);
// conv_slab.hip — per-image LDS-slab conv kernels (H=W=14, Cin=32, Cout=64)
bool conv_slab_supported(int H, int W, int Cin, int Cout);
void launch_conv_fwd_slab(const unsigned short* x, const unsigned short* w,
                          const float* bias, unsigned short* y, uint8_t* amax,
                          int NB, int H, int W, int Cin, int Cout, hipStream_t);
void launch_conv_dx_slab(const unsigned short* dact, const unsigned short* w,
                         unsigned short* dx, int NB, int H, int W, int Cin,
                         int Cout, hipStream_t);
void launch_conv_dw_slab(const unsigned short* x, const unsigned short* dact,
                         float* dw, int NB, int H, int W, int Cin, int Cout,
                         hipStream_t);
bool conv1_slab_supported(int H, int W, int Cin, int Cout);
void launch_conv1_dw_slab(const unsigned short* x, const unsigned short* dact,
                          float* dw, int NB, int H, int W, int Cout,
                          hipStream_t);
// ops_misc.hip — direct-VALU conv1 dW (Cin==1, H,W<=28, Cout<=32)
void launch_conv1_dw_direct(const unsigned short* x,
                            const unsigned short* dact, float* dw, int NB,
                            int H, int W, int Cout, hipStream_t s);
// gemm_tile.hip — split-K fwd slice entries + their summing epilogue
void gemm_fwd_slices_64(const GemmParams&, hipStream_t);
void gemm_fwd_slices_64_bt(const GemmParams&, hipStream_t);
// ops_misc.hip — fwd epilogue: y = [drop][relu](sum_slices + bias), bf16
void launch_fwd_epilogue(const float* acc, const float* bias,
                         unsigned short* y, int M, int N, int slices,
                         int relu, float p_keep, uint64_t seed,
                         uint64_t offset, const long* offset_dev,
                         hipStream_t);
