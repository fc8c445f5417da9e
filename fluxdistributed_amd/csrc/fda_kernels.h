// Launcher API for the gfx950 kernels. bf16 tensors cross this boundary as
// raw unsigned short bit patterns (at::BFloat16 is bit-compatible).
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>

namespace fda {

enum class DT { F32 = 0, BF16 = 1 };

// fused logit cross-entropy: mean loss over N rows + dlogits in one pass
void ce_fwd_launch(const void* logits, const int64_t* target, float* loss,
                   void* dlogits, int N, int C, int ldl, DT dt,
                   hipStream_t s);

// out = relu(x + r)
void add_relu_fwd_launch(const void* x, const void* r, void* out, int64_t n,
                         DT dt, hipStream_t s);
// gx = gout * (out > 0)
void add_relu_bwd_launch(const void* gout, const void* out, void* gx, int64_t n,
                         DT dt, hipStream_t s);

// BatchNorm(+residual)(+ReLU), NHWC. ws layout (floats):
//   [0,C)    sum        [C,2C)  sumsq
//   [2C,3C)  scale      [3C,4C) shift
// save_mean/save_invstd are separate C-float buffers.
int bn_stats_partial_floats(int C, int64_t rows, DT dt);
// training: stats partials + fused reduce/finalize (running-stat update,
// scale/shift into ws[2C..4C))
void bn_stats_launch(const void* x, float* ws, float* part,
                     const float* weight, const float* bias,
                     float* running_mean, float* running_var, float* save_mean,
                     float* save_invstd, int64_t rows, int C, float momentum,
                     float eps, DT dt, hipStream_t s);
// pre-fold big conv-epilogue partial sets to NB2 rows at full grid width
void bn_partial_prefold_launch(const float* part, float* out, int NB,
                               int NB2, int chunkC, hipStream_t s);
// finalize from conv-epilogue partials ([NB][2][C])
void bn_finalize_from_partials_launch(
    const float* part, int NB, const float* weight, const float* bias,
    float* rm, float* rv, float* save_mean, float* save_invstd, float* ws,
    int64_t rows, int C, float momentum, float eps, hipStream_t s);

// eval: scale/shift from running stats
void bn_finalize_launch(float* ws, const float* weight, const float* bias,
                        float* running_mean, float* running_var,
                        float* save_mean, float* save_invstd, int64_t rows,
                        int C, bool training, float momentum, float eps,
                        hipStream_t s);
void bn_apply_launch(const void* x, const void* residual, void* out,
                     const float* ws /*scale/shift*/, int64_t rows, int C,
                     bool relu, DT dt, hipStream_t s);

// backward. ws layout (floats): [0,C) sum_g  [C,2C) sum_g_xhat
//   [2C,3C) k1  [3C,4C) k2   (k's folded with invstd*gamma in finalize)
// bwd: stats partials + fused reduce/finalize (gw, gb, k1/k2 into ws)
void bn_bwd_stats_launch(const void* gout, const void* x, const void* out,
                         const float* save_mean, const float* save_invstd,
                         float* ws, float* part, float* gw, float* gb,
                         int64_t rows, int C, bool relu, bool training,
                         bool accum_g, DT dt, hipStream_t s);
void bn_bwd_apply_launch(const void* gout, const void* x, const void* out,
                         const float* save_mean, const float* save_invstd,
                         const float* weight, const float* ws, void* gx,
                         void* gres, int64_t rows, int C, bool relu,
                         bool training, DT dt, hipStream_t s);

// MaxPool2d NHWC with saved argmax byte per output element
void maxpool_fwd_launch(const void* x, void* out, uint8_t* idx, int N, int H,
                        int W, int C, int HO, int WO, int KH, int KW, int S,
                        int P, DT dt, hipStream_t s);
void maxpool_bwd_launch(const void* gout, const uint8_t* idx, void* gx, int N,
                        int H, int W, int C, int HO, int WO, int KH, int KW,
                        int S, int P, DT dt, hipStream_t s);

// global average pool (NHWC): y[n][c] = mean_hw x; gx = gy/HW broadcast
void gap_fwd_launch(const void* x, void* y, int N, int HW, int C, DT dt,
                    hipStream_t s);
void gap_bwd_launch(const void* gy, void* gx, int N, int HW, int C, DT dt,
                    hipStream_t s);

// Implicit-GEMM conv, NHWC bf16 only (see conv_igemm.hip).
// fwd:   src=x [N,H,W,C], wgt=w [K][R*S*C], out=y [N*P*Q][K]
// dgrad: src=dy [N,P,Q,K], wgt=wt [R*S*C][K] (pre-transposed), out=dx
void conv_igemm_launch(const void* src, const void* wgt, void* out,
                       int N, int H, int W, int C, int K, int P, int Q,
                       int R, int S, int sy, int sx, int py, int px,
                       bool dgrad, hipStream_t stream,
                       float* stats = nullptr, float* skpart = nullptr,
                       int SK = 1, unsigned* cnt = nullptr,
                       const void* accsrc = nullptr);

// true when split-K uses the in-launch last-arriver combine (cnt tickets)
// instead of the separate conv_skcombine kernel (FLUXDIST_CONV_INLSK).
bool conv_use_inlsk();

// Tile/split-K plan for a conv launch — the single source of truth the
// bindings use to size the stats/skpart workspaces. M = output rows (per
// z-class for dgrad), OC = output channels, T = K-loop depth in BK=64
// steps, zbase = sy*sx for strided dgrad else 1.
void conv_igemm_plan(long M, int OC, long T, int zbase,
                     int* bm, int* bn, int* sk);

// split-K combine: y = bf16(sum over SK fp32 partials) + optional BN
// stats partials ([nblocks][2][OC])
int conv_skcombine_blocks(long M, int OC);
void conv_skcombine_launch(const float* part, void* y, float* stats, long M,
                           int OC, int SK, int nblocks, hipStream_t stream,
                           const void* accsrc = nullptr);

// stem conv (small C via channel-pad to 8, spatially pre-padded input)
void conv_stem_fwd_launch(const void* src, const void* wgt, void* out,
                          int N, int Hp, int Wp, int K, int P, int Q,
                          int R, int sy, int sx, hipStream_t stream,
                          float* stats = nullptr);
void conv_stem_wgrad_launch(const void* dy, const void* x, float* ws,
                            int N, int Hp, int Wp, int K, int P, int Q,
                            int R, int sy, int sx, hipStream_t stream);

// wgrad: ws[K][RS*C] fp32 (pre-zeroed) += dy^T @ im2col(x), atomic chunks.
// Two-phase mode (part != nullptr): chunk blocks plain-store into the
// [nch][K][RS*C] slab `part` (uninitialized OK) instead of DRAM atomics;
// follow with wgrad_combine_launch(ws, part, K*RS*C, nch).
void conv_wgrad_launch(const void* dy, const void* x, float* ws,
                       int N, int H, int W, int C, int K, int P, int Q,
                       int R, int S, int sy, int sx, int py, int px,
                       hipStream_t stream, float* part = nullptr);
// tile/chunk plan (single source of truth with the bindings): M = N*P*Q
void conv_wgrad_plan(long M, int C, int K, int R, int S,
                     int* ft, int* mch, int* nch);
bool conv_wgrad_two_phase();   // FLUXDIST_WGRAD_2PH knob
void wgrad_combine_launch(float* ws, const float* part, long n, int nch,
                          hipStream_t stream);

// batched conv-weight transpose: w[k][rc] -> wt[rc][k] for all tensors in
// one launch (device pointer arrays)
void wt_transpose_launch(const long* src_ptrs, long* dst_ptrs, const int* Ks,
                         const int* RCs, const int* tile_counts, int ntensors,
                         int max_tiles, hipStream_t stream);

// direct-grad flush: G_bf16 += cast(ws_f32), one conv-weight slice
void grad_accum_bf16_launch(void* g, const float* ws, long n, hipStream_t s);
// batched flush: all pending slices in one launch (device pointer arrays)
void grad_accum_batch_launch(const long* g_ptrs, const long* ws_ptrs,
                             const long* ns, int ntensors, long max_n,
                             hipStream_t s);

// FC-head padding helpers (Dense on the conv kernels, ops/linear.py)
void pad_rows_bf16_launch(void* dst, const void* src, long M, int C, int ldl,
                          hipStream_t s);
void bias_add_rows_bf16_launch(void* y, const void* bias, long M, int ldl,
                               hipStream_t s);
void colsum_accum_bf16_launch(void* g, const void* dy, long M, int ldl,
                              int Cvalid, hipStream_t s);

// flat fused optimizers. P: param dtype; M/V/S fp32; G param dtype.
void sgd_step_launch(void* P, const void* G, float* M, float* V, int64_t n,
                     float lr, float mom, float wd, bool nesterov,
                     bool has_master, DT dt, hipStream_t s);
void adam_step_launch(void* P, const void* G, float* M, float* V, float* S,
                      int64_t n, float lr, float b1, float b2, float eps,
                      float wd, float bc1, float bc2, bool has_master, DT dt,
                      hipStream_t s);

}  // namespace fda
