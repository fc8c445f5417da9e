// Implicit-GEMM 2D convolution for gfx950 (CDNA4), NHWC bf16.
//
// MI355X-native replacement for the conv layer the reference gets for free
// from cuDNN via NNlibCUDA (/root/reference -> Flux conv):
// hand-written MFMA kernels — v_mfma_f32_16x16x32_bf16 tiles, LDS staging
// via global_load_lds (direct HBM->LDS DMA), source-side XOR swizzle for
// bank-conflict-free ds_read_b128 fragment reads (guide T2 / rule 21),
// double-buffered with the staging DMA in flight across the compute phase.
//
// GEMM view (forward):
//   M = N*P*Q output pixels, Nd = K output channels, Kd = R*S*C
//   y[m][k] = sum_kd A[m][kd] * B[kd][k]
//   A = im2col gather of x (never materialized: the per-lane source
//       addresses of the LDS DMA do the gather; out-of-bounds rows read a
//       16-B zero page)
//   B = w[k][r][s][c] (torch channels_last conv weight = [K][R*S*C] rows)
//
// DGRAD: dx[n,h,w,c] = sum_{r,s,k} dy[n,(h+py-r)/sy,(w+px-s)/sx,k]*w[k,r,s,c].
// For stride > 1 the output pixels are partitioned by (h%sy, w%sx) parity
// class (blockIdx.z); each class iterates only its own valid filter taps,
// so no MFMA work is spent on divisibility-masked zero rows (a 4x saving
// for 3x3 stride-2). B = pre-transposed weights wt[rs*C + c][k]
// (k-contiguous rows).
//
// Tile geometry (templated; wave tile fixed at 64x64 = 4x4 fragments of
// 16x16 so every config runs 32 MFMAs per K-step per wave — the
// MFMA-per-glds ratio is what sets throughput, guide §5 ladder):
//   OC % 128 == 0 : BM=128 x BN=128, 4 waves as 2x2, LDS 2x32 KiB
//   OC % 128 != 0 : BM=256 x BN=64,  4 waves as 4x1, LDS 2x40 KiB
// fp32 accumulate, bf16 store. Staging source offsets advance
// incrementally within a filter tap (full address math only on tap
// changes). The fwd epilogue can also emit per-channel sum/sumsq
// partials of the rounded output for the downstream BatchNorm (`stats`),
// and small-M deep-K shapes split the K loop over blockIdx.z into fp32
// partials (`skpart`) folded by conv_skcombine_kernel — layer4-sized
// grids otherwise fill only ~30% of the 256 CUs.
//
// This file also contains: the CONV_STEM mode (small-C stems on a
// channel-padded C=8 image, r-only tap loop), the wgrad kernel
// (reduction along the pixel axis via ds_read_tr16_b64 hardware
// transpose reads from an m4-grouped LDS image, window positions
// permuted for bank-conflict-free half-wave reads, fp32 atomic chunk
// accumulation), and the batched weight transposer for the dgrad B
// layout.
//
// Constraints (host wrapper): staged reduction channels (C fwd / K dgrad)
// and output channels multiples of 64, dilation 1, groups 1; everything
// else (grouped/dilated convs) falls back to the library path.

#include <hip/hip_runtime.h>
#include <cstdlib>
#include "fda_common.h"

namespace fda {

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float floatx4;

__device__ __align__(16) static const unsigned short conv_zero16[8] = {0};

#define FDA_GLDS16(gptr, lptr)                                              \
    __builtin_amdgcn_global_load_lds(                                       \
        (const __attribute__((address_space(1))) void*)(gptr),              \
        (__attribute__((address_space(3))) void*)(lptr), 16, 0, 0)

enum ConvMode { CONV_FWD = 0, CONV_DGRAD = 1, CONV_STEM = 2 };
// CONV_STEM: small-C stem conv on a channel-padded (C=8) pre-padded input.
// Taps are r-only: each K-step covers one filter row r as 64 virtual
// channels = 8 pixels (s=0..7, s==7 zero-padded in wpad) x 8 channels
// (c>=3 zero). B reads wpad[k][r][64]; A granules are whole 16-B pixels.

constexpr int BK = 64;

// NW = waves per block = (BM/64)*(BN/64): 4 for the 128x128 / 256x64
// configs (256 threads, 2 blocks/CU), 8 for the 256x128 big tile (512
// threads, 1 block/CU — fewer glds per wave per K-step: 6 vs 8, same 32
// MFMAs, so a higher MFMA:staging ratio on the deep-K layers).
template <int MODE, int BM, int BN, int WN, int NBUF = 2, int BKT = BK>
__global__ __launch_bounds__((BM / 64) * (BN / 64) * 64,
                             512 / ((BM / 64) * (BN / 64) * 64))
void conv_igemm_kernel(
    const unsigned short* __restrict__ src,
    const unsigned short* __restrict__ wgt,
    unsigned short* __restrict__ out,
    int N, int H, int W, int C,
    int K, int P, int Q,
    int R, int S, int sy, int sx, int py, int px,
    float* __restrict__ stats, /* [mtiles][2][OC] or null: per-channel
                                 sum/sumsq of the rounded output — feeds the
                                 BN reduce+finalize directly (the separate
                                 bn_stats read pass is skipped) */
    float* __restrict__ skpart, /* split-K: fp32 partial output
                                 [SK][M][OC]; reduced either by the
                                 separate combine kernel or in-launch by
                                 the last-arriving slice (cnt != null).
                                 Small-M late layers fill only ~30% of the
                                 chip otherwise. */
    int SK,
    unsigned* __restrict__ cnt, /* in-launch combine tickets, one per
                                 (m,n,zclass) tile, memset to 0 before the
                                 launch; null = separate combine kernel */
    const unsigned short* __restrict__ accsrc
        /* non-null (dgrad): bf16 tensor with the OUTPUT's layout added into
           the result before the store — the residual-junction grad
           (d/d identity) fused into dx so autograd's separate
           CUDAFunctor_add pass disappears (ops/conv.py junction stash) */
    ) {
    constexpr int NW = (BM / 64) * (BN / 64);   // waves per block
    constexpr int A_ELEMS = BM * BKT;
    constexpr int B_ELEMS = BN * BKT;
    constexpr int BUF_ELEMS = A_ELEMS + B_ELEMS;
    // one glds = 64 lanes x 16 B = 512/BKT rows of BKT elements
    constexpr int RPG = 512 / BKT;       // rows per glds (8 @64, 16 @32)
    constexpr int GPR = BKT / 8;         // 16-B granules per row (8 / 4)
    constexpr int KH = BKT / 32;         // mfma k-halves per staged tile
    // granule swizzle: BK64 keeps the proven row&7; BK32 folds row>>2 in
    // (plain row&3 repeats banks every 4 rows: 4-way b128 conflicts; the
    // fold reaches the 2-way floor of 4 slots over 8 same-phase rows)
    auto swz = [](int row) {
        return BKT == 64 ? (row & 7) : ((row ^ (row >> 2)) & (GPR - 1));
    };
    constexpr int AI = BM / (RPG * NW);  // A glds per wave per tile
    constexpr int BI = BN / (RPG * NW);  // B glds per wave per tile

    const int OC = (MODE == CONV_DGRAD) ? C : K;
    const int RC = (MODE == CONV_FWD) ? C : ((MODE == CONV_STEM) ? 64 : K);

    int a = 0, b = 0, OH, OW, r0 = 0, s0 = 0, nR = R, nS = S;
    int sk = 0, zrest = blockIdx.z;
    if (SK > 1) { sk = zrest % SK; zrest /= SK; }
    if (MODE != CONV_DGRAD) {
        OH = P; OW = Q;
        if (MODE == CONV_STEM) nS = 1;   // taps iterate r only
    } else {
        a = zrest / sx;  b = zrest % sx;
        OH = (H - a + sy - 1) / sy;
        OW = (W - b + sx - 1) / sx;
        r0 = (a + py) % sy;  nR = (R - r0 + sy - 1) / sy;
        s0 = (b + px) % sx;  nS = (S - s0 + sx - 1) / sx;
        if (OH <= 0 || OW <= 0) return;
        if (nR < 0) nR = 0;
        if (nS < 0) nS = 0;
    }
    const long M = (long)N * OH * OW;
    const long m0 = (long)blockIdx.x * BM;
    if (m0 >= M) return;
    const int n0 = blockIdx.y * BN;

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int wm = wid / WN;             // wave row (64-pixel granularity)
    const int wn = wid % WN;             // wave col (64-channel granularity)

    extern __shared__ unsigned short lds[];   // [NBUF][BUF_ELEMS]
    constexpr int GPW = AI + BI;              // glds per wave per tile

    // ---- per-lane staging descriptors ------------------------------------
    int a_row[AI];
    long a_pix[AI];
    int a_hb[AI], a_wb[AI];
    bool a_mok[AI];
    const int cslot = lane % GPR;
    #pragma unroll
    for (int i = 0; i < AI; ++i) {
        const int row = (wid * AI + i) * RPG + lane / GPR;
        a_row[i] = row;
        const long m = m0 + row;
        const bool mok = m < M;
        const long mm = mok ? m : 0;
        const int ow = (int)(mm % OW);
        const int oh = (int)((mm / OW) % OH);
        const int n = (int)(mm / ((long)OW * OH));
        a_mok[i] = mok;
        if (MODE == CONV_STEM) {
            a_hb[i] = oh * sy;           // input pre-padded: no -py
            a_wb[i] = ow * sx;
            a_pix[i] = ((long)n * H) * W * 8;
        } else if (MODE == CONV_FWD) {
            a_hb[i] = oh * sy - py;
            a_wb[i] = ow * sx - px;
            a_pix[i] = ((long)n * H) * W * C;
        } else {
            a_hb[i] = oh + (a + py) / sy;
            a_wb[i] = ow + (b + px) / sx;
            a_pix[i] = ((long)n * P) * Q * K;
        }
    }
    int b_row[BI];
    #pragma unroll
    for (int i = 0; i < BI; ++i)
        b_row[i] = (wid * BI + i) * RPG + lane / GPR;

    const int cblocks = RC / BKT;
    const int T = nR * nS * cblocks;

    // Incremental staging offsets: stage() is called with strictly
    // increasing `it` (prologue, then it+NBUF-1), so within one filter tap
    // the source offsets just advance by BK; the full per-row address math
    // (multiply + bounds check) runs only on tap changes — it was
    // comparable to the whole MFMA phase per K-step otherwise.
    long a_goff[AI];
    bool a_okc[AI];
    long b_goff[BI];
    int last_rsi = -1;
    auto stage = [&](int buf, int it) {
        const int rsi = it / cblocks;
        const int ri = rsi / nS, si = rsi % nS;
        unsigned short* base = lds + buf * BUF_ELEMS;
        if (rsi != last_rsi) {
            last_rsi = rsi;
            const int cb = (it % cblocks) * BKT;  // 0 except NBUF>1 prologue
            #pragma unroll
            for (int i = 0; i < AI; ++i) {
                const int cs = (cslot ^ swz(a_row[i])) * 8;
                bool ok = a_mok[i];
                long off = 0;
                if (MODE == CONV_STEM) {
                    off = a_pix[i] + ((long)(a_hb[i] + ri) * W + a_wb[i]) * 8 + cs;
                } else if (MODE == CONV_FWD) {
                    const int h = a_hb[i] + ri, w = a_wb[i] + si;
                    ok = ok && (unsigned)h < (unsigned)H && (unsigned)w < (unsigned)W;
                    off = a_pix[i] + ((long)h * W + w) * C + cb + cs;
                } else {
                    const int p = a_hb[i] - ri, q = a_wb[i] - si;
                    ok = ok && (unsigned)p < (unsigned)P && (unsigned)q < (unsigned)Q;
                    off = a_pix[i] + ((long)p * Q + q) * K + cb + cs;
                }
                a_goff[i] = off;
                a_okc[i] = ok;
            }
            const int r = r0 + ri * ((MODE == CONV_FWD) ? 1 : sy);
            const int s = s0 + si * ((MODE == CONV_FWD) ? 1 : sx);
            const int rs = r * S + s;
            #pragma unroll
            for (int i = 0; i < BI; ++i) {
                const int row = b_row[i];
                const int cs = (cslot ^ swz(row)) * 8;
                if (MODE == CONV_STEM)
                    b_goff[i] = ((long)(n0 + row) * R + ri) * 64 + cs;
                else if (MODE == CONV_FWD)
                    b_goff[i] = (long)(n0 + row) * R * S * C + (long)rs * C + cb + cs;
                else
                    b_goff[i] = (long)((long)rs * C + n0 + row) * K + cb + cs;
            }
        } else {
            #pragma unroll
            for (int i = 0; i < AI; ++i) a_goff[i] += BKT;
            #pragma unroll
            for (int i = 0; i < BI; ++i) b_goff[i] += BKT;
        }
        #pragma unroll
        for (int i = 0; i < AI; ++i) {
            const unsigned short* sp = a_okc[i] ? src + a_goff[i] : conv_zero16;
            FDA_GLDS16(sp, base + (wid * AI + i) * RPG * BKT);
        }
        #pragma unroll
        for (int i = 0; i < BI; ++i) {
            FDA_GLDS16(wgt + b_goff[i],
                       base + A_ELEMS + (wid * BI + i) * RPG * BKT);
        }
    };

    // ---- fragment read offsets (elements into an lds buffer) -------------
    int a_off[4][KH], b_off[4][KH];
    {
        const int fr = lane & 15, fq = lane >> 4;
        #pragma unroll
        for (int mi = 0; mi < 4; ++mi)
            #pragma unroll
            for (int kh = 0; kh < KH; ++kh) {
                const int row = wm * 64 + mi * 16 + fr;
                const int slot = ((kh * 4 + fq) ^ swz(row)) & (GPR - 1);
                a_off[mi][kh] = row * BKT + slot * 8;
            }
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni)
            #pragma unroll
            for (int kh = 0; kh < KH; ++kh) {
                const int row = wn * 64 + ni * 16 + fr;
                const int slot = ((kh * 4 + fq) ^ swz(row)) & (GPR - 1);
                b_off[ni][kh] = A_ELEMS + row * BKT + slot * 8;
            }
    }

    floatx4 acc[4][4];
    #pragma unroll
    for (int mi = 0; mi < 4; ++mi)
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni) acc[mi][ni] = floatx4{0.f, 0.f, 0.f, 0.f};

    // ---- main loop: double buffer; tile t+1's DMA in flight over tile t's
    // compute, drained at the iteration boundary (guide T3 minimum form) ---
    int it0 = 0, itN = T;
    if (SK > 1) {
        const int chunk = (T + SK - 1) / SK;
        it0 = sk * chunk;
        itN = it0 + chunk < T ? it0 + chunk : T;
        if (it0 >= itN) itN = it0;       // empty slice: epilogue writes 0
    }
    if (it0 < itN) stage(it0 % NBUF, it0);
    if (NBUF > 2 && it0 + 1 < itN) stage((it0 + 1) % NBUF, it0 + 1);
    for (int it = it0; it < itN; ++it) {
        // tile `it` landed chip-wide: each wave drains its own DMA, the
        // barrier joins all waves. This is the loop's ONLY barrier — the
        // next K-step's staging targets the buffer every wave finished
        // reading before it arrived here. With NBUF==3 the counted wait
        // leaves tile it+1's DMA in flight across the barrier (T4).
        if (NBUF > 2 && it + 1 < itN) {
            if constexpr (GPW == 4)
                asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
            else if constexpr (GPW == 5)
                asm volatile("s_waitcnt vmcnt(5)" ::: "memory");
            else if constexpr (GPW == 6)
                asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
            else if constexpr (GPW == 8)
                asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
            else
                asm volatile("s_waitcnt vmcnt(10)" ::: "memory");
        } else {
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        }
        __builtin_amdgcn_s_barrier();
        // stage-before-reads: tried the other order in r2 (reads first so
        // their latency hides under stage's address math) — measured
        // +314 us/step on fwd+dgrad: delaying the glds issue shrinks the
        // DMA's flight window over the MFMA phase, which costs more than
        // the lgkm stall it saves. (The wgrad kernel is the opposite case:
        // hipcc force-drains DMA before tr16 reads, so there reads-first
        // is required — see conv_wgrad_kernel.)
        if (it + NBUF - 1 < itN) stage((it + NBUF - 1) % NBUF, it + NBUF - 1);
        const unsigned short* buf = lds + (it % NBUF) * BUF_ELEMS;
        short8 af[4][KH], bf[4][KH];
        #pragma unroll
        for (int kh = 0; kh < KH; ++kh) {
            #pragma unroll
            for (int mi = 0; mi < 4; ++mi)
                af[mi][kh] = *(const short8*)(buf + a_off[mi][kh]);
            #pragma unroll
            for (int ni = 0; ni < 4; ++ni)
                bf[ni][kh] = *(const short8*)(buf + b_off[ni][kh]);
        }
        __builtin_amdgcn_s_setprio(1);
        #pragma unroll
        for (int kh = 0; kh < KH; ++kh)
            #pragma unroll
            for (int mi = 0; mi < 4; ++mi)
                #pragma unroll
                for (int ni = 0; ni < 4; ++ni)
                    acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        af[mi][kh], bf[ni][kh], acc[mi][ni], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
        // no end-of-iteration barrier: the next iteration's vmcnt + top
        // barrier already orders buffer reuse (a wave reaches that barrier
        // only after its lgkm-waited fragment reads of this buffer), and
        // the final iteration needs no sync before the epilogue.
    }

    // ---- epilogue: C/D map col=lane&15, row=(lane>>4)*4+j ----------------
    const int fcol = lane & 15, frow0 = (lane >> 4) * 4;
    float ssum[4] = {0.f, 0.f, 0.f, 0.f};   // per-ni channel sums (rounded y)
    float sq[4] = {0.f, 0.f, 0.f, 0.f};
    #pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
            const long m = m0 + wm * 64 + mi * 16 + frow0 + j;
            if (m >= M) continue;
            long obase;
            if (MODE != CONV_DGRAD) {
                obase = m * OC;
            } else {
                const int ww = (int)(m % OW);
                const int hh = (int)((m / OW) % OH);
                const int n = (int)(m / ((long)OW * OH));
                obase = (((long)n * H + a + (long)sy * hh) * W + b +
                         (long)sx * ww) * C;
            }
            if (skpart != nullptr) {
                // split-K: fp32 partials, linear by m (the dgrad split path
                // is stride-1 only, where obase == m*OC)
                float* prow = skpart + ((long)sk * M + m) * OC + n0 + wn * 64;
                #pragma unroll
                for (int ni = 0; ni < 4; ++ni)
                    prow[ni * 16 + fcol] = acc[mi][ni][j];
                continue;
            }
            unsigned short* orow = out + obase + n0 + wn * 64;
            const unsigned short* arow =
                accsrc ? accsrc + obase + n0 + wn * 64 : nullptr;
            #pragma unroll
            for (int ni = 0; ni < 4; ++ni) {
                float v = acc[mi][ni][j];
                if (arow) v += bf16bits_to_f32(arow[ni * 16 + fcol]);
                const unsigned short us = f32_to_bf16bits(v);
                orow[ni * 16 + fcol] = us;
                if (MODE != CONV_DGRAD && stats != nullptr) {
                    const float v = bf16bits_to_f32(us);
                    ssum[ni] += v;
                    sq[ni] += v * v;
                }
            }
        }
    }
    if (skpart != nullptr && cnt != nullptr) {
        // ---- in-launch split-K seam (guide §6 Guideline 16, counter form):
        // publish this slice's fp32 slab with an agent-scope release and
        // take a ticket; the last-arriving slice re-reads all SK slabs,
        // writes the bf16 tile and accumulates the BN stats partials.
        // Saves the separate combine kernel's launch boundary + the slab
        // round trip on shallow-K small-M shapes, which is why split-K can
        // trigger there at all (the round-trip combine measured a net loss
        // below T=48).
        const long tileid =
            ((long)zrest * gridDim.y + blockIdx.y) * gridDim.x + blockIdx.x;
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __syncthreads();
        unsigned* flag = (unsigned*)lds;      // reuse the ONE shared array
        if (threadIdx.x == 0) {
            __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
            // restated wait: ROCm 7.2 drops the post-wbl2 vmcnt when its
            // scoreboard says this wave has nothing outstanding
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
            const unsigned t = __hip_atomic_fetch_add(
                &cnt[tileid], 1u, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
            flag[0] = (t == (unsigned)SK - 1) ? 1u : 0u;
        }
        __syncthreads();
        const bool im_last = flag[0] != 0;
        __syncthreads();          // flag consumed before any lds reuse
        if (!im_last) return;
        if (threadIdx.x == 0)
            __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
        __syncthreads();
        #pragma unroll
        for (int mi = 0; mi < 4; ++mi) {
            #pragma unroll
            for (int j = 0; j < 4; ++j) {
                const long m = m0 + wm * 64 + mi * 16 + frow0 + j;
                if (m >= M) continue;
                long obase;
                if (MODE != CONV_DGRAD) {
                    obase = m * OC;
                } else {
                    const int ww = (int)(m % OW);
                    const int hh = (int)((m / OW) % OH);
                    const int n = (int)(m / ((long)OW * OH));
                    obase = (((long)n * H + a + (long)sy * hh) * W + b +
                             (long)sx * ww) * C;
                }
                const float* prow = skpart + m * OC + n0 + wn * 64;
                unsigned short* orow = out + obase + n0 + wn * 64;
                #pragma unroll
                for (int ni = 0; ni < 4; ++ni) {
                    float v = 0.f;
                    for (int z = 0; z < SK; ++z)
                        v += prow[(long)z * M * OC + ni * 16 + fcol];
                    const unsigned short us = f32_to_bf16bits(v);
                    orow[ni * 16 + fcol] = us;
                    if (MODE != CONV_DGRAD && stats != nullptr) {
                        const float vr = bf16bits_to_f32(us);
                        ssum[ni] += vr;
                        sq[ni] += vr * vr;
                    }
                }
            }
        }
    }

    if (MODE != CONV_DGRAD && stats != nullptr) {
        // fold the four 16-row lane groups, then the waves sharing this
        // channel column, then write this m-tile's [2][OC] partial slice
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
            #pragma unroll
            for (int off = 16; off < 64; off <<= 1) {
                ssum[ni] += __shfl_xor(ssum[ni], off, 64);
                sq[ni] += __shfl_xor(sq[ni], off, 64);
            }
        }
        float* sf = (float*)lds;             // staging LDS is free now
        __builtin_amdgcn_s_barrier();
        if ((lane >> 4) == 0) {
            #pragma unroll
            for (int ni = 0; ni < 4; ++ni) {
                sf[(wid * 4 + ni) * 16 + fcol] = ssum[ni];
                sf[NW * 64 + (wid * 4 + ni) * 16 + fcol] = sq[ni];
            }
        }
        __builtin_amdgcn_s_barrier();
        if (wm == 0 && (lane >> 4) == 0) {
            float* prow = stats + (long)blockIdx.x * 2 * OC + n0 + wn * 64;
            #pragma unroll
            for (int ni = 0; ni < 4; ++ni) {
                float s = ssum[ni], z = sq[ni];
                for (int w2 = wid + WN; w2 < NW; w2 += WN) {
                    s += sf[(w2 * 4 + ni) * 16 + fcol];
                    z += sf[NW * 64 + (w2 * 4 + ni) * 16 + fcol];
                }
                prow[ni * 16 + fcol] = s;
                prow[OC + ni * 16 + fcol] = z;
            }
        }
    }
}

template <int MODE, int BM, int BN, int WN, int NBUF = 2, int BKT = BK>
static void launch_cfg(const void* src, const void* wgt, void* out,
                       int N, int H, int W, int C, int K, int P, int Q,
                       int R, int S, int sy, int sx, int py, int px,
                       hipStream_t stream, float* stats = nullptr,
                       float* skpart = nullptr, int SK = 1,
                       unsigned* cnt = nullptr,
                       const void* accsrc = nullptr) {
    const int OC = (MODE == CONV_DGRAD) ? C : K;
    const long M = (MODE != CONV_DGRAD)
        ? (long)N * P * Q
        : (long)N * ((H + sy - 1) / sy) * ((W + sx - 1) / sx);
    const unsigned zbase = (MODE != CONV_DGRAD) ? 1u : (unsigned)(sy * sx);
    constexpr unsigned NTHREADS = (BM / 64) * (BN / 64) * 64;
    dim3 grid((unsigned)((M + BM - 1) / BM), (unsigned)(OC / BN),
              zbase * (unsigned)SK);
    const size_t shmem = NBUF * (BM * BKT + BN * BKT) * sizeof(unsigned short);
    if (shmem > 65536) {
        static bool raised = [] {
            hipFuncSetAttribute(
                (const void*)&conv_igemm_kernel<MODE, BM, BN, WN, NBUF, BKT>,
                hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
            return true;
        }();
        (void)raised;
    }
    hipLaunchKernelGGL((conv_igemm_kernel<MODE, BM, BN, WN, NBUF, BKT>), grid,
                       dim3(NTHREADS), shmem, stream,
                       (const unsigned short*)src,
                       (const unsigned short*)wgt, (unsigned short*)out,
                       N, H, W, C, K, P, Q, R, S, sy, sx, py, px, stats,
                       skpart, SK, cnt, (const unsigned short*)accsrc);
}

// ---- split-K combine: y = bf16(sum_sk part) (+ BN stats partials) --------
__global__ __launch_bounds__(256) void conv_skcombine_kernel(
    const float* __restrict__ part, unsigned short* __restrict__ y,
    float* __restrict__ stats, long M, int OC, int SK,
    const unsigned short* __restrict__ accsrc) {
    __shared__ float smem[2048];          // 256 threads x 8 lanes
    const int V = 8;
    const int tpr = OC / V;               // threads per row (OC <= 2048)
    const int rpb = 256 / tpr;
    const int lane_c = threadIdx.x % tpr;
    const int sub_r = threadIdx.x / tpr;
    const int c0 = lane_c * V;
    float s[V], q[V];
    #pragma unroll
    for (int k = 0; k < V; ++k) s[k] = q[k] = 0.f;
    for (long r = (long)blockIdx.x * rpb + sub_r; r < M;
         r += (long)gridDim.x * rpb) {
        float v[V];
        #pragma unroll
        for (int k = 0; k < V; ++k) v[k] = 0.f;
        for (int z = 0; z < SK; ++z) {
            const float* p = part + ((long)z * M + r) * OC + c0;
            float4 a = *(const float4*)p;
            float4 b = *(const float4*)(p + 4);
            v[0] += a.x; v[1] += a.y; v[2] += a.z; v[3] += a.w;
            v[4] += b.x; v[5] += b.y; v[6] += b.z; v[7] += b.w;
        }
        if (accsrc != nullptr) {
            unsigned short av[V];
            *(uint4*)av = *(const uint4*)(accsrc + r * OC + c0);
            #pragma unroll
            for (int k = 0; k < V; ++k) v[k] += bf16bits_to_f32(av[k]);
        }
        unsigned short o[V];
        #pragma unroll
        for (int k = 0; k < V; ++k) {
            o[k] = f32_to_bf16bits(v[k]);
            if (stats != nullptr) {
                const float vr = bf16bits_to_f32(o[k]);
                s[k] += vr;
                q[k] += vr * vr;
            }
        }
        *(uint4*)(y + r * OC + c0) = *(uint4*)o;
    }
    if (stats == nullptr) return;
    // fold the rpb row-groups, write this block's [2][OC] partial slice
    float* outp = stats + (long)blockIdx.x * 2 * OC;
    #pragma unroll
    for (int pass = 0; pass < 2; ++pass) {
        float* loc = pass == 0 ? s : q;
        #pragma unroll
        for (int k = 0; k < V; ++k) smem[threadIdx.x * V + k] = loc[k];
        __syncthreads();
        if (sub_r == 0) {
            float acc[V];
            #pragma unroll
            for (int k = 0; k < V; ++k) acc[k] = 0.f;
            for (int rr = 0; rr < rpb; ++rr) {
                const float* sp = smem + (rr * tpr + lane_c) * V;
                #pragma unroll
                for (int k = 0; k < V; ++k) acc[k] += sp[k];
            }
            #pragma unroll
            for (int k = 0; k < V; ++k)
                outp[pass * OC + lane_c * V + k] = acc[k];
        }
        __syncthreads();
    }
}

int conv_skcombine_blocks(long M, int OC) {
    const int rpb = 256 / (OC / 8);
    long b = (M + rpb - 1) / rpb;
    return (int)(b < 512 ? b : 512);
}

void conv_skcombine_launch(const float* part, void* y, float* stats, long M,
                           int OC, int SK, int nblocks, hipStream_t stream,
                           const void* accsrc) {
    hipLaunchKernelGGL(conv_skcombine_kernel, dim3(nblocks), dim3(256), 0,
                       stream, part, (unsigned short*)y, stats, M, OC, SK,
                       (const unsigned short*)accsrc);
}


// ---- 8-phase fine-interleave variant (guide §5 "256² 8-phase template",
// adapted to the conv gather): 256x128 tile, 8 waves (4M x 2N), 512
// threads, 1 block/CU, 3 LDS buffers (144 KiB). Each K-step runs as 4
// phases of [ds-read subtile | stage chunk of tile t+2 | raw barrier |
// lgkmcnt(0) | 8 MFMAs | raw barrier]; the tile drain is ONE counted
// s_waitcnt vmcnt(12) per K-step (2 newer tiles x 6 glds/wave stay in
// flight), so DMA latency spans ~4-8 phases of MFMA work instead of
// stalling at a single per-step barrier. FLUXDIST_CONV8 gates dispatch.
template <int MODE>
__global__ __launch_bounds__(512, 1) void conv_igemm8_kernel(
    const unsigned short* __restrict__ src,
    const unsigned short* __restrict__ wgt,
    unsigned short* __restrict__ out,
    int N, int H, int W, int C,
    int K, int P, int Q,
    int R, int S, int sy, int sx, int py, int px,
    float* __restrict__ stats,
    const unsigned short* __restrict__ accsrc) {
    constexpr int BM = 256, BN = 128, NBUF = 3;
    constexpr int BKT = BK;              // this variant is BK=64 only
    constexpr int RPG = 8, GPR = 8;      // rows / 16-B granules per glds
    constexpr int KH = 2;                // mfma k-halves (BK=64)
    auto swz = [](int row) { return row & 7; };
    constexpr int A_ELEMS = BM * BKT;
    constexpr int B_ELEMS = BN * BKT;
    constexpr int BUF_ELEMS = A_ELEMS + B_ELEMS;
    constexpr int AI = 4, BI = 2;        // glds per wave per tile
    constexpr int NW = 8, WN = 2;

    const int OC = (MODE == CONV_DGRAD) ? C : K;
    const int RC = (MODE == CONV_FWD) ? C : K;

    int a = 0, b = 0, OH, OW, r0 = 0, s0 = 0, nR = R, nS = S;
    if (MODE != CONV_DGRAD) {
        OH = P; OW = Q;
    } else {
        a = blockIdx.z / sx;  b = blockIdx.z % sx;
        OH = (H - a + sy - 1) / sy;
        OW = (W - b + sx - 1) / sx;
        r0 = (a + py) % sy;  nR = (R - r0 + sy - 1) / sy;
        s0 = (b + px) % sx;  nS = (S - s0 + sx - 1) / sx;
        if (OH <= 0 || OW <= 0) return;
        if (nR < 0) nR = 0;
        if (nS < 0) nS = 0;
    }
    const long M = (long)N * OH * OW;
    const long m0 = (long)blockIdx.x * BM;
    if (m0 >= M) return;
    const int n0 = blockIdx.y * BN;

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int wm = wid >> 1;
    const int wn = wid & 1;

    extern __shared__ unsigned short lds[];   // [NBUF][BUF_ELEMS]

    // ---- per-lane staging descriptors (same gather as conv_igemm_kernel)
    int a_row[AI];
    long a_pix[AI];
    int a_hb[AI], a_wb[AI];
    bool a_mok[AI];
    const int cslot = lane % GPR;
    #pragma unroll
    for (int i = 0; i < AI; ++i) {
        const int row = (wid * AI + i) * RPG + lane / GPR;
        a_row[i] = row;
        const long m = m0 + row;
        const bool mok = m < M;
        const long mm = mok ? m : 0;
        const int ow = (int)(mm % OW);
        const int oh = (int)((mm / OW) % OH);
        const int n = (int)(mm / ((long)OW * OH));
        a_mok[i] = mok;
        if (MODE == CONV_FWD) {
            a_hb[i] = oh * sy - py;
            a_wb[i] = ow * sx - px;
            a_pix[i] = ((long)n * H) * W * C;
        } else {
            a_hb[i] = oh + (a + py) / sy;
            a_wb[i] = ow + (b + px) / sx;
            a_pix[i] = ((long)n * P) * Q * K;
        }
    }
    int b_row[BI];
    #pragma unroll
    for (int i = 0; i < BI; ++i)
        b_row[i] = (wid * BI + i) * RPG + lane / GPR;

    const int cblocks = RC / BKT;
    const int T = nR * nS * cblocks;

    long a_goff[AI];
    bool a_okc[AI];
    long b_goff[BI];
    int last_rsi = -1;
    auto advance = [&](int it) {
        const int rsi = it / cblocks;
        const int ri = rsi / nS, si = rsi % nS;
        if (rsi != last_rsi) {
            last_rsi = rsi;
            const int cb = (it % cblocks) * BK;
            #pragma unroll
            for (int i = 0; i < AI; ++i) {
                bool ok = a_mok[i];
                long off = 0;
                if (MODE == CONV_FWD) {
                    const int h = a_hb[i] + ri, w = a_wb[i] + si;
                    ok = ok && (unsigned)h < (unsigned)H && (unsigned)w < (unsigned)W;
                    off = a_pix[i] + ((long)h * W + w) * C + cb + (cslot ^ (a_row[i] & 7)) * 8;
                } else {
                    const int pp = a_hb[i] - ri, qq = a_wb[i] - si;
                    ok = ok && (unsigned)pp < (unsigned)P && (unsigned)qq < (unsigned)Q;
                    off = a_pix[i] + ((long)pp * Q + qq) * K + cb + (cslot ^ (a_row[i] & 7)) * 8;
                }
                a_goff[i] = off;
                a_okc[i] = ok;
            }
            const int r = r0 + ri * ((MODE == CONV_FWD) ? 1 : sy);
            const int sss = s0 + si * ((MODE == CONV_FWD) ? 1 : sx);
            const int rs = r * S + sss;
            #pragma unroll
            for (int i = 0; i < BI; ++i) {
                const int row = b_row[i];
                const int cs = (cslot ^ (row & (GPR - 1))) * 8;
                if (MODE == CONV_FWD)
                    b_goff[i] = (long)(n0 + row) * R * S * C + (long)rs * C + cb + cs;
                else
                    b_goff[i] = (long)((long)rs * C + n0 + row) * K + cb + cs;
            }
        } else {
            #pragma unroll
            for (int i = 0; i < AI; ++i) a_goff[i] += BK;
            #pragma unroll
            for (int i = 0; i < BI; ++i) b_goff[i] += BK;
        }
    };

    // fragment read offsets
    int a_off[4][KH], b_off[4][KH];
    {
        const int fr = lane & 15, fq = lane >> 4;
        #pragma unroll
        for (int mi = 0; mi < 4; ++mi)
            #pragma unroll
            for (int kh = 0; kh < KH; ++kh) {
                const int row = wm * 64 + mi * 16 + fr;
                const int slot = ((kh * 4 + fq) ^ swz(row)) & (GPR - 1);
                a_off[mi][kh] = row * BKT + slot * 8;
            }
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni)
            #pragma unroll
            for (int kh = 0; kh < KH; ++kh) {
                const int row = wn * 64 + ni * 16 + fr;
                const int slot = ((kh * 4 + fq) ^ swz(row)) & (GPR - 1);
                b_off[ni][kh] = A_ELEMS + row * BKT + slot * 8;
            }
    }

    floatx4 acc[4][4];
    #pragma unroll
    for (int mi = 0; mi < 4; ++mi)
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni) acc[mi][ni] = floatx4{0.f, 0.f, 0.f, 0.f};

    auto issueA = [&](int buf, int i) {
        unsigned short* base = lds + buf * BUF_ELEMS;
        const unsigned short* sp = a_okc[i] ? src + a_goff[i] : conv_zero16;
        FDA_GLDS16(sp, base + (wid * AI + i) * 8 * BK);
    };
    auto issueB = [&](int buf, int i) {
        unsigned short* base = lds + buf * BUF_ELEMS;
        FDA_GLDS16(wgt + b_goff[i], base + A_ELEMS + (wid * BI + i) * 8 * BK);
    };
    auto stage_all = [&](int buf, int it) {   // prologue only
        advance(it);
        #pragma unroll
        for (int i = 0; i < AI; ++i) issueA(buf, i);
        #pragma unroll
        for (int i = 0; i < BI; ++i) issueB(buf, i);
    };

    if (T > 0) stage_all(0, 0);
    if (T > 1) stage_all(1, 1);

    for (int it = 0; it < T; ++it) {
        const int buf = it % NBUF;
        const unsigned short* rbuf = lds + buf * BUF_ELEMS;
        const int sbuf = (it + 2) % NBUF;
        const bool do_stage = it + 2 < T;
        if (do_stage) advance(it + 2);
        // drain THIS tile's 6 glds; leave the 2 newer tiles (<=12) in flight
        if (it + 1 < T)
            asm volatile("s_waitcnt vmcnt(12)" ::: "memory");
        else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();

        short8 af0[4], af1[4], bf[4][2];
        // ---- phase 0: kh=0 (16 MFMAs) ----------------------------------
        // 2 phases of 16 MFMAs: the 4x8-MFMA grain measured ~10% SLOWER
        // (per-phase barrier cost not amortized at this tile size;
        // profiles/ab_conv8.md).
        #pragma unroll
        for (int mi = 0; mi < 4; ++mi)
            af0[mi] = *(const short8*)(rbuf + a_off[mi][0]);
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni)
            bf[ni][0] = *(const short8*)(rbuf + b_off[ni][0]);
        if (do_stage) { issueA(sbuf, 0); issueA(sbuf, 1); issueA(sbuf, 2); }
        __builtin_amdgcn_s_barrier();
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_setprio(1);
        #pragma unroll
        for (int mi = 0; mi < 4; ++mi)
            #pragma unroll
            for (int ni = 0; ni < 4; ++ni)
                acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    af0[mi], bf[ni][0], acc[mi][ni], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
        __builtin_amdgcn_s_barrier();
        // ---- phase 1: kh=1 (16 MFMAs) ----------------------------------
        #pragma unroll
        for (int mi = 0; mi < 4; ++mi)
            af1[mi] = *(const short8*)(rbuf + a_off[mi][1]);
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni)
            bf[ni][1] = *(const short8*)(rbuf + b_off[ni][1]);
        if (do_stage) { issueA(sbuf, 3); issueB(sbuf, 0); issueB(sbuf, 1); }
        __builtin_amdgcn_s_barrier();
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_setprio(1);
        #pragma unroll
        for (int mi = 0; mi < 4; ++mi)
            #pragma unroll
            for (int ni = 0; ni < 4; ++ni)
                acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    af1[mi], bf[ni][1], acc[mi][ni], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
        __builtin_amdgcn_s_barrier();
    }

    // ---- epilogue (same mapping as conv_igemm_kernel) --------------------
    const int fcol = lane & 15, frow0 = (lane >> 4) * 4;
    float ssum[4] = {0.f, 0.f, 0.f, 0.f};
    float sq[4] = {0.f, 0.f, 0.f, 0.f};
    #pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
            const long m = m0 + wm * 64 + mi * 16 + frow0 + j;
            if (m >= M) continue;
            long obase;
            if (MODE != CONV_DGRAD) {
                obase = m * OC;
            } else {
                const int ww = (int)(m % OW);
                const int hh = (int)((m / OW) % OH);
                const int n = (int)(m / ((long)OW * OH));
                obase = (((long)n * H + a + (long)sy * hh) * W + b +
                         (long)sx * ww) * C;
            }
            unsigned short* orow = out + obase + n0 + wn * 64;
            const unsigned short* arow =
                accsrc ? accsrc + obase + n0 + wn * 64 : nullptr;
            #pragma unroll
            for (int ni = 0; ni < 4; ++ni) {
                float v = acc[mi][ni][j];
                if (arow) v += bf16bits_to_f32(arow[ni * 16 + fcol]);
                const unsigned short us = f32_to_bf16bits(v);
                orow[ni * 16 + fcol] = us;
                if (MODE != CONV_DGRAD && stats != nullptr) {
                    const float vr = bf16bits_to_f32(us);
                    ssum[ni] += vr;
                    sq[ni] += vr * vr;
                }
            }
        }
    }
    if (MODE != CONV_DGRAD && stats != nullptr) {
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
            #pragma unroll
            for (int off = 16; off < 64; off <<= 1) {
                ssum[ni] += __shfl_xor(ssum[ni], off, 64);
                sq[ni] += __shfl_xor(sq[ni], off, 64);
            }
        }
        float* sf = (float*)lds;
        __builtin_amdgcn_s_barrier();
        if ((lane >> 4) == 0) {
            #pragma unroll
            for (int ni = 0; ni < 4; ++ni) {
                sf[(wid * 4 + ni) * 16 + fcol] = ssum[ni];
                sf[NW * 64 + (wid * 4 + ni) * 16 + fcol] = sq[ni];
            }
        }
        __builtin_amdgcn_s_barrier();
        if (wm == 0 && (lane >> 4) == 0) {
            float* prow = stats + (long)blockIdx.x * 2 * OC + n0 + wn * 64;
            #pragma unroll
            for (int ni = 0; ni < 4; ++ni) {
                float s2 = ssum[ni], z = sq[ni];
                for (int w2 = wid + WN; w2 < NW; w2 += WN) {
                    s2 += sf[(w2 * 4 + ni) * 16 + fcol];
                    z += sf[NW * 64 + (w2 * 4 + ni) * 16 + fcol];
                }
                prow[ni * 16 + fcol] = s2;
                prow[OC + ni * 16 + fcol] = z;
            }
        }
    }
}

int conv8_enabled() {
    // 8-phase variant A/B knob: FLUXDIST_CONV8=1 routes eligible shapes
    // (OC%128, SK==1, grid fills at 1 block/CU) through conv_igemm8_kernel.
    static int v = [] {
        const char* e = getenv("FLUXDIST_CONV8");
        return e ? atoi(e) : 0;
    }();
    return v;
}

template <int MODE>
static void launch_cfg8(const void* src, const void* wgt, void* out,
                        int N, int H, int W, int C, int K, int P, int Q,
                        int R, int S, int sy, int sx, int py, int px,
                        hipStream_t stream, float* stats,
                        const void* accsrc) {
    const int OC = (MODE == CONV_DGRAD) ? C : K;
    const long M = (MODE != CONV_DGRAD)
        ? (long)N * P * Q
        : (long)N * ((H + sy - 1) / sy) * ((W + sx - 1) / sx);
    const unsigned zbase = (MODE != CONV_DGRAD) ? 1u : (unsigned)(sy * sx);
    dim3 grid((unsigned)((M + 255) / 256), (unsigned)(OC / 128), zbase);
    const size_t shmem = 3 * (256 * BK + 128 * BK) * sizeof(unsigned short);
    static bool raised = [] {
        hipFuncSetAttribute((const void*)&conv_igemm8_kernel<MODE>,
                            hipFuncAttributeMaxDynamicSharedMemorySize,
                            160 * 1024);
        return true;
    }();
    (void)raised;
    hipLaunchKernelGGL((conv_igemm8_kernel<MODE>), grid, dim3(512), shmem,
                       stream, (const unsigned short*)src,
                       (const unsigned short*)wgt, (unsigned short*)out,
                       N, H, W, C, K, P, Q, R, S, sy, sx, py, px, stats,
                       (const unsigned short*)accsrc);
}

void conv_stem_fwd_launch(const void* src, const void* wgt, void* out,
                          int N, int Hp, int Wp, int K, int P, int Q,
                          int R, int sy, int sx, hipStream_t stream,
                          float* stats) {
    // src: channel-padded (C=8) spatially pre-padded input [N,Hp,Wp,8];
    // wgt: wpad [K][R][64]; out: [N*P*Q][K]
    launch_cfg<CONV_STEM, 256, 64, 1>(src, wgt, out, N, Hp, Wp, /*C=*/8, K,
                                      P, Q, R, /*S=*/1, sy, sx, 0, 0, stream,
                                      stats);
}

static int conv_nbuf() {
    // A/B knob: FLUXDIST_CONV_NBUF=3 -> 3-buffer counted ring (1 block/CU
    // for the 128x128 config; 96 KiB LDS) vs default 2 (2 blocks/CU).
    static int v = [] {
        const char* e = getenv("FLUXDIST_CONV_NBUF");
        return (e && e[0] == '3') ? 3 : 2;
    }();
    return v;
}

int conv_bk32_knob() {
    // BK=32 x NBUF=3 for the 128x128 config: 48 KB LDS/block -> 3
    // blocks/CU (12 waves) vs the default's 2. Measured (r2 A/B,
    // profiles/ab_conv8.md addendum): -9..-28% on the big-M layer-1/2
    // legs, +17% on the small-M 14x14 legs (294 blocks can't feed 3
    // blocks/CU), so dispatch is grid-size-gated below. 0 = never,
    // 1 = gated by fill (default), 2 = force everywhere (A/B).
    static int v = [] {
        const char* e = getenv("FLUXDIST_CONV_BK32");
        return e ? atoi(e) : 1;
    }();
    return v;
}

int conv_bk32_check(long blocks64);

int conv_bk32_check(long blocks64) {
    // 128x64 x BK32 gate — measured a large LOSS on layer 1 (fwd 61->99,
    // dgrad 76->100 us: the 2-wave block drops the MFMA:glds ratio to
    // 2.7:1 and these legs are staging-bandwidth-bound). Only the forced
    // A/B mode (FLUXDIST_CONV_BK32=2) takes it; the 128x128 BK32 config
    // keeps its own fill gate (the measured +3.8% win).
    extern int conv_bk32_knob();
    (void)blocks64;
    return conv_bk32_knob() >= 2;
}

static int conv_bigtile() {
    // 256x128 512-thread tile (8 waves, 6 glds/wave/K-step at the same 32
    // MFMAs). A/B knob FLUXDIST_CONV_BIGTILE: 0 = never, 1 = whenever
    // OC%128==0 and the grid still fills at 1 block/CU. Default 0: the r2
    // microbench measured it 3-13% SLOWER on every eligible leg (block-
    // level overlap lost at 1 block/CU — same mechanism as NBUF=3,
    // profiles/ab_bigtile.md).
    static int v = [] {
        const char* e = getenv("FLUXDIST_CONV_BIGTILE");
        return e ? atoi(e) : 0;
    }();
    return v;
}

int conv8_enabled();   // defined below (8-phase variant knob)

bool conv_use_inlsk() {
    // in-launch split-K combine (last-arriver seam) vs the separate
    // combine kernel. Default OFF: measured 2-2.6x SLOWER than the
    // round-trip combine on every SK shape (r2 A/B, profiles/ab_inlsk.md)
    // — the per-block agent-release fence (buffer_wbl2 L2 writeback) and
    // the reducer's serial slab re-read dwarf the saved launch boundary at
    // these 64 KB/tile slab sizes. Kept behind FLUXDIST_CONV_INLSK=1 for
    // future shapes with small slabs.
    static bool v = [] {
        const char* e = getenv("FLUXDIST_CONV_INLSK");
        return e && e[0] == '1';
    }();
    return v;
}

// One place that decides tile geometry + split-K for a conv launch; the
// torch bindings call this too so the stats/skpart workspace shapes always
// match what the kernel will write.
//   M   = output rows (per z-class for dgrad), OC = output channels,
//   T   = K-loop depth in BK=64 steps (nR*nS*RC/64), zbase = sy*sx classes
//         for strided dgrad else 1.
// split-K only when the launch would underfill the 256-CU chip AND the
// K-loop is deep enough to amortize the fp32 partial round-trip (T < 48
// measured a net loss: 14x14 / 1x1 shapes regressed up to 3x on a blanket
// trigger).
void conv_igemm_plan(long M, int OC, long T, int zbase,
                     int* bm, int* bn, int* sk) {
    const bool big = OC % 128 == 0;
    int BM = big ? 128 : 256;
    int BN = big ? 128 : 64;
    int SK = 1;
    // T >= 48 only: shallow-K shapes measured a net loss under split-K on
    // BOTH combine flavors (round-trip r1.14; in-launch seam r2 A/B — the
    // 4-6x slab traffic dominates at shallow T, these legs are
    // traffic-bound, not fill-bound).
    if (T >= 48) {
        const long blocks = ((M + BM - 1) / BM) * (OC / BN) * zbase;
        if (blocks < 192) SK = 4;
        else if (blocks < 384) SK = 2;
    }
    if (big && SK == 1 && (conv_bigtile() || conv8_enabled())) {
        const long blocks256 = ((M + 255) / 256) * (OC / 128) * zbase;
        if (blocks256 >= 256) { BM = 256; BN = 128; }
    }
    if (!big) {
        // 64-channel shapes may take the 128x64 x BK32 config (4 blocks/
        // CU); BM=128 here so the stats workspace matches the launch
        const long blocks64 = ((M + 127) / 128) * (OC / 64) * zbase * SK;
        extern int conv_bk32_check(long blocks64);
        if (conv_bk32_check(blocks64)) BM = 128;
    }
    *bm = BM; *bn = BN; *sk = SK;
}

void conv_igemm_launch(const void* src, const void* wgt, void* out,
                       int N, int H, int W, int C, int K, int P, int Q,
                       int R, int S, int sy, int sx, int py, int px,
                       bool dgrad, hipStream_t stream, float* stats,
                       float* skpart, int SK, unsigned* cnt,
                       const void* accsrc) {
    const int OC = dgrad ? C : K;
    const bool nb3 = conv_nbuf() == 3;
    const long Mv = !dgrad
        ? (long)N * P * Q
        : (long)N * ((H + sy - 1) / sy) * ((W + sx - 1) / sx);
    const int zbase = dgrad ? sy * sx : 1;
    const long RC = dgrad ? K : C;
    // T here mirrors the bindings' SK trigger; the plan's BM/BN choice is
    // what this function must obey so workspace shapes match
    int BM, BN, SKp;
    conv_igemm_plan(Mv, OC, (long)R * S * (RC / 64), zbase, &BM, &BN, &SKp);
    const bool big = BN == 128;
    if (BM == 256 && BN == 128) {
        if (conv8_enabled()) {     // 8-phase fine-interleave variant
            if (dgrad)
                launch_cfg8<CONV_DGRAD>(src, wgt, out, N, H, W, C, K, P, Q,
                                        R, S, sy, sx, py, px, stream,
                                        nullptr, accsrc);
            else
                launch_cfg8<CONV_FWD>(src, wgt, out, N, H, W, C, K, P, Q,
                                      R, S, sy, sx, py, px, stream, stats,
                                      accsrc);
            return;
        }
        if (dgrad)
            launch_cfg<CONV_DGRAD, 256, 128, 2, 2>(
                src, wgt, out, N, H, W, C, K, P, Q, R, S, sy, sx, py, px,
                stream, nullptr, skpart, SK, cnt, accsrc);
        else
            launch_cfg<CONV_FWD, 256, 128, 2, 2>(
                src, wgt, out, N, H, W, C, K, P, Q, R, S, sy, sx, py, px,
                stream, stats, skpart, SK, cnt, accsrc);
        return;
    }
    // the 3-blocks/CU BK32 config needs >=~576 blocks to fill; below
    // that the lost per-block depth outweighs the extra block overlap
    const long blocks128 =
        ((Mv + 127) / 128) * (OC / 128) * (long)zbase * SK;
    const bool bk32 = big && conv_bk32_knob() &&
                      (conv_bk32_knob() >= 2 || blocks128 >= 576);
    // 64-channel shapes (layer 1): 128x64 x BK32 x NBUF3 = 36 KB LDS ->
    // 4 blocks/CU of 2 waves; the plan already switched BM to 128 when
    // the grid qualifies (stats workspace sizing must match)
    const bool bk32s = !big && BM == 128;
    // 256x64 x BK32 x 3-ring: SAME 2 blocks/CU and the same 3.2:1
    // MFMA:glds ratio as the BK64 default, but with a second tile in
    // flight (60 KB x 2 fits; 3 x 80 KB at BK64 never did)
    const long blocks256x64 =
        ((Mv + 255) / 256) * (OC / 64) * (long)zbase * SK;
    const bool bk32r = !big && BM == 256 && conv_bk32_knob() &&
                       (conv_bk32_knob() >= 2 || blocks256x64 >= 512);
    if (dgrad) {
        if (bk32)
            launch_cfg<CONV_DGRAD, 128, 128, 2, 3, 32>(
                src, wgt, out, N, H, W, C, K, P, Q, R, S, sy, sx, py, px,
                stream, nullptr, skpart, SK, cnt, accsrc);
        else if (big && nb3)
            launch_cfg<CONV_DGRAD, 128, 128, 2, 3>(src, wgt, out, N, H, W, C,
                                                   K, P, Q, R, S, sy, sx, py,
                                                   px, stream, nullptr,
                                                   skpart, SK, cnt, accsrc);
        else if (big)
            launch_cfg<CONV_DGRAD, 128, 128, 2>(src, wgt, out, N, H, W, C, K,
                                                P, Q, R, S, sy, sx, py, px,
                                                stream, nullptr, skpart, SK,
                                                cnt, accsrc);
        else if (bk32s)
            launch_cfg<CONV_DGRAD, 128, 64, 1, 3, 32>(
                src, wgt, out, N, H, W, C, K, P, Q, R, S, sy, sx, py, px,
                stream, nullptr, skpart, SK, cnt, accsrc);
        else if (bk32r)
            launch_cfg<CONV_DGRAD, 256, 64, 1, 3, 32>(
                src, wgt, out, N, H, W, C, K, P, Q, R, S, sy, sx, py, px,
                stream, nullptr, skpart, SK, cnt, accsrc);
        else
            launch_cfg<CONV_DGRAD, 256, 64, 1>(src, wgt, out, N, H, W, C, K,
                                               P, Q, R, S, sy, sx, py, px,
                                               stream, nullptr, skpart, SK,
                                               cnt, accsrc);
    } else {
        if (bk32)
            launch_cfg<CONV_FWD, 128, 128, 2, 3, 32>(
                src, wgt, out, N, H, W, C, K, P, Q, R, S, sy, sx, py, px,
                stream, stats, skpart, SK, cnt, accsrc);
        else if (big && nb3)
            launch_cfg<CONV_FWD, 128, 128, 2, 3>(src, wgt, out, N, H, W, C,
                                                 K, P, Q, R, S, sy, sx, py,
                                                 px, stream, stats, skpart,
                                                 SK, cnt);
        else if (big)
            launch_cfg<CONV_FWD, 128, 128, 2>(src, wgt, out, N, H, W, C, K,
                                              P, Q, R, S, sy, sx, py, px,
                                              stream, stats, skpart, SK,
                                              cnt);
        else if (bk32s)
            launch_cfg<CONV_FWD, 128, 64, 1, 3, 32>(
                src, wgt, out, N, H, W, C, K, P, Q, R, S, sy, sx, py, px,
                stream, stats, skpart, SK, cnt, accsrc);
        else if (bk32r)
            launch_cfg<CONV_FWD, 256, 64, 1, 3, 32>(
                src, wgt, out, N, H, W, C, K, P, Q, R, S, sy, sx, py, px,
                stream, stats, skpart, SK, cnt, accsrc);
        else
            launch_cfg<CONV_FWD, 256, 64, 1>(src, wgt, out, N, H, W, C, K,
                                             P, Q, R, S, sy, sx, py, px,
                                             stream, stats, skpart, SK,
                                             cnt);
    }
}

}  // namespace fda

namespace fda {

// ---- batched conv-weight transpose ----------------------------------------
// One launch transposes every conv weight w[k][rs*C+c] -> wt[rs*C+c][k] for
// the dgrad B-tiles (replaces 36 per-layer permute kernels per step).
// Per tensor: 2D 64x64 LDS-tiled transpose, read coalesced along rc, write
// coalesced along k. All conv weights have K and RS*C multiples of 64.
__global__ __launch_bounds__(256) void wt_transpose_kernel(
    const long* __restrict__ src_ptrs,   // device addresses of w tensors
    long* __restrict__ dst_ptrs,         // device addresses of wt slices
    const int* __restrict__ Ks, const int* __restrict__ RCs,
    const int* __restrict__ tile_counts) {
    const int t = blockIdx.y;
    const int K = Ks[t], RC = RCs[t];
    const int kt = K / 64, rt = RC / 64;
    if ((int)blockIdx.x >= tile_counts[t]) return;
    const int tk = blockIdx.x % kt, trc = blockIdx.x / kt;
    (void)rt;
    const unsigned short* src = (const unsigned short*)src_ptrs[t];
    unsigned short* dst = (unsigned short*)dst_ptrs[t];

    __shared__ unsigned short tile[64][64 + 8];  // +8 bf16 pad: no bank dup
    const int tid = threadIdx.x;
    // read: 64 k-rows x 64 rc; thread reads 16 elems, 8-contig along rc
    #pragma unroll
    for (int i = 0; i < 2; ++i) {
        const int kk = (tid / 8) + i * 32;           // 0..63
        const int rc = (tid % 8) * 8;                // 0..56 step 8
        const unsigned short* sp =
            src + (long)(tk * 64 + kk) * RC + trc * 64 + rc;
        #pragma unroll
        for (int e = 0; e < 8; ++e) tile[kk][rc + e] = sp[e];
    }
    __syncthreads();
    // write: 64 rc-rows x 64 k; thread writes 16 elems, 8-contig along k
    #pragma unroll
    for (int i = 0; i < 2; ++i) {
        const int rc = (tid / 8) + i * 32;
        const int kk = (tid % 8) * 8;
        unsigned short* dp =
            dst + (long)(trc * 64 + rc) * K + tk * 64 + kk;
        #pragma unroll
        for (int e = 0; e < 8; ++e) dp[e] = tile[kk + e][rc];
    }
}

void wt_transpose_launch(const long* src_ptrs, long* dst_ptrs, const int* Ks,
                         const int* RCs, const int* tile_counts, int ntensors,
                         int max_tiles, hipStream_t stream) {
    dim3 grid((unsigned)max_tiles, (unsigned)ntensors);
    hipLaunchKernelGGL(wt_transpose_kernel, grid, dim3(256), 0, stream,
                       src_ptrs, dst_ptrs, Ks, RCs, tile_counts);
}

}  // namespace fda

namespace fda {

typedef __attribute__((ext_vector_type(4))) short short4_;

// ---- wgrad: dw[k][rs][c] = sum_m dy[m][k] * x_gather[m][c] ----------------
// GEMM with the reduction along the M (pixel) axis — both operands are
// [m][channel] in memory, so the MFMA fragments (which want 8 elements
// along the reduction per lane) are served by gfx950's ds_read_tr16_b64
// hardware transpose-read from an "m4-grouped" LDS image:
//   image element (m, ch) at  kb*1024 + (ch&15) + (m&3)*16 + (m>>2)*64
//   (kb = ch/16); built directly by global_load_lds (16-B chunks are
//   8-channel runs of one m row); read back with per-16-lane-group window
//   addressing: lane addr = kb*2048B + (ks*8 + (lane>>4)*2)*128B +
//   (lane&15)*8B, second half of the fragment at immediate offset +128B.
// Each block owns a (32*FT)k x (32*FT)c output tile (FT = fragments per
// wave axis: 2 -> 64x64 tile / 8 MFMAs per K-step, 4 -> 128x128 tile /
// 32 MFMAs — chosen by channel divisibility; the bigger tile quadruples
// the MFMA-per-glds ratio) for one (r,s) tap and a 2048-pixel M-chunk;
// chunks accumulate into an fp32 workspace with atomicAdd.
constexpr int WG_BM = 64;      // m per K-step
constexpr int WG_MCH = 2048;   // pixels per block (chunk)

template <int FT, bool STEM = false, bool ASMRD = false, int MB = WG_BM,
          bool TP = false>
__global__ __launch_bounds__(256, 2) void conv_wgrad_kernel(
    const unsigned short* __restrict__ dy,   // [M][K] (NHWC out grad)
    const unsigned short* __restrict__ x,    // [N,H,W,C]
    float* __restrict__ ws,  /* TP=false: [K][RS*C] fp32, pre-zeroed,
                                accumulated with atomicAdd across chunks.
                                TP=true (two-phase): [nch][K][RS*C] fp32
                                partial slab, uninitialized — every chunk
                                block plain-stores its full tile (PMC
                                showed 100% of the atomics go to DRAM as
                                serialized RMWs, docs/wgrad_study.md);
                                wgrad_combine_kernel folds the slab. */
    int N, int H, int W, int C, int K, int P, int Q,
    int R, int S, int sy, int sx, int py, int px, int nch, int mch) {
    constexpr int TCH = 32 * FT;             // tile channels per operand
    constexpr int TILE_ELEMS = MB * TCH;     // one operand tile
    constexpr int GI = MB * TCH / 2048;      // glds per wave per operand
    constexpr int NWIN = MB / 4;             // tr16 windows (4 m x 16 ch)
    constexpr int KSN = MB / 32;             // MFMA m-halves per K-step
    const int rs = blockIdx.z / nch;
    const int chunk = blockIdx.z % nch;
    const int r = rs / S, s = rs % S;
    const int k0 = blockIdx.x * TCH;
    const int c0 = blockIdx.y * TCH;
    const long M = (long)N * P * Q;
    const long mb0 = (long)chunk * mch;
    const long mend = (mb0 + mch < M) ? mb0 + mch : M;

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int wk = wid >> 1;          // 2x2 waves over [TCH k][TCH c]
    const int wc = wid & 1;

    extern __shared__ unsigned short lds[];   // [ring][2*TILE_ELEMS]

    const floatx4 zero4 = {0.f, 0.f, 0.f, 0.f};
    floatx4 acc[FT][FT];
    #pragma unroll
    for (int ki = 0; ki < FT; ++ki)
        #pragma unroll
        for (int ci = 0; ci < FT; ++ci) acc[ki][ci] = zero4;

    // Per-lane staging coordinates, advanced INCREMENTALLY: stage() is
    // called with strictly sequential m-bases (prologue 0,1 then it+2), so
    // each lane tracks its pixel (n,p,q) with constant-delta carries — no
    // per-iteration division (those were ~3x the MFMA issue time here).
    int st_q[GI], st_p[GI], st_n[GI], st_ch[GI];
    long st_dyoff[GI];
    long st_m[GI];
    #pragma unroll
    for (int i = 0; i < GI; ++i) {
        const int ln = (wid * GI + i) * 64 + lane;
        const int e8 = ln * 8;
        const int kb = e8 / (MB * 16);
        const int rr = e8 % (MB * 16);
        // window position -> window index: evens in the low half of the
        // positions, odds in the high half, so the two windows a
        // half-wave tr-reads simultaneously sit on different 128-B bank
        // halves (2-way conflict measured at 6.3% of wave cycles with
        // the linear layout)
        const int pos = rr >> 6;
        const int v = (pos < NWIN / 2) ? pos * 2
                                       : (pos - NWIN / 2) * 2 + 1;
        const int ml = (v << 2) + ((rr & 63) >> 4);
        const int ch = (kb << 4) + (rr & 15);
        st_ch[i] = ch;
        const long m = mb0 + ml;
        st_m[i] = m;
        st_q[i] = (int)(m % Q);
        st_p[i] = (int)((m / Q) % P);
        st_n[i] = (int)(m / ((long)Q * P));
        st_dyoff[i] = m * K + k0 + ch;
    }
    const int dQ = MB % Q, dP = (MB / Q) % P, dN0 = MB / (Q * P);
    const long dDY = (long)MB * K;

    auto stage = [&](int buf) {
        unsigned short* base = lds + buf * 2 * TILE_ELEMS;
        #pragma unroll
        for (int i = 0; i < GI; ++i) {
            const unsigned short* sp =
                (st_m[i] < mend) ? dy + st_dyoff[i] : conv_zero16;
            FDA_GLDS16(sp, base + (wid * GI + i) * 512);
        }
        #pragma unroll
        for (int i = 0; i < GI; ++i) {
            const unsigned short* sp = conv_zero16;
            if (st_m[i] < mend) {
                if (STEM) {
                    // pre-padded C=8 image; virtual channel = pixel s x 8ch
                    const int hh = st_p[i] * sy + r;
                    const int ww = st_q[i] * sx;
                    sp = x + (((long)st_n[i] * H + hh) * W + ww) * 8 +
                         st_ch[i];
                } else {
                    const int hh = st_p[i] * sy - py + r;
                    const int ww = st_q[i] * sx - px + s;
                    if ((unsigned)hh < (unsigned)H && (unsigned)ww < (unsigned)W)
                        sp = x + (((long)st_n[i] * H + hh) * W + ww) * C +
                             c0 + st_ch[i];
                }
            }
            FDA_GLDS16(sp, base + TILE_ELEMS + (wid * GI + i) * 512);
        }
        // advance MB pixels (bounded carries; dP < P, so p needs at most
        // two conditional wraps after the q carry)
        #pragma unroll
        for (int i = 0; i < GI; ++i) {
            st_m[i] += MB;
            st_dyoff[i] += dDY;
            int q = st_q[i] + dQ;
            int p = st_p[i] + dP;
            int n = st_n[i] + dN0;
            if (q >= Q) { q -= Q; ++p; }
            if (p >= P) { p -= P; ++n; }
            if (p >= P) { p -= P; ++n; }
            st_q[i] = q; st_p[i] = p; st_n[i] = n;
        }
    };

    const int l15 = lane & 15, lg = lane >> 4;

    // ring of RING tile-pairs, counted vmcnt: tile t+2's DMA stays in
    // flight across the barrier while tile t computes (2*FT glds per wave
    // per tile-pair). FT=4 uses a 2-deep ring (96 KiB would exceed 1
    // block/CU headroom at 3).
    constexpr int RING = (FT == 2 || MB == 32) ? 3 : 2;
    const int nsteps = (int)((mend - mb0 + MB - 1) / MB);
    if (nsteps > 0) stage(0);
    if (RING > 2 && nsteps > 1) stage(1);
    for (int it = 0; it < nsteps; ++it) {
        if (RING > 2 && it + 1 < nsteps) {
            if constexpr (2 * GI == 4)
                asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
            else
                asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
        } else {
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        }
        __builtin_amdgcn_s_barrier();
        // NOTE: the next tile's stage() is issued AFTER the tr reads below.
        // hipcc (ROCm 7.2) conservatively emits `s_waitcnt vmcnt(0)` before
        // every ds_read_b64_tr_b16 cluster (the intrinsic is ordered
        // against pending LDS-DMA writes it cannot alias-analyze), so a
        // stage issued before the reads is fully drained before any MFMA
        // runs — the r1 kernel had ZERO DMA/compute overlap here (the
        // measured WAIT_INST 38.6%). Reads-then-stage makes the inserted
        // drain wait only for data this iteration needs anyway, and the
        // DMA now flies over the MFMA cluster.
        const unsigned short* buf = lds + (it % RING) * 2 * TILE_ELEMS;
        short4_ a[FT][KSN][2], b[FT][KSN][2];   // [fi][ks][half]
        if constexpr (ASMRD) {
            // tr16 reads via inline asm, INVISIBLE to hipcc's wait
            // inserter: the compiler otherwise (a) force-drains all
            // outstanding LDS-DMA with vmcnt(0) before every visible
            // tr-read cluster and (b) waits lgkmcnt(0) over ALL 8*FT
            // reads before the first MFMA. Here the waits are counted by
            // hand: ks=1's reads stay in flight under ks=0's MFMAs.
            // stage() goes FIRST (max DMA flight time): its stray
            // compiler lgkmcnt(0) then fires while nothing is
            // outstanding, instead of draining the tr reads.
            if (it + RING - 1 < nsteps) stage((it + RING - 1) % RING);
            __builtin_amdgcn_sched_barrier(0);
            #pragma unroll
            for (int ks = 0; ks < KSN; ++ks)
                #pragma unroll
                for (int fi = 0; fi < FT; ++fi) {
                    const int kb_a = wk * FT + fi;
                    const int kb_b = wc * FT + fi;
                    const int v0 = ks * 8 + lg * 2;
                    const int p0 = (v0 >> 1);
                    const int p1 = p0 + NWIN / 2;
                    const unsigned short* pa =
                        buf + kb_a * (MB * 16) + l15 * 4;
                    const unsigned short* pb =
                        buf + TILE_ELEMS + kb_b * (MB * 16) + l15 * 4;
                    const unsigned aa0 = (unsigned)(unsigned long)
                        (const __attribute__((address_space(3))) char*)
                        (const char*)(pa + p0 * 64);
                    const unsigned aa1 = (unsigned)(unsigned long)
                        (const __attribute__((address_space(3))) char*)
                        (const char*)(pa + p1 * 64);
                    const unsigned ab0 = (unsigned)(unsigned long)
                        (const __attribute__((address_space(3))) char*)
                        (const char*)(pb + p0 * 64);
                    const unsigned ab1 = (unsigned)(unsigned long)
                        (const __attribute__((address_space(3))) char*)
                        (const char*)(pb + p1 * 64);
                    asm volatile("ds_read_b64_tr_b16 %0, %1"
                                 : "=v"(a[fi][ks][0]) : "v"(aa0));
                    asm volatile("ds_read_b64_tr_b16 %0, %1"
                                 : "=v"(a[fi][ks][1]) : "v"(aa1));
                    asm volatile("ds_read_b64_tr_b16 %0, %1"
                                 : "=v"(b[fi][ks][0]) : "v"(ab0));
                    asm volatile("ds_read_b64_tr_b16 %0, %1"
                                 : "=v"(b[fi][ks][1]) : "v"(ab1));
                }
            __builtin_amdgcn_s_setprio(1);
            #pragma unroll
            for (int ks = 0; ks < KSN; ++ks) {
                // sched_barrier(0) pins the MFMA clusters BETWEEN the
                // counted waits: MFMAs are pure-register ops, so without
                // it LLVM floats both clusters past both waits — and the
                // asm-read results carry no wait-dependency at all (the
                // hand-counted waits ARE the only correctness fence).
                __builtin_amdgcn_sched_barrier(0);
                if (ks == 0 && KSN == 2) {
                    // lgkmcnt is 4-bit (max 15): for FT=4 the ideal
                    // "leave 16 in flight" clamps to 15 (ds_reads return
                    // in order, so this waits one extra ks=1 read)
                    if constexpr (FT == 4)
                        asm volatile("s_waitcnt lgkmcnt(15)" ::: "memory");
                    else
                        asm volatile("s_waitcnt lgkmcnt(8)" ::: "memory");
                } else {
                    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
                }
                __builtin_amdgcn_sched_barrier(0);
                #pragma unroll
                for (int ki = 0; ki < FT; ++ki)
                    #pragma unroll
                    for (int ci = 0; ci < FT; ++ci) {
                        short8 af, bf;
                        #pragma unroll
                        for (int e = 0; e < 4; ++e) {
                            af[e] = a[ki][ks][0][e];
                            af[e + 4] = a[ki][ks][1][e];
                            bf[e] = b[ci][ks][0][e];
                            bf[e + 4] = b[ci][ks][1][e];
                        }
                        acc[ki][ci] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            af, bf, acc[ki][ci], 0, 0, 0);
                    }
            }
            __builtin_amdgcn_s_setprio(0);
        } else {
        #pragma unroll
        for (int fi = 0; fi < FT; ++fi)
            #pragma unroll
            for (int ks = 0; ks < KSN; ++ks) {
                const int kb_a = wk * FT + fi;
                const int kb_b = wc * FT + fi;
                const int v0 = ks * 8 + lg * 2;          // window indices
                const int p0 = (v0 >> 1);                // v0 even -> low half
                const int p1 = p0 + NWIN / 2;            // v0+1 odd -> high
                const unsigned short* pa =
                    buf + kb_a * (MB * 16) + l15 * 4;
                const unsigned short* pb =
                    buf + TILE_ELEMS + kb_b * (MB * 16) + l15 * 4;
                a[fi][ks][0] = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
                    (__attribute__((address_space(3))) short4_*)(pa + p0 * 64));
                a[fi][ks][1] = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
                    (__attribute__((address_space(3))) short4_*)(pa + p1 * 64));
                b[fi][ks][0] = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
                    (__attribute__((address_space(3))) short4_*)(pb + p0 * 64));
                b[fi][ks][1] = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
                    (__attribute__((address_space(3))) short4_*)(pb + p1 * 64));
            }
        if (it + RING - 1 < nsteps) stage((it + RING - 1) % RING);
        __builtin_amdgcn_s_setprio(1);
        #pragma unroll
        for (int ks = 0; ks < KSN; ++ks)
            #pragma unroll
            for (int ki = 0; ki < FT; ++ki)
                #pragma unroll
                for (int ci = 0; ci < FT; ++ci) {
                    short8 af, bf;
                    #pragma unroll
                    for (int e = 0; e < 4; ++e) {
                        af[e] = a[ki][ks][0][e];
                        af[e + 4] = a[ki][ks][1][e];
                        bf[e] = b[ci][ks][0][e];
                        bf[e + 4] = b[ci][ks][1][e];
                    }
                    acc[ki][ci] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        af, bf, acc[ki][ci], 0, 0, 0);
                }
        __builtin_amdgcn_s_setprio(0);
        }
        // end barrier dropped: the next iteration's wait + top barrier
        // orders buffer reuse for every ring depth.
    }

    // epilogue: out[i=k][j=c]; C/D map col=lane&15, row=(lane>>4)*4+jj
    const int fcol = lane & 15, frow0 = (lane >> 4) * 4;
    const long RSC = (long)R * S * C;
    float* const wsb = TP ? ws + (long)chunk * K * RSC : ws;
    #pragma unroll
    for (int ki = 0; ki < FT; ++ki)
        #pragma unroll
        for (int ci = 0; ci < FT; ++ci)
            #pragma unroll
            for (int jj = 0; jj < 4; ++jj) {
                const int kk = k0 + wk * 16 * FT + ki * 16 + frow0 + jj;
                const int cc = c0 + wc * 16 * FT + ci * 16 + fcol;
                if constexpr (TP)
                    wsb[kk * RSC + (long)rs * C + cc] = acc[ki][ci][jj];
                else
                    atomicAdd(&wsb[kk * RSC + (long)rs * C + cc],
                              acc[ki][ci][jj]);
            }
}

// two-phase combine: ws[i] += sum_chunk part[chunk][i] (i over K*RSC,
// float4-vectorized — C%64==0 so n%4==0). Streaming reads replace the
// nch-deep DRAM atomic fan-in per address.
__global__ __launch_bounds__(256) void wgrad_combine_kernel(
    float* __restrict__ ws, const float* __restrict__ part, long n,
    int nch) {
    const long i = ((long)blockIdx.x * 256 + threadIdx.x) * 4;
    if (i >= n) return;
    floatx4 s = *(const floatx4*)(part + i);
    for (int j = 1; j < nch; ++j) {
        const floatx4 p = *(const floatx4*)(part + (long)j * n + i);
        s.x += p.x; s.y += p.y; s.z += p.z; s.w += p.w;
    }
    floatx4* w = (floatx4*)(ws + i);
    const floatx4 w0 = *w;
    s.x += w0.x; s.y += w0.y; s.z += w0.z; s.w += w0.w;
    *w = s;
}

void wgrad_combine_launch(float* ws, const float* part, long n, int nch,
                          hipStream_t stream) {
    const long lanes = n / 4;
    dim3 grid((unsigned)((lanes + 255) / 256));
    hipLaunchKernelGGL(wgrad_combine_kernel, grid, dim3(256), 0, stream,
                       ws, part, n, nch);
}

static bool wgrad_asm() {
    // A/B knob FLUXDIST_WGRAD_ASM=1: hand-counted lgkm waits via
    // inline-asm tr16 reads (ks-split MFMA overlap). Default OFF:
    // measured -4.5% on the weighted wgrad total (2382 vs 2279 us) —
    // the sched_barrier fences cost more scheduling freedom than the
    // counted waits recover (profiles/ab_conv8.md campaign).
    static bool v = [] {
        const char* e = getenv("FLUXDIST_WGRAD_ASM");
        return e && e[0] == '1';
    }();
    return v;
}

void conv_stem_wgrad_launch(const void* dy, const void* x, float* ws,
                            int N, int Hp, int Wp, int K, int P, int Q,
                            int R, int sy, int sx, hipStream_t stream) {
    // dy [M][K]; x padded C=8 image; ws [K][R*64] fp32 pre-zeroed.
    // Output is tiny (K x R*64), so small chunks would hammer the same
    // fp32 addresses with atomics (588 hits/address measured at 2048):
    // use 16k-pixel chunks.
    const long M = (long)N * P * Q;
    constexpr int STEM_MCH = 8 * WG_MCH;    // 16k pixels: balances atomic fan-in per address against block count (32k measured 472 us/step vs 254 at 16k)
    const int nch = (int)((M + STEM_MCH - 1) / STEM_MCH);
    dim3 grid((unsigned)(K / 64), 1u, (unsigned)(R * nch));
    const size_t shmem = 3 * 2 * (WG_BM * 64) * sizeof(unsigned short);
    if (wgrad_asm())
        hipLaunchKernelGGL((conv_wgrad_kernel<2, true, true>), grid,
                           dim3(256), shmem, stream,
                           (const unsigned short*)dy,
                           (const unsigned short*)x, ws, N, Hp, Wp, 64, K,
                           P, Q, R, 1, sy, sx, 0, 0, nch, STEM_MCH);
    else
        hipLaunchKernelGGL((conv_wgrad_kernel<2, true>), grid, dim3(256),
                           shmem, stream, (const unsigned short*)dy,
                           (const unsigned short*)x, ws, N, Hp, Wp, 64, K,
                           P, Q, R, 1, sy, sx, 0, 0, nch, STEM_MCH);
}

// pick the pixel-chunk size so the grid lands near `target` blocks:
// enough parallelism to fill the chip, few enough chunks that the fp32
// atomic fan-in per output address stays small.
static int pick_mch(long M, long tiles, int target) {
    long nch_t = target / (tiles > 0 ? tiles : 1);
    if (nch_t < 1) nch_t = 1;
    long mch = (M + nch_t - 1) / nch_t;
    mch = ((mch + WG_MCH - 1) / WG_MCH) * WG_MCH;   // multiple of 2048
    if (mch < WG_MCH) mch = WG_MCH;
    return (int)mch;
}

// tile/chunk plan — the single source of truth shared with the bindings
// (which size the two-phase partial slab from nch; same pattern as
// conv_igemm_plan after the fwd-stats workspace near-miss).
void conv_wgrad_plan(long M, int C, int K, int R, int S,
                     int* ft, int* mch, int* nch) {
    // A/B: FLUXDIST_WGRAD_FT2=1 forces the 64x64 tile everywhere (its
    // 48 KB 3-ring fits 3 blocks/CU vs FT4's 2 — the BK32 lesson)
    static const bool force_ft2 = [] {
        const char* e = getenv("FLUXDIST_WGRAD_FT2");
        return e && e[0] == '1';
    }();
    // FT=4 quarters the block count; only worth it when the grid still
    // fills the 256 CUs (small-M 1x1 shapes measured 1.4x slower on it)
    const long tiles4 = (long)(K / 128) * (C / 128) * R * S;
    const long blocks4_max = tiles4 * ((M + WG_MCH - 1) / WG_MCH);
    if (!force_ft2 && K % 128 == 0 && C % 128 == 0 && blocks4_max >= 192) {
        *ft = 4;
        *mch = pick_mch(M, tiles4, 768);
    } else {
        *ft = 2;
        *mch = pick_mch(M, (long)(K / 64) * (C / 64) * R * S, 768);
    }
    *nch = (int)((M + *mch - 1) / *mch);
}

bool conv_wgrad_two_phase() {
    // FLUXDIST_WGRAD_2PH=1: chunk blocks plain-store private partials,
    // one combine kernel folds them (no DRAM atomics, no fill needed on
    // the slab — docs/wgrad_study.md option 1). Mutually exclusive with
    // the MB32 A/B knob, whose launch path ignores the slab — combining
    // an unwritten slab would corrupt ws.
    static const bool v = [] {
        const char* e = getenv("FLUXDIST_WGRAD_2PH");
        const char* m = getenv("FLUXDIST_WGRAD_MB32");
        return e && e[0] == '1' && !(m && m[0] == '1');
    }();
    return v;
}

void conv_wgrad_launch(const void* dy, const void* x, float* ws,
                       int N, int H, int W, int C, int K, int P, int Q,
                       int R, int S, int sy, int sx, int py, int px,
                       hipStream_t stream, float* part) {
    const long M = (long)N * P * Q;
    int FTp, mch, nch;
    conv_wgrad_plan(M, C, K, R, S, &FTp, &mch, &nch);
    if (FTp == 4) {
        dim3 grid((unsigned)(K / 128), (unsigned)(C / 128),
                  (unsigned)(R * S * nch));
        static const bool mb32 = [] {
            // A/B knob: FT4 with a 32-pixel K-step — 48 KB 3-ring at 3
            // blocks/CU, same 4:1 MFMA:glds ratio (the BK32 recipe)
            const char* e = getenv("FLUXDIST_WGRAD_MB32");
            return e && e[0] == '1';
        }();
        if (mb32) {
            const size_t shmem = 3 * 2 * (32 * 128) * sizeof(unsigned short);
            hipLaunchKernelGGL((conv_wgrad_kernel<4, false, false, 32>),
                               grid, dim3(256), shmem, stream,
                               (const unsigned short*)dy,
                               (const unsigned short*)x, ws, N, H, W, C, K,
                               P, Q, R, S, sy, sx, py, px, nch, mch);
            return;
        }
        const size_t shmem = 2 * 2 * (WG_BM * 128) * sizeof(unsigned short);
        if (part)
            hipLaunchKernelGGL((conv_wgrad_kernel<4, false, false, WG_BM,
                                                  true>),
                               grid, dim3(256), shmem, stream,
                               (const unsigned short*)dy,
                               (const unsigned short*)x, part, N, H, W, C,
                               K, P, Q, R, S, sy, sx, py, px, nch, mch);
        else if (wgrad_asm())
            hipLaunchKernelGGL((conv_wgrad_kernel<4, false, true>), grid,
                               dim3(256), shmem, stream,
                               (const unsigned short*)dy,
                               (const unsigned short*)x, ws, N, H, W, C, K,
                               P, Q, R, S, sy, sx, py, px, nch, mch);
        else
            hipLaunchKernelGGL(conv_wgrad_kernel<4>, grid, dim3(256), shmem,
                               stream, (const unsigned short*)dy,
                               (const unsigned short*)x, ws, N, H, W, C, K,
                               P, Q, R, S, sy, sx, py, px, nch, mch);
    } else {
        dim3 grid((unsigned)(K / 64), (unsigned)(C / 64),
                  (unsigned)(R * S * nch));
        const size_t shmem = 3 * 2 * (WG_BM * 64) * sizeof(unsigned short);
        if (part)
            hipLaunchKernelGGL((conv_wgrad_kernel<2, false, false, WG_BM,
                                                  true>),
                               grid, dim3(256), shmem, stream,
                               (const unsigned short*)dy,
                               (const unsigned short*)x, part, N, H, W, C,
                               K, P, Q, R, S, sy, sx, py, px, nch, mch);
        else if (wgrad_asm())
            hipLaunchKernelGGL((conv_wgrad_kernel<2, false, true>), grid,
                               dim3(256), shmem, stream,
                               (const unsigned short*)dy,
                               (const unsigned short*)x, ws, N, H, W, C, K,
                               P, Q, R, S, sy, sx, py, px, nch, mch);
        else
            hipLaunchKernelGGL(conv_wgrad_kernel<2>, grid, dim3(256), shmem,
                               stream, (const unsigned short*)dy,
                               (const unsigned short*)x, ws, N, H, W, C, K,
                               P, Q, R, S, sy, sx, py, px, nch, mch);
    }
}

}  // namespace fda
