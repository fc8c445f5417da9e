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
// fp32 accumulate, bf16 store.
//
// Constraints (host wrapper): staged reduction channels (C fwd / K dgrad)
// and output channels multiples of 64, dilation 1, groups 1. The ResNet
// stem (C=3) falls back to the library path.

#include <hip/hip_runtime.h>
#include "fda_common.h"

namespace fda {

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float floatx4;

__device__ __align__(16) static const unsigned short conv_zero16[8] = {0};

#define FDA_GLDS16(gptr, lptr)                                              \
    __builtin_amdgcn_global_load_lds(                                       \
        (const __attribute__((address_space(1))) void*)(gptr),              \
        (__attribute__((address_space(3))) void*)(lptr), 16, 0, 0)

enum ConvMode { CONV_FWD = 0, CONV_DGRAD = 1 };

constexpr int BK = 64;

template <int MODE, int BM, int BN, int WN>
__global__ __launch_bounds__(256, 2) void conv_igemm_kernel(
    const unsigned short* __restrict__ src,
    const unsigned short* __restrict__ wgt,
    unsigned short* __restrict__ out,
    int N, int H, int W, int C,
    int K, int P, int Q,
    int R, int S, int sy, int sx, int py, int px) {
    constexpr int A_ELEMS = BM * BK;
    constexpr int B_ELEMS = BN * BK;
    constexpr int BUF_ELEMS = A_ELEMS + B_ELEMS;
    constexpr int AI = BM / 32;          // A glds per wave per tile
    constexpr int BI = BN / 32;          // B glds per wave per tile

    const int OC = (MODE == CONV_FWD) ? K : C;
    const int RC = (MODE == CONV_FWD) ? C : K;

    int a = 0, b = 0, OH, OW, r0 = 0, s0 = 0, nR = R, nS = S;
    if (MODE == CONV_FWD) {
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
    const int wm = wid / WN;             // wave row (64-pixel granularity)
    const int wn = wid % WN;             // wave col (64-channel granularity)

    extern __shared__ unsigned short lds[];   // [2][BUF_ELEMS]

    // ---- per-lane staging descriptors ------------------------------------
    int a_row[AI];
    long a_pix[AI];
    int a_hb[AI], a_wb[AI];
    bool a_mok[AI];
    const int cslot = lane & 7;
    #pragma unroll
    for (int i = 0; i < AI; ++i) {
        const int row = (wid * AI + i) * 8 + (lane >> 3);
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
    for (int i = 0; i < BI; ++i) b_row[i] = (wid * BI + i) * 8 + (lane >> 3);

    const int cblocks = RC / BK;
    const int T = nR * nS * cblocks;

    auto stage = [&](int buf, int it) {
        const int rsi = it / cblocks;
        const int cb = (it % cblocks) * BK;
        const int ri = rsi / nS, si = rsi % nS;
        unsigned short* base = lds + buf * BUF_ELEMS;
        #pragma unroll
        for (int i = 0; i < AI; ++i) {
            const int row = a_row[i];
            const int cs = (cslot ^ (row & 7)) * 8;
            const unsigned short* sp;
            bool ok = a_mok[i];
            long off = 0;
            if (MODE == CONV_FWD) {
                const int h = a_hb[i] + ri, w = a_wb[i] + si;
                ok = ok && (unsigned)h < (unsigned)H && (unsigned)w < (unsigned)W;
                off = a_pix[i] + ((long)h * W + w) * C + cb + cs;
            } else {
                const int p = a_hb[i] - ri, q = a_wb[i] - si;
                ok = ok && (unsigned)p < (unsigned)P && (unsigned)q < (unsigned)Q;
                off = a_pix[i] + ((long)p * Q + q) * K + cb + cs;
            }
            sp = ok ? src + off : conv_zero16;
            FDA_GLDS16(sp, base + (wid * AI + i) * 8 * BK);
        }
        const int r = r0 + ri * ((MODE == CONV_FWD) ? 1 : sy);
        const int s = s0 + si * ((MODE == CONV_FWD) ? 1 : sx);
        const int rs = r * S + s;
        #pragma unroll
        for (int i = 0; i < BI; ++i) {
            const int row = b_row[i];
            const int cs = (cslot ^ (row & 7)) * 8;
            const unsigned short* sp;
            if (MODE == CONV_FWD) {
                sp = wgt + ((long)(n0 + row) * R * S * C + (long)rs * C + cb + cs);
            } else {
                sp = wgt + ((long)((long)rs * C + n0 + row) * K + cb + cs);
            }
            FDA_GLDS16(sp, base + A_ELEMS + (wid * BI + i) * 8 * BK);
        }
    };

    // ---- fragment read offsets (elements into an lds buffer) -------------
    int a_off[4][2], b_off[4][2];
    {
        const int fr = lane & 15, fq = lane >> 4;
        #pragma unroll
        for (int mi = 0; mi < 4; ++mi)
            #pragma unroll
            for (int kh = 0; kh < 2; ++kh) {
                const int row = wm * 64 + mi * 16 + fr;
                const int slot = (kh * 4 + fq) ^ (row & 7);
                a_off[mi][kh] = row * BK + slot * 8;
            }
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni)
            #pragma unroll
            for (int kh = 0; kh < 2; ++kh) {
                const int row = wn * 64 + ni * 16 + fr;
                const int slot = (kh * 4 + fq) ^ (row & 7);
                b_off[ni][kh] = A_ELEMS + row * BK + slot * 8;
            }
    }

    floatx4 acc[4][4];
    #pragma unroll
    for (int mi = 0; mi < 4; ++mi)
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni) acc[mi][ni] = floatx4{0.f, 0.f, 0.f, 0.f};

    // ---- main loop: double buffer; tile t+1's DMA in flight over tile t's
    // compute, drained at the iteration boundary (guide T3 minimum form) ---
    if (T > 0) stage(0, 0);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    for (int it = 0; it < T; ++it) {
        if (it + 1 < T) stage((it + 1) & 1, it + 1);
        const unsigned short* buf = lds + (it & 1) * BUF_ELEMS;
        short8 af[4][2], bf[4][2];
        #pragma unroll
        for (int mi = 0; mi < 4; ++mi)
            #pragma unroll
            for (int kh = 0; kh < 2; ++kh)
                af[mi][kh] = *(const short8*)(buf + a_off[mi][kh]);
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni)
            #pragma unroll
            for (int kh = 0; kh < 2; ++kh)
                bf[ni][kh] = *(const short8*)(buf + b_off[ni][kh]);
        __builtin_amdgcn_s_setprio(1);
        #pragma unroll
        for (int kh = 0; kh < 2; ++kh)
            #pragma unroll
            for (int mi = 0; mi < 4; ++mi)
                #pragma unroll
                for (int ni = 0; ni < 4; ++ni)
                    acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        af[mi][kh], bf[ni][kh], acc[mi][ni], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
    }

    // ---- epilogue: C/D map col=lane&15, row=(lane>>4)*4+j ----------------
    const int fcol = lane & 15, frow0 = (lane >> 4) * 4;
    #pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
            const long m = m0 + wm * 64 + mi * 16 + frow0 + j;
            if (m >= M) continue;
            long obase;
            if (MODE == CONV_FWD) {
                obase = m * OC;
            } else {
                const int ww = (int)(m % OW);
                const int hh = (int)((m / OW) % OH);
                const int n = (int)(m / ((long)OW * OH));
                obase = (((long)n * H + a + (long)sy * hh) * W + b +
                         (long)sx * ww) * C;
            }
            unsigned short* orow = out + obase + n0 + wn * 64;
            #pragma unroll
            for (int ni = 0; ni < 4; ++ni)
                orow[ni * 16 + fcol] = f32_to_bf16bits(acc[mi][ni][j]);
        }
    }
}

template <int MODE, int BM, int BN, int WN>
static void launch_cfg(const void* src, const void* wgt, void* out,
                       int N, int H, int W, int C, int K, int P, int Q,
                       int R, int S, int sy, int sx, int py, int px,
                       hipStream_t stream) {
    const int OC = (MODE == CONV_FWD) ? K : C;
    const long M = (MODE == CONV_FWD)
        ? (long)N * P * Q
        : (long)N * ((H + sy - 1) / sy) * ((W + sx - 1) / sx);
    dim3 grid((unsigned)((M + BM - 1) / BM), (unsigned)(OC / BN),
              (MODE == CONV_FWD) ? 1u : (unsigned)(sy * sx));
    const size_t shmem = 2 * (BM * BK + BN * BK) * sizeof(unsigned short);
    hipLaunchKernelGGL((conv_igemm_kernel<MODE, BM, BN, WN>), grid, dim3(256),
                       shmem, stream, (const unsigned short*)src,
                       (const unsigned short*)wgt, (unsigned short*)out,
                       N, H, W, C, K, P, Q, R, S, sy, sx, py, px);
}

void conv_igemm_launch(const void* src, const void* wgt, void* out,
                       int N, int H, int W, int C, int K, int P, int Q,
                       int R, int S, int sy, int sx, int py, int px,
                       bool dgrad, hipStream_t stream) {
    const int OC = dgrad ? C : K;
    if (dgrad) {
        if (OC % 128 == 0)
            launch_cfg<CONV_DGRAD, 128, 128, 2>(src, wgt, out, N, H, W, C, K,
                                                P, Q, R, S, sy, sx, py, px,
                                                stream);
        else
            launch_cfg<CONV_DGRAD, 256, 64, 1>(src, wgt, out, N, H, W, C, K,
                                               P, Q, R, S, sy, sx, py, px,
                                               stream);
    } else {
        if (OC % 128 == 0)
            launch_cfg<CONV_FWD, 128, 128, 2>(src, wgt, out, N, H, W, C, K,
                                              P, Q, R, S, sy, sx, py, px,
                                              stream);
        else
            launch_cfg<CONV_FWD, 256, 64, 1>(src, wgt, out, N, H, W, C, K,
                                             P, Q, R, S, sy, sx, py, px,
                                             stream);
    }
}

}  // namespace fda
