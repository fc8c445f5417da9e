// Implicit-GEMM 2D convolution for gfx950 (CDNA4), NHWC bf16.
//
// MI355X-native replacement for the conv layer the reference gets for free
// from cuDNN via NNlibCUDA (/root/reference -> Flux conv):
// here it is a hand-written MFMA kernel — v_mfma_f32_16x16x32_bf16 tiles,
// LDS double-buffered staging via global_load_lds (direct HBM->LDS DMA),
// source-side XOR swizzle for bank-conflict-free ds_read_b128 fragment
// reads (guide T2/rule 21), one raw-barrier 2-phase pipeline per K-step.
//
// GEMM view (forward):
//   M = N*P*Q output pixels, Nd = K output channels, Kd = R*S*C
//   y[m][k] = sum_kd A[m][kd] * B[kd][k]
//   A = im2col gather of x (never materialized: per-lane source addresses
//       of the LDS DMA do the gather; out-of-bounds rows read a zero page)
//   B = w[k][r][s][c] (torch channels_last conv weight = [K][R*S*C] rows)
//
// DGRAD is the same kernel with A = gather of dy (stride-divisibility
// masked) and B = pre-transposed weights wt[rs][c][k] (k-contiguous rows).
//
// Tiles: BM=128 x BN=64 x BK=64, 256 threads (4 waves as 2x2), per-wave
// 64x32 output = 4x2 fragments of 16x16, fp32 accumulate, bf16 store.
// LDS: (128*64 + 64*64) bf16 * 2 buffers = 48 KiB -> up to 3 blocks/CU.
//
// Constraints (enforced by the host wrapper): C % 64 == 0 (fwd) or
// K % 64 == 0 (dgrad) for the staged operand, dilation 1, groups 1.
// The ResNet stem (C=3) falls back to the library path.

#include <hip/hip_runtime.h>
#include "fda_common.h"

namespace fda {

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float floatx4;

// 16-byte zero page for out-of-bounds im2col rows (glds has no predication;
// invalid lanes redirect their source address here).
__device__ __align__(16) static const unsigned short conv_zero16[8] = {0};

#define FDA_GLDS16(gptr, lptr)                                              \
    __builtin_amdgcn_global_load_lds(                                       \
        (const __attribute__((address_space(1))) void*)(gptr),              \
        (__attribute__((address_space(3))) void*)(lptr), 16, 0, 0)

enum ConvMode { CONV_FWD = 0, CONV_DGRAD = 1 };

// One K-step tile pair in LDS (bf16): A [BM][BK] then B [BN][BK], both
// row-major 128-B rows, slot s of row r holds global slot (s ^ (r&7)).
constexpr int BM = 128, BN = 64, BK = 64;
constexpr int A_ELEMS = BM * BK;          // 8192 bf16 = 16 KiB
constexpr int B_ELEMS = BN * BK;          // 4096 bf16 = 8 KiB
constexpr int BUF_ELEMS = A_ELEMS + B_ELEMS;

template <int MODE>
__global__ __launch_bounds__(256, 2) void conv_igemm_kernel(
    const unsigned short* __restrict__ src,   // x (fwd) / dy (dgrad), NHWC
    const unsigned short* __restrict__ wgt,   // w [K][RS*C] (fwd) / wt [RS*C][K] (dgrad)
    unsigned short* __restrict__ out,         // y [M][K] (fwd) / dx [M][C] (dgrad)
    int N, int H, int W, int C,               // input tensor dims (fwd view)
    int K, int P, int Q,                      // output channels & spatial
    int R, int S, int sy, int sx, int py, int px) {
    // Output-pixel space of THIS kernel: fwd -> (P,Q,K over C-reduction);
    // dgrad -> out pixels are (H,W) with channels C, reduction over K.
    const int OH = (MODE == CONV_FWD) ? P : H;
    const int OW = (MODE == CONV_FWD) ? Q : W;
    const int OC = (MODE == CONV_FWD) ? K : C;   // Nd of the GEMM
    const int RC = (MODE == CONV_FWD) ? C : K;   // staged reduction channels
    const long M = (long)N * OH * OW;

    const int mtile = blockIdx.x;             // BM rows of output pixels
    const int ntile = blockIdx.y;             // BN output channels
    const long m0 = (long)mtile * BM;
    const int n0 = ntile * BN;

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;                 // 4 waves: 2x2 tiling
    const int wm = wid >> 1;                  // wave row (0..1): 64 pixels
    const int wn = wid & 1;                   // wave col (0..1): 32 channels

    extern __shared__ unsigned short lds[];   // [2][BUF_ELEMS]

    // ---- per-lane staging descriptors (computed once) ---------------------
    // A staging: wave issues 4 glds, instr i covers rows (wid*4+i)*8 + lane/8.
    // Each lane loads 16 B (8 bf16) from channel-slot ((lane%8) ^ (row&7)).
    int a_row[4];       // tile-row handled by this lane per instr
    long a_pix[4];      // base offset of the row's pixel, in elements
    int a_hb[4], a_wb[4];  // fwd: h/w base; dgrad: raw h+py / w+px
    bool a_mok[4];
    const int cslot = lane & 7;
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
        const int row = (wid * 4 + i) * 8 + (lane >> 3);
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
            a_pix[i] = ((long)n * H) * W * C;  // + (h*W + w)*C later
        } else {
            a_hb[i] = oh + py;                 // h + py (subtract r later)
            a_wb[i] = ow + px;
            a_pix[i] = ((long)n * P) * Q * K;
        }
    }
    // B staging: wave issues 2 glds, instr i covers rows (wid*2+i)*8 + lane/8.
    int b_row[2];
    #pragma unroll
    for (int i = 0; i < 2; ++i) b_row[i] = (wid * 2 + i) * 8 + (lane >> 3);

    const int cblocks = RC / BK;              // reduction-channel blocks
    const int T = R * S * cblocks;            // total K-steps

    // ---- staging of one K-step tile into lds[buf] -------------------------
    auto stage = [&](int buf, int it) {
        const int rs = it / cblocks;
        const int cb = (it % cblocks) * BK;
        const int r = rs / S, s = rs % S;
        unsigned short* base = lds + buf * BUF_ELEMS;
        // A tile
        #pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int row = a_row[i];
            const int cs = (cslot ^ (row & 7)) * 8;  // swizzled source slot
            const unsigned short* sp;
            bool ok = a_mok[i];
            long off = 0;
            if (MODE == CONV_FWD) {
                const int h = a_hb[i] + r, w = a_wb[i] + s;
                ok = ok && (unsigned)h < (unsigned)H && (unsigned)w < (unsigned)W;
                off = a_pix[i] + ((long)h * W + w) * C + cb + cs;
            } else {
                const int hn = a_hb[i] - r, wn2 = a_wb[i] - s;
                const int p = hn / sy, q = wn2 / sx;
                ok = ok && hn >= 0 && wn2 >= 0 && (sy == 1 || (hn % sy) == 0)
                        && (sx == 1 || (wn2 % sx) == 0) && p < P && q < Q;
                off = a_pix[i] + ((long)p * Q + q) * K + cb + cs;
            }
            sp = ok ? src + off : conv_zero16;
            // dest: row-major [BM][BK]; wave-uniform base, lane*16B appended
            // by hardware: rows (wid*4+i)*8 .. +8
            FDA_GLDS16(sp, base + (wid * 4 + i) * 8 * BK);
        }
        // B tile
        #pragma unroll
        for (int i = 0; i < 2; ++i) {
            const int row = b_row[i];
            const int cs = (cslot ^ (row & 7)) * 8;
            const unsigned short* sp;
            if (MODE == CONV_FWD) {
                // B[kd=c][j=k] staged as rows [k][c]: w[k][rs*C + cb + c]
                sp = wgt + ((long)(n0 + row) * R * S * C + (long)rs * C + cb + cs);
            } else {
                // B[kd=k][j=c] staged as rows [c][k]: wt[rs*C + c][k]
                sp = wgt + ((long)((long)rs * C + n0 + row) * K + cb + cs);
            }
            FDA_GLDS16(sp, base + A_ELEMS + (wid * 2 + i) * 8 * BK);
        }
    };

    // ---- fragment read offsets (bytes into an lds buffer) -----------------
    // a_frag[mi][kh]: row = wm*64 + mi*16 + (lane&15), slot = kh*4 + lane>>4
    // b_frag[ni][kh]: row = wn*32 + ni*16 + (lane&15) (+A_ELEMS base)
    int a_off[4][2], b_off[2][2];
    {
        const int fr = lane & 15, fq = lane >> 4;
        #pragma unroll
        for (int mi = 0; mi < 4; ++mi)
            #pragma unroll
            for (int kh = 0; kh < 2; ++kh) {
                const int row = wm * 64 + mi * 16 + fr;
                const int slot = (kh * 4 + fq) ^ (row & 7);
                a_off[mi][kh] = row * BK + slot * 8;      // elements
            }
        #pragma unroll
        for (int ni = 0; ni < 2; ++ni)
            #pragma unroll
            for (int kh = 0; kh < 2; ++kh) {
                const int row = wn * 32 + ni * 16 + fr;
                const int slot = (kh * 4 + fq) ^ (row & 7);
                b_off[ni][kh] = A_ELEMS + row * BK + slot * 8;
            }
    }

    floatx4 acc[4][2];
    #pragma unroll
    for (int mi = 0; mi < 4; ++mi)
        #pragma unroll
        for (int ni = 0; ni < 2; ++ni) acc[mi][ni] = floatx4{0.f, 0.f, 0.f, 0.f};

    // ---- main loop: 2-phase glds pipeline (guide §5.5 T3 minimum form) ----
    stage(0, 0);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    int cur = 0;
    for (int it = 0; it < T; ++it) {
        if (it + 1 < T) stage(cur ^ 1, it + 1);
        const unsigned short* buf = lds + cur * BUF_ELEMS;
        short8 a[4][2], b[2][2];
        #pragma unroll
        for (int mi = 0; mi < 4; ++mi)
            #pragma unroll
            for (int kh = 0; kh < 2; ++kh)
                a[mi][kh] = *(const short8*)(buf + a_off[mi][kh]);
        #pragma unroll
        for (int ni = 0; ni < 2; ++ni)
            #pragma unroll
            for (int kh = 0; kh < 2; ++kh)
                b[ni][kh] = *(const short8*)(buf + b_off[ni][kh]);
        __builtin_amdgcn_s_setprio(1);
        #pragma unroll
        for (int kh = 0; kh < 2; ++kh)
            #pragma unroll
            for (int mi = 0; mi < 4; ++mi)
                #pragma unroll
                for (int ni = 0; ni < 2; ++ni)
                    acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a[mi][kh], b[ni][kh], acc[mi][ni], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        cur ^= 1;
    }

    // ---- epilogue: C/D map col=lane&15, row=(lane>>4)*4+j (16x16) ---------
    const int fcol = lane & 15, frow0 = (lane >> 4) * 4;
    #pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
            const long m = m0 + wm * 64 + mi * 16 + frow0 + j;
            if (m >= M) continue;
            unsigned short* orow = out + m * OC + n0 + wn * 32;
            #pragma unroll
            for (int ni = 0; ni < 2; ++ni)
                orow[ni * 16 + fcol] = f32_to_bf16bits(acc[mi][ni][j]);
        }
    }
}

void conv_igemm_launch(const void* src, const void* wgt, void* out,
                       int N, int H, int W, int C, int K, int P, int Q,
                       int R, int S, int sy, int sx, int py, int px,
                       bool dgrad, hipStream_t stream) {
    const long M = dgrad ? (long)N * H * W : (long)N * P * Q;
    const int OC = dgrad ? C : K;
    dim3 grid((unsigned)((M + BM - 1) / BM), (unsigned)(OC / BN));
    dim3 block(256);
    const size_t shmem = 2 * BUF_ELEMS * sizeof(unsigned short);
    if (dgrad)
        hipLaunchKernelGGL((conv_igemm_kernel<CONV_DGRAD>), grid, block, shmem,
                           stream, (const unsigned short*)src,
                           (const unsigned short*)wgt, (unsigned short*)out,
                           N, H, W, C, K, P, Q, R, S, sy, sx, py, px);
    else
        hipLaunchKernelGGL((conv_igemm_kernel<CONV_FWD>), grid, block, shmem,
                           stream, (const unsigned short*)src,
                           (const unsigned short*)wgt, (unsigned short*)out,
                           N, H, W, C, K, P, Q, R, S, sy, sx, py, px);
}

}  // namespace fda
