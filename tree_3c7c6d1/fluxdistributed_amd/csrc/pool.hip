// MaxPool2d NHWC fwd (+argmax) / bwd (gather over candidate windows).
// Reference site: ResNet stem 3x3 stride-2 pool (SURVEY.md §2.4 — "hand
// kernel + index mask"). Generic in kernel/stride/pad; ResNet uses 3x3 s2 p1.
//
// fwd: one thread computes V consecutive channels of one output pixel,
//      storing the window-position argmax (0..KH*KW-1) as one byte/channel.
// bwd: one thread computes V channels of one INPUT pixel by gathering the
//      <= ceil(K/S)^2 output windows that cover it — no atomics.
#include "fda_common.h"
#include "fda_kernels.h"

namespace fda {

template <typename T, int V>
__global__ void maxpool_fwd_kernel(const T* __restrict__ x, T* __restrict__ out,
                                   uint8_t* __restrict__ idx, int N, int H,
                                   int W, int C, int HO, int WO, int KH, int KW,
                                   int S, int P) {
    const int64_t total = (int64_t)N * HO * WO * (C / V);
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        const int cv = (int)(i % (C / V));
        int64_t t = i / (C / V);
        const int wo = (int)(t % WO);
        t /= WO;
        const int ho = (int)(t % HO);
        const int n = (int)(t / HO);
        const int c0 = cv * V;

        float best[V];
        uint8_t barg[V];
        #pragma unroll
        for (int k = 0; k < V; ++k) {
            best[k] = -INFINITY;
            barg[k] = 0;
        }
        const int h0 = ho * S - P, w0 = wo * S - P;
        for (int kh = 0; kh < KH; ++kh) {
            const int hi = h0 + kh;
            if (hi < 0 || hi >= H) continue;
            for (int kw = 0; kw < KW; ++kw) {
                const int wi = w0 + kw;
                if (wi < 0 || wi >= W) continue;
                T xv[V];
                *(uint4*)xv = *(const uint4*)(
                    x + (((int64_t)n * H + hi) * W + wi) * C + c0);
                const uint8_t p = (uint8_t)(kh * KW + kw);
                #pragma unroll
                for (int k = 0; k < V; ++k) {
                    const float v = load_f32(xv + k);
                    if (v > best[k]) {
                        best[k] = v;
                        barg[k] = p;
                    }
                }
            }
        }
        T ov[V];
        #pragma unroll
        for (int k = 0; k < V; ++k) store_f32(ov + k, best[k]);
        const int64_t o = (((int64_t)n * HO + ho) * WO + wo) * C + c0;
        *(uint4*)(out + o) = *(uint4*)ov;
        #pragma unroll
        for (int k = 0; k < V; ++k) idx[o + k] = barg[k];
    }
}

template <typename T, int V>
__global__ void maxpool_bwd_kernel(const T* __restrict__ gout,
                                   const uint8_t* __restrict__ idx,
                                   T* __restrict__ gx, int N, int H, int W,
                                   int C, int HO, int WO, int KH, int KW, int S,
                                   int P) {
    const int64_t total = (int64_t)N * H * W * (C / V);
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        const int cv = (int)(i % (C / V));
        int64_t t = i / (C / V);
        const int wi = (int)(t % W);
        t /= W;
        const int hi = (int)(t % H);
        const int n = (int)(t / H);
        const int c0 = cv * V;

        float acc[V];
        #pragma unroll
        for (int k = 0; k < V; ++k) acc[k] = 0.f;

        // output windows covering (hi, wi): ho*S - P <= hi < ho*S - P + KH
        const int ho_lo = max(0, (hi + P - KH + S) / S);
        const int ho_hi = min(HO - 1, (hi + P) / S);
        const int wo_lo = max(0, (wi + P - KW + S) / S);
        const int wo_hi = min(WO - 1, (wi + P) / S);
        for (int ho = ho_lo; ho <= ho_hi; ++ho) {
            const int kh = hi - (ho * S - P);
            if (kh < 0 || kh >= KH) continue;
            for (int wo = wo_lo; wo <= wo_hi; ++wo) {
                const int kw = wi - (wo * S - P);
                if (kw < 0 || kw >= KW) continue;
                const uint8_t p = (uint8_t)(kh * KW + kw);
                const int64_t o = (((int64_t)n * HO + ho) * WO + wo) * C + c0;
                T gv[V];
                *(uint4*)gv = *(const uint4*)(gout + o);
                // V idx bytes in one word (per-byte loads were issue-bound)
                uint64_t ib;
                if constexpr (V == 8)
                    ib = *(const uint64_t*)(idx + o);
                else
                    ib = *(const uint32_t*)(idx + o);
                #pragma unroll
                for (int k = 0; k < V; ++k)
                    if (((ib >> (8 * k)) & 0xffu) == p)
                        acc[k] += load_f32(gv + k);
            }
        }
        T rv[V];
        #pragma unroll
        for (int k = 0; k < V; ++k) store_f32(rv + k, acc[k]);
        *(uint4*)(gx + (((int64_t)n * H + hi) * W + wi) * C + c0) = *(uint4*)rv;
    }
}

static inline int pool_grid(int64_t total) {
    int64_t b = (total + 255) / 256;
    return (int)(b < 1 ? 1 : (b > 4096 ? 4096 : b));
}

void maxpool_fwd_launch(const void* x, void* out, uint8_t* idx, int N, int H,
                        int W, int C, int HO, int WO, int KH, int KW, int S,
                        int P, DT dt, hipStream_t s) {
    if (dt == DT::BF16)
        hipLaunchKernelGGL((maxpool_fwd_kernel<unsigned short, 8>),
                           dim3(pool_grid((int64_t)N * HO * WO * C / 8)),
                           dim3(256), 0, s, (const unsigned short*)x,
                           (unsigned short*)out, idx, N, H, W, C, HO, WO, KH,
                           KW, S, P);
    else
        hipLaunchKernelGGL((maxpool_fwd_kernel<float, 4>),
                           dim3(pool_grid((int64_t)N * HO * WO * C / 4)),
                           dim3(256), 0, s, (const float*)x, (float*)out, idx,
                           N, H, W, C, HO, WO, KH, KW, S, P);
}

void maxpool_bwd_launch(const void* gout, const uint8_t* idx, void* gx, int N,
                        int H, int W, int C, int HO, int WO, int KH, int KW,
                        int S, int P, DT dt, hipStream_t s) {
    if (dt == DT::BF16)
        hipLaunchKernelGGL((maxpool_bwd_kernel<unsigned short, 8>),
                           dim3(pool_grid((int64_t)N * H * W * C / 8)),
                           dim3(256), 0, s, (const unsigned short*)gout, idx,
                           (unsigned short*)gx, N, H, W, C, HO, WO, KH, KW, S,
                           P);
    else
        hipLaunchKernelGGL((maxpool_bwd_kernel<float, 4>),
                           dim3(pool_grid((int64_t)N * H * W * C / 4)),
                           dim3(256), 0, s, (const float*)gout, idx,
                           (float*)gx, N, H, W, C, HO, WO, KH, KW, S, P);
}

}  // namespace fda

namespace fda {

// ---- global average pool (AdaptiveMeanPool 1x1), NHWC -------------------
// "AdaptiveMeanPool / global avg pool fwd/bwd — warp
// reduction". fwd: y[n][c] = mean_hw x[n][h][w][c]; bwd: gx = gy/HW bcast.
template <typename T, int V>
__global__ __launch_bounds__(256) void gap_fwd_kernel(
    const T* __restrict__ x, T* __restrict__ y, int N, int HW, int C) {
    // one thread per (n, c-vec): strided column reduction, fp32 accum
    const long nv = (long)N * (C / V);
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nv;
         i += (long)gridDim.x * blockDim.x) {
        const int n = (int)(i / (C / V));
        const int c0 = (int)(i % (C / V)) * V;
        const T* base = x + ((long)n * HW) * C + c0;
        float acc[V];
        #pragma unroll
        for (int e = 0; e < V; ++e) acc[e] = 0.f;
        for (int r = 0; r < HW; ++r) {
            T v[V];
            *(uint4*)v = *(const uint4*)(base + (long)r * C);
            #pragma unroll
            for (int e = 0; e < V; ++e) acc[e] += load_f32(v + e);
        }
        T o[V];
        const float inv = 1.f / (float)HW;
        #pragma unroll
        for (int e = 0; e < V; ++e) store_f32(o + e, acc[e] * inv);
        *(uint4*)(y + (long)n * C + c0) = *(uint4*)o;
    }
}

template <typename T, int V>
__global__ __launch_bounds__(256) void gap_bwd_kernel(
    const T* __restrict__ gy, T* __restrict__ gx, int N, int HW, int C) {
    const long nv = (long)N * HW * (C / V);
    const float inv = 1.f / (float)HW;
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nv;
         i += (long)gridDim.x * blockDim.x) {
        const long row = i / (C / V);          // (n, hw)
        const int n = (int)(row / HW);
        const int c0 = (int)(i % (C / V)) * V;
        T g[V];
        *(uint4*)g = *(const uint4*)(gy + (long)n * C + c0);
        T o[V];
        #pragma unroll
        for (int e = 0; e < V; ++e) store_f32(o + e, load_f32(g + e) * inv);
        *(uint4*)(gx + row * C + c0) = *(uint4*)o;
    }
}

void gap_fwd_launch(const void* x, void* y, int N, int HW, int C, DT dt,
                    hipStream_t s) {
    const int V = dt == DT::BF16 ? 8 : 4;
    const long nv = (long)N * (C / V);
    dim3 grid((unsigned)((nv + 255) / 256 < 1024 ? (nv + 255) / 256 : 1024));
    if (dt == DT::BF16)
        hipLaunchKernelGGL((gap_fwd_kernel<unsigned short, 8>), grid,
                           dim3(256), 0, s, (const unsigned short*)x,
                           (unsigned short*)y, N, HW, C);
    else
        hipLaunchKernelGGL((gap_fwd_kernel<float, 4>), grid, dim3(256), 0, s,
                           (const float*)x, (float*)y, N, HW, C);
}

void gap_bwd_launch(const void* gy, void* gx, int N, int HW, int C, DT dt,
                    hipStream_t s) {
    const int V = dt == DT::BF16 ? 8 : 4;
    const long nv = (long)N * HW * (C / V);
    dim3 grid((unsigned)((nv + 255) / 256 < 4096 ? (nv + 255) / 256 : 4096));
    if (dt == DT::BF16)
        hipLaunchKernelGGL((gap_bwd_kernel<unsigned short, 8>), grid,
                           dim3(256), 0, s, (const unsigned short*)gy,
                           (unsigned short*)gx, N, HW, C);
    else
        hipLaunchKernelGGL((gap_bwd_kernel<float, 4>), grid, dim3(256), 0, s,
                           (const float*)gy, (float*)gx, N, HW, C);
}

}  // namespace fda
