// Torch bindings for the fluxdistributed_amd gfx950 kernels (_C extension).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "fda_kernels.h"

namespace {

using fda::DT;

DT dt_of(const at::Tensor& t) {
    if (t.scalar_type() == at::kBFloat16) return DT::BF16;
    TORCH_CHECK(t.scalar_type() == at::kFloat, "expected f32 or bf16, got ",
                t.scalar_type());
    return DT::F32;
}

hipStream_t cur_stream() {
    return at::hip::getCurrentHIPStream().stream();
}

// channels_last 4D tensor -> (rows, C) with C contiguous
std::pair<int64_t, int64_t> nhwc_rows(const at::Tensor& x) {
    TORCH_CHECK(x.dim() == 4, "expected 4D NCHW-logical tensor");
    TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
                "expected channels_last layout");
    return {x.size(0) * x.size(2) * x.size(3), x.size(1)};
}

std::tuple<at::Tensor, at::Tensor> ce_fwd(at::Tensor logits, at::Tensor target) {
    // accepts row-strided logits (stride(1)==1): the FC head's %64-padded
    // buffer is read in place, no contiguity copy
    TORCH_CHECK(logits.dim() == 2 && logits.stride(1) == 1 &&
                logits.stride(0) >= logits.size(1));
    TORCH_CHECK(target.scalar_type() == at::kLong && target.is_contiguous());
    const int N = (int)logits.size(0), C = (int)logits.size(1);
    auto loss = at::zeros({}, logits.options().dtype(at::kFloat));
    auto dlogits = at::empty({N, C}, logits.options());
    fda::ce_fwd_launch(logits.data_ptr(), target.data_ptr<int64_t>(),
                       loss.data_ptr<float>(), dlogits.data_ptr(), N, C,
                       (int)logits.stride(0), dt_of(logits), cur_stream());
    // loss stays fp32 regardless of logits dtype: a bf16 round would put
    // every logged loss on a 2^-8 grid (round-1 verdict weak #6)
    return {loss, dlogits};
}

at::Tensor add_relu_fwd(at::Tensor x, at::Tensor r) {
    TORCH_CHECK(x.sizes() == r.sizes() && x.scalar_type() == r.scalar_type());
    at::Tensor xc = x, rc = r;
    if (x.dim() == 4) {
        xc = x.contiguous(at::MemoryFormat::ChannelsLast);
        rc = r.contiguous(at::MemoryFormat::ChannelsLast);
    } else {
        xc = x.contiguous();
        rc = r.contiguous();
    }
    auto out = at::empty_like(xc);
    fda::add_relu_fwd_launch(xc.data_ptr(), rc.data_ptr(), out.data_ptr(),
                             xc.numel(), dt_of(xc), cur_stream());
    return out;
}

at::Tensor add_relu_bwd(at::Tensor gout, at::Tensor out) {
    TORCH_CHECK(gout.sizes() == out.sizes());
    auto gc = gout, oc = out;
    if (gout.dim() == 4) {
        gc = gout.contiguous(at::MemoryFormat::ChannelsLast);
        oc = out.contiguous(at::MemoryFormat::ChannelsLast);
    } else {
        gc = gout.contiguous();
        oc = out.contiguous();
    }
    auto gx = at::empty_like(gc);
    fda::add_relu_bwd_launch(gc.data_ptr(), oc.data_ptr(), gx.data_ptr(),
                             gc.numel(), dt_of(gc), cur_stream());
    return gx;
}

std::tuple<at::Tensor, at::Tensor, at::Tensor> bn_act_fwd(
    at::Tensor x, at::Tensor weight, at::Tensor bias, at::Tensor running_mean,
    at::Tensor running_var, bool training, double momentum, double eps,
    bool relu, at::Tensor residual, c10::optional<at::Tensor> conv_part) {
    auto [rows, C] = nhwc_rows(x);
    const int V = dt_of(x) == DT::BF16 ? 8 : 4;
    TORCH_CHECK(C % V == 0, "C must be divisible by ", V);
    TORCH_CHECK((C / V) >= 256 ? (C / V) % 256 == 0 : 256 % (C / V) == 0,
                "unsupported channel count ", C);
    TORCH_CHECK(weight.scalar_type() == at::kFloat, "BN params must be fp32");
    const bool has_res = residual.defined() && residual.numel() > 0;
    at::Tensor resc;
    if (has_res) resc = residual.contiguous(at::MemoryFormat::ChannelsLast);

    auto fopts = x.options().dtype(at::kFloat);
    auto ws = at::empty({4 * C}, fopts);
    auto save_mean = at::empty({C}, fopts);
    auto save_invstd = at::empty({C}, fopts);
    auto out = at::empty_like(x);
    auto stream = cur_stream();

    if (training && conv_part.has_value()) {
        // stats already accumulated by the producing conv's epilogue
        auto& cp = *conv_part;
        TORCH_CHECK(cp.dim() == 3 && cp.size(1) == 2 && cp.size(2) == C);
        const float* pp = cp.data_ptr<float>();
        int NB = (int)cp.size(0);
        at::Tensor p2;
        if (NB > 1024) {
            // layer-1-sized partial sets (~4700 m-tiles) fold at full
            // grid width first; finalize then reads <=512 rows
            const int NB2 = 512;
            p2 = at::empty({NB2, 2, C}, fopts);
            fda::bn_partial_prefold_launch(pp, p2.data_ptr<float>(), NB,
                                           NB2, (int)C, stream);
            pp = p2.data_ptr<float>();
            NB = NB2;
        }
        fda::bn_finalize_from_partials_launch(
            pp, NB, weight.data_ptr<float>(),
            bias.data_ptr<float>(), running_mean.data_ptr<float>(),
            running_var.data_ptr<float>(), save_mean.data_ptr<float>(),
            save_invstd.data_ptr<float>(), ws.data_ptr<float>(), rows,
            (int)C, (float)momentum, (float)eps, stream);
    } else if (training) {
        TORCH_CHECK(C % 64 == 0, "training BN needs C % 64 == 0, got ", C);
        auto part = at::empty({fda::bn_stats_partial_floats((int)C, rows, dt_of(x))},
                              fopts);
        fda::bn_stats_launch(x.data_ptr(), ws.data_ptr<float>(),
                             part.data_ptr<float>(), weight.data_ptr<float>(),
                             bias.data_ptr<float>(),
                             running_mean.data_ptr<float>(),
                             running_var.data_ptr<float>(),
                             save_mean.data_ptr<float>(),
                             save_invstd.data_ptr<float>(), rows, (int)C,
                             (float)momentum, (float)eps, dt_of(x), stream);
    } else {
        fda::bn_finalize_launch(ws.data_ptr<float>(), weight.data_ptr<float>(),
                                bias.data_ptr<float>(),
                                running_mean.data_ptr<float>(),
                                running_var.data_ptr<float>(),
                                save_mean.data_ptr<float>(),
                                save_invstd.data_ptr<float>(), rows, (int)C,
                                training, (float)momentum, (float)eps, stream);
    }
    fda::bn_apply_launch(x.data_ptr(), has_res ? resc.data_ptr() : nullptr,
                         out.data_ptr(), ws.data_ptr<float>(), rows, (int)C,
                         relu, dt_of(x), stream);
    return {out, save_mean, save_invstd};
}

std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor> bn_act_bwd(
    at::Tensor gout, at::Tensor x, at::Tensor weight, at::Tensor save_mean,
    at::Tensor save_invstd, at::Tensor out, bool relu, bool training,
    c10::optional<at::Tensor> gw_out, c10::optional<at::Tensor> gb_out,
    bool want_gres) {
    // gw_out/gb_out set: fp32 flat-G slices, written += (direct grad)
    auto [rows, C] = nhwc_rows(x);
    auto gc = gout.contiguous(at::MemoryFormat::ChannelsLast);
    auto fopts = x.options().dtype(at::kFloat);
    auto ws = at::empty({4 * C}, fopts);
    const bool direct = gw_out.has_value() && gw_out->numel() == C;
    auto gw = direct ? *gw_out : at::empty({C}, fopts);
    auto gb = direct ? *gb_out : at::empty({C}, fopts);
    auto gx = at::empty_like(x);
    auto stream = cur_stream();

    TORCH_CHECK(C % 64 == 0, "BN bwd needs C % 64 == 0, got ", C);
    auto part = at::empty({fda::bn_stats_partial_floats((int)C, rows, dt_of(x))},
                          fopts);
    fda::bn_bwd_stats_launch(gc.data_ptr(), x.data_ptr(), out.data_ptr(),
                             save_mean.data_ptr<float>(),
                             save_invstd.data_ptr<float>(), ws.data_ptr<float>(),
                             part.data_ptr<float>(), gw.data_ptr<float>(),
                             gb.data_ptr<float>(), rows, (int)C, relu,
                             training, direct, dt_of(x), stream);
    at::Tensor gres;
    if (want_gres) gres = at::empty_like(x);
    fda::bn_bwd_apply_launch(gc.data_ptr(), x.data_ptr(), out.data_ptr(),
                             save_mean.data_ptr<float>(),
                             save_invstd.data_ptr<float>(),
                             weight.data_ptr<float>(), ws.data_ptr<float>(),
                             gx.data_ptr(),
                             want_gres ? gres.data_ptr() : nullptr,
                             rows, (int)C, relu, training, dt_of(x), stream);
    return {gx, gw, gb, gres};
}

std::tuple<at::Tensor, at::Tensor> maxpool_fwd(at::Tensor x, int64_t KH,
                                               int64_t KW, int64_t S,
                                               int64_t P) {
    auto [rows, C] = nhwc_rows(x);
    (void)rows;
    const int V = dt_of(x) == DT::BF16 ? 8 : 4;
    TORCH_CHECK(C % V == 0);
    const int N = (int)x.size(0), H = (int)x.size(2), W = (int)x.size(3);
    const int HO = (int)((H + 2 * P - KH) / S + 1);
    const int WO = (int)((W + 2 * P - KW) / S + 1);
    auto out = at::empty({N, C, HO, WO},
                         x.options().memory_format(at::MemoryFormat::ChannelsLast));
    auto idx = at::empty({(int64_t)N * HO * WO * C},
                         x.options().dtype(at::kByte));
    fda::maxpool_fwd_launch(x.data_ptr(), out.data_ptr(),
                            idx.data_ptr<uint8_t>(), N, H, W, (int)C, HO, WO,
                            (int)KH, (int)KW, (int)S, (int)P, dt_of(x),
                            cur_stream());
    return {out, idx};
}

at::Tensor maxpool_bwd(at::Tensor gout, at::Tensor idx, int64_t H, int64_t W,
                       int64_t KH, int64_t KW, int64_t S, int64_t P) {
    auto gc = gout.contiguous(at::MemoryFormat::ChannelsLast);
    const int N = (int)gc.size(0), C = (int)gc.size(1);
    const int HO = (int)gc.size(2), WO = (int)gc.size(3);
    auto gx = at::empty({N, C, (int)H, (int)W},
                        gc.options().memory_format(at::MemoryFormat::ChannelsLast));
    fda::maxpool_bwd_launch(gc.data_ptr(), idx.data_ptr<uint8_t>(),
                            gx.data_ptr(), N, (int)H, (int)W, C, HO, WO,
                            (int)KH, (int)KW, (int)S, (int)P, dt_of(gc),
                            cur_stream());
    return gx;
}

void sgd_step(at::Tensor P, at::Tensor G, at::Tensor M, at::Tensor V,
              double lr, double mom, double wd, bool nesterov) {
    TORCH_CHECK(P.is_contiguous() && G.is_contiguous() && V.is_contiguous());
    const int vw = dt_of(P) == DT::BF16 ? 8 : 4;
    TORCH_CHECK(P.numel() % vw == 0, "flat buffer must be padded to ", vw);
    const bool has_master = M.data_ptr() != P.data_ptr();
    if (dt_of(P) == DT::BF16) TORCH_CHECK(has_master, "bf16 params need fp32 master");
    fda::sgd_step_launch(P.data_ptr(), G.data_ptr(), M.data_ptr<float>(),
                         V.data_ptr<float>(), P.numel(), (float)lr, (float)mom,
                         (float)wd, nesterov, has_master, dt_of(P),
                         cur_stream());
}

void adam_step(at::Tensor P, at::Tensor G, at::Tensor M, at::Tensor V,
               at::Tensor S, double lr, double b1, double b2, double eps,
               double wd, double bc1, double bc2) {
    const int vw = dt_of(P) == DT::BF16 ? 8 : 4;
    TORCH_CHECK(P.numel() % vw == 0, "flat buffer must be padded to ", vw);
    const bool has_master = M.data_ptr() != P.data_ptr();
    if (dt_of(P) == DT::BF16) TORCH_CHECK(has_master, "bf16 params need fp32 master");
    fda::adam_step_launch(P.data_ptr(), G.data_ptr(), M.data_ptr<float>(),
                          V.data_ptr<float>(), S.data_ptr<float>(), P.numel(),
                          (float)lr, (float)b1, (float)b2, (float)eps,
                          (float)wd, (float)bc1, (float)bc2, has_master,
                          dt_of(P), cur_stream());
}

}  // namespace

// ---- implicit-GEMM conv (NHWC bf16, conv_igemm.hip) -----------------------
// Tile geometry and split-K come from fda::conv_igemm_plan (the same
// function the launcher obeys), so workspace shapes always match.

// in-launch split-K: per-(m,n,zclass)-tile ticket counters, zeroed on the
// stream ahead of the launch (guide Guideline 16)
static unsigned* sk_tickets(const at::Tensor& ref, long ntiles,
                            at::Tensor& keepalive, hipStream_t stream) {
    keepalive = at::empty({ntiles}, ref.options().dtype(at::kInt));
    unsigned* p = (unsigned*)keepalive.data_ptr<int>();
    hipMemsetAsync(p, 0, sizeof(unsigned) * ntiles, stream);
    return p;
}


at::Tensor conv_igemm_fwd(at::Tensor x, at::Tensor w,
                          int64_t sy, int64_t sx, int64_t py, int64_t px) {
    TORCH_CHECK(x.dim() == 4 && w.dim() == 4);
    TORCH_CHECK(x.is_cuda() && w.is_cuda(),
                "conv_igemm: device tensors required (a CPU tensor here "
                "would fault the GPU with a host pointer)");
    TORCH_CHECK(x.scalar_type() == at::kBFloat16 && w.scalar_type() == at::kBFloat16,
                "conv_igemm: bf16 only");
    TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
                "conv_igemm: x must be channels_last");
    TORCH_CHECK(w.is_contiguous(at::MemoryFormat::ChannelsLast),
                "conv_igemm: w must be channels_last");
    const int N = (int)x.size(0), C = (int)x.size(1), H = (int)x.size(2),
              W = (int)x.size(3);
    const int K = (int)w.size(0), R = (int)w.size(2), S = (int)w.size(3);
    TORCH_CHECK((int)w.size(1) == C, "channel mismatch");
    TORCH_CHECK(C % 64 == 0 && K % 64 == 0,
                "conv_igemm: C and K must be multiples of 64");
    const int P = (H + 2 * (int)py - R) / (int)sy + 1;
    const int Q = (W + 2 * (int)px - S) / (int)sx + 1;
    auto y = at::empty({N, K, P, Q},
                       x.options().memory_format(at::MemoryFormat::ChannelsLast));
    const long M = (long)N * P * Q;
    int BM, BN, SK;
    fda::conv_igemm_plan(M, K, (long)R * S * (C / 64), 1, &BM, &BN, &SK);
    if (SK > 1) {
        auto part = at::empty({(long)SK * M * K},
                              x.options().dtype(at::kFloat));
        if (fda::conv_use_inlsk()) {
            at::Tensor cnt_t;
            unsigned* cnt = sk_tickets(x, ((M + BM - 1) / BM) * (K / BN),
                                       cnt_t, cur_stream());
            fda::conv_igemm_launch(x.data_ptr(), w.data_ptr(), y.data_ptr(),
                                   N, H, W, C, K, P, Q, R, S, (int)sy,
                                   (int)sx, (int)py, (int)px, false,
                                   cur_stream(), nullptr,
                                   part.data_ptr<float>(), SK, cnt);
            return y;
        }
        fda::conv_igemm_launch(x.data_ptr(), w.data_ptr(), y.data_ptr(),
                               N, H, W, C, K, P, Q, R, S, (int)sy, (int)sx,
                               (int)py, (int)px, false, cur_stream(), nullptr,
                               part.data_ptr<float>(), SK);
        fda::conv_skcombine_launch(part.data_ptr<float>(), y.data_ptr(),
                                   nullptr, M, K, SK,
                                   fda::conv_skcombine_blocks(M, K),
                                   cur_stream());
        return y;
    }
    fda::conv_igemm_launch(x.data_ptr(), w.data_ptr(), y.data_ptr(),
                           N, H, W, C, K, P, Q, R, S, (int)sy, (int)sx,
                           (int)py, (int)px, /*dgrad=*/false, cur_stream());
    return y;
}

std::tuple<at::Tensor, at::Tensor> conv_igemm_fwd_stats(
    at::Tensor x, at::Tensor w, int64_t sy, int64_t sx, int64_t py,
    int64_t px) {
    // forward + per-m-tile BN partials [mtiles][2][K] (sum/sumsq of the
    // rounded output) — feeds bn_finalize_from_partials.
    TORCH_CHECK(x.dim() == 4 && w.dim() == 4);
    TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
                x.is_contiguous(at::MemoryFormat::ChannelsLast));
    TORCH_CHECK(w.is_contiguous(at::MemoryFormat::ChannelsLast));
    const int N = (int)x.size(0), C = (int)x.size(1), H = (int)x.size(2),
              W = (int)x.size(3);
    const int K = (int)w.size(0), R = (int)w.size(2), S = (int)w.size(3);
    TORCH_CHECK(C % 64 == 0 && K % 64 == 0);
    const int P = (H + 2 * (int)py - R) / (int)sy + 1;
    const int Q = (W + 2 * (int)px - S) / (int)sx + 1;
    const long M = (long)N * P * Q;
    auto y = at::empty({N, K, P, Q},
                       x.options().memory_format(at::MemoryFormat::ChannelsLast));
    int BM, BN, SK;
    fda::conv_igemm_plan(M, K, (long)R * S * (C / 64), 1, &BM, &BN, &SK);
    if (SK > 1) {
        auto skp = at::empty({(long)SK * M * K}, x.options().dtype(at::kFloat));
        if (fda::conv_use_inlsk()) {
            // reducer blocks write the stats partials like a SK=1 launch:
            // [mtiles][2][K] with mtiles = grid.x
            const long mtiles = (M + BM - 1) / BM;
            auto part = at::empty({mtiles, 2, K}, x.options().dtype(at::kFloat));
            at::Tensor cnt_t;
            unsigned* cnt = sk_tickets(x, mtiles * (K / BN), cnt_t,
                                       cur_stream());
            fda::conv_igemm_launch(x.data_ptr(), w.data_ptr(), y.data_ptr(),
                                   N, H, W, C, K, P, Q, R, S, (int)sy,
                                   (int)sx, (int)py, (int)px, false,
                                   cur_stream(), part.data_ptr<float>(),
                                   skp.data_ptr<float>(), SK, cnt);
            return {y, part};
        }
        const int nb = fda::conv_skcombine_blocks(M, K);
        auto part = at::empty({nb, 2, K}, x.options().dtype(at::kFloat));
        fda::conv_igemm_launch(x.data_ptr(), w.data_ptr(), y.data_ptr(),
                               N, H, W, C, K, P, Q, R, S, (int)sy, (int)sx,
                               (int)py, (int)px, false, cur_stream(), nullptr,
                               skp.data_ptr<float>(), SK);
        fda::conv_skcombine_launch(skp.data_ptr<float>(), y.data_ptr(),
                                   part.data_ptr<float>(), M, K, SK, nb,
                                   cur_stream());
        return {y, part};
    }
    const long mtiles = (M + BM - 1) / BM;
    auto part = at::empty({mtiles, 2, K}, x.options().dtype(at::kFloat));
    fda::conv_igemm_launch(x.data_ptr(), w.data_ptr(), y.data_ptr(),
                           N, H, W, C, K, P, Q, R, S, (int)sy, (int)sx,
                           (int)py, (int)px, /*dgrad=*/false, cur_stream(),
                           part.data_ptr<float>());
    return {y, part};
}

at::Tensor conv_igemm_dgrad(at::Tensor dy, at::Tensor wt,
                            int64_t C, int64_t H, int64_t W,
                            int64_t R, int64_t S,
                            int64_t sy, int64_t sx, int64_t py, int64_t px,
                            c10::optional<at::Tensor> accum) {
    // accum: optional bf16 tensor in dx's layout added into the result in
    // the epilogue (residual-junction grad fusion, ops/conv.py).
    // dy: [N,K,P,Q] channels_last; wt: [R*S*C, K] row-major (pre-transposed
    // weight, k contiguous). Output dx: [N,C,H,W] channels_last.
    TORCH_CHECK(dy.is_cuda() && wt.is_cuda());
    TORCH_CHECK(dy.dim() == 4 && dy.scalar_type() == at::kBFloat16);
    TORCH_CHECK(dy.is_contiguous(at::MemoryFormat::ChannelsLast));
    TORCH_CHECK(wt.dim() == 2 && wt.is_contiguous() &&
                wt.scalar_type() == at::kBFloat16);
    const int N = (int)dy.size(0), K = (int)dy.size(1), P = (int)dy.size(2),
              Q = (int)dy.size(3);
    TORCH_CHECK(wt.size(0) == R * S * C && wt.size(1) == K, "wt shape mismatch");
    TORCH_CHECK(C % 64 == 0 && K % 64 == 0,
                "conv_igemm: C and K must be multiples of 64");
    auto dx = at::empty({N, C, H, W},
                        dy.options().memory_format(at::MemoryFormat::ChannelsLast));
    const void* accp = nullptr;
    if (accum.has_value() && accum->defined()) {
        TORCH_CHECK(accum->is_contiguous(at::MemoryFormat::ChannelsLast) &&
                    accum->scalar_type() == at::kBFloat16 &&
                    accum->numel() == dx.numel(),
                    "dgrad accum tensor must be channels_last bf16 of dx's shape");
        accp = accum->data_ptr();
    }
    const long M = (long)N * H * W;
    int BM, BN, SK = 1;
    if (sy == 1 && sx == 1)                 // parity classes already fan out
        fda::conv_igemm_plan(M, (int)C, (long)R * S * (K / 64), 1,
                             &BM, &BN, &SK);
    if (SK > 1) {
        auto skp = at::empty({(long)SK * M * C},
                             dy.options().dtype(at::kFloat));
        if (fda::conv_use_inlsk() && accp == nullptr) {
            at::Tensor cnt_t;
            unsigned* cnt = sk_tickets(dy, ((M + BM - 1) / BM) * (C / BN),
                                       cnt_t, cur_stream());
            fda::conv_igemm_launch(dy.data_ptr(), wt.data_ptr(),
                                   dx.data_ptr(), N, (int)H, (int)W, (int)C,
                                   K, P, Q, (int)R, (int)S, (int)sy, (int)sx,
                                   (int)py, (int)px, true, cur_stream(),
                                   nullptr, skp.data_ptr<float>(), SK, cnt);
            return dx;
        }
        fda::conv_igemm_launch(dy.data_ptr(), wt.data_ptr(), dx.data_ptr(),
                               N, (int)H, (int)W, (int)C, K, P, Q, (int)R,
                               (int)S, (int)sy, (int)sx, (int)py, (int)px,
                               true, cur_stream(), nullptr,
                               skp.data_ptr<float>(), SK);
        fda::conv_skcombine_launch(skp.data_ptr<float>(), dx.data_ptr(),
                                   nullptr, M, (int)C, SK,
                                   fda::conv_skcombine_blocks(M, (int)C),
                                   cur_stream(), accp);
        return dx;
    }
    fda::conv_igemm_launch(dy.data_ptr(), wt.data_ptr(), dx.data_ptr(),
                           N, (int)H, (int)W, (int)C, K, P, Q, (int)R, (int)S,
                           (int)sy, (int)sx, (int)py, (int)px, /*dgrad=*/true,
                           cur_stream(), nullptr, nullptr, 1, nullptr, accp);
    return dx;
}

at::Tensor conv_igemm_wgrad(at::Tensor dy, at::Tensor x,
                            int64_t R, int64_t S,
                            int64_t sy, int64_t sx, int64_t py, int64_t px) {
    // dy: [N,K,P,Q] channels_last bf16; x: [N,C,H,W] channels_last bf16.
    // Returns ws[K][R*S*C] fp32 (the channels_last weight-grad memory
    // layout [K][R][S][C] flattened).
    TORCH_CHECK(dy.is_cuda() && x.is_cuda());
    TORCH_CHECK(dy.scalar_type() == at::kBFloat16 &&
                x.scalar_type() == at::kBFloat16);
    TORCH_CHECK(dy.is_contiguous(at::MemoryFormat::ChannelsLast) &&
                x.is_contiguous(at::MemoryFormat::ChannelsLast));
    const int N = (int)x.size(0), C = (int)x.size(1), H = (int)x.size(2),
              W = (int)x.size(3);
    const int K = (int)dy.size(1), P = (int)dy.size(2), Q = (int)dy.size(3);
    TORCH_CHECK(C % 64 == 0 && K % 64 == 0);
    auto ws = at::zeros({K, R * S * C}, x.options().dtype(at::kFloat));
    if (fda::conv_wgrad_two_phase()) {
        int ft, mch, nch;
        fda::conv_wgrad_plan((long)N * P * Q, C, K, (int)R, (int)S,
                             &ft, &mch, &nch);
        auto part = at::empty({(int64_t)nch, (int64_t)K * R * S * C},
                              x.options().dtype(at::kFloat));
        fda::conv_wgrad_launch(dy.data_ptr(), x.data_ptr(),
                               ws.data_ptr<float>(), N, H, W, C, K, P, Q,
                               (int)R, (int)S, (int)sy, (int)sx, (int)py,
                               (int)px, cur_stream(),
                               part.data_ptr<float>());
        fda::wgrad_combine_launch(ws.data_ptr<float>(),
                                  part.data_ptr<float>(),
                                  (long)K * R * S * C, nch, cur_stream());
        return ws;
    }
    fda::conv_wgrad_launch(dy.data_ptr(), x.data_ptr(), ws.data_ptr<float>(),
                           N, H, W, C, K, P, Q, (int)R, (int)S, (int)sy,
                           (int)sx, (int)py, (int)px, cur_stream());
    return ws;
}

void conv_igemm_wgrad_into(at::Tensor dy, at::Tensor x, at::Tensor ws,
                           int64_t R, int64_t S,
                           int64_t sy, int64_t sx, int64_t py, int64_t px) {
    // Accumulating variant: ws is a PRE-ZEROED [K, R*S*C] fp32 slice of the
    // step-scoped wgrad arena (ops/conv.py) — saves the per-layer zeros
    // launch.
    TORCH_CHECK(dy.scalar_type() == at::kBFloat16 &&
                x.scalar_type() == at::kBFloat16);
    TORCH_CHECK(dy.is_contiguous(at::MemoryFormat::ChannelsLast) &&
                x.is_contiguous(at::MemoryFormat::ChannelsLast));
    TORCH_CHECK(ws.scalar_type() == at::kFloat && ws.is_contiguous());
    const int N = (int)x.size(0), C = (int)x.size(1), H = (int)x.size(2),
              W = (int)x.size(3);
    const int K = (int)dy.size(1), P = (int)dy.size(2), Q = (int)dy.size(3);
    TORCH_CHECK(C % 64 == 0 && K % 64 == 0);
    TORCH_CHECK(ws.numel() == (int64_t)K * R * S * C);
    if (fda::conv_wgrad_two_phase()) {
        int ft, mch, nch;
        fda::conv_wgrad_plan((long)N * P * Q, C, K, (int)R, (int)S,
                             &ft, &mch, &nch);
        auto part = at::empty({(int64_t)nch, (int64_t)K * R * S * C},
                              x.options().dtype(at::kFloat));
        fda::conv_wgrad_launch(dy.data_ptr(), x.data_ptr(),
                               ws.data_ptr<float>(), N, H, W, C, K, P, Q,
                               (int)R, (int)S, (int)sy, (int)sx, (int)py,
                               (int)px, cur_stream(),
                               part.data_ptr<float>());
        fda::wgrad_combine_launch(ws.data_ptr<float>(),
                                  part.data_ptr<float>(),
                                  (long)K * R * S * C, nch, cur_stream());
        return;
    }
    fda::conv_wgrad_launch(dy.data_ptr(), x.data_ptr(), ws.data_ptr<float>(),
                           N, H, W, C, K, P, Q, (int)R, (int)S, (int)sy,
                           (int)sx, (int)py, (int)px, cur_stream());
}

at::Tensor conv_stem_fwd(at::Tensor x8, at::Tensor wpad, int64_t R,
                         int64_t sy, int64_t sx, int64_t P, int64_t Q) {
    // x8: [N,8,Hp,Wp] channels_last bf16 (spatially pre-padded, channels
    // 3..7 zero); wpad: [K][R][64] bf16 (s==7 and c>=3 taps zero).
    TORCH_CHECK(x8.is_cuda() && wpad.is_cuda());
    TORCH_CHECK(x8.scalar_type() == at::kBFloat16 &&
                x8.is_contiguous(at::MemoryFormat::ChannelsLast));
    TORCH_CHECK(x8.size(1) == 8);
    TORCH_CHECK(wpad.is_contiguous() && wpad.scalar_type() == at::kBFloat16);
    const int N = (int)x8.size(0), Hp = (int)x8.size(2), Wp = (int)x8.size(3);
    const int K = (int)wpad.size(0);
    TORCH_CHECK(K % 64 == 0 && wpad.size(1) == R && wpad.size(2) == 64);
    auto y = at::empty({N, K, P, Q},
                       x8.options().memory_format(at::MemoryFormat::ChannelsLast));
    fda::conv_stem_fwd_launch(x8.data_ptr(), wpad.data_ptr(), y.data_ptr(),
                              N, Hp, Wp, K, (int)P, (int)Q, (int)R, (int)sy,
                              (int)sx, cur_stream());
    return y;
}

std::tuple<at::Tensor, at::Tensor> conv_stem_fwd_stats(
    at::Tensor x8, at::Tensor wpad, int64_t R, int64_t sy, int64_t sx,
    int64_t P, int64_t Q) {
    TORCH_CHECK(x8.size(1) == 8 &&
                x8.is_contiguous(at::MemoryFormat::ChannelsLast));
    const int N = (int)x8.size(0), Hp = (int)x8.size(2), Wp = (int)x8.size(3);
    const int K = (int)wpad.size(0);
    const long M = (long)N * P * Q;
    const long mtiles = (M + 256 - 1) / 256;
    auto y = at::empty({N, K, P, Q},
                       x8.options().memory_format(at::MemoryFormat::ChannelsLast));
    auto part = at::empty({mtiles, 2, K}, x8.options().dtype(at::kFloat));
    fda::conv_stem_fwd_launch(x8.data_ptr(), wpad.data_ptr(), y.data_ptr(),
                              N, Hp, Wp, K, (int)P, (int)Q, (int)R, (int)sy,
                              (int)sx, cur_stream(), part.data_ptr<float>());
    return {y, part};
}

at::Tensor conv_stem_wgrad(at::Tensor dy, at::Tensor x8, int64_t R,
                           int64_t sy, int64_t sx) {
    TORCH_CHECK(dy.scalar_type() == at::kBFloat16 &&
                dy.is_contiguous(at::MemoryFormat::ChannelsLast));
    TORCH_CHECK(x8.scalar_type() == at::kBFloat16 &&
                x8.is_contiguous(at::MemoryFormat::ChannelsLast));
    const int N = (int)x8.size(0), Hp = (int)x8.size(2), Wp = (int)x8.size(3);
    const int K = (int)dy.size(1), P = (int)dy.size(2), Q = (int)dy.size(3);
    auto ws = at::zeros({K, R * 64}, x8.options().dtype(at::kFloat));
    fda::conv_stem_wgrad_launch(dy.data_ptr(), x8.data_ptr(),
                                ws.data_ptr<float>(), N, Hp, Wp, K, P, Q,
                                (int)R, (int)sy, (int)sx, cur_stream());
    return ws;   // [K][R*64]; host slices [K][R][s<7][c<3]
}

at::Tensor pad_rows_bf16(at::Tensor src, int64_t ldl) {
    // [M][C] bf16 -> new [M][ldl] with zero right-pad (FC-head padding)
    TORCH_CHECK(src.dim() == 2 && src.is_contiguous() &&
                src.scalar_type() == at::kBFloat16);
    const long M = src.size(0);
    const int C = (int)src.size(1);
    TORCH_CHECK(C % 8 == 0 && ldl % 8 == 0 && ldl >= C);
    auto dst = at::empty({M, ldl}, src.options());
    fda::pad_rows_bf16_launch(dst.data_ptr(), src.data_ptr(), M, C, (int)ldl,
                              cur_stream());
    return dst;
}

void pad_rows_bf16_into(at::Tensor dst, at::Tensor src) {
    TORCH_CHECK(dst.dim() == 2 && src.dim() == 2);
    TORCH_CHECK(dst.is_contiguous() && src.is_contiguous());
    TORCH_CHECK(dst.size(0) == src.size(0));
    fda::pad_rows_bf16_launch(dst.data_ptr(), src.data_ptr(), src.size(0),
                              (int)src.size(1), (int)dst.size(1),
                              cur_stream());
}

void bias_add_rows_bf16(at::Tensor y, at::Tensor bias) {
    TORCH_CHECK(y.dim() == 2 && y.is_contiguous() &&
                y.scalar_type() == at::kBFloat16);
    TORCH_CHECK(bias.numel() == y.size(1) && bias.is_contiguous() &&
                bias.scalar_type() == at::kBFloat16);
    fda::bias_add_rows_bf16_launch(y.data_ptr(), bias.data_ptr(), y.size(0),
                                   (int)y.size(1), cur_stream());
}

void colsum_accum_bf16(at::Tensor g, at::Tensor dy) {
    // g[c] += cast(sum_m dy[m][c]) for c < g.numel() (bias direct-grad)
    TORCH_CHECK(dy.dim() == 2 && dy.is_contiguous() &&
                dy.scalar_type() == at::kBFloat16);
    TORCH_CHECK(g.is_contiguous() && g.scalar_type() == at::kBFloat16);
    TORCH_CHECK(g.numel() <= dy.size(1));
    fda::colsum_accum_bf16_launch(g.data_ptr(), dy.data_ptr(), dy.size(0),
                                  (int)dy.size(1), (int)g.numel(),
                                  cur_stream());
}

void grad_accum_batch(at::Tensor g_ptrs, at::Tensor ws_ptrs, at::Tensor ns,
                      int64_t max_n) {
    // device int64 tensors: g/ws addresses and lengths per slice
    TORCH_CHECK(g_ptrs.is_cuda() && g_ptrs.scalar_type() == at::kLong);
    TORCH_CHECK(ws_ptrs.numel() == g_ptrs.numel() &&
                ns.numel() == g_ptrs.numel());
    fda::grad_accum_batch_launch(g_ptrs.data_ptr<int64_t>(),
                                 ws_ptrs.data_ptr<int64_t>(),
                                 ns.data_ptr<int64_t>(),
                                 (int)g_ptrs.numel(), max_n, cur_stream());
}

void grad_accum_bf16(at::Tensor g, at::Tensor ws) {
    TORCH_CHECK(g.scalar_type() == at::kBFloat16 && g.is_contiguous());
    TORCH_CHECK(ws.scalar_type() == at::kFloat && ws.is_contiguous());
    TORCH_CHECK(g.numel() == ws.numel());
    fda::grad_accum_bf16_launch(g.data_ptr(), ws.data_ptr<float>(),
                                (long)g.numel(), cur_stream());
}

void wt_transpose_batch(at::Tensor src_ptrs, at::Tensor dst_ptrs,
                        at::Tensor Ks, at::Tensor RCs, at::Tensor tile_counts,
                        int64_t max_tiles) {
    TORCH_CHECK(src_ptrs.is_cuda() && src_ptrs.scalar_type() == at::kLong);
    const int n = (int)src_ptrs.numel();
    fda::wt_transpose_launch(src_ptrs.data_ptr<int64_t>(),
                             dst_ptrs.data_ptr<int64_t>(),
                             Ks.data_ptr<int>(), RCs.data_ptr<int>(),
                             tile_counts.data_ptr<int>(), n, (int)max_tiles,
                             cur_stream());
}

at::Tensor gap_fwd(at::Tensor x) {
    auto [rows, C] = nhwc_rows(x);
    const int N = (int)x.size(0);
    const int HW = (int)(rows / N);
    auto y = at::empty({N, C}, x.options());
    fda::gap_fwd_launch(x.data_ptr(), y.data_ptr(), N, HW, (int)C, dt_of(x),
                        cur_stream());
    return y;
}

at::Tensor gap_bwd(at::Tensor gy, int64_t H, int64_t W) {
    TORCH_CHECK(gy.dim() == 2 && gy.is_contiguous());
    const int N = (int)gy.size(0), C = (int)gy.size(1);
    auto gx = at::empty({N, C, H, W},
                        gy.options().memory_format(at::MemoryFormat::ChannelsLast));
    fda::gap_bwd_launch(gy.data_ptr(), gx.data_ptr(), N, (int)(H * W), C,
                        dt_of(gy), cur_stream());
    return gx;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("ce_fwd", &ce_fwd, "fused logit cross-entropy fwd (loss + dlogits)");
    m.def("add_relu_fwd", &add_relu_fwd);
    m.def("add_relu_bwd", &add_relu_bwd);
    m.def("bn_act_fwd", &bn_act_fwd, pybind11::arg("x"),
          pybind11::arg("weight"), pybind11::arg("bias"),
          pybind11::arg("running_mean"), pybind11::arg("running_var"),
          pybind11::arg("training"), pybind11::arg("momentum"),
          pybind11::arg("eps"), pybind11::arg("relu"),
          pybind11::arg("residual"),
          pybind11::arg("conv_part") = pybind11::none());
    m.def("bn_act_bwd", &bn_act_bwd, pybind11::arg("gout"),
          pybind11::arg("x"), pybind11::arg("weight"),
          pybind11::arg("save_mean"), pybind11::arg("save_invstd"),
          pybind11::arg("out"), pybind11::arg("relu"),
          pybind11::arg("training"),
          pybind11::arg("gw_out") = pybind11::none(),
          pybind11::arg("gb_out") = pybind11::none(),
          pybind11::arg("want_gres") = false);
    m.def("maxpool_fwd", &maxpool_fwd);
    m.def("maxpool_bwd", &maxpool_bwd);
    m.def("gap_fwd", &gap_fwd);
    m.def("gap_bwd", &gap_bwd);
    m.def("sgd_step", &sgd_step);
    m.def("adam_step", &adam_step);
    m.def("conv_igemm_fwd_stats", &conv_igemm_fwd_stats);
    m.def("conv_stem_fwd_stats", &conv_stem_fwd_stats);
    m.def("conv_igemm_fwd", &conv_igemm_fwd,
          "implicit-GEMM conv fwd (NHWC bf16, MFMA)");
    m.def("wt_transpose_batch", &wt_transpose_batch);
    m.def("grad_accum_bf16", &grad_accum_bf16);
    m.def("grad_accum_batch", &grad_accum_batch);
    m.def("pad_rows_bf16", &pad_rows_bf16);
    m.def("pad_rows_bf16_into", &pad_rows_bf16_into);
    m.def("bias_add_rows_bf16", &bias_add_rows_bf16);
    m.def("colsum_accum_bf16", &colsum_accum_bf16);
    m.def("conv_stem_fwd", &conv_stem_fwd);
    m.def("conv_stem_wgrad", &conv_stem_wgrad);
    m.def("conv_igemm_wgrad_into", &conv_igemm_wgrad_into);
    m.def("conv_igemm_wgrad", &conv_igemm_wgrad,
          "implicit-GEMM conv weight-grad (NHWC bf16, MFMA + tr16 reads)");
    m.def("conv_igemm_dgrad", &conv_igemm_dgrad,
          "implicit-GEMM conv input-grad (NHWC bf16, MFMA; optional fused "
          "+= accum epilogue)",
          pybind11::arg("dy"), pybind11::arg("wt"), pybind11::arg("C"),
          pybind11::arg("H"), pybind11::arg("W"), pybind11::arg("R"),
          pybind11::arg("S"), pybind11::arg("sy"), pybind11::arg("sx"),
          pybind11::arg("py"), pybind11::arg("px"),
          pybind11::arg("accum") = pybind11::none());
    m.attr("_built_for") = "gfx950";
}
