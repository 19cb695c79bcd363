// Torch bindings for the raft_amd gfx950 kernels (native HIP, no CUDA path).
// Host-side plumbing only: tensor checks, allocation, launch on the current
// HIP stream. Device code lives in the .hip files; each exposes an
// extern "C" launcher so this TU needs no cross-TU kernel symbols.
#include <torch/extension.h>

#include <ATen/hip/HIPContext.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime_api.h>

#include <cmath>
#include <vector>

extern "C" {
void launch_corr_volume_f32(const float*, const float*, float*, int, int,
                            int, int, float, hipStream_t);
void launch_corr_pool2x_f32(const float*, float*, int, int, int, int,
                            long long, hipStream_t);
void launch_corr_lookup_fwd_f32(const float* const*, const int*, const int*,
                                const float*, float*, int, int, int, int,
                                int, hipStream_t);
void launch_corr_lookup_bwd_f32(float* const*, const int*, const int*,
                                const float*, const float*, int, int, int,
                                int, int, hipStream_t);
void launch_gru_gates_fwd_f32(const float*, const float*, const float*,
                              float*, long long, hipStream_t);
void launch_gru_gates_bwd_f32(const float*, const float*, const float*,
                              const float*, float*, float*, float*,
                              long long, hipStream_t);
void launch_convex_upsample_fwd_f32(const float*, const float*, float*, int,
                                    int, int, hipStream_t);
void launch_convex_upsample_bwd_f32(const float*, const float*, const float*,
                                    float*, float*, int, int, int,
                                    hipStream_t);
}

namespace {

hipStream_t current_stream() {
    return c10::hip::getCurrentHIPStream().stream();
}

#define CHECK_DEV(x) TORCH_CHECK((x).is_cuda(), #x " must be on GPU")
#define CHECK_CONT(x) TORCH_CHECK((x).is_contiguous(), #x " must be contiguous")

at::Tensor to_f32(const at::Tensor& t) {
    return t.scalar_type() == at::kFloat ? t : t.to(at::kFloat);
}

at::Tensor corr_volume(at::Tensor fmap1, at::Tensor fmap2) {
    CHECK_DEV(fmap1); CHECK_DEV(fmap2);
    CHECK_CONT(fmap1); CHECK_CONT(fmap2);
    TORCH_CHECK(fmap1.sizes() == fmap2.sizes(), "fmap shape mismatch");
    auto f1 = to_f32(fmap1);
    auto f2 = to_f32(fmap2);
    const int B = f1.size(0), C = f1.size(1), H = f1.size(2), W = f1.size(3);
    const int M = H * W;
    auto out = at::empty({B, M, H, W}, f1.options());
    launch_corr_volume_f32(f1.data_ptr<float>(), f2.data_ptr<float>(),
                           out.data_ptr<float>(), B, M, M, C,
                           1.0f / std::sqrt((float)C), current_stream());
    return out;
}

at::Tensor corr_pool2x(at::Tensor corr) {
    CHECK_DEV(corr); CHECK_CONT(corr);
    TORCH_CHECK(corr.scalar_type() == at::kFloat, "corr must be fp32");
    const int B = corr.size(0), Q = corr.size(1);
    const int H = corr.size(2), W = corr.size(3);
    const int Ho = H / 2, Wo = W / 2;
    auto out = at::empty({B, Q, Ho, Wo}, corr.options());
    const long long total = (long long)B * Q * Ho * Wo;
    launch_corr_pool2x_f32(corr.data_ptr<float>(), out.data_ptr<float>(), H,
                           W, Ho, Wo, total, current_stream());
    return out;
}

at::Tensor corr_lookup(std::vector<at::Tensor> levels, at::Tensor coords,
                       int64_t radius) {
    TORCH_CHECK(!levels.empty() && levels.size() <= 4, "1..4 pyramid levels");
    CHECK_DEV(coords); CHECK_CONT(coords);
    TORCH_CHECK(coords.scalar_type() == at::kFloat, "coords must be fp32");
    const int B = coords.size(0), H = coords.size(1), W = coords.size(2);
    const int L = (int)levels.size();
    const int K = 2 * (int)radius + 1;
    const float* ptrs[4] = {nullptr, nullptr, nullptr, nullptr};
    int hs[4] = {0}, ws[4] = {0};
    for (int i = 0; i < L; ++i) {
        CHECK_DEV(levels[i]); CHECK_CONT(levels[i]);
        TORCH_CHECK(levels[i].scalar_type() == at::kFloat,
                    "pyramid must be fp32");
        TORCH_CHECK(levels[i].size(1) == (int64_t)H * W,
                    "level query dim mismatch");
        ptrs[i] = levels[i].data_ptr<float>();
        hs[i] = levels[i].size(2);
        ws[i] = levels[i].size(3);
    }
    auto out = at::empty({B, (int64_t)L * K * K, H, W}, coords.options());
    launch_corr_lookup_fwd_f32(ptrs, hs, ws, coords.data_ptr<float>(),
                               out.data_ptr<float>(), B, H, W, L,
                               (int)radius, current_stream());
    return out;
}

std::vector<at::Tensor> corr_lookup_backward(
    at::Tensor grad_out, at::Tensor coords, int64_t radius,
    std::vector<std::vector<int64_t>> level_shapes) {
    CHECK_DEV(grad_out); CHECK_CONT(grad_out);
    CHECK_DEV(coords); CHECK_CONT(coords);
    const int B = coords.size(0), H = coords.size(1), W = coords.size(2);
    const int L = (int)level_shapes.size();
    std::vector<at::Tensor> grads;
    float* ptrs[4] = {nullptr, nullptr, nullptr, nullptr};
    int hs[4] = {0}, ws[4] = {0};
    for (int i = 0; i < L; ++i) {
        grads.push_back(at::zeros(level_shapes[i],
                                  grad_out.options().dtype(at::kFloat)));
        ptrs[i] = grads[i].data_ptr<float>();
        hs[i] = (int)level_shapes[i][2];
        ws[i] = (int)level_shapes[i][3];
    }
    auto go = to_f32(grad_out);
    launch_corr_lookup_bwd_f32(ptrs, hs, ws, coords.data_ptr<float>(),
                               go.data_ptr<float>(), B, H, W, L, (int)radius,
                               current_stream());
    return grads;
}

at::Tensor gru_gates_fwd(at::Tensor h, at::Tensor z, at::Tensor q) {
    CHECK_DEV(h); CHECK_CONT(h); CHECK_CONT(z); CHECK_CONT(q);
    auto hf = to_f32(h); auto zf = to_f32(z); auto qf = to_f32(q);
    auto out = at::empty_like(hf);
    launch_gru_gates_fwd_f32(hf.data_ptr<float>(), zf.data_ptr<float>(),
                             qf.data_ptr<float>(), out.data_ptr<float>(),
                             hf.numel(), current_stream());
    return out.scalar_type() == h.scalar_type() ? out : out.to(h.scalar_type());
}

std::vector<at::Tensor> gru_gates_bwd(at::Tensor go, at::Tensor h,
                                      at::Tensor z, at::Tensor q) {
    auto gof = to_f32(go); auto hf = to_f32(h);
    auto zf = to_f32(z); auto qf = to_f32(q);
    auto gh = at::empty_like(hf);
    auto gz = at::empty_like(hf);
    auto gq = at::empty_like(hf);
    launch_gru_gates_bwd_f32(gof.data_ptr<float>(), hf.data_ptr<float>(),
                             zf.data_ptr<float>(), qf.data_ptr<float>(),
                             gh.data_ptr<float>(), gz.data_ptr<float>(),
                             gq.data_ptr<float>(), hf.numel(),
                             current_stream());
    return {gh.to(h.scalar_type()), gz.to(z.scalar_type()),
            gq.to(q.scalar_type())};
}

at::Tensor convex_upsample(at::Tensor flow, at::Tensor mask) {
    CHECK_DEV(flow); CHECK_CONT(flow); CHECK_DEV(mask); CHECK_CONT(mask);
    const int B = flow.size(0), H = flow.size(2), W = flow.size(3);
    TORCH_CHECK(mask.size(1) == 576, "mask must have 64*9 channels");
    auto ff = to_f32(flow); auto mf = to_f32(mask);
    auto out = at::empty({B, 2, 8 * H, 8 * W}, ff.options());
    launch_convex_upsample_fwd_f32(ff.data_ptr<float>(), mf.data_ptr<float>(),
                                   out.data_ptr<float>(), B, H, W,
                                   current_stream());
    return out;
}

std::vector<at::Tensor> convex_upsample_backward(at::Tensor grad_up,
                                                 at::Tensor flow,
                                                 at::Tensor mask) {
    const int B = flow.size(0), H = flow.size(2), W = flow.size(3);
    auto gf = to_f32(grad_up).contiguous();
    auto ff = to_f32(flow).contiguous();
    auto mf = to_f32(mask).contiguous();
    auto grad_flow = at::zeros_like(ff);
    auto grad_mask = at::empty_like(mf);
    launch_convex_upsample_bwd_f32(gf.data_ptr<float>(), ff.data_ptr<float>(),
                                   mf.data_ptr<float>(),
                                   grad_flow.data_ptr<float>(),
                                   grad_mask.data_ptr<float>(), B, H, W,
                                   current_stream());
    return {grad_flow.to(flow.scalar_type()), grad_mask.to(mask.scalar_type())};
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("corr_volume", &corr_volume, "all-pairs correlation volume (MFMA)");
    m.def("corr_pool2x", &corr_pool2x, "2x2/2 avg-pool one pyramid level");
    m.def("corr_lookup", &corr_lookup, "multi-scale window lookup");
    m.def("corr_lookup_backward", &corr_lookup_backward);
    m.def("gru_gates_fwd", &gru_gates_fwd, "fused GRU gate pointwise");
    m.def("gru_gates_bwd", &gru_gates_bwd);
    m.def("convex_upsample", &convex_upsample, "8x convex upsample");
    m.def("convex_upsample_backward", &convex_upsample_backward);
}
