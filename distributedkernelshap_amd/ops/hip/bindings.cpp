// Torch bindings for the CDNA4 KernelSHAP kernels (kshap_kernels.hip).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <c10/hip/HIPStream.h>

#include <cstdint>

extern "C" void launch_fill_random_masks(
    uint8_t* masks, int B, int S, int M, int ne, int n_random,
    const float* cdf, const int* sizes, int n_sizes, int num_paired,
    uint32_t seed, const int32_t* inst_ids, hipStream_t stream);

extern "C" int launch_fused_predict_linear(
    const uint8_t* masksU, const float* diff, const float* base, const float* wbg,
    float* ey, int B, int S, int M, int Mpad, int Npad, int n_out, int act,
    hipStream_t stream);

extern "C" void launch_synth_chunk(
    const uint8_t* masks, const float* x, const float* bg, const int* col_group,
    void* out, int out_bf16, int S, int M, int N, int D, int b_lo, int b_hi,
    int s_lo, int s_hi, hipStream_t stream);

// bf16 device pointers are opaque 16-bit words host-side (extern "C")
extern "C" void launch_pack_masks(
    const uint8_t* masks, uint64_t* packed, int B, int S, int M,
    hipStream_t stream);

extern "C" int launch_fused_predict_bf16(
    const uint8_t* masksU, const uint16_t* diffB,
    const float* base, const float* wbg, float* ey, int B, int S, int M,
    int Npad, int n_out, int act, int split, hipStream_t stream);

extern "C" void launch_build_diff_f32(
    const float* xp, const float* bgp, const int64_t* vidx, float* out,
    int B, int G, int O, int N, int m, int Mpad, int Npad, hipStream_t stream);

extern "C" void launch_build_diff_bf16(
    const float* xp, const float* bgp, const int64_t* vidx, uint16_t* out,
    int B, int G, int O, int N, int m, int Npad, int split, hipStream_t stream);

extern "C" int launch_wls_solve(
    const uint8_t* masks, const uint64_t* packed, const float* kw,
    const float* ey_adj, const float* total, float* phi, int B, int S, int M,
    int n_out, hipStream_t stream);

extern "C" void launch_pack_masks_words(
    const uint8_t* masks, uint64_t* packed, int B, int S, int M, int W,
    hipStream_t stream);

extern "C" int launch_wls_gram(
    const uint64_t* packed, const float* kw, const float* ey_adj,
    const float* total, double* A64, double* rhs64, int B, int S, int M,
    int W, int n_out, hipStream_t stream);

extern "C" int launch_fused_predict_tiled(
    const uint8_t* masksU, const float* diff, const float* base,
    const float* wbg, float* partial, float* ey, int B, int S, int M,
    int Mpad, int Npad, int n_out, int act, hipStream_t stream);

extern "C" void launch_build_diff_bf16_tiled(
    const float* xp, const float* bgp, const int64_t* vidx, uint16_t* out,
    int B, int G, int O, int N, int m, int Mpad, int Npad, int split,
    hipStream_t stream);

extern "C" int launch_fused_predict_tiled_bf16(
    const uint8_t* masksU, const uint16_t* diffB, const float* base,
    const float* wbg, float* partial, float* ey, int B, int S, int M,
    int Mpad, int Npad, int n_out, int act, int split, hipStream_t stream);

namespace {

#define CHECK_DEV(t) TORCH_CHECK((t).is_cuda() && (t).is_contiguous(), #t " must be a contiguous device tensor")

hipStream_t current_stream() {
    return c10::hip::getCurrentHIPStream().stream();
}

void fill_random_masks(
    torch::Tensor masks, int64_t ne, int64_t n_random,
    torch::Tensor cdf, torch::Tensor sizes, int64_t num_paired, int64_t seed,
    torch::Tensor inst_ids) {
    CHECK_DEV(masks); CHECK_DEV(cdf); CHECK_DEV(sizes); CHECK_DEV(inst_ids);
    TORCH_CHECK(masks.dtype() == torch::kUInt8, "masks must be uint8");
    TORCH_CHECK(masks.dim() == 3, "masks must be (B,S,M)");
    TORCH_CHECK(inst_ids.dtype() == torch::kInt32, "inst_ids must be int32");
    int B = masks.size(0), S = masks.size(1), M = masks.size(2);
    TORCH_CHECK(inst_ids.size(0) == B, "inst_ids length");
    TORCH_CHECK(M <= 256, "fill_random_masks supports M <= 256");
    launch_fill_random_masks(
        masks.data_ptr<uint8_t>(), B, S, M, (int)ne, (int)n_random,
        cdf.data_ptr<float>(), sizes.data_ptr<int>(), (int)cdf.size(0),
        (int)num_paired, (uint32_t)seed, inst_ids.data_ptr<int32_t>(),
        current_stream());
}

void fused_predict_linear(
    torch::Tensor masks, torch::Tensor diff, torch::Tensor base,
    torch::Tensor wbg, torch::Tensor ey, int64_t act) {
    CHECK_DEV(masks); CHECK_DEV(diff); CHECK_DEV(base); CHECK_DEV(wbg); CHECK_DEV(ey);
    TORCH_CHECK(masks.dtype() == torch::kUInt8, "masks must be u8 (B,S,M)");
    int B = masks.size(0), S = masks.size(1), M = masks.size(2);
    int Mpad = diff.size(2), Npad = diff.size(3);
    int n_out = ey.size(2);                    // act 3 uses 1 diff image for 2 outputs
    int oimg = (act == 3) ? 1 : n_out;
    TORCH_CHECK(diff.size(1) == oimg, "diff image count vs act");
    TORCH_CHECK(M <= Mpad, "M vs Mpad");
    TORCH_CHECK(ey.size(0) == B && ey.size(1) == S, "ey shape");
    TORCH_CHECK(base.size(0) == oimg && base.size(1) == Npad, "base shape");
    TORCH_CHECK(wbg.size(0) == Npad, "wbg shape");
    int rc = launch_fused_predict_linear(
        masks.data_ptr<uint8_t>(), diff.data_ptr<float>(), base.data_ptr<float>(),
        wbg.data_ptr<float>(), ey.data_ptr<float>(), B, S, M, Mpad, Npad,
        n_out, (int)act, current_stream());
    TORCH_CHECK(rc == 0, "fused_predict_linear: unsupported shape (Mpad<=64, Npad%16==0, Npad<=128, n_out in {1,2,4})");
}

void synth_chunk(
    torch::Tensor masks, torch::Tensor x, torch::Tensor bg,
    torch::Tensor col_group, torch::Tensor out, int64_t b_lo, int64_t b_hi,
    int64_t s_lo, int64_t s_hi) {
    CHECK_DEV(masks); CHECK_DEV(x); CHECK_DEV(bg); CHECK_DEV(col_group); CHECK_DEV(out);
    int S = masks.size(1), M = masks.size(2);
    int N = bg.size(0), D = bg.size(1);
    TORCH_CHECK(out.size(0) == (b_hi - b_lo) * (s_hi - s_lo) * N && out.size(1) == D,
                "out shape");
    const bool bf16 = out.dtype() == torch::kBFloat16;
    TORCH_CHECK(bf16 || out.dtype() == torch::kFloat32,
                "out must be float32 or bfloat16");
    launch_synth_chunk(
        masks.data_ptr<uint8_t>(), x.data_ptr<float>(), bg.data_ptr<float>(),
        col_group.data_ptr<int>(),
        bf16 ? (void*)out.data_ptr<at::BFloat16>()
             : (void*)out.data_ptr<float>(),
        bf16 ? 1 : 0, S, M, N, D,
        (int)b_lo, (int)b_hi, (int)s_lo, (int)s_hi, current_stream());
}

void wls_solve(
    torch::Tensor masks, torch::Tensor kw, torch::Tensor ey_adj,
    torch::Tensor total, torch::Tensor phi,
    c10::optional<torch::Tensor> packed) {
    CHECK_DEV(masks); CHECK_DEV(kw); CHECK_DEV(ey_adj); CHECK_DEV(total); CHECK_DEV(phi);
    int B = masks.size(0), S = masks.size(1), M = masks.size(2);
    int n_out = ey_adj.size(2);
    TORCH_CHECK(phi.size(0) == B && phi.size(1) == M && phi.size(2) == n_out, "phi shape");
    const uint64_t* pk = nullptr;
    if (packed.has_value()) {
        CHECK_DEV(packed.value());
        TORCH_CHECK(packed->size(0) == B && packed->size(1) == S, "packed shape");
        pk = reinterpret_cast<const uint64_t*>(packed->data_ptr<int64_t>());
    }
    int rc = launch_wls_solve(
        masks.data_ptr<uint8_t>(), pk, kw.data_ptr<float>(), ey_adj.data_ptr<float>(),
        total.data_ptr<float>(), phi.data_ptr<float>(), B, S, M, n_out,
        current_stream());
    TORCH_CHECK(rc == 0, "wls_solve: unsupported shape (2<=M<=64, n_out<=8)");
}

void pack_masks(torch::Tensor masks, torch::Tensor packed) {
    CHECK_DEV(masks); CHECK_DEV(packed);
    TORCH_CHECK(masks.dtype() == torch::kUInt8, "masks must be u8");
    int B = masks.size(0), S = masks.size(1), M = masks.size(2);
    TORCH_CHECK(M <= 64, "pack_masks supports M <= 64");
    TORCH_CHECK(packed.size(0) == B && packed.size(1) == S, "packed shape");
    launch_pack_masks(
        masks.data_ptr<uint8_t>(),
        reinterpret_cast<uint64_t*>(packed.data_ptr<int64_t>()),
        B, S, M, current_stream());
}

void fused_predict_bf16(
    torch::Tensor masks, torch::Tensor diffB, torch::Tensor base,
    torch::Tensor wbg, torch::Tensor ey, int64_t act) {
    CHECK_DEV(masks); CHECK_DEV(diffB); CHECK_DEV(base); CHECK_DEV(wbg); CHECK_DEV(ey);
    TORCH_CHECK(masks.dtype() == torch::kUInt8 && diffB.dtype() == torch::kBFloat16);
    int B = masks.size(0), S = masks.size(1), M = masks.size(2);
    int split = diffB.size(1), Npad = diffB.size(3);
    int n_out = ey.size(2);                    // act 3 uses 1 diff image for 2 outputs
    TORCH_CHECK(diffB.size(2) == ((act == 3) ? 1 : n_out), "diffB image count vs act");
    TORCH_CHECK(diffB.size(0) == B && diffB.size(4) == 40, "diffB (B,split,o,Npad,40)");
    TORCH_CHECK(ey.size(0) == B && ey.size(1) == S, "ey shape");
    int rc = launch_fused_predict_bf16(
        masks.data_ptr<uint8_t>(),
        reinterpret_cast<const uint16_t*>(diffB.data_ptr<at::BFloat16>()),
        base.data_ptr<float>(), wbg.data_ptr<float>(), ey.data_ptr<float>(),
        B, S, M, Npad, n_out, (int)act, split, current_stream());
    TORCH_CHECK(rc == 0, "fused_predict_bf16: unsupported shape (M<=32, Npad<=128, n_out in {1,2,4}, split in {1,2})");
}

void build_diff_f32(torch::Tensor xp, torch::Tensor bgp, torch::Tensor vidx,
                    torch::Tensor out) {
    CHECK_DEV(xp); CHECK_DEV(bgp); CHECK_DEV(vidx); CHECK_DEV(out);
    int B = xp.size(0), G = xp.size(1), O = xp.size(2);
    int N = bgp.size(0), m = vidx.size(0);
    int Mpad = out.size(2), Npad = out.size(3);
    TORCH_CHECK(out.size(0) == B && out.size(1) == O && m <= Mpad && N <= Npad);
    launch_build_diff_f32(
        xp.data_ptr<float>(), bgp.data_ptr<float>(), vidx.data_ptr<int64_t>(),
        out.data_ptr<float>(), B, G, O, N, m, Mpad, Npad, current_stream());
}

void build_diff_bf16(torch::Tensor xp, torch::Tensor bgp, torch::Tensor vidx,
                     torch::Tensor out) {
    CHECK_DEV(xp); CHECK_DEV(bgp); CHECK_DEV(vidx); CHECK_DEV(out);
    TORCH_CHECK(out.dtype() == torch::kBFloat16 && out.size(4) == 40);
    int B = xp.size(0), G = xp.size(1), O = xp.size(2);
    int N = bgp.size(0), m = vidx.size(0);
    int split = out.size(1), Npad = out.size(3);
    TORCH_CHECK(out.size(0) == B && out.size(2) == O && N <= Npad && m <= 32);
    launch_build_diff_bf16(
        xp.data_ptr<float>(), bgp.data_ptr<float>(), vidx.data_ptr<int64_t>(),
        reinterpret_cast<uint16_t*>(out.data_ptr<at::BFloat16>()), B, G, O, N,
        m, Npad, split, current_stream());
}

void pack_masks_words(torch::Tensor masks, torch::Tensor packed) {
    CHECK_DEV(masks); CHECK_DEV(packed);
    TORCH_CHECK(masks.dtype() == torch::kUInt8, "masks must be u8");
    TORCH_CHECK(packed.dim() == 3, "packed must be (B,S,W)");
    int B = masks.size(0), S = masks.size(1), M = masks.size(2);
    int W = packed.size(2);
    TORCH_CHECK(packed.size(0) == B && packed.size(1) == S, "packed shape");
    TORCH_CHECK(W * 64 >= M, "packed word count vs M");
    launch_pack_masks_words(
        masks.data_ptr<uint8_t>(),
        reinterpret_cast<uint64_t*>(packed.data_ptr<int64_t>()),
        B, S, M, W, current_stream());
}

void wls_gram(
    torch::Tensor packed, torch::Tensor kw, torch::Tensor ey_adj,
    torch::Tensor total, torch::Tensor A64, torch::Tensor rhs64) {
    CHECK_DEV(packed); CHECK_DEV(kw); CHECK_DEV(ey_adj); CHECK_DEV(total);
    CHECK_DEV(A64); CHECK_DEV(rhs64);
    TORCH_CHECK(packed.dim() == 3, "packed must be (B,S,W)");
    TORCH_CHECK(A64.dtype() == torch::kFloat64 && rhs64.dtype() == torch::kFloat64);
    int B = packed.size(0), S = packed.size(1), W = packed.size(2);
    int mm = A64.size(1), n_out = ey_adj.size(2);
    TORCH_CHECK(A64.size(0) == B && A64.size(2) == mm, "A64 shape");
    TORCH_CHECK(rhs64.size(0) == B && rhs64.size(1) == mm
                && rhs64.size(2) == n_out, "rhs64 shape");
    TORCH_CHECK(W * 64 >= mm + 1, "packed word count vs M");
    int rc = launch_wls_gram(
        reinterpret_cast<const uint64_t*>(packed.data_ptr<int64_t>()),
        kw.data_ptr<float>(), ey_adj.data_ptr<float>(),
        total.data_ptr<float>(), A64.data_ptr<double>(),
        rhs64.data_ptr<double>(), B, S, mm + 1, W, n_out, current_stream());
    TORCH_CHECK(rc == 0, "wls_gram: unsupported shape (M<=513, n_out<=8)");
}

void fused_predict_tiled(
    torch::Tensor masks, torch::Tensor diff, torch::Tensor base,
    torch::Tensor wbg, torch::Tensor partial, torch::Tensor ey, int64_t act) {
    CHECK_DEV(masks); CHECK_DEV(diff); CHECK_DEV(base); CHECK_DEV(wbg);
    CHECK_DEV(partial); CHECK_DEV(ey);
    TORCH_CHECK(masks.dtype() == torch::kUInt8, "masks must be u8 (B,S,M)");
    int B = masks.size(0), S = masks.size(1), M = masks.size(2);
    int Mpad = diff.size(2), Npad = diff.size(3);
    int n_out = ey.size(2);
    int oimg = (act == 3) ? 1 : n_out;
    int nacc = n_out;   // act 3 dual-accumulates p0 and p1
    int n_ntiles = (Npad + 127) / 128;
    TORCH_CHECK(diff.size(1) == oimg, "diff image count vs act");
    TORCH_CHECK(M <= Mpad, "M vs Mpad");
    TORCH_CHECK(ey.size(0) == B && ey.size(1) == S, "ey shape");
    TORCH_CHECK(base.size(0) == oimg && base.size(1) == Npad, "base shape");
    TORCH_CHECK(wbg.size(0) == Npad, "wbg shape");
    TORCH_CHECK(partial.size(0) == B && partial.size(1) == n_ntiles
                && partial.size(2) == S && partial.size(3) == nacc,
                "partial shape (B, n_ntiles, S, nacc)");
    int rc = launch_fused_predict_tiled(
        masks.data_ptr<uint8_t>(), diff.data_ptr<float>(),
        base.data_ptr<float>(), wbg.data_ptr<float>(),
        partial.data_ptr<float>(), ey.data_ptr<float>(), B, S, M, Mpad,
        Npad, n_out, (int)act, current_stream());
    TORCH_CHECK(rc == 0, "fused_predict_tiled: unsupported shape "
                         "(Mpad%4==0, Npad%16==0, n_out in {1,2,4})");
}

void build_diff_bf16_tiled(torch::Tensor xp, torch::Tensor bgp,
                           torch::Tensor vidx, torch::Tensor out) {
    CHECK_DEV(xp); CHECK_DEV(bgp); CHECK_DEV(vidx); CHECK_DEV(out);
    TORCH_CHECK(out.dtype() == torch::kBFloat16 && out.dim() == 5,
                "out must be bf16 (B,split,O,Npad,Mpad)");
    int B = xp.size(0), G = xp.size(1), O = xp.size(2);
    int N = bgp.size(0), mcount = vidx.size(0);
    int split = out.size(1), Npad = out.size(3), Mpad = out.size(4);
    TORCH_CHECK(out.size(0) == B && out.size(2) == O && N <= Npad
                && mcount <= Mpad);
    launch_build_diff_bf16_tiled(
        xp.data_ptr<float>(), bgp.data_ptr<float>(), vidx.data_ptr<int64_t>(),
        reinterpret_cast<uint16_t*>(out.data_ptr<at::BFloat16>()), B, G, O, N,
        mcount, Mpad, Npad, split, current_stream());
}

void fused_predict_tiled_bf16(
    torch::Tensor masks, torch::Tensor diffB, torch::Tensor base,
    torch::Tensor wbg, torch::Tensor partial, torch::Tensor ey, int64_t act) {
    CHECK_DEV(masks); CHECK_DEV(diffB); CHECK_DEV(base); CHECK_DEV(wbg);
    CHECK_DEV(partial); CHECK_DEV(ey);
    TORCH_CHECK(masks.dtype() == torch::kUInt8
                && diffB.dtype() == torch::kBFloat16);
    int B = masks.size(0), S = masks.size(1), M = masks.size(2);
    int split = diffB.size(1), Npad = diffB.size(3), Mpad = diffB.size(4);
    int n_out = ey.size(2);
    int oimg = (act == 3) ? 1 : n_out;
    int n_ntiles = (Npad + 127) / 128;
    TORCH_CHECK(diffB.size(2) == oimg, "diffB image count vs act");
    TORCH_CHECK(ey.size(0) == B && ey.size(1) == S, "ey shape");
    TORCH_CHECK(base.size(0) == oimg && base.size(1) == Npad, "base shape");
    TORCH_CHECK(wbg.size(0) == Npad, "wbg shape");
    TORCH_CHECK(partial.size(0) == B && partial.size(1) == n_ntiles
                && partial.size(2) == S && partial.size(3) == n_out,
                "partial shape");
    TORCH_CHECK((size_t)split * oimg * 128 * 40 * 2 <= 64 * 1024,
                "bf16 tiled LDS footprint (split*oimg too large)");
    int rc = launch_fused_predict_tiled_bf16(
        masks.data_ptr<uint8_t>(),
        reinterpret_cast<const uint16_t*>(diffB.data_ptr<at::BFloat16>()),
        base.data_ptr<float>(), wbg.data_ptr<float>(),
        partial.data_ptr<float>(), ey.data_ptr<float>(), B, S, M, Mpad, Npad,
        n_out, (int)act, split, current_stream());
    TORCH_CHECK(rc == 0, "fused_predict_tiled_bf16: unsupported shape "
                         "(Mpad%32==0, Npad%16==0, n_out in {1,2,4})");
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("fill_random_masks", &fill_random_masks,
          "Philox coalition sampling (K2)");
    m.def("fused_predict_linear", &fused_predict_linear,
          "MFMA fused mask@diff GEMM + activation + background reduce (K3-K6)");
    m.def("build_diff_f32", &build_diff_f32,
          "diff image in the f32 fused-kernel layout");
    m.def("build_diff_bf16", &build_diff_bf16,
          "hi(+lo) diff images in the bf16 fused-kernel layout");
    m.def("pack_masks", &pack_masks,
          "masks u8 -> packed u64 bits (for the MFMA WLS Gram build)");
    m.def("fused_predict_bf16", &fused_predict_bf16,
          "bf16 matrix-core fused predict (single or hi+lo split), A-operand "
          "converted in-register from the raw u8 masks");
    m.def("synth_chunk", &synth_chunk,
          "masked-background perturbation synthesis tile (K3')");
    m.def("wls_solve", &wls_solve,
          "batched constrained WLS Shapley solve (K7)",
          pybind11::arg("masks"), pybind11::arg("kw"), pybind11::arg("ey_adj"),
          pybind11::arg("total"), pybind11::arg("phi"),
          pybind11::arg("packed") = pybind11::none());
    m.def("pack_masks_words", &pack_masks_words,
          "masks u8 -> (B,S,W) packed u64 words (wide-M Gram build)");
    m.def("wls_gram", &wls_gram,
          "tiled MFMA Gram+rhs build (fp64 via chunked promotion) for the "
          "stress WLS shapes, M up to 513");
    m.def("fused_predict_tiled", &fused_predict_tiled,
          "tiled MFMA fused predict for Mpad>64 / Npad>128 (LDS-streamed "
          "diff chunks, per-column-tile partials + deterministic reduce)");
    m.def("build_diff_bf16_tiled", &build_diff_bf16_tiled,
          "hi(+lo) k-contiguous diff image for the bf16 tiled predict");
    m.def("fused_predict_tiled_bf16", &fused_predict_tiled_bf16,
          "bf16 matrix-core tiled fused predict (v_mfma_f32_16x16x32_bf16 "
          "per 32-deep k block; hi+lo split for fp32-grade)");
}
