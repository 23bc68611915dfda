// Torch bindings for the fmda_amd HIP kernels (MI355X / gfx950).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <vector>

extern "C" int fmda_gru_fwd_launch(int is_bf16, int Hp, const void* gi,
                                   const void* w, const float* bhh, void* out,
                                   float* hlast, int B, int Tseq, int n_dir,
                                   const float* h0, void* out_drop,
                                   unsigned int drop_thr, float drop_scale,
                                   unsigned long long drop_seed,
                                   hipStream_t stream);
extern "C" int fmda_gru_fwd_cs_launch(const void* gi, const void* w,
                                      const float* bhh, void* out,
                                      float* hlast, void* hpub,
                                      unsigned int* cnt, int B, int Tseq,
                                      int n_dir, void* out_drop,
                                      unsigned int drop_thr,
                                      float drop_scale,
                                      unsigned long long drop_seed,
                                      hipStream_t stream);
extern "C" int fmda_gru_bwd_cs_launch(const void* gi, const void* w,
                                      const void* wt, const float* bhh,
                                      const void* out, const void* dout,
                                      const float* dhT, void* dgi, void* dgh,
                                      float* dh0, float* dbhh, void* gpub,
                                      unsigned int* cnt, int B, int Tseq,
                                      int n_dir, unsigned int drop_thr,
                                      float drop_scale,
                                      unsigned long long drop_seed,
                                      hipStream_t stream);
extern "C" int fmda_gru_bwd_launch(int is_bf16, int Hp, const void* gi,
                                   const void* w, const void* wt,
                                   const float* bhh,
                                   const void* out, const void* dout,
                                   const float* dhT, void* dgi, void* dgh,
                                   float* dh0, float* dbhh, int B, int Tseq,
                                   int n_dir, unsigned int drop_thr,
                                   float drop_scale,
                                   unsigned long long drop_seed,
                                   const float* h0, void* dgh0,
                                   hipStream_t stream);
extern "C" int fmda_mfma_selftest_launch(const void* A, const void* Bm,
                                         float* C, hipStream_t stream);
extern "C" int fmda_pool_fwd_launch(int is_bf16, const void* out, float* maxv,
                                    float* avgv, int* amax, int B, int Tseq,
                                    int H, int n_dir, hipStream_t stream);
extern "C" int fmda_pool_bwd_launch(int is_bf16, const float* dmax,
                                    const float* davg, const int* amax,
                                    void* dout, int B, int Tseq, int H,
                                    int n_dir, hipStream_t stream);
extern "C" int fmda_dropout_launch(int is_bf16, const void* x, void* y,
                                   long n, float p, unsigned long long seed,
                                   hipStream_t stream);
extern "C" int fmda_spatial_dropout_launch(int is_bf16, const void* x,
                                           void* y, long B, long Tlen, long F,
                                           float p, unsigned long long seed,
                                           hipStream_t stream);
extern "C" int fmda_ingest_row_launch(void* ring, const float* row,
                                      const float* xmin, const float* xrng,
                                      int Tseq, int F, hipStream_t stream);
extern "C" int fmda_pool_concat_launch(int is_bf16, const void* out,
                                       const float* hlast, void* feat, int B,
                                       int Tseq, int H, int Hp, int n_dir,
                                       hipStream_t stream);
extern "C" int fmda_head_sigmoid_launch(int is_bf16, const void* x,
                                        const void* W, const void* bias,
                                        float* probs, int B, int K, int C,
                                        hipStream_t stream);
extern "C" int fmda_head_fwd_launch(int is_bf16, const void* x, const void* W,
                                    const void* bias, const float* y,
                                    const float* wgt, const float* pw,
                                    float* logits, float* sig,
                                    float* loss_sum, int B, int K, int C,
                                    hipStream_t stream);
extern "C" int fmda_head_bwd_launch(int is_bf16, const float* sig,
                                    const float* y, const float* wgt,
                                    const float* pw, const float* gscale,
                                    const void* W, float* dlogits, void* dx,
                                    int B, int K, int C, hipStream_t stream);
extern "C" int fmda_opt_norm2_launch(const void* chunks, int n_chunks,
                                     float* out, hipStream_t stream);
namespace fmda_ckpt {
void save_state_dict(const std::string& path,
                     const std::vector<std::string>& keys,
                     const std::vector<torch::Tensor>& tensors);
std::vector<std::pair<std::string, torch::Tensor>> load_state_dict(
        const std::string& path);
}  // namespace fmda_ckpt

extern "C" int fmda_opt_adam_launch(const void* chunks, int n_chunks,
                                    const float* norm2, float clip, float lr,
                                    float beta1, float beta2, float eps,
                                    float bc1, float bc2, hipStream_t stream);

namespace {

void check_common(const torch::Tensor& gi, const torch::Tensor& w,
                  const torch::Tensor& bhh, int& B, int& T, int& n_dir,
                  int& Hp, bool& is_bf16) {
    TORCH_CHECK(gi.is_cuda() && w.is_cuda() && bhh.is_cuda(),
                "fmda gru: tensors must be on GPU");
    TORCH_CHECK(gi.is_contiguous() && w.is_contiguous() && bhh.is_contiguous(),
                "fmda gru: tensors must be contiguous");
    TORCH_CHECK(gi.dim() == 3 && w.dim() == 3, "fmda gru: bad ranks");
    TORCH_CHECK(w.scalar_type() == gi.scalar_type(), "fmda gru: dtype mismatch");
    TORCH_CHECK(bhh.scalar_type() == torch::kFloat32, "bhh must be fp32");
    n_dir = w.size(0);
    Hp = w.size(2);
    TORCH_CHECK(w.size(1) == 3 * Hp, "w must be (n_dir, 3H, H)");
    B = gi.size(0);
    T = gi.size(1);
    TORCH_CHECK(gi.size(2) == (int64_t)n_dir * 3 * Hp, "gi last dim mismatch");
    TORCH_CHECK(n_dir == 1 || n_dir == 2, "n_dir must be 1 or 2");
    is_bf16 = gi.scalar_type() == torch::kBFloat16;
    TORCH_CHECK(is_bf16 || gi.scalar_type() == torch::kFloat32,
                "fmda gru: dtype must be bf16 or fp32");
    // the bf16 recurrence tiles MFMA K=32: Hp=16 would silently drop the
    // recurrent term (KK = Hp/32 = 0) — pad to Hp >= 32 instead
    TORCH_CHECK(!(is_bf16 && Hp < 32),
                "fmda gru: bf16 requires Hp >= 32 (MFMA K=32); pad the "
                "hidden size");
}

}  // namespace

std::vector<torch::Tensor> gru_fwd(torch::Tensor gi, torch::Tensor w,
                                   torch::Tensor bhh,
                                   c10::optional<torch::Tensor> h0_opt,
                                   double drop_p, int64_t drop_seed) {
    int B, T, n_dir, Hp;
    bool is_bf16;
    check_common(gi, w, bhh, B, T, n_dir, Hp, is_bf16);
    const float* h0 = nullptr;
    if (h0_opt.has_value()) {
        auto& h = h0_opt.value();
        TORCH_CHECK(h.is_cuda() && h.is_contiguous() &&
                    h.scalar_type() == torch::kFloat32 &&
                    h.sizes() == torch::IntArrayRef({n_dir, B, Hp}),
                    "h0 must be contiguous fp32 (n_dir, B, Hp)");
        h0 = h.data_ptr<float>();
    }
    auto out = torch::empty({B, T, (int64_t)n_dir * Hp}, gi.options());
    auto hlast = torch::empty({n_dir, B, Hp},
                              gi.options().dtype(torch::kFloat32));
    // fused forward dropout: emit out AND the dropped copy (the layer
    // above's input) from the store epilogue, replacing the separate
    // full-tensor dropout pass. Quantized exactly like dropout_kernel.
    const unsigned int drop_thr =
        (drop_p > 0.0) ? (unsigned int)((float)drop_p * 256.0f) : 0u;
    const float drop_scale =
        (drop_p > 0.0) ? (float)(1.0 / (1.0 - drop_p)) : 1.0f;
    torch::Tensor out_drop;
    void* out_drop_ptr = nullptr;
    if (drop_thr != 0u) {
        TORCH_CHECK(is_bf16 && (Hp == 128 || Hp == 512),
                    "fused fwd dropout requires bf16 Hp=128/512");
        TORCH_CHECK(Hp == 128 || h0 == nullptr,
                    "fused fwd dropout with h0 requires Hp=128");
        out_drop = torch::empty_like(out);
        out_drop_ptr = out_drop.data_ptr();
    }
    auto stream = at::hip::getCurrentHIPStream();
    int rc;
    // the column-split Hp=512 kernel pre-publishes h_{-1}=0; with a real
    // h0 the batch-parallel kernel takes the (rare) call instead
    if (is_bf16 && Hp == 512 && h0 == nullptr) {
        // column-split persistent kernel: publication ring + group counters
        const int BR = 256;
        const int GB = (B + BR - 1) / BR;
        const int G = GB * n_dir;
        auto hpub = torch::empty({2, G, BR, (int64_t)Hp}, gi.options());
        auto cnt = torch::zeros({G}, gi.options().dtype(torch::kUInt32));
        rc = fmda_gru_fwd_cs_launch(gi.data_ptr(), w.data_ptr(),
                                    bhh.data_ptr<float>(), out.data_ptr(),
                                    hlast.data_ptr<float>(), hpub.data_ptr(),
                                    (unsigned int*)cnt.data_ptr(), B, T,
                                    n_dir, out_drop_ptr, drop_thr,
                                    drop_scale,
                                    (unsigned long long)drop_seed,
                                    stream.stream());
    } else {
        rc = fmda_gru_fwd_launch(is_bf16 ? 1 : 0, Hp, gi.data_ptr(),
                                 w.data_ptr(), bhh.data_ptr<float>(),
                                 out.data_ptr(), hlast.data_ptr<float>(), B, T,
                                 n_dir, h0, out_drop_ptr, drop_thr,
                                 drop_scale,
                                 (unsigned long long)drop_seed,
                                 stream.stream());
    }
    TORCH_CHECK(rc == 0, "fmda gru_fwd launch failed rc=", rc, " Hp=", Hp);
    if (out_drop_ptr != nullptr)
        return {out, hlast, out_drop};
    return {out, hlast};
}

std::vector<torch::Tensor> gru_bwd(torch::Tensor gi, torch::Tensor w,
                                   torch::Tensor bhh, torch::Tensor out,
                                   torch::Tensor dout, torch::Tensor dhT,
                                   double drop_p, int64_t drop_seed,
                                   c10::optional<torch::Tensor> h0_opt) {
    int B, T, n_dir, Hp;
    bool is_bf16;
    check_common(gi, w, bhh, B, T, n_dir, Hp, is_bf16);
    const float* h0 = nullptr;
    torch::Tensor dgh0;
    void* dgh0_ptr = nullptr;
    if (h0_opt.has_value()) {
        auto& h = h0_opt.value();
        TORCH_CHECK(h.is_cuda() && h.is_contiguous() &&
                    h.scalar_type() == torch::kFloat32 &&
                    h.sizes() == torch::IntArrayRef({n_dir, B, Hp}),
                    "h0 must be contiguous fp32 (n_dir, B, Hp)");
        h0 = h.data_ptr<float>();
        dgh0 = torch::empty({n_dir, B, (int64_t)3 * Hp}, gi.options());
        dgh0_ptr = dgh0.data_ptr();
    }
    TORCH_CHECK(out.is_contiguous() && dout.is_contiguous() &&
                dhT.is_contiguous(), "fmda gru_bwd: tensors must be contiguous");
    TORCH_CHECK(dhT.scalar_type() == torch::kFloat32, "dhT must be fp32");
    // fp32 BPTT at Hp=512 needs ~264 KB of LDS staging — architecturally
    // out of budget (160 KB/CU); bf16 (the training dtype) covers H>256.
    TORCH_CHECK(is_bf16 || Hp <= 256,
                "fp32 backward unsupported for Hp > 256 (LDS budget); "
                "train in bf16 for hidden sizes above 256");
    torch::Tensor dgi, dgh;
    if (getenv("FMDA_ZERO_GRADS")) {   // debug: expose unwritten positions
        dgi = torch::zeros_like(gi);
        dgh = torch::zeros_like(gi);
    } else {
        dgi = torch::empty_like(gi);
        dgh = torch::empty_like(gi);
    }
    auto dh0 = torch::empty({n_dir, B, Hp},
                            gi.options().dtype(torch::kFloat32));
    // packed bias-grad sums: [dr, dz, dhn | dn] per direction; db_hh and
    // db_ih are assembled from slices below (dr/dz shared).
    auto dbsum = torch::zeros({n_dir, 4 * Hp},
                              gi.options().dtype(torch::kFloat32));
    // W^T for the carry GEMM's B fragments (v3 kernel streams them from
    // L2 as contiguous bf16x8 rows instead of hoisting 48 VGPRs).
    torch::Tensor wt;
    const void* wt_ptr = nullptr;
    if (is_bf16 && (Hp == 128 || Hp == 512)) {
        wt = w.transpose(1, 2).contiguous();
        wt_ptr = wt.data_ptr();
    }
    auto stream = at::hip::getCurrentHIPStream();
    int rc;
    if (is_bf16 && Hp == 512 && h0 == nullptr) {
        const unsigned int cs_thr =
            (drop_p > 0.0) ? (unsigned int)((float)drop_p * 256.0f) : 0u;
        const float cs_scale =
            (drop_p > 0.0) ? (float)(1.0 / (1.0 - drop_p)) : 1.0f;
        const int BR = 256;
        const int GB = (B + BR - 1) / BR;
        const int G = GB * n_dir;
        auto gpub = torch::empty({2, G, BR, (int64_t)3 * Hp}, gi.options());
        auto cnt = torch::zeros({G}, gi.options().dtype(torch::kUInt32));
        rc = fmda_gru_bwd_cs_launch(
            gi.data_ptr(), w.data_ptr(), wt_ptr, bhh.data_ptr<float>(),
            out.data_ptr(), dout.data_ptr(), dhT.data_ptr<float>(),
            dgi.data_ptr(), dgh.data_ptr(), dh0.data_ptr<float>(),
            dbsum.data_ptr<float>(), gpub.data_ptr(),
            (unsigned int*)cnt.data_ptr(), B, T, n_dir, cs_thr, cs_scale,
            (unsigned long long)drop_seed, stream.stream());
    } else {
        // fused inter-layer dropout backward (d_out is w.r.t. the DROPPED
        // activations; the kernel recomputes the counter-based mask at the
        // dout read instead of a separate full-tensor pass)
        // quantized EXACTLY like dropout_kernel's forward threshold
        // ((unsigned)(float_p * 256.0f)) so the recomputed backward mask
        // can never disagree with the forward one at a rounding boundary
        const unsigned int drop_thr =
            (drop_p > 0.0) ? (unsigned int)((float)drop_p * 256.0f) : 0u;
        const float drop_scale =
            (drop_p > 0.0) ? (float)(1.0 / (1.0 - drop_p)) : 1.0f;
        TORCH_CHECK(drop_thr == 0u || (is_bf16 && Hp == 128),
                    "fused dropout-backward on this path requires bf16 "
                    "Hp=128");
        rc = fmda_gru_bwd_launch(is_bf16 ? 1 : 0, Hp, gi.data_ptr(),
                                 w.data_ptr(), wt_ptr, bhh.data_ptr<float>(),
                                 out.data_ptr(), dout.data_ptr(),
                                 dhT.data_ptr<float>(), dgi.data_ptr(),
                                 dgh.data_ptr(), dh0.data_ptr<float>(),
                                 dbsum.data_ptr<float>(), B, T,
                                 n_dir, drop_thr, drop_scale,
                                 (unsigned long long)drop_seed, h0, dgh0_ptr,
                                 stream.stream());
    }
    TORCH_CHECK(rc == 0, "fmda gru_bwd launch failed rc=", rc, " Hp=", Hp);
    using torch::indexing::Slice;
    auto dbhh = dbsum.index({Slice(), Slice(0, 3 * Hp)});
    auto dbih = torch::cat({dbsum.index({Slice(), Slice(0, 2 * Hp)}),
                            dbsum.index({Slice(), Slice(3 * Hp, 4 * Hp)})}, 1);
    if (dgh0_ptr != nullptr)
        return {dgi, dgh, dh0, dbhh.contiguous(), dbih, dgh0};
    return {dgi, dgh, dh0, dbhh.contiguous(), dbih};
}

torch::Tensor mfma_selftest(torch::Tensor A, torch::Tensor Bm) {
    TORCH_CHECK(A.is_cuda() && A.scalar_type() == torch::kBFloat16);
    TORCH_CHECK(A.sizes() == torch::IntArrayRef({16, 32}) &&
                Bm.sizes() == torch::IntArrayRef({32, 16}));
    auto C = torch::empty({16, 16}, A.options().dtype(torch::kFloat32));
    auto stream = at::hip::getCurrentHIPStream();
    int rc = fmda_mfma_selftest_launch(A.contiguous().data_ptr(),
                                       Bm.contiguous().data_ptr(),
                                       C.data_ptr<float>(), stream.stream());
    TORCH_CHECK(rc == 0, "mfma selftest launch failed");
    return C;
}

std::vector<torch::Tensor> pool_fwd(torch::Tensor out, int64_t n_dir) {
    TORCH_CHECK(out.is_cuda() && out.is_contiguous() && out.dim() == 3);
    const bool bf16 = out.scalar_type() == torch::kBFloat16;
    TORCH_CHECK(bf16 || out.scalar_type() == torch::kFloat32);
    const int B = out.size(0), T = out.size(1);
    const int H = out.size(2) / n_dir;
    TORCH_CHECK(H % 2 == 0, "pool_fwd pairs columns: H must be even "
                "(odd H uses the eager path)");
    auto f32 = out.options().dtype(torch::kFloat32);
    auto maxv = torch::empty({B, H}, f32);
    auto avgv = torch::empty({B, H}, f32);
    auto amax = torch::empty({B, H}, out.options().dtype(torch::kInt32));
    auto stream = at::hip::getCurrentHIPStream();
    int rc = fmda_pool_fwd_launch(bf16 ? 1 : 0, out.data_ptr(),
                                  maxv.data_ptr<float>(), avgv.data_ptr<float>(),
                                  amax.data_ptr<int>(), B, T, H, (int)n_dir,
                                  stream.stream());
    TORCH_CHECK(rc == 0, "pool_fwd launch failed");
    return {maxv, avgv, amax};
}

torch::Tensor pool_bwd(torch::Tensor dmax, torch::Tensor davg,
                       torch::Tensor amax, int64_t T, int64_t n_dir,
                       torch::ScalarType dtype) {
    TORCH_CHECK(dmax.is_cuda() && dmax.is_contiguous() && davg.is_contiguous());
    const int B = dmax.size(0), H = dmax.size(1);
    const bool bf16 = dtype == torch::kBFloat16;
    auto dout = torch::empty({B, T, n_dir * H}, dmax.options().dtype(dtype));
    auto stream = at::hip::getCurrentHIPStream();
    int rc = fmda_pool_bwd_launch(bf16 ? 1 : 0, dmax.data_ptr<float>(),
                                  davg.data_ptr<float>(), amax.data_ptr<int>(),
                                  dout.data_ptr(), B, (int)T, H, (int)n_dir,
                                  stream.stream());
    TORCH_CHECK(rc == 0, "pool_bwd launch failed");
    return dout;
}

// chunks: int64 tensor (n_chunks, 5) holding {p, g, m, v, n} built by
// fmda_amd.optim.FusedClipAdam (device buffer; pointers as int64).
torch::Tensor fused_clip_adam(torch::Tensor chunks, int64_t n_chunks,
                              double clip, double lr, double beta1,
                              double beta2, double eps, int64_t step) {
    TORCH_CHECK(chunks.is_cuda() && chunks.scalar_type() == torch::kInt64);
    auto norm2 = torch::zeros({1}, chunks.options().dtype(torch::kFloat32));
    auto stream = at::hip::getCurrentHIPStream();
    if (clip > 0) {
        int rc = fmda_opt_norm2_launch(chunks.data_ptr(), (int)n_chunks,
                                       norm2.data_ptr<float>(),
                                       stream.stream());
        TORCH_CHECK(rc == 0, "opt norm2 launch failed");
    }
    const float bc1 = 1.0f - powf((float)beta1, (float)step);
    const float bc2 = 1.0f - powf((float)beta2, (float)step);
    int rc = fmda_opt_adam_launch(chunks.data_ptr(), (int)n_chunks,
                                  norm2.data_ptr<float>(), (float)clip,
                                  (float)lr, (float)beta1, (float)beta2,
                                  (float)eps, bc1, bc2, stream.stream());
    TORCH_CHECK(rc == 0, "opt adam launch failed");
    return norm2;
}

std::vector<torch::Tensor> head_loss_fwd(torch::Tensor x, torch::Tensor W,
                                         torch::Tensor b, torch::Tensor y,
                                         torch::Tensor wgt, torch::Tensor pw) {
    TORCH_CHECK(x.is_cuda() && x.is_contiguous() && W.is_contiguous());
    const bool bf16 = x.scalar_type() == torch::kBFloat16;
    const int B = x.size(0), K = x.size(1), C = W.size(0);
    auto f32 = x.options().dtype(torch::kFloat32);
    auto logits = torch::empty({B, C}, f32);
    auto sig = torch::empty({B, C}, f32);
    auto loss = torch::zeros({1}, f32);
    auto stream = at::hip::getCurrentHIPStream();
    int rc = fmda_head_fwd_launch(
        bf16 ? 1 : 0, x.data_ptr(), W.data_ptr(), b.data_ptr(),
        y.data_ptr<float>(), wgt.data_ptr<float>(), pw.data_ptr<float>(),
        logits.data_ptr<float>(), sig.data_ptr<float>(),
        loss.data_ptr<float>(), B, K, C, stream.stream());
    TORCH_CHECK(rc == 0, "head fwd launch failed");
    return {logits, sig, loss};
}

std::vector<torch::Tensor> head_loss_bwd(torch::Tensor sig, torch::Tensor y,
                                         torch::Tensor wgt, torch::Tensor pw,
                                         torch::Tensor gscale,
                                         torch::Tensor W,
                                         torch::ScalarType xdtype) {
    const int B = sig.size(0), C = sig.size(1), K = W.size(1);
    const bool bf16 = xdtype == torch::kBFloat16;
    auto dlogits = torch::empty_like(sig);
    auto dx = torch::empty({B, K}, sig.options().dtype(xdtype));
    auto stream = at::hip::getCurrentHIPStream();
    int rc = fmda_head_bwd_launch(
        bf16 ? 1 : 0, sig.data_ptr<float>(), y.data_ptr<float>(),
        wgt.data_ptr<float>(), pw.data_ptr<float>(),
        gscale.data_ptr<float>(), W.data_ptr(), dlogits.data_ptr<float>(),
        dx.data_ptr(), B, K, C, stream.stream());
    TORCH_CHECK(rc == 0, "head bwd launch failed");
    return {dlogits, dx};
}

torch::Tensor dropout_fused(torch::Tensor x, double p, int64_t seed) {
    TORCH_CHECK(x.is_cuda() && x.is_contiguous() &&
                x.scalar_type() == torch::kBFloat16);
    auto y = torch::empty_like(x);
    auto stream = at::hip::getCurrentHIPStream();
    int rc = fmda_dropout_launch(1, x.data_ptr(), y.data_ptr(), x.numel(),
                                 (float)p, (unsigned long long)seed,
                                 stream.stream());
    TORCH_CHECK(rc == 0, "dropout launch failed");
    return y;
}

torch::Tensor spatial_dropout_fused(torch::Tensor x, double p,
                                    int64_t seed) {
    TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 3 &&
                x.scalar_type() == torch::kBFloat16);
    auto y = torch::empty_like(x);
    auto stream = at::hip::getCurrentHIPStream();
    int rc = fmda_spatial_dropout_launch(
        1, x.data_ptr(), y.data_ptr(), x.size(0), x.size(1), x.size(2),
        (float)p, (unsigned long long)seed, stream.stream());
    TORCH_CHECK(rc == 0, "spatial dropout launch failed");
    return y;
}

void ingest_row(torch::Tensor ring, torch::Tensor row, torch::Tensor xmin,
                torch::Tensor xrng) {
    TORCH_CHECK(ring.is_cuda() && ring.is_contiguous() && ring.dim() == 2 &&
                ring.scalar_type() == torch::kBFloat16,
                "ring must be (T, F) bf16 on GPU");
    TORCH_CHECK(row.is_cuda() && row.is_contiguous() &&
                row.scalar_type() == torch::kFloat32 &&
                row.numel() == ring.size(1));
    TORCH_CHECK(xmin.is_cuda() && xmin.is_contiguous() &&
                xmin.scalar_type() == torch::kFloat32 &&
                xmin.numel() == ring.size(1), "xmin must be fp32 (F) on GPU");
    TORCH_CHECK(xrng.is_cuda() && xrng.is_contiguous() &&
                xrng.scalar_type() == torch::kFloat32 &&
                xrng.numel() == ring.size(1), "xrng must be fp32 (F) on GPU");
    auto stream = at::hip::getCurrentHIPStream();
    int rc = fmda_ingest_row_launch(
        ring.data_ptr(), row.data_ptr<float>(), xmin.data_ptr<float>(),
        xrng.data_ptr<float>(), ring.size(0), ring.size(1), stream.stream());
    TORCH_CHECK(rc == 0, "ingest_row launch failed rc=", rc);
}

torch::Tensor pool_concat_infer(torch::Tensor out, torch::Tensor hlast,
                                int64_t H) {
    TORCH_CHECK(out.is_cuda() && out.is_contiguous() && out.dim() == 3);
    TORCH_CHECK(hlast.is_contiguous() &&
                hlast.scalar_type() == torch::kFloat32);
    const bool bf16 = out.scalar_type() == torch::kBFloat16;
    const int B = out.size(0), T = out.size(1);
    const int n_dir = hlast.size(0);
    const int Hp = hlast.size(2);
    auto feat = torch::empty({B, 3 * H}, out.options());
    auto stream = at::hip::getCurrentHIPStream();
    int rc = fmda_pool_concat_launch(bf16 ? 1 : 0, out.data_ptr(),
                                     hlast.data_ptr<float>(), feat.data_ptr(),
                                     B, T, (int)H, Hp, n_dir,
                                     stream.stream());
    TORCH_CHECK(rc == 0, "pool_concat launch failed");
    return feat;
}

torch::Tensor head_sigmoid(torch::Tensor x, torch::Tensor W,
                           torch::Tensor b) {
    TORCH_CHECK(x.is_cuda() && x.is_contiguous() && W.is_contiguous() &&
                b.is_contiguous());
    TORCH_CHECK(x.scalar_type() == W.scalar_type() &&
                x.scalar_type() == b.scalar_type());
    const bool bf16 = x.scalar_type() == torch::kBFloat16;
    const int B = x.size(0), K = x.size(1), C = W.size(0);
    auto probs = torch::empty({B, C}, x.options().dtype(torch::kFloat32));
    auto stream = at::hip::getCurrentHIPStream();
    int rc = fmda_head_sigmoid_launch(bf16 ? 1 : 0, x.data_ptr(),
                                      W.data_ptr(), b.data_ptr(),
                                      probs.data_ptr<float>(), B, K, C,
                                      stream.stream());
    TORCH_CHECK(rc == 0, "head_sigmoid launch failed");
    return probs;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("gru_fwd", &gru_fwd, "fused biGRU recurrence forward (HIP/CDNA4)",
          py::arg("gi"), py::arg("w"), py::arg("bhh"),
          py::arg("h0") = py::none(), py::arg("drop_p") = 0.0,
          py::arg("drop_seed") = 0);
    m.def("gru_bwd", &gru_bwd, "fused biGRU recurrence backward (HIP/CDNA4)",
          py::arg("gi"), py::arg("w"), py::arg("bhh"), py::arg("out"),
          py::arg("dout"), py::arg("dhT"), py::arg("drop_p") = 0.0,
          py::arg("drop_seed") = 0, py::arg("h0") = py::none());
    m.def("mfma_selftest", &mfma_selftest, "mfma fragment layout self-test");
    m.def("dropout_fused", &dropout_fused,
          "counter-based dropout (mask recomputed in backward)");
    m.def("spatial_dropout_fused", &spatial_dropout_fused,
          "channel (Dropout2d) dropout over (B, T, F) without permutes");
    m.def("ingest_row", &ingest_row,
          "streaming ingest: shift GPU window ring + normalize new row");
    m.def("pool_concat_infer", &pool_concat_infer,
          "fused [dirsum(h_last) | max | avg] concat for inference");
    m.def("head_sigmoid", &head_sigmoid,
          "linear head + sigmoid emitting probabilities");
    m.def("pool_fwd", &pool_fwd, "fused dirsum+max/avg pooling forward");
    m.def("pool_bwd", &pool_bwd, "fused pooling backward (d_out assembly)");
    m.def("head_loss_fwd", &head_loss_fwd, "fused head GEMM + BCE loss");
    m.def("head_loss_bwd", &head_loss_bwd, "fused head/loss backward");
    m.def("fused_clip_adam", &fused_clip_adam,
          "fused multi-tensor grad-clip + Adam step");
    m.def("save_state_dict_native", &fmda_ckpt::save_state_dict,
          "write a torch.load-compatible zip state_dict (native C++)");
    m.def("load_state_dict_native", &fmda_ckpt::load_state_dict,
          "read a torch.save zip state_dict (native C++)");
}
