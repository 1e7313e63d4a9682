// Torch bindings for the bee2bee_amd CDNA4 kernel set.
//
// Validation lives here (shape/dtype/device checks) so the kernels stay
// branch-free. All entry points are hipGraph-capture safe: no allocation
// beyond torch's caching allocator, no synchronization, no host reads.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

using torch::Tensor;

extern "C" {
void launch_rmsnorm(const unsigned short*, const unsigned short*,
                    unsigned short*, long, int, float, hipStream_t);
void launch_fused_add_rmsnorm(const unsigned short*, const unsigned short*,
                              const unsigned short*, unsigned short*,
                              unsigned short*, long, int, float, hipStream_t);
void launch_layernorm(const unsigned short*, const unsigned short*,
                      const unsigned short*, unsigned short*, long, int,
                      float, hipStream_t);
void launch_fused_add_layernorm(const unsigned short*, const unsigned short*,
                                const unsigned short*, const unsigned short*,
                                unsigned short*, unsigned short*, long, int,
                                float, hipStream_t);
void launch_gelu(const unsigned short*, unsigned short*, long, hipStream_t);
void launch_rope(unsigned short*, unsigned short*, const int*, const float*,
                 const float*, int, int, int, int, long, long, hipStream_t);
void launch_kv_store(const unsigned short*, const unsigned short*,
                     unsigned short*, unsigned short*, const int*, int, int,
                     int, int, long, long, hipStream_t);
void launch_kv_store_fp8(const unsigned short*, const unsigned short*,
                         unsigned char*, unsigned char*, const int*, int,
                         int, int, int, long, long, hipStream_t);
void launch_swiglu(const unsigned short*, unsigned short*, long, long,
                   hipStream_t);
void launch_attn_decode(const unsigned short*, const void*, const void*,
                        const int*, const int*, unsigned short*, float*,
                        float*, unsigned short*, float*, int, int, int, int,
                        int, int, int, long, float, int, hipStream_t);
void launch_attn_prefill(const unsigned short*, const unsigned short*,
                         const unsigned short*, const int*, unsigned short*,
                         int, int, int, int, long, long, long, int, float,
                         int, hipStream_t);
void launch_attn_prefill_paged(const unsigned short*, const void*,
                               const void*, const int*, const int*,
                               const int*, unsigned short*, int, int, int,
                               int, long, int, int, int, float, int,
                               hipStream_t);
void launch_mfma_probe(const unsigned short*, const unsigned short*, float*,
                       hipStream_t);
void launch_mfma_probe_fp8(const unsigned char*, const unsigned char*,
                           float*, hipStream_t);
void launch_grouped_gemm(const unsigned short*, const unsigned short*,
                         const int*, unsigned short*, int, int, int, int,
                         hipStream_t);
void launch_bw_linear(const unsigned short*, long, float*, hipStream_t);
void launch_bw_rowperlane(const unsigned short*, long, int, int, float*,
                          hipStream_t);
void launch_bw_rowperinstr(const unsigned short*, long, int, int, float*,
                           hipStream_t);
void launch_bw_paged(const unsigned short*, const int*, long, int, int,
                     float*, hipStream_t);
}

namespace {

inline hipStream_t stream() {
    return at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

inline bool hd_fp8_ok(long hd) { return hd % 8 == 0; }

inline const unsigned short* bf16p(const Tensor& t) {
    return reinterpret_cast<const unsigned short*>(t.data_ptr());
}

inline unsigned short* bf16p_mut(Tensor& t) {
    return reinterpret_cast<unsigned short*>(t.data_ptr());
}

void check_bf16(const Tensor& t, const char* name) {
    TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
    TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
}

Tensor rmsnorm(const Tensor& x, const Tensor& w, double eps) {
    check_bf16(x, "x");
    check_bf16(w, "w");
    TORCH_CHECK(x.is_contiguous() && w.is_contiguous());
    const long T = x.numel() / x.size(-1);
    const int H = x.size(-1);
    TORCH_CHECK(H % 8 == 0, "hidden size must be a multiple of 8");
    Tensor y = torch::empty_like(x);
    launch_rmsnorm(bf16p(x), bf16p(w), bf16p_mut(y), T, H, (float)eps,
                   stream());
    return y;
}

std::vector<Tensor> fused_add_rmsnorm(const Tensor& x, const Tensor& resid,
                                      const Tensor& w, double eps) {
    check_bf16(x, "x");
    check_bf16(resid, "resid");
    TORCH_CHECK(x.is_contiguous() && resid.is_contiguous() && w.is_contiguous());
    TORCH_CHECK(x.sizes() == resid.sizes());
    const long T = x.numel() / x.size(-1);
    const int H = x.size(-1);
    TORCH_CHECK(H % 8 == 0);
    Tensor y = torch::empty_like(x);
    Tensor resid_out = torch::empty_like(x);
    launch_fused_add_rmsnorm(bf16p(x), bf16p(resid), bf16p(w), bf16p_mut(y),
                             bf16p_mut(resid_out), T, H, (float)eps, stream());
    return {y, resid_out};
}

void rope_inplace(Tensor& q, Tensor& k, const Tensor& pos, const Tensor& cos_t,
                  const Tensor& sin_t) {
    check_bf16(q, "q");
    check_bf16(k, "k");
    TORCH_CHECK(pos.scalar_type() == torch::kInt32);
    TORCH_CHECK(cos_t.scalar_type() == torch::kFloat32);
    const int T = q.size(0);
    const int nq = q.size(1), nkv = k.size(1), hd = q.size(2);
    TORCH_CHECK(k.size(2) == hd && k.size(0) == T);
    // heads/dims contiguous within a token; token stride may be larger
    TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == hd);
    TORCH_CHECK(k.stride(2) == 1 && k.stride(1) == hd);
    TORCH_CHECK(cos_t.size(1) == hd / 2 && cos_t.is_contiguous() &&
                sin_t.is_contiguous());
    launch_rope(bf16p_mut(q), bf16p_mut(k), pos.data_ptr<int>(),
                cos_t.data_ptr<float>(), sin_t.data_ptr<float>(), T, nq, nkv,
                hd, q.stride(0), k.stride(0), stream());
}

void kv_cache_store(const Tensor& k, const Tensor& v, Tensor& k_cache,
                    Tensor& v_cache, const Tensor& slots) {
    check_bf16(k, "k");
    TORCH_CHECK(slots.scalar_type() == torch::kInt32);
    const int T = k.size(0), nkv = k.size(1), hd = k.size(2);
    const int bs = k_cache.size(2);
    TORCH_CHECK(k_cache.size(1) == nkv && k_cache.size(3) == hd);
    TORCH_CHECK(k_cache.is_contiguous() && v_cache.is_contiguous());
    TORCH_CHECK(k.stride(2) == 1 && k.stride(1) == hd);
    TORCH_CHECK(v.stride(2) == 1 && v.stride(1) == hd);
    TORCH_CHECK(hd % 2 == 0);
    if (k_cache.scalar_type() == torch::kUInt8) {
        // fp8 (OCP e4m3) cache: quantize at store time
        launch_kv_store_fp8(bf16p(k), bf16p(v),
                            k_cache.data_ptr<unsigned char>(),
                            v_cache.data_ptr<unsigned char>(),
                            slots.data_ptr<int>(), T, nkv, hd, bs,
                            k.stride(0), v.stride(0), stream());
        return;
    }
    check_bf16(k_cache, "k_cache");
    launch_kv_store(bf16p(k), bf16p(v), bf16p_mut(k_cache), bf16p_mut(v_cache),
                    slots.data_ptr<int>(), T, nkv, hd, bs, k.stride(0),
                    v.stride(0), stream());
}

Tensor layernorm(const Tensor& x, const Tensor& w, const Tensor& b,
                 double eps) {
    check_bf16(x, "x");
    check_bf16(w, "w");
    check_bf16(b, "b");
    TORCH_CHECK(x.is_contiguous() && w.is_contiguous() && b.is_contiguous());
    const long T = x.numel() / x.size(-1);
    const int H = x.size(-1);
    TORCH_CHECK(H % 8 == 0);
    Tensor y = torch::empty_like(x);
    launch_layernorm(bf16p(x), bf16p(w), bf16p(b), bf16p_mut(y), T, H,
                     (float)eps, stream());
    return y;
}

std::vector<Tensor> fused_add_layernorm(const Tensor& x, const Tensor& resid,
                                        const Tensor& w, const Tensor& b,
                                        double eps) {
    check_bf16(x, "x");
    check_bf16(resid, "resid");
    TORCH_CHECK(x.is_contiguous() && resid.is_contiguous());
    TORCH_CHECK(x.sizes() == resid.sizes());
    const long T = x.numel() / x.size(-1);
    const int H = x.size(-1);
    TORCH_CHECK(H % 8 == 0);
    Tensor y = torch::empty_like(x);
    Tensor resid_out = torch::empty_like(x);
    launch_fused_add_layernorm(bf16p(x), bf16p(resid), bf16p(w), bf16p(b),
                               bf16p_mut(y), bf16p_mut(resid_out), T, H,
                               (float)eps, stream());
    return {y, resid_out};
}

Tensor gelu(const Tensor& x) {
    check_bf16(x, "x");
    TORCH_CHECK(x.is_contiguous());
    TORCH_CHECK(x.numel() % 8 == 0);
    Tensor y = torch::empty_like(x);
    launch_gelu(bf16p(x), bf16p_mut(y), x.numel(), stream());
    return y;
}

Tensor swiglu(const Tensor& gu) {
    check_bf16(gu, "gate_up");
    TORCH_CHECK(gu.is_contiguous());
    const long I = gu.size(-1) / 2;
    const long T = gu.numel() / gu.size(-1);
    TORCH_CHECK(I % 8 == 0);
    Tensor out = torch::empty({gu.size(0), I}, gu.options());
    launch_swiglu(bf16p(gu), bf16p_mut(out), T, I, stream());
    return out;
}

std::vector<Tensor> attn_decode_impl(const Tensor& q, const Tensor& k_cache,
                                     const Tensor& v_cache,
                                     const Tensor& block_table,
                                     const Tensor& seq_lens, double scale,
                                     bool want_lse) {
    check_bf16(q, "q");
    const bool fp8 = k_cache.scalar_type() == torch::kUInt8;
    if (!fp8) check_bf16(k_cache, "k_cache");
    TORCH_CHECK(fp8 ? hd_fp8_ok(q.size(2)) : true,
                "fp8 KV decode needs head_dim % 8 == 0");
    TORCH_CHECK(block_table.scalar_type() == torch::kInt32 &&
                seq_lens.scalar_type() == torch::kInt32);
    TORCH_CHECK(block_table.is_contiguous());
    const int B = q.size(0), nq = q.size(1), hd = q.size(2);
    const int nkv = k_cache.size(1), bs = k_cache.size(2);
    const int W = block_table.size(1);
    TORCH_CHECK(hd <= 128 && hd % 2 == 0, "decode kernel supports hd<=128");
    TORCH_CHECK(nq % nkv == 0);
    const int G = nq / nkv;
    TORCH_CHECK(G >= 1 && G <= 8 || G == 16,
                "unsupported GQA group size ", G);
    TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == hd);
    Tensor out = torch::empty({B, nq, hd}, q.options());
    // flash-decoding split: one partial per 256-key chunk, then combine
    const int kDecChunk = 256;  // keep in sync with DEC_CHUNK
    const int C = (W * bs + kDecChunk - 1) / kDecChunk;
    TORCH_CHECK(C <= 128, "decode supports up to 32768-token contexts for now");
    auto fopt = q.options().dtype(torch::kFloat32);
    Tensor p_buf = torch::empty({B, nkv, C, kDecChunk, G}, q.options());
    Tensor part_o = torch::empty({B, nkv, C, G, hd}, fopt);
    Tensor part_ml = torch::empty({B, nkv, C, G, 2}, fopt);
    Tensor out_ml;
    float* out_ml_p = nullptr;
    if (want_lse) {
        out_ml = torch::empty({B, nq, 2}, fopt);
        out_ml_p = out_ml.data_ptr<float>();
    }
    launch_attn_decode(bf16p(q), k_cache.data_ptr(), v_cache.data_ptr(),
                       block_table.data_ptr<int>(), seq_lens.data_ptr<int>(),
                       bf16p_mut(p_buf), part_o.data_ptr<float>(),
                       part_ml.data_ptr<float>(),
                       bf16p_mut(out), out_ml_p, B, nkv, G, W, bs, hd, C,
                       q.stride(0), (float)scale, fp8 ? 1 : 0, stream());
    if (want_lse) return {out, out_ml};
    return {out};
}

Tensor attn_decode(const Tensor& q, const Tensor& k_cache,
                   const Tensor& v_cache, const Tensor& block_table,
                   const Tensor& seq_lens, double scale) {
    return attn_decode_impl(q, k_cache, v_cache, block_table, seq_lens,
                            scale, false)[0];
}

// out + per-(seq, q-head) (m, l): the flash merge state that context
// parallelism exchanges across ranks (parallel/cp.py)
std::vector<Tensor> attn_decode_lse(const Tensor& q, const Tensor& k_cache,
                                    const Tensor& v_cache,
                                    const Tensor& block_table,
                                    const Tensor& seq_lens, double scale) {
    return attn_decode_impl(q, k_cache, v_cache, block_table, seq_lens,
                            scale, true);
}

Tensor attn_prefill(const Tensor& q, const Tensor& k, const Tensor& v,
                    const Tensor& cu_seqlens, long max_seqlen, double scale,
                    bool causal) {
    check_bf16(q, "q");
    check_bf16(k, "k");
    TORCH_CHECK(cu_seqlens.scalar_type() == torch::kInt32 &&
                cu_seqlens.is_contiguous());
    const int T = q.size(0), nq = q.size(1), hd = q.size(2);
    const int nkv = k.size(1);
    const int nseq = cu_seqlens.size(0) - 1;
    TORCH_CHECK(nq % nkv == 0);
    TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == hd);
    TORCH_CHECK(k.stride(2) == 1 && k.stride(1) == hd);
    TORCH_CHECK(v.stride(2) == 1 && v.stride(1) == hd);
    Tensor out = torch::empty({T, nq, hd}, q.options());
    launch_attn_prefill(bf16p(q), bf16p(k), bf16p(v),
                        cu_seqlens.data_ptr<int>(), bf16p_mut(out), nseq, nq,
                        nkv, hd, q.stride(0), k.stride(0), v.stride(0),
                        (int)max_seqlen, (float)scale, causal ? 1 : 0,
                        stream());
    return out;
}

Tensor attn_prefill_paged(const Tensor& q, const Tensor& k_cache,
                          const Tensor& v_cache, const Tensor& block_table,
                          const Tensor& seq_lens, const Tensor& cu_q,
                          long max_qlen, double scale) {
    check_bf16(q, "q");
    const bool fp8 = k_cache.scalar_type() == torch::kUInt8;
    if (!fp8) check_bf16(k_cache, "k_cache");
    TORCH_CHECK(block_table.scalar_type() == torch::kInt32 &&
                seq_lens.scalar_type() == torch::kInt32 &&
                cu_q.scalar_type() == torch::kInt32);
    TORCH_CHECK(block_table.is_contiguous() && cu_q.is_contiguous());
    const int T = q.size(0), nq = q.size(1), hd = q.size(2);
    const int nkv = k_cache.size(1), bs = k_cache.size(2);
    const int W = block_table.size(1);
    const int nseq = cu_q.size(0) - 1;
    TORCH_CHECK(nq % nkv == 0);
    TORCH_CHECK(hd % 8 == 0, "paged prefill needs hd % 8 == 0");
    TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == hd);
    TORCH_CHECK(k_cache.is_contiguous() && v_cache.is_contiguous());
    TORCH_CHECK(block_table.size(0) == nseq && seq_lens.size(0) == nseq);
    Tensor out = torch::empty({T, nq, hd}, q.options());
    launch_attn_prefill_paged(bf16p(q), k_cache.data_ptr(),
                              v_cache.data_ptr(), cu_q.data_ptr<int>(),
                              block_table.data_ptr<int>(),
                              seq_lens.data_ptr<int>(), bf16p_mut(out), nseq,
                              nq, nkv, hd, q.stride(0), W, bs, (int)max_qlen,
                              (float)scale, fp8 ? 1 : 0, stream());
    return out;
}

Tensor grouped_gemm(const Tensor& x, const Tensor& w, const Tensor& offs) {
    check_bf16(x, "x");
    check_bf16(w, "w");
    TORCH_CHECK(offs.scalar_type() == torch::kInt32 && offs.is_contiguous());
    TORCH_CHECK(x.is_contiguous() && w.is_contiguous());
    const int S = x.size(0), K = x.size(1);
    const int E = w.size(0), N = w.size(1);
    TORCH_CHECK(w.size(2) == K, "K mismatch");
    TORCH_CHECK(offs.size(0) == E + 1);
    TORCH_CHECK(N % 64 == 0, "N must be a multiple of 64");
    TORCH_CHECK(K % 128 == 0, "K must be a multiple of 128");
    Tensor out = torch::empty({S, N}, x.options());
    if (S > 0)
        launch_grouped_gemm(bf16p(x), bf16p(w), offs.data_ptr<int>(),
                            bf16p_mut(out), E, S, N, K, stream());
    return out;
}

Tensor mfma_probe_fp8(const Tensor& A, const Tensor& B) {
    TORCH_CHECK(A.scalar_type() == torch::kUInt8 &&
                B.scalar_type() == torch::kUInt8);
    TORCH_CHECK(A.sizes() == torch::IntArrayRef({16, 32}));
    TORCH_CHECK(B.sizes() == torch::IntArrayRef({32, 16}));
    TORCH_CHECK(A.is_contiguous() && B.is_contiguous());
    Tensor C = torch::empty({16, 16}, A.options().dtype(torch::kFloat32));
    launch_mfma_probe_fp8(A.data_ptr<unsigned char>(),
                          B.data_ptr<unsigned char>(), C.data_ptr<float>(),
                          stream());
    return C;
}

Tensor mfma_probe(const Tensor& A, const Tensor& B) {
    check_bf16(A, "A");
    check_bf16(B, "B");
    TORCH_CHECK(A.sizes() == torch::IntArrayRef({16, 32}));
    TORCH_CHECK(B.sizes() == torch::IntArrayRef({32, 16}));
    TORCH_CHECK(A.is_contiguous() && B.is_contiguous());
    Tensor C = torch::empty({16, 16},
                            A.options().dtype(torch::kFloat32));
    launch_mfma_probe(bf16p(A), bf16p(B), C.data_ptr<float>(), stream());
    return C;
}

void bw_probe_paged(const Tensor& pool, const Tensor& table, int64_t hd,
                    int64_t rows_per_page) {
    check_bf16(pool, "pool");
    Tensor sink = torch::zeros({1}, pool.options().dtype(torch::kFloat32));
    launch_bw_paged(bf16p(pool), table.data_ptr<int>(), pool.numel() / hd,
                    (int)hd, (int)rows_per_page, sink.data_ptr<float>(),
                    stream());
}

void bw_probe(const Tensor& pool, int64_t mode, int64_t hd, int64_t arg) {
    check_bf16(pool, "pool");
    Tensor sink = torch::zeros({1}, pool.options().dtype(torch::kFloat32));
    const long n = pool.numel();
    if (mode == 0)
        launch_bw_linear(bf16p(pool), n, sink.data_ptr<float>(), stream());
    else if (mode == 1)
        launch_bw_rowperlane(bf16p(pool), n / hd, hd, (int)arg,
                             sink.data_ptr<float>(), stream());
    else
        launch_bw_rowperinstr(bf16p(pool), n / hd, hd, (int)arg,
                              sink.data_ptr<float>(), stream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("rmsnorm", &rmsnorm, "fused RMSNorm (bf16)");
    m.def("fused_add_rmsnorm", &fused_add_rmsnorm,
          "residual += x; y = rmsnorm(residual)");
    m.def("rope_inplace", &rope_inplace, "llama RoPE in place");
    m.def("kv_cache_store", &kv_cache_store, "paged KV scatter");
    m.def("swiglu", &swiglu, "silu(g) * u");
    m.def("layernorm", &layernorm, "LayerNorm (bf16, weight+bias)");
    m.def("fused_add_layernorm", &fused_add_layernorm,
          "residual add + LayerNorm");
    m.def("gelu", &gelu, "gelu_new (tanh approximation)");
    m.def("attn_decode", &attn_decode, "paged GQA decode attention");
    m.def("attn_decode_lse", &attn_decode_lse,
          "paged GQA decode attention + per-head (m, l) merge state");
    m.def("attn_prefill", &attn_prefill, "varlen causal flash prefill");
    m.def("attn_prefill_paged", &attn_prefill_paged,
          "chunked prefill vs paged history");
    m.def("mfma_probe", &mfma_probe, "16x16x32 bf16 MFMA layout probe");
    m.def("mfma_probe_fp8", &mfma_probe_fp8,
          "16x16x32 fp8 MFMA layout probe");
    m.def("grouped_gemm", &grouped_gemm, "per-expert segment GEMM (MoE)");
    m.def("bw_probe", &bw_probe, "bandwidth pattern probe");
    m.def("bw_probe_paged", &bw_probe_paged, "paged row-per-lane probe");
}
