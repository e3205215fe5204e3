// pybind11 module smg_amd._core — host C++ tree + gfx950 GPU tree + hashing.
//
// Deliberately torch-free: device pointers/streams cross the boundary as
// integers, so the .so builds with plain hipcc and loads in any process.
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cstdint>
#include <stdexcept>
#include <vector>

#include "host_tree.cpp"

namespace py = pybind11;

// ---- GPU tree C API (gpu_tree.hip) ---------------------------------------
extern "C" {
void* smg_gpu_tree_create(int device, uint32_t node_cap, uint32_t table_size, uint32_t page_size,
                          uint32_t max_pages, uint32_t max_batch_reqs, uint32_t max_batch_tokens);
void smg_gpu_tree_destroy(void* p);
int smg_gpu_tree_run(void* p, int n_reqs, unsigned long long healthy_mask, const int* loads,
                     const int* processed, int n_workers, float cache_threshold, int imbalanced,
                     int do_insert, int forced_tenant, int mode, int* out_selected,
                     uint32_t* out_matched, uint32_t* out_tenant, uint32_t* out_depths);
void smg_gpu_tree_staging(void* p, uint32_t** tokens, uint32_t** offsets);
int smg_gpu_tree_remove_tenant(void* p, int slot);
int smg_gpu_tree_clear_entries(void* p, const unsigned long long* keys, int n, int slot);
int smg_gpu_tree_evict_older(void* p, uint32_t cutoff);
long long smg_gpu_tree_reclaim(void* p);
int smg_gpu_tree_stats(void* p, unsigned long long* out);
int smg_gpu_tree_clear(void* p);
int smg_hip_device_count();
// ---- BPE (bpe.hip) ----
void* smg_bpe_create(const unsigned long long* pair_keys, const unsigned long long* pair_vals,
                     uint32_t n_pairs, const uint32_t* byte_to_tok, uint32_t max_pieces,
                     uint32_t max_bytes, int use_gpu);
void smg_bpe_destroy(void* p);
int smg_bpe_on_gpu(void* p);
int smg_bpe_encode(void* p, const uint8_t* bytes, const uint32_t* offsets, uint32_t n_pieces,
                   uint32_t* out_tokens, uint32_t* out_counts);
uint32_t smg_bpe_pretokenize(const uint8_t* s, uint32_t n, uint32_t* piece_offsets, uint32_t max_pieces);
uint32_t smg_bpe_encode_long(void* p, const uint8_t* bytes, uint32_t len, uint32_t* out_tokens, uint32_t cap);
// ---- image (image.hip) ----
void* smg_img_create(int use_gpu);
void smg_img_destroy(void* p);
int smg_img_on_gpu(void* p);
int smg_img_resize_normalize(void* p, const uint8_t* in, int in_w, int in_h, int channels,
                             int out_w, int out_h, const float* mean, const float* stddev,
                             uint8_t* out_u8, float* out_f32);
// ---- decode attention (attn_decode.hip) ----
int smg_attn_decode_launch(const void* q, const void* k, const void* v, const void* pos,
                           void* out, int n_slots, int n_heads, int max_seq, int head_dim,
                           float scale, void* stream);
int smg_attn_decode_launch_gqa(const void* q, const void* k, const void* v, const void* pos,
                               void* out, int n_slots, int n_heads, int n_kv_heads, int max_seq,
                               int head_dim, float scale, void* stream, int kv_fp8);
int smg_attn_decode_launch_gqa2(const void* q, const void* k, const void* v, const void* pos,
                                void* out, int n_slots, int n_heads, int n_kv_heads, int max_seq,
                                int head_dim, float scale, void* stream, int kv_fp8);
int smg_rope_kv_store_launch_gqa(const void* qkv, const void* freqs, const void* pos,
                                 void* k_cache, void* v_cache, void* q_out, int n_slots,
                                 int n_heads, int n_kv_heads, int max_seq, int head_dim,
                                 void* stream, int kv_fp8);
int smg_rope_prefill_launch_gqa(const void* qkv, const void* freqs, const void* slots,
                                const void* starts, void* k_cache, void* v_cache, void* q_out,
                                void* k_out, void* v_out, int B, int L, int n_heads,
                                int n_kv_heads, int max_seq, int head_dim, void* stream,
                                int kv_fp8);
int smg_rms_gemm_launch(const void* a, const void* wt, const void* invrms, void* c, int M,
                        int K, int N, void* stream);
int smg_row_invrms_launch(const void* a, void* out, int m, int k, float eps, void* stream);
int smg_mfma_probe_launch(const void* afrag, const void* bfrag, void* out, void* stream);
int smg_attn_decode_launch_split(const void* q, const void* k, const void* v, const void* pos,
                                 void* out, void* part_out, void* part_ml, int n_slots,
                                 int n_heads, int n_kv_heads, int n_split, int max_seq,
                                 int head_dim, float scale, void* stream, int kv_fp8);
int smg_attn_decode_launch_ex(const void* q, const void* k, const void* v, const void* pos,
                              void* out, int n_slots, int n_heads, int max_seq, int head_dim,
                              float scale, void* stream, int kv_fp8);
// ---- fused decode elementwise (fused_decode.hip) ----
int smg_rope_kv_store_launch(const void* qkv, const void* freqs, const void* pos,
                             void* k_cache, void* v_cache, void* q_out,
                             int n_slots, int n_heads, int max_seq, int head_dim, void* stream);
int smg_rope_kv_store_launch_ex(const void* qkv, const void* freqs, const void* pos,
                                void* k_cache, void* v_cache, void* q_out,
                                int n_slots, int n_heads, int max_seq, int head_dim, void* stream,
                                int kv_fp8);
int smg_rope_prefill_launch_ex(const void* qkv, const void* freqs, const void* slots,
                               const void* starts, void* k_cache, void* v_cache, void* q_out,
                               void* k_out, void* v_out, int B, int L, int n_heads, int max_seq,
                               int head_dim, void* stream, int kv_fp8);
int smg_silu_mul_launch(const void* gu, void* out, long long rows, long long inner, void* stream);
int smg_lse_merge_launch(const void* o1, const void* o2, const void* lse1, const void* lse2,
                         void* out, long long rows, int head_dim, void* stream);
int smg_rope_prefill_launch(const void* qkv, const void* freqs, const void* slots,
                            const void* starts, void* k_cache, void* v_cache, void* q_out,
                            void* k_out, void* v_out, int B, int L, int n_heads, int max_seq,
                            int head_dim, void* stream);
}

#define BPE_MAX_PIECE 64

namespace {

class PyHostTree {
   public:
    explicit PyHostTree(uint32_t page_size) : tree_(page_size) {}

    py::tuple match(py::array_t<uint32_t, py::array::c_style | py::array::forcecast> toks, bool touch) {
        auto r = tree_.match(toks.data(), (uint32_t)toks.size(), touch);
        return py::make_tuple(r.tenant, r.matched, r.input);
    }
    uint32_t insert(py::array_t<uint32_t, py::array::c_style | py::array::forcecast> toks, int tenant) {
        return tree_.insert(toks.data(), (uint32_t)toks.size(), tenant);
    }
    void remove_tenant(int t) { tree_.remove_tenant(t); }
    uint32_t evict(size_t max_nodes) { return tree_.evict(max_nodes); }
    void clear() { tree_.clear(); }
    size_t size() const { return tree_.live_count(); }
    uint64_t tenant_tokens(int t) { return tree_.tenant_tokens(t); }
    uint32_t page_size() const { return tree_.page_size(); }

   private:
    smg::HostTree tree_;
};

class PyGpuTree {
   public:
    PyGpuTree(int device, uint32_t node_cap, uint32_t table_size, uint32_t page_size,
              uint32_t max_pages, uint32_t max_batch_reqs, uint32_t max_batch_tokens)
        : max_batch_reqs_(max_batch_reqs), max_batch_tokens_(max_batch_tokens) {
        h_ = smg_gpu_tree_create(device, node_cap, table_size, page_size, max_pages, max_batch_reqs,
                                 max_batch_tokens);
        if (!h_) throw std::runtime_error("smg_gpu_tree_create failed (no GPU / OOM)");
    }
    ~PyGpuTree() { smg_gpu_tree_destroy(h_); }

    // Batched match/decide/insert.  `tokens_flat` int32/uint32, `offsets`
    // int32 (n+1).  Returns (selected[int32], matched[uint32], tenant[uint32]).
    py::tuple run(py::array_t<uint32_t, py::array::c_style | py::array::forcecast> tokens_flat,
                  py::array_t<uint32_t, py::array::c_style | py::array::forcecast> offsets,
                  unsigned long long healthy_mask, std::vector<int> loads, std::vector<int> processed,
                  int n_workers, float cache_threshold, bool imbalanced, bool do_insert,
                  int forced_tenant, int mode) {
        int n_reqs = (int)offsets.size() - 1;
        if (n_reqs < 1 || (uint32_t)n_reqs > max_batch_reqs_)
            throw std::runtime_error("batch size out of range");
        if ((uint32_t)tokens_flat.size() > max_batch_tokens_)
            throw std::runtime_error("batch token count out of range");
        uint32_t *stg_tokens, *stg_offsets;
        smg_gpu_tree_staging(h_, &stg_tokens, &stg_offsets);
        std::memcpy(stg_tokens, tokens_flat.data(), tokens_flat.size() * 4);
        std::memcpy(stg_offsets, offsets.data(), ((size_t)n_reqs + 1) * 4);
        loads.resize(64, 0);
        processed.resize(64, 0);
        py::array_t<int> sel(n_reqs);
        py::array_t<uint32_t> matched(n_reqs);
        py::array_t<uint32_t> tenant(n_reqs);
        py::array_t<uint32_t> depths(mode == 1 ? n_reqs * 64 : 0);
        int rc;
        {
            py::gil_scoped_release nogil;
            rc = smg_gpu_tree_run(h_, n_reqs, healthy_mask, loads.data(), processed.data(),
                                  n_workers, cache_threshold, imbalanced ? 1 : 0, do_insert ? 1 : 0,
                                  forced_tenant, mode, sel.mutable_data(), matched.mutable_data(),
                                  tenant.mutable_data(), mode == 1 ? depths.mutable_data() : nullptr);
        }
        if (rc != 0) throw std::runtime_error("smg_gpu_tree_run failed rc=" + std::to_string(rc));
        return py::make_tuple(sel, matched, tenant, depths);
    }

    void remove_tenant(int slot) {
        if (smg_gpu_tree_remove_tenant(h_, slot)) throw std::runtime_error("remove_tenant failed");
    }
    void clear_entries(py::array_t<uint64_t, py::array::c_style | py::array::forcecast> keys, int slot) {
        if (smg_gpu_tree_clear_entries(h_, (const unsigned long long*)keys.data(), (int)keys.size(), slot))
            throw std::runtime_error("clear_entries failed");
    }
    void evict_older(uint32_t cutoff) {
        if (smg_gpu_tree_evict_older(h_, cutoff)) throw std::runtime_error("evict failed");
    }
    long long reclaim() {
        long long n = smg_gpu_tree_reclaim(h_);
        if (n < 0) throw std::runtime_error("reclaim failed");
        return n;
    }
    py::dict stats() {
        unsigned long long out[68];
        if (smg_gpu_tree_stats(h_, out)) throw std::runtime_error("stats failed");
        py::dict d;
        py::list per_tenant;
        for (int i = 0; i < 64; ++i) per_tenant.append(out[i]);
        d["tenant_nodes"] = per_tenant;
        d["live_nodes"] = out[64];
        d["allocated_nodes"] = out[65];
        d["clock"] = out[66];
        d["free_nodes"] = out[67];
        return d;
    }
    void clear() {
        if (smg_gpu_tree_clear(h_)) throw std::runtime_error("clear failed");
    }

   private:
    void* h_;
    uint32_t max_batch_reqs_, max_batch_tokens_;
};

uint64_t py_page_hash(py::array_t<uint32_t, py::array::c_style | py::array::forcecast> toks) {
    smg::HostTree t(1);
    return t.page_hash(toks.data(), (uint32_t)toks.size());
}

class PyBpe {
   public:
    PyBpe(py::array_t<uint64_t, py::array::c_style | py::array::forcecast> pair_keys,
          py::array_t<uint64_t, py::array::c_style | py::array::forcecast> pair_vals,
          py::array_t<uint32_t, py::array::c_style | py::array::forcecast> byte_to_tok,
          uint32_t max_pieces, uint32_t max_bytes, bool use_gpu) {
        if (byte_to_tok.size() != 256) throw std::runtime_error("byte_to_tok must have 256 entries");
        h_ = smg_bpe_create((const unsigned long long*)pair_keys.data(),
                            (const unsigned long long*)pair_vals.data(), (uint32_t)pair_keys.size(),
                            byte_to_tok.data(), max_pieces, max_bytes, use_gpu ? 1 : 0);
        if (!h_) throw std::runtime_error("smg_bpe_create failed");
        max_pieces_ = max_pieces;
    }
    ~PyBpe() { smg_bpe_destroy(h_); }

    bool on_gpu() const { return smg_bpe_on_gpu(h_) != 0; }

    // pieces: flattened bytes + offsets -> per-piece token lists.  Pieces
    // longer than BPE_MAX_PIECE take the uncapped host path; the rest run in
    // one GPU batch (or the identical host loop without a GPU).
    py::tuple encode_pieces(py::bytes data,
                            py::array_t<uint32_t, py::array::c_style | py::array::forcecast> offsets) {
        std::string buf = data;
        const uint8_t* bytes = (const uint8_t*)buf.data();
        uint32_t n_pieces = (uint32_t)offsets.size() - 1;
        const uint32_t* offs = offsets.data();
        // split short/long
        std::vector<uint32_t> short_offsets{0};
        std::vector<uint8_t> short_bytes;
        std::vector<int> kind(n_pieces);  // index into short list, or -1-longIdx
        std::vector<std::vector<uint32_t>> long_tokens;
        for (uint32_t i = 0; i < n_pieces; ++i) {
            uint32_t len = offs[i + 1] - offs[i];
            if (len > BPE_MAX_PIECE) {
                std::vector<uint32_t> out(len);
                uint32_t n;
                {
                    py::gil_scoped_release nogil;
                    n = smg_bpe_encode_long(h_, bytes + offs[i], len, out.data(), len);
                }
                out.resize(n);
                kind[i] = -1 - (int)long_tokens.size();
                long_tokens.push_back(std::move(out));
            } else {
                kind[i] = (int)(short_offsets.size() - 1);
                short_bytes.insert(short_bytes.end(), bytes + offs[i], bytes + offs[i + 1]);
                short_offsets.push_back((uint32_t)short_bytes.size());
            }
        }
        uint32_t n_short = (uint32_t)short_offsets.size() - 1;
        std::vector<uint32_t> out_tokens((size_t)n_short * BPE_MAX_PIECE);
        std::vector<uint32_t> short_counts(n_short);
        if (n_short) {
            int rc;
            {
                py::gil_scoped_release nogil;
                rc = smg_bpe_encode(h_, short_bytes.data(), short_offsets.data(), n_short,
                                    out_tokens.data(), short_counts.data());
            }
            if (rc != 0) throw std::runtime_error("smg_bpe_encode rc=" + std::to_string(rc));
        }
        // reassemble in original piece order
        py::array_t<uint32_t> counts(n_pieces);
        uint32_t total = 0;
        for (uint32_t i = 0; i < n_pieces; ++i) {
            uint32_t c = kind[i] >= 0 ? short_counts[kind[i]] : (uint32_t)long_tokens[-1 - kind[i]].size();
            counts.mutable_data()[i] = c;
            total += c;
        }
        py::array_t<uint32_t> flat(total);
        uint32_t w = 0;
        for (uint32_t i = 0; i < n_pieces; ++i) {
            if (kind[i] >= 0) {
                uint32_t* src = out_tokens.data() + (size_t)kind[i] * BPE_MAX_PIECE;
                for (uint32_t j = 0; j < short_counts[kind[i]]; ++j) flat.mutable_data()[w++] = src[j];
            } else {
                for (uint32_t v : long_tokens[-1 - kind[i]]) flat.mutable_data()[w++] = v;
            }
        }
        return py::make_tuple(flat, counts);
    }

    py::array_t<uint32_t> pretokenize(py::bytes data) {
        std::string buf = data;
        std::vector<uint32_t> offs(max_pieces_ + 2);
        uint32_t np = smg_bpe_pretokenize((const uint8_t*)buf.data(), (uint32_t)buf.size(),
                                          offs.data(), max_pieces_);
        py::array_t<uint32_t> out(np + 1);
        std::memcpy(out.mutable_data(), offs.data(), sizeof(uint32_t) * (np + 1));
        return out;
    }

   private:
    void* h_;
    uint32_t max_pieces_;
};

class PyImg {
   public:
    explicit PyImg(bool use_gpu) { h_ = smg_img_create(use_gpu ? 1 : 0); }
    ~PyImg() { smg_img_destroy(h_); }
    bool on_gpu() const { return smg_img_on_gpu(h_) != 0; }

    // image: u8 HWC array (H, W, C); returns (u8 HWC resized | None,
    // f32 CHW normalized | None)
    py::tuple resize_normalize(py::array_t<uint8_t, py::array::c_style | py::array::forcecast> image,
                               int out_w, int out_h, py::object mean, py::object std,
                               bool want_u8, bool want_f32) {
        if (image.ndim() != 3) throw std::runtime_error("image must be HWC u8");
        int in_h = (int)image.shape(0), in_w = (int)image.shape(1), ch = (int)image.shape(2);
        if (ch < 1 || ch > 4) throw std::runtime_error("1..4 channels supported");
        std::vector<float> mv(ch, 0.f), sv(ch, 1.f);
        if (!mean.is_none()) {
            auto m = mean.cast<std::vector<float>>();
            for (int i = 0; i < ch && i < (int)m.size(); ++i) mv[i] = m[i];
        }
        if (!std.is_none()) {
            auto s = std.cast<std::vector<float>>();
            for (int i = 0; i < ch && i < (int)s.size(); ++i) sv[i] = s[i];
        }
        py::array_t<uint8_t> out_u8;
        py::array_t<float> out_f32;
        uint8_t* pu8 = nullptr;
        float* pf32 = nullptr;
        if (want_u8) {
            out_u8 = py::array_t<uint8_t>({out_h, out_w, ch});
            pu8 = out_u8.mutable_data();
        }
        if (want_f32) {
            out_f32 = py::array_t<float>({ch, out_h, out_w});
            pf32 = out_f32.mutable_data();
        }
        int rc;
        {
            py::gil_scoped_release nogil;
            rc = smg_img_resize_normalize(h_, image.data(), in_w, in_h, ch, out_w, out_h,
                                          mv.data(), sv.data(), pu8, pf32);
        }
        if (rc != 0) throw std::runtime_error("resize failed rc=" + std::to_string(rc));
        return py::make_tuple(want_u8 ? py::object(out_u8) : py::none(),
                              want_f32 ? py::object(out_f32) : py::none());
    }

   private:
    void* h_;
};

}  // namespace

PYBIND11_MODULE(_core, m) {
    m.doc() = "smg_amd native core: host C++ radix tree + gfx950 GPU radix tree";
    m.def("hip_device_count", &smg_hip_device_count);
    m.def("page_hash", &py_page_hash);
    // decode attention: raw device pointers + stream (ints from torch)
    m.def("attn_decode",
          [](uintptr_t q, uintptr_t k, uintptr_t v, uintptr_t pos, uintptr_t out, int n_slots,
             int n_heads, int max_seq, int head_dim, float scale, uintptr_t stream, int kv_fp8,
             int n_kv_heads) {
              if (n_kv_heads <= 0) n_kv_heads = n_heads;
              int rc = smg_attn_decode_launch_gqa((const void*)q, (const void*)k, (const void*)v,
                                                  (const void*)pos, (void*)out, n_slots, n_heads,
                                                  n_kv_heads, max_seq, head_dim, scale,
                                                  (void*)stream, kv_fp8);
              if (rc != 0) throw std::runtime_error("attn_decode launch failed rc=" + std::to_string(rc));
          },
          py::arg("q"), py::arg("k"), py::arg("v"), py::arg("pos"), py::arg("out"),
          py::arg("n_slots"), py::arg("n_heads"), py::arg("max_seq"), py::arg("head_dim"),
          py::arg("scale"), py::arg("stream"), py::arg("kv_fp8") = 0, py::arg("n_kv_heads") = 0);
    // v9 decode attention: two-tile softmax + v_dot2c_f32_bf16 K phase
    m.def("attn_decode2",
          [](uintptr_t q, uintptr_t k, uintptr_t v, uintptr_t pos, uintptr_t out, int n_slots,
             int n_heads, int max_seq, int head_dim, float scale, uintptr_t stream, int kv_fp8,
             int n_kv_heads) {
              if (n_kv_heads <= 0) n_kv_heads = n_heads;
              int rc = smg_attn_decode_launch_gqa2((const void*)q, (const void*)k, (const void*)v,
                                                   (const void*)pos, (void*)out, n_slots, n_heads,
                                                   n_kv_heads, max_seq, head_dim, scale,
                                                   (void*)stream, kv_fp8);
              if (rc != 0) throw std::runtime_error("attn_decode2 launch failed rc=" + std::to_string(rc));
          },
          py::arg("q"), py::arg("k"), py::arg("v"), py::arg("pos"), py::arg("out"),
          py::arg("n_slots"), py::arg("n_heads"), py::arg("max_seq"), py::arg("head_dim"),
          py::arg("scale"), py::arg("stream"), py::arg("kv_fp8") = 0, py::arg("n_kv_heads") = 0);
    // fused rms_norm + MFMA GEMM (rms_gemm.hip): C = invrms ⊙ (A @ Wt^T)
    m.def("row_invrms",
          [](uintptr_t a, uintptr_t out, int m, int k, float eps, uintptr_t stream) {
              int rc = smg_row_invrms_launch((const void*)a, (void*)out, m, k, eps, (void*)stream);
              if (rc != 0) throw std::runtime_error("row_invrms launch failed rc=" + std::to_string(rc));
          },
          py::arg("a"), py::arg("out"), py::arg("m"), py::arg("k"), py::arg("eps"),
          py::arg("stream"));
    m.def("mfma_probe",
          [](uintptr_t afrag, uintptr_t bfrag, uintptr_t out, uintptr_t stream) {
              int rc = smg_mfma_probe_launch((const void*)afrag, (const void*)bfrag,
                                             (void*)out, (void*)stream);
              if (rc != 0) throw std::runtime_error("mfma_probe launch failed");
          },
          py::arg("afrag"), py::arg("bfrag"), py::arg("out"), py::arg("stream"));
    m.def("rms_gemm",
          [](uintptr_t a, uintptr_t wt, uintptr_t invrms, uintptr_t c, int M, int K, int N,
             uintptr_t stream) {
              int rc = smg_rms_gemm_launch((const void*)a, (const void*)wt, (const void*)invrms,
                                           (void*)c, M, K, N, (void*)stream);
              if (rc != 0) throw std::runtime_error("rms_gemm launch failed rc=" + std::to_string(rc));
          },
          py::arg("a"), py::arg("wt"), py::arg("invrms"), py::arg("c"), py::arg("M"),
          py::arg("K"), py::arg("N"), py::arg("stream"));
    // T-split (flash-decoding) decode attention + LSE merge (v8): restores
    // chip occupancy when n_slots*n_kv_heads alone under-fills the CUs
    m.def("attn_decode_split",
          [](uintptr_t q, uintptr_t k, uintptr_t v, uintptr_t pos, uintptr_t out,
             uintptr_t part_out, uintptr_t part_ml, int n_slots, int n_heads, int n_kv_heads,
             int n_split, int max_seq, int head_dim, float scale, uintptr_t stream, int kv_fp8) {
              if (n_kv_heads <= 0) n_kv_heads = n_heads;
              int rc = smg_attn_decode_launch_split(
                  (const void*)q, (const void*)k, (const void*)v, (const void*)pos, (void*)out,
                  (void*)part_out, (void*)part_ml, n_slots, n_heads, n_kv_heads, n_split,
                  max_seq, head_dim, scale, (void*)stream, kv_fp8);
              if (rc != 0)
                  throw std::runtime_error("attn_decode_split launch failed rc=" + std::to_string(rc));
          },
          py::arg("q"), py::arg("k"), py::arg("v"), py::arg("pos"), py::arg("out"),
          py::arg("part_out"), py::arg("part_ml"), py::arg("n_slots"), py::arg("n_heads"),
          py::arg("n_kv_heads"), py::arg("n_split"), py::arg("max_seq"), py::arg("head_dim"),
          py::arg("scale"), py::arg("stream"), py::arg("kv_fp8") = 0);
    // fused rope + KV-store + q-pack (fused_decode.hip)
    m.def("rope_kv_store",
          [](uintptr_t qkv, uintptr_t freqs, uintptr_t pos, uintptr_t k_cache, uintptr_t v_cache,
             uintptr_t q_out, int n_slots, int n_heads, int max_seq, int head_dim, uintptr_t stream,
             int kv_fp8, int n_kv_heads) {
              if (n_kv_heads <= 0) n_kv_heads = n_heads;
              int rc = smg_rope_kv_store_launch_gqa((const void*)qkv, (const void*)freqs,
                                                    (const void*)pos, (void*)k_cache, (void*)v_cache,
                                                    (void*)q_out, n_slots, n_heads, n_kv_heads,
                                                    max_seq, head_dim, (void*)stream, kv_fp8);
              if (rc != 0) throw std::runtime_error("rope_kv_store launch failed rc=" + std::to_string(rc));
          },
          py::arg("qkv"), py::arg("freqs"), py::arg("pos"), py::arg("k_cache"), py::arg("v_cache"),
          py::arg("q_out"), py::arg("n_slots"), py::arg("n_heads"), py::arg("max_seq"),
          py::arg("head_dim"), py::arg("stream"), py::arg("kv_fp8") = 0, py::arg("n_kv_heads") = 0);
    m.def("rope_prefill",
          [](uintptr_t qkv, uintptr_t freqs, uintptr_t slots, uintptr_t starts, uintptr_t k_cache,
             uintptr_t v_cache, uintptr_t q_out, uintptr_t k_out, uintptr_t v_out, int B, int L,
             int n_heads, int max_seq, int head_dim, uintptr_t stream, int kv_fp8, int n_kv_heads) {
              if (n_kv_heads <= 0) n_kv_heads = n_heads;
              int rc = smg_rope_prefill_launch_gqa(
                  (const void*)qkv, (const void*)freqs, (const void*)slots, (const void*)starts,
                  (void*)k_cache, (void*)v_cache, (void*)q_out, (void*)k_out, (void*)v_out, B, L,
                  n_heads, n_kv_heads, max_seq, head_dim, (void*)stream, kv_fp8);
              if (rc != 0) throw std::runtime_error("rope_prefill launch failed rc=" + std::to_string(rc));
          },
          py::arg("qkv"), py::arg("freqs"), py::arg("slots"), py::arg("starts"), py::arg("k_cache"),
          py::arg("v_cache"), py::arg("q_out"), py::arg("k_out"), py::arg("v_out"), py::arg("B"),
          py::arg("L"), py::arg("n_heads"), py::arg("max_seq"), py::arg("head_dim"),
          py::arg("stream"), py::arg("kv_fp8") = 0, py::arg("n_kv_heads") = 0);
    m.def("lse_merge",
          [](uintptr_t o1, uintptr_t o2, uintptr_t lse1, uintptr_t lse2, uintptr_t out,
             long long rows, int head_dim, uintptr_t stream) {
              int rc = smg_lse_merge_launch((const void*)o1, (const void*)o2, (const void*)lse1,
                                            (const void*)lse2, (void*)out, rows, head_dim,
                                            (void*)stream);
              if (rc != 0) throw std::runtime_error("lse_merge launch failed rc=" + std::to_string(rc));
          },
          py::arg("o1"), py::arg("o2"), py::arg("lse1"), py::arg("lse2"), py::arg("out"),
          py::arg("rows"), py::arg("head_dim"), py::arg("stream"));
    m.def("silu_mul",
          [](uintptr_t gu, uintptr_t out, long long rows, long long inner, uintptr_t stream) {
              int rc = smg_silu_mul_launch((const void*)gu, (void*)out, rows, inner, (void*)stream);
              if (rc != 0) throw std::runtime_error("silu_mul launch failed rc=" + std::to_string(rc));
          },
          py::arg("gu"), py::arg("out"), py::arg("rows"), py::arg("inner"), py::arg("stream"));

    py::class_<PyHostTree>(m, "HostTokenTree")
        .def(py::init<uint32_t>(), py::arg("page_size") = 16)
        .def("match", &PyHostTree::match, py::arg("tokens"), py::arg("touch") = true)
        .def("insert", &PyHostTree::insert)
        .def("remove_tenant", &PyHostTree::remove_tenant)
        .def("evict", &PyHostTree::evict)
        .def("clear", &PyHostTree::clear)
        .def("tenant_tokens", &PyHostTree::tenant_tokens)
        .def("page_size", &PyHostTree::page_size)
        .def("__len__", &PyHostTree::size);

    py::class_<PyGpuTree>(m, "GpuTree")
        .def(py::init<int, uint32_t, uint32_t, uint32_t, uint32_t, uint32_t, uint32_t>(),
             py::arg("device") = 0, py::arg("node_cap") = 1u << 22, py::arg("table_size") = 1u << 23,
             py::arg("page_size") = 16, py::arg("max_pages") = 1024, py::arg("max_batch_reqs") = 4096,
             py::arg("max_batch_tokens") = 1u << 22)
        .def("run", &PyGpuTree::run, py::arg("tokens_flat"), py::arg("offsets"),
             py::arg("healthy_mask"), py::arg("loads"), py::arg("processed"), py::arg("n_workers"),
             py::arg("cache_threshold") = 0.3f, py::arg("imbalanced") = false,
             py::arg("do_insert") = true, py::arg("forced_tenant") = -1, py::arg("mode") = 0)
        .def("remove_tenant", &PyGpuTree::remove_tenant)
        .def("clear_entries", &PyGpuTree::clear_entries)
        .def("evict_older", &PyGpuTree::evict_older)
        .def("reclaim", &PyGpuTree::reclaim)
        .def("stats", &PyGpuTree::stats)
        .def("clear", &PyGpuTree::clear);

    py::class_<PyBpe>(m, "Bpe")
        .def(py::init<py::array_t<uint64_t, py::array::c_style | py::array::forcecast>,
                      py::array_t<uint64_t, py::array::c_style | py::array::forcecast>,
                      py::array_t<uint32_t, py::array::c_style | py::array::forcecast>, uint32_t,
                      uint32_t, bool>(),
             py::arg("pair_keys"), py::arg("pair_vals"), py::arg("byte_to_tok"),
             py::arg("max_pieces") = 1u << 20, py::arg("max_bytes") = 1u << 24,
             py::arg("use_gpu") = true)
        .def("on_gpu", &PyBpe::on_gpu)
        .def("encode_pieces", &PyBpe::encode_pieces)
        .def("pretokenize", &PyBpe::pretokenize);

    py::class_<PyImg>(m, "ImageProcessor")
        .def(py::init<bool>(), py::arg("use_gpu") = true)
        .def("on_gpu", &PyImg::on_gpu)
        .def("resize_normalize", &PyImg::resize_normalize, py::arg("image"), py::arg("out_w"),
             py::arg("out_h"), py::arg("mean") = py::none(), py::arg("std") = py::none(),
             py::arg("want_u8") = true, py::arg("want_f32") = true);
}
