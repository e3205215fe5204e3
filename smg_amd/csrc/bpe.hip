// Batch byte-level BPE tokenization on gfx950 (MI355X).
//
// Re-designs the reference's batch tokenizer (crates/tokenizer/src/
// huggingface.rs:485 encode_batch + rayon parallel cache/mod.rs:247) as a
// GPU pipeline:
//
//   host:   GPT-2-style pre-tokenization scan (C++: letter/digit/punct/space
//           classes + contraction suffixes) -> byte pieces;
//   device: BPE merge loop per piece — one thread per piece, piece bytes
//           staged in LDS, merge-pair ranks probed in a device hash table
//           (u64 key = left<<32|right -> rank<<32|merged id).
//
// Pieces from byte-level pre-tokenization are short (mean 4-8 bytes, capped
// at MAX_PIECE), so thread-per-piece gives tens of thousands of concurrent
// merge loops; the table is read-only after upload (no visibility hazards).
// Host C++ runs the IDENTICAL merge loop for the CPU fallback and the
// differential tests.
#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>
#include <cstring>
#include <string>
#include <vector>

#define BPE_MAX_PIECE 64

// ---------------------------------------------------------------------------
// shared merge-table lookup (host + device via HIP __host__ __device__)
// ---------------------------------------------------------------------------
struct BpeTable {
    const unsigned long long* keys;  // 0 = empty
    const unsigned long long* vals;  // rank<<32 | merged_id
    uint32_t mask;
};

__host__ __device__ inline unsigned long long bpe_mix(unsigned long long x) {
    x += 0x9E3779B97F4A7C15ull;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
    return x ^ (x >> 31);
}

__host__ __device__ inline unsigned long long bpe_lookup(const BpeTable& T, uint32_t a, uint32_t b) {
    unsigned long long key = ((unsigned long long)a << 32) | b;
    unsigned long long h = bpe_mix(key);
    for (uint32_t i = 0; i < 64; ++i) {
        uint32_t slot = (uint32_t)(h + i) & T.mask;
        unsigned long long k = T.keys[slot];
        if (k == 0) return ~0ull;       // empty: no such merge
        if (k == key + 1) return T.vals[slot];  // stored key+1 so 0 stays "empty"
    }
    return ~0ull;
}

// The merge loop, shared verbatim between the device kernel and the host
// fallback: greedy lowest-rank adjacent merge (canonical BPE).
template <typename TokT>
__host__ __device__ inline int bpe_merge_loop(const BpeTable& T, TokT* t, int n) {
    while (n > 1) {
        unsigned long long best = ~0ull;
        int best_i = -1;
        for (int i = 0; i + 1 < n; ++i) {
            unsigned long long v = bpe_lookup(T, (uint32_t)t[i], (uint32_t)t[i + 1]);
            if (v < best) {
                best = v;
                best_i = i;
            }
        }
        if (best_i < 0) break;
        t[best_i] = (TokT)(best & 0xffffffffu);
        for (int j = best_i + 1; j + 1 < n; ++j) t[j] = t[j + 1];
        --n;
    }
    return n;
}

// ---------------------------------------------------------------------------
// kernel: one thread per piece
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256) smg_bpe_encode_pieces(
    BpeTable T,
    const uint8_t* bytes,        // flattened piece bytes
    const uint32_t* offsets,     // n_pieces + 1
    const uint32_t* byte_to_tok, // 256-entry initial alphabet
    int n_pieces,
    uint32_t* out_tokens,        // [n_pieces * BPE_MAX_PIECE]
    uint32_t* out_counts) {      // [n_pieces]
    int p = blockIdx.x * blockDim.x + threadIdx.x;
    if (p >= n_pieces) return;
    uint32_t beg = offsets[p], end = offsets[p + 1];
    int n = (int)(end - beg);
    if (n > BPE_MAX_PIECE) n = BPE_MAX_PIECE;
    uint32_t t[BPE_MAX_PIECE];
    for (int i = 0; i < n; ++i) t[i] = byte_to_tok[bytes[beg + i]];
    n = bpe_merge_loop(T, t, n);
    uint32_t* out = out_tokens + (size_t)p * BPE_MAX_PIECE;
    for (int i = 0; i < n; ++i) out[i] = t[i];
    out_counts[p] = (uint32_t)n;
}

// ---------------------------------------------------------------------------
// host-side state
// ---------------------------------------------------------------------------
struct BpeHost {
    // device
    unsigned long long* d_keys = nullptr;
    unsigned long long* d_vals = nullptr;
    uint32_t* d_byte_to_tok = nullptr;
    uint8_t* d_bytes = nullptr;
    uint32_t* d_offsets = nullptr;
    uint32_t* d_out = nullptr;
    uint32_t* d_counts = nullptr;
    // host copy (CPU fallback + tests)
    std::vector<unsigned long long> h_keys, h_vals;
    std::vector<uint32_t> h_byte_to_tok;
    uint32_t mask = 0;
    uint32_t max_pieces = 0, max_bytes = 0;
    hipStream_t stream{};
    bool on_gpu = false;
};

extern "C" void* smg_bpe_create(const unsigned long long* pair_keys, const unsigned long long* pair_vals,
                                uint32_t n_pairs, const uint32_t* byte_to_tok,
                                uint32_t max_pieces, uint32_t max_bytes, int use_gpu) {
    BpeHost* h = new BpeHost();
    uint32_t table_size = 1;
    while (table_size < n_pairs * 2 + 16) table_size <<= 1;
    h->mask = table_size - 1;
    h->h_keys.assign(table_size, 0);
    h->h_vals.assign(table_size, 0);
    for (uint32_t i = 0; i < n_pairs; ++i) {
        unsigned long long key = pair_keys[i];
        unsigned long long hsh = bpe_mix(key);
        for (uint32_t j = 0;; ++j) {
            uint32_t slot = (uint32_t)(hsh + j) & h->mask;
            if (h->h_keys[slot] == 0) {
                h->h_keys[slot] = key + 1;
                h->h_vals[slot] = pair_vals[i];
                break;
            }
        }
    }
    h->h_byte_to_tok.assign(byte_to_tok, byte_to_tok + 256);
    h->max_pieces = max_pieces;
    h->max_bytes = max_bytes;
    if (use_gpu) {
        int ndev = 0;
        if (hipGetDeviceCount(&ndev) == hipSuccess && ndev > 0) {
            bool ok = true;
            auto chk = [&](hipError_t e) { if (e != hipSuccess) ok = false; };
            chk(hipStreamCreate(&h->stream));
            chk(hipMalloc(&h->d_keys, sizeof(unsigned long long) * table_size));
            chk(hipMalloc(&h->d_vals, sizeof(unsigned long long) * table_size));
            chk(hipMalloc(&h->d_byte_to_tok, sizeof(uint32_t) * 256));
            chk(hipMalloc(&h->d_bytes, max_bytes));
            chk(hipMalloc(&h->d_offsets, sizeof(uint32_t) * (max_pieces + 1)));
            chk(hipMalloc(&h->d_out, sizeof(uint32_t) * (size_t)max_pieces * BPE_MAX_PIECE));
            chk(hipMalloc(&h->d_counts, sizeof(uint32_t) * max_pieces));
            if (ok) {
                chk(hipMemcpy(h->d_keys, h->h_keys.data(), sizeof(unsigned long long) * table_size, hipMemcpyHostToDevice));
                chk(hipMemcpy(h->d_vals, h->h_vals.data(), sizeof(unsigned long long) * table_size, hipMemcpyHostToDevice));
                chk(hipMemcpy(h->d_byte_to_tok, byte_to_tok, sizeof(uint32_t) * 256, hipMemcpyHostToDevice));
            }
            h->on_gpu = ok;
        }
    }
    return h;
}

extern "C" void smg_bpe_destroy(void* p) {
    BpeHost* h = (BpeHost*)p;
    if (!h) return;
    if (h->on_gpu) {
        hipFree(h->d_keys); hipFree(h->d_vals); hipFree(h->d_byte_to_tok);
        hipFree(h->d_bytes); hipFree(h->d_offsets); hipFree(h->d_out); hipFree(h->d_counts);
        hipStreamDestroy(h->stream);
    }
    delete h;
}

extern "C" int smg_bpe_on_gpu(void* p) { return ((BpeHost*)p)->on_gpu ? 1 : 0; }

// Encode pieces; returns 0 on success.  out_tokens is [n_pieces*BPE_MAX_PIECE],
// out_counts [n_pieces].  Runs on GPU when available, host loop otherwise.
extern "C" int smg_bpe_encode(void* p, const uint8_t* bytes, const uint32_t* offsets,
                              uint32_t n_pieces, uint32_t* out_tokens, uint32_t* out_counts) {
    BpeHost* h = (BpeHost*)p;
    uint32_t n_bytes = offsets[n_pieces];
    BpeTable T{h->h_keys.data(), h->h_vals.data(), h->mask};
    if (h->on_gpu && n_pieces > 0) {
        if (n_pieces > h->max_pieces || n_bytes > h->max_bytes) return -1;
        hipMemcpyAsync(h->d_bytes, bytes, n_bytes, hipMemcpyHostToDevice, h->stream);
        hipMemcpyAsync(h->d_offsets, offsets, sizeof(uint32_t) * (n_pieces + 1), hipMemcpyHostToDevice, h->stream);
        BpeTable D{h->d_keys, h->d_vals, h->mask};
        int threads = 256;
        int blocks = (int)((n_pieces + threads - 1) / threads);
        hipLaunchKernelGGL(smg_bpe_encode_pieces, dim3(blocks), dim3(threads), 0, h->stream,
                           D, h->d_bytes, h->d_offsets, h->d_byte_to_tok, (int)n_pieces,
                           h->d_out, h->d_counts);
        hipMemcpyAsync(out_tokens, h->d_out, sizeof(uint32_t) * (size_t)n_pieces * BPE_MAX_PIECE,
                       hipMemcpyDeviceToHost, h->stream);
        hipMemcpyAsync(out_counts, h->d_counts, sizeof(uint32_t) * n_pieces, hipMemcpyDeviceToHost, h->stream);
        return hipStreamSynchronize(h->stream) == hipSuccess ? 0 : -2;
    }
    // host fallback: identical merge loop
    for (uint32_t i = 0; i < n_pieces; ++i) {
        uint32_t beg = offsets[i], end = offsets[i + 1];
        int n = (int)(end - beg);
        if (n > BPE_MAX_PIECE) n = BPE_MAX_PIECE;
        uint32_t t[BPE_MAX_PIECE];
        for (int j = 0; j < n; ++j) t[j] = h->h_byte_to_tok[bytes[beg + j]];
        n = bpe_merge_loop(T, t, n);
        for (int j = 0; j < n; ++j) out_tokens[(size_t)i * BPE_MAX_PIECE + j] = t[j];
        out_counts[i] = (uint32_t)n;
    }
    return 0;
}

// Arbitrary-length host encode for pieces longer than BPE_MAX_PIECE (rare:
// long words / character runs).  Identical merge loop, heap storage.
extern "C" uint32_t smg_bpe_encode_long(void* p, const uint8_t* bytes, uint32_t len,
                                        uint32_t* out_tokens, uint32_t cap) {
    BpeHost* h = (BpeHost*)p;
    BpeTable T{h->h_keys.data(), h->h_vals.data(), h->mask};
    std::vector<uint32_t> t(len);
    for (uint32_t i = 0; i < len; ++i) t[i] = h->h_byte_to_tok[bytes[i]];
    int n = bpe_merge_loop(T, t.data(), (int)len);
    uint32_t out_n = (uint32_t)n < cap ? (uint32_t)n : cap;
    for (uint32_t i = 0; i < out_n; ++i) out_tokens[i] = t[i];
    return out_n;
}

// ---------------------------------------------------------------------------
// GPT-2-style pre-tokenization (host C++): approximates the regex
//   's|'t|'re|'ve|'m|'ll|'d| ?\p{L}+| ?\p{N}+| ?[^\s\p{L}\p{N}]+|\s+(?!\S)|\s+
// over UTF-8 (non-ASCII treated as letters).  Emits piece byte offsets.
// ---------------------------------------------------------------------------
static inline bool is_letter(uint8_t c) {
    return (c >= 'a' && c <= 'z') || (c >= 'A' && c <= 'Z') || c >= 0x80;
}
static inline bool is_digit(uint8_t c) { return c >= '0' && c <= '9'; }
static inline bool is_space(uint8_t c) { return c == ' ' || c == '\t' || c == '\n' || c == '\r' || c == '\f' || c == '\v'; }

static inline uint32_t run_end(const uint8_t* s, uint32_t n, uint32_t j) {
    // consume one homogeneous run starting at j (letters | digits | punct)
    if (j >= n) return j;
    if (is_letter(s[j])) {
        while (j < n && is_letter(s[j])) j++;
    } else if (is_digit(s[j])) {
        while (j < n && is_digit(s[j])) j++;
    } else {
        while (j < n && !is_space(s[j]) && !is_letter(s[j]) && !is_digit(s[j])) j++;
    }
    return j;
}

extern "C" uint32_t smg_bpe_pretokenize(const uint8_t* s, uint32_t n, uint32_t* piece_offsets,
                                        uint32_t max_pieces) {
    uint32_t np = 0;
    uint32_t i = 0;
    auto push = [&](uint32_t end) {
        if (np < max_pieces) piece_offsets[np + 1] = end;
        np++;
        return end;
    };
    piece_offsets[0] = 0;
    while (i < n && np < max_pieces) {
        uint8_t c = s[i];
        // contraction suffixes 's 't 'm 'd 're 've 'll
        if (c == '\'' && i + 1 < n) {
            uint8_t c1 = s[i + 1] | 0x20;
            if (c1 == 's' || c1 == 't' || c1 == 'm' || c1 == 'd') { i = push(i + 2); continue; }
            if (i + 2 < n) {
                uint8_t c2 = s[i + 2] | 0x20;
                if ((c1 == 'r' && c2 == 'e') || (c1 == 'v' && c2 == 'e') || (c1 == 'l' && c2 == 'l')) {
                    i = push(i + 3);
                    continue;
                }
            }
        }
        if (is_space(c)) {
            uint32_t j = i;
            while (j < n && is_space(s[j])) j++;
            bool next_is_run = j < n && s[j] != '\'';
            if (next_is_run && s[j - 1] == ' ') {
                if (j - 1 > i) {
                    i = push(j - 1);  // whitespace run minus the space that joins the next piece
                } else {
                    i = push(run_end(s, n, j));  // " word": single space + run
                }
            } else {
                i = push(j);  // pure whitespace piece (or space before contraction)
            }
            continue;
        }
        i = push(run_end(s, n, i));
    }
    if (i < n && np < max_pieces) {
        piece_offsets[np + 1] = n;
        np++;
    }
    return np > max_pieces ? max_pieces : np;
}
