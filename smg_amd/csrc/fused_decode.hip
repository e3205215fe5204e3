// Fused decode-step elementwise kernels — gfx950.
//
// The graphed decode loop spends ~20% of its device time in the elementwise
// "soup" between GEMMs (rope complex-mul chains, KV index_put scatter, the
// q-contiguous pack, silu+mul).  Two kernels remove ~10 launches per layer:
//
//   smg_rope_kv_store: consumes the fused QKV GEMM output [S, 3*D] directly —
//     applies the interleaved-pair rotary rotation to q and k (reading
//     cos/sin from the engine's complex64 freqs table, which is (re,im)
//     float pairs in memory), stores rotated k and raw v into the KV arena
//     at each slot's write position, and emits q packed [S, H, hd] for the
//     decode-attention kernel.  Replaces: 2x rope chains (float cast,
//     complex view/mul, real view, dtype cast), 2x index_put, freqs gather,
//     and the q .contiguous() copy.
//
//   smg_silu_mul: silu(g) * u from the fused gate+up GEMM output [N, 2F]
//     (one kernel instead of silu + mul; the producing GEMM stays hipBLASLt).
//
// Layer layout contract (TorchEngine):
//   qkv:     [S, 3*D] bf16, D = n_heads * head_dim (q | k | v blocks)
//   freqs:   complex64 [max_seq, head_dim/2] -> float2 (cos, sin)
//   pos:     [S] int32 — the write position of each slot this step
//   k,v:     [S, n_heads, max_seq, head_dim] bf16 (one layer's arena slice)
//   q_out:   [S, n_heads, head_dim] bf16
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdint>

#define WAVE 64

// KV8: store the cache as OCP e4m3 fp8 (native v_cvt_pk_fp8_f32 pack of each
// rotated pair) — the decode-attention kernel then streams half the bytes.
__device__ __forceinline__ void store_pair_fp8(unsigned char* dst, float a, float b) {
    unsigned int packed = 0;
    packed = __builtin_amdgcn_cvt_pk_fp8_f32(a, b, packed, false);
    *(unsigned short*)dst = (unsigned short)(packed & 0xFFFFu);
}

// GQA: qkv rows are [q(D) | k(KD) | v(KD)] with D = n_heads*hd and
// KD = n_kv_heads*hd.  One wave per (slot, q head); the wave whose head id is
// < n_kv_heads also ropes+stores that KV head's k/v (q heads kvh*G..kvh*G+G-1
// share KV head kvh — repeat_interleave convention, matched by attn_decode).
template <bool KV8>
__global__ void __launch_bounds__(WAVE) smg_rope_kv_store_t(
    const __hip_bfloat16* __restrict__ qkv,
    const float2* __restrict__ freqs,
    const int* __restrict__ pos,
    void* __restrict__ k_cache,
    void* __restrict__ v_cache,
    __hip_bfloat16* __restrict__ q_out,
    int n_slots, int n_heads, int n_kv_heads, int max_seq, int head_dim) {
    const int sh = blockIdx.x;
    const int slot = sh / n_heads;
    const int head = sh % n_heads;
    if (slot >= n_slots) return;
    const int lane = threadIdx.x;
    const int p = pos[slot];
    const int pairs = head_dim >> 1;
    const int D = n_heads * head_dim;
    const int KD = n_kv_heads * head_dim;
    const bool do_kv = head < n_kv_heads;

    const __hip_bfloat16* qrow = qkv + (size_t)slot * (D + 2 * KD) + (size_t)head * head_dim;
    const __hip_bfloat16* krow = qkv + (size_t)slot * (D + 2 * KD) + D + (size_t)head * head_dim;
    const __hip_bfloat16* vrow = krow + KD;
    const float2* f = freqs + (size_t)p * pairs;

    const size_t row_off = ((size_t)slot * n_kv_heads + head) * max_seq + p;
    __hip_bfloat16* kdst = (__hip_bfloat16*)k_cache + row_off * head_dim;
    __hip_bfloat16* vdst = (__hip_bfloat16*)v_cache + row_off * head_dim;
    unsigned char* kdst8 = (unsigned char*)k_cache + row_off * head_dim;
    unsigned char* vdst8 = (unsigned char*)v_cache + row_off * head_dim;
    __hip_bfloat16* qdst = q_out + ((size_t)slot * n_heads + head) * head_dim;

    for (int i = lane; i < pairs; i += WAVE) {
        const float2 cs = f[i];  // (cos, sin)
        const int e = 2 * i;
        const float q0 = __bfloat162float(qrow[e]);
        const float q1 = __bfloat162float(qrow[e + 1]);
        qdst[e] = __float2bfloat16(q0 * cs.x - q1 * cs.y);
        qdst[e + 1] = __float2bfloat16(q0 * cs.y + q1 * cs.x);
        if (!do_kv) continue;
        const float k0 = __bfloat162float(krow[e]);
        const float k1 = __bfloat162float(krow[e + 1]);
        const float kr0 = k0 * cs.x - k1 * cs.y;
        const float kr1 = k0 * cs.y + k1 * cs.x;
        const float v0 = __bfloat162float(vrow[e]);
        const float v1 = __bfloat162float(vrow[e + 1]);
        if constexpr (KV8) {
            store_pair_fp8(kdst8 + e, kr0, kr1);
            store_pair_fp8(vdst8 + e, v0, v1);
        } else {
            kdst[e] = __float2bfloat16(kr0);
            kdst[e + 1] = __float2bfloat16(kr1);
            vdst[e] = vrow[e];
            vdst[e + 1] = vrow[e + 1];
        }
    }
}

extern "C" __global__ void __launch_bounds__(256) smg_silu_mul(
    const __hip_bfloat16* __restrict__ gu,
    __hip_bfloat16* __restrict__ out,
    long long rows, long long inner) {
    const long long total = rows * inner;
    for (long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
         idx += (long long)gridDim.x * blockDim.x) {
        const long long row = idx / inner;
        const long long col = idx - row * inner;
        const float g = __bfloat162float(gu[row * 2 * inner + col]);
        const float u = __bfloat162float(gu[row * 2 * inner + inner + col]);
        const float s = g / (1.0f + __expf(-g));  // silu
        out[idx] = __float2bfloat16(s * u);
    }
}

// Prefill variant: one workgroup-wave per (request, position, head).  Applies
// the rotary rotation at absolute position start[b]+t, scatters K/V into each
// request's slot window in the KV arena, and emits q/k/v in the [B, H, L, hd]
// layout sdpa wants (so the flash prefill consumes them with zero transpose
// copies).  Replaces the per-layer torch chain: 2x rope complex-mul chains,
// 2x advanced-index KV scatter, and the implicit transpose-contiguous copies.
template <bool KV8>
__global__ void __launch_bounds__(WAVE) smg_rope_prefill_t(
    const __hip_bfloat16* __restrict__ qkv,   // [B, L, D + 2*KD]
    const float2* __restrict__ freqs,         // [max_seq, hd/2]
    const int* __restrict__ slots,            // [B]
    const int* __restrict__ starts,           // [B]
    void* __restrict__ k_cache,               // [n_slots, KVH, max_seq, hd]
    void* __restrict__ v_cache,
    __hip_bfloat16* __restrict__ q_out,       // [B, H, L, hd] (always bf16)
    __hip_bfloat16* __restrict__ k_out,       // [B, KVH, L, hd]
    __hip_bfloat16* __restrict__ v_out,       // [B, KVH, L, hd]
    int B, int L, int n_heads, int n_kv_heads, int max_seq, int head_dim) {
    const int idx = blockIdx.x;
    const int h = idx % n_heads;
    const int bt = idx / n_heads;
    const int t = bt % L;
    const int b = bt / L;
    if (b >= B) return;
    const int lane = threadIdx.x;
    const int pairs = head_dim >> 1;
    const int D = n_heads * head_dim;
    const int KD = n_kv_heads * head_dim;
    const bool do_kv = h < n_kv_heads;
    const int p = starts[b] + t;
    const int slot = slots[b];

    const __hip_bfloat16* qrow = qkv + ((size_t)b * L + t) * (D + 2 * KD) + (size_t)h * head_dim;
    const __hip_bfloat16* krow = qkv + ((size_t)b * L + t) * (D + 2 * KD) + D + (size_t)h * head_dim;
    const __hip_bfloat16* vrow = krow + KD;
    const float2* f = freqs + (size_t)p * pairs;

    const size_t crow = ((size_t)slot * n_kv_heads + h) * max_seq + p;
    __hip_bfloat16* kc = (__hip_bfloat16*)k_cache + crow * head_dim;
    __hip_bfloat16* vc = (__hip_bfloat16*)v_cache + crow * head_dim;
    unsigned char* kc8 = (unsigned char*)k_cache + crow * head_dim;
    unsigned char* vc8 = (unsigned char*)v_cache + crow * head_dim;
    __hip_bfloat16* qo = q_out + (((size_t)b * n_heads + h) * L + t) * head_dim;
    const size_t kv_off = (((size_t)b * n_kv_heads + h) * L + t) * head_dim;
    __hip_bfloat16* ko = k_out + kv_off;
    __hip_bfloat16* vo = v_out + kv_off;

    for (int i = lane; i < pairs; i += WAVE) {
        const float2 cs = f[i];
        const int e = 2 * i;
        const float q0 = __bfloat162float(qrow[e]);
        const float q1 = __bfloat162float(qrow[e + 1]);
        qo[e] = __float2bfloat16(q0 * cs.x - q1 * cs.y);
        qo[e + 1] = __float2bfloat16(q0 * cs.y + q1 * cs.x);
        if (!do_kv) continue;
        const float k0 = __bfloat162float(krow[e]);
        const float k1 = __bfloat162float(krow[e + 1]);
        const float krf0 = k0 * cs.x - k1 * cs.y;
        const float krf1 = k0 * cs.y + k1 * cs.x;
        const __hip_bfloat16 kr0 = __float2bfloat16(krf0);
        const __hip_bfloat16 kr1 = __float2bfloat16(krf1);
        ko[e] = kr0;
        ko[e + 1] = kr1;
        const __hip_bfloat16 v0 = vrow[e];
        const __hip_bfloat16 v1 = vrow[e + 1];
        vo[e] = v0;
        vo[e + 1] = v1;
        if constexpr (KV8) {
            store_pair_fp8(kc8 + e, krf0, krf1);
            store_pair_fp8(vc8 + e, __bfloat162float(v0), __bfloat162float(v1));
        } else {
            kc[e] = kr0;
            kc[e + 1] = kr1;
            vc[e] = v0;
            vc[e + 1] = v1;
        }
    }
}

extern "C" int smg_rope_prefill_launch_gqa(
    const void* qkv, const void* freqs, const void* slots, const void* starts,
    void* k_cache, void* v_cache, void* q_out, void* k_out, void* v_out,
    int B, int L, int n_heads, int n_kv_heads, int max_seq, int head_dim, void* stream,
    int kv_fp8) {
    if (head_dim % 2 != 0 || head_dim > 256) return 1;
    if (n_kv_heads <= 0 || n_heads % n_kv_heads) return 2;
    dim3 grid((unsigned)B * L * n_heads);
    if (kv_fp8) {
        hipLaunchKernelGGL(smg_rope_prefill_t<true>, grid, dim3(WAVE), 0, (hipStream_t)stream,
                           (const __hip_bfloat16*)qkv, (const float2*)freqs, (const int*)slots,
                           (const int*)starts, k_cache, v_cache, (__hip_bfloat16*)q_out,
                           (__hip_bfloat16*)k_out, (__hip_bfloat16*)v_out,
                           B, L, n_heads, n_kv_heads, max_seq, head_dim);
    } else {
        hipLaunchKernelGGL(smg_rope_prefill_t<false>, grid, dim3(WAVE), 0, (hipStream_t)stream,
                           (const __hip_bfloat16*)qkv, (const float2*)freqs, (const int*)slots,
                           (const int*)starts, k_cache, v_cache, (__hip_bfloat16*)q_out,
                           (__hip_bfloat16*)k_out, (__hip_bfloat16*)v_out,
                           B, L, n_heads, n_kv_heads, max_seq, head_dim);
    }
    return (int)hipGetLastError();
}

extern "C" int smg_rope_prefill_launch_ex(
    const void* qkv, const void* freqs, const void* slots, const void* starts,
    void* k_cache, void* v_cache, void* q_out, void* k_out, void* v_out,
    int B, int L, int n_heads, int max_seq, int head_dim, void* stream, int kv_fp8) {
    return smg_rope_prefill_launch_gqa(qkv, freqs, slots, starts, k_cache, v_cache, q_out,
                                       k_out, v_out, B, L, n_heads, n_heads, max_seq, head_dim,
                                       stream, kv_fp8);
}

extern "C" int smg_rope_prefill_launch(
    const void* qkv, const void* freqs, const void* slots, const void* starts,
    void* k_cache, void* v_cache, void* q_out, void* k_out, void* v_out,
    int B, int L, int n_heads, int max_seq, int head_dim, void* stream) {
    return smg_rope_prefill_launch_ex(qkv, freqs, slots, starts, k_cache, v_cache, q_out,
                                      k_out, v_out, B, L, n_heads, max_seq, head_dim, stream, 0);
}

extern "C" int smg_rope_kv_store_launch_gqa(
    const void* qkv, const void* freqs, const void* pos,
    void* k_cache, void* v_cache, void* q_out,
    int n_slots, int n_heads, int n_kv_heads, int max_seq, int head_dim, void* stream,
    int kv_fp8) {
    if (head_dim % 2 != 0 || head_dim > 256) return 1;
    if (n_kv_heads <= 0 || n_heads % n_kv_heads) return 2;
    dim3 grid(n_slots * n_heads);
    if (kv_fp8) {
        hipLaunchKernelGGL(smg_rope_kv_store_t<true>, grid, dim3(WAVE), 0, (hipStream_t)stream,
                           (const __hip_bfloat16*)qkv, (const float2*)freqs, (const int*)pos,
                           k_cache, v_cache, (__hip_bfloat16*)q_out,
                           n_slots, n_heads, n_kv_heads, max_seq, head_dim);
    } else {
        hipLaunchKernelGGL(smg_rope_kv_store_t<false>, grid, dim3(WAVE), 0, (hipStream_t)stream,
                           (const __hip_bfloat16*)qkv, (const float2*)freqs, (const int*)pos,
                           k_cache, v_cache, (__hip_bfloat16*)q_out,
                           n_slots, n_heads, n_kv_heads, max_seq, head_dim);
    }
    return (int)hipGetLastError();
}

extern "C" int smg_rope_kv_store_launch_ex(
    const void* qkv, const void* freqs, const void* pos,
    void* k_cache, void* v_cache, void* q_out,
    int n_slots, int n_heads, int max_seq, int head_dim, void* stream, int kv_fp8) {
    return smg_rope_kv_store_launch_gqa(qkv, freqs, pos, k_cache, v_cache, q_out, n_slots,
                                        n_heads, n_heads, max_seq, head_dim, stream, kv_fp8);
}

extern "C" int smg_rope_kv_store_launch(
    const void* qkv, const void* freqs, const void* pos,
    void* k_cache, void* v_cache, void* q_out,
    int n_slots, int n_heads, int max_seq, int head_dim, void* stream) {
    return smg_rope_kv_store_launch_ex(qkv, freqs, pos, k_cache, v_cache, q_out,
                                       n_slots, n_heads, max_seq, head_dim, stream, 0);
}

// Suffix-prefill flash-LSE merge: o = o1*exp(lse1-lse) + o2*exp(lse2-lse),
// lse = logaddexp(lse1, lse2) — one kernel instead of the ~6 torch
// elementwise launches per layer the two-pass flash combination costs.
extern "C" __global__ void __launch_bounds__(WAVE) smg_lse_merge(
    const __hip_bfloat16* __restrict__ o1,
    const __hip_bfloat16* __restrict__ o2,
    const float* __restrict__ lse1,
    const float* __restrict__ lse2,
    __hip_bfloat16* __restrict__ out,
    long long rows, int head_dim) {
    const long long r = blockIdx.x;
    if (r >= rows) return;
    const float a = lse1[r];
    const float b = lse2[r];
    const float m = fmaxf(a, b);
    const float lse = m + __logf(__expf(a - m) + __expf(b - m));
    const float w1 = __expf(a - lse);
    const float w2 = __expf(b - lse);
    const __hip_bfloat16* r1 = o1 + r * head_dim;
    const __hip_bfloat16* r2 = o2 + r * head_dim;
    __hip_bfloat16* ro = out + r * head_dim;
    for (int i = threadIdx.x; i < head_dim; i += WAVE) {
        ro[i] = __float2bfloat16(w1 * __bfloat162float(r1[i]) + w2 * __bfloat162float(r2[i]));
    }
}

extern "C" int smg_lse_merge_launch(const void* o1, const void* o2, const void* lse1,
                                    const void* lse2, void* out, long long rows, int head_dim,
                                    void* stream) {
    if (rows < 1) return 0;
    hipLaunchKernelGGL(smg_lse_merge, dim3((unsigned)rows), dim3(WAVE), 0, (hipStream_t)stream,
                       (const __hip_bfloat16*)o1, (const __hip_bfloat16*)o2, (const float*)lse1,
                       (const float*)lse2, (__hip_bfloat16*)out, rows, head_dim);
    return (int)hipGetLastError();
}

extern "C" int smg_silu_mul_launch(const void* gu, void* out, long long rows,
                                   long long inner, void* stream) {
    long long total = rows * inner;
    int blocks = (int)((total + 2047) / 2048);  // 8 elems/thread target
    if (blocks < 1) blocks = 1;
    if (blocks > 65535) blocks = 65535;
    hipLaunchKernelGGL(smg_silu_mul, dim3(blocks), dim3(256), 0, (hipStream_t)stream,
                       (const __hip_bfloat16*)gu, (__hip_bfloat16*)out, rows, inner);
    return (int)hipGetLastError();
}
