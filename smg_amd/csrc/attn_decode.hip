// Fused single-token (decode) attention for the GPU worker engine — gfx950.
//
// One wave per (slot, head): online-softmax over the slot's live KV window
// (kv[slot][:pos+1]); idle/short slots cost nothing, unlike sdpa's
// rectangular [slots, maxlen] window.
//
// v5 layout notes (measured on MI355X):
//   * K dot phase: each lane streams ITS OWN timestep's K row with dwordx4
//     loads — a row is 256 B = 4 consecutive cache lines, so per-lane
//     streaming is line-efficient, and 64 lanes x 8 wide loads keep >500
//     lines in flight with no LDS round-trip or barrier;
//   * P·V phase: probabilities via LDS broadcast, V rows loaded 4 B/lane
//     coalesced, no branch between shfl and load so loads pipeline;
//   * zero LDS tiles -> no occupancy cap from shared memory.
//
// Layout contract (the engine's KV arena, one layer):
//   K, V:  [n_slots, n_heads, max_seq, head_dim]  bf16, contiguous
//   q:     [n_slots, n_heads, head_dim]           bf16, contiguous
//   pos:   [n_slots] int32 — attend kpos <= pos[slot]
//   out:   [n_slots, n_heads, head_dim]           bf16
// head_dim <= 128 and divisible by 8.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdint>

#define WAVE 64

__device__ __forceinline__ float bf16_to_f32(unsigned short u) {
    union {
        unsigned int i;
        float f;
    } c;
    c.i = ((unsigned int)u) << 16;
    return c.f;
}

__device__ __forceinline__ void fp8x4_to_f32(unsigned int w, float* o) {
    // byte-select must be a literal for v_cvt_f32_fp8
    o[0] = __builtin_amdgcn_cvt_f32_fp8(w, 0);
    o[1] = __builtin_amdgcn_cvt_f32_fp8(w, 1);
    o[2] = __builtin_amdgcn_cvt_f32_fp8(w, 2);
    o[3] = __builtin_amdgcn_cvt_f32_fp8(w, 3);
}

// fp8 (OCP e4m3) KV-cache variant support: gfx950 has native fp8<->f32
// converts (v_cvt_f32_fp8); the cache is stored quantized by the fused
// rope/KV-store kernels and dequantized inline here — halving the HBM bytes
// the decode-attention streams per step.
template <bool KV8>
__global__ void __launch_bounds__(WAVE) smg_attn_decode_t(
    const __hip_bfloat16* __restrict__ q,
    const void* __restrict__ k,
    const void* __restrict__ v,
    const int* __restrict__ pos,
    __hip_bfloat16* __restrict__ out,
    int n_slots, int n_heads, int max_seq, int head_dim, float scale) {
    int sh = blockIdx.x;
    int slot = sh / n_heads;
    int head = sh % n_heads;
    if (slot >= n_slots) return;
    int lane = threadIdx.x;
    int T = pos[slot] + 1;  // inclusive current position
    if (T > max_seq) T = max_seq;

    const size_t head_base = ((size_t)slot * n_heads + head) * (size_t)max_seq * head_dim;
    const __hip_bfloat16* kh = (const __hip_bfloat16*)k + head_base;
    const __hip_bfloat16* vh = (const __hip_bfloat16*)v + head_base;
    const unsigned char* kh8 = (const unsigned char*)k + head_base;
    const unsigned char* vh8 = (const unsigned char*)v + head_base;
    const __hip_bfloat16* qh = q + ((size_t)slot * n_heads + head) * head_dim;

    // q staged once in LDS as f32 (read by every lane's dot)
    __shared__ float s_q[128];
    __shared__ float s_p[WAVE];
    for (int i = lane; i < head_dim; i += WAVE) s_q[i] = (float)qh[i];
    __syncthreads();

    const int vec_n = head_dim / 8;  // dwordx4 chunks per row (<=16)
    // v6 P·V mapping: lane -> (row-group rgrp = lane/chunks, chunk = lane%chunks)
    // so one wave instruction loads dwordx4 from `rows_per` different V rows —
    // 1 KB per instruction (vs 256 B with the old 4 B/lane scheme), matching
    // the K phase's streaming width.  Each lane accumulates 8 dims of its
    // chunk over rows rgrp, rgrp+rows_per, ...; a log2(rows_per)-step
    // shfl_xor tree folds the row groups at the end.
    const int chunks = vec_n;             // dwordx4 chunks per row (8 or 16)
    const int rows_per = WAVE / chunks;   // rows covered per instruction
    const int chunk = lane % chunks;
    const int rgrp = lane / chunks;
    float accv[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) accv[j] = 0.f;
    float m = -1e30f, l = 0.f;

    for (int base = 0; base < T; base += WAVE) {
        int lim = min(WAVE, T - base);
        int t = base + lane;
        float score = -1e30f;
        if (t < T) {
            float d = 0.f;
            if constexpr (KV8) {
                const uint4* row = (const uint4*)(kh8 + (size_t)t * head_dim);
#pragma unroll 4
                for (int c = 0; c < vec_n / 2; ++c) {
                    uint4 w = row[c];  // 16 fp8
                    const unsigned int* wp = (const unsigned int*)&w;
                    int ib = c * 16;
                    float dec[4];
#pragma unroll
                    for (int wi = 0; wi < 4; ++wi) {
                        fp8x4_to_f32(wp[wi], dec);
#pragma unroll
                        for (int s = 0; s < 4; ++s) d += s_q[ib + wi * 4 + s] * dec[s];
                    }
                }
            } else {
                const uint4* row = (const uint4*)(kh + (size_t)t * head_dim);
#pragma unroll 4
                for (int c = 0; c < vec_n; ++c) {
                    uint4 w = row[c];  // 8 bf16
                    const unsigned short* hsp = (const unsigned short*)&w;
                    int ib = c * 8;
#pragma unroll
                    for (int j = 0; j < 8; ++j) d += s_q[ib + j] * bf16_to_f32(hsp[j]);
                }
            }
            score = d * scale;
        }
        // online softmax across the wave's scores
        float mr = score;
#pragma unroll
        for (int off = 32; off > 0; off >>= 1) mr = fmaxf(mr, __shfl_xor(mr, off, WAVE));
        float m_new = fmaxf(m, mr);
        float alpha = __expf(m - m_new);
        float p = (t < T) ? __expf(score - m_new) : 0.f;
        s_p[lane] = p;
        float pr = p;
#pragma unroll
        for (int off = 32; off > 0; off >>= 1) pr += __shfl_xor(pr, off, WAVE);
        l = l * alpha + pr;
#pragma unroll
        for (int j = 0; j < 8; ++j) accv[j] *= alpha;
        __syncthreads();
        // P·V: each lane streams dwordx4 of its chunk from every rows_per-th
        // row.  Full tiles take the unrolled fast path (compile-time trip
        // count exposes all 16 loads to the scheduler); only the final
        // partial tile pays the runtime loop.
        if constexpr (KV8) {
            const unsigned char* vtile8 = vh8 + (size_t)base * head_dim + chunk * 8;
            if (lim == WAVE && head_dim == 128) {
#pragma unroll
                for (int it = 0; it < 16; ++it) {
                    const int j = rgrp + it * 4;
                    const uint2 w = *(const uint2*)(vtile8 + (size_t)j * head_dim);
                    float pj = s_p[j];
                    float dec[4];
                    fp8x4_to_f32(w.x, dec);
#pragma unroll
                    for (int s = 0; s < 4; ++s) accv[s] += pj * dec[s];
                    fp8x4_to_f32(w.y, dec);
#pragma unroll
                    for (int s = 0; s < 4; ++s) accv[4 + s] += pj * dec[s];
                }
            } else {
                for (int j = rgrp; j < lim; j += rows_per) {
                    const uint2 w = *(const uint2*)(vtile8 + (size_t)j * head_dim);
                    float pj = s_p[j];
                    float dec[4];
                    fp8x4_to_f32(w.x, dec);
#pragma unroll
                    for (int s = 0; s < 4; ++s) accv[s] += pj * dec[s];
                    fp8x4_to_f32(w.y, dec);
#pragma unroll
                    for (int s = 0; s < 4; ++s) accv[4 + s] += pj * dec[s];
                }
            }
        } else {
            const __hip_bfloat16* vtile = vh + (size_t)base * head_dim + chunk * 8;
            if (lim == WAVE && head_dim == 128) {
#pragma unroll
                for (int it = 0; it < 16; ++it) {
                    const int j = rgrp + it * 4;  // rows_per == 4 when chunks == 16
                    const uint4 w = *(const uint4*)(vtile + (size_t)j * head_dim);
                    const unsigned short* hsp = (const unsigned short*)&w;
                    float pj = s_p[j];
#pragma unroll
                    for (int jj = 0; jj < 8; ++jj) accv[jj] += pj * bf16_to_f32(hsp[jj]);
                }
            } else {
                for (int j = rgrp; j < lim; j += rows_per) {
                    const uint4 w = *(const uint4*)(vtile + (size_t)j * head_dim);
                    const unsigned short* hsp = (const unsigned short*)&w;
                    float pj = s_p[j];
#pragma unroll
                    for (int jj = 0; jj < 8; ++jj) accv[jj] += pj * bf16_to_f32(hsp[jj]);
                }
            }
        }
        __syncthreads();
        m = m_new;
    }
    // fold the row groups: lanes sharing `chunk` differ in bits >= log2(chunks)
    for (int off = chunks; off < WAVE; off <<= 1) {
#pragma unroll
        for (int jj = 0; jj < 8; ++jj) accv[jj] += __shfl_xor(accv[jj], off, WAVE);
    }
    float inv = l > 0.f ? 1.f / l : 0.f;
    __hip_bfloat16* orow = out + ((size_t)slot * n_heads + head) * head_dim;
    if (rgrp == 0) {
#pragma unroll
        for (int jj = 0; jj < 8; ++jj) orow[chunk * 8 + jj] = (__hip_bfloat16)(accv[jj] * inv);
    }
}

extern "C" int smg_attn_decode_launch_ex(const void* q, const void* k, const void* v,
                                         const void* pos, void* out, int n_slots, int n_heads,
                                         int max_seq, int head_dim, float scale, void* stream,
                                         int kv_fp8) {
    if (head_dim > 128 || (head_dim & 7)) return -1;
    int chunks = head_dim / 8;  // v6 P·V row-group mapping needs 2^k chunks
    if (chunks & (chunks - 1)) return -1;
    if (kv_fp8 && (head_dim & 15)) return -1;  // fp8 K streaming is 16-wide
    dim3 grid(n_slots * n_heads);
    if (kv_fp8) {
        hipLaunchKernelGGL(smg_attn_decode_t<true>, grid, dim3(WAVE), 0, (hipStream_t)stream,
                           (const __hip_bfloat16*)q, k, v, (const int*)pos,
                           (__hip_bfloat16*)out, n_slots, n_heads, max_seq, head_dim, scale);
    } else {
        hipLaunchKernelGGL(smg_attn_decode_t<false>, grid, dim3(WAVE), 0, (hipStream_t)stream,
                           (const __hip_bfloat16*)q, k, v, (const int*)pos,
                           (__hip_bfloat16*)out, n_slots, n_heads, max_seq, head_dim, scale);
    }
    return hipGetLastError() == hipSuccess ? 0 : -2;
}

extern "C" int smg_attn_decode_launch(const void* q, const void* k, const void* v,
                                      const void* pos, void* out, int n_slots, int n_heads,
                                      int max_seq, int head_dim, float scale, void* stream) {
    return smg_attn_decode_launch_ex(q, k, v, pos, out, n_slots, n_heads, max_seq, head_dim,
                                     scale, stream, 0);
}
