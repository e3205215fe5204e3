// Fused single-token (decode) attention for the GPU worker engine — gfx950.
//
// One wave per (slot, KV head): online-softmax over the slot's live KV window
// (kv[slot][:pos+1]); idle/short slots cost nothing, unlike sdpa's
// rectangular [slots, maxlen] window.
//
// v7 adds GQA: the wave carries all G = n_heads/n_kv_heads query heads of the
// group, so each K/V row is streamed from HBM ONCE and dotted/FMAed against G
// q vectors — decode attention is bandwidth-bound (~80% of achievable HBM BW
// at MHA), so GQA cuts its device time by ~G.  G in {1,2,4,8} is a template
// parameter (full unrolling); G=1 reproduces the v6 MHA kernel exactly.
//
// v5/v6 layout notes (measured on MI355X):
//   * K dot phase: each lane streams ITS OWN timestep's K row with dwordx4
//     loads — a row is 256 B = 4 consecutive cache lines, so per-lane
//     streaming is line-efficient, and 64 lanes x 8 wide loads keep >500
//     lines in flight with no LDS round-trip or barrier;
//   * P·V phase: probabilities via LDS broadcast, V rows loaded dwordx4 with
//     a lane->(row-group, chunk) mapping so one wave instruction covers 4
//     rows (1 KB); unrolled full-tile path; shfl_xor row-group fold;
//   * q staged in LDS (wave-uniform broadcast reads), no LDS K/V tiles ->
//     no occupancy cap from shared memory.
//
// Layout contract (the engine's KV arena, one layer):
//   K, V:  [n_slots, n_kv_heads, max_seq, head_dim]  bf16 or fp8-e4m3
//   q:     [n_slots, n_heads, head_dim]              bf16, contiguous
//   pos:   [n_slots] int32 — attend kpos <= pos[slot]
//   out:   [n_slots, n_heads, head_dim]              bf16
// head_dim <= 128, divisible by 8, power-of-two chunk count; q heads
// kvh*G .. kvh*G+G-1 share KV head kvh (repeat_interleave convention).
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

typedef float __f32x2 __attribute__((__vector_size__(8)));

__device__ __forceinline__ void fp8x4_to_f32(unsigned int w, float* o) {
    // packed converts: v_cvt_pk_f32_fp8 yields 2 floats per instruction
    // (word-select low=bytes 0-1, high=bytes 2-3) — half the convert ops of
    // per-byte v_cvt_f32_fp8, bit-identical results (fp8->f32 is exact)
    __f32x2 lo = __builtin_amdgcn_cvt_pk_f32_fp8(w, false);
    __f32x2 hi = __builtin_amdgcn_cvt_pk_f32_fp8(w, true);
    o[0] = lo[0];
    o[1] = lo[1];
    o[2] = hi[0];
    o[3] = hi[1];
}

// fp8 (OCP e4m3) KV-cache variant support: gfx950 has native fp8<->f32
// converts (v_cvt_f32_fp8); the cache is stored quantized by the fused
// rope/KV-store kernels and dequantized inline here — halving the HBM bytes
// the decode-attention streams per step (composes with GQA's 1/G).
template <bool KV8, int G>
__global__ void __launch_bounds__(WAVE) smg_attn_decode_t(
    const __hip_bfloat16* __restrict__ q,
    const void* __restrict__ k,
    const void* __restrict__ v,
    const int* __restrict__ pos,
    __hip_bfloat16* __restrict__ out,
    int n_slots, int n_heads, int n_kv_heads, int max_seq, int head_dim, float scale) {
    int sh = blockIdx.x;
    int slot = sh / n_kv_heads;
    int kvh = sh % n_kv_heads;
    if (slot >= n_slots) return;
    int lane = threadIdx.x;
    int T = pos[slot] + 1;  // inclusive current position
    if (T > max_seq) T = max_seq;

    const size_t head_base = ((size_t)slot * n_kv_heads + kvh) * (size_t)max_seq * head_dim;
    const __hip_bfloat16* kh = (const __hip_bfloat16*)k + head_base;
    const __hip_bfloat16* vh = (const __hip_bfloat16*)v + head_base;
    const unsigned char* kh8 = (const unsigned char*)k + head_base;
    const unsigned char* vh8 = (const unsigned char*)v + head_base;
    // the group's G query rows are contiguous: heads kvh*G .. kvh*G+G-1
    const __hip_bfloat16* qh = q + ((size_t)slot * n_heads + (size_t)kvh * G) * head_dim;

    // q staged once in LDS as f32 (wave-uniform broadcast reads in the dot)
    __shared__ float s_q[G * 128];
    __shared__ float s_p[G][WAVE];
    for (int i = lane; i < G * head_dim; i += WAVE) s_q[i] = (float)qh[i];
    __syncthreads();

    const int vec_n = head_dim / 8;  // dwordx4 chunks per row (<=16)
    const int chunks = vec_n;             // dwordx4 chunks per row (8 or 16)
    const int rows_per = WAVE / chunks;   // rows covered per instruction
    const int chunk = lane % chunks;
    const int rgrp = lane / chunks;
    float accv[G][8];
#pragma unroll
    for (int g = 0; g < G; ++g)
#pragma unroll
        for (int j = 0; j < 8; ++j) accv[g][j] = 0.f;
    float m[G], l[G];
#pragma unroll
    for (int g = 0; g < G; ++g) { m[g] = -1e30f; l[g] = 0.f; }

    for (int base = 0; base < T; base += WAVE) {
        int lim = min(WAVE, T - base);
        int t = base + lane;
        float d[G];
#pragma unroll
        for (int g = 0; g < G; ++g) d[g] = 0.f;
        if (t < T) {
            // stream this lane's K row ONCE; dot against all G q vectors
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
                        for (int s = 0; s < 4; ++s) {
                            const float kv = dec[s];
                            const int di = ib + wi * 4 + s;
#pragma unroll
                            for (int g = 0; g < G; ++g) d[g] += s_q[g * head_dim + di] * kv;
                        }
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
                    for (int j = 0; j < 8; ++j) {
                        const float kv = bf16_to_f32(hsp[j]);
#pragma unroll
                        for (int g = 0; g < G; ++g) d[g] += s_q[g * head_dim + ib + j] * kv;
                    }
                }
            }
        }
        // online softmax per query head across the wave's scores
        float alpha[G];
#pragma unroll
        for (int g = 0; g < G; ++g) {
            float score = (t < T) ? d[g] * scale : -1e30f;
            float mr = score;
#pragma unroll
            for (int off = 32; off > 0; off >>= 1) mr = fmaxf(mr, __shfl_xor(mr, off, WAVE));
            float m_new = fmaxf(m[g], mr);
            alpha[g] = __expf(m[g] - m_new);
            float p = (t < T) ? __expf(score - m_new) : 0.f;
            s_p[g][lane] = p;
            float pr = p;
#pragma unroll
            for (int off = 32; off > 0; off >>= 1) pr += __shfl_xor(pr, off, WAVE);
            l[g] = l[g] * alpha[g] + pr;
            m[g] = m_new;
#pragma unroll
            for (int j = 0; j < 8; ++j) accv[g][j] *= alpha[g];
        }
        __syncthreads();
        // P·V: each lane streams dwordx4 of its chunk from every rows_per-th
        // row — ONE V load feeds all G accumulators.  Full tiles take the
        // unrolled fast path; only the final partial tile pays the runtime loop.
        if constexpr (KV8) {
            const unsigned char* vtile8 = vh8 + (size_t)base * head_dim + chunk * 8;
            if (lim == WAVE && head_dim == 128) {
#pragma unroll
                for (int it = 0; it < 16; ++it) {
                    const int j = rgrp + it * 4;
                    const uint2 w = *(const uint2*)(vtile8 + (size_t)j * head_dim);
                    float dec[8];
                    fp8x4_to_f32(w.x, dec);
                    fp8x4_to_f32(w.y, dec + 4);
#pragma unroll
                    for (int g = 0; g < G; ++g) {
                        const float pj = s_p[g][j];
#pragma unroll
                        for (int s = 0; s < 8; ++s) accv[g][s] += pj * dec[s];
                    }
                }
            } else {
                for (int j = rgrp; j < lim; j += rows_per) {
                    const uint2 w = *(const uint2*)(vtile8 + (size_t)j * head_dim);
                    float dec[8];
                    fp8x4_to_f32(w.x, dec);
                    fp8x4_to_f32(w.y, dec + 4);
#pragma unroll
                    for (int g = 0; g < G; ++g) {
                        const float pj = s_p[g][j];
#pragma unroll
                        for (int s = 0; s < 8; ++s) accv[g][s] += pj * dec[s];
                    }
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
#pragma unroll
                    for (int g = 0; g < G; ++g) {
                        const float pj = s_p[g][j];
#pragma unroll
                        for (int jj = 0; jj < 8; ++jj) accv[g][jj] += pj * bf16_to_f32(hsp[jj]);
                    }
                }
            } else {
                for (int j = rgrp; j < lim; j += rows_per) {
                    const uint4 w = *(const uint4*)(vtile + (size_t)j * head_dim);
                    const unsigned short* hsp = (const unsigned short*)&w;
#pragma unroll
                    for (int g = 0; g < G; ++g) {
                        const float pj = s_p[g][j];
#pragma unroll
                        for (int jj = 0; jj < 8; ++jj) accv[g][jj] += pj * bf16_to_f32(hsp[jj]);
                    }
                }
            }
        }
        __syncthreads();
    }
    // fold the row groups: lanes sharing `chunk` differ in bits >= log2(chunks)
#pragma unroll
    for (int g = 0; g < G; ++g) {
        for (int off = chunks; off < WAVE; off <<= 1) {
#pragma unroll
            for (int jj = 0; jj < 8; ++jj) accv[g][jj] += __shfl_xor(accv[g][jj], off, WAVE);
        }
    }
    if (rgrp == 0) {
#pragma unroll
        for (int g = 0; g < G; ++g) {
            float inv = l[g] > 0.f ? 1.f / l[g] : 0.f;
            __hip_bfloat16* orow =
                out + ((size_t)slot * n_heads + (size_t)kvh * G + g) * head_dim;
#pragma unroll
            for (int jj = 0; jj < 8; ++jj) orow[chunk * 8 + jj] = (__hip_bfloat16)(accv[g][jj] * inv);
        }
    }
}

// ---------------------------------------------------------------------------
// v8: T-split ("flash-decoding") variant.  GQA cut the wave count to
// n_slots * n_kv_heads, leaving the chip under-occupied (measured ~40%
// occupancy, HBM at ~59% of peak vs ~80% for the MHA kernel).  Splitting each
// slot's KV window over n_split waves restores the memory-level parallelism;
// a small merge kernel LSE-combines the partials.  Scratch layout:
//   part_out: [n_slots, n_kv_heads, n_split, G, head_dim] f32 (unnormalized)
//   part_ml:  [n_slots, n_kv_heads, n_split, G, 2] f32 (running max, sum)
// ---------------------------------------------------------------------------
template <bool KV8, int G>
__global__ void __launch_bounds__(WAVE) smg_attn_decode_split_t(
    const __hip_bfloat16* __restrict__ q,
    const void* __restrict__ k,
    const void* __restrict__ v,
    const int* __restrict__ pos,
    float* __restrict__ part_out,
    float* __restrict__ part_ml,
    int n_slots, int n_heads, int n_kv_heads, int n_split, int max_seq, int head_dim,
    float scale) {
    int idx = blockIdx.x;
    int split = idx % n_split;
    int kvh = (idx / n_split) % n_kv_heads;
    int slot = idx / (n_split * n_kv_heads);
    if (slot >= n_slots) return;
    int lane = threadIdx.x;
    int T = pos[slot] + 1;
    if (T > max_seq) T = max_seq;
    // this wave's [t0, t1) range: contiguous WAVE-aligned chunks
    int chunk_len = ((T + n_split - 1) / n_split + WAVE - 1) / WAVE * WAVE;
    int t0 = split * chunk_len;
    int t1 = min(T, t0 + chunk_len);

    const size_t head_base = ((size_t)slot * n_kv_heads + kvh) * (size_t)max_seq * head_dim;
    const __hip_bfloat16* kh = (const __hip_bfloat16*)k + head_base;
    const __hip_bfloat16* vh = (const __hip_bfloat16*)v + head_base;
    const unsigned char* kh8 = (const unsigned char*)k + head_base;
    const unsigned char* vh8 = (const unsigned char*)v + head_base;
    const __hip_bfloat16* qh = q + ((size_t)slot * n_heads + (size_t)kvh * G) * head_dim;

    __shared__ float s_q[G * 128];
    __shared__ float s_p[G][WAVE];
    for (int i = lane; i < G * head_dim; i += WAVE) s_q[i] = (float)qh[i];
    __syncthreads();

    const int vec_n = head_dim / 8;
    const int chunks = vec_n;
    const int rows_per = WAVE / chunks;
    const int chunk = lane % chunks;
    const int rgrp = lane / chunks;
    float accv[G][8];
#pragma unroll
    for (int g = 0; g < G; ++g)
#pragma unroll
        for (int j = 0; j < 8; ++j) accv[g][j] = 0.f;
    float m[G], l[G];
#pragma unroll
    for (int g = 0; g < G; ++g) { m[g] = -1e30f; l[g] = 0.f; }

    for (int base = t0; base < t1; base += WAVE) {
        int lim = min(WAVE, t1 - base);
        int t = base + lane;
        float d[G];
#pragma unroll
        for (int g = 0; g < G; ++g) d[g] = 0.f;
        if (t < t1) {
            if constexpr (KV8) {
                const uint4* row = (const uint4*)(kh8 + (size_t)t * head_dim);
#pragma unroll 4
                for (int c = 0; c < vec_n / 2; ++c) {
                    uint4 w = row[c];
                    const unsigned int* wp = (const unsigned int*)&w;
                    int ib = c * 16;
                    float dec[4];
#pragma unroll
                    for (int wi = 0; wi < 4; ++wi) {
                        fp8x4_to_f32(wp[wi], dec);
#pragma unroll
                        for (int s = 0; s < 4; ++s) {
                            const float kv = dec[s];
                            const int di = ib + wi * 4 + s;
#pragma unroll
                            for (int g = 0; g < G; ++g) d[g] += s_q[g * head_dim + di] * kv;
                        }
                    }
                }
            } else {
                const uint4* row = (const uint4*)(kh + (size_t)t * head_dim);
#pragma unroll 4
                for (int c = 0; c < vec_n; ++c) {
                    uint4 w = row[c];
                    const unsigned short* hsp = (const unsigned short*)&w;
                    int ib = c * 8;
#pragma unroll
                    for (int j = 0; j < 8; ++j) {
                        const float kv = bf16_to_f32(hsp[j]);
#pragma unroll
                        for (int g = 0; g < G; ++g) d[g] += s_q[g * head_dim + ib + j] * kv;
                    }
                }
            }
        }
        float alpha[G];
#pragma unroll
        for (int g = 0; g < G; ++g) {
            float score = (t < t1) ? d[g] * scale : -1e30f;
            float mr = score;
#pragma unroll
            for (int off = 32; off > 0; off >>= 1) mr = fmaxf(mr, __shfl_xor(mr, off, WAVE));
            float m_new = fmaxf(m[g], mr);
            alpha[g] = __expf(m[g] - m_new);
            float p = (t < t1) ? __expf(score - m_new) : 0.f;
            s_p[g][lane] = p;
            float pr = p;
#pragma unroll
            for (int off = 32; off > 0; off >>= 1) pr += __shfl_xor(pr, off, WAVE);
            l[g] = l[g] * alpha[g] + pr;
            m[g] = m_new;
#pragma unroll
            for (int j = 0; j < 8; ++j) accv[g][j] *= alpha[g];
        }
        __syncthreads();
        if constexpr (KV8) {
            const unsigned char* vtile8 = vh8 + (size_t)base * head_dim + chunk * 8;
            for (int j = rgrp; j < lim; j += rows_per) {
                const uint2 w = *(const uint2*)(vtile8 + (size_t)j * head_dim);
                float dec[8];
                fp8x4_to_f32(w.x, dec);
                fp8x4_to_f32(w.y, dec + 4);
#pragma unroll
                for (int g = 0; g < G; ++g) {
                    const float pj = s_p[g][j];
#pragma unroll
                    for (int s = 0; s < 8; ++s) accv[g][s] += pj * dec[s];
                }
            }
        } else {
            const __hip_bfloat16* vtile = vh + (size_t)base * head_dim + chunk * 8;
            if (lim == WAVE && head_dim == 128) {
#pragma unroll
                for (int it = 0; it < 16; ++it) {
                    const int j = rgrp + it * 4;
                    const uint4 w = *(const uint4*)(vtile + (size_t)j * head_dim);
                    const unsigned short* hsp = (const unsigned short*)&w;
#pragma unroll
                    for (int g = 0; g < G; ++g) {
                        const float pj = s_p[g][j];
#pragma unroll
                        for (int jj = 0; jj < 8; ++jj) accv[g][jj] += pj * bf16_to_f32(hsp[jj]);
                    }
                }
            } else {
                for (int j = rgrp; j < lim; j += rows_per) {
                    const uint4 w = *(const uint4*)(vtile + (size_t)j * head_dim);
                    const unsigned short* hsp = (const unsigned short*)&w;
#pragma unroll
                    for (int g = 0; g < G; ++g) {
                        const float pj = s_p[g][j];
#pragma unroll
                        for (int jj = 0; jj < 8; ++jj) accv[g][jj] += pj * bf16_to_f32(hsp[jj]);
                    }
                }
            }
        }
        __syncthreads();
    }
#pragma unroll
    for (int g = 0; g < G; ++g) {
        for (int off = chunks; off < WAVE; off <<= 1) {
#pragma unroll
            for (int jj = 0; jj < 8; ++jj) accv[g][jj] += __shfl_xor(accv[g][jj], off, WAVE);
        }
    }
    // partial write: unnormalized acc (f32) + (m, l) per query head
    const size_t pbase = (((size_t)slot * n_kv_heads + kvh) * n_split + split) * G;
    if (rgrp == 0) {
#pragma unroll
        for (int g = 0; g < G; ++g) {
            float* orow = part_out + (pbase + g) * head_dim;
#pragma unroll
            for (int jj = 0; jj < 8; ++jj) orow[chunk * 8 + jj] = accv[g][jj];
        }
    }
    if (lane == 0) {
#pragma unroll
        for (int g = 0; g < G; ++g) {
            part_ml[(pbase + g) * 2] = m[g];
            part_ml[(pbase + g) * 2 + 1] = l[g];
        }
    }
}

// LSE-merge the n_split partials: one wave per (slot, kv head) folds all G
// query heads (G*head_dim <= 1024 accumulators across the wave's lanes).
template <int G>
__global__ void __launch_bounds__(WAVE) smg_attn_merge_t(
    const float* __restrict__ part_out,
    const float* __restrict__ part_ml,
    __hip_bfloat16* __restrict__ out,
    int n_slots, int n_heads, int n_kv_heads, int n_split, int head_dim) {
    int sh = blockIdx.x;
    int slot = sh / n_kv_heads;
    int kvh = sh % n_kv_heads;
    if (slot >= n_slots) return;
    int lane = threadIdx.x;
    const size_t pbase0 = ((size_t)slot * n_kv_heads + kvh) * n_split * G;
#pragma unroll 1
    for (int g = 0; g < G; ++g) {
        // global max over splits (splits with l==0 never contribute)
        float m_star = -1e30f;
        for (int s = 0; s < n_split; ++s) {
            float lm = part_ml[(pbase0 + (size_t)s * G + g) * 2];
            float ll = part_ml[(pbase0 + (size_t)s * G + g) * 2 + 1];
            if (ll > 0.f) m_star = fmaxf(m_star, lm);
        }
        float l_star = 0.f;
        for (int s = 0; s < n_split; ++s) {
            float lm = part_ml[(pbase0 + (size_t)s * G + g) * 2];
            float ll = part_ml[(pbase0 + (size_t)s * G + g) * 2 + 1];
            if (ll > 0.f) l_star += ll * __expf(lm - m_star);
        }
        float inv = l_star > 0.f ? 1.f / l_star : 0.f;
        for (int i = lane; i < head_dim; i += WAVE) {
            float acc = 0.f;
            for (int s = 0; s < n_split; ++s) {
                float lm = part_ml[(pbase0 + (size_t)s * G + g) * 2];
                float ll = part_ml[(pbase0 + (size_t)s * G + g) * 2 + 1];
                if (ll > 0.f)
                    acc += part_out[(pbase0 + (size_t)s * G + g) * head_dim + i] *
                           __expf(lm - m_star);
            }
            out[((size_t)slot * n_heads + (size_t)kvh * G + g) * head_dim + i] =
                (__hip_bfloat16)(acc * inv);
        }
    }
}

template <bool KV8>
static int launch_split_g(const void* q, const void* k, const void* v, const void* pos,
                          void* out, float* part_out, float* part_ml, int n_slots, int n_heads,
                          int n_kv_heads, int n_split, int max_seq, int head_dim, float scale,
                          hipStream_t stream) {
    dim3 grid((unsigned)n_slots * n_kv_heads * n_split);
    dim3 mgrid((unsigned)n_slots * n_kv_heads);
    const int G = n_heads / n_kv_heads;
#define SPLIT_CASE(GG)                                                                          \
    case GG:                                                                                    \
        hipLaunchKernelGGL((smg_attn_decode_split_t<KV8, GG>), grid, dim3(WAVE), 0, stream,     \
                           (const __hip_bfloat16*)q, k, v, (const int*)pos, part_out, part_ml,  \
                           n_slots, n_heads, n_kv_heads, n_split, max_seq, head_dim, scale);    \
        hipLaunchKernelGGL((smg_attn_merge_t<GG>), mgrid, dim3(WAVE), 0, stream, part_out,      \
                           part_ml, (__hip_bfloat16*)out, n_slots, n_heads, n_kv_heads,         \
                           n_split, head_dim);                                                  \
        break;
    switch (G) {
        SPLIT_CASE(1)
        SPLIT_CASE(2)
        SPLIT_CASE(4)
        SPLIT_CASE(8)
        default:
            return -4;
    }
#undef SPLIT_CASE
    return hipGetLastError() == hipSuccess ? 0 : -2;
}

extern "C" int smg_attn_decode_launch_split(const void* q, const void* k, const void* v,
                                            const void* pos, void* out, void* part_out,
                                            void* part_ml, int n_slots, int n_heads,
                                            int n_kv_heads, int n_split, int max_seq,
                                            int head_dim, float scale, void* stream,
                                            int kv_fp8) {
    if (head_dim > 128 || (head_dim & 7)) return -1;
    int chunks = head_dim / 8;
    if (chunks & (chunks - 1)) return -1;
    if (kv_fp8 && (head_dim & 15)) return -1;
    if (n_kv_heads <= 0 || n_heads % n_kv_heads || n_split < 1) return -4;
    if (kv_fp8)
        return launch_split_g<true>(q, k, v, pos, out, (float*)part_out, (float*)part_ml,
                                    n_slots, n_heads, n_kv_heads, n_split, max_seq, head_dim,
                                    scale, (hipStream_t)stream);
    return launch_split_g<false>(q, k, v, pos, out, (float*)part_out, (float*)part_ml,
                                 n_slots, n_heads, n_kv_heads, n_split, max_seq, head_dim,
                                 scale, (hipStream_t)stream);
}

template <bool KV8>
static int launch_g(const void* q, const void* k, const void* v, const void* pos, void* out,
                    int n_slots, int n_heads, int n_kv_heads, int max_seq, int head_dim,
                    float scale, hipStream_t stream) {
    dim3 grid(n_slots * n_kv_heads);
    const int G = n_heads / n_kv_heads;
#define LAUNCH_CASE(GG)                                                                        \
    case GG:                                                                                   \
        hipLaunchKernelGGL((smg_attn_decode_t<KV8, GG>), grid, dim3(WAVE), 0, stream,          \
                           (const __hip_bfloat16*)q, k, v, (const int*)pos,                    \
                           (__hip_bfloat16*)out, n_slots, n_heads, n_kv_heads, max_seq,        \
                           head_dim, scale);                                                   \
        break;
    switch (G) {
        LAUNCH_CASE(1)
        LAUNCH_CASE(2)
        LAUNCH_CASE(4)
        LAUNCH_CASE(8)
        default:
            return -4;
    }
#undef LAUNCH_CASE
    return hipGetLastError() == hipSuccess ? 0 : -2;
}

extern "C" int smg_attn_decode_launch_gqa(const void* q, const void* k, const void* v,
                                          const void* pos, void* out, int n_slots, int n_heads,
                                          int n_kv_heads, int max_seq, int head_dim, float scale,
                                          void* stream, int kv_fp8) {
    if (head_dim > 128 || (head_dim & 7)) return -1;
    int chunks = head_dim / 8;  // v6 P·V row-group mapping needs 2^k chunks
    if (chunks & (chunks - 1)) return -1;
    if (kv_fp8 && (head_dim & 15)) return -1;  // fp8 K streaming is 16-wide
    if (n_kv_heads <= 0 || n_heads % n_kv_heads) return -4;
    if (kv_fp8)
        return launch_g<true>(q, k, v, pos, out, n_slots, n_heads, n_kv_heads, max_seq, head_dim,
                              scale, (hipStream_t)stream);
    return launch_g<false>(q, k, v, pos, out, n_slots, n_heads, n_kv_heads, max_seq, head_dim,
                           scale, (hipStream_t)stream);
}

// ---------------------------------------------------------------------------
// v9 (bf16 only): packed v_dot2c_f32_bf16 K phase.
// Measured motivation: the v7 kernel runs at 2.8 TB/s (35% of HBM peak) at
// bench shapes and is latency/VALU-bound, not bandwidth-bound — the K dot
// spends 5 VALU ops per bf16 pair (2 converts + 2 FMA + addressing).  Here
// the K dot is one v_dot2c_f32_bf16 per pair against q pre-packed as bf16x2
// in LDS (2.5x fewer K-phase VALU ops).  Also tried and MEASURED WORSE:
// a two-tile (128-timestep) softmax round (256 VGPRs + 56 spilled) and
// prefetching 6 V rows across the softmax phase (242 VGPRs, no spill, but
// 198 -> 205 us — the register pressure costs more scheduling freedom than
// the overlap buys), so rounds stay 64 timesteps with inline V loads.
// ---------------------------------------------------------------------------
typedef short __bf16x2 __attribute__((__vector_size__(2 * sizeof(short))));

__device__ __forceinline__ float dot2_bf16(unsigned int a, unsigned int b, float acc) {
    union {
        unsigned int u;
        __bf16x2 v;
    } ca{a}, cb{b};
    return __builtin_amdgcn_fdot2_f32_bf16(ca.v, cb.v, acc, false);
}

__device__ __forceinline__ void fp8x4_to_bf16x2(unsigned int w, unsigned int* p) {
    // 4 fp8 -> 2 packed bf16 pairs: v_cvt_pk_f32_fp8 + v_cvt_pk_bf16_f32,
    // EXACT (e4m3 mantissa/exponent embed in bf16), 4 ops per 4 values vs 4
    // converts + widening — and the result feeds v_dot2c directly
    __f32x2 lo = __builtin_amdgcn_cvt_pk_f32_fp8(w, false);
    __f32x2 hi = __builtin_amdgcn_cvt_pk_f32_fp8(w, true);
    __hip_bfloat162 a = __float22bfloat162_rn(float2{lo[0], lo[1]});
    __hip_bfloat162 b = __float22bfloat162_rn(float2{hi[0], hi[1]});
    p[0] = *(const unsigned int*)&a;
    p[1] = *(const unsigned int*)&b;
}

template <bool KV8, int G>
__global__ void __launch_bounds__(WAVE, 2) smg_attn_decode2_t(
    const __hip_bfloat16* __restrict__ q,
    const void* __restrict__ k,
    const void* __restrict__ v,
    const int* __restrict__ pos,
    __hip_bfloat16* __restrict__ out,
    int n_slots, int n_heads, int n_kv_heads, int max_seq, int head_dim, float scale) {
    int sh = blockIdx.x;
    int slot = sh / n_kv_heads;
    int kvh = sh % n_kv_heads;
    if (slot >= n_slots) return;
    int lane = threadIdx.x;
    int T = pos[slot] + 1;
    if (T > max_seq) T = max_seq;

    const size_t head_base = ((size_t)slot * n_kv_heads + kvh) * (size_t)max_seq * head_dim;
    const __hip_bfloat16* kh = (const __hip_bfloat16*)k + head_base;
    const __hip_bfloat16* vh = (const __hip_bfloat16*)v + head_base;
    const unsigned char* kh8 = (const unsigned char*)k + head_base;
    const unsigned char* vh8 = (const unsigned char*)v + head_base;
    const __hip_bfloat16* qh = q + ((size_t)slot * n_heads + (size_t)kvh * G) * head_dim;

    // q staged as packed bf16 pairs (dot2 operands)
    __shared__ unsigned int s_q2[G * 64];
    __shared__ float s_p[G][WAVE];
    {
        const unsigned int* q32 = (const unsigned int*)qh;  // 256 B-aligned rows
        for (int i = lane; i < G * (head_dim / 2); i += WAVE) s_q2[i] = q32[i];
    }
    __syncthreads();

    const int vec_n = head_dim / 8;
    const int chunks = vec_n;
    const int rows_per = WAVE / chunks;
    const int chunk = lane % chunks;
    const int rgrp = lane / chunks;
    float accv[G][8];
#pragma unroll
    for (int g = 0; g < G; ++g)
#pragma unroll
        for (int j = 0; j < 8; ++j) accv[g][j] = 0.f;
    float m[G], l[G];
#pragma unroll
    for (int g = 0; g < G; ++g) { m[g] = -1e30f; l[g] = 0.f; }

    for (int base = 0; base < T; base += WAVE) {
        const int lim = min(WAVE, T - base);
        const int t = base + lane;
        float d[G];
#pragma unroll
        for (int g = 0; g < G; ++g) d[g] = 0.f;
        if (t < T) {
            if constexpr (KV8) {
                // fp8 K row: packed fp8->f32->bf16x2 repack (exact), then the
                // same dot2 stream as bf16 — 6 ops per 4 elems vs 9 for the
                // per-element convert+FMA form
                const uint4* row = (const uint4*)(kh8 + (size_t)t * head_dim);
#pragma unroll 4
                for (int c = 0; c < vec_n / 2; ++c) {
                    uint4 w = row[c];  // 16 fp8
                    const unsigned int* wp = (const unsigned int*)&w;
#pragma unroll
                    for (int wi = 0; wi < 4; ++wi) {
                        unsigned int pk[2];
                        fp8x4_to_bf16x2(wp[wi], pk);
                        const int qb = c * 8 + wi * 2;  // bf16-pair index of elem c*16+wi*4
#pragma unroll
                        for (int g = 0; g < G; ++g) {
                            d[g] = dot2_bf16(pk[0], s_q2[g * (head_dim / 2) + qb], d[g]);
                            d[g] = dot2_bf16(pk[1], s_q2[g * (head_dim / 2) + qb + 1], d[g]);
                        }
                    }
                }
            } else {
                const uint4* row = (const uint4*)(kh + (size_t)t * head_dim);
#pragma unroll 4
                for (int c = 0; c < vec_n; ++c) {
                    uint4 w = row[c];
                    const unsigned int* wp = (const unsigned int*)&w;
#pragma unroll
                    for (int wi = 0; wi < 4; ++wi)
#pragma unroll
                        for (int g = 0; g < G; ++g)
                            d[g] = dot2_bf16(wp[wi], s_q2[g * (head_dim / 2) + c * 4 + wi], d[g]);
                }
            }
        }
        float alpha[G];
#pragma unroll
        for (int g = 0; g < G; ++g) {
            float score = (t < T) ? d[g] * scale : -1e30f;
            float mr = score;
#pragma unroll
            for (int off = 32; off > 0; off >>= 1) mr = fmaxf(mr, __shfl_xor(mr, off, WAVE));
            float m_new = fmaxf(m[g], mr);
            alpha[g] = __expf(m[g] - m_new);
            float p = (t < T) ? __expf(score - m_new) : 0.f;
            s_p[g][lane] = p;
            float pr = p;
#pragma unroll
            for (int off = 32; off > 0; off >>= 1) pr += __shfl_xor(pr, off, WAVE);
            l[g] = l[g] * alpha[g] + pr;
            m[g] = m_new;
#pragma unroll
            for (int j = 0; j < 8; ++j) accv[g][j] *= alpha[g];
        }
        __syncthreads();
        if constexpr (KV8) {
            const unsigned char* vtile8 = vh8 + (size_t)base * head_dim + chunk * 8;
            for (int j = rgrp; j < lim; j += rows_per) {
                const uint2 w = *(const uint2*)(vtile8 + (size_t)j * head_dim);
                float dec[8];
                fp8x4_to_f32(w.x, dec);
                fp8x4_to_f32(w.y, dec + 4);
#pragma unroll
                for (int g = 0; g < G; ++g) {
                    const float pj = s_p[g][j];
#pragma unroll
                    for (int s = 0; s < 8; ++s) accv[g][s] += pj * dec[s];
                }
            }
        } else {
            const __hip_bfloat16* vtile = vh + (size_t)base * head_dim + chunk * 8;
            if (lim == WAVE && head_dim == 128) {
#pragma unroll
                for (int it = 0; it < 16; ++it) {
                    const int j = rgrp + it * 4;  // rows_per == 4 at hd128
                    const uint4 w = *(const uint4*)(vtile + (size_t)j * head_dim);
                    const unsigned short* hsp = (const unsigned short*)&w;
#pragma unroll
                    for (int g = 0; g < G; ++g) {
                        const float pj = s_p[g][j];
#pragma unroll
                        for (int jj = 0; jj < 8; ++jj) accv[g][jj] += pj * bf16_to_f32(hsp[jj]);
                    }
                }
            } else {
                for (int j = rgrp; j < lim; j += rows_per) {
                    const uint4 w = *(const uint4*)(vtile + (size_t)j * head_dim);
                    const unsigned short* hsp = (const unsigned short*)&w;
#pragma unroll
                    for (int g = 0; g < G; ++g) {
                        const float pj = s_p[g][j];
#pragma unroll
                        for (int jj = 0; jj < 8; ++jj) accv[g][jj] += pj * bf16_to_f32(hsp[jj]);
                    }
                }
            }
        }
        __syncthreads();
    }
#pragma unroll
    for (int g = 0; g < G; ++g) {
        for (int off = chunks; off < WAVE; off <<= 1) {
#pragma unroll
            for (int jj = 0; jj < 8; ++jj) accv[g][jj] += __shfl_xor(accv[g][jj], off, WAVE);
        }
    }
    if (rgrp == 0) {
#pragma unroll
        for (int g = 0; g < G; ++g) {
            float inv = l[g] > 0.f ? 1.f / l[g] : 0.f;
            __hip_bfloat16* orow =
                out + ((size_t)slot * n_heads + (size_t)kvh * G + g) * head_dim;
#pragma unroll
            for (int jj = 0; jj < 8; ++jj) orow[chunk * 8 + jj] = (__hip_bfloat16)(accv[g][jj] * inv);
        }
    }
}

extern "C" int smg_attn_decode_launch_gqa2(const void* q, const void* k, const void* v,
                                           const void* pos, void* out, int n_slots, int n_heads,
                                           int n_kv_heads, int max_seq, int head_dim, float scale,
                                           void* stream, int kv_fp8) {
    if (head_dim > 128 || (head_dim & 7)) return -1;
    int chunks = head_dim / 8;
    if (chunks & (chunks - 1)) return -1;
    if (kv_fp8 && (head_dim & 15)) return -1;  // fp8 K streaming is 16-wide
    if (n_kv_heads <= 0 || n_heads % n_kv_heads) return -4;
    dim3 grid(n_slots * n_kv_heads);
    const int G = n_heads / n_kv_heads;
    hipStream_t s = (hipStream_t)stream;
#define LAUNCH2_CASE(K8, GG)                                                                   \
    hipLaunchKernelGGL((smg_attn_decode2_t<K8, GG>), grid, dim3(WAVE), 0, s,                   \
                       (const __hip_bfloat16*)q, k, v, (const int*)pos,                        \
                       (__hip_bfloat16*)out, n_slots, n_heads, n_kv_heads, max_seq,            \
                       head_dim, scale);                                                       \
    break;
    if (kv_fp8) {
        switch (G) {
            case 1: LAUNCH2_CASE(true, 1)
            case 2: LAUNCH2_CASE(true, 2)
            case 4: LAUNCH2_CASE(true, 4)
            case 8: LAUNCH2_CASE(true, 8)
            default: return -4;
        }
    } else {
        switch (G) {
            case 1: LAUNCH2_CASE(false, 1)
            case 2: LAUNCH2_CASE(false, 2)
            case 4: LAUNCH2_CASE(false, 4)
            case 8: LAUNCH2_CASE(false, 8)
            default: return -4;
        }
    }
#undef LAUNCH2_CASE
    return hipGetLastError() == hipSuccess ? 0 : -2;
}

extern "C" int smg_attn_decode_launch_ex(const void* q, const void* k, const void* v,
                                         const void* pos, void* out, int n_slots, int n_heads,
                                         int max_seq, int head_dim, float scale, void* stream,
                                         int kv_fp8) {
    return smg_attn_decode_launch_gqa(q, k, v, pos, out, n_slots, n_heads, n_heads, max_seq,
                                      head_dim, scale, stream, kv_fp8);
}

extern "C" int smg_attn_decode_launch(const void* q, const void* k, const void* v,
                                      const void* pos, void* out, int n_slots, int n_heads,
                                      int max_seq, int head_dim, float scale, void* stream) {
    return smg_attn_decode_launch_ex(q, k, v, pos, out, n_slots, n_heads, max_seq, head_dim,
                                     scale, stream, 0);
}
