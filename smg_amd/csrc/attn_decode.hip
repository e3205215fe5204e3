// Fused single-token (decode) attention for the GPU worker engine — gfx950.
//
// One wave per (slot, head): online-softmax over the slot's KV window.
// Replaces torch sdpa in the engine's full-arena decode, where sdpa must
// read the rectangular [slots, maxlen] window for every slot; this kernel
// reads only kv[slot][:pos+1], so idle/short slots cost nothing — the decode
// step becomes KV-bandwidth bound on the ACTIVE tokens.
//
// Layout contract (the engine's KV arena, one layer):
//   K, V:  [n_slots, n_heads, max_seq, head_dim]  bf16, contiguous
//   q:     [n_slots, n_heads, head_dim]           bf16, contiguous
//   pos:   [n_slots] int32 — attend kpos <= pos[slot] (the current token's
//          K/V must already be written at pos[slot])
//   out:   [n_slots, n_heads, head_dim]           bf16
//
// head_dim <= 128 (2 elements per lane).  Round of 64 timesteps per
// iteration: lane t computes the full q·K[t] dot (K rows stream per lane),
// then the P·V accumulation broadcasts each lane's probability with shfl
// while V rows load coalesced (4 B per lane).
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdint>

#define WAVE 64

extern "C" __global__ void __launch_bounds__(WAVE) smg_attn_decode(
    const __hip_bfloat16* __restrict__ q,
    const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v,
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
    const __hip_bfloat16* kh = k + head_base;
    const __hip_bfloat16* vh = v + head_base;
    const __hip_bfloat16* qh = q + ((size_t)slot * n_heads + head) * head_dim;

    // stage q in LDS (whole wave reads it every dot)
    __shared__ float s_q[128];
    // K/V tiles: 64 rows, stride 136 bf16 = 17 16-byte slots — rows stay
    // 16 B aligned for dwordx4 loads and the odd slot stride staggers banks
    __shared__ __hip_bfloat16 s_k[WAVE][136];
    for (int i = lane; i < head_dim; i += WAVE) s_q[i] = (float)qh[i];
    __syncthreads();

    // accumulator: 2 output elements per lane (head_dim <= 128)
    float acc0 = 0.f, acc1 = 0.f;
    float m = -1e30f, l = 0.f;
    int e0 = lane * 2, e1 = lane * 2 + 1;

    __shared__ __hip_bfloat16 s_v[WAVE][136];
    __shared__ float s_p[WAVE];
    // vectorized tile fill: 16 B per lane, 4 rows per instruction
    const int vec_per_row = head_dim / 8;  // 16-byte chunks per row (<=16)
    const int rows_per_iter = WAVE / vec_per_row;
    const int sub = lane % vec_per_row;
    const int rofs = lane / vec_per_row;
    for (int base = 0; base < T; base += WAVE) {
        int lim = min(WAVE, T - base);
        for (int r0 = 0; r0 < lim; r0 += rows_per_iter) {
            int r = r0 + rofs;
            if (r < lim) {
                const uint4* krow = (const uint4*)(kh + (size_t)(base + r) * head_dim);
                const uint4* vrow = (const uint4*)(vh + (size_t)(base + r) * head_dim);
                ((uint4*)&s_k[r][0])[sub] = krow[sub];
                ((uint4*)&s_v[r][0])[sub] = vrow[sub];
            }
        }
        __syncthreads();
        int t = base + lane;
        float score = -1e30f;
        if (t < T) {
            float d = 0.f;
#pragma unroll 8
            for (int i = 0; i < head_dim; i += 2)
                d += s_q[i] * (float)s_k[lane][i] + s_q[i + 1] * (float)s_k[lane][i + 1];
            score = d * scale;
        }
        // online softmax across the wave's 64 scores
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
        acc0 *= alpha;
        acc1 *= alpha;
        __syncthreads();
        // P·V from LDS: lane owns output elements (e0, e1)
        if (e0 < head_dim) {
#pragma unroll 8
            for (int j = 0; j < lim; ++j) {
                float pj = s_p[j];
                acc0 += pj * (float)s_v[j][e0];
                acc1 += pj * (float)s_v[j][e1];
            }
        }
        __syncthreads();
        m = m_new;
    }
    float inv = l > 0.f ? 1.f / l : 0.f;
    __hip_bfloat16* orow = out + ((size_t)slot * n_heads + head) * head_dim;
    if (e0 < head_dim) {
        orow[e0] = (__hip_bfloat16)(acc0 * inv);
        orow[e1] = (__hip_bfloat16)(acc1 * inv);
    }
}

extern "C" int smg_attn_decode_launch(const void* q, const void* k, const void* v,
                                      const void* pos, void* out, int n_slots, int n_heads,
                                      int max_seq, int head_dim, float scale, void* stream) {
    if (head_dim > 128 || (head_dim & 7)) return -1;
    dim3 grid(n_slots * n_heads);
    hipLaunchKernelGGL(smg_attn_decode, grid, dim3(WAVE), 0, (hipStream_t)stream,
                       (const __hip_bfloat16*)q, (const __hip_bfloat16*)k,
                       (const __hip_bfloat16*)v, (const int*)pos, (__hip_bfloat16*)out,
                       n_slots, n_heads, max_seq, head_dim, scale);
    return hipGetLastError() == hipSuccess ? 0 : -2;
}
