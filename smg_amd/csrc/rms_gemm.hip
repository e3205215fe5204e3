// Fused rms_norm + GEMM for the decode projections — gfx950 MFMA.
//
// Computes C[M,N] = rowscale(A) ⊙ (A @ Wt^T) in bf16 with fp32 accumulate,
// where rowscale = 1/rms(row) and the rms elementwise weight g is FOLDED
// into the stored weight (W'_kn = g_k · W_kn, prepared once at init):
//
//     rms_norm(x; g) @ W  ==  invrms(x) ⊙ (x @ (g ⊙ W))
//
// so the kernel never materializes the normalized activations — one HBM
// round trip of A and one layer_norm launch disappear per projection
// (measured: the decode QKV/W13 GEMMs run at 190-800 TF in hipBLASLt on the
// tall-skinny [S≈520, 2048]×[2048, N] shapes, far under both the MFMA and
// bandwidth ceilings — small-M tiles leave the chip idle).
//
// Weights are stored PRE-TRANSPOSED [N, K] so that BOTH MFMA operands read
// contiguous K-runs: the v_mfma_f32_16x16x32_bf16 fragment wants lane
// (i + 16*ko) to hold elements [row i][k = 8*ko + 0..8) — a 16-byte
// ds_read_b128 from a row-major [row][k] LDS tile.  No transposes anywhere.
//
// Shapes: M arbitrary, K % 64 == 0, N % 128 == 0 (the engine pads weights).
// Tiling: WG = 4 waves = M64 × N128 tile; wave = M64 × N32 (4×2 MFMA tiles,
// 32 f32 acc VGPRs); K-loop staged through double-buffered LDS
// (A 64×64, W 128×64, +8-element row pad → conflict-free b128 reads).
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdint>

#define WAVE 64

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

// K per MFMA instruction for 16x16x32
#define MFMA_K 32
#define KSTEP 32           // K per LDS stage (1 MFMA k-step)
#define APAD 8             // row pad (bf16 elems) -> conflict-free b128
#define ALD (KSTEP + APAD)

extern "C" __global__ void __launch_bounds__(256)
smg_row_invrms(const __hip_bfloat16* __restrict__ a, float* __restrict__ out,
               int m, int k, float eps) {
    // one wave per row: mean(x^2) over k, out = rsqrt(mean + eps)
    int row = blockIdx.x * 4 + (threadIdx.x / WAVE);
    if (row >= m) return;
    int lane = threadIdx.x % WAVE;
    const __hip_bfloat16* r = a + (size_t)row * k;
    float s = 0.f;
    for (int i = lane * 8; i < k; i += WAVE * 8) {
        bf16x8 v8 = *(const bf16x8*)(r + i);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float x = (float)v8[j];
            s += x * x;
        }
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) s += __shfl_xor(s, off, WAVE);
    if (lane == 0) out[row] = rsqrtf(s / (float)k + eps);
}

#define GEMM_THREADS 512  // 8 waves: WG tile 128Mx128N, wave 64Mx32N, ~6 waves/SIMD

extern "C" __global__ void __launch_bounds__(GEMM_THREADS)
smg_rms_gemm(const __hip_bfloat16* __restrict__ a,   // [M, K] raw activations
             const __hip_bfloat16* __restrict__ wt,  // [N, K] g-folded, transposed
             const float* __restrict__ invrms,       // [M]
             __hip_bfloat16* __restrict__ c,         // [M, N]
             int M, int K, int N) {
    const int m0 = blockIdx.x * 128;  // this WG's M tile
    const int n0 = blockIdx.y * 128;  // this WG's N tile
    const int tid = threadIdx.x;
    const int wave = tid / WAVE;
    const int lane = tid % WAVE;

    __shared__ __hip_bfloat16 sA[2][128][ALD];
    __shared__ __hip_bfloat16 sW[2][128][ALD];

    // 8 waves tile the 128x128 output as 2(M) x 4(N): wave (wm, wn) covers
    // rows [wm*64, +64) x cols [wn*32, +32) -> 4x2 MFMA tiles, 32 acc VGPRs
    const int wave_m = (wave >> 2) * 64;
    const int wave_n = (wave & 3) * 32;
    f32x4 acc[4][2];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j) acc[i][j] = (f32x4)0.f;

    // staging: A and W tiles are each 128x32 = 512 bf16x8 chunks -> 1/thread
    const int a_row0 = tid / 4;          // 0..127
    const int a_col8 = (tid % 4) * 8;
    const int w_row0 = tid / 4;
    const int w_col8 = (tid % 4) * 8;

    // software pipeline: global loads ISSUE before the MFMA section (their
    // latency hides under compute); the vmcnt wait lands at the LDS write
    // AFTER the MFMAs, not before them
    bf16x8 pa, pw;
    auto load_global = [&](int k0) {
        int gr = m0 + a_row0;
        pa = (gr < M)
            ? *(const bf16x8*)(a + (size_t)gr * K + k0 + a_col8)
            : (bf16x8)(__bf16)0.f;
        pw = *(const bf16x8*)(wt + (size_t)(n0 + w_row0) * K + k0 + w_col8);
    };
    auto write_lds = [&](int buf) {
        *(bf16x8*)&sA[buf][a_row0][a_col8] = pa;
        *(bf16x8*)&sW[buf][w_row0][w_col8] = pw;
    };

    load_global(0);
    write_lds(0);
    __syncthreads();

    const int frag_row = lane & 15;        // MFMA i / j
    const int frag_ko = (lane >> 4) * 8;   // k octet within the 32-k step

    for (int k0 = 0; k0 < K; k0 += KSTEP) {
        const int buf = (k0 / KSTEP) & 1;
        const bool more = k0 + KSTEP < K;
        if (more) {
            load_global(k0 + KSTEP);  // issue now, consumed after the MFMAs
        }
        {
            const int kb = frag_ko;
            bf16x8 afrag[4], wfrag[2];
#pragma unroll
            for (int i = 0; i < 4; ++i)
                afrag[i] = *(const bf16x8*)&sA[buf][wave_m + i * 16 + frag_row][kb];
#pragma unroll
            for (int j = 0; j < 2; ++j)
                wfrag[j] = *(const bf16x8*)&sW[buf][wave_n + j * 16 + frag_row][kb];
#pragma unroll
            for (int i = 0; i < 4; ++i)
#pragma unroll
                for (int j = 0; j < 2; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        afrag[i], wfrag[j], acc[i][j], 0, 0, 0);
        }
        if (more) {
            write_lds(buf ^ 1);  // vm wait here, after the compute phase
        }
        __syncthreads();
    }

    // epilogue: C/D mapping for 16x16x32: col = lane&15, row = (lane>>4)*4 + reg
    const int c_col = lane & 15;
    const int c_row0 = (lane >> 4) * 4;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
        for (int j = 0; j < 2; ++j) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = m0 + wave_m + i * 16 + c_row0 + r;
                if (row >= M) continue;
                int col = n0 + wave_n + j * 16 + c_col;
                c[(size_t)row * N + col] =
                    (__hip_bfloat16)(acc[i][j][r] * invrms[row]);
            }
        }
    }
}

// Layout probe: one mfma_f32_16x16x32_bf16 on caller-provided per-lane
// fragments.  afrag/bfrag: [64 lanes][8] bf16; out: [64 lanes][4] f32 in the
// raw accumulator register order (host decodes with the C map).  Lets a test
// determine the true A/B k-ordering empirically instead of trusting docs.
extern "C" __global__ void __launch_bounds__(WAVE)
smg_mfma_probe(const __hip_bfloat16* __restrict__ afrag,
               const __hip_bfloat16* __restrict__ bfrag,
               float* __restrict__ out) {
    int lane = threadIdx.x;
    bf16x8 a = *(const bf16x8*)(afrag + lane * 8);
    bf16x8 b = *(const bf16x8*)(bfrag + lane * 8);
    f32x4 c = (f32x4)0.f;
    c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
#pragma unroll
    for (int r = 0; r < 4; ++r) out[lane * 4 + r] = c[r];
}

extern "C" int smg_mfma_probe_launch(const void* afrag, const void* bfrag, void* out,
                                     void* stream) {
    hipLaunchKernelGGL(smg_mfma_probe, dim3(1), dim3(WAVE), 0, (hipStream_t)stream,
                       (const __hip_bfloat16*)afrag, (const __hip_bfloat16*)bfrag,
                       (float*)out);
    return hipGetLastError() == hipSuccess ? 0 : -2;
}

extern "C" int smg_rms_gemm_launch(const void* a, const void* wt, const void* invrms,
                                   void* c, int M, int K, int N, void* stream) {
    if (K % KSTEP || N % 128) return -1;
    dim3 grid((M + 127) / 128, N / 128);
    hipLaunchKernelGGL(smg_rms_gemm, grid, dim3(GEMM_THREADS), 0, (hipStream_t)stream,
                       (const __hip_bfloat16*)a, (const __hip_bfloat16*)wt,
                       (const float*)invrms, (__hip_bfloat16*)c, M, K, N);
    return hipGetLastError() == hipSuccess ? 0 : -2;
}

extern "C" int smg_row_invrms_launch(const void* a, void* out, int m, int k, float eps,
                                     void* stream) {
    if (k % (WAVE * 8)) return -1;
    hipLaunchKernelGGL(smg_row_invrms, dim3((m + 3) / 4), dim3(256), 0, (hipStream_t)stream,
                       (const __hip_bfloat16*)a, (float*)out, m, k, eps);
    return hipGetLastError() == hipSuccess ? 0 : -2;
}
