// Fused bf16 Linear (+bias +ReLU/Tanh) MFMA GEMM for CDNA4/gfx950.
//
// C[M,N] = act(A[M,K] · W[N,K]^T + bias[N]) — the projection shape of every
// MLP layer in the GNN (nn.Linear weight layout is (out,in) = (N,K), so both
// operands are read K-major: a TN GEMM).  bf16 inputs, fp32 accumulation
// (v_mfma_f32_16x16x32_bf16), bias+activation fused into the epilogue so the
// inter-layer activations never round-trip HBM as separate kernels.
//
// Structure (the CDNA4 "step-3" recipe): 128×128 block tile, BK=64,
// 256 threads = 4 waves in 2×2, each wave computing a 64×64 sub-tile as
// 4×4 MFMA fragments; double-buffered LDS staged with
// __builtin_amdgcn_global_load_lds (16 B per lane), with the st_16x32 XOR
// swizzle applied on the *global source address* (glds destinations are
// lane-linear) and on the ds_read offsets, killing the 128-B-row bank
// conflicts.  Requires M%128==0 (callers bucket M to 256), N%128==0,
// K%64==0 (callers zero-pad K).
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define BM 128
#define BN 128
#define BK 64
#define THREADS 256

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define ACT_NONE 0
#define ACT_RELU 1
#define ACT_TANH 2

// st_16x32 swizzle within each 1024-B subtile: XOR byte-bit-5 with bit-9.
__device__ __forceinline__ unsigned swz(unsigned byte_off) {
    return byte_off ^ (((byte_off >> 9) & 1u) << 5);
}

// One tile (BM×BK bf16, row-major logical) staged to LDS via glds with the
// swizzle pre-applied on the global source.  `gbase` points at
// tile(row0, k0); grow = K elements per row.  Each of the 4 waves issues
// 4 glds (16 B/lane): instruction q of wave w covers logical bytes
// [ (w*4+q)*1024, +1024 ).
__device__ __forceinline__ void stage_tile(const __hip_bfloat16* __restrict__ g,
                                           char* lds_base,
                                           int grow, int wid, int lane) {
#pragma unroll
    for (int q = 0; q < 4; ++q) {
        const unsigned p_phys = (unsigned)(wid * 4 + q) * 1024u
                                + (unsigned)lane * 16u;
        const unsigned p_log = swz(p_phys);
        const int row = p_log >> 7;          // 128 B per logical row
        const int colb = p_log & 127;        // byte within row
        const __hip_bfloat16* src = g + (size_t)row * grow + (colb >> 1);
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int*)src,
            (__attribute__((address_space(3))) unsigned int*)(
                lds_base + p_phys),
            16, 0, 0);
    }
}

extern "C" __global__ __launch_bounds__(THREADS, 2) void fused_linear_bf16(
        const __hip_bfloat16* __restrict__ A,   // (M, K)
        const __hip_bfloat16* __restrict__ W,   // (N, K)
        const float* __restrict__ bias,         // (N,) or null
        __hip_bfloat16* __restrict__ Cb,        // (M, N) out when !out_f32
        float* __restrict__ Cf,                 // (M, N) out when out_f32
        int M, int N, int K, int act, int out_f32) {
    __shared__ __attribute__((aligned(16))) char smem[2 * 2 * BM * BK * 2];
    // layout: [buf][A|W][BM*BK bf16]
    const unsigned A_off[2] = {0u, 2u * 2 * BM * BK};
    const unsigned W_off[2] = {(unsigned)(BM * BK * 2),
                               (unsigned)(2 * 2 * BM * BK + BM * BK * 2)};

    const int tid = threadIdx.x;
    const int wid = tid / WAVE, lane = tid % WAVE;
    const int bm = blockIdx.x * BM;
    const int bn = blockIdx.y * BN;
    // wave tile: 2×2 waves, each 64×64
    const int wr = (wid >> 1) * 64;   // row offset within block tile
    const int wc = (wid & 1) * 64;    // col offset

    f32x4 acc[4][4] = {};

    const int n_tiles = K / BK;
    // prologue: stage tile 0
    stage_tile(A + (size_t)bm * K, smem + A_off[0], K, wid, lane);
    stage_tile(W + (size_t)bn * K, smem + W_off[0], K, wid, lane);

    // fragment ds_read offsets (logical → swizzled physical), 16 B each:
    // lane reads row (lane&15) of its 16-row fragment block, k-span
    // (lane>>4)*8.
    const int frow = lane & 15;
    const int fk = (lane >> 4) * 8;

    for (int t = 0; t < n_tiles; ++t) {
        const int buf = t & 1;
        if (t + 1 < n_tiles) {
            // prefetch next tile into the other buffer (glds queues are
            // drained by the barrier below — simple 2-stage pipeline)
            stage_tile(A + (size_t)bm * K + (t + 1) * BK,
                       smem + A_off[buf ^ 1], K, wid, lane);
            stage_tile(W + (size_t)bn * K + (t + 1) * BK,
                       smem + W_off[buf ^ 1], K, wid, lane);
        }
        __syncthreads();

#pragma unroll
        for (int kk = 0; kk < BK; kk += 32) {
            bf16x8 a_frag[4], b_frag[4];
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const unsigned a_log = (unsigned)((wr + r * 16 + frow) * BK
                                                  + kk + fk) * 2u;
                a_frag[r] = *(const bf16x8*)(smem + A_off[buf] + swz(a_log));
                const unsigned b_log = (unsigned)((wc + r * 16 + frow) * BK
                                                  + kk + fk) * 2u;
                b_frag[r] = *(const bf16x8*)(smem + W_off[buf] + swz(b_log));
            }
#pragma unroll
            for (int mi = 0; mi < 4; ++mi)
#pragma unroll
                for (int ni = 0; ni < 4; ++ni)
                    acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
        }
        __syncthreads();
    }

    // epilogue: bias + activation + store.  C/D lane map (16x16x32):
    // col = lane&15, row = (lane>>4)*4 + reg.
    const int ccol = lane & 15;
    const int crow = (lane >> 4) * 4;
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
            const int col = bn + wc + ni * 16 + ccol;
            const float b = bias ? bias[col] : 0.f;
#pragma unroll
            for (int rg = 0; rg < 4; ++rg) {
                const int row = bm + wr + mi * 16 + crow + rg;
                float v = acc[mi][ni][rg] + b;
                if (act == ACT_RELU) v = fmaxf(v, 0.f);
                else if (act == ACT_TANH) v = tanhf(v);
                if (out_f32)
                    Cf[(size_t)row * N + col] = v;
                else
                    Cb[(size_t)row * N + col] = __float2bfloat16(v);
            }
        }
    }
}

extern "C" void launch_fused_linear_bf16(const void* A, const void* W,
                                         const float* bias, void* Cb,
                                         float* Cf, int M, int N, int K,
                                         int act, int out_f32,
                                         hipStream_t stream) {
    dim3 grid(M / BM, N / BN);
    hipLaunchKernelGGL(fused_linear_bf16, grid, dim3(THREADS), 0, stream,
                       (const __hip_bfloat16*)A, (const __hip_bfloat16*)W,
                       bias, (__hip_bfloat16*)Cb, Cf, M, N, K, act, out_f32);
}
