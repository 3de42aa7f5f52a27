// Segment (per-destination) attention aggregation kernels for CDNA4/gfx950.
//
// Replaces the reference's torch_scatter-based PyG AttentionalAggregation
// (reference gcbf/nn/gnn.py:17-19): scatter-softmax of a per-edge gate over
// incoming edges of each node, then the softmax-weighted sum of per-edge
// messages.  Edge lists are destination-sorted (CSR via `ptr`), so each
// segment is contiguous: no atomics, deterministic reduction order, bitwise
// reproducible across runs.
//
// Decomposition: one 256-thread workgroup (4 waves of 64) per destination
// node.  The per-edge att / dot-product scratch is staged through the output
// buffers themselves (same-workgroup global RAW after __syncthreads is
// coherent), so no degree-dependent LDS: any segment size works and
// occupancy stays high.
#include <hip/hip_runtime.h>
#include <cfloat>

#define BLOCK 256
#define WAVE 64

__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
    for (int off = WAVE / 2; off > 0; off >>= 1)
        v += __shfl_down(v, off, WAVE);
    return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
    for (int off = WAVE / 2; off > 0; off >>= 1)
        v = fmaxf(v, __shfl_down(v, off, WAVE));
    return v;
}

__device__ __forceinline__ float block_reduce_sum(float v, float* red,
                                                  int wid, int lane) {
    v = wave_reduce_sum(v);
    __syncthreads();
    if (lane == 0) red[wid] = v;
    __syncthreads();
    return red[0] + red[1] + red[2] + red[3];
}

// ---------------------------------------------------------------- forward
extern "C" __global__ void seg_attn_fwd(
        const float* __restrict__ msg,    // (E, D)
        const float* __restrict__ gate,   // (E,)
        const int* __restrict__ ptr,      // (N+1,)
        float* __restrict__ att,          // (E,)  out
        float* __restrict__ out,          // (N, D) out
        int N, int D) {
    const int n = blockIdx.x;
    if (n >= N) return;
    const int lo = ptr[n], hi = ptr[n + 1];
    const int deg = hi - lo;
    const int tid = threadIdx.x;
    const int wid = tid / WAVE, lane = tid % WAVE;

    __shared__ __attribute__((aligned(16))) float red[4];

    float* out_row = out + (size_t)n * D;
    if (deg == 0) {
        for (int d = tid; d < D; d += BLOCK) out_row[d] = 0.f;
        return;
    }

    // 1) segment max of gate
    float m = -FLT_MAX;
    for (int e = tid; e < deg; e += BLOCK) m = fmaxf(m, gate[lo + e]);
    m = wave_reduce_max(m);
    if (lane == 0) red[wid] = m;
    __syncthreads();
    m = fmaxf(fmaxf(red[0], red[1]), fmaxf(red[2], red[3]));

    // 2) exp, stage into att, segment sum
    float s = 0.f;
    for (int e = tid; e < deg; e += BLOCK) {
        const float x = __expf(gate[lo + e] - m);
        att[lo + e] = x;
        s += x;
    }
    s = block_reduce_sum(s, red, wid, lane);
    const float inv = 1.f / (s + 1e-16f);

    // 3) normalize in place
    for (int e = tid; e < deg; e += BLOCK) att[lo + e] *= inv;
    __syncthreads();

    // 4) out[n][:] = sum_e att[e] * msg[e][:]  (lanes own feature columns)
    if ((D & 3) == 0) {
        const int D4 = D >> 2;
        for (int d4 = tid; d4 < D4; d4 += BLOCK) {
            float4 acc = make_float4(0.f, 0.f, 0.f, 0.f);
            for (int e = 0; e < deg; ++e) {
                const float w = att[lo + e];
                const float4 v = reinterpret_cast<const float4*>(
                    msg + (size_t)(lo + e) * D)[d4];
                acc.x += w * v.x; acc.y += w * v.y;
                acc.z += w * v.z; acc.w += w * v.w;
            }
            reinterpret_cast<float4*>(out_row)[d4] = acc;
        }
    } else {
        for (int d = tid; d < D; d += BLOCK) {
            float acc = 0.f;
            for (int e = 0; e < deg; ++e)
                acc += att[lo + e] * msg[(size_t)(lo + e) * D + d];
            out_row[d] = acc;
        }
    }
}

// ---------------------------------------------------------------- backward
// dmsg[e] = att[e] * g_n ;  s_e = <msg_e, g_n> ;
// dgate[e] = att[e] * (s_e - sum_{e' in n} att_e' s_e')
// s_e is staged in dgate itself between the two passes.
extern "C" __global__ void seg_attn_bwd(
        const float* __restrict__ grad_out,  // (N, D)
        const float* __restrict__ msg,       // (E, D)
        const float* __restrict__ att,       // (E,)
        const int* __restrict__ ptr,         // (N+1,)
        float* __restrict__ dmsg,            // (E, D) out
        float* __restrict__ dgate,           // (E,)  out
        int N, int D) {
    const int n = blockIdx.x;
    if (n >= N) return;
    const int lo = ptr[n], hi = ptr[n + 1];
    const int deg = hi - lo;
    if (deg == 0) return;
    const int tid = threadIdx.x;
    const int wid = tid / WAVE, lane = tid % WAVE;
    const int n_waves = BLOCK / WAVE;

    __shared__ __attribute__((aligned(16))) float red[4];

    const float* g = grad_out + (size_t)n * D;

    // pass 1: per-edge dmsg write + dot product (one wave per edge)
    const bool vec4 = (D & 3) == 0;
    for (int e = wid; e < deg; e += n_waves) {
        const float ae = att[lo + e];
        const float* me = msg + (size_t)(lo + e) * D;
        float* dme = dmsg + (size_t)(lo + e) * D;
        float dot = 0.f;
        if (vec4) {
            const int D4 = D >> 2;
            for (int d4 = lane; d4 < D4; d4 += WAVE) {
                const float4 gv = reinterpret_cast<const float4*>(g)[d4];
                const float4 mv = reinterpret_cast<const float4*>(me)[d4];
                float4 dv;
                dv.x = ae * gv.x; dv.y = ae * gv.y;
                dv.z = ae * gv.z; dv.w = ae * gv.w;
                reinterpret_cast<float4*>(dme)[d4] = dv;
                dot += mv.x * gv.x + mv.y * gv.y + mv.z * gv.z + mv.w * gv.w;
            }
        } else {
            for (int d = lane; d < D; d += WAVE) {
                const float gv = g[d];
                dme[d] = ae * gv;
                dot += me[d] * gv;
            }
        }
        dot = wave_reduce_sum(dot);
        if (lane == 0) dgate[lo + e] = dot;   // stage s_e
    }
    __syncthreads();

    // seg = sum_e att_e * s_e
    float part = 0.f;
    for (int e = tid; e < deg; e += BLOCK) part += att[lo + e] * dgate[lo + e];
    const float seg = block_reduce_sum(part, red, wid, lane);

    // pass 2: dgate (overwrites the staged s_e)
    for (int e = tid; e < deg; e += BLOCK)
        dgate[lo + e] = att[lo + e] * (dgate[lo + e] - seg);
}

// ------------------------------------------------------------- launchers
extern "C" void launch_seg_attn_fwd(const float* msg, const float* gate,
                                    const int* ptr, float* att, float* out,
                                    int N, int D, hipStream_t stream) {
    hipLaunchKernelGGL(seg_attn_fwd, dim3(N), dim3(BLOCK), 0, stream,
                       msg, gate, ptr, att, out, N, D);
}

extern "C" void launch_seg_attn_bwd(const float* grad_out, const float* msg,
                                    const float* att, const int* ptr,
                                    float* dmsg, float* dgate, int N, int D,
                                    hipStream_t stream) {
    hipLaunchKernelGGL(seg_attn_bwd, dim3(N), dim3(BLOCK), 0, stream,
                       grad_out, msg, att, ptr, dmsg, dgate, N, D);
}
