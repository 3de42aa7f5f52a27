// Segment max over dst-sorted edges for CDNA4/gfx950 (MACBF's aggr='max',
// reference gcbf/nn/gnn.py:117 via torch_scatter scatter_max).
//
// One 256-thread workgroup per destination node; lanes own feature columns
// and scan the CSR segment, tracking the argmax for the backward pass.
// Empty segments produce 0 (PyG's fill value).  Deterministic: first
// maximal edge wins ties (scan order).
#include <hip/hip_runtime.h>
#include <cfloat>

#define BLOCK 256

extern "C" __global__ void seg_max_fwd(
        const float* __restrict__ values,  // (E, D)
        const int* __restrict__ ptr,       // (N+1,)
        float* __restrict__ out,           // (N, D)
        int* __restrict__ argmax,          // (N, D) edge index or -1
        int N, int D) {
    const int n = blockIdx.x;
    if (n >= N) return;
    const int lo = ptr[n], hi = ptr[n + 1];
    const int tid = threadIdx.x;
    float* orow = out + (size_t)n * D;
    int* arow = argmax + (size_t)n * D;
    for (int d = tid; d < D; d += BLOCK) {
        float m = -FLT_MAX;
        int am = -1;
        for (int e = lo; e < hi; ++e) {
            const float v = values[(size_t)e * D + d];
            if (v > m) {
                m = v;
                am = e;
            }
        }
        orow[d] = (am < 0) ? 0.f : m;
        arow[d] = am;
    }
}

extern "C" __global__ void seg_max_bwd(
        const float* __restrict__ grad_out,  // (N, D)
        const int* __restrict__ argmax,      // (N, D)
        float* __restrict__ dval,            // (E, D), pre-zeroed
        int N, int D) {
    const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= (long)N * D) return;
    const int e = argmax[idx];
    if (e >= 0)
        dval[(size_t)e * D + (idx % D)] = grad_out[idx];
}

extern "C" void launch_seg_max_fwd(const float* values, const int* ptr,
                                   float* out, int* argmax, int N, int D,
                                   hipStream_t stream) {
    hipLaunchKernelGGL(seg_max_fwd, dim3(N), dim3(BLOCK), 0, stream, values,
                       ptr, out, argmax, N, D);
}

extern "C" void launch_seg_max_bwd(const float* grad_out, const int* argmax,
                                   float* dval, int N, int D,
                                   hipStream_t stream) {
    const long total = (long)N * D;
    const int blocks = (int)((total + 255) / 256);
    hipLaunchKernelGGL(seg_max_bwd, dim3(blocks), dim3(256), 0, stream,
                       grad_out, argmax, dval, N, D);
}
