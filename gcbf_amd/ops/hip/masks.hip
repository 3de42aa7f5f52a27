// Fused batched safety-mask kernels for CDNA4/gfx950.
//
// Computes the reference's per-agent safe / unsafe / collision masks
// (reference gcbf/env/simple_car.py:306-387, dubins_car.py:818-923,
// simple_drone.py:379-465) in ONE pass over the (B, n_agents, N) pairwise
// geometry — the reference loops graphs in Python and launches dozens of
// broadcast kernels per call.  One wave per agent row; lanes stride the
// sender dimension and reduce with ballots.
//
// Env kinds (mask geometry):
//   0 car2d:   P=2, cone dir = velocity (vx,vy)/|v|, warn 4r, safe thr 4r
//   1 dubins:  P=2, cone dir = heading (cosθ, sinθ), warn 3r, safe thr 3r
//   2 drone3d: P=3, cone dir = (vx/|v|, vy/|v|, vz)  [vz quirk], warn 4r,
//              safe thr 4r, unsafe-collision diag 2r+1
#include <hip/hip_runtime.h>
#include <cfloat>

#define WAVE 64
#define ROWS_PER_BLOCK 4

#define ENV_CAR 0
#define ENV_DUBINS 1
#define ENV_DRONE 2

extern "C" __global__ void fused_masks(
        const float* __restrict__ states,  // (B*N, S)
        bool* __restrict__ safe,           // (B*n_rec,) or null
        bool* __restrict__ unsafe,         // (B*n_rec,) or null
        bool* __restrict__ collision,      // (B*n_rec,) or null
        int B, int N, int n_rec, int S, float r, int kind) {
    const int row = blockIdx.x * ROWS_PER_BLOCK + threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    if (row >= B * n_rec) return;
    const int b = row / n_rec, i = row % n_rec;
    const int gbase = b * N;
    const int P = (kind == ENV_DRONE) ? 3 : 2;

    const float* si = states + (size_t)(gbase + i) * S;

    // per-row cone direction
    float dir0 = 0.f, dir1 = 0.f, dir2 = 0.f;
    if (kind == ENV_CAR) {
        const float v = sqrtf(si[2] * si[2] + si[3] * si[3]) + 1e-5f;
        dir0 = si[2] / v;
        dir1 = si[3] / v;
    } else if (kind == ENV_DUBINS) {
        dir0 = cosf(si[2]);
        dir1 = sinf(si[2]);
    } else {
        const float v = sqrtf(si[3] * si[3] + si[4] * si[4] +
                              si[5] * si[5]) + 1e-5f;
        dir0 = si[3] / v;
        dir1 = si[4] / v;
        dir2 = si[5];  // reference quirk: vz not normalized
    }

    const float safe_thr = (kind == ENV_DUBINS) ? 3.f * r : 4.f * r;
    const float warn = (kind == ENV_DUBINS) ? 3.f * r : 4.f * r;
    // diagonal offsets per mask (match the reference's +eye*(c+1) exactly)
    const float diag_safe = 4.f * r + 1.f;
    const float diag_unsafe = (kind == ENV_DRONE) ? 2.f * r + 1.f
                                                  : 4.f * r + 1.f;
    const float diag_coll = 2.f * r + 1.f;

    bool all_safe = true;
    bool any_unsafe = false;
    bool any_coll = false;

    for (int j = lane; j < N; j += WAVE) {
        const float* sj = states + (size_t)(gbase + j) * S;
        float pd[3] = {0.f, 0.f, 0.f};
        float d2 = 0.f;
        for (int c = 0; c < P; ++c) {
            pd[c] = si[c] - sj[c];
            d2 += pd[c] * pd[c];
        }
        const float d = sqrtf(d2);
        const float dself = (i == j) ? 1.f : 0.f;  // i==j only when j<n_rec

        const float d_safe = d + dself * diag_safe;
        const float d_uns = d + dself * diag_unsafe;
        const float d_coll = d + dself * diag_coll;

        all_safe &= (d_safe > safe_thr);
        any_coll |= (d_coll < 2.f * r);

        // heading/velocity cone (unsafe direction)
        bool cone = false;
        {
            const float inv = 1.f / (d + 1e-4f);
            // pos_vec = -(pd)/(|pd|+1e-4) : direction i -> j
            const float inner = -(pd[0] * dir0 + pd[1] * dir1 +
                                  pd[2] * dir2) * inv;
            const float ratio = 2.f * r / (d_uns + 1e-7f);
            // ratio > 1 -> asin NaN -> comparison false (matches torch)
            const float thr = cosf(asinf(ratio));
            cone = (inner > thr) && (d_uns < warn);
        }
        any_unsafe |= (d_uns < 2.f * r) || cone;
    }

    const bool w_all_safe = __all(all_safe);
    const bool w_any_uns = __any(any_unsafe);
    const bool w_any_coll = __any(any_coll);
    if (lane == 0) {
        if (safe) safe[row] = w_all_safe;
        if (unsafe) unsafe[row] = w_any_uns;
        if (collision) collision[row] = w_any_coll;
    }
}

extern "C" void launch_fused_masks(const float* states, bool* safe,
                                   bool* unsafe, bool* collision, int B,
                                   int N, int n_rec, int S, float r, int kind,
                                   hipStream_t stream) {
    const int rows = B * n_rec;
    const int blocks = (rows + ROWS_PER_BLOCK - 1) / ROWS_PER_BLOCK;
    hipLaunchKernelGGL(fused_masks, dim3(blocks),
                       dim3(ROWS_PER_BLOCK * WAVE), 0, stream,
                       states, safe, unsafe, collision, B, N, n_rec, S, r,
                       kind);
}
