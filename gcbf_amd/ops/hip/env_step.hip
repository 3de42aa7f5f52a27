// Fused single-graph environment step kernels for CDNA4/gfx950.
//
// One launch per env.step() replaces the reference's ~40 small ATen calls
// (u_ref PID/LQR, action clamp, dynamics, Euler, reach/collision tests,
// reward assembly — reference gcbf/env/dubins_car.py:522-615,
// simple_car.py:146-176, simple_drone.py:191-234).  It also emits the NEXT
// step's u_ref (the trainer attaches u_ref to the new graph immediately, so
// fusing it here makes that attach free).
//
// Key structural fact exploited: in all three envs the position update is
// action-independent (positions integrate the CURRENT velocity/heading), so
// each agent-wave can reconstruct every neighbor's t+1 position on the fly
// for the pairwise collision test — the whole step is one kernel.
//
// Decomposition: one 64-lane wave per agent; obstacles are integrated
// round-robin by the waves after their agent work.
#include <hip/hip_runtime.h>
#include <cfloat>

#define WAVE 64
#define ROWS_PER_BLOCK 4

#define PI_F 3.14159265358979323846f
#define TWO_PI_F 6.28318530717958647692f

__device__ __forceinline__ float pmod2pi(float x) {
    // python-style modulo: result in [0, 2*pi)
    float r = fmodf(x, TWO_PI_F);
    if (r < 0.f) r += TWO_PI_F;
    return r;
}

__device__ __forceinline__ float signf(float x) {
    return (x > 0.f) - (x < 0.f);
}

// ---------------------------------------------------------------- DubinsCar
// states [x, y, th, v]; action [omega/10, a]; obstacles move at constant
// heading/speed.  PID reference controller (reference dubins_car.py:764-816).
__device__ void dubins_u_ref(const float* s, const float* g, float sl,
                             float* out) {
    const float dx = s[0] - g[0], dy = s[1] - g[1];
    const float dist = sqrtf(dx * dx + dy * dy);
    const float k_omega = 0.2f, k_v = 0.3f, k_a = 0.6f;

    float arg = -dx / (dist + 1e-4f);
    arg = fminf(fmaxf(arg, -1.f), 1.f);
    const float theta_t = pmod2pi(acosf(arg) * signf(-dy));
    const float theta = pmod2pi(s[2]);
    const float td = theta_t - theta;
    const float c = cosf(theta), sn = sinf(theta);
    float inner = (-dx) * c + (-dy) * sn;
    inner = fminf(fmaxf(inner / (dist + 1e-4f), -1.f), 1.f);
    const float tb = acosf(inner);

    const bool anti = (td < PI_F) && (td >= 0.f);
    const bool clock = (td > -PI_F) && (td <= 0.f);
    const float sgn = (theta <= PI_F) ? (anti ? 1.f : -1.f)
                                      : (clock ? -1.f : 1.f);
    out[0] = fminf(fmaxf(sgn * k_omega * tb, -5.f), 5.f);

    float a = -k_a * s[3] + k_v * dist;
    if (s[3] > sl) a = fminf(a, 0.f);
    if (s[3] < -sl) a = fmaxf(a, 0.f);
    out[1] = a;
}

// t+1 position of node j (agents freeze on reach; obstacles drift)
__device__ void dubins_next_pos(const float* s, const float* goal, int j,
                                int n, float sl, float d2g, float dt,
                                float* px, float* py) {
    float x = s[0], y = s[1];
    bool frozen = false;
    if (j < n) {
        const float gx = goal[0], gy = goal[1];
        const float dx = x - gx, dy = y - gy;
        frozen = sqrtf(dx * dx + dy * dy) < d2g;
    }
    if (!frozen) {
        const float vc = fminf(s[3], sl);
        x += vc * cosf(s[2]) * dt;
        y += vc * sinf(s[2]) * dt;
    }
    *px = x;
    *py = y;
}

extern "C" __global__ void dubins_step(
        const float* __restrict__ states,   // (N, 4) at t
        const float* __restrict__ goal,     // (n, 4)
        const float* __restrict__ action,   // (n, 2) residual
        float* __restrict__ new_states,     // (N, 4) out
        float* __restrict__ u_ref_next,     // (n, 2) out (for t+1 graph)
        float* __restrict__ reward,         // (n,)  out
        bool* __restrict__ reach,           // (n,)  out (at t+1)
        bool* __restrict__ collision,       // (n,)  out (at t+1)
        int N, int n, float dt, float r_car, float sl, float d2g,
        float act_lim) {
    const int i = blockIdx.x * ROWS_PER_BLOCK + threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int n_waves_total = gridDim.x * ROWS_PER_BLOCK;
    if (i >= n) return;

    const float* si = states + (size_t)i * 4;
    const float* gi = goal + (size_t)i * 4;

    // u_ref at t (bitwise-identical to the previous launch's u_ref_next)
    float ur[2];
    dubins_u_ref(si, gi, sl, ur);
    float u0 = fminf(fmaxf(action[i * 2 + 0] + ur[0], -act_lim), act_lim);
    float u1 = fminf(fmaxf(action[i * 2 + 1] + ur[1], -act_lim), act_lim);

    // prev reach + dynamics (+ freeze) + Euler
    const float pdx = si[0] - gi[0], pdy = si[1] - gi[1];
    const bool prev_reach = sqrtf(pdx * pdx + pdy * pdy) < d2g;
    float ns[4];
    if (prev_reach) {
        ns[0] = si[0]; ns[1] = si[1]; ns[2] = si[2]; ns[3] = si[3];
    } else {
        const float vc = fminf(si[3], sl);
        ns[0] = si[0] + vc * cosf(si[2]) * dt;
        ns[1] = si[1] + vc * sinf(si[2]) * dt;
        ns[2] = si[2] + u0 * 10.f * dt;
        ns[3] = si[3] + u1 * dt;
    }
    if (lane == 0) {
        for (int c = 0; c < 4; ++c) new_states[i * 4 + c] = ns[c];
    }

    // reach at t+1
    const float ndx = ns[0] - gi[0], ndy = ns[1] - gi[1];
    const bool reach_i = sqrtf(ndx * ndx + ndy * ndy) < d2g;

    // pairwise collision at t+1 (reconstruct every j's next position)
    bool any_coll = false;
    for (int j = lane; j < N; j += WAVE) {
        if (j == i) continue;
        float pjx, pjy;
        dubins_next_pos(states + (size_t)j * 4, goal + (size_t)j * 4, j, n,
                        sl, d2g, dt, &pjx, &pjy);
        const float dx = ns[0] - pjx, dy = ns[1] - pjy;
        any_coll |= sqrtf(dx * dx + dy * dy) < 2.f * r_car;
    }
    any_coll = __any(any_coll);

    // scalar action-norm penalty: sum over ALL agents (deterministic:
    // each wave reduces the same full list)
    float asum = 0.f;
    for (int k = lane; k < n; k += WAVE) {
        const float a0 = action[k * 2 + 0], a1 = action[k * 2 + 1];
        asum += sqrtf(a0 * a0 + a1 * a1);
    }
#pragma unroll
    for (int off = WAVE / 2; off > 0; off >>= 1)
        asum += __shfl_down(asum, off, WAVE);
    asum = __shfl(asum, 0, WAVE);

    if (lane == 0) {
        reach[i] = reach_i;
        collision[i] = any_coll;
        reward[i] = ((float)reach_i - (float)prev_reach) * 10.f
                    - (float)any_coll * 0.1f - 1e-4f - 0.01f * asum;
        // u_ref for t+1 from the new state
        float urn[2];
        dubins_u_ref(ns, gi, sl, urn);
        u_ref_next[i * 2 + 0] = urn[0];
        u_ref_next[i * 2 + 1] = urn[1];
    }

    // obstacles: integrated round-robin by agent wave i (constant drift)
    for (int o = n + i; o < N; o += n) {
        if (lane == 0) {
            const float* so = states + (size_t)o * 4;
            const float vc = fminf(so[3], sl);
            new_states[o * 4 + 0] = so[0] + vc * cosf(so[2]) * dt;
            new_states[o * 4 + 1] = so[1] + vc * sinf(so[2]) * dt;
            new_states[o * 4 + 2] = so[2];
            new_states[o * 4 + 3] = so[3];
        }
    }
    (void)n_waves_total;
}

// ---------------------------------------------------------------- SimpleCar
// states [x, y, vx, vy]; action [ax, ay]; LQR u_ref = -K (x - goal) with an
// over-speed penalty (reference simple_car.py:270-304).  No obstacles.
__device__ void car_u_ref(const float* s, const float* g2,
                          const float* K,  // (2, 4) row-major
                          float sl, float* out) {
    float d[4] = {s[0] - g2[0], s[1] - g2[1], s[2], s[3]};
    float a0 = 0.f, a1 = 0.f;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
        a0 -= K[c] * d[c];
        a1 -= K[4 + c] * d[c];
    }
    const float v = sqrtf(s[2] * s[2] + s[3] * s[3]);
    if (v - sl > 0.f) {
        const float scale = (v - sl) * 50.f / v;
        a0 -= scale * s[2];
        a1 -= scale * s[3];
    }
    out[0] = a0;
    out[1] = a1;
}

extern "C" __global__ void car_step(
        const float* __restrict__ states,   // (N, 4)
        const float* __restrict__ goal,     // (N, 2)
        const float* __restrict__ action,   // (N, 2)
        const float* __restrict__ K,        // (2, 4)
        float* __restrict__ new_states, float* __restrict__ u_ref_next,
        float* __restrict__ reward, bool* __restrict__ reach,
        bool* __restrict__ collision,
        int N, float dt, float r_car, float sl, float d2g, float act_lim) {
    const int i = blockIdx.x * ROWS_PER_BLOCK + threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    if (i >= N) return;

    const float* si = states + (size_t)i * 4;
    const float* gi = goal + (size_t)i * 2;

    float ur[2];
    car_u_ref(si, gi, K, sl, ur);
    const float u0 = fminf(fmaxf(action[i * 2] + ur[0], -act_lim), act_lim);
    const float u1 = fminf(fmaxf(action[i * 2 + 1] + ur[1], -act_lim),
                           act_lim);

    const float pdx = si[0] - gi[0], pdy = si[1] - gi[1];
    const bool prev_reach = sqrtf(pdx * pdx + pdy * pdy) < d2g;

    float ns[4];
    ns[0] = si[0] + si[2] * dt;
    ns[1] = si[1] + si[3] * dt;
    ns[2] = si[2] + u0 * dt;
    ns[3] = si[3] + u1 * dt;
    if (lane == 0)
        for (int c = 0; c < 4; ++c) new_states[i * 4 + c] = ns[c];

    const float ndx = ns[0] - gi[0], ndy = ns[1] - gi[1];
    const bool reach_i = sqrtf(ndx * ndx + ndy * ndy) < d2g;

    bool any_coll = false;
    for (int j = lane; j < N; j += WAVE) {
        if (j == i) continue;
        const float* sj = states + (size_t)j * 4;
        const float pjx = sj[0] + sj[2] * dt, pjy = sj[1] + sj[3] * dt;
        const float dx = ns[0] - pjx, dy = ns[1] - pjy;
        any_coll |= sqrtf(dx * dx + dy * dy) < 2.f * r_car;
    }
    any_coll = __any(any_coll);

    if (lane == 0) {
        const float a0 = action[i * 2], a1 = action[i * 2 + 1];
        reach[i] = reach_i;
        collision[i] = any_coll;
        reward[i] = ((float)reach_i - (float)prev_reach) * 4.f
                    - (float)any_coll * 2.f - 0.01f
                    - 1e-4f * sqrtf(a0 * a0 + a1 * a1);
        float urn[2];
        car_u_ref(ns, gi, K, sl, urn);
        u_ref_next[i * 2] = urn[0];
        u_ref_next[i * 2 + 1] = urn[1];
    }
}

// --------------------------------------------------------------- SimpleDrone
// states [x,y,z,vx,vy,vz]; xdot = A x + B u with diagonal damping
// (A: pos<-vel identity, vel damping -1.1/-1.1/-6; B: 1.1/1.1/6);
// LQR u_ref with over-speed penalty *10 (reference simple_drone.py:85-120,
// 349-377).  Obstacles static; agents freeze on reach.
__device__ void drone_u_ref(const float* s, const float* g,
                            const float* K,  // (3, 6) row-major
                            float sl, float* out) {
    float d[6];
#pragma unroll
    for (int c = 0; c < 6; ++c) d[c] = s[c] - g[c];
    float a[3] = {0.f, 0.f, 0.f};
#pragma unroll
    for (int u = 0; u < 3; ++u)
#pragma unroll
        for (int c = 0; c < 6; ++c) a[u] -= K[u * 6 + c] * d[c];
    const float v = sqrtf(s[3] * s[3] + s[4] * s[4] + s[5] * s[5]);
    if (v - sl > 0.f) {
        const float scale = (v - sl) * 10.f / v;
        a[0] -= scale * s[3];
        a[1] -= scale * s[4];
        a[2] -= scale * s[5];
    }
    out[0] = a[0]; out[1] = a[1]; out[2] = a[2];
}

extern "C" __global__ void drone_step(
        const float* __restrict__ states,   // (N, 6)
        const float* __restrict__ goal,     // (n, 6)
        const float* __restrict__ action,   // (n, 3)
        const float* __restrict__ K,        // (3, 6)
        float* __restrict__ new_states, float* __restrict__ u_ref_next,
        float* __restrict__ reward, bool* __restrict__ reach,
        bool* __restrict__ collision,
        int N, int n, float dt, float r_drone, float sl, float d2g,
        float act_lim) {
    const int i = blockIdx.x * ROWS_PER_BLOCK + threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    if (i >= n) return;

    const float* si = states + (size_t)i * 6;
    const float* gi = goal + (size_t)i * 6;

    float ur[3];
    drone_u_ref(si, gi, K, sl, ur);
    float u[3];
#pragma unroll
    for (int c = 0; c < 3; ++c)
        u[c] = fminf(fmaxf(action[i * 3 + c] + ur[c], -act_lim), act_lim);

    const float pdx = si[0] - gi[0], pdy = si[1] - gi[1],
                pdz = si[2] - gi[2];
    const bool prev_reach = sqrtf(pdx * pdx + pdy * pdy + pdz * pdz) < d2g;

    float ns[6];
    if (prev_reach) {
#pragma unroll
        for (int c = 0; c < 6; ++c) ns[c] = si[c];
    } else {
        // xdot = A x + B u
        ns[0] = si[0] + si[3] * dt;
        ns[1] = si[1] + si[4] * dt;
        ns[2] = si[2] + si[5] * dt;
        ns[3] = si[3] + (-1.1f * si[3] + 1.1f * u[0]) * dt;
        ns[4] = si[4] + (-1.1f * si[4] + 1.1f * u[1]) * dt;
        ns[5] = si[5] + (-6.0f * si[5] + 6.0f * u[2]) * dt;
    }
    if (lane == 0)
        for (int c = 0; c < 6; ++c) new_states[i * 6 + c] = ns[c];

    const float ndx = ns[0] - gi[0], ndy = ns[1] - gi[1],
                ndz = ns[2] - gi[2];
    const bool reach_i = sqrtf(ndx * ndx + ndy * ndy + ndz * ndz) < d2g;

    bool any_coll = false;
    for (int j = lane; j < N; j += WAVE) {
        if (j == i) continue;
        const float* sj = states + (size_t)j * 6;
        float pjx = sj[0], pjy = sj[1], pjz = sj[2];
        if (j < n) {
            const float* gj = goal + (size_t)j * 6;
            const float dx = sj[0] - gj[0], dy = sj[1] - gj[1],
                        dz = sj[2] - gj[2];
            if (sqrtf(dx * dx + dy * dy + dz * dz) >= d2g) {
                pjx += sj[3] * dt;
                pjy += sj[4] * dt;
                pjz += sj[5] * dt;
            }
        }  // obstacles static
        const float dx = ns[0] - pjx, dy = ns[1] - pjy, dz = ns[2] - pjz;
        any_coll |= sqrtf(dx * dx + dy * dy + dz * dz) < 2.f * r_drone;
    }
    any_coll = __any(any_coll);

    if (lane == 0) {
        const float a0 = action[i * 3], a1 = action[i * 3 + 1],
                    a2 = action[i * 3 + 2];
        reach[i] = reach_i;
        collision[i] = any_coll;
        reward[i] = ((float)reach_i - (float)prev_reach) * 10.f
                    - (float)any_coll - 0.01f
                    - 1e-3f * sqrtf(a0 * a0 + a1 * a1 + a2 * a2);
        float urn[3];
        drone_u_ref(ns, gi, K, sl, urn);
#pragma unroll
        for (int c = 0; c < 3; ++c) u_ref_next[i * 3 + c] = urn[c];
    }

    // obstacles are static: copy rows round-robin
    for (int o = n + i; o < N; o += n) {
        if (lane == 0)
            for (int c = 0; c < 6; ++c)
                new_states[o * 6 + c] = states[o * 6 + c];
    }
}

// ------------------------------------------------------------- launchers
extern "C" void launch_dubins_step(const float* states, const float* goal,
                                   const float* action, float* new_states,
                                   float* u_ref_next, float* reward,
                                   bool* reach, bool* collision, int N, int n,
                                   float dt, float r, float sl, float d2g,
                                   float act_lim, hipStream_t stream) {
    const int blocks = (n + ROWS_PER_BLOCK - 1) / ROWS_PER_BLOCK;
    hipLaunchKernelGGL(dubins_step, dim3(blocks),
                       dim3(ROWS_PER_BLOCK * WAVE), 0, stream, states, goal,
                       action, new_states, u_ref_next, reward, reach,
                       collision, N, n, dt, r, sl, d2g, act_lim);
}

extern "C" void launch_car_step(const float* states, const float* goal,
                                const float* action, const float* K,
                                float* new_states, float* u_ref_next,
                                float* reward, bool* reach, bool* collision,
                                int N, float dt, float r, float sl, float d2g,
                                float act_lim, hipStream_t stream) {
    const int blocks = (N + ROWS_PER_BLOCK - 1) / ROWS_PER_BLOCK;
    hipLaunchKernelGGL(car_step, dim3(blocks), dim3(ROWS_PER_BLOCK * WAVE),
                       0, stream, states, goal, action, K, new_states,
                       u_ref_next, reward, reach, collision, N, dt, r, sl,
                       d2g, act_lim);
}

extern "C" void launch_drone_step(const float* states, const float* goal,
                                  const float* action, const float* K,
                                  float* new_states, float* u_ref_next,
                                  float* reward, bool* reach,
                                  bool* collision, int N, int n, float dt,
                                  float r, float sl, float d2g,
                                  float act_lim, hipStream_t stream) {
    const int blocks = (n + ROWS_PER_BLOCK - 1) / ROWS_PER_BLOCK;
    hipLaunchKernelGGL(drone_step, dim3(blocks), dim3(ROWS_PER_BLOCK * WAVE),
                       0, stream, states, goal, action, K, new_states,
                       u_ref_next, reward, reach, collision, N, n, dt, r, sl,
                       d2g, act_lim);
}
