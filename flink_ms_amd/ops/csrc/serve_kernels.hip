// Serving-path kernels for MI355X (gfx950): batched factor dots and online
// SGD updates against the GPU-resident model store.
//
//   K5: score(u,i) = dot(U[u], V[i])   (reference ALSPredict.java:74-83,
//       MSE.java:150-154, ALSPredictRandom.java:89-92)
//   K4: online SGD step (reference SGD.java:182-207, v1 "simultaneous"
//       semantics: both factor updates computed from the OLD vectors).
//
// One wave per query; factors are bf16 rows of the store, math in fp32.

#include "common.hip.h"

__launch_bounds__(256)
__global__ void k_predict_dot(const unsigned short* __restrict__ U,
                              const unsigned short* __restrict__ V,
                              const long long* __restrict__ u_idx,
                              const long long* __restrict__ i_idx,
                              float* __restrict__ out,
                              long long nq, int k) {
    const int lane = threadIdx.x & 63;
    const long long wave = (long long)blockIdx.x * 4 + (threadIdx.x >> 6);
    const long long nwaves = (long long)gridDim.x * 4;
    for (long long q = wave; q < nq; q += nwaves) {
        const unsigned short* up = U + u_idx[q] * (long long)k;
        const unsigned short* vp = V + i_idx[q] * (long long)k;
        float part = 0.0f;
        for (int c = lane; c < k; c += WAVE)
            part += bf2f(up[c]) * bf2f(vp[c]);
        const float dot = wave_reduce_sum(part);
        if (lane == 0) out[q] = dot;
    }
}

// In-place SGD factor update.  Collisions inside a batch follow
// last-writer-wins, the same unsynchronized read-modify-write contract the
// reference's SGD/serving-job cycle has.
__launch_bounds__(256)
__global__ void k_sgd_update(unsigned short* __restrict__ U,
                             unsigned short* __restrict__ V,
                             const long long* __restrict__ u_idx,
                             const long long* __restrict__ i_idx,
                             const float* __restrict__ r,
                             float* __restrict__ err_out,
                             long long nq, int k, float lr,
                             float user_reg, float item_reg) {
    const int lane = threadIdx.x & 63;
    const long long wave = (long long)blockIdx.x * 4 + (threadIdx.x >> 6);
    const long long nwaves = (long long)gridDim.x * 4;
    for (long long q = wave; q < nq; q += nwaves) {
        unsigned short* up = U + u_idx[q] * (long long)k;
        unsigned short* vp = V + i_idx[q] * (long long)k;
        // read old vectors (each lane owns columns lane, lane+64, ...)
        float pu[2] = {0, 0}, qi[2] = {0, 0};
        float part = 0.0f;
#pragma unroll 2
        for (int c = lane, s = 0; c < k; c += WAVE, ++s) {
            pu[s] = bf2f(up[c]);
            qi[s] = bf2f(vp[c]);
            part += pu[s] * qi[s];
        }
        const float err = r[q] - wave_reduce_sum(part);
        if (lane == 0 && err_out) err_out[q] = err;
#pragma unroll 2
        for (int c = lane, s = 0; c < k; c += WAVE, ++s) {
            const float pn = pu[s] + lr * (err * qi[s] - user_reg * pu[s]);
            const float qn = qi[s] + lr * (err * pu[s] - item_reg * qi[s]);
            up[c] = f2bf(pn);
            vp[c] = f2bf(qn);
        }
    }
}

extern "C" hipError_t fma_predict_dot(
    const unsigned short* U, const unsigned short* V, const long long* u_idx,
    const long long* i_idx, float* out, long long nq, int k,
    hipStream_t stream) {
    if (nq <= 0 || k <= 0 || k > 128) return hipErrorInvalidValue;
    long long waves = (nq + 3) / 4;
    unsigned grid = (unsigned)(waves < 2048 ? waves : 2048);
    k_predict_dot<<<dim3(grid), dim3(256), 0, stream>>>(
        U, V, u_idx, i_idx, out, nq, k);
    return hipGetLastError();
}

extern "C" hipError_t fma_sgd_update(
    unsigned short* U, unsigned short* V, const long long* u_idx,
    const long long* i_idx, const float* r, float* err_out, long long nq,
    int k, float lr, float user_reg, float item_reg, hipStream_t stream) {
    if (nq <= 0 || k <= 0 || k > 128) return hipErrorInvalidValue;
    long long waves = (nq + 3) / 4;
    unsigned grid = (unsigned)(waves < 2048 ? waves : 2048);
    k_sgd_update<<<dim3(grid), dim3(256), 0, stream>>>(
        U, V, u_idx, i_idx, r, err_out, nq, k, lr, user_reg, item_reg);
    return hipGetLastError();
}
