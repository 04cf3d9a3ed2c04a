// CoCoA-SVM local-solver kernels for MI355X (gfx950).
//
// Rebuilds flink-ml's CoCoA localSDCA inner loop (driven by the reference's
// flink-svm/src/main/scala/de/tub/it4bi/SVMImpl.scala:24-29; SURVEY.md §2.5
// K3) as a wave-per-sample hinge-loss dual coordinate ascent:
//
//   per sample i:  g      = (1 - y_i <v, x_i>) / (||x_i||^2 / (lambda n))
//                  dalpha = clip_[0,1](alpha_i + g) - alpha_i
//                  alpha_i += dalpha ;  v += dalpha * y_i * x_i / (lambda n)
//
// v is the local primal image (w + local delta, fp32 dense); updates are
// asynchronous ("hogwild") across the in-flight waves — the MI355X-native
// replacement for flink-ml's sequential per-block pass, with the same
// convergence contract CoCoA needs (an inexact local solver).  Each wave
// owns a disjoint sample subset, so the alpha[i] read-modify-write itself is
// race-free; only v is shared.
//
// xGMI note: the outer CoCoA aggregate (reference: reduce of per-block
// deltas) runs as an RCCL all-reduce in flink_ms_amd/parallel/.

#include "common.hip.h"

__launch_bounds__(256)
__global__ void k_sdca_pass(const long long* __restrict__ indptr,
                            const int* __restrict__ indices,
                            const float* __restrict__ values,
                            const float* __restrict__ y,
                            const float* __restrict__ norms_sq,
                            const int* __restrict__ perm,  // may be null
                            float* __restrict__ alpha,
                            float* __restrict__ v,
                            long long nrows, float scale /* 1/(lambda*n) */) {
    const int lane = threadIdx.x & 63;
    const long long wave = (long long)blockIdx.x * 4 + (threadIdx.x >> 6);
    const long long nwaves = (long long)gridDim.x * 4;
    if (wave >= nrows) return;

    // The per-sample chain is 3-4 dependent memory round trips (perm ->
    // indptr -> indices -> v); prefetching the NEXT sample's metadata and
    // first 64 nonzeros while the current sample computes hides most of it
    // (the v gather itself stays at compute time for hogwild freshness).
    struct Meta {
        long long i, e0, e1;
        float nsq, yi, a, val;
        int idx;
    };
    auto fetch = [&](long long s, Meta& m) {
        if (s >= nrows) { m.e0 = m.e1 = 0; m.idx = -1; return; }
        m.i = perm ? (long long)perm[s] : s;
        m.e0 = indptr[m.i];
        m.e1 = indptr[m.i + 1];
        m.nsq = norms_sq[m.i];
        m.yi = y[m.i];
        m.a = alpha[m.i];
        const long long t = m.e0 + lane;
        m.idx = (t < m.e1) ? indices[t] : -1;
        m.val = (t < m.e1) ? values[t] : 0.0f;
    };

    Meta cur, nxt;
    fetch(wave, cur);
    for (long long s = wave; s < nrows; s = s + nwaves) {
        fetch(s + nwaves, nxt);   // in flight across the current compute
        if (cur.e1 > cur.e0 && cur.nsq != 0.0f) {
            float part = (cur.idx >= 0) ? cur.val * v[cur.idx] : 0.0f;
            for (long long t = cur.e0 + 64 + lane; t < cur.e1; t += WAVE)
                part += values[t] * v[indices[t]];
            const float dot = wave_reduce_sum(part);
            const float grad = (1.0f - cur.yi * dot) / (cur.nsq * scale);
            float a_new = cur.a + grad;
            a_new = a_new < 0.0f ? 0.0f : (a_new > 1.0f ? 1.0f : a_new);
            const float dalpha = a_new - cur.a;
            if (dalpha != 0.0f) {
                if (lane == 0) alpha[cur.i] = a_new;
                const float c = dalpha * cur.yi * scale;
                if (cur.idx >= 0) atomicAdd(&v[cur.idx], c * cur.val);
                for (long long t = cur.e0 + 64 + lane; t < cur.e1; t += WAVE)
                    atomicAdd(&v[indices[t]], c * values[t]);
            }
        }
        cur = nxt;
    }
}

// Batched decision values <w, x_i> (evaluation / serving batch scoring).
__launch_bounds__(256)
__global__ void k_svm_margins(const long long* __restrict__ indptr,
                              const int* __restrict__ indices,
                              const float* __restrict__ values,
                              const float* __restrict__ w,
                              float* __restrict__ out,
                              long long nrows) {
    const int lane = threadIdx.x & 63;
    const long long wave = (long long)blockIdx.x * 4 + (threadIdx.x >> 6);
    const long long nwaves = (long long)gridDim.x * 4;
    for (long long i = wave; i < nrows; i += nwaves) {
        float part = 0.0f;
        for (long long t = indptr[i] + lane; t < indptr[i + 1]; t += WAVE)
            part += values[t] * w[indices[t]];
        const float dot = wave_reduce_sum(part);
        if (lane == 0) out[i] = dot;
    }
}

extern "C" hipError_t fma_sdca_pass(
    const long long* indptr, const int* indices, const float* values,
    const float* y, const float* norms_sq, const int* perm, float* alpha,
    float* v, long long nrows, float scale, hipStream_t stream) {
    if (nrows <= 0) return hipErrorInvalidValue;
    // keep >=16 samples sequential per wave: full-width hogwild degenerates
    // to a synchronous full-batch step and stalls dual convergence
    long long waves = nrows / 16;
    if (waves < 4) waves = 4;
    if (waves > 8192) waves = 8192;
    unsigned grid = (unsigned)((waves + 3) / 4);
    k_sdca_pass<<<dim3(grid), dim3(256), 0, stream>>>(
        indptr, indices, values, y, norms_sq, perm, alpha, v, nrows, scale);
    return hipGetLastError();
}

extern "C" hipError_t fma_svm_margins(
    const long long* indptr, const int* indices, const float* values,
    const float* w, float* out, long long nrows, hipStream_t stream) {
    if (nrows <= 0) return hipErrorInvalidValue;
    long long waves = (nrows + 3) / 4;
    unsigned grid = (unsigned)(waves < 2048 ? waves : 2048);
    k_svm_margins<<<dim3(grid), dim3(256), 0, stream>>>(
        indptr, indices, values, w, out, nrows);
    return hipGetLastError();
}
