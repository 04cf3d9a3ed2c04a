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
    // r2: 16-LANE GROUPS, 4 concurrent samples per wave.  The r1 wave-
    // per-sample shape wasted ~45% of its lanes on RCV1-length rows
    // (74 nnz vs 64 lanes: a full second round with 10 live lanes) and
    // exposed every gather round trip; groups cover a 74-nnz row in 5
    // ~93%-utilized rounds and quadruple the samples in flight per wave.
    // Still hogwild: groups own disjoint sample subsets, only v is shared.
    const int lane = threadIdx.x & 63;
    const int gl = lane & 15;
    const long long grp =
        ((long long)blockIdx.x * 4 + (threadIdx.x >> 6)) * 4 + (lane >> 4);
    const long long ngrps = (long long)gridDim.x * 16;
    if (grp >= nrows) return;
    for (long long s = grp; s < nrows; s += ngrps) {
        const long long i = perm ? (long long)perm[s] : s;
        const long long e0 = indptr[i];
        const long long e1 = indptr[i + 1];
        const float nsq = norms_sq[i];
        if (e1 <= e0 || nsq == 0.0f) continue;
        float part = 0.0f;
        for (long long t = e0 + gl; t < e1; t += 16)
            part += values[t] * v[indices[t]];
#pragma unroll
        for (int off = 8; off; off >>= 1)
            part += __shfl_xor(part, off, 16);
        const float yi = y[i];
        const float grad = (1.0f - yi * part) / (nsq * scale);
        const float a = alpha[i];
        float a_new = a + grad;
        a_new = a_new < 0.0f ? 0.0f : (a_new > 1.0f ? 1.0f : a_new);
        const float dalpha = a_new - a;
        if (dalpha != 0.0f) {
            if (gl == 0) alpha[i] = a_new;
            const float c = dalpha * yi * scale;
            for (long long t = e0 + gl; t < e1; t += 16)
                atomicAdd(&v[indices[t]], c * values[t]);
        }
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
    // keep >=16 samples sequential per GROUP: full-width hogwild
    // degenerates to a synchronous full-batch step and stalls dual
    // convergence.  4 groups per wave, 4 waves per block.
    long long groups = nrows / 16;
    if (groups < 16) groups = 16;
    if (groups > 65536) groups = 65536;
    unsigned grid = (unsigned)((groups + 15) / 16);
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
