// Debug/introspection kernels (not used in production paths): dump the
// staged LDS image, the MFMA fragment registers, and the raw bhi/blo
// accumulator columns of the Gramian EXT tile for kernel bring-up.

#include "common.hip.h"

// re-instantiate the geometry/device functions
#include "als_kernels_device.inc"

// Dump the staged Gt LDS image (one chunk): out[(K+16)][32] bf16.
template <int KT>
__launch_bounds__(256)
__global__ void k_stage_dump(const long long* __restrict__ indptr,
                             const int* __restrict__ indices,
                             const float* __restrict__ values,
                             const unsigned short* __restrict__ factors,
                             unsigned short* __restrict__ out) {
    constexpr int K = Geo<KT>::K;
    constexpr int TROW = Geo<KT>::TROW;
    __shared__ __align__(16) char smem[Geo<KT>::SMEM];
    const long long p0 = indptr[0];
    const int n = (int)(indptr[1] - p0);
    stage_zero_pad_all<KT, false, 1>(smem);
    if (threadIdx.x < 64)
        stage_chunk_w<KT, false>(smem, indices, values, factors, p0, n,
                                 threadIdx.x);
    __syncthreads();
    for (int i = threadIdx.x; i < (K + 16) * 32; i += 256) {
        const int row = i / 32, col = i % 32;
        out[i] = *(const unsigned short*)(smem + (long long)row * TROW
                                          + col * 2);
    }
}

// Dump every wave's fragment registers: out[wave][tile][lane][j] bf16.
template <int KT>
__launch_bounds__(256)
__global__ void k_frag_dump(const long long* __restrict__ indptr,
                            const int* __restrict__ indices,
                            const float* __restrict__ values,
                            const unsigned short* __restrict__ factors,
                            unsigned short* __restrict__ out) {
    __shared__ __align__(16) char smem[Geo<KT>::SMEM];
    const long long p0 = indptr[0];
    const int n = (int)(indptr[1] - p0);
    stage_zero_pad_all<KT, false, 1>(smem);
    if (threadIdx.x < 64)
        stage_chunk_w<KT, false>(smem, indices, values, factors, p0, n,
                                 threadIdx.x);
    __syncthreads();
    const int lane = threadIdx.x & 63, w = threadIdx.x >> 6;
    bf16x8 frag[KT + 1];
    read_frags<KT, false>(smem, lane, frag);
    for (int t = 0; t <= KT; ++t)
        for (int j = 0; j < 8; ++j)
            out[((w * (KT + 1) + t) * 64 + lane) * 8 + j] =
                (unsigned short)frag[t][j];
}

// Gramian with separated b columns: bhi -> b_out, blo -> blo_out.
template <int KT>
__launch_bounds__(256)
__global__ void k_gramian_dbg(const long long* __restrict__ indptr,
                              const int* __restrict__ indices,
                              const float* __restrict__ values,
                              const unsigned short* __restrict__ factors,
                              float* __restrict__ A_out,
                              float* __restrict__ bhi_out,
                              float* __restrict__ blo_out,
                              long long nrows, float reg) {
    constexpr int K = Geo<KT>::K;
    __shared__ __align__(16) char smem[Geo<KT>::SMEM];
    const long long row = blockIdx.x;
    if (row >= nrows) return;
    const int tid = threadIdx.x;
    const int n = gramian_to_lds<KT, /*combine=*/false>(
        smem, indptr, indices, values, factors, row, reg);
    float* A = (float*)smem;
    float* bhi = A + K * (K + 1);
    float* blo = bhi + K;
    if (n == 0) return;
    for (int i = tid; i < K * K; i += 256)
        A_out[row * K * K + i] = A[(i / K) * (K + 1) + (i % K)];
    for (int c = tid; c < K; c += 256) {
        bhi_out[row * K + c] = bhi[c];
        blo_out[row * K + c] = blo[c];
    }
}

extern "C" hipError_t fma_dbg_stage_dump(
    int k, const long long* indptr, const int* indices, const float* values,
    const unsigned short* factors, unsigned short* out, hipStream_t stream) {
    DISPATCH_KT(k, (k_stage_dump<KT><<<dim3(1), dim3(256), 0, stream>>>(
        indptr, indices, values, factors, out)));
    return hipGetLastError();
}

extern "C" hipError_t fma_dbg_frag_dump(
    int k, const long long* indptr, const int* indices, const float* values,
    const unsigned short* factors, unsigned short* out, hipStream_t stream) {
    DISPATCH_KT(k, (k_frag_dump<KT><<<dim3(1), dim3(256), 0, stream>>>(
        indptr, indices, values, factors, out)));
    return hipGetLastError();
}

extern "C" hipError_t fma_dbg_gramian(
    int k, const long long* indptr, const int* indices, const float* values,
    const unsigned short* factors, float* A_out, float* bhi_out,
    float* blo_out, long long nrows, float reg, hipStream_t stream) {
    DISPATCH_KT(k, (k_gramian_dbg<KT><<<dim3((unsigned)nrows), dim3(256), 0,
                                        stream>>>(
        indptr, indices, values, factors, A_out, bhi_out, blo_out, nrows,
        reg)));
    return hipGetLastError();
}
