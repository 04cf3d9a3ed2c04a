// ALS hot-path kernels for MI355X (gfx950, CDNA4).
//
// Rebuilds the compute core of flink-ml's blocked ALS (driven by the
// reference's flink-als/src/main/scala/de/tub/it4bi/ALSImpl.scala:47-52) as
// MI355X-native HIP:
//
//   K1  per-entity normal equations  A_u = sum_{i in R(u)} q_i q_i^T
//       + lambda*n_u*I,  b_u = sum r_ui q_i
//       -> MFMA rank-32 updates: the entity's rated factor rows are staged
//          through LDS in 32-row chunks and contracted with
//          v_mfma_f32_16x16x32_bf16 into 16x16 fp32 accumulator tiles
//          (upper triangle only; A is symmetric).  The rating value rides in
//          an extra staged 16-column block as a bf16 hi/lo pair, so b falls
//          out of the same MFMAs (columns 0/1 of the EXT tiles).
//   K2  p_u = A_u^{-1} b_u
//       -> fused in-LDS Cholesky factorization + wave-level triangular
//          solves, so A never round-trips through HBM.
//
// One workgroup (256 threads = 4 waves) per entity; grid = #entities.
// Wave w owns accumulator tiles t = w, w+4, w+8, ... of the upper-triangle +
// EXT tile list.  Factor rank k = 16*KT, KT in 1..8 (wrappers pad).
//
// LDS layout per block (union; stage dies before A is born):
//   stage: [32][SP] bf16 rows (SP = roundup(K+32,32)), columns 0..K-1 the
//          gathered factor row, K..K+15 the [r_hi, r_lo, 0...] rating block.
//          Byte addresses are XOR'd by 32 on rows with bit 3 set so the two
//          16-lane halves of a b16 lane-group land on disjoint banks
//          (cdna_hip_programming.md §6 Guideline 4).
//   A:     [K][K+1] fp32 (padded leading dim -> conflict-free column walks)
//   b_hi/b_lo: [K] fp32 each, after A.

#include "common.hip.h"

// ---------------------------------------------------------------- geometry

template <int KT> struct Geo {
    static constexpr int K = 16 * KT;
    static constexpr int SP = ((K + 32 + 31) / 32) * 32;     // stage row stride (elements)
    static constexpr int NA = KT * (KT + 1) / 2;             // upper-triangle tiles
    static constexpr int TILES = NA + KT;                    // + EXT (b) tiles
    static constexpr int SLOTS = (TILES + 3) / 4;            // acc tiles per wave
    static constexpr int STAGE_BYTES = 32 * SP * 2;
    static constexpr int A_BYTES = K * (K + 1) * 4;
    static constexpr int SMEM = (STAGE_BYTES > A_BYTES + 8 * K)
                                    ? STAGE_BYTES : A_BYTES + 8 * K;
};

DEV_INLINE unsigned stage_xor(int row) { return (row & 8) ? 32u : 0u; }

// ------------------------------------------------------------------ stage
// Gather the chunk's 32 factor rows (+ rating hi/lo block) into LDS.
// nrem = ratings left in this entity (rows >= nrem are zero-filled).
template <int KT>
DEV_INLINE void stage_chunk(char* smem,
                            const int* __restrict__ indices,
                            const float* __restrict__ values,
                            const unsigned short* __restrict__ factors,
                            long long base, int nrem) {
    constexpr int K = Geo<KT>::K, SP = Geo<KT>::SP;
    constexpr int LPR = K / 8;              // 16B loads per factor row
    const int tid = threadIdx.x;
    // factor-row tasks
    for (int t = tid; t < 32 * LPR; t += 256) {
        const int row = t / LPR, seg = t % LPR;
        uint4 v = {0, 0, 0, 0};
        if (row < nrem) {
            const long long col = indices[base + row];
            v = *(const uint4*)(factors + col * (long long)K + seg * 8);
        }
        unsigned byte = (unsigned)(row * SP + seg * 8) * 2u ^ stage_xor(row);
        *(uint4*)(smem + byte) = v;
    }
    // rating EXT block: columns K..K+15 = [hi, lo, 0 x14]
    for (int t = tid; t < 64; t += 256) {
        const int row = t >> 1, half = t & 1;
        uint4 v = {0, 0, 0, 0};
        if (half == 0 && row < nrem) {
            const float r = values[base + row];
            const unsigned short hi = f2bf(r);
            const unsigned short lo = f2bf(r - bf2f(hi));
            v.x = (unsigned)hi | ((unsigned)lo << 16);
        }
        unsigned byte = (unsigned)(row * SP + K + half * 8) * 2u ^ stage_xor(row);
        *(uint4*)(smem + byte) = v;
    }
}

// ------------------------------------------------------------- frag reads
// A- and B-operand fragments of v_mfma_f32_16x16x32_bf16 share one layout
// here: lane l holds G[(l>>4)*8 + j][tile*16 + (l&15)] for j=0..7.  The
// Gramian sums over the staged-row (contraction) axis, so any consistent
// lane->k mapping of the hardware yields the same A (both operands use the
// same map); only the 16-row/16-col lane maps and the C/D layout
// (col=lane&15, row=(lane>>4)*4+reg: cdna_hip_programming.md §3) must match
// the hardware, which tests/test_gpu_mfma.py verifies.
template <int KT>
DEV_INLINE void read_frags(const char* smem, int lane, bf16x8* frag) {
    constexpr int SP = Geo<KT>::SP;
    const int g = lane >> 4, li = lane & 15;
    const unsigned xorb = (g & 1) ? 32u : 0u;
#pragma unroll
    for (int t = 0; t <= KT; ++t) {
        unsigned short e[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            unsigned byte = (unsigned)((8 * g + j) * SP + t * 16 + li) * 2u;
            e[j] = *(const unsigned short*)(smem + (byte ^ xorb));
        }
        bf16x8 f;
#pragma unroll
        for (int j = 0; j < 8; ++j) f[j] = (short)e[j];
        frag[t] = f;
    }
}

// --------------------------------------------------------------- MFMA loop

template <int KT, int W, int S = 0>
DEV_INLINE void mfma_tiles(const bf16x8* frag, f32x4* acc) {
    if constexpr (S < Geo<KT>::SLOTS) {
        constexpr int t = W + 4 * S;
        if constexpr (t < Geo<KT>::TILES) {
            constexpr int NA = Geo<KT>::NA;
            constexpr int p = (t < NA) ? up_tile_p(t, KT) : (t - NA);
            constexpr int q = (t < NA) ? up_tile_q((t < NA) ? t : 0, KT) : KT;
            acc[S] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                frag[p], frag[q], acc[S], 0, 0, 0);
        }
        mfma_tiles<KT, W, S + 1>(frag, acc);
    }
}

// Spill accumulators to the LDS A/b images (C/D map: row=(l>>4)*4+r, col=l&15).
template <int KT, int W, int S = 0>
DEV_INLINE void write_acc(const f32x4* acc, float* A, float* bhi, float* blo,
                          int lane) {
    if constexpr (S < Geo<KT>::SLOTS) {
        constexpr int t = W + 4 * S;
        if constexpr (t < Geo<KT>::TILES) {
            constexpr int K = Geo<KT>::K;
            constexpr int NA = Geo<KT>::NA;
            constexpr int p = (t < NA) ? up_tile_p(t, KT) : (t - NA);
            constexpr int q = (t < NA) ? up_tile_q((t < NA) ? t : 0, KT) : KT;
            const int g = lane >> 4, li = lane & 15;
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int row = p * 16 + g * 4 + r;
                if constexpr (q == KT) {       // EXT: b tile, cols 0/1 = hi/lo
                    if (li == 0) bhi[row] = acc[S][r];
                    else if (li == 1) blo[row] = acc[S][r];
                } else {
                    const int col = q * 16 + li;
                    A[row * (K + 1) + col] = acc[S][r];
                    if constexpr (p != q) A[col * (K + 1) + row] = acc[S][r];
                }
            }
        }
        write_acc<KT, W, S + 1>(acc, A, bhi, blo, lane);
    }
}

// ------------------------------------------------------ in-LDS Cholesky/solve

// Right-looking Cholesky of the [K][K+1] LDS image, lower triangle in place.
// Thread (r = tid%K, s = tid/K) owns row r, column-slice s.
template <int K>
DEV_INLINE void cholesky_lds(float* A) {
    constexpr int NSL = 256 / K;
    const int tid = threadIdx.x;
    const int r = tid % K, s = tid / K;
    const bool owner = tid < NSL * K;
    for (int j = 0; j < K; ++j) {
        const float d = sqrtf(A[j * (K + 1) + j]);
        const float dinv = 1.0f / d;
        if (owner && s == 0 && r > j) A[r * (K + 1) + j] *= dinv;
        __syncthreads();
        if (owner && s == 0 && r == j) A[j * (K + 1) + j] = d;
        if (owner && r > j) {
            const float lrj = A[r * (K + 1) + j];
            for (int c = j + 1 + s; c <= r; c += NSL)
                A[r * (K + 1) + c] -= lrj * A[c * (K + 1) + j];
        }
        __syncthreads();
    }
}

// Triangular solves L L^T x = b on wave 0; lane owns rows lane and lane+64.
// Returns x in (x0, x1).
template <int K>
DEV_INLINE void solve_lds(const float* A, const float* b, int lane,
                          float& x0, float& x1) {
    const bool lv = lane < K;                       // low row valid
    const bool hv = (K > 64) && (lane + 64 < K);    // high row valid
    x0 = lv ? b[lane] : 0.0f;
    x1 = hv ? b[lane + 64] : 0.0f;
    const float id0 = lv ? 1.0f / A[lane * (K + 1) + lane] : 0.0f;
    const float id1 = hv ? 1.0f / A[(lane + 64) * (K + 1) + lane + 64] : 0.0f;
    // forward: L y = b
    for (int j = 0; j < K; ++j) {
        const bool hi = j >= 64;
        const float yj = __shfl(hi ? x1 : x0, j & 63, WAVE)
                       * __shfl(hi ? id1 : id0, j & 63, WAVE);
        if (!hi) {
            if (lane == j) x0 = yj;
            else if (lane > j && lv) x0 -= A[lane * (K + 1) + j] * yj;
            if (hv) x1 -= A[(lane + 64) * (K + 1) + j] * yj;
        } else {
            if (lane + 64 == j) x1 = yj;
            else if (lane + 64 > j && hv) x1 -= A[(lane + 64) * (K + 1) + j] * yj;
        }
    }
    // backward: L^T x = y  (L^T[r][j] = A[j][r])
    for (int j = K - 1; j >= 0; --j) {
        const bool hi = j >= 64;
        const float xj = __shfl(hi ? x1 : x0, j & 63, WAVE)
                       * __shfl(hi ? id1 : id0, j & 63, WAVE);
        if (hi) {
            if (lane + 64 == j) x1 = xj;
            else if (lane + 64 < j && hv) x1 -= A[j * (K + 1) + lane + 64] * xj;
            if (lv) x0 -= A[j * (K + 1) + lane] * xj;
        } else {
            if (lane == j) x0 = xj;
            else if (lane < j && lv) x0 -= A[j * (K + 1) + lane] * xj;
        }
    }
}

// ------------------------------------------------- Gramian accumulation core
// Runs the chunked stage->frag->MFMA loop and leaves A (mirrored, with
// lambda*n*I) + combined b in LDS.  Returns n (ratings of this entity).
template <int KT>
DEV_INLINE int gramian_to_lds(char* smem,
                              const long long* __restrict__ indptr,
                              const int* __restrict__ indices,
                              const float* __restrict__ values,
                              const unsigned short* __restrict__ factors,
                              long long row, float reg) {
    constexpr int K = Geo<KT>::K;
    const int tid = threadIdx.x, lane = tid & 63, w = tid >> 6;
    const long long p0 = indptr[row];
    const int n = (int)(indptr[row + 1] - p0);
    if (n == 0) return 0;

    f32x4 acc[Geo<KT>::SLOTS];
#pragma unroll
    for (int s = 0; s < Geo<KT>::SLOTS; ++s) acc[s] = f32x4{0, 0, 0, 0};

    const int nchunks = (n + 31) >> 5;
    for (int ch = 0; ch < nchunks; ++ch) {
        stage_chunk<KT>(smem, indices, values, factors, p0 + ch * 32,
                        n - ch * 32);
        __syncthreads();
        bf16x8 frag[KT + 1];
        read_frags<KT>(smem, lane, frag);
        switch (w) {
            case 0: mfma_tiles<KT, 0>(frag, acc); break;
            case 1: mfma_tiles<KT, 1>(frag, acc); break;
            case 2: mfma_tiles<KT, 2>(frag, acc); break;
            default: mfma_tiles<KT, 3>(frag, acc); break;
        }
        __syncthreads();
    }

    float* A = (float*)smem;
    float* bhi = A + K * (K + 1);
    float* blo = bhi + K;
    switch (w) {
        case 0: write_acc<KT, 0>(acc, A, bhi, blo, lane); break;
        case 1: write_acc<KT, 1>(acc, A, bhi, blo, lane); break;
        case 2: write_acc<KT, 2>(acc, A, bhi, blo, lane); break;
        default: write_acc<KT, 3>(acc, A, bhi, blo, lane); break;
    }
    __syncthreads();
    if (tid < K) {
        bhi[tid] += blo[tid];                       // combined b
        float dd = A[tid * (K + 1) + tid] + reg * (float)n;
        if (dd <= 0.0f) dd = 1.0f;                  // degenerate guard
        A[tid * (K + 1) + tid] = dd;
    }
    __syncthreads();
    return n;
}

template <int KT>
DEV_INLINE void write_zero_row(float* out_f32, unsigned short* out_bf16,
                               long long row) {
    constexpr int K = Geo<KT>::K;
    for (int c = threadIdx.x; c < K; c += 256) {
        out_f32[row * K + c] = 0.0f;
        if (out_bf16) out_bf16[row * K + c] = 0;
    }
}

// -------------------------------------------------------------- kernels

// Fused K1+K2: normal equations + Cholesky solve, one entity per block.
template <int KT>
__launch_bounds__(256)
__global__ void k_als_solve_fused(const long long* __restrict__ indptr,
                                  const int* __restrict__ indices,
                                  const float* __restrict__ values,
                                  const unsigned short* __restrict__ factors,
                                  float* __restrict__ out_f32,
                                  unsigned short* __restrict__ out_bf16,
                                  const int* __restrict__ row_order,
                                  long long nrows, float reg) {
    constexpr int K = Geo<KT>::K;
    __shared__ __align__(16) char smem[Geo<KT>::SMEM];
    long long row = blockIdx.x;
    if (row >= nrows) return;
    if (row_order) row = row_order[row];

    const int n = gramian_to_lds<KT>(smem, indptr, indices, values, factors,
                                     row, reg);
    if (n == 0) { write_zero_row<KT>(out_f32, out_bf16, row); return; }

    float* A = (float*)smem;
    float* b = A + K * (K + 1);
    cholesky_lds<K>(A);

    const int lane = threadIdx.x & 63;
    if (threadIdx.x < 64) {
        float x0, x1;
        solve_lds<K>(A, b, lane, x0, x1);
        if (lane < K) {
            out_f32[row * K + lane] = x0;
            if (out_bf16) out_bf16[row * K + lane] = f2bf(x0);
        }
        if (K > 64 && lane + 64 < K) {
            out_f32[row * K + lane + 64] = x1;
            if (out_bf16) out_bf16[row * K + lane + 64] = f2bf(x1);
        }
    }
}

// Standalone K1 (for parity tests / modular path): writes dense A and b.
template <int KT>
__launch_bounds__(256)
__global__ void k_gramian(const long long* __restrict__ indptr,
                          const int* __restrict__ indices,
                          const float* __restrict__ values,
                          const unsigned short* __restrict__ factors,
                          float* __restrict__ A_out,   // [nrows][K][K]
                          float* __restrict__ b_out,   // [nrows][K]
                          long long nrows, float reg) {
    constexpr int K = Geo<KT>::K;
    __shared__ __align__(16) char smem[Geo<KT>::SMEM];
    const long long row = blockIdx.x;
    if (row >= nrows) return;
    const int tid = threadIdx.x;

    const int n = gramian_to_lds<KT>(smem, indptr, indices, values, factors,
                                     row, reg);
    float* A = (float*)smem;
    float* b = A + K * (K + 1);
    if (n == 0) {
        for (int i = tid; i < K * K; i += 256) A_out[row * K * K + i] = 0.0f;
        for (int c = tid; c < K; c += 256) b_out[row * K + c] = 0.0f;
        return;
    }
    for (int i = tid; i < K * K; i += 256)
        A_out[row * K * K + i] = A[(i / K) * (K + 1) + (i % K)];
    for (int c = tid; c < K; c += 256) b_out[row * K + c] = b[c];
}

// Standalone K2: batched SPD solve from dense global A/b.
template <int KT>
__launch_bounds__(256)
__global__ void k_cholesky_solve(const float* __restrict__ A_in,  // [n][K][K]
                                 const float* __restrict__ b_in,  // [n][K]
                                 float* __restrict__ x_out,       // [n][K]
                                 long long nrows) {
    constexpr int K = Geo<KT>::K;
    __shared__ __align__(16) char smem[Geo<KT>::SMEM];
    const long long row = blockIdx.x;
    if (row >= nrows) return;
    float* A = (float*)smem;
    float* b = A + K * (K + 1);
    const int tid = threadIdx.x;
    for (int i = tid; i < K * K; i += 256)
        A[(i / K) * (K + 1) + (i % K)] = A_in[row * K * K + i];
    for (int c = tid; c < K; c += 256) b[c] = b_in[row * K + c];
    __syncthreads();
    cholesky_lds<K>(A);
    if (tid < 64) {
        float x0, x1;
        solve_lds<K>(A, b, tid, x0, x1);
        if (tid < K) x_out[row * K + tid] = x0;
        if (K > 64 && tid + 64 < K) x_out[row * K + tid + 64] = x1;
    }
}

// ------------------------------------------------------------- MFMA probes
// Layout-validation kernels for tests/test_gpu_mfma.py.

// f32 16x16x4 probe with the guide-documented operand maps
// (cdna_hip_programming.md §3): definitive C/D-layout check.
__global__ void k_mfma_probe_f32(const float* __restrict__ Amat,  // [16][4]
                                 const float* __restrict__ Bmat,  // [4][16]
                                 float* __restrict__ D) {         // [16][16]
    const int lane = threadIdx.x;
    const float a = Amat[(lane & 15) * 4 + (lane >> 4)];
    const float b = Bmat[(lane >> 4) * 16 + (lane & 15)];
    f32x4 acc{0, 0, 0, 0};
    acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
#pragma unroll
    for (int r = 0; r < 4; ++r)
        D[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = acc[r];
}

// bf16 16x16x32 probe through the exact stage/frag machinery of the Gramian:
// C[16][16] = Xt^T @ Yt for Xt, Yt [32][16] bf16.
__global__ void k_mfma_probe_bf16(const unsigned short* __restrict__ Xt,
                                  const unsigned short* __restrict__ Yt,
                                  float* __restrict__ C) {
    constexpr int SP = 32;  // 2 tiles of 16 cols; SP*2 = 64B, XOR-closed
    __shared__ __align__(16) unsigned short st[32 * SP];
    const int lane = threadIdx.x;  // launched with 64 threads
    {
        const int row = lane & 31, tile = lane >> 5;
        const unsigned short* src = (tile == 0 ? Xt : Yt) + row * 16;
        for (int h = 0; h < 2; ++h) {
            unsigned byte = (unsigned)(row * SP + tile * 16 + h * 8) * 2u
                            ^ stage_xor(row);
            *(uint4*)((char*)st + byte) = *(const uint4*)(src + h * 8);
        }
    }
    __syncthreads();
    const int g = lane >> 4, li = lane & 15;
    const unsigned xorb = (g & 1) ? 32u : 0u;
    bf16x8 frag[2];
#pragma unroll
    for (int t = 0; t < 2; ++t) {
        bf16x8 f;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            unsigned byte = (unsigned)((8 * g + j) * SP + t * 16 + li) * 2u;
            f[j] = *(const short*)((const char*)st + (byte ^ xorb));
        }
        frag[t] = f;
    }
    f32x4 acc{0, 0, 0, 0};
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(frag[0], frag[1], acc, 0, 0, 0);
#pragma unroll
    for (int r = 0; r < 4; ++r)
        C[(g * 4 + r) * 16 + li] = acc[r];
}

// -------------------------------------------------------------- launchers

#define DISPATCH_KT(k, expr)                                                  \
    switch ((k) / 16) {                                                       \
        case 1: { constexpr int KT = 1; expr; break; }                        \
        case 2: { constexpr int KT = 2; expr; break; }                        \
        case 3: { constexpr int KT = 3; expr; break; }                        \
        case 4: { constexpr int KT = 4; expr; break; }                        \
        case 5: { constexpr int KT = 5; expr; break; }                        \
        case 6: { constexpr int KT = 6; expr; break; }                        \
        case 7: { constexpr int KT = 7; expr; break; }                        \
        case 8: { constexpr int KT = 8; expr; break; }                        \
        default: return hipErrorInvalidValue;                                 \
    }

extern "C" hipError_t fma_als_solve_fused(
    int k, const long long* indptr, const int* indices, const float* values,
    const unsigned short* factors, float* out_f32, unsigned short* out_bf16,
    const int* row_order, long long nrows, float reg, hipStream_t stream) {
    if (k % 16 || k < 16 || k > 128 || nrows <= 0) return hipErrorInvalidValue;
    dim3 grid((unsigned)nrows), block(256);
    DISPATCH_KT(k, (k_als_solve_fused<KT><<<grid, block, 0, stream>>>(
        indptr, indices, values, factors, out_f32, out_bf16, row_order,
        nrows, reg)));
    return hipGetLastError();
}

extern "C" hipError_t fma_gramian(
    int k, const long long* indptr, const int* indices, const float* values,
    const unsigned short* factors, float* A_out, float* b_out,
    long long nrows, float reg, hipStream_t stream) {
    if (k % 16 || k < 16 || k > 128 || nrows <= 0) return hipErrorInvalidValue;
    dim3 grid((unsigned)nrows), block(256);
    DISPATCH_KT(k, (k_gramian<KT><<<grid, block, 0, stream>>>(
        indptr, indices, values, factors, A_out, b_out, nrows, reg)));
    return hipGetLastError();
}

extern "C" hipError_t fma_cholesky_solve(
    int k, const float* A_in, const float* b_in, float* x_out,
    long long nrows, hipStream_t stream) {
    if (k % 16 || k < 16 || k > 128 || nrows <= 0) return hipErrorInvalidValue;
    dim3 grid((unsigned)nrows), block(256);
    DISPATCH_KT(k, (k_cholesky_solve<KT><<<grid, block, 0, stream>>>(
        A_in, b_in, x_out, nrows)));
    return hipGetLastError();
}

extern "C" hipError_t fma_mfma_probe_f32(const float* A, const float* B,
                                         float* D, hipStream_t stream) {
    k_mfma_probe_f32<<<dim3(1), dim3(64), 0, stream>>>(A, B, D);
    return hipGetLastError();
}

extern "C" hipError_t fma_mfma_probe_bf16(const unsigned short* Xt,
                                          const unsigned short* Yt, float* C,
                                          hipStream_t stream) {
    k_mfma_probe_bf16<<<dim3(1), dim3(64), 0, stream>>>(Xt, Yt, C);
    return hipGetLastError();
}

extern "C" const char* fma_err_str(int err) {
    return hipGetErrorString((hipError_t)err);
}
