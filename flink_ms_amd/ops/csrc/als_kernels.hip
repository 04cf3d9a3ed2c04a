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
//       -> in-LDS LDL^T factorization (register panels + f32-MFMA trailing
//          updates) and substitution solves: wave-per-entity for k <= 64
//          (k_ldl_solve_wave, the modular default) or 256-thread blocks up
//          to k = 128 (cholesky_lds/solve_lds_block, also fused with K1 in
//          k_als_solve_fused so A never round-trips through HBM).
//
// Gramian geometry: one workgroup (256 threads = 4 waves) per entity;
// grid = #entities; wave w owns accumulator tiles t = w, w+4, ... of the
// upper-triangle + EXT tile list.  Factor rank k = 16*KT, KT 1..8
// (wrappers pad).
//
// LDS layout per block (union; the stage dies before A is born):
//   stage: TRANSPOSED Gt[K+16 rows][32 ratings] bf16, 96-B row stride
//          (24 dwords: 16 used + 8 pad).  Rows 0..K-1 hold the gathered
//          factor columns, rows K..K+1 the rating hi/lo pair, K+2..K+15
//          zeros.  Each MFMA fragment is ONE bank-conflict-free
//          ds_read_b128; staging transposes 4x4 bf16 blocks in registers
//          (als_kernels_device.inc).
//   A:     [K][K+1] fp32 (padded leading dim -> conflict-free column walks)
//   b_hi/b_lo: [K] fp32 each, then a 16x16 pivot-column scratch.

#include "common.hip.h"

#include "als_kernels_device.inc"

// -------------------------------------------------------------- kernels

// Fused K1+K2: normal equations + Cholesky solve, one entity per block.
template <int KT>
__launch_bounds__(256)
// waves_per_eu(3): without it the allocator splits 93 VGPR + 88 AGPR
// (181 total -> 2 waves/SIMD); constrained it finds a 127-VGPR, zero-AGPR,
// zero-spill allocation -> 3-4 waves/SIMD (measured: the K=128 fused
// kernel is the whole 1B-config iteration)
__attribute__((amdgpu_waves_per_eu(3)))
__global__ void k_als_solve_fused(const long long* __restrict__ indptr,
                                  const int* __restrict__ indices,
                                  const float* __restrict__ values,
                                  const unsigned short* __restrict__ factors,
                                  float* __restrict__ out_f32,
                                  unsigned short* __restrict__ out_bf16,
                                  const int* __restrict__ row_order,
                                  long long nrows, float reg) {
    constexpr int K = Geo<KT>::K;
    __shared__ __align__(16) char smem[Geo<KT>::SMEM_TRI];
    long long row = blockIdx.x;
    if (row >= nrows) return;
    if (row_order) row = row_order[row];

    const int n = gramian_to_lds<KT, true, true>(smem, indptr, indices,
                                                 values, factors, row, reg);
    if (n == 0) { write_zero_row<KT>(out_f32, out_bf16, row); return; }

    float* A = (float*)smem;
    float* b = A + Geo<KT>::A_TRI_FLOATS;
    cholesky_lds_tri<K>(A, b + 2 * K);
    solve_lds_block_tri<K>(A, b, b + 2 * K, out_f32 + row * K,
                           out_bf16 ? out_bf16 + row * K : nullptr);
}

// Standalone K1 (for parity tests / modular path): writes dense A and b.
// row_order (optional): degree-descending launch schedule — block i
// processes entity row_order[i] and writes A_out/b_out at THAT row, so the
// downstream batched solve stays order-agnostic (heavy entities launch
// first; avoids tail stragglers at the end of the grid).
template <int KT>
__launch_bounds__(256)
__global__ void k_gramian(const long long* __restrict__ indptr,
                          const int* __restrict__ indices,
                          const float* __restrict__ values,
                          const unsigned short* __restrict__ factors,
                          float* __restrict__ A_out,   // [nrows][K][K]
                          float* __restrict__ b_out,   // [nrows][K]
                          const int* __restrict__ row_order,
                          long long nrows, float reg) {
    constexpr int K = Geo<KT>::K;
    __shared__ __align__(16) char smem[Geo<KT>::SMEM];
    long long row = blockIdx.x;
    if (row >= nrows) return;
    if (row_order) row = row_order[row];
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

// fp8 twins of the fused and standalone-K1 kernels: e4m3 factor gathers
// (one cache line per k<=64 row) through gramian_to_lds_fp8; everything
// downstream (fp32 A/b, LDL solve) is identical.  out_fp8 writes the next
// half-iteration's e4m3 factor image alongside fp32 (and optional bf16).
template <int KT>
__launch_bounds__(256)
// waves_per_eu(3): without it the allocator splits 93 VGPR + 88 AGPR
// (181 total -> 2 waves/SIMD); constrained it finds a 127-VGPR, zero-AGPR,
// zero-spill allocation -> 3-4 waves/SIMD (measured: the K=128 fused
// kernel is the whole 1B-config iteration)
__attribute__((amdgpu_waves_per_eu(3)))
__global__ void k_als_solve_fused_fp8(const long long* __restrict__ indptr,
                                      const int* __restrict__ indices,
                                      const float* __restrict__ values,
                                      const unsigned char* __restrict__ factors,
                                      float* __restrict__ out_f32,
                                      unsigned char* __restrict__ out_fp8,
                                      const int* __restrict__ row_order,
                                      long long nrows, float reg) {
    constexpr int K = Geo<KT>::K;
    __shared__ __align__(16) char smem[Geo<KT>::SMEM8_TRI];
    long long row = blockIdx.x;
    if (row >= nrows) return;
    if (row_order) row = row_order[row];

    const int n = gramian_to_lds<KT, true, true, true>(smem, indptr,
                                                       indices, values,
                                                       factors, row, reg);
    if (n == 0) {
        for (int c = threadIdx.x; c < K; c += 256) {
            out_f32[row * K + c] = 0.0f;
            if (out_fp8) out_fp8[row * K + c] = 0;
        }
        return;
    }
    float* A = (float*)smem;
    float* b = A + Geo<KT>::A_TRI_FLOATS;
    cholesky_lds_tri<K>(A, b + 2 * K);
    solve_lds_block_tri<K>(A, b, b + 2 * K, out_f32 + row * K, nullptr,
                           out_fp8 ? out_fp8 + row * K : nullptr);
}

template <int KT>
__launch_bounds__(256)
__global__ void k_gramian_fp8(const long long* __restrict__ indptr,
                              const int* __restrict__ indices,
                              const float* __restrict__ values,
                              const unsigned char* __restrict__ factors,
                              float* __restrict__ A_out,   // [nrows][K][K]
                              float* __restrict__ b_out,   // [nrows][K]
                              const int* __restrict__ row_order,
                              long long nrows, float reg) {
    constexpr int K = Geo<KT>::K;
    __shared__ __align__(16) char smem[Geo<KT>::SMEM8];
    long long row = blockIdx.x;
    if (row >= nrows) return;
    if (row_order) row = row_order[row];
    const int tid = threadIdx.x;

    const int n = gramian_to_lds<KT, true, false, true>(
        smem, indptr, indices, values, factors, row, reg);
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
// phases bitmask (debug/ablation): 1 = eliminate, 2 = solve
template <int KT>
__launch_bounds__(256, 2)
__global__ void k_cholesky_solve(const float* __restrict__ A_in,  // [n][K][K]
                                 const float* __restrict__ b_in,  // [n][K]
                                 float* __restrict__ x_out,       // [n][K]
                                 long long nrows, int phases) {
    constexpr int K = Geo<KT>::K;
    constexpr int NTRI = Geo<KT>::NA;
    __shared__ __align__(16) char smem[Geo<KT>::CHOL_TRI_BYTES];
    const long long row = blockIdx.x;
    if (row >= nrows) return;
    float* A = (float*)smem;
    float* b = A + Geo<KT>::A_TRI_FLOATS;
    const int tid = threadIdx.x;
    {   // load the lower tiles of the (symmetric) square A into the tri
        // image: one whole 16x16 tile per iteration of the 256 threads
        const float* src = A_in + row * (long long)(K * K);
        const int r = tid >> 4, c = tid & 15;
        int I = 0, J = 0;
        for (int t = 0; t < NTRI; ++t) {
            A[t * Geo<KT>::TSZ + r * Geo<KT>::LDT + c] =
                src[(I * 16 + r) * K + J * 16 + c];
            if (++J > I) { ++I; J = 0; }
        }
    }
    for (int c = tid; c < K; c += 256) b[c] = b_in[row * K + c];
    __syncthreads();
    if (phases & 1) cholesky_lds_tri<K>(A, b + 2 * K);
    if (phases & 2) {
        solve_lds_block_tri<K>(A, b, b + 2 * K, x_out + row * K, nullptr);
    } else if (tid < K) {
        x_out[row * K + tid] = b[tid];
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
    auto stage_xor = [](int row) -> unsigned { return (row & 8) ? 32u : 0u; };
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

// fp8 e4m3 16x16x32 probe through the fp8 stage geometry (TROW8 rows,
// ds_read_b64 fragments): C[16][16] = Xt^T @ Yt for Xt, Yt [32][16] e4m3
// bytes.  Validates the (lane, element) map + C/D layout assumption of
// k_gramian_fp8 against a torch reference (tests/test_gpu_mfma.py).
__global__ void k_mfma_probe_fp8(const unsigned char* __restrict__ Xt,
                                 const unsigned char* __restrict__ Yt,
                                 float* __restrict__ C) {
    constexpr int TROW8 = 40;
    __shared__ __align__(8) char st[32 * TROW8];
    const int lane = threadIdx.x;  // launched with 64 threads
    for (int i = lane; i < 2 * 16 * 32; i += 64) {
        const int tile = i >> 9, rem = i & 511;
        const int r = rem >> 4, c = rem & 15;   // rating r, column c
        st[(tile * 16 + c) * TROW8 + r] = (tile == 0 ? Xt : Yt)[r * 16 + c];
    }
    __syncthreads();
    const int g = lane >> 4, li = lane & 15;
    const fp8x8 fx = *(const fp8x8*)(st + (li)*TROW8 + g * 8);
    const fp8x8 fy = *(const fp8x8*)(st + (16 + li) * TROW8 + g * 8);
    f32x4 acc{0, 0, 0, 0};
    acc = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(fx, fy, acc, 0, 0, 0);
#pragma unroll
    for (int r = 0; r < 4; ++r)
        C[(g * 4 + r) * 16 + li] = acc[r];
}

// -------------------------------------------------------------- launchers

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
    const int* row_order, long long nrows, float reg, hipStream_t stream) {
    if (k % 16 || k < 16 || k > 128 || nrows <= 0) return hipErrorInvalidValue;
    dim3 grid((unsigned)nrows), block(256);
    DISPATCH_KT(k, (k_gramian<KT><<<grid, block, 0, stream>>>(
        indptr, indices, values, factors, A_out, b_out, row_order, nrows,
        reg)));
    return hipGetLastError();
}

extern "C" hipError_t fma_gramian_fp8(
    int k, const long long* indptr, const int* indices, const float* values,
    const unsigned char* factors, float* A_out, float* b_out,
    const int* row_order, long long nrows, float reg, hipStream_t stream) {
    if (k % 16 || k < 16 || k > 128 || nrows <= 0) return hipErrorInvalidValue;
    dim3 grid((unsigned)nrows), block(256);
    DISPATCH_KT(k, (k_gramian_fp8<KT><<<grid, block, 0, stream>>>(
        indptr, indices, values, factors, A_out, b_out, row_order, nrows,
        reg)));
    return hipGetLastError();
}

extern "C" hipError_t fma_als_solve_fused_fp8(
    int k, const long long* indptr, const int* indices, const float* values,
    const unsigned char* factors, float* out_f32, unsigned char* out_fp8,
    const int* row_order, long long nrows, float reg, hipStream_t stream) {
    if (k % 16 || k < 16 || k > 128 || nrows <= 0) return hipErrorInvalidValue;
    dim3 grid((unsigned)nrows), block(256);
    DISPATCH_KT(k, (k_als_solve_fused_fp8<KT><<<grid, block, 0, stream>>>(
        indptr, indices, values, factors, out_f32, out_fp8, row_order,
        nrows, reg)));
    return hipGetLastError();
}

extern "C" hipError_t fma_cholesky_solve_ph(
    int k, const float* A_in, const float* b_in, float* x_out,
    long long nrows, int phases, hipStream_t stream) {
    if (k % 16 || k < 16 || k > 128 || nrows <= 0) return hipErrorInvalidValue;
    dim3 grid((unsigned)nrows), block(256);
    DISPATCH_KT(k, (k_cholesky_solve<KT><<<grid, block, 0, stream>>>(
        A_in, b_in, x_out, nrows, phases)));
    return hipGetLastError();
}

extern "C" hipError_t fma_cholesky_solve(
    int k, const float* A_in, const float* b_in, float* x_out,
    long long nrows, hipStream_t stream) {
    return fma_cholesky_solve_ph(k, A_in, b_in, x_out, nrows, 3, stream);
}


// -------- wave-per-entity LDL solver (k <= 64) --------
// The block-per-entity factorization is issue-bound: ~100 instructions of
// work between 2x64 barriers.  For k <= 64 a single wave owns one entity's
// whole [K][K+1] LDS image instead: no barriers (wave-internal LDS ordering
// suffices), 4 independent entities per block, solve fully in registers.
template <int KT>
__launch_bounds__(256)
__global__ void k_ldl_solve_wave(const float* __restrict__ A_in,
                                 const float* __restrict__ b_in,
                                 float* __restrict__ x_out,
                                 unsigned short* __restrict__ x_bf16,
                                 unsigned char* __restrict__ x_fp8,
                                 long long nrows) {
    constexpr int K = KT * 16;
    static_assert(K <= 64, "wave solver handles k <= 64");
    // A lives in LDS as LOWER-TRIANGLE 16x16 tiles only (tile (I,J), I>=J,
    // at triangular offset I(I+1)/2+J).  The LDL never reads above the
    // block diagonal; the panel factorization's upper-scratch trick only
    // needs the diagonal tiles' upper halves, which the full 16x16 tiles
    // keep.  At K=64 this is 10/16 of the square: LDS 66.5 -> 46 KB per
    // block (2 -> 3 waves/SIMD) and 10/16 of the HBM read of A.
    // Tile row stride 18: 8-byte-aligned rows, conflict-free column access
    // (18*i mod 32 distinct over i=0..15).
    constexpr int NP = K / 16;
    constexpr int NTRI = NP * (NP + 1) / 2;
    constexpr int LDT = 18;
    constexpr int TSZ = 16 * LDT;
    __shared__ __align__(16) float As[4][NTRI * TSZ];
    const int w = __builtin_amdgcn_readfirstlane(threadIdx.x >> 6);
    const int lane = threadIdx.x & 63;
    const long long e = (long long)blockIdx.x * 4 + w;
    if (e >= nrows) return;
    float* A = As[w];
    const int li = lane & 15, g4 = lane >> 4;
    {   // load the lower tiles: one whole tile per iteration (16 rows x
        // 4 float4 quads = 64 lanes), contiguous 16-byte global reads
        const float* src = A_in + e * (long long)(K * K);
        const int r = lane >> 2, c4 = lane & 3;
        int I = 0, J = 0;
        for (int t = 0; t < NTRI; ++t) {
            const f32x4 v =
                *(const f32x4*)(src + (I * 16 + r) * K + J * 16 + c4 * 4);
            float* dst = A + t * TSZ + r * LDT + c4 * 4;
            dst[0] = v[0]; dst[1] = v[1]; dst[2] = v[2]; dst[3] = v[3];
            if (++J > I) { ++I; J = 0; }
        }
    }
    float x0 = (lane < K) ? b_in[e * K + lane] : 0.0f;
    // In-place LDL^T elimination, 16-column panels with MFMA trailing
    // updates (cdna_hip_programming.md §6 G10: the trailing update
    // A22 -= L21 D^-1 L21^T is matmul-shaped, so it runs on the f32 matrix
    // cores: v_mfma_f32_16x16x4_f32, exact fp32).  Per panel:
    //   (a) factor the panel slab in place, column by column, each lane
    //       updating its own row's panel segment (columns stay raw L*D;
    //       over-length quads CLAMP to the panel edge -- duplicate
    //       same-value writes, never out-of-segment ones);
    //   (b) subtract the panel's outer product from the trailing block as
    //       16x16 MFMA tiles (B operand scaled by -1/D on load).  Tiles on
    //       or above the diagonal write upper-triangle scratch only.
    // Lanes run in wave lockstep, so cross-lane panel dependencies are
    // ordered by program order; no barriers.
    // NOTE: the MFMA tiles need ALL 64 lanes live (every lane carries one
    // (i,k) fragment element regardless of K), so only the row-ownership
    // panel factorization is masked to lane < K; every address in the
    // MFMA phase is bounded by K for all lanes.
    {
        auto tri = [](int I, int J) { return (I * (I + 1)) / 2 + J; };
        // rowseg: lane = global row; its 16-col panel-pi segment lives in
        // tile (lane>>4, pi).  Only dereferenced when P0 <= lane < K.
        for (int pi = 0; pi < NP; ++pi) {
            const int P0 = 16 * pi;
            // (a) panel factorization, fully in registers: each lane holds
            // its row's 16-column panel segment; the pivot column A[c][j]
            // lives in lane c's seg[jj], broadcast by __shfl -- the inner
            // rank-1 update is shfl+fma with ZERO LDS traffic.  Rows above
            // the panel (lane < P0) have no storage in the triangular
            // image and sit the phase out (one exec mask per panel, not
            // per element).
            if (lane >= P0 && lane < K) {
                float* rowseg =
                    A + tri(lane >> 4, pi) * TSZ + (lane & 15) * LDT;
                float seg[16];
#pragma unroll
                for (int cc = 0; cc < 16; ++cc) seg[cc] = rowseg[cc];
#pragma unroll
                for (int jj = 0; jj < 15; ++jj) {
                    const float dj = __shfl(seg[jj], P0 + jj, WAVE);
                    const float dinv = dj > 0.0f ? 1.0f / dj : 0.0f;
                    const float lscl = seg[jj] * dinv;
#pragma unroll
                    for (int cc = jj + 1; cc < 16; ++cc) {
                        const float colj = __shfl(seg[jj], P0 + cc, WAVE);
                        seg[cc] -= lscl * colj;
                    }
                }
#pragma unroll
                for (int cc = 0; cc < 16; ++cc) rowseg[cc] = seg[cc];
            }
            if (pi == NP - 1) break;
            // (b) MFMA trailing update, whole wave.  -1/D per contraction
            // column, local panel col 4*kk + g4.
            const float* Tpp = A + tri(pi, pi) * TSZ;
            float ndk[4];
#pragma unroll
            for (int kk = 0; kk < 4; ++kk) {
                const int c = 4 * kk + g4;
                const float d = Tpp[c * LDT + c];
                ndk[kk] = d > 0.0f ? -1.0f / d : 0.0f;
            }
            for (int rb = pi + 1; rb < NP; ++rb) {
                const float* TA = A + tri(rb, pi) * TSZ;  // L21 rows rb
                for (int cb = pi + 1; cb <= rb; ++cb) {
                    const float* TB = A + tri(cb, pi) * TSZ;
                    float* TC = A + tri(rb, cb) * TSZ;
                    f32x4 acc;   // C tile: D map row=(l>>4)*4+r, col=l&15
#pragma unroll
                    for (int r = 0; r < 4; ++r)
                        acc[r] = TC[(g4 * 4 + r) * LDT + li];
#pragma unroll
                    for (int kk = 0; kk < 4; ++kk) {
                        const int pc = 4 * kk + g4;
                        const float a = TA[li * LDT + pc];
                        const float b = TB[li * LDT + pc] * ndk[kk];
                        acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc,
                                                                   0, 0, 0);
                    }
#pragma unroll
                    for (int r = 0; r < 4; ++r)
                        TC[(g4 * 4 + r) * LDT + li] = acc[r];
                }
            }
        }
    }
    // solve (I+Ls) y = b ; w = y/D ; (I+Ls^T) x = w, all in registers.
    // Row `lane`'s tiles sit at triangular offsets rowbase + J for J <=
    // lane>>4 (rowbase fixed per lane).
    const int Irow = lane >> 4;
    const float* rowbase = A + ((Irow * (Irow + 1)) / 2) * TSZ + li * LDT;
    const float d0 = (lane < K) ? rowbase[Irow * TSZ + li] : 1.0f;
    const float id0 = d0 > 0.0f ? 1.0f / d0 : 0.0f;
    for (int j = 0; j < K - 1; ++j) {
        const float zj = __shfl(x0, j, WAVE) * __shfl(id0, j, WAVE);
        if (lane > j && lane < K)
            x0 -= rowbase[(j >> 4) * TSZ + (j & 15)] * zj;
    }
    x0 *= id0;
    for (int c = K - 1; c >= 1; --c) {
        const float xc = __shfl(x0, c, WAVE);
        // A(c, lane): row c's tile at column block lane>>4
        if (lane < c)
            x0 -= A[(((c >> 4) * ((c >> 4) + 1)) / 2 + (lane >> 4)) * TSZ
                    + (c & 15) * LDT + li] * id0 * xc;
    }
    if (lane < K) {
        x_out[e * K + lane] = x0;
        if (x_bf16) x_bf16[e * K + lane] = f2bf(x0);
        if (x_fp8) x_fp8[e * K + lane] = f2fp8(x0);
    }
}


// -------- register-resident wave LDL (k <= 64), v2 --------
// A lives in REGISTERS as the 10 lower-triangle 16x16 MFMA C-fragments
// (lane holds (row = (l>>4)*4 + r, col = l&15) of each tile); a 4.35 KB
// per-entity LDS scratch re-shapes one 16-column panel (or one tile
// row/column for the substitution) at a time.  LDS per 4-entity block
// drops 46 -> 17.4 KB and the trailing-update C tiles never touch LDS.
// All tile indices are COMPILE-TIME (constexpr recursion): runtime
// indexing of the fragment array would spill to scratch (guide rule 20).
// Scratch strides: 17 (panel rows, conflict-free columns), 65 (tile-row
// image for the backward substitution).
// WREG_FENCE: the scratch bounces are cross-lane LDS write->read chains
// with no barrier (single wave).  The LDS unit processes a wave's ds ops
// in order, but the COMPILER must not reorder the reads above the masked
// writes (observed miscompile: reader lanes in other row-quads got stale
// panel values) -- a compiler memory fence plus lgkmcnt drain pins the
// order at negligible cost (a handful per entity).
#define WREG_FENCE() __asm__ volatile("s_waitcnt lgkmcnt(0)" ::: "memory")


template <int KT, int PI, int I>
DEV_INLINE void wreg_dump_col(const f32x4* T, float* scr, int g4, int li) {
    if constexpr (I < KT) {
        constexpr int t = tri_off(I, PI);
#pragma unroll
        for (int r = 0; r < 4; ++r)
            scr[((I - PI) * 16 + g4 * 4 + r) * 17 + li] = T[t][r];
        wreg_dump_col<KT, PI, I + 1>(T, scr, g4, li);
    }
}

template <int KT, int PI, int I>
DEV_INLINE void wreg_load_col(f32x4* T, const float* scr, int g4, int li) {
    if constexpr (I < KT) {
        constexpr int t = tri_off(I, PI);
#pragma unroll
        for (int r = 0; r < 4; ++r)
            T[t][r] = scr[((I - PI) * 16 + g4 * 4 + r) * 17 + li];
        wreg_load_col<KT, PI, I + 1>(T, scr, g4, li);
    }
}

template <int KT, int PI, int RB, int CB>
DEV_INLINE void wreg_trail(f32x4* T, const float* scr, const float* ndk,
                           int g4, int li) {
    if constexpr (RB < KT) {
        if constexpr (CB > RB) {
            wreg_trail<KT, PI, RB + 1, PI + 1>(T, scr, ndk, g4, li);
        } else {
            constexpr int t = tri_off(RB, CB);
            f32x4 acc = T[t];
#pragma unroll
            for (int kk = 0; kk < 4; ++kk) {
                const int pc = 4 * kk + g4;
                const float a = scr[((RB - PI) * 16 + li) * 17 + pc];
                const float b =
                    scr[((CB - PI) * 16 + li) * 17 + pc] * ndk[kk];
                acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0,
                                                           0);
            }
            T[t] = acc;
            wreg_trail<KT, PI, RB, CB + 1>(T, scr, ndk, g4, li);
        }
    }
}

template <int KT, int PI>
DEV_INLINE void wreg_panels(f32x4* T, float* scr, int lane, int g4, int li,
                            float& d0) {
    if constexpr (PI < KT) {
        constexpr int K = KT * 16;
        constexpr int P0 = 16 * PI;
        wreg_dump_col<KT, PI, PI>(T, scr, g4, li);
        WREG_FENCE();
        // register factorization of the panel: lane = row P0+lane
        if (lane < K - P0) {
            float seg[16];
#pragma unroll
            for (int cc = 0; cc < 16; ++cc) seg[cc] = scr[lane * 17 + cc];
#pragma unroll
            for (int jj = 0; jj < 15; ++jj) {
                const float dj = __shfl(seg[jj], jj, WAVE);
                const float dinv = dj > 0.0f ? 1.0f / dj : 0.0f;
                const float lscl = seg[jj] * dinv;
#pragma unroll
                for (int cc = jj + 1; cc < 16; ++cc) {
                    const float colj = __shfl(seg[jj], cc, WAVE);
                    seg[cc] -= lscl * colj;
                }
            }
#pragma unroll
            for (int cc = 0; cc < 16; ++cc) scr[lane * 17 + cc] = seg[cc];
        }
        WREG_FENCE();
        if (lane >= P0 && lane < P0 + 16)
            d0 = scr[(lane - P0) * 17 + (lane - P0)];
        wreg_load_col<KT, PI, PI>(T, scr, g4, li);
        if constexpr (PI + 1 < KT) {
            float ndk[4];
#pragma unroll
            for (int kk = 0; kk < 4; ++kk) {
                const int c = 4 * kk + g4;
                const float d = scr[c * 17 + c];
                ndk[kk] = d > 0.0f ? -1.0f / d : 0.0f;
            }
            wreg_trail<KT, PI, PI + 1, PI + 1>(T, scr, ndk, g4, li);
        }
        wreg_panels<KT, PI + 1>(T, scr, lane, g4, li, d0);
    }
}

template <int KT, int J>
DEV_INLINE void wreg_forward(const f32x4* T, float* scr, float& x0,
                             float id0, int lane, int g4, int li) {
    if constexpr (J < KT) {
        constexpr int K = KT * 16;
        constexpr int B0 = 16 * J;
        wreg_dump_col<KT, J, J>(T, scr, g4, li);
        WREG_FENCE();
#pragma unroll
        for (int t = 0; t < 16; ++t) {
            const int j = B0 + t;
            if (j >= K - 1) break;
            const float zj = __shfl(x0, j, WAVE) * __shfl(id0, j, WAVE);
            if (lane > j && lane < K)
                x0 -= scr[(lane - B0) * 17 + t] * zj;
        }
        wreg_forward<KT, J + 1>(T, scr, x0, id0, lane, g4, li);
    }
}

template <int KT, int I, int J>
DEV_INLINE void wreg_dump_row(const f32x4* T, float* scr, int g4, int li) {
    if constexpr (J <= I) {
        constexpr int t = tri_off(I, J);
#pragma unroll
        for (int r = 0; r < 4; ++r)
            scr[(g4 * 4 + r) * 65 + J * 16 + li] = T[t][r];
        wreg_dump_row<KT, I, J + 1>(T, scr, g4, li);
    }
}

template <int KT, int I>
DEV_INLINE void wreg_backward(const f32x4* T, float* scr, float& x0,
                              float id0, int lane, int g4, int li) {
    if constexpr (I >= 0) {
        constexpr int B0 = 16 * I;
        wreg_dump_row<KT, I, 0>(T, scr, g4, li);
        WREG_FENCE();
#pragma unroll
        for (int t = 15; t >= 0; --t) {
            const int c = B0 + t;
            if (c < 1) break;
            const float xc = __shfl(x0, c, WAVE);
            if (lane < c)
                x0 -= scr[t * 65 + lane] * id0 * xc;
        }
        wreg_backward<KT, I - 1>(T, scr, x0, id0, lane, g4, li);
    }
}

template <int KT, int t>
DEV_INLINE void wreg_load_A(f32x4* T, const float* src, int g4, int li) {
    if constexpr (t < KT * (KT + 1) / 2) {
        constexpr int K = KT * 16;
        constexpr int I = lo_tile_i(t);
        constexpr int J = t - (I * (I + 1)) / 2;
#pragma unroll
        for (int r = 0; r < 4; ++r)
            T[t][r] = src[(I * 16 + g4 * 4 + r) * K + J * 16 + li];
        wreg_load_A<KT, t + 1>(T, src, g4, li);
    }
}

template <int KT>
__launch_bounds__(256)
__global__ void k_ldl_solve_wave_reg(const float* __restrict__ A_in,
                                     const float* __restrict__ b_in,
                                     float* __restrict__ x_out,
                                     unsigned short* __restrict__ x_bf16,
                                     unsigned char* __restrict__ x_fp8,
                                     long long nrows) {
    constexpr int K = KT * 16;
    static_assert(K <= 64, "register wave solver handles k <= 64");
    constexpr int NTRI = KT * (KT + 1) / 2;
    constexpr int SCR = K * 17;            // covers 16*65=1040 too (K=64)
    __shared__ __align__(16) float As[4][SCR > 1040 ? SCR : 1040];
    const int w = __builtin_amdgcn_readfirstlane(threadIdx.x >> 6);
    const int lane = threadIdx.x & 63;
    const long long e = (long long)blockIdx.x * 4 + w;
    if (e >= nrows) return;
    float* scr = As[w];
    const int li = lane & 15, g4 = lane >> 4;

    f32x4 T[NTRI];
    wreg_load_A<KT, 0>(T, A_in + e * (long long)(K * K), g4, li);
    float x0 = (lane < K) ? b_in[e * K + lane] : 0.0f;
    float d0 = 1.0f;
    wreg_panels<KT, 0>(T, scr, lane, g4, li, d0);
    const float id0 = d0 > 0.0f ? 1.0f / d0 : 0.0f;
    wreg_forward<KT, 0>(T, scr, x0, id0, lane, g4, li);
    x0 *= id0;
    wreg_backward<KT, KT - 1>(T, scr, x0, id0, lane, g4, li);
    if (lane < K) {
        x_out[e * K + lane] = x0;
        if (x_bf16) x_bf16[e * K + lane] = f2bf(x0);
        if (x_fp8) x_fp8[e * K + lane] = f2fp8(x0);
    }
}

extern "C" hipError_t fma_ldl_solve_wave_reg(
    int k, const float* A_in, const float* b_in, float* x_out,
    unsigned short* x_bf16, unsigned char* x_fp8, long long nrows,
    hipStream_t stream) {
    if (k % 16 || k < 16 || k > 64 || nrows <= 0) return hipErrorInvalidValue;
    dim3 grid((unsigned)((nrows + 3) / 4)), block(256);
    switch (k / 16) {
        case 1: k_ldl_solve_wave_reg<1><<<grid, block, 0, stream>>>(A_in, b_in, x_out, x_bf16, x_fp8, nrows); break;
        case 2: k_ldl_solve_wave_reg<2><<<grid, block, 0, stream>>>(A_in, b_in, x_out, x_bf16, x_fp8, nrows); break;
        case 3: k_ldl_solve_wave_reg<3><<<grid, block, 0, stream>>>(A_in, b_in, x_out, x_bf16, x_fp8, nrows); break;
        case 4: k_ldl_solve_wave_reg<4><<<grid, block, 0, stream>>>(A_in, b_in, x_out, x_bf16, x_fp8, nrows); break;
        default: return hipErrorInvalidValue;
    }
    return hipGetLastError();
}

// -------- wave-fused ALS solve (k <= 64): Gramian + LDL in ONE wave --------
// r2 profiling: with the v2 multi-chunk gramian the modular path became
// A-round-trip bound — the gramian writes nrows*K*K fp32 to HBM (~2.9 TB/s)
// and the wave solver reads the lower tiles back (~1.5 TB/s).  This kernel
// removes that traffic entirely: one WAVE owns one entity end to end and
// the Gramian MFMAs accumulate DIRECTLY into the lower-triangle C-fragment
// tiles T[] that the register LDL (wreg_panels/forward/backward) consumes.
// No barriers anywhere (wave-local LDS + WREG_FENCE ordering); gather
// stalls of one wave overlap solver compute of the others.
//
// Tile algebra: lower tile (I,J) = mfma(frag[I], frag[J]) — D[row][col] =
// sum_r G[r][I*16+row]*G[r][J*16+col] = A[I*16+row][J*16+col], exactly the
// (g4*4+r, li) layout wreg_load_A produced in the modular path.  b comes
// from KT EXT tiles (cols 0/1 = rating hi/lo), redistributed to
// x0 = b[lane] through a 128-float LDS bounce.

template <int KT, bool FP8, int t = 0>
DEV_INLINE void mfma_lower_tiles(const typename FragT<FP8>::type* frag,
                                 f32x4* T) {
    if constexpr (t < KT * (KT + 1) / 2) {
        constexpr int I = lo_tile_i(t);
        constexpr int J = t - (I * (I + 1)) / 2;
        if constexpr (FP8)
            T[t] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
                frag[I], frag[J], T[t], 0, 0, 0);
        else
            T[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                frag[I], frag[J], T[t], 0, 0, 0);
        mfma_lower_tiles<KT, FP8, t + 1>(frag, T);
    }
}

template <int KT, bool FP8, int P = 0>
DEV_INLINE void mfma_ext_tiles(const typename FragT<FP8>::type* frag,
                               f32x4* E) {
    if constexpr (P < KT) {
        if constexpr (FP8)
            E[P] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
                frag[P], frag[KT], E[P], 0, 0, 0);
        else
            E[P] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                frag[P], frag[KT], E[P], 0, 0, 0);
        mfma_ext_tiles<KT, FP8, P + 1>(frag, E);
    }
}

template <int KT, int P = 0>
DEV_INLINE void wavefused_dump_b(const f32x4* E, float* scr, int g4, int li) {
    if constexpr (P < KT) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            if (li == 0) scr[P * 16 + g4 * 4 + r] = E[P][r];
            else if (li == 1) scr[64 + P * 16 + g4 * 4 + r] = E[P][r];
        }
        wavefused_dump_b<KT, P + 1>(E, scr, g4, li);
    }
}

template <int KT, int I = 0>
DEV_INLINE void wavefused_diag(f32x4* T, float regn, int g4, int li) {
    if constexpr (I < KT) {
        constexpr int t = tri_off(I, I);
#pragma unroll
        for (int r = 0; r < 4; ++r)
            if (li == g4 * 4 + r) {
                const float dd = T[t][r] + regn;
                T[t][r] = dd <= 0.0f ? 1.0f : dd;   // degenerate guard
            }
        wavefused_diag<KT, I + 1>(T, regn, g4, li);
    }
}

// Split staging for the double-buffered wave-fused variant: issue the
// gathers for chunk ch+1 (registers only, no wait) BEFORE chunk ch's
// MFMAs, commit the transpose+LDS writes after — the load latency hides
// behind the matrix work instead of stalling the wave.
template <int KT, bool FP8>
struct StagedRegs {
    uint4 v[4];
    unsigned ext_h, ext_l;
};

template <int KT, bool FP8>
DEV_INLINE void stage_loads(StagedRegs<KT, FP8>& sr,
                            const int* __restrict__ indices,
                            const float* __restrict__ values,
                            const void* __restrict__ factors_v,
                            long long base, int nrem, int lane) {
    constexpr int K = Geo<KT>::K;
    static_assert(FP8, "double-buffer staging: fp8 path only");
    const unsigned char* factors = (const unsigned char*)factors_v;
    constexpr int NT = 8 * (K / 16);
    if (lane < NT) {
        const int q = lane & 7, c16 = lane >> 3;
#pragma unroll
        for (int m = 0; m < 4; ++m) {
            uint4 x = {0, 0, 0, 0};
            const int row = 4 * q + m;
            if (row < nrem) {
                const long long col = indices[base + row];
                x = *(const uint4*)(factors + col * (long long)K + c16 * 16);
            }
            sr.v[m] = x;
        }
    }
    if (lane < 8) {
        unsigned h = 0, l = 0;
#pragma unroll
        for (int m = 0; m < 4; ++m) {
            const int row = 4 * lane + m;
            const float r = (row < nrem) ? values[base + row] : 0.0f;
            const unsigned char hb = f2fp8(r);
            const unsigned char lb = f2fp8(r - fp82f(hb));
            h |= ((unsigned)hb) << (8 * m);
            l |= ((unsigned)lb) << (8 * m);
        }
        sr.ext_h = h;
        sr.ext_l = l;
    }
}

template <int KT, bool FP8>
DEV_INLINE void stage_commit(const StagedRegs<KT, FP8>& sr, char* buf,
                             int lane) {
    constexpr int K = Geo<KT>::K;
    constexpr int TROW = Geo<KT>::TROW8;
    constexpr int NT = 8 * (K / 16);
    if (lane < NT) {
        const int q = lane & 7, c16 = lane >> 3;
        char* rb = buf + (long long)(16 * c16) * TROW + q * 4;
#pragma unroll
        for (int cc = 0; cc < 16; ++cc) {
            unsigned wd = 0;
#pragma unroll
            for (int m = 0; m < 4; ++m)
                wd |= (((&sr.v[m].x)[cc >> 2] >> (8 * (cc & 3))) & 0xffu)
                      << (8 * m);
            *(unsigned*)(rb + (long long)cc * TROW) = wd;
        }
    }
    if (lane < 8) {
        *(unsigned*)(buf + (long long)K * TROW + lane * 4) = sr.ext_h;
        *(unsigned*)(buf + (long long)(K + 1) * TROW + lane * 4) = sr.ext_l;
    }
}

// Double-buffered wave-fused ALS solve (fp8, k <= 64): same math as
// k_als_solve_wavefused, software-pipelined staging.
template <int KT>
__launch_bounds__(256)
__attribute__((amdgpu_waves_per_eu(4)))
__global__ void k_als_solve_wavefused_db(const long long* __restrict__ indptr,
                                         const int* __restrict__ indices,
                                         const float* __restrict__ values,
                                         const void* __restrict__ factors,
                                         float* __restrict__ out_f32,
                                         unsigned short* __restrict__ out_bf16,
                                         unsigned char* __restrict__ out_fp8,
                                         const int* __restrict__ row_order,
                                         long long nrows, float reg) {
    constexpr int K = KT * 16;
    static_assert(K <= 64, "wave-fused path handles k <= 64");
    constexpr int NA = KT * (KT + 1) / 2;
    constexpr int TROW = Geo<KT>::TROW8;
    constexpr int SB = (K + 16) * TROW;
    constexpr int SCRF = (K * 17 > 1040) ? K * 17 : 1040;
    constexpr int WB0 = (2 * SB > SCRF * 4) ? 2 * SB : SCRF * 4;
    constexpr int WB = (WB0 + 15) & ~15;
    __shared__ __align__(16) char smem[4 * WB];
    const int w = __builtin_amdgcn_readfirstlane(threadIdx.x >> 6);
    const int lane = threadIdx.x & 63;
    const long long e0 = (long long)blockIdx.x * 4 + w;
    if (e0 >= nrows) return;
    const long long e = row_order ? row_order[e0] : e0;
    char* base = smem + (long long)w * WB;
    float* scr = (float*)base;
    const int g4 = lane >> 4, li = lane & 15;
    const long long p0 = indptr[e];
    const int n = (int)(indptr[e + 1] - p0);
    if (n == 0) {
        if (lane < K) {
            out_f32[e * K + lane] = 0.0f;
            if (out_bf16) out_bf16[e * K + lane] = 0;
            if (out_fp8) out_fp8[e * K + lane] = 0;
        }
        return;
    }
    for (int b = 0; b < 2; ++b)
        for (int i = lane; i < 14 * 8; i += WAVE) {
            const int row = K + 2 + i / 8, seg = i % 8;
            *(unsigned*)(base + b * SB + (long long)row * TROW + seg * 4) =
                0u;
        }
    f32x4 T[NA], E[KT];
#pragma unroll
    for (int t = 0; t < NA; ++t) T[t] = f32x4{0, 0, 0, 0};
#pragma unroll
    for (int p = 0; p < KT; ++p) E[p] = f32x4{0, 0, 0, 0};
    const int nchunks = (n + 31) >> 5;
    StagedRegs<KT, true> sr;
    stage_loads<KT, true>(sr, indices, values, factors, p0, n, lane);
    stage_commit<KT, true>(sr, base, lane);
#pragma unroll 1
    for (int ch = 0; ch < nchunks; ++ch) {
        if (ch + 1 < nchunks)
            stage_loads<KT, true>(sr, indices, values, factors,
                                  p0 + (long long)(ch + 1) * 32,
                                  n - (ch + 1) * 32, lane);
        WREG_FENCE();
        typename FragT<true>::type frag[KT + 1];
        read_frags<KT, true>(base + (ch & 1) * SB, lane, frag);
        WREG_FENCE();
        mfma_lower_tiles<KT, true>(frag, T);
        mfma_ext_tiles<KT, true>(frag, E);
        if (ch + 1 < nchunks)
            stage_commit<KT, true>(sr, base + ((ch + 1) & 1) * SB, lane);
    }
    wavefused_dump_b<KT>(E, scr, g4, li);
    WREG_FENCE();
    float x0 = (lane < K) ? scr[lane] + scr[64 + lane] : 0.0f;
    WREG_FENCE();
    wavefused_diag<KT>(T, reg * (float)n, g4, li);
    float d0 = 1.0f;
    wreg_panels<KT, 0>(T, scr, lane, g4, li, d0);
    const float id0 = d0 > 0.0f ? 1.0f / d0 : 0.0f;
    wreg_forward<KT, 0>(T, scr, x0, id0, lane, g4, li);
    x0 *= id0;
    wreg_backward<KT, KT - 1>(T, scr, x0, id0, lane, g4, li);
    if (lane < K) {
        out_f32[e * K + lane] = x0;
        if (out_bf16) out_bf16[e * K + lane] = f2bf(x0);
        if (out_fp8) out_fp8[e * K + lane] = f2fp8(x0);
    }
}

extern "C" hipError_t fma_als_solve_wavefused_db(
    int k, const long long* indptr, const int* indices, const float* values,
    const void* factors, float* out_f32, unsigned short* out_bf16,
    unsigned char* out_fp8, const int* row_order, long long nrows,
    float reg, hipStream_t stream) {
    if (k % 16 || k < 16 || k > 64 || nrows <= 0) return hipErrorInvalidValue;
    dim3 grid((unsigned)((nrows + 3) / 4)), block(256);
    switch (k / 16) {
        case 1: k_als_solve_wavefused_db<1><<<grid, block, 0, stream>>>(indptr, indices, values, factors, out_f32, out_bf16, out_fp8, row_order, nrows, reg); break;
        case 2: k_als_solve_wavefused_db<2><<<grid, block, 0, stream>>>(indptr, indices, values, factors, out_f32, out_bf16, out_fp8, row_order, nrows, reg); break;
        case 3: k_als_solve_wavefused_db<3><<<grid, block, 0, stream>>>(indptr, indices, values, factors, out_f32, out_bf16, out_fp8, row_order, nrows, reg); break;
        case 4: k_als_solve_wavefused_db<4><<<grid, block, 0, stream>>>(indptr, indices, values, factors, out_f32, out_bf16, out_fp8, row_order, nrows, reg); break;
        default: return hipErrorInvalidValue;
    }
    return hipGetLastError();
}

template <int KT, bool FP8>
__launch_bounds__(256)
// waves_per_eu(5): 96 VGPRs with ~12-17 spilled (52-60 B/lane scratch) buys
// a 5th wave/SIMD over the natural 105-reg allocation — measured 3.23 ->
// 2.65 ms/iter on the rank-64 headline (the spill traffic hides behind the
// same gather stalls the extra wave fills)
__attribute__((amdgpu_waves_per_eu(5)))
__global__ void k_als_solve_wavefused(const long long* __restrict__ indptr,
                                      const int* __restrict__ indices,
                                      const float* __restrict__ values,
                                      const void* __restrict__ factors,
                                      float* __restrict__ out_f32,
                                      unsigned short* __restrict__ out_bf16,
                                      unsigned char* __restrict__ out_fp8,
                                      const int* __restrict__ row_order,
                                      long long nrows, float reg) {
    constexpr int K = KT * 16;
    static_assert(K <= 64, "wave-fused path handles k <= 64");
    constexpr int NA = KT * (KT + 1) / 2;
    constexpr int TROW = FP8 ? Geo<KT>::TROW8 : Geo<KT>::TROW;
    constexpr int SB = (K + 16) * TROW;            // stage bytes per wave
    // split-wave staging: a chunk's gather tasks only need NT lanes (fp8
    // K=64: 32); when two chunks' tasks fit in one wave, the upper half
    // stages chunk ch+1 into a second buffer in the SAME instructions —
    // double the gathers in flight at zero extra registers or occupancy
    constexpr int NT = FP8 ? 8 * (K / 16) : K;
    constexpr int NCHW = (2 * NT <= WAVE && NT >= 8) ? 2 : 1;
    constexpr int SCRF = (K * 17 > 1040) ? K * 17 : 1040;  // solver scratch
    constexpr int WB0 = (NCHW * SB > SCRF * 4) ? NCHW * SB : SCRF * 4;
    constexpr int WB = (WB0 + 15) & ~15;
    __shared__ __align__(16) char smem[4 * WB];
    const int w = __builtin_amdgcn_readfirstlane(threadIdx.x >> 6);
    const int lane = threadIdx.x & 63;
    const long long e0 = (long long)blockIdx.x * 4 + w;
    if (e0 >= nrows) return;
    const long long e = row_order ? row_order[e0] : e0;
    char* buf = smem + (long long)w * WB;
    float* scr = (float*)buf;          // union: stage dies before the solve
    const int g4 = lane >> 4, li = lane & 15;
    const long long p0 = indptr[e];
    const int n = (int)(indptr[e + 1] - p0);
    if (n == 0) {
        if (lane < K) {
            out_f32[e * K + lane] = 0.0f;
            if (out_bf16) out_bf16[e * K + lane] = 0;
            if (out_fp8) out_fp8[e * K + lane] = 0;
        }
        return;
    }
    {   // zero the EXT pad rows K+2..K+15 of my NCHW buffers (wave-local)
        constexpr int SEGS = FP8 ? 8 : 16;
        for (int i = lane; i < NCHW * 14 * SEGS; i += WAVE) {
            const int b = i / (14 * SEGS), r2 = i % (14 * SEGS);
            const int row = K + 2 + r2 / SEGS, seg = r2 % SEGS;
            *(unsigned*)(buf + b * SB + (long long)row * TROW + seg * 4) =
                0u;
        }
    }
    f32x4 T[NA], E[KT];
#pragma unroll
    for (int t = 0; t < NA; ++t) T[t] = f32x4{0, 0, 0, 0};
#pragma unroll
    for (int p = 0; p < KT; ++p) E[p] = f32x4{0, 0, 0, 0};
    const int nchunks = (n + 31) >> 5;
    for (int ch = 0; ch < nchunks; ch += NCHW) {
        if constexpr (NCHW == 2) {
            const int sub = (lane >= NT) ? 1 : 0;
            const int lch = ch + sub;
            if (lch < nchunks && lane < 2 * NT)
                stage_chunk_w<KT, FP8, NT>(buf + sub * SB, indices, values,
                                           factors,
                                           p0 + (long long)lch * 32,
                                           n - lch * 32, lane - sub * NT);
        } else {
            stage_chunk_w<KT, FP8>(buf, indices, values, factors,
                                   p0 + (long long)ch * 32, n - ch * 32,
                                   lane);
        }
        WREG_FENCE();                  // cross-lane LDS write -> read
        {
            typename FragT<FP8>::type frag[KT + 1];
            read_frags<KT, FP8>(buf, lane, frag);
            mfma_lower_tiles<KT, FP8>(frag, T);
            mfma_ext_tiles<KT, FP8>(frag, E);
        }
        if constexpr (NCHW == 2) {
            if (ch + 1 < nchunks) {
                typename FragT<FP8>::type frag[KT + 1];
                read_frags<KT, FP8>(buf + SB, lane, frag);
                mfma_lower_tiles<KT, FP8>(frag, T);
                mfma_ext_tiles<KT, FP8>(frag, E);
            }
        }
        WREG_FENCE();                  // reads drained before next stage
    }
    // b = bhi + blo redistributed to lane = row; then lambda*n*I
    wavefused_dump_b<KT>(E, scr, g4, li);
    WREG_FENCE();
    float x0 = (lane < K) ? scr[lane] + scr[64 + lane] : 0.0f;
    WREG_FENCE();                      // b read before panel dumps reuse scr
    wavefused_diag<KT>(T, reg * (float)n, g4, li);
    // in-register LDL + substitution (shared with k_ldl_solve_wave_reg)
    float d0 = 1.0f;
    wreg_panels<KT, 0>(T, scr, lane, g4, li, d0);
    const float id0 = d0 > 0.0f ? 1.0f / d0 : 0.0f;
    wreg_forward<KT, 0>(T, scr, x0, id0, lane, g4, li);
    x0 *= id0;
    wreg_backward<KT, KT - 1>(T, scr, x0, id0, lane, g4, li);
    if (lane < K) {
        out_f32[e * K + lane] = x0;
        if (out_bf16) out_bf16[e * K + lane] = f2bf(x0);
        if (out_fp8) out_fp8[e * K + lane] = f2fp8(x0);
    }
}

extern "C" hipError_t fma_als_solve_wavefused(
    int k, int fp8, const long long* indptr, const int* indices,
    const float* values, const void* factors, float* out_f32,
    unsigned short* out_bf16, unsigned char* out_fp8, const int* row_order,
    long long nrows, float reg, hipStream_t stream) {
    if (k % 16 || k < 16 || k > 64 || nrows <= 0) return hipErrorInvalidValue;
    dim3 grid((unsigned)((nrows + 3) / 4)), block(256);
#define WF_CASE(KT)                                                           \
    case KT:                                                                  \
        if (fp8)                                                              \
            k_als_solve_wavefused<KT, true><<<grid, block, 0, stream>>>(      \
                indptr, indices, values, factors, out_f32, out_bf16,          \
                out_fp8, row_order, nrows, reg);                              \
        else                                                                  \
            k_als_solve_wavefused<KT, false><<<grid, block, 0, stream>>>(     \
                indptr, indices, values, factors, out_f32, out_bf16,          \
                out_fp8, row_order, nrows, reg);                              \
        break;
    switch (k / 16) {
        WF_CASE(1)
        WF_CASE(2)
        WF_CASE(3)
        WF_CASE(4)
        default: return hipErrorInvalidValue;
    }
#undef WF_CASE
    return hipGetLastError();
}

extern "C" hipError_t fma_ldl_solve_wave(
    int k, const float* A_in, const float* b_in, float* x_out,
    unsigned short* x_bf16, unsigned char* x_fp8, long long nrows,
    hipStream_t stream) {
    if (k % 16 || k < 16 || k > 64 || nrows <= 0) return hipErrorInvalidValue;
    dim3 grid((unsigned)((nrows + 3) / 4)), block(256);
    switch (k / 16) {
        case 1: k_ldl_solve_wave<1><<<grid, block, 0, stream>>>(A_in, b_in, x_out, x_bf16, x_fp8, nrows); break;
        case 2: k_ldl_solve_wave<2><<<grid, block, 0, stream>>>(A_in, b_in, x_out, x_bf16, x_fp8, nrows); break;
        case 3: k_ldl_solve_wave<3><<<grid, block, 0, stream>>>(A_in, b_in, x_out, x_bf16, x_fp8, nrows); break;
        case 4: k_ldl_solve_wave<4><<<grid, block, 0, stream>>>(A_in, b_in, x_out, x_bf16, x_fp8, nrows); break;
        default: return hipErrorInvalidValue;
    }
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

extern "C" hipError_t fma_mfma_probe_fp8(const unsigned char* Xt,
                                         const unsigned char* Yt, float* C,
                                         hipStream_t stream) {
    k_mfma_probe_fp8<<<dim3(1), dim3(64), 0, stream>>>(Xt, Yt, C);
    return hipGetLastError();
}

extern "C" const char* fma_err_str(int err) {
    return hipGetErrorString((hipError_t)err);
}

// ------- wave-QUAD fused ALS solve (64 < k <= 128, fp8 gathers) -------
// The k=128 block-fused kernel serializes its LDL at Gramian occupancy and
// the modular path pays an 80 GB A round trip; this kernel extends the
// k<=64 wave-fused design: all FOUR waves of a block own ONE entity, each
// holding a quarter of the lower-triangle MFMA C-fragment tiles (t % 4 ==
// wave).  The Gramian accumulates straight into those registers (4 staged
// chunks in flight per entity), then an in-register LDL runs with block
// barriers at phase edges: panel columns bounce through LDS scratch,
// 16-col panels factor on wave 0 (2 rows/lane), trailing updates run on
// every wave's own tiles via f32 MFMA, and the substitutions keep x as
// TWO registers per lane (rows lane and lane+64) redundantly on each wave
// so the serial chain needs no cross-wave traffic.  One entity per block
// keeps every barrier trivially uniform (n==0 exits whole-block).

template <int KT, int P, int t = 0>
DEV_INLINE void wq_mfma(const fp8x8* frag, f32x4* T, f32x4* E) {
    constexpr int NA = KT * (KT + 1) / 2;
    if constexpr (t < NA) {
        if constexpr ((t & 3) == P) {
            constexpr int I = lo_tile_i(t);
            constexpr int J = t - (I * (I + 1)) / 2;
            T[t >> 2] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
                frag[I], frag[J], T[t >> 2], 0, 0, 0);
        }
        wq_mfma<KT, P, t + 1>(frag, T, E);
    } else if constexpr (t < NA + KT) {
        constexpr int Pt = t - NA;
        if constexpr ((Pt & 3) == P)
            E[Pt >> 2] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
                frag[Pt], frag[KT], E[Pt >> 2], 0, 0, 0);
        wq_mfma<KT, P, t + 1>(frag, T, E);
    }
}

template <int KT, int P, int Pt = 0>
DEV_INLINE void wq_dump_b(const f32x4* E, float* scr, int g4, int li) {
    constexpr int K = KT * 16;
    if constexpr (Pt < KT) {
        if constexpr ((Pt & 3) == P) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                if (li == 0) scr[Pt * 16 + g4 * 4 + r] = E[Pt >> 2][r];
                else if (li == 1)
                    scr[K + Pt * 16 + g4 * 4 + r] = E[Pt >> 2][r];
            }
        }
        wq_dump_b<KT, P, Pt + 1>(E, scr, g4, li);
    }
}

template <int KT, int P, int I = 0>
DEV_INLINE void wq_diag(f32x4* T, float regn, int g4, int li) {
    if constexpr (I < KT) {
        constexpr int t = tri_off(I, I);
        if constexpr ((t & 3) == P) {
#pragma unroll
            for (int r = 0; r < 4; ++r)
                if (li == g4 * 4 + r) {
                    const float dd = T[t >> 2][r] + regn;
                    T[t >> 2][r] = dd <= 0.0f ? 1.0f : dd;
                }
        }
        wq_diag<KT, P, I + 1>(T, regn, g4, li);
    }
}

// dump / load the PI-th 16-column panel (tiles (I,PI), I>=PI) to/from scr
// rows relative to P0, stride 17 — each wave handles its owned tiles
template <int KT, int PI, int P, int I = PI>
DEV_INLINE void wq_dump_col(const f32x4* T, float* scr, int g4, int li) {
    if constexpr (I < KT) {
        constexpr int t = tri_off(I, PI);
        if constexpr ((t & 3) == P) {
#pragma unroll
            for (int r = 0; r < 4; ++r)
                scr[((I - PI) * 16 + g4 * 4 + r) * 17 + li] = T[t >> 2][r];
        }
        wq_dump_col<KT, PI, P, I + 1>(T, scr, g4, li);
    }
}

template <int KT, int PI, int P, int I = PI>
DEV_INLINE void wq_load_col(f32x4* T, const float* scr, int g4, int li) {
    if constexpr (I < KT) {
        constexpr int t = tri_off(I, PI);
        if constexpr ((t & 3) == P) {
#pragma unroll
            for (int r = 0; r < 4; ++r)
                T[t >> 2][r] = scr[((I - PI) * 16 + g4 * 4 + r) * 17 + li];
        }
        wq_load_col<KT, PI, P, I + 1>(T, scr, g4, li);
    }
}

template <int KT, int PI, int P, int RB, int CB>
DEV_INLINE void wq_trail(f32x4* T, const float* scr, const float* ndk,
                         int g4, int li) {
    if constexpr (RB < KT) {
        if constexpr (CB > RB) {
            wq_trail<KT, PI, P, RB + 1, PI + 1>(T, scr, ndk, g4, li);
        } else {
            constexpr int t = tri_off(RB, CB);
            if constexpr ((t & 3) == P) {
                f32x4 acc = T[t >> 2];
#pragma unroll
                for (int kk = 0; kk < 4; ++kk) {
                    const int pc = 4 * kk + g4;
                    const float a = scr[((RB - PI) * 16 + li) * 17 + pc];
                    const float b =
                        scr[((CB - PI) * 16 + li) * 17 + pc] * ndk[kk];
                    acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc,
                                                               0, 0, 0);
                }
                T[t >> 2] = acc;
            }
            wq_trail<KT, PI, P, RB, CB + 1>(T, scr, ndk, g4, li);
        }
    }
}

// LDL panel loop: dump column -> factor (wave 0, 2 rows/lane) -> load the
// factored column back -> trailing MFMA on owned tiles.  Uniform barriers.
template <int KT, int P, int PI = 0>
DEV_INLINE void wq_panels(f32x4* T, float* scr, int lane, int g4, int li,
                          int w, float& da, float& db) {
    if constexpr (PI < KT) {
        constexpr int K = KT * 16;
        constexpr int P0 = 16 * PI;
        constexpr int NR = K - P0;          // rows in this panel image
        wq_dump_col<KT, PI, P>(T, scr, g4, li);
        __syncthreads();
        if (w == 0) {   // factor: lane owns panel rows rel lane, lane+64
            float sa[16], sb[16];
            const bool av = lane < NR, bv = lane + 64 < NR;
            if (av) {
#pragma unroll
                for (int cc = 0; cc < 16; ++cc) sa[cc] = scr[lane * 17 + cc];
            }
            if (bv) {
#pragma unroll
                for (int cc = 0; cc < 16; ++cc)
                    sb[cc] = scr[(lane + 64) * 17 + cc];
            }
#pragma unroll
            for (int jj = 0; jj < 15; ++jj) {
                const float dj = __shfl(sa[jj], jj, WAVE);
                const float dinv = dj > 0.0f ? 1.0f / dj : 0.0f;
                const float la = sa[jj] * dinv, lb = sb[jj] * dinv;
#pragma unroll
                for (int cc = jj + 1; cc < 16; ++cc) {
                    const float colj = __shfl(sa[jj], cc, WAVE);
                    sa[cc] -= la * colj;
                    sb[cc] -= lb * colj;
                }
            }
            if (av) {
#pragma unroll
                for (int cc = 0; cc < 16; ++cc) scr[lane * 17 + cc] = sa[cc];
            }
            if (bv) {
#pragma unroll
                for (int cc = 0; cc < 16; ++cc)
                    scr[(lane + 64) * 17 + cc] = sb[cc];
            }
        }
        __syncthreads();
        // raw diagonal of this panel's rows -> per-lane d registers
        if ((lane >> 4) == PI)
            da = scr[(lane - P0) * 17 + (lane - P0)];
        if constexpr (P0 >= 64) {
            if ((lane >> 4) + 4 == PI)
                db = scr[(lane + 64 - P0) * 17 + (lane + 64 - P0)];
        }
        // load the FACTORED panel column back into the register tiles (the
        // substitutions dump L from T; without this they would read raw A)
        wq_load_col<KT, PI, P>(T, scr, g4, li);
        if constexpr (PI + 1 < KT) {
            float ndk[4];
#pragma unroll
            for (int kk = 0; kk < 4; ++kk) {
                const float d = scr[(4 * kk + g4) * 17 + (4 * kk + g4)];
                ndk[kk] = d > 0.0f ? -1.0f / d : 0.0f;
            }
            wq_trail<KT, PI, P, PI + 1, PI + 1>(T, scr, ndk, g4, li);
        }
        __syncthreads();   // trailing reads done before next panel's dump
        wq_panels<KT, P, PI + 1>(T, scr, lane, g4, li, w, da, db);
    }
}

// forward substitution: x as 2 regs/lane (rows lane, 64+lane), redundant
// on every wave; column-J tiles bounce through scr per block of 16
template <int KT, int P, int J = 0>
DEV_INLINE void wq_forward(const f32x4* T, float* scr, float& xa, float& xb,
                           float ida, float idb, int lane, int g4, int li) {
    if constexpr (J < KT) {
        constexpr int K = KT * 16;
        constexpr int B0 = 16 * J;
        wq_dump_col<KT, J, P>(T, scr, g4, li);
        __syncthreads();
#pragma unroll
        for (int t = 0; t < 16; ++t) {
            const int j = B0 + t;
            if (j >= K - 1) break;
            float zj;
            if constexpr (B0 < 64) {
                zj = __shfl(xa, j, WAVE) * __shfl(ida, j, WAVE);
            } else {
                zj = __shfl(xb, j - 64, WAVE) * __shfl(idb, j - 64, WAVE);
            }
            if (lane > j) xa -= scr[(lane - B0) * 17 + t] * zj;
            if (64 + lane > j && 64 + lane < K)
                xb -= scr[(64 + lane - B0) * 17 + t] * zj;
        }
        __syncthreads();
        wq_forward<KT, P, J + 1>(T, scr, xa, xb, ida, idb, lane, g4, li);
    }
}

// backward: row image of tile-row I (16 rows x K cols, stride K+1)
template <int KT, int I, int P, int J = 0>
DEV_INLINE void wq_dump_row(const f32x4* T, float* scr, int g4, int li) {
    if constexpr (J <= I) {
        constexpr int K = KT * 16;
        constexpr int t = tri_off(I, J);
        if constexpr ((t & 3) == P) {
#pragma unroll
            for (int r = 0; r < 4; ++r)
                scr[(g4 * 4 + r) * (K + 1) + J * 16 + li] = T[t >> 2][r];
        }
        wq_dump_row<KT, I, P, J + 1>(T, scr, g4, li);
    }
}

template <int KT, int P, int I = KT - 1>
DEV_INLINE void wq_backward(const f32x4* T, float* scr, float& xa, float& xb,
                            float ida, float idb, int lane, int g4, int li) {
    if constexpr (I >= 0) {
        constexpr int K = KT * 16;
        constexpr int B0 = 16 * I;
        wq_dump_row<KT, I, P>(T, scr, g4, li);
        __syncthreads();
#pragma unroll
        for (int t = 15; t >= 0; --t) {
            const int c = B0 + t;
            if (c < 1) break;
            float xc;
            if constexpr (B0 >= 64) {
                xc = __shfl(xb, c - 64, WAVE);
            } else {
                xc = __shfl(xa, c, WAVE);
            }
            if (lane < c) xa -= scr[t * (K + 1) + lane] * ida * xc;
            if (64 + lane < c && 64 + lane < K)
                xb -= scr[t * (K + 1) + 64 + lane] * idb * xc;
        }
        __syncthreads();
        wq_backward<KT, P, I - 1>(T, scr, xa, xb, ida, idb, lane, g4, li);
    }
}

template <int KT>
__launch_bounds__(256)
__global__ void k_als_solve_wavefused2(const long long* __restrict__ indptr,
                                       const int* __restrict__ indices,
                                       const float* __restrict__ values,
                                       const unsigned char* __restrict__ factors,
                                       float* __restrict__ out_f32,
                                       unsigned char* __restrict__ out_fp8,
                                       const int* __restrict__ row_order,
                                       long long nrows, float reg) {
    constexpr int K = KT * 16;
    static_assert(K > 64 && K <= 128, "wave-quad path: 64 < k <= 128");
    constexpr int NA = KT * (KT + 1) / 2;
    constexpr int NOWN = (NA + 3) / 4;
    constexpr int NE = (KT + 3) / 4;
    constexpr int TROW = Geo<KT>::TROW8;
    constexpr int SB = (K + 16) * TROW;
    constexpr int SCRF = (K * 17 > 16 * (K + 1)) ? K * 17 : 16 * (K + 1);
    constexpr int BB0 = (4 * SB > SCRF * 4) ? 4 * SB : SCRF * 4;
    constexpr int BB = (BB0 + 15) & ~15;
    __shared__ __align__(16) char smem[BB];
    const int w = __builtin_amdgcn_readfirstlane(threadIdx.x >> 6);
    const int lane = threadIdx.x & 63;
    const int g4 = lane >> 4, li = lane & 15;
    float* scr = (float*)smem;
    long long e = blockIdx.x;
    if (e >= nrows) return;
    if (row_order) e = row_order[e];
    const long long p0 = indptr[e];
    const int n = (int)(indptr[e + 1] - p0);
    if (n == 0) {   // block-uniform: every wave exits together
        for (int c = threadIdx.x; c < K; c += 256) {
            out_f32[e * K + c] = 0.0f;
            if (out_fp8) out_fp8[e * K + c] = 0;
        }
        return;
    }
    {   // zero EXT pad rows of my stage buffer
        char* buf = smem + (long long)w * SB;
        for (int i = lane; i < 14 * 8; i += WAVE) {
            const int row = K + 2 + i / 8, seg = i % 8;
            *(unsigned*)(buf + (long long)row * TROW + seg * 4) = 0u;
        }
    }
    f32x4 T[NOWN], E[NE];
#pragma unroll
    for (int t = 0; t < NOWN; ++t) T[t] = f32x4{0, 0, 0, 0};
#pragma unroll
    for (int t = 0; t < NE; ++t) E[t] = f32x4{0, 0, 0, 0};
    const int nch = (n + 31) >> 5;
    for (int r0 = 0; r0 < nch; r0 += 4) {
        if (r0 + w < nch)
            stage_chunk_w<KT, true>(smem + (long long)w * SB, indices,
                                    values, factors,
                                    p0 + (long long)(r0 + w) * 32,
                                    n - (r0 + w) * 32, lane);
        __syncthreads();
        const int ns = min(4, nch - r0);
#pragma unroll 1
        for (int sc = 0; sc < ns; ++sc) {
            fp8x8 frag[KT + 1];
            read_frags<KT, true>(smem + (long long)sc * SB, lane, frag);
            switch (w) {
                case 0: wq_mfma<KT, 0>(frag, T, E); break;
                case 1: wq_mfma<KT, 1>(frag, T, E); break;
                case 2: wq_mfma<KT, 2>(frag, T, E); break;
                default: wq_mfma<KT, 3>(frag, T, E); break;
            }
        }
        __syncthreads();
    }
    // b = hi + lo via scr rows [0,K) hi, [K,2K) lo
    switch (w) {
        case 0: wq_dump_b<KT, 0>(E, scr, g4, li); break;
        case 1: wq_dump_b<KT, 1>(E, scr, g4, li); break;
        case 2: wq_dump_b<KT, 2>(E, scr, g4, li); break;
        default: wq_dump_b<KT, 3>(E, scr, g4, li); break;
    }
    __syncthreads();
    float xa = scr[lane] + scr[K + lane];
    float xb = (64 + lane < K) ? scr[64 + lane] + scr[K + 64 + lane] : 0.0f;
    __syncthreads();   // b read before the panel dumps reuse scr
    float da = 1.0f, db = 1.0f;
    const float regn = reg * (float)n;
    switch (w) {
        case 0:
            wq_diag<KT, 0>(T, regn, g4, li);
            wq_panels<KT, 0>(T, scr, lane, g4, li, w, da, db);
            break;
        case 1:
            wq_diag<KT, 1>(T, regn, g4, li);
            wq_panels<KT, 1>(T, scr, lane, g4, li, w, da, db);
            break;
        case 2:
            wq_diag<KT, 2>(T, regn, g4, li);
            wq_panels<KT, 2>(T, scr, lane, g4, li, w, da, db);
            break;
        default:
            wq_diag<KT, 3>(T, regn, g4, li);
            wq_panels<KT, 3>(T, scr, lane, g4, li, w, da, db);
            break;
    }
    const float ida = da > 0.0f ? 1.0f / da : 0.0f;
    const float idb = db > 0.0f ? 1.0f / db : 0.0f;
    switch (w) {
        case 0: wq_forward<KT, 0>(T, scr, xa, xb, ida, idb, lane, g4, li); break;
        case 1: wq_forward<KT, 1>(T, scr, xa, xb, ida, idb, lane, g4, li); break;
        case 2: wq_forward<KT, 2>(T, scr, xa, xb, ida, idb, lane, g4, li); break;
        default: wq_forward<KT, 3>(T, scr, xa, xb, ida, idb, lane, g4, li); break;
    }
    xa *= ida;
    xb *= idb;
    switch (w) {
        case 0: wq_backward<KT, 0>(T, scr, xa, xb, ida, idb, lane, g4, li); break;
        case 1: wq_backward<KT, 1>(T, scr, xa, xb, ida, idb, lane, g4, li); break;
        case 2: wq_backward<KT, 2>(T, scr, xa, xb, ida, idb, lane, g4, li); break;
        default: wq_backward<KT, 3>(T, scr, xa, xb, ida, idb, lane, g4, li); break;
    }
    if (w == 0) {
        out_f32[e * K + lane] = xa;
        if (out_fp8) out_fp8[e * K + lane] = f2fp8(xa);
        if (64 + lane < K) {
            out_f32[e * K + 64 + lane] = xb;
            if (out_fp8) out_fp8[e * K + 64 + lane] = f2fp8(xb);
        }
    }
}

extern "C" hipError_t fma_als_solve_wavefused2(
    int k, const long long* indptr, const int* indices, const float* values,
    const unsigned char* factors, float* out_f32, unsigned char* out_fp8,
    const int* row_order, long long nrows, float reg, hipStream_t stream) {
    if (k % 16 || k <= 64 || k > 128 || nrows <= 0)
        return hipErrorInvalidValue;
    dim3 grid((unsigned)nrows), block(256);
    switch (k / 16) {
        case 5: k_als_solve_wavefused2<5><<<grid, block, 0, stream>>>(indptr, indices, values, factors, out_f32, out_fp8, row_order, nrows, reg); break;
        case 6: k_als_solve_wavefused2<6><<<grid, block, 0, stream>>>(indptr, indices, values, factors, out_f32, out_fp8, row_order, nrows, reg); break;
        case 7: k_als_solve_wavefused2<7><<<grid, block, 0, stream>>>(indptr, indices, values, factors, out_f32, out_fp8, row_order, nrows, reg); break;
        case 8: k_als_solve_wavefused2<8><<<grid, block, 0, stream>>>(indptr, indices, values, factors, out_f32, out_fp8, row_order, nrows, reg); break;
        default: return hipErrorInvalidValue;
    }
    return hipGetLastError();
}
