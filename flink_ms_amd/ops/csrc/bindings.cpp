// Python bindings for the flink_ms_amd HIP kernels.
//
// This translation unit is deliberately HIP-header-free: kernels live in the
// .hip files behind extern "C" launchers (enum hipError_t <-> int,
// hipStream_t <-> void*), so only hipcc ever sees device code and this file
// only sees torch tensors.  The current HIP stream is passed in from Python
// (torch.cuda.current_stream().cuda_stream).

#include <torch/extension.h>

#include <charconv>
#include <cstdint>
#include <cstdlib>
#include <thread>
#include <vector>

extern "C" {
int fma_als_solve_fused(int k, const int64_t* indptr, const int* indices,
                        const float* values, const unsigned short* factors,
                        float* out_f32, unsigned short* out_bf16,
                        const int* row_order, int64_t nrows, float reg,
                        void* stream);
int fma_gramian(int k, const int64_t* indptr, const int* indices,
                const float* values, const unsigned short* factors,
                float* A_out, float* b_out, const int* row_order,
                int64_t nrows, float reg, void* stream);
int fma_gramian_fp8(int k, const int64_t* indptr, const int* indices,
                    const float* values, const unsigned char* factors,
                    float* A_out, float* b_out, const int* row_order,
                    int64_t nrows, float reg, void* stream);
int fma_als_solve_fused_fp8(int k, const int64_t* indptr, const int* indices,
                            const float* values, const unsigned char* factors,
                            float* out_f32, unsigned char* out_fp8,
                            const int* row_order, int64_t nrows, float reg,
                            void* stream);
int fma_cholesky_solve(int k, const float* A_in, const float* b_in,
                       float* x_out, int64_t nrows, void* stream);
int fma_cholesky_solve_ph(int k, const float* A_in, const float* b_in,
                          float* x_out, int64_t nrows, int phases,
                          void* stream);
int fma_sdca_pass(const int64_t* indptr, const int* indices,
                  const float* values, const float* y, const float* norms_sq,
                  const int* perm, float* alpha, float* v, int64_t nrows,
                  float scale, void* stream);
int fma_svm_margins(const int64_t* indptr, const int* indices,
                    const float* values, const float* w, float* out,
                    int64_t nrows, void* stream);
int fma_predict_dot(const unsigned short* U, const unsigned short* V,
                    const int64_t* u_idx, const int64_t* i_idx,
                    float* out, int64_t nq, int k, void* stream);
int fma_sgd_update(unsigned short* U, unsigned short* V,
                   const int64_t* u_idx, const int64_t* i_idx,
                   const float* r, float* err_out, int64_t nq, int k,
                   float lr, float user_reg, float item_reg, void* stream);
int fma_als_solve_wavefused_db(int k, const int64_t* indptr,
                               const int* indices, const float* values,
                               const void* factors, float* out_f32,
                               unsigned short* out_bf16,
                               unsigned char* out_fp8,
                               const int* row_order, int64_t nrows,
                               float reg, void* stream);
int fma_als_solve_wavefused(int k, int fp8, const int64_t* indptr,
                            const int* indices, const float* values,
                            const void* factors, float* out_f32,
                            unsigned short* out_bf16, unsigned char* out_fp8,
                            const int* row_order, int64_t nrows, float reg,
                            void* stream);
int fma_als_solve_wavefused2(int k, const int64_t* indptr,
                             const int* indices, const float* values,
                             const unsigned char* factors, float* out_f32,
                             unsigned char* out_fp8, const int* row_order,
                             int64_t nrows, float reg, void* stream);
int fma_ldl_solve_wave(int k, const float* A_in, const float* b_in,
                       float* x_out, unsigned short* x_bf16,
                       unsigned char* x_fp8, int64_t nrows, void* stream);
int fma_ldl_solve_wave_reg(int k, const float* A_in, const float* b_in,
                           float* x_out, unsigned short* x_bf16,
                           unsigned char* x_fp8, int64_t nrows,
                           void* stream);
int fma_mfma_probe_f32(const float* A, const float* B, float* D, void* stream);
int fma_mfma_probe_bf16(const unsigned short* Xt, const unsigned short* Yt,
                        float* C, void* stream);
int fma_mfma_probe_fp8(const unsigned char* Xt, const unsigned char* Yt,
                       float* C, void* stream);
int fma_dbg_stage_dump(int k, const int64_t* indptr, const int* indices,
                       const float* values, const unsigned short* factors,
                       unsigned short* out, void* stream);
int fma_dbg_frag_dump(int k, const int64_t* indptr, const int* indices,
                      const float* values, const unsigned short* factors,
                      unsigned short* out, void* stream);
int fma_dbg_gramian(int k, const int64_t* indptr, const int* indices,
                    const float* values, const unsigned short* factors,
                    float* A_out, float* bhi_out, float* blo_out,
                    int64_t nrows, float reg, void* stream);
const char* fma_err_str(int err);
}

namespace {

// ---------------------------------------------------------------- ingest
// Threaded parser for ALS model-row blocks ("<id>,<U|I>,<f;f;...>\n" x n):
// the native replacement for the reference's Kafka-consumer row decode
// (ALSKafkaConsumer.java:73-82).  One pass splits lines across threads;
// each thread strtoll/strtof-parses its share in place.  Returns
// (ids int64[n], kinds uint8[n] (0=U,1=I), factors fp32[n][k],
//  payload byte offsets int64[n], payload byte lengths int64[n], n_bad).
// Rows with a wrong factor count or malformed fields are marked kind=255
// and skipped by the caller's scalar fallback.
py::tuple parse_als_block(py::object text_o, int64_t k) {
    // raw CPython buffer protocol: accepts bytes AND mmap objects
    // (zero-copy parse of a disk-resident model file — the
    // larger-than-memory serving path)
    Py_buffer view;
    if (PyObject_GetBuffer(text_o.ptr(), &view, PyBUF_SIMPLE) != 0)
        throw py::error_already_set();
    char* base = (char*)view.buf;
    Py_ssize_t total = view.len;
    // line index (single pass; cheap relative to float parsing)
    std::vector<std::pair<int64_t, int64_t>> lines;
    int64_t start = 0;
    for (int64_t i = 0; i < total; ++i) {
        if (base[i] == '\n') {
            if (i > start) lines.emplace_back(start, i);
            start = i + 1;
        }
    }
    if (start < total) lines.emplace_back(start, total);
    const int64_t n = (int64_t)lines.size();
    auto ids = torch::empty({n}, torch::kInt64);
    auto kinds = torch::empty({n}, torch::kUInt8);
    auto facs = torch::empty({n, k}, torch::kFloat32);
    auto poffs = torch::empty({n}, torch::kInt64);
    auto plens = torch::empty({n}, torch::kInt64);
    int64_t* idp = ids.data_ptr<int64_t>();
    uint8_t* kp = kinds.data_ptr<uint8_t>();
    float* fp = facs.data_ptr<float>();
    int64_t* op = poffs.data_ptr<int64_t>();
    int64_t* lp = plens.data_ptr<int64_t>();
    std::atomic<int64_t> bad{0};
    const int nthreads = (int)std::min<int64_t>(
        std::max<int64_t>(1, n / 20000),
        (int64_t)std::thread::hardware_concurrency());
    auto work = [&](int64_t lo, int64_t hi) {
        int64_t my_bad = 0;
        for (int64_t r = lo; r < hi; ++r) {
            // std::from_chars everywhere: bounded by `end` (the input may
            // be an mmap with no trailing NUL) and faster than strtod
            const char* q = base + lines[r].first;
            const char* end = base + lines[r].second;
            kp[r] = 255;
            long long id;
            auto ir = std::from_chars(q, end, id, 10);
            if (ir.ec != std::errc() || ir.ptr >= end || *ir.ptr != ',') {
                ++my_bad;
                continue;
            }
            q = ir.ptr + 1;
            uint8_t kind;
            if (*q == 'U') kind = 0;
            else if (*q == 'I') kind = 1;
            else { ++my_bad; continue; }
            ++q;
            if (q >= end || *q != ',') { ++my_bad; continue; }
            ++q;
            op[r] = q - base;
            lp[r] = end - q;
            float* frow = fp + r * k;
            int64_t c = 0;
            bool ok = true;
            while (q < end && c < k) {
                // parse as double then narrow: float-subnormal payloads
                // (e.g. 4.9e-324) must flush like strtof, not reject
                double dv;
                auto fr = std::from_chars(q, end, dv,
                                          std::chars_format::general);
                if (fr.ec == std::errc::result_out_of_range) {
                    dv = 0.0;  // double-range underflow only
                } else if (fr.ec != std::errc()) {
                    ok = false;
                    break;
                }
                frow[c] = (float)dv;
                q = fr.ptr;
                ++c;
                if (q < end) {
                    if (*q == ';') ++q;
                    else { ok = false; break; }
                }
            }
            if (!ok || c != k || q < end) { ++my_bad; continue; }
            idp[r] = (int64_t)id;
            kp[r] = kind;
        }
        bad += my_bad;
    };
    if (nthreads <= 1) {
        work(0, n);
    } else {
        std::vector<std::thread> ts;
        const int64_t per = (n + nthreads - 1) / nthreads;
        for (int t = 0; t < nthreads; ++t)
            ts.emplace_back(work, t * per,
                            std::min<int64_t>(n, (t + 1) * per));
        for (auto& t : ts) t.join();
    }
    PyBuffer_Release(&view);
    return py::make_tuple(ids, kinds, facs, poffs, plens, (int64_t)bad);
}


void check_hip(int err, const char* what) {
    TORCH_CHECK(err == 0, what, " failed: ", fma_err_str(err));
}

void check_t(const torch::Tensor& t, torch::ScalarType dt, const char* name) {
    TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
    TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
    TORCH_CHECK(t.scalar_type() == dt, name, " has wrong dtype");
}

const unsigned short* bf16_ptr(const torch::Tensor& t) {
    TORCH_CHECK(t.scalar_type() == torch::kBFloat16,
                "expected a bf16 tensor at the kernel boundary");
    return reinterpret_cast<const unsigned short*>(t.data_ptr());
}
unsigned short* bf16_ptr_mut(torch::Tensor& t) {
    TORCH_CHECK(t.scalar_type() == torch::kBFloat16,
                "expected a bf16 output tensor at the kernel boundary");
    return reinterpret_cast<unsigned short*>(t.data_ptr());
}

// e4m3 factor images travel as uint8 byte tensors
const unsigned char* fp8_ptr(const torch::Tensor& t) {
    TORCH_CHECK(t.scalar_type() == torch::kUInt8,
                "expected a uint8 (e4m3) tensor at the kernel boundary");
    return reinterpret_cast<const unsigned char*>(t.data_ptr());
}
unsigned char* fp8_ptr_mut(torch::Tensor& t) {
    TORCH_CHECK(t.scalar_type() == torch::kUInt8,
                "expected a uint8 (e4m3) output tensor at the kernel boundary");
    return reinterpret_cast<unsigned char*>(t.data_ptr());
}

const int* order_ptr(const torch::Tensor& row_order, int64_t nrows) {
    if (row_order.numel() == 0) return nullptr;
    TORCH_CHECK(row_order.scalar_type() == torch::kInt32 &&
                row_order.is_cuda() && row_order.is_contiguous(),
                "row_order must be contiguous int32 on GPU");
    TORCH_CHECK(row_order.numel() == nrows, "row_order size");
    return row_order.data_ptr<int>();
}

void als_solve_fused(torch::Tensor indptr, torch::Tensor indices,
                     torch::Tensor values, torch::Tensor factors,
                     torch::Tensor out_f32, torch::Tensor out_bf16,
                     torch::Tensor row_order, double reg, int64_t stream) {
    check_t(indptr, torch::kInt64, "indptr");
    check_t(indices, torch::kInt32, "indices");
    check_t(values, torch::kFloat32, "values");
    check_t(factors, torch::kBFloat16, "factors");
    check_t(out_f32, torch::kFloat32, "out_f32");
    const int k = (int)factors.size(1);
    const long long nrows = indptr.size(0) - 1;
    TORCH_CHECK(out_f32.size(0) >= nrows && out_f32.size(1) == k,
                "out_f32 shape mismatch");
    unsigned short* ob = nullptr;
    if (out_bf16.numel() > 0) {
        check_t(out_bf16, torch::kBFloat16, "out_bf16");
        TORCH_CHECK(out_bf16.numel() >= out_f32.numel(), "out_bf16 shape");
        ob = bf16_ptr_mut(out_bf16);
    }
    const int* order = nullptr;
    if (row_order.numel() > 0) {
        check_t(row_order, torch::kInt32, "row_order");
        TORCH_CHECK(row_order.numel() == nrows, "row_order size");
        order = row_order.data_ptr<int>();
    }
    check_hip(fma_als_solve_fused(
                  k, indptr.data_ptr<int64_t>(), indices.data_ptr<int>(),
                  values.data_ptr<float>(), bf16_ptr(factors),
                  out_f32.data_ptr<float>(), ob, order, nrows, (float)reg,
                  (void*)stream),
              "als_solve_fused");
}

void gramian(torch::Tensor indptr, torch::Tensor indices, torch::Tensor values,
             torch::Tensor factors, torch::Tensor A_out, torch::Tensor b_out,
             torch::Tensor row_order, double reg, int64_t stream) {
    check_t(indptr, torch::kInt64, "indptr");
    check_t(indices, torch::kInt32, "indices");
    check_t(values, torch::kFloat32, "values");
    check_t(factors, torch::kBFloat16, "factors");
    check_t(A_out, torch::kFloat32, "A_out");
    check_t(b_out, torch::kFloat32, "b_out");
    const int k = (int)factors.size(1);
    const long long nrows = indptr.size(0) - 1;
    check_hip(fma_gramian(k, indptr.data_ptr<int64_t>(),
                          indices.data_ptr<int>(), values.data_ptr<float>(),
                          bf16_ptr(factors), A_out.data_ptr<float>(),
                          b_out.data_ptr<float>(),
                          order_ptr(row_order, nrows), nrows, (float)reg,
                          (void*)stream),
              "gramian");
}

void gramian_fp8(torch::Tensor indptr, torch::Tensor indices,
                 torch::Tensor values, torch::Tensor factors,
                 torch::Tensor A_out, torch::Tensor b_out,
                 torch::Tensor row_order, double reg, int64_t stream) {
    check_t(indptr, torch::kInt64, "indptr");
    check_t(indices, torch::kInt32, "indices");
    check_t(values, torch::kFloat32, "values");
    check_t(factors, torch::kUInt8, "factors");
    check_t(A_out, torch::kFloat32, "A_out");
    check_t(b_out, torch::kFloat32, "b_out");
    const int k = (int)factors.size(1);
    const long long nrows = indptr.size(0) - 1;
    check_hip(fma_gramian_fp8(
                  k, indptr.data_ptr<int64_t>(), indices.data_ptr<int>(),
                  values.data_ptr<float>(), fp8_ptr(factors),
                  A_out.data_ptr<float>(), b_out.data_ptr<float>(),
                  order_ptr(row_order, nrows), nrows, (float)reg,
                  (void*)stream),
              "gramian_fp8");
}

void als_solve_fused_fp8(torch::Tensor indptr, torch::Tensor indices,
                         torch::Tensor values, torch::Tensor factors,
                         torch::Tensor out_f32, torch::Tensor out_fp8,
                         torch::Tensor row_order, double reg,
                         int64_t stream) {
    check_t(indptr, torch::kInt64, "indptr");
    check_t(indices, torch::kInt32, "indices");
    check_t(values, torch::kFloat32, "values");
    check_t(factors, torch::kUInt8, "factors");
    check_t(out_f32, torch::kFloat32, "out_f32");
    const int k = (int)factors.size(1);
    const long long nrows = indptr.size(0) - 1;
    TORCH_CHECK(out_f32.size(0) >= nrows && out_f32.size(1) == k,
                "out_f32 shape mismatch");
    unsigned char* o8 = nullptr;
    if (out_fp8.numel() > 0) {
        check_t(out_fp8, torch::kUInt8, "out_fp8");
        TORCH_CHECK(out_fp8.numel() >= out_f32.numel(), "out_fp8 shape");
        o8 = fp8_ptr_mut(out_fp8);
    }
    check_hip(fma_als_solve_fused_fp8(
                  k, indptr.data_ptr<int64_t>(), indices.data_ptr<int>(),
                  values.data_ptr<float>(), fp8_ptr(factors),
                  out_f32.data_ptr<float>(), o8, order_ptr(row_order, nrows),
                  nrows, (float)reg, (void*)stream),
              "als_solve_fused_fp8");
}

void cholesky_solve_ph(torch::Tensor A, torch::Tensor b, torch::Tensor x,
                       int64_t phases, int64_t stream) {
    check_hip(fma_cholesky_solve_ph((int)A.size(1), A.data_ptr<float>(),
                                    b.data_ptr<float>(), x.data_ptr<float>(),
                                    A.size(0), (int)phases, (void*)stream),
              "cholesky_solve_ph");
}

// single-kernel k<=64 path: Gramian + register LDL in one wave per entity
// (no A materialization in HBM).  factors dtype selects bf16 vs e4m3.
void als_solve_wavefused(torch::Tensor indptr, torch::Tensor indices,
                         torch::Tensor values, torch::Tensor factors,
                         torch::Tensor out_f32, torch::Tensor out_bf16,
                         torch::Tensor out_fp8, torch::Tensor row_order,
                         double reg, int64_t stream) {
    check_t(indptr, torch::kInt64, "indptr");
    check_t(indices, torch::kInt32, "indices");
    check_t(values, torch::kFloat32, "values");
    check_t(out_f32, torch::kFloat32, "out_f32");
    const bool fp8 = factors.scalar_type() == torch::kUInt8;
    if (!fp8) check_t(factors, torch::kBFloat16, "factors");
    TORCH_CHECK(factors.is_cuda() && factors.is_contiguous(),
                "factors must be contiguous on GPU");
    const int k = (int)factors.size(1);
    const long long nrows = indptr.size(0) - 1;
    TORCH_CHECK(out_f32.size(0) >= nrows && out_f32.size(1) == k,
                "out_f32 shape mismatch");
    unsigned short* ob = nullptr;
    if (out_bf16.numel() > 0) {
        TORCH_CHECK(out_bf16.numel() >= out_f32.numel(), "out_bf16 shape");
        ob = bf16_ptr_mut(out_bf16);
    }
    unsigned char* o8 = nullptr;
    if (out_fp8.numel() > 0) {
        TORCH_CHECK(out_fp8.numel() >= out_f32.numel(), "out_fp8 shape");
        o8 = fp8_ptr_mut(out_fp8);
    }
    check_hip(fma_als_solve_wavefused(
                  k, fp8 ? 1 : 0, indptr.data_ptr<int64_t>(),
                  indices.data_ptr<int>(), values.data_ptr<float>(),
                  factors.data_ptr(), out_f32.data_ptr<float>(), ob, o8,
                  order_ptr(row_order, nrows), nrows, (float)reg,
                  (void*)stream),
              "als_solve_wavefused");
}

// double-buffered variant (fp8 only): software-pipelined staging
void als_solve_wavefused_db(torch::Tensor indptr, torch::Tensor indices,
                            torch::Tensor values, torch::Tensor factors,
                            torch::Tensor out_f32, torch::Tensor out_bf16,
                            torch::Tensor out_fp8, torch::Tensor row_order,
                            double reg, int64_t stream) {
    check_t(indptr, torch::kInt64, "indptr");
    check_t(indices, torch::kInt32, "indices");
    check_t(values, torch::kFloat32, "values");
    check_t(factors, torch::kUInt8, "factors");
    check_t(out_f32, torch::kFloat32, "out_f32");
    const int k = (int)factors.size(1);
    const long long nrows = indptr.size(0) - 1;
    TORCH_CHECK(out_f32.size(0) >= nrows && out_f32.size(1) == k,
                "out_f32 shape mismatch");
    unsigned short* ob = nullptr;
    if (out_bf16.numel() > 0) ob = bf16_ptr_mut(out_bf16);
    unsigned char* o8 = nullptr;
    if (out_fp8.numel() > 0) o8 = fp8_ptr_mut(out_fp8);
    check_hip(fma_als_solve_wavefused_db(
                  k, indptr.data_ptr<int64_t>(), indices.data_ptr<int>(),
                  values.data_ptr<float>(), factors.data_ptr(),
                  out_f32.data_ptr<float>(), ob, o8,
                  order_ptr(row_order, nrows), nrows, (float)reg,
                  (void*)stream),
              "als_solve_wavefused_db");
}

// wave-pair fused path for 64 < k <= 128 (fp8 gathers only)
void als_solve_wavefused2(torch::Tensor indptr, torch::Tensor indices,
                          torch::Tensor values, torch::Tensor factors,
                          torch::Tensor out_f32, torch::Tensor out_fp8,
                          torch::Tensor row_order, double reg,
                          int64_t stream) {
    check_t(indptr, torch::kInt64, "indptr");
    check_t(indices, torch::kInt32, "indices");
    check_t(values, torch::kFloat32, "values");
    check_t(factors, torch::kUInt8, "factors");
    check_t(out_f32, torch::kFloat32, "out_f32");
    const int k = (int)factors.size(1);
    const long long nrows = indptr.size(0) - 1;
    TORCH_CHECK(out_f32.size(0) == nrows && out_f32.size(1) == k,
                "out_f32 shape mismatch");
    unsigned char* o8 = nullptr;
    if (out_fp8.numel() > 0) {
        TORCH_CHECK(out_fp8.numel() == out_f32.numel(), "out_fp8 shape");
        o8 = fp8_ptr_mut(out_fp8);
    }
    check_hip(fma_als_solve_wavefused2(
                  k, indptr.data_ptr<int64_t>(), indices.data_ptr<int>(),
                  values.data_ptr<float>(), fp8_ptr(factors),
                  out_f32.data_ptr<float>(), o8, order_ptr(row_order, nrows),
                  nrows, (float)reg, (void*)stream),
              "als_solve_wavefused2");
}

void ldl_solve_wave(torch::Tensor A, torch::Tensor b, torch::Tensor x,
                    torch::Tensor x_bf16, torch::Tensor x_fp8,
                    int64_t stream) {
    unsigned short* xb = x_bf16.numel() > 0 ? bf16_ptr_mut(x_bf16) : nullptr;
    unsigned char* x8 = x_fp8.numel() > 0 ? fp8_ptr_mut(x_fp8) : nullptr;
    check_hip(fma_ldl_solve_wave((int)A.size(1), A.data_ptr<float>(),
                                 b.data_ptr<float>(), x.data_ptr<float>(),
                                 xb, x8, A.size(0), (void*)stream),
              "ldl_solve_wave");
}

// register-resident v2 (A in MFMA fragments, panel scratch in LDS)
void ldl_solve_wave_reg(torch::Tensor A, torch::Tensor b, torch::Tensor x,
                        torch::Tensor x_bf16, torch::Tensor x_fp8,
                        int64_t stream) {
    unsigned short* xb = x_bf16.numel() > 0 ? bf16_ptr_mut(x_bf16) : nullptr;
    unsigned char* x8 = x_fp8.numel() > 0 ? fp8_ptr_mut(x_fp8) : nullptr;
    check_hip(fma_ldl_solve_wave_reg((int)A.size(1), A.data_ptr<float>(),
                                     b.data_ptr<float>(), x.data_ptr<float>(),
                                     xb, x8, A.size(0), (void*)stream),
              "ldl_solve_wave_reg");
}

void cholesky_solve(torch::Tensor A, torch::Tensor b, torch::Tensor x,
                    int64_t stream) {
    check_t(A, torch::kFloat32, "A");
    check_t(b, torch::kFloat32, "b");
    check_t(x, torch::kFloat32, "x");
    const long long nrows = A.size(0);
    const int k = (int)A.size(1);
    check_hip(fma_cholesky_solve(k, A.data_ptr<float>(), b.data_ptr<float>(),
                                 x.data_ptr<float>(), nrows, (void*)stream),
              "cholesky_solve");
}

void sdca_pass(torch::Tensor indptr, torch::Tensor indices,
               torch::Tensor values, torch::Tensor y, torch::Tensor norms_sq,
               torch::Tensor perm, torch::Tensor alpha, torch::Tensor v,
               double scale, int64_t stream) {
    check_t(indptr, torch::kInt64, "indptr");
    check_t(indices, torch::kInt32, "indices");
    check_t(values, torch::kFloat32, "values");
    check_t(y, torch::kFloat32, "y");
    check_t(norms_sq, torch::kFloat32, "norms_sq");
    check_t(alpha, torch::kFloat32, "alpha");
    check_t(v, torch::kFloat32, "v");
    const int* p = nullptr;
    if (perm.numel() > 0) {
        check_t(perm, torch::kInt32, "perm");
        p = perm.data_ptr<int>();
    }
    check_hip(fma_sdca_pass(indptr.data_ptr<int64_t>(),
                            indices.data_ptr<int>(), values.data_ptr<float>(),
                            y.data_ptr<float>(), norms_sq.data_ptr<float>(),
                            p, alpha.data_ptr<float>(), v.data_ptr<float>(),
                            indptr.size(0) - 1, (float)scale, (void*)stream),
              "sdca_pass");
}

void svm_margins(torch::Tensor indptr, torch::Tensor indices,
                 torch::Tensor values, torch::Tensor w, torch::Tensor out,
                 int64_t stream) {
    check_t(indptr, torch::kInt64, "indptr");
    check_t(indices, torch::kInt32, "indices");
    check_t(values, torch::kFloat32, "values");
    check_t(w, torch::kFloat32, "w");
    check_t(out, torch::kFloat32, "out");
    check_hip(fma_svm_margins(indptr.data_ptr<int64_t>(),
                              indices.data_ptr<int>(),
                              values.data_ptr<float>(), w.data_ptr<float>(),
                              out.data_ptr<float>(), indptr.size(0) - 1,
                              (void*)stream),
              "svm_margins");
}

void predict_dot(torch::Tensor U, torch::Tensor V, torch::Tensor u_idx,
                 torch::Tensor i_idx, torch::Tensor out, int64_t stream) {
    check_t(U, torch::kBFloat16, "U");
    check_t(V, torch::kBFloat16, "V");
    check_t(u_idx, torch::kInt64, "u_idx");
    check_t(i_idx, torch::kInt64, "i_idx");
    check_t(out, torch::kFloat32, "out");
    TORCH_CHECK(U.size(1) == V.size(1), "rank mismatch");
    check_hip(fma_predict_dot(bf16_ptr(U), bf16_ptr(V),
                              u_idx.data_ptr<int64_t>(),
                              i_idx.data_ptr<int64_t>(),
                              out.data_ptr<float>(), u_idx.numel(),
                              (int)U.size(1), (void*)stream),
              "predict_dot");
}

void sgd_update(torch::Tensor U, torch::Tensor V, torch::Tensor u_idx,
                torch::Tensor i_idx, torch::Tensor r, torch::Tensor err_out,
                double lr, double user_reg, double item_reg, int64_t stream) {
    check_t(U, torch::kBFloat16, "U");
    check_t(V, torch::kBFloat16, "V");
    check_t(u_idx, torch::kInt64, "u_idx");
    check_t(i_idx, torch::kInt64, "i_idx");
    check_t(r, torch::kFloat32, "r");
    float* ep = nullptr;
    if (err_out.numel() > 0) {
        check_t(err_out, torch::kFloat32, "err_out");
        ep = err_out.data_ptr<float>();
    }
    check_hip(fma_sgd_update(bf16_ptr_mut(U), bf16_ptr_mut(V),
                             u_idx.data_ptr<int64_t>(),
                             i_idx.data_ptr<int64_t>(), r.data_ptr<float>(),
                             ep, u_idx.numel(), (int)U.size(1), (float)lr,
                             (float)user_reg, (float)item_reg, (void*)stream),
              "sgd_update");
}

void mfma_probe_f32(torch::Tensor A, torch::Tensor B, torch::Tensor D,
                    int64_t stream) {
    check_t(A, torch::kFloat32, "A");
    check_t(B, torch::kFloat32, "B");
    check_t(D, torch::kFloat32, "D");
    check_hip(fma_mfma_probe_f32(A.data_ptr<float>(), B.data_ptr<float>(),
                                 D.data_ptr<float>(), (void*)stream),
              "mfma_probe_f32");
}

void mfma_probe_bf16(torch::Tensor Xt, torch::Tensor Yt, torch::Tensor C,
                     int64_t stream) {
    check_t(Xt, torch::kBFloat16, "Xt");
    check_t(Yt, torch::kBFloat16, "Yt");
    check_t(C, torch::kFloat32, "C");
    check_hip(fma_mfma_probe_bf16(bf16_ptr(Xt), bf16_ptr(Yt),
                                  C.data_ptr<float>(), (void*)stream),
              "mfma_probe_bf16");
}

void mfma_probe_fp8(torch::Tensor Xt, torch::Tensor Yt, torch::Tensor C,
                    int64_t stream) {
    check_t(Xt, torch::kUInt8, "Xt");
    check_t(Yt, torch::kUInt8, "Yt");
    check_t(C, torch::kFloat32, "C");
    check_hip(fma_mfma_probe_fp8(fp8_ptr(Xt), fp8_ptr(Yt),
                                 C.data_ptr<float>(), (void*)stream),
              "mfma_probe_fp8");
}

void dbg_stage_dump(torch::Tensor indptr, torch::Tensor indices,
                    torch::Tensor values, torch::Tensor factors,
                    torch::Tensor out, int64_t stream) {
    check_hip(fma_dbg_stage_dump((int)factors.size(1),
                                 indptr.data_ptr<int64_t>(),
                                 indices.data_ptr<int>(),
                                 values.data_ptr<float>(), bf16_ptr(factors),
                                 bf16_ptr_mut(out), (void*)stream),
              "dbg_stage_dump");
}

void dbg_frag_dump(torch::Tensor indptr, torch::Tensor indices,
                   torch::Tensor values, torch::Tensor factors,
                   torch::Tensor out, int64_t stream) {
    check_hip(fma_dbg_frag_dump((int)factors.size(1),
                                indptr.data_ptr<int64_t>(),
                                indices.data_ptr<int>(),
                                values.data_ptr<float>(), bf16_ptr(factors),
                                bf16_ptr_mut(out), (void*)stream),
              "dbg_frag_dump");
}

void dbg_gramian(torch::Tensor indptr, torch::Tensor indices,
                 torch::Tensor values, torch::Tensor factors,
                 torch::Tensor A_out, torch::Tensor bhi, torch::Tensor blo,
                 double reg, int64_t stream) {
    check_hip(fma_dbg_gramian((int)factors.size(1),
                              indptr.data_ptr<int64_t>(),
                              indices.data_ptr<int>(),
                              values.data_ptr<float>(), bf16_ptr(factors),
                              A_out.data_ptr<float>(), bhi.data_ptr<float>(),
                              blo.data_ptr<float>(), indptr.size(0) - 1,
                              (float)reg, (void*)stream),
              "dbg_gramian");
}

}  // namespace

void register_kvserver(pybind11::module_& m);  // serving/csrc/kvserver.cpp

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.doc() = "flink_ms_amd MI355X (gfx950) HIP kernels";
    register_kvserver(m);
    m.def("als_solve_fused", &als_solve_fused);
    m.def("gramian", &gramian);
    m.def("gramian_fp8", &gramian_fp8);
    m.def("als_solve_fused_fp8", &als_solve_fused_fp8);
    m.def("als_solve_wavefused", &als_solve_wavefused);
    m.def("als_solve_wavefused2", &als_solve_wavefused2);
    m.def("als_solve_wavefused_db", &als_solve_wavefused_db);
    m.def("cholesky_solve", &cholesky_solve);
    m.def("cholesky_solve_ph", &cholesky_solve_ph);
    m.def("ldl_solve_wave", &ldl_solve_wave);
    m.def("ldl_solve_wave_reg", &ldl_solve_wave_reg);
    m.def("sdca_pass", &sdca_pass);
    m.def("svm_margins", &svm_margins);
    m.def("predict_dot", &predict_dot);
    m.def("sgd_update", &sgd_update);
    m.def("dbg_stage_dump", &dbg_stage_dump);
    m.def("dbg_frag_dump", &dbg_frag_dump);
    m.def("dbg_gramian", &dbg_gramian);
    m.def("mfma_probe_f32", &mfma_probe_f32);
    m.def("mfma_probe_bf16", &mfma_probe_bf16);
    m.def("mfma_probe_fp8", &mfma_probe_fp8);
    m.def("parse_als_block", &parse_als_block);
}
