// Fused CDNA4 (gfx950) kernels for the CODA hot path.
//
// The acquisition hot loop (reference: coda/coda.py:77-119 + :235-281) is,
// per candidate row, a Beta-grid P(best) integral coupled over the model
// axis H:
//   p_h = integral pdf_h(x) * prod_{h'!=h} cdf_{h'}(x) dx  on a P=256 grid,
// followed by a log2-entropy EIG assembly. The reference materializes six
// (R, H, P) fp32 tensors per chunk and runs a SEQUENTIAL Python loop over P
// for the trapezoid CDF; these kernels keep the whole pipeline per row
// inside one workgroup: grid point p <-> thread p, trapezoid CDF as an
// LDS/wave inclusive scan, the H-coupling as a running per-thread register
// (slog), and the EIG entropy fused into the epilogue. Global traffic is
// 2*R*H floats in, R*H (or B*C) floats out - ~3 orders of magnitude less
// than the eager formulation.
//
// Numerics: log-pdf evaluated in f64 (2 FMA per point; the f32
// cancellation at large Beta counts is the reference's main error source),
// exp/log in f32, the clamp ladder of the reference preserved exactly
// (cdf clamp 1e-30, log-space clamp +-80, entropy clamp 1e-12).
//
// Workgroup = 256 threads = 4 waves (wave64); P == blockDim == 256.

#include <torch/extension.h>
#include <hip/hip_runtime.h>
#include <c10/hip/HIPStream.h>

#define P_POINTS 256
#define BLOCK 256

namespace {

constexpr double kGridLo = 1e-6;
constexpr double kGridHi = 1.0 - 1e-6;
constexpr float kEps = 1e-30f;
constexpr float kLogClamp = 80.0f;

// Inclusive scan of v across the 256-thread block. Uses 4 floats of
// scratch + two barriers. Scratch may be reused after the call returns
// (a trailing barrier protects it).
__device__ __forceinline__ float block_inclusive_scan(float v, float* scr) {
    const int lane = threadIdx.x & 63;
    const int wave = threadIdx.x >> 6;
#pragma unroll
    for (int off = 1; off < 64; off <<= 1) {
        float n = __shfl_up(v, off, 64);
        if (lane >= off) v += n;
    }
    if (lane == 63) scr[wave] = v;
    __syncthreads();
    float prefix = 0.f;
#pragma unroll
    for (int w = 0; w < 4; ++w) {
        float s = scr[w];
        if (w < wave) prefix += s;
    }
    v += prefix;
    __syncthreads();
    return v;
}

// Sum of v across the block; every thread returns the total.
__device__ __forceinline__ float block_reduce_sum(float v, float* scr) {
    const int lane = threadIdx.x & 63;
    const int wave = threadIdx.x >> 6;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
    if (lane == 0) scr[wave] = v;
    __syncthreads();
    float total = scr[0] + scr[1] + scr[2] + scr[3];
    __syncthreads();
    return total;
}

// One Beta pdf value + its trapezoid-cumulative cdf at this thread's grid
// point, for row Beta(a, b). lnB precomputed in f64. The pdf of the
// PREVIOUS grid point comes through LDS (pdfbuf).
__device__ __forceinline__ void beta_pdf_cdf(
        double a, double b, double lnB, double lx, double l1mx, float dxf,
        float* pdfbuf, float* scr, float& pdf, float& cdf) {
    double t = (a - 1.0) * lx + (b - 1.0) * l1mx - lnB;
    pdf = __expf((float)t);
    pdfbuf[threadIdx.x] = pdf;
    __syncthreads();
    float prev = (threadIdx.x > 0) ? pdfbuf[threadIdx.x - 1] : 0.f;
    float tr = (threadIdx.x > 0) ? 0.5f * (pdf + prev) * dxf : 0.f;
    cdf = block_inclusive_scan(tr, scr);
}

// ---------------------------------------------------------------------------
// Kernel 1: generic P(best) over rows. alpha/beta: (R, H) -> out: (R, H).
// One workgroup per row; two passes over H (pass A accumulates
// slog_p = sum_h log cdf_h(p) in a register; pass B recomputes pdf/cdf and
// integrates).  LDS: a, b (f32 H) + lnB (f64 H) + pb (f32 H) + pdfbuf(256)
// + scratch.
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(BLOCK)
pbest_kernel(const float* __restrict__ alpha, const float* __restrict__ beta,
             float* __restrict__ out, int R, int H) {
    extern __shared__ __attribute__((aligned(16))) char smem_raw[];
    double* s_lnB = reinterpret_cast<double*>(smem_raw);         // H f64
    float* s_a = reinterpret_cast<float*>(s_lnB + H);            // H
    float* s_b = s_a + H;                                        // H
    float* s_pb = s_b + H;                                       // H
    float* s_pdf = s_pb + H;                                     // 256
    float* s_scr = s_pdf + P_POINTS;                             // 8

    const int r = blockIdx.x;
    if (r >= R) return;
    const int tid = threadIdx.x;

    for (int h = tid; h < H; h += BLOCK) {
        float a = alpha[(size_t)r * H + h];
        float b = beta[(size_t)r * H + h];
        s_a[h] = a;
        s_b[h] = b;
        s_lnB[h] = lgamma((double)a) + lgamma((double)b)
                 - lgamma((double)a + (double)b);
    }
    __syncthreads();

    const double step = (kGridHi - kGridLo) / (P_POINTS - 1);
    const double x = kGridLo + (double)tid * step;
    const double lx = log(x), l1mx = log1p(-x);
    const float dxf = (float)step;

    // pass A: slog_p = sum_h log cdf_h(p)
    float slog = 0.f;
    for (int h = 0; h < H; ++h) {
        float pdf, cdf;
        beta_pdf_cdf(s_a[h], s_b[h], s_lnB[h], lx, l1mx, dxf,
                     s_pdf, s_scr, pdf, cdf);
        slog += __logf(fmaxf(cdf, kEps));
    }

    // pass B: integrate pdf_h * exp(clamp(slog - log cdf_h))
    const float w_trapz = (tid == 0 || tid == P_POINTS - 1) ? 0.5f : 1.0f;
    for (int h = 0; h < H; ++h) {
        float pdf, cdf;
        beta_pdf_cdf(s_a[h], s_b[h], s_lnB[h], lx, l1mx, dxf,
                     s_pdf, s_scr, pdf, cdf);
        float lc = __logf(fmaxf(cdf, kEps));
        float pe = __expf(fminf(fmaxf(slog - lc, -kLogClamp), kLogClamp));
        float total = block_reduce_sum(pdf * pe * w_trapz * dxf, s_scr);
        if (tid == 0) s_pb[h] = total;
        __syncthreads();
    }

    // normalize over H and write out
    float part = 0.f;
    for (int h = tid; h < H; h += BLOCK) part += s_pb[h];
    float total = block_reduce_sum(part, s_scr);
    float inv = 1.0f / fmaxf(total, kEps);
    for (int h = tid; h < H; h += BLOCK)
        out[(size_t)r * H + h] = s_pb[h] * inv;
}

// ---------------------------------------------------------------------------
// Kernel 2: fused hypothetical-update P(best) + entropy epilogue for EIG.
// One workgroup per (candidate b, hypothesized class c) row:
//   a_h = alpha_t[c,h] + w*[cls[b,h]==c];  b_h = beta_t[c,h] + w*[!=]
//   pb = pbest(a, b)          (normalized over H)
//   m_h = mixture0[h] + pi_hat[c] * (pb_h - pbest_before[c,h])
//   H_after[b,c] = -sum_h clamp(m,1e-12) log2 m
// (reference: coda/coda.py:150-168 + :267-276). EIG itself is a trivial
// (B,C) contraction done by the host.
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(BLOCK)
eig_hyp_kernel(const float* __restrict__ alpha_t,   // (C, H)
               const float* __restrict__ beta_t,    // (C, H)
               const int* __restrict__ cls,         // (B, H)
               const float* __restrict__ pbest_before,  // (C, H)
               const float* __restrict__ pi_hat,    // (C,)
               const float* __restrict__ mixture0,  // (H,)
               float* __restrict__ h_after,         // (B, C)
               float update_weight, int B, int C, int H) {
    extern __shared__ __attribute__((aligned(16))) char smem_raw[];
    double* s_lnB = reinterpret_cast<double*>(smem_raw);         // H f64
    float* s_a = reinterpret_cast<float*>(s_lnB + H);            // H
    float* s_b = s_a + H;                                        // H
    float* s_pb = s_b + H;                                       // H
    float* s_pdf = s_pb + H;                                     // 256
    float* s_scr = s_pdf + P_POINTS;                             // 8

    const int rid = blockIdx.x;
    if (rid >= B * C) return;
    const int b = rid / C;
    const int c = rid - b * C;
    const int tid = threadIdx.x;

    for (int h = tid; h < H; h += BLOCK) {
        int cl = cls[(size_t)b * H + h];
        float add = (cl == c) ? update_weight : 0.f;
        float a = alpha_t[(size_t)c * H + h] + add;
        float bb = beta_t[(size_t)c * H + h] + (update_weight - add);
        s_a[h] = a;
        s_b[h] = bb;
        s_lnB[h] = lgamma((double)a) + lgamma((double)bb)
                 - lgamma((double)a + (double)bb);
    }
    __syncthreads();

    const double step = (kGridHi - kGridLo) / (P_POINTS - 1);
    const double x = kGridLo + (double)tid * step;
    const double lx = log(x), l1mx = log1p(-x);
    const float dxf = (float)step;

    float slog = 0.f;
    for (int h = 0; h < H; ++h) {
        float pdf, cdf;
        beta_pdf_cdf(s_a[h], s_b[h], s_lnB[h], lx, l1mx, dxf,
                     s_pdf, s_scr, pdf, cdf);
        slog += __logf(fmaxf(cdf, kEps));
    }

    const float w_trapz = (tid == 0 || tid == P_POINTS - 1) ? 0.5f : 1.0f;
    for (int h = 0; h < H; ++h) {
        float pdf, cdf;
        beta_pdf_cdf(s_a[h], s_b[h], s_lnB[h], lx, l1mx, dxf,
                     s_pdf, s_scr, pdf, cdf);
        float lc = __logf(fmaxf(cdf, kEps));
        float pe = __expf(fminf(fmaxf(slog - lc, -kLogClamp), kLogClamp));
        float total = block_reduce_sum(pdf * pe * w_trapz * dxf, s_scr);
        if (tid == 0) s_pb[h] = total;
        __syncthreads();
    }

    float part = 0.f;
    for (int h = tid; h < H; h += BLOCK) part += s_pb[h];
    float total = block_reduce_sum(part, s_scr);
    float inv = 1.0f / fmaxf(total, kEps);

    // entropy epilogue: -sum_h m log2 m, m = mixture0 + pi_c*(pb - before)
    const float pi_c = pi_hat[c];
    float ent = 0.f;
    for (int h = tid; h < H; h += BLOCK) {
        float pb = s_pb[h] * inv;
        float m = mixture0[h] + pi_c * (pb - pbest_before[(size_t)c * H + h]);
        m = fmaxf(m, 1e-12f);
        ent += -m * __log2f(m);
    }
    float ent_total = block_reduce_sum(ent, s_scr);
    if (tid == 0) h_after[rid] = ent_total;
}

size_t smem_bytes(int H) {
    return (size_t)H * sizeof(double) + (size_t)(3 * H) * sizeof(float)
         + (P_POINTS + 8) * sizeof(float);
}

}  // namespace

// ---------------------------------------------------------------------------
// Host bindings
// ---------------------------------------------------------------------------

static void check_f32_cuda(const torch::Tensor& t, const char* name) {
    TORCH_CHECK(t.is_cuda(), name, " must be on a ROCm device");
    TORCH_CHECK(t.scalar_type() == torch::kFloat32, name, " must be fp32");
    TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

torch::Tensor pbest_from_beta(torch::Tensor alpha, torch::Tensor beta,
                              int64_t num_points) {
    check_f32_cuda(alpha, "alpha");
    check_f32_cuda(beta, "beta");
    TORCH_CHECK(num_points == P_POINTS,
                "HIP pbest kernel is compiled for P=256");
    TORCH_CHECK(alpha.dim() == 2 && alpha.sizes() == beta.sizes(),
                "alpha/beta must be (R, H)");
    const int R = alpha.size(0), H = alpha.size(1);
    auto out = torch::empty_like(alpha);
    if (R == 0) return out;
    size_t smem = smem_bytes(H);
    TORCH_CHECK(smem <= 160 * 1024, "H too large for LDS: ", H);
    auto stream = c10::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(pbest_kernel, dim3(R), dim3(BLOCK), smem,
                       stream.stream(), alpha.data_ptr<float>(),
                       beta.data_ptr<float>(), out.data_ptr<float>(), R, H);
    C10_HIP_CHECK(hipGetLastError());
    return out;
}

torch::Tensor eig_chunk(torch::Tensor alpha_cc, torch::Tensor beta_cc,
                        torch::Tensor chunk_classes,
                        torch::Tensor pbest_before, torch::Tensor pi_hat,
                        torch::Tensor pi_hat_xi, torch::Tensor mixture0,
                        double h_before, double update_weight,
                        int64_t num_points) {
    check_f32_cuda(pbest_before, "pbest_before");
    check_f32_cuda(pi_hat, "pi_hat");
    check_f32_cuda(pi_hat_xi, "pi_hat_xi");
    check_f32_cuda(mixture0, "mixture0");
    TORCH_CHECK(chunk_classes.scalar_type() == torch::kInt32,
                "chunk_classes must be int32");
    TORCH_CHECK(num_points == P_POINTS,
                "HIP eig kernel is compiled for P=256");
    const int H = alpha_cc.size(0), C = alpha_cc.size(1);
    const int B = chunk_classes.size(0);
    // (H, C) -> (C, H) contiguous rows for per-class streaming
    auto alpha_t = alpha_cc.t().contiguous();
    auto beta_t = beta_cc.t().contiguous();
    check_f32_cuda(alpha_t, "alpha_cc");
    check_f32_cuda(beta_t, "beta_cc");

    auto h_after = torch::empty({B, C}, alpha_t.options());
    size_t smem = smem_bytes(H);
    TORCH_CHECK(smem <= 160 * 1024, "H too large for LDS: ", H);
    auto stream = c10::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(eig_hyp_kernel, dim3(B * C), dim3(BLOCK), smem,
                       stream.stream(), alpha_t.data_ptr<float>(),
                       beta_t.data_ptr<float>(),
                       chunk_classes.data_ptr<int>(),
                       pbest_before.data_ptr<float>(),
                       pi_hat.data_ptr<float>(), mixture0.data_ptr<float>(),
                       h_after.data_ptr<float>(), (float)update_weight,
                       B, C, H);
    C10_HIP_CHECK(hipGetLastError());
    // EIG = H_before - sum_c pi_hat_xi[b,c] * H_after[b,c]
    return h_before - (pi_hat_xi * h_after).sum(-1);
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.doc() = "coda_amd fused gfx950 kernels";
    m.def("pbest_from_beta", &pbest_from_beta,
          "Beta-grid P(best) per row: (R,H),(R,H) -> (R,H)");
    m.def("eig_chunk", &eig_chunk,
          "Fused hypothetical P(best) + entropy EIG for a candidate chunk");
}
