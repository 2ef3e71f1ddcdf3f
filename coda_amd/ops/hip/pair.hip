// v3 pair-factored EIG kernels for gfx950 (see coda_amd/ops/pair.py for
// the math and the sparsity argument; reference semantics:
// coda/coda.py:150-168 + :235-281 restricted to hit cells).
//
// Two kernels per acquisition step, covering the WHOLE candidate pool:
//
//   pair_dsum_es_kernel    one wave per (candidate, class) hit pair:
//                          sums the pair's selected log2-cdf delta
//                          curves and exponentiates -> the bf16 MFMA A
//                          operand (K, P). The baseline curve
//                          exp2(s_base)*w is folded into the B operand
//                          (egw, built host-side once per step delta).
//
//   pair_gemm_entropy_kernel
//                          one workgroup per 16-pair class-uniform
//                          tile: bf16 MFMA (16 x 2H x P) pairing GEMM
//                          with the variant-select + normalize + log2
//                          entropy epilogue fused (the (K, 2H) M tensor
//                          never reaches global memory). A operand
//                          staged in LDS; B operand (egw rows of the
//                          tile's class) streamed from L2 - tiles are
//                          class-sorted so each XCD reads a class's
//                          131 KB table once.
//
// MFMA: v_mfma_f32_16x16x32_bf16. Lane mapping (cdna_hip_programming.md
// section 3): A[i][k] i=lane&15, k=(lane>>4)*8+e; B[k][j] j=lane&15,
// same k; C/D col=lane&15, row=(lane>>4)*4+reg. The B operand is stored
// row-major (j, p) = egw (C, 2H, P) so both fragments load 16
// contiguous bytes per lane. Layout verified by the mfma_probe test
// (tests/test_gpu.py).

#include <torch/extension.h>
#include <hip/hip_runtime.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_bfloat16.h>

#define P_POINTS 256
#define BLOCK 256
#define PAIR_TILE 16

namespace pairops {

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

__device__ __forceinline__ float wave_reduce(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        v += __shfl_down(v, off, 64);
    return __shfl(v, 0, 64);
}

// ---------------------------------------------------------------------
// A-operand build: a16[k, p] = 2^(sum_{h in seg(k)} delta[c_k, h, p])
// ---------------------------------------------------------------------
__global__ void __launch_bounds__(BLOCK)
pair_dsum_es_kernel(const float* __restrict__ delta,   // (C, H, P)
                    const int* __restrict__ pair_c,    // (K,)
                    const int* __restrict__ seg_off,   // (K+1,)
                    const int* __restrict__ seg_h,     // (S,)
                    hip_bfloat16* __restrict__ a16,  // (K, P)
                    int K, int H) {
    const int k = blockIdx.x * 4 + (threadIdx.x >> 6);
    if (k >= K) return;
    const int lane = threadIdx.x & 63;
    const int p0 = lane * 4;
    const int c = pair_c[k];
    const int s0 = seg_off[k], s1 = seg_off[k + 1];
    const size_t dbase = (size_t)c * H * P_POINTS + p0;

    // 4 accumulator sets keep 4 delta-row loads in flight (fp32 adds
    // are not reassociated by the compiler)
    float a0[4] = {0.f, 0.f, 0.f, 0.f}, a1[4] = {0.f, 0.f, 0.f, 0.f},
          a2[4] = {0.f, 0.f, 0.f, 0.f}, a3[4] = {0.f, 0.f, 0.f, 0.f};
    int s = s0;
    for (; s + 3 < s1; s += 4) {
        const int ha = seg_h[s], hb = seg_h[s + 1];
        const int hc = seg_h[s + 2], hd = seg_h[s + 3];
        const float4 da = *reinterpret_cast<const float4*>(
            delta + dbase + (size_t)ha * P_POINTS);
        const float4 db = *reinterpret_cast<const float4*>(
            delta + dbase + (size_t)hb * P_POINTS);
        const float4 dc = *reinterpret_cast<const float4*>(
            delta + dbase + (size_t)hc * P_POINTS);
        const float4 dd = *reinterpret_cast<const float4*>(
            delta + dbase + (size_t)hd * P_POINTS);
        a0[0] += da.x; a0[1] += da.y; a0[2] += da.z; a0[3] += da.w;
        a1[0] += db.x; a1[1] += db.y; a1[2] += db.z; a1[3] += db.w;
        a2[0] += dc.x; a2[1] += dc.y; a2[2] += dc.z; a2[3] += dc.w;
        a3[0] += dd.x; a3[1] += dd.y; a3[2] += dd.z; a3[3] += dd.w;
    }
    for (; s < s1; ++s) {
        const int h = seg_h[s];
        const float4 d = *reinterpret_cast<const float4*>(
            delta + dbase + (size_t)h * P_POINTS);
        a0[0] += d.x; a0[1] += d.y; a0[2] += d.z; a0[3] += d.w;
    }
    ushort4 out;
    unsigned short* o = reinterpret_cast<unsigned short*>(&out);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
        const float v = exp2f(a0[j] + (a1[j] + a2[j]) + a3[j]);
        o[j] = hip_bfloat16(v).data;
    }
    *reinterpret_cast<ushort4*>(a16 + (size_t)k * P_POINTS + p0) = out;
}

// ---------------------------------------------------------------------
// Pairing GEMM + fused entropy epilogue.
// Tile: PAIR_TILE pairs (class-uniform by construction) x 2H columns,
// K-loop over P in steps of 32. M stays in LDS; h_after (K,) out.
// ---------------------------------------------------------------------
__global__ void __launch_bounds__(BLOCK)
pair_gemm_entropy_kernel(const hip_bfloat16* __restrict__ a16,  // (K, P)
                         const hip_bfloat16* __restrict__ egw,  // (C, 2H, P)
                         const int* __restrict__ pair_b,          // (K,)
                         const int* __restrict__ pair_c,          // (K,)
                         const int* __restrict__ cls,             // (B, H)
                         const float* __restrict__ pi_hat,        // (C,)
                         const float* __restrict__ pbest_before,  // (C, H)
                         const float* __restrict__ mixture0,      // (H,)
                         float* __restrict__ h_after,             // (K,)
                         int H, int mstride) {
    extern __shared__ char smem[];
    hip_bfloat16* a_lds = reinterpret_cast<hip_bfloat16*>(smem);
    float* m_tile = reinterpret_cast<float*>(
        smem + PAIR_TILE * P_POINTS * sizeof(hip_bfloat16));

    const int k0 = blockIdx.x * PAIR_TILE;
    const int c = pair_c[k0];
    const int tid = threadIdx.x;
    const int wave = tid >> 6, lane = tid & 63;
    const int row16 = lane & 15;       // A row / B col / D col
    const int kgrp = lane >> 4;        // K sub-group (0..3)

    {   // stage A tile: 16 rows x 256 bf16 = 8 KB, 16 B per thread x2
        const uint4* g = reinterpret_cast<const uint4*>(
            a16 + (size_t)k0 * P_POINTS);
        uint4* d = reinterpret_cast<uint4*>(a_lds);
        d[tid] = g[tid];
        d[tid + BLOCK] = g[tid + BLOCK];
    }
    __syncthreads();

    const int twoH = 2 * H;
    const int JT = (twoH + 15) / 16;
    for (int jt = wave; jt < JT; jt += 4) {
        const int j = jt * 16 + row16;
        const bool jvalid = j < twoH;
        const hip_bfloat16* brow =
            egw + ((size_t)c * twoH + (jvalid ? j : 0)) * P_POINTS;
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int kk = 0; kk < P_POINTS; kk += 32) {
            const bf16x8 afrag = *reinterpret_cast<const bf16x8*>(
                a_lds + row16 * P_POINTS + kk + kgrp * 8);
            bf16x8 bfrag = {};
            if (jvalid)
                bfrag = *reinterpret_cast<const bf16x8*>(
                    brow + kk + kgrp * 8);
            acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag, bfrag, acc, 0, 0, 0);
        }
        if (jvalid) {
#pragma unroll
            for (int r = 0; r < 4; ++r)
                m_tile[(kgrp * 4 + r) * mstride + j] = acc[r];
        }
    }
    __syncthreads();

    // epilogue: 4 pairs per wave; v-select + normalize + entropy
    for (int pi = wave; pi < PAIR_TILE; pi += 4) {
        const int k = k0 + pi;
        const int b = pair_b[k];
        const float* mrow = m_tile + (size_t)pi * mstride;
        float tot = 0.f;
        for (int h = lane; h < H; h += 64) {
            const int v = (b >= 0 && cls[(size_t)b * H + h] == c) ? 1 : 0;
            tot += mrow[2 * h + v];
        }
        tot = wave_reduce(tot);
        const float inv = 1.0f / fmaxf(tot, 1e-30f);
        const float pic = pi_hat[c];
        float ent = 0.f;
        for (int h = lane; h < H; h += 64) {
            const int v = (b >= 0 && cls[(size_t)b * H + h] == c) ? 1 : 0;
            const float pb = mrow[2 * h + v] * inv;
            const float mm = fmaxf(
                mixture0[h] + pic * (pb - pbest_before[(size_t)c * H + h]),
                1e-12f);
            ent += -mm * __log2f(mm);
        }
        ent = wave_reduce(ent);
        if (lane == 0) h_after[k] = ent;
    }
}

// MFMA layout probe (correctness insurance, not a production op):
// C (16,16) = A (16,32) x B stored row-major as BT (16 cols x 32 k).
__global__ void mfma_probe_kernel(const float* __restrict__ a,   // (16,32)
                                  const float* __restrict__ bt,  // (16,32)
                                  float* __restrict__ out) {     // (16,16)
    if (threadIdx.x >= 64) return;
    const int lane = threadIdx.x;
    const int row16 = lane & 15, kgrp = lane >> 4;
    bf16x8 afrag, bfrag;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
        afrag[e] = (__bf16)a[row16 * 32 + kgrp * 8 + e];
        bfrag[e] = (__bf16)bt[row16 * 32 + kgrp * 8 + e];
    }
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc,
                                                  0, 0, 0);
#pragma unroll
    for (int r = 0; r < 4; ++r)
        out[(kgrp * 4 + r) * 16 + row16] = acc[r];
}

}  // namespace pairops

// ---------------------------------------------------------------------
// Host bindings
// ---------------------------------------------------------------------

torch::Tensor pair_dsum_es(torch::Tensor delta, torch::Tensor pair_c,
                           torch::Tensor seg_off, torch::Tensor seg_h) {
    TORCH_CHECK(delta.is_cuda() && delta.dtype() == torch::kFloat32);
    TORCH_CHECK(delta.size(-1) == P_POINTS);
    const int C = delta.size(0), H = delta.size(1);
    const int K = pair_c.size(0);
    (void)C;
    auto a16 = torch::empty({K, P_POINTS},
                            delta.options().dtype(torch::kBFloat16));
    auto stream = c10::hip::getCurrentHIPStream();
    dim3 grid((K + 3) / 4);
    hipLaunchKernelGGL(pairops::pair_dsum_es_kernel, grid, dim3(BLOCK), 0,
                       stream.stream(),
                       delta.data_ptr<float>(), pair_c.data_ptr<int>(),
                       seg_off.data_ptr<int>(), seg_h.data_ptr<int>(),
                       reinterpret_cast<hip_bfloat16*>(a16.data_ptr()),
                       K, H);
    return a16;
}

torch::Tensor pair_gemm_entropy(torch::Tensor a16, torch::Tensor egw,
                                torch::Tensor pair_b, torch::Tensor pair_c,
                                torch::Tensor cls, torch::Tensor pi_hat,
                                torch::Tensor pbest_before,
                                torch::Tensor mixture0) {
    TORCH_CHECK(a16.is_cuda() && a16.dtype() == torch::kBFloat16);
    TORCH_CHECK(egw.dtype() == torch::kBFloat16);
    const int K = a16.size(0);
    const int H = mixture0.size(0);
    TORCH_CHECK(K % PAIR_TILE == 0, "pair count must be tile-padded");
    TORCH_CHECK(egw.size(1) == 2 * H);
    const int mstride = 2 * H + 4;  // LDS row pad against bank conflicts
    const size_t shmem = PAIR_TILE * P_POINTS * sizeof(hip_bfloat16)
                       + (size_t)PAIR_TILE * mstride * sizeof(float);
    TORCH_CHECK(shmem <= 160 * 1024, "H too large for the fused pair "
                "kernel (use the table engine beyond H=1024)");
    auto h_after = torch::empty({K}, pi_hat.options());
    auto stream = c10::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(pairops::pair_gemm_entropy_kernel,
                       dim3(K / PAIR_TILE), dim3(BLOCK), shmem,
                       stream.stream(),
                       reinterpret_cast<const hip_bfloat16*>(
                           a16.data_ptr()),
                       reinterpret_cast<const hip_bfloat16*>(
                           egw.data_ptr()),
                       pair_b.data_ptr<int>(), pair_c.data_ptr<int>(),
                       cls.data_ptr<int>(), pi_hat.data_ptr<float>(),
                       pbest_before.data_ptr<float>(),
                       mixture0.data_ptr<float>(),
                       h_after.data_ptr<float>(), H, mstride);
    return h_after;
}

torch::Tensor mfma_probe(torch::Tensor a, torch::Tensor bt) {
    TORCH_CHECK(a.sizes() == torch::IntArrayRef({16, 32}));
    TORCH_CHECK(bt.sizes() == torch::IntArrayRef({16, 32}));
    auto out = torch::zeros({16, 16}, a.options());
    auto stream = c10::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(pairops::mfma_probe_kernel, dim3(1), dim3(64), 0,
                       stream.stream(), a.data_ptr<float>(),
                       bt.data_ptr<float>(), out.data_ptr<float>());
    return out;
}

void register_pair_ops(pybind11::module_& m) {
    m.def("pair_dsum_es", &pair_dsum_es,
          "v3 pair A-operand: 2^(summed delta curves) -> (K, P) bf16");
    m.def("pair_gemm_entropy", &pair_gemm_entropy,
          "v3 fused pairing MFMA GEMM + entropy epilogue -> (K,)");
    m.def("mfma_probe", &mfma_probe,
          "16x16x32 bf16 MFMA fragment-layout probe");
}
