// Standalone variant sweep for the pi_marginal kernel (the scaled
// column-sum pi = (1/rowsum) @ adjusted, (N,C) fp32 -> (C,)).
//
// Shipping kernel (ops/hip/pbest.hip pi_marginal_kernel) measures
// 173 us at N=50k, C=1000 vs a ~25 us streaming floor; suspects are
// (a) only 4 float4 loads in flight per lane and (b) 1.5M atomicAdds
// onto 1000 float addresses. This probe times the current schedule
// against deeper unrolls, different grid sizes and a two-stage
// partial-buffer reduction, so only an evidenced winner gets ported.
//
// Build + run ON the GPU box (scripts/pi_marginal_probe.py drives it).
#include <torch/extension.h>
#include <hip/hip_runtime.h>
#include <c10/hip/HIPStream.h>

#define BLOCK 256

namespace {

// ---- v0: the shipping schedule (4-row unroll, 2 acc sets, atomics) ----
template <int ROWS>
__global__ void pim_atomic_kernel(const float* __restrict__ adjusted,
                                  const float* __restrict__ row_sums,
                                  float* __restrict__ out,
                                  long long N, int C) {
    const int tid = threadIdx.x;
    const long long rows_per_block = (N + gridDim.x - 1) / gridDim.x;
    const long long n0 = (long long)blockIdx.x * rows_per_block;
    const long long n1 = min(n0 + rows_per_block, N);
    const int c4 = tid * 4;
    if (c4 >= C) return;
    float4 acc = {0.f, 0.f, 0.f, 0.f};
    float4 acc2 = {0.f, 0.f, 0.f, 0.f};
    long long n = n0;
    for (; n + ROWS - 1 < n1; n += ROWS) {
        float inv[ROWS];
        float4 v[ROWS];
#pragma unroll
        for (int r = 0; r < ROWS; ++r)
            inv[r] = 1.0f / fmaxf(row_sums[n + r], 1e-12f);
#pragma unroll
        for (int r = 0; r < ROWS; ++r)
            v[r] = *reinterpret_cast<const float4*>(
                adjusted + (n + r) * C + c4);
#pragma unroll
        for (int r = 0; r < ROWS; r += 2) {
            acc.x += v[r].x * inv[r];  acc2.x += v[r + 1].x * inv[r + 1];
            acc.y += v[r].y * inv[r];  acc2.y += v[r + 1].y * inv[r + 1];
            acc.z += v[r].z * inv[r];  acc2.z += v[r + 1].z * inv[r + 1];
            acc.w += v[r].w * inv[r];  acc2.w += v[r + 1].w * inv[r + 1];
        }
    }
    for (; n < n1; ++n) {
        const float inv = 1.0f / fmaxf(row_sums[n], 1e-12f);
        const float4 v = *reinterpret_cast<const float4*>(
            adjusted + n * C + c4);
        acc.x += v.x * inv; acc.y += v.y * inv;
        acc.z += v.z * inv; acc.w += v.w * inv;
    }
    atomicAdd(out + c4 + 0, acc.x + acc2.x);
    atomicAdd(out + c4 + 1, acc.y + acc2.y);
    atomicAdd(out + c4 + 2, acc.z + acc2.z);
    atomicAdd(out + c4 + 3, acc.w + acc2.w);
}

// ---- two-stage: coalesced float4 partial stores + tiny reduce ----
template <int ROWS>
__global__ void pim_partial_kernel(const float* __restrict__ adjusted,
                                   const float* __restrict__ row_sums,
                                   float* __restrict__ partial, // (G, Cpad)
                                   long long N, int C, int Cpad) {
    const int tid = threadIdx.x;
    const long long rows_per_block = (N + gridDim.x - 1) / gridDim.x;
    const long long n0 = (long long)blockIdx.x * rows_per_block;
    const long long n1 = min(n0 + rows_per_block, N);
    const int c4 = tid * 4;
    if (c4 >= C) return;
    float4 acc = {0.f, 0.f, 0.f, 0.f};
    float4 acc2 = {0.f, 0.f, 0.f, 0.f};
    long long n = n0;
    for (; n + ROWS - 1 < n1; n += ROWS) {
        float inv[ROWS];
        float4 v[ROWS];
#pragma unroll
        for (int r = 0; r < ROWS; ++r)
            inv[r] = 1.0f / fmaxf(row_sums[n + r], 1e-12f);
#pragma unroll
        for (int r = 0; r < ROWS; ++r)
            v[r] = *reinterpret_cast<const float4*>(
                adjusted + (n + r) * C + c4);
#pragma unroll
        for (int r = 0; r < ROWS; r += 2) {
            acc.x += v[r].x * inv[r];  acc2.x += v[r + 1].x * inv[r + 1];
            acc.y += v[r].y * inv[r];  acc2.y += v[r + 1].y * inv[r + 1];
            acc.z += v[r].z * inv[r];  acc2.z += v[r + 1].z * inv[r + 1];
            acc.w += v[r].w * inv[r];  acc2.w += v[r + 1].w * inv[r + 1];
        }
    }
    for (; n < n1; ++n) {
        const float inv = 1.0f / fmaxf(row_sums[n], 1e-12f);
        const float4 v = *reinterpret_cast<const float4*>(
            adjusted + n * C + c4);
        acc.x += v.x * inv; acc.y += v.y * inv;
        acc.z += v.z * inv; acc.w += v.w * inv;
    }
    acc.x += acc2.x; acc.y += acc2.y; acc.z += acc2.z; acc.w += acc2.w;
    *reinterpret_cast<float4*>(
        partial + (long long)blockIdx.x * Cpad + c4) = acc;
}

// parallel reduce: grid (ceil(C/1024), GSPLIT); each block owns 256
// float4 columns x a G/GSPLIT slice, 8 independent accumulators deep,
// then ONE atomicAdd per (column, split) - 4*C*GSPLIT atomics total.
__global__ void pim_reduce_kernel(const float* __restrict__ partial,
                                  float* __restrict__ out,
                                  int G, int C, int Cpad) {
    const int c4 = (blockIdx.x * BLOCK + threadIdx.x) * 4;
    if (c4 >= C) return;
    const int gs = (G + gridDim.y - 1) / gridDim.y;
    const int g0 = blockIdx.y * gs, g1 = min(g0 + gs, G);
    float4 acc[2] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};
    int g = g0;
    for (; g + 1 < g1; g += 2) {
        const float4 a = *reinterpret_cast<const float4*>(
            partial + (long long)g * Cpad + c4);
        const float4 b = *reinterpret_cast<const float4*>(
            partial + (long long)(g + 1) * Cpad + c4);
        acc[0].x += a.x; acc[0].y += a.y; acc[0].z += a.z; acc[0].w += a.w;
        acc[1].x += b.x; acc[1].y += b.y; acc[1].z += b.z; acc[1].w += b.w;
    }
    if (g < g1) {
        const float4 a = *reinterpret_cast<const float4*>(
            partial + (long long)g * Cpad + c4);
        acc[0].x += a.x; acc[0].y += a.y; acc[0].z += a.z; acc[0].w += a.w;
    }
    if (gridDim.y == 1) {
        *reinterpret_cast<float4*>(out + c4) = {
            acc[0].x + acc[1].x, acc[0].y + acc[1].y,
            acc[0].z + acc[1].z, acc[0].w + acc[1].w};
    } else {
        atomicAdd(out + c4 + 0, acc[0].x + acc[1].x);
        atomicAdd(out + c4 + 1, acc[0].y + acc[1].y);
        atomicAdd(out + c4 + 2, acc[0].z + acc[1].z);
        atomicAdd(out + c4 + 3, acc[0].w + acc[1].w);
    }
}

}  // namespace

torch::Tensor pim_atomic(torch::Tensor adjusted, torch::Tensor row_sums,
                         int64_t blocks, int64_t rows_unroll) {
    const long long N = adjusted.size(0);
    const int C = adjusted.size(1);
    auto out = torch::zeros({C}, adjusted.options());
    auto stream = c10::hip::getCurrentHIPStream();
    if (rows_unroll == 8)
        hipLaunchKernelGGL(pim_atomic_kernel<8>, dim3((int)blocks),
                           dim3(BLOCK), 0, stream.stream(),
                           adjusted.data_ptr<float>(),
                           row_sums.data_ptr<float>(),
                           out.data_ptr<float>(), N, C);
    else
        hipLaunchKernelGGL(pim_atomic_kernel<4>, dim3((int)blocks),
                           dim3(BLOCK), 0, stream.stream(),
                           adjusted.data_ptr<float>(),
                           row_sums.data_ptr<float>(),
                           out.data_ptr<float>(), N, C);
    C10_HIP_CHECK(hipGetLastError());
    return out;
}

torch::Tensor pim_twostage(torch::Tensor adjusted, torch::Tensor row_sums,
                           torch::Tensor partial, int64_t rows_unroll,
                           int64_t gsplit) {
    const long long N = adjusted.size(0);
    const int C = adjusted.size(1);
    const int G = partial.size(0), Cpad = partial.size(1);
    auto out = gsplit > 1 ? torch::zeros({C}, adjusted.options())
                          : torch::empty({C}, adjusted.options());
    auto stream = c10::hip::getCurrentHIPStream();
    if (rows_unroll == 8)
        hipLaunchKernelGGL(pim_partial_kernel<8>, dim3(G), dim3(BLOCK), 0,
                           stream.stream(), adjusted.data_ptr<float>(),
                           row_sums.data_ptr<float>(),
                           partial.data_ptr<float>(), N, C, Cpad);
    else
        hipLaunchKernelGGL(pim_partial_kernel<4>, dim3(G), dim3(BLOCK), 0,
                           stream.stream(), adjusted.data_ptr<float>(),
                           row_sums.data_ptr<float>(),
                           partial.data_ptr<float>(), N, C, Cpad);
    hipLaunchKernelGGL(pim_reduce_kernel,
                       dim3((C + 4 * BLOCK - 1) / (4 * BLOCK), (int)gsplit),
                       dim3(BLOCK), 0, stream.stream(),
                       partial.data_ptr<float>(), out.data_ptr<float>(),
                       G, C, Cpad);
    C10_HIP_CHECK(hipGetLastError());
    return out;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("pim_atomic", &pim_atomic);
    m.def("pim_twostage", &pim_twostage);
}
