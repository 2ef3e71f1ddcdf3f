// v3 pair-factored EIG kernels for gfx950 (see coda_amd/ops/pair.py for
// the math and the sparsity argument; reference semantics:
// coda/coda.py:150-168 + :235-281 restricted to hit cells).
//
// Three kernels per acquisition step, covering the WHOLE candidate pool:
//
//   pair_dsum_es_kernel    one wave per (candidate, class) hit pair:
//                          sums the pair's selected log2-cdf delta
//                          curves (fp16 table - traffic-bound) and
//                          exponentiates -> the bf16 MFMA A operand
//                          (K, P). The baseline curve exp2(s_base)*w is
//                          folded into the B operand (egw).
//
//   pair_gemm_entropy{16,64}_kernel
//                          one workgroup per class-uniform pair tile:
//                          bf16 MFMA (tile x 2H x P) pairing GEMM with
//                          the variant-select + normalize + log2
//                          entropy epilogue fused (the (K, 2H) M tensor
//                          never reaches global memory). The 64-pair
//                          variant (2H <= 512) stages both operands in
//                          LDS, quartering the dominant B traffic vs
//                          the 16-pair tile; the 16-pair variant covers
//                          H up to 1024.
//
//   pair_eig_finalize_kernel
//                          one wave per candidate: EIG[b] = H_before -
//                          (pi_xi row) . h_base - sum over the
//                          candidate's hit pairs of pi_xi*(h_after -
//                          h_base), in a fixed lane-strided order
//                          (deterministic - no atomics).
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

namespace pairops {

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef _Float16 half4v __attribute__((ext_vector_type(4)));

#ifdef CODA_DPP_SCAN
// Wave64 sum via the GCN DPP row_shr/row_bcast ladder: 6 VALU ops at
// 1-2 cycle dependent latency each, vs ~7 dependent ds_bpermute rounds
// (~30+ cycles each) for the __shfl ladder. Same ladder as pbest.hip.
template <int CTRL, int ROW_MASK>
__device__ __forceinline__ float dpp_add(float x) {
    int moved = __builtin_amdgcn_update_dpp(0, __float_as_int(x), CTRL,
                                            ROW_MASK, 0xf, true);
    return x + __int_as_float(moved);
}

__device__ __forceinline__ float wave_reduce(float v) {
    v = dpp_add<0x111, 0xf>(v);  // row_shr:1
    v = dpp_add<0x112, 0xf>(v);  // row_shr:2
    v = dpp_add<0x114, 0xf>(v);  // row_shr:4
    v = dpp_add<0x118, 0xf>(v);  // row_shr:8
    v = dpp_add<0x142, 0xa>(v);  // row_bcast:15
    v = dpp_add<0x143, 0xc>(v);  // row_bcast:31
    return __int_as_float(__builtin_amdgcn_readlane(__float_as_int(v),
                                                    63));
}
#else
__device__ __forceinline__ float wave_reduce(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        v += __shfl_down(v, off, 64);
    return __shfl(v, 0, 64);
}
#endif

// ---------------------------------------------------------------------
// A-operand build: a16[k, p] = 2^(sum_{h in seg(k)} delta16[c_k, h, p])
// ---------------------------------------------------------------------
__global__ void __launch_bounds__(BLOCK)
pair_dsum_es_kernel(const _Float16* __restrict__ delta16,  // (C, H, P)
                    const float* __restrict__ dall,        // (C, P)
                    const int* __restrict__ pair_c,        // (K,)
                    const int* __restrict__ pair_neg,      // (K,)
                    const int* __restrict__ seg_off,       // (K+1,)
                    const int* __restrict__ seg_h,         // (S,)
                    hip_bfloat16* __restrict__ a16,        // (K, P)
                    int K, int H) {
    // 16 lanes per pair (the hit distribution is singleton-heavy:
    // avg segment ~2 models - a wave per pair would mostly idle on one
    // 512-B load). Each lane owns 16 grid points as 4x half4v loads.
    const int k = blockIdx.x * 16 + (threadIdx.x >> 4);
    if (k >= K) return;
    const int sublane = threadIdx.x & 15;
    const int p0 = sublane * 16;
    const int c = pair_c[k];
    const int s0 = seg_off[k], s1 = seg_off[k + 1];
    const size_t dbase = (size_t)c * H * P_POINTS + p0;

    float acc[16] = {};
    float acc2[16] = {};
    int s = s0;
    // 2-row software pipeline: the single-row loop is a dependent
    // seg_h -> 4-load -> FMA chain (the table lives in the LLC, so
    // this kernel is latency-bound, not HBM-bound); two rows with
    // split accumulators keep 8 loads in flight per lane (a 4-row
    // variant measured the same)
    for (; s + 1 < s1; s += 2) {
        const _Float16* ra = delta16 + dbase
            + (size_t)seg_h[s] * P_POINTS;
        const _Float16* rb = delta16 + dbase
            + (size_t)seg_h[s + 1] * P_POINTS;
        const half4v a0 = *reinterpret_cast<const half4v*>(ra);
        const half4v a1 = *reinterpret_cast<const half4v*>(ra + 4);
        const half4v a2 = *reinterpret_cast<const half4v*>(ra + 8);
        const half4v a3 = *reinterpret_cast<const half4v*>(ra + 12);
        const half4v b0 = *reinterpret_cast<const half4v*>(rb);
        const half4v b1 = *reinterpret_cast<const half4v*>(rb + 4);
        const half4v b2 = *reinterpret_cast<const half4v*>(rb + 8);
        const half4v b3 = *reinterpret_cast<const half4v*>(rb + 12);
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            acc[j] += (float)a0[j];      acc2[j] += (float)b0[j];
            acc[4 + j] += (float)a1[j];  acc2[4 + j] += (float)b1[j];
            acc[8 + j] += (float)a2[j];  acc2[8 + j] += (float)b2[j];
            acc[12 + j] += (float)a3[j]; acc2[12 + j] += (float)b3[j];
        }
    }
    if (s < s1) {
        const _Float16* row = delta16 + dbase + (size_t)seg_h[s] * P_POINTS;
        const half4v d0 = *reinterpret_cast<const half4v*>(row);
        const half4v d1 = *reinterpret_cast<const half4v*>(row + 4);
        const half4v d2 = *reinterpret_cast<const half4v*>(row + 8);
        const half4v d3 = *reinterpret_cast<const half4v*>(row + 12);
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            acc[j] += (float)d0[j];
            acc[4 + j] += (float)d1[j];
            acc[8 + j] += (float)d2[j];
            acc[12 + j] += (float)d3[j];
        }
    }
#pragma unroll
    for (int j = 0; j < 16; ++j) acc[j] += acc2[j];
    if (pair_neg[k]) {
        // complement segment: dsum = (sum over ALL models) - partial
        const float4* da = reinterpret_cast<const float4*>(
            dall + (size_t)c * P_POINTS + p0);
#pragma unroll
        for (int q = 0; q < 4; ++q) {
            const float4 d = da[q];
            acc[4 * q] = d.x - acc[4 * q];
            acc[4 * q + 1] = d.y - acc[4 * q + 1];
            acc[4 * q + 2] = d.z - acc[4 * q + 2];
            acc[4 * q + 3] = d.w - acc[4 * q + 3];
        }
    }
    ushort4 out[4];
    unsigned short* o = reinterpret_cast<unsigned short*>(out);
#pragma unroll
    for (int j = 0; j < 16; ++j)
        o[j] = hip_bfloat16(exp2f(acc[j])).data;
    ushort4* dst = reinterpret_cast<ushort4*>(
        a16 + (size_t)k * P_POINTS + p0);
    dst[0] = out[0]; dst[1] = out[1]; dst[2] = out[2]; dst[3] = out[3];
}

// ---------------------------------------------------------------------
// 16-pair tile GEMM+entropy (any H <= 1024): B streamed from L2.
// ---------------------------------------------------------------------
__global__ void __launch_bounds__(BLOCK)
pair_gemm_entropy16_kernel(const hip_bfloat16* __restrict__ a16,
                           const hip_bfloat16* __restrict__ egw,
                           const unsigned* __restrict__ vmask,
                           const int* __restrict__ pair_c,
                           int W,
                           const float* __restrict__ pi_hat,
                           const float* __restrict__ pbest_before,
                           const float* __restrict__ mixture0,
                           float* __restrict__ h_after,
                           int H, int mstride) {
    extern __shared__ char smem[];
    hip_bfloat16* a_lds = reinterpret_cast<hip_bfloat16*>(smem);
    float* m_tile = reinterpret_cast<float*>(
        smem + 16 * P_POINTS * sizeof(hip_bfloat16));

    const int k0 = blockIdx.x * 16;
    const int c = pair_c[k0];
    const int tid = threadIdx.x;
    const int wave = tid >> 6, lane = tid & 63;
    const int row16 = lane & 15;
    const int kgrp = lane >> 4;

    {   // stage A tile: 16 rows x 256 bf16 = 8 KB
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

    for (int pi = wave; pi < 16; pi += 4) {
        const int k = k0 + pi;
        const unsigned* vm = vmask + (size_t)k * W;
        const float* mrow = m_tile + (size_t)pi * mstride;
        float tot = 0.f;
        for (int h = lane; h < H; h += 64) {
            const int v = (vm[h >> 5] >> (h & 31)) & 1;
            tot += mrow[2 * h + v];
        }
        tot = wave_reduce(tot);
        const float inv = 1.0f / fmaxf(tot, 1e-30f);
        const float pic = pi_hat[c];
        float ent = 0.f;
        for (int h = lane; h < H; h += 64) {
            const int v = (vm[h >> 5] >> (h & 31)) & 1;
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

// ---------------------------------------------------------------------
// 128-pair B-resident tile GEMM+entropy (2H <= 272): the tile's whole
// egw[c] table (2H x P bf16) is staged into LDS ONCE with a coalesced
// row-major copy (stride 264 elems - a 16-B-aligned row start is
// required for the b128 fragment loads; an odd-word "bank-ideal"
// stride measurably loses more to misalignment than it wins back in
// conflicts), then 8 waves run the whole K loop against it
// barrier-free with the A operand streamed from L2. The epilogue is REGISTER-RESIDENT: each lane holds
// pairs kgrp*4+r at columns jt*16+row16, the static vmask selects the
// v variant per (pair, h) (the mask word index (8*jt)>>5 is
// compile-time), and 16-lane DPP row reductions produce tot/entropy -
// the (128, 2H) M tile never exists in LDS or HBM.
// ---------------------------------------------------------------------
#define BSTRIDE 272  // B row stride in halfs: 136 words, 8-word aligned (132
                     // put odd-row b128 reads at =4 mod 8 words: +60
                     // stall cycles each on CDNA4)
#define BLOCK2 512

__device__ __forceinline__ float row16_reduce(float v) {
#ifdef CODA_DPP_SCAN
    v = dpp_add<0x111, 0xf>(v);  // row_shr:1
    v = dpp_add<0x112, 0xf>(v);  // row_shr:2
    v = dpp_add<0x114, 0xf>(v);  // row_shr:4
    v = dpp_add<0x118, 0xf>(v);  // row_shr:8 -> lane15 of each row
#else
#pragma unroll
    for (int off = 1; off < 16; off <<= 1) {
        float n = __shfl_up(v, off, 16);
        if ((threadIdx.x & 15) >= off) v += n;
    }
#endif
    return __shfl(v, 15, 16);    // broadcast row total to all 16 lanes
}

template <int JT>
__global__ void __launch_bounds__(BLOCK2)
pair_gemm_entropy128_kernel(const hip_bfloat16* __restrict__ a16,
                            const hip_bfloat16* __restrict__ egw,
                            const unsigned* __restrict__ vmask,
                            const int* __restrict__ pair_c,
                            int W,
                            const float* __restrict__ pi_hat,
                            const float* __restrict__ pbest_before,
                            const float* __restrict__ mixture0,
                            float* __restrict__ h_after,
                            const int* __restrict__ grp_off,
                            int H, int mstride) {
    extern __shared__ char smem[];
    const int twoH = 2 * H;
    hip_bfloat16* b_lds = reinterpret_cast<hip_bfloat16*>(smem);

    // block owns tiles [t0, t1) - all the same class, so the 131 KB
    // B stage amortizes over up to GRP_MAX tiles (classes average
    // ~2.4 tiles at the headline hit distribution; B staging was
    // ~60% of this kernel's global traffic)
    const int t0 = grp_off[blockIdx.x], t1 = grp_off[blockIdx.x + 1];
    const int c = pair_c[t0 * 128];
    const int tid = threadIdx.x;
    const int wave = tid >> 6, lane = tid & 63;
    const int row16 = lane & 15;
    const int kgrp = lane >> 4;
    const hip_bfloat16* egw_c = egw + (size_t)c * twoH * P_POINTS;

    // stage the whole B table: row j (512 B) split in 2 halves; thread
    // t copies half (t&1) of row (t>>1) - fully coalesced 16-B loads
    for (int t = tid; t < 2 * twoH; t += BLOCK2) {
        const int j = t >> 1, half = t & 1;
        const uint4* g = reinterpret_cast<const uint4*>(
            egw_c + (size_t)j * P_POINTS + half * 128);
        uint4* d = reinterpret_cast<uint4*>(
            reinterpret_cast<char*>(b_lds)
            + (size_t)j * BSTRIDE * 2 + half * 256);
#pragma unroll
        for (int i = 0; i < 16; ++i) d[i] = g[i];
    }
    __syncthreads();

    for (int tile = t0; tile < t1; ++tile) {
    const int k0 = tile * 128;
    f32x4 acc[JT];
#pragma unroll
    for (int jt = 0; jt < JT; ++jt) acc[jt] = {0.f, 0.f, 0.f, 0.f};

    const hip_bfloat16* arow =
        a16 + (size_t)(k0 + wave * 16 + row16) * P_POINTS;
#pragma unroll
    for (int kk = 0; kk < P_POINTS; kk += 32) {
        const bf16x8 afrag = *reinterpret_cast<const bf16x8*>(
            arow + kk + kgrp * 8);
#pragma unroll
        for (int jt = 0; jt < JT; ++jt) {
            const int j = jt * 16 + row16;
            bf16x8 bfrag = {};
            if (j < twoH)
                bfrag = *reinterpret_cast<const bf16x8*>(
                    b_lds + (size_t)j * BSTRIDE + kk + kgrp * 8);
            acc[jt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag, bfrag, acc[jt], 0, 0, 0);
        }
    }

    // register-resident epilogue: lane owns pairs kbase..kbase+3 (the
    // accumulator rows) at columns jt*16+row16. For column j: h = j>>1
    // and it contributes to pair r iff vmask bit h == (j&1).
    const int kbase = k0 + wave * 16 + kgrp * 4;
    unsigned vmw[4][(JT + 3) / 4];
#pragma unroll
    for (int r = 0; r < 4; ++r)
#pragma unroll
        for (int w = 0; w < (JT + 3) / 4; ++w)
            vmw[r][w] = (w < W)
                ? vmask[(size_t)(kbase + r) * W + w] : 0u;

    float tot[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int jt = 0; jt < JT; ++jt) {
        const int j = jt * 16 + row16;
        if (j >= twoH) continue;
        const int h = j >> 1, v = j & 1;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int bit = (vmw[r][jt >> 2] >> (h & 31)) & 1;
            if (bit == v) tot[r] += acc[jt][r];
        }
    }
    float inv[4];
#pragma unroll
    for (int r = 0; r < 4; ++r)
        inv[r] = 1.0f / fmaxf(row16_reduce(tot[r]), 1e-30f);

    const float pic = pi_hat[c];
    float ent[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int jt = 0; jt < JT; ++jt) {
        const int j = jt * 16 + row16;
        if (j >= twoH) continue;
        const int h = j >> 1, v = j & 1;
        const float mix = mixture0[h];
        const float pbb = pbest_before[(size_t)c * H + h];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int bit = (vmw[r][jt >> 2] >> (h & 31)) & 1;
            if (bit == v) {
                const float pb = acc[jt][r] * inv[r];
                const float mm = fmaxf(mix + pic * (pb - pbb), 1e-12f);
                ent[r] += -mm * __log2f(mm);
            }
        }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        const float e = row16_reduce(ent[r]);
        if (row16 == 0) h_after[kbase + r] = e;
    }
    }  // tile loop
    (void)mstride;
}

// ---------------------------------------------------------------------
// Wide-H pipeline (288 < 2H <= 4096, i.e. the multi-GPU H=256..1024
// pools): the fused B-resident tile cannot hold egw[c] or the M tile in
// LDS, so the pairing GEMM writes M (K, 2H) bf16 to global with a
// standard 128x128 double-buffered MFMA tile (B traffic divided by 128
// vs the 16-pair fused tile), and a wave-per-pair entropy kernel
// consumes it with the same vmask epilogue.
// ---------------------------------------------------------------------
#define WSTRIDE 40   // 32 k-elems + 8 pad (16-B-aligned rows)

__global__ void __launch_bounds__(BLOCK)
pair_gemm_wide_kernel(const hip_bfloat16* __restrict__ a16,  // (K, P)
                      const hip_bfloat16* __restrict__ egw,  // (C, 2H, P)
                      const int* __restrict__ pair_c,        // (K,)
                      hip_bfloat16* __restrict__ mout,       // (K, 2H)
                      int twoH) {
    extern __shared__ char smem[];
    // [2 buffers][128 rows][WSTRIDE] for A and B chunks
    hip_bfloat16* a_lds = reinterpret_cast<hip_bfloat16*>(smem);
    hip_bfloat16* b_lds = a_lds + 2 * 128 * WSTRIDE;

    const int k0 = blockIdx.x * 128;
    const int j0 = blockIdx.y * 128;
    const int c = pair_c[k0];
    const int tid = threadIdx.x;
    const int wave = tid >> 6, lane = tid & 63;
    const int row16 = lane & 15, kgrp = lane >> 4;
    const hip_bfloat16* egw_c = egw + (size_t)c * twoH * P_POINTS;

    // stage one 128x32 chunk of A or B: thread t loads 16 B (8 elems)
    auto stage = [&](hip_bfloat16* dst, const hip_bfloat16* src,
                     int kk, bool valid_rows) {
        const int row = tid >> 1, half = tid & 1;
        const uint4* g = reinterpret_cast<const uint4*>(
            src + (size_t)row * P_POINTS + kk + half * 16);
        uint4* d = reinterpret_cast<uint4*>(
            reinterpret_cast<char*>(dst)
            + (size_t)row * WSTRIDE * 2 + half * 32);
        if (valid_rows) { d[0] = g[0]; d[1] = g[1]; }
    };

    stage(a_lds, a16 + (size_t)k0 * P_POINTS, 0, true);
    stage(b_lds, egw_c + (size_t)j0 * P_POINTS, 0,
          j0 + (tid >> 1) < twoH);
    __syncthreads();

    f32x4 acc[2][8];
#pragma unroll
    for (int r = 0; r < 2; ++r)
#pragma unroll
        for (int jt = 0; jt < 8; ++jt) acc[r][jt] = {0.f, 0.f, 0.f, 0.f};

    for (int kk = 0; kk < P_POINTS; kk += 32) {
        const int cur = (kk >> 5) & 1;
        if (kk + 32 < P_POINTS) {
            stage(a_lds + (cur ^ 1) * 128 * WSTRIDE,
                  a16 + (size_t)k0 * P_POINTS, kk + 32, true);
            stage(b_lds + (cur ^ 1) * 128 * WSTRIDE,
                  egw_c + (size_t)j0 * P_POINTS, kk + 32,
                  j0 + (threadIdx.x >> 1) < twoH);
        }
        const hip_bfloat16* ab = a_lds + cur * 128 * WSTRIDE;
        const hip_bfloat16* bb = b_lds + cur * 128 * WSTRIDE;
#pragma unroll
        for (int r = 0; r < 2; ++r) {
            // wave w owns pair rows [wave*32 + r*16 + row16]
            const bf16x8 afrag = *reinterpret_cast<const bf16x8*>(
                ab + (size_t)(wave * 32 + r * 16 + row16) * WSTRIDE
                + kgrp * 8);
#pragma unroll
            for (int jt = 0; jt < 8; ++jt) {
                const bf16x8 bfrag = *reinterpret_cast<const bf16x8*>(
                    bb + (size_t)(jt * 16 + row16) * WSTRIDE + kgrp * 8);
                acc[r][jt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    afrag, bfrag, acc[r][jt], 0, 0, 0);
            }
        }
        __syncthreads();
    }

    // write M tile: D col = row16 (the B row j), D rows = kgrp*4+e
#pragma unroll
    for (int r = 0; r < 2; ++r) {
#pragma unroll
        for (int jt = 0; jt < 8; ++jt) {
            const int j = j0 + jt * 16 + row16;
            if (j >= twoH) continue;
#pragma unroll
            for (int e = 0; e < 4; ++e) {
                const int k = k0 + wave * 32 + r * 16 + kgrp * 4 + e;
                mout[(size_t)k * twoH + j] =
                    hip_bfloat16(acc[r][jt][e]);
            }
        }
    }
}

__global__ void __launch_bounds__(BLOCK)
pair_entropy_wide_kernel(const hip_bfloat16* __restrict__ m,  // (K, 2H)
                         const unsigned* __restrict__ vmask,  // (K, W)
                         const int* __restrict__ pair_c,      // (K,)
                         int W,
                         const float* __restrict__ pi_hat,
                         const float* __restrict__ pbest_before,
                         const float* __restrict__ mixture0,
                         float* __restrict__ h_after,         // (K,)
                         int K, int H) {
    const int k = blockIdx.x * 4 + (threadIdx.x >> 6);
    if (k >= K) return;
    const int lane = threadIdx.x & 63;
    const int c = pair_c[k];
    const unsigned* vm = vmask + (size_t)k * W;
    const hip_bfloat16* mrow = m + (size_t)k * 2 * H;

    float val[32];   // H <= 2048 -> <= 32 h per lane
    float tot = 0.f;
    int i = 0;
    for (int h = lane; h < H; h += 64, ++i) {
        const int v = (vm[h >> 5] >> (h & 31)) & 1;
        val[i] = (float)mrow[2 * h + v];
        tot += val[i];
    }
    tot = wave_reduce(tot);
    const float inv = 1.0f / fmaxf(tot, 1e-30f);
    const float pic = pi_hat[c];
    float ent = 0.f;
    i = 0;
    for (int h = lane; h < H; h += 64, ++i) {
        const float pb = val[i] * inv;
        const float mm = fmaxf(
            mixture0[h] + pic * (pb - pbest_before[(size_t)c * H + h]),
            1e-12f);
        ent += -mm * __log2f(mm);
    }
    ent = wave_reduce(ent);
    if (lane == 0) h_after[k] = ent;
}

// ---------------------------------------------------------------------
// Finalize: q[b] = H_before - (1/rowsum) * (arow . h_base
//                 + sum_{pairs of b} arow[c]*(h_after - h_base[c]))
// One wave per candidate; fixed reduction order (deterministic).
// ---------------------------------------------------------------------
__global__ void __launch_bounds__(BLOCK)
pair_eig_finalize_kernel(const float* __restrict__ h_after,   // (K,)
                         const float* __restrict__ h_base,    // (C,)
                         const int* __restrict__ pair_c,      // (K,)
                         const int* __restrict__ cand_off,    // (B+1,)
                         const long long* __restrict__ cand_ck,  // (hits,)
                         const long* __restrict__ cand_ids,   // (B,)
                         const float* __restrict__ adjusted,  // (N, C)
                         const float* __restrict__ row_sums,  // (N,)
                         float H_before,
                         float* __restrict__ q,               // (B,)
                         int B, int C) {
    const int b = blockIdx.x * 4 + (threadIdx.x >> 6);
    if (b >= B) return;
    const int lane = threadIdx.x & 63;
    const long id = cand_ids[b];
    const float inv = 1.0f / fmaxf(row_sums[id], 1e-12f);
    const float* arow = adjusted + (size_t)id * C;
    float base = 0.f;
    int cc = lane * 4;
    for (; cc + 3 < C; cc += 256) {   // float4 over the class axis
        const float4 a = *reinterpret_cast<const float4*>(arow + cc);
        const float4 hb = *reinterpret_cast<const float4*>(h_base + cc);
        base += a.x * hb.x + a.y * hb.y + a.z * hb.z + a.w * hb.w;
    }
    for (; cc < C; ++cc) base += arow[cc] * h_base[cc];
    float corr = 0.f;
    const int s1 = cand_off[b + 1];
    // (pair, class) packed per hit: one load decodes both, so the
    // chase is packed -> {h_after, h_base, arow} (2 dependent levels,
    // the last three loads independent) instead of
    // cand_pairs -> pair_c -> gathers (3 levels)
    for (int s = cand_off[b] + lane; s < s1; s += 64) {
        const long long pk = cand_ck[s];
        const int k = (int)(pk >> 32);
        const int c2 = (int)(pk & 0xffffffff);
        corr += arow[c2] * (h_after[k] - h_base[c2]);
    }
    (void)pair_c;
    const float tot = wave_reduce(base + corr);
    if (lane == 0) q[b] = H_before - tot * inv;
}

// cls-based wide entropy (H beyond the vmask regime, e.g. 10k-model
// pools): v-select reads the candidate's argmax classes directly (the
// B x H cls block is small for prefiltered candidate sets and
// L2-resident); no per-lane value cache (H unbounded), so M is read
// twice.
__global__ void __launch_bounds__(BLOCK)
pair_entropy_wide_cls_kernel(const hip_bfloat16* __restrict__ m,  // (K, 2H)
                             const int* __restrict__ cls,         // (B, H)
                             const int* __restrict__ pair_b,      // (K,)
                             const int* __restrict__ pair_c,      // (K,)
                             const float* __restrict__ pi_hat,
                             const float* __restrict__ pbest_before,
                             const float* __restrict__ mixture0,
                             float* __restrict__ h_after,         // (K,)
                             int K, int H) {
    const int k = blockIdx.x * 4 + (threadIdx.x >> 6);
    if (k >= K) return;
    const int lane = threadIdx.x & 63;
    const int c = pair_c[k];
    const int b = pair_b[k];
    const hip_bfloat16* mrow = m + (size_t)k * 2 * H;

    float tot = 0.f;
    for (int h = lane; h < H; h += 64) {
        const int v = (b >= 0 && cls[(size_t)b * H + h] == c) ? 1 : 0;
        tot += (float)mrow[2 * h + v];
    }
    tot = wave_reduce(tot);
    const float inv = 1.0f / fmaxf(tot, 1e-30f);
    const float pic = pi_hat[c];
    float ent = 0.f;
    for (int h = lane; h < H; h += 64) {
        const int v = (b >= 0 && cls[(size_t)b * H + h] == c) ? 1 : 0;
        const float pb = (float)mrow[2 * h + v] * inv;
        const float mm = fmaxf(
            mixture0[h] + pic * (pb - pbest_before[(size_t)c * H + h]),
            1e-12f);
        ent += -mm * __log2f(mm);
    }
    ent = wave_reduce(ent);
    if (lane == 0) h_after[k] = ent;
}

// ---------------------------------------------------------------------
// Fused acquisition epilogue.  After pair_eig_finalize the graphed
// get_next tail was ~10 small torch launches (add + where(-inf) + max
// + isclose chain + sum + qbuf copy + stack/copy, ~120 us/step at
// B=50k): three kernels replace it.
//   K1: qbuf = active ? h0 + q0 : -inf, per-block (max, first-idx)
//   K2: one block combines the partials -> out[0..1]
//   K3: tie count |q - best| <= atol + rtol*|best| over active
//       candidates (torch.isclose semantics), f64 atomic into out[2]
//       (integer-valued doubles: exact, order-independent)
// ---------------------------------------------------------------------
#define ACQ_GRID 256

__global__ void __launch_bounds__(BLOCK)
acq_select_part_kernel(const float* __restrict__ q0,
                       const float* __restrict__ h0,     // scalar (1,)
                       const bool* __restrict__ active,
                       float* __restrict__ qbuf,
                       float* __restrict__ pmax, int* __restrict__ pidx,
                       int B) {
    const int per = (B + gridDim.x - 1) / gridDim.x;
    const int b0 = blockIdx.x * per;
    const int b1 = min(b0 + per, B);
    const float h = h0[0];
    float best = -INFINITY;
    int bidx = -1;
    for (int i = b0 + threadIdx.x; i < b1; i += BLOCK) {
        const float q = active[i] ? h + q0[i] : -INFINITY;
        qbuf[i] = q;
        if (q > best) { best = q; bidx = i; }   // strict: first index wins
    }
    __shared__ float sv[BLOCK];
    __shared__ int si[BLOCK];
    sv[threadIdx.x] = best;
    si[threadIdx.x] = bidx;
    __syncthreads();
    for (int s = BLOCK / 2; s > 0; s >>= 1) {
        if (threadIdx.x < s) {
            const float ov = sv[threadIdx.x + s];
            const int oi = si[threadIdx.x + s];
            if (ov > sv[threadIdx.x]
                || (ov == sv[threadIdx.x] && oi != -1
                    && (si[threadIdx.x] == -1 || oi < si[threadIdx.x]))) {
                sv[threadIdx.x] = ov;
                si[threadIdx.x] = oi;
            }
        }
        __syncthreads();
    }
    if (threadIdx.x == 0) {
        pmax[blockIdx.x] = sv[0];
        pidx[blockIdx.x] = si[0];
    }
}

__global__ void __launch_bounds__(ACQ_GRID)
acq_select_combine_kernel(const float* __restrict__ pmax,
                          const int* __restrict__ pidx,
                          double* __restrict__ out,     // (3,) f64
                          long long* __restrict__ tbuf) {
    __shared__ float sv[ACQ_GRID];
    __shared__ int si[ACQ_GRID];
    sv[threadIdx.x] = pmax[threadIdx.x];
    si[threadIdx.x] = pidx[threadIdx.x];
    __syncthreads();
    for (int s = ACQ_GRID / 2; s > 0; s >>= 1) {
        if (threadIdx.x < s) {
            const float ov = sv[threadIdx.x + s];
            const int oi = si[threadIdx.x + s];
            if (ov > sv[threadIdx.x]
                || (ov == sv[threadIdx.x] && oi != -1
                    && (si[threadIdx.x] == -1 || oi < si[threadIdx.x]))) {
                sv[threadIdx.x] = ov;
                si[threadIdx.x] = oi;
            }
        }
        __syncthreads();
    }
    if (threadIdx.x == 0) {
        out[0] = (double)sv[0];
        out[1] = (double)si[0];
        out[2] = 0.0;
        tbuf[0] = 0;                       // reset the tie-slot counter
    }
}

__global__ void __launch_bounds__(BLOCK)
acq_select_ties_kernel(const float* __restrict__ qbuf,
                       const bool* __restrict__ active,
                       double* __restrict__ out,
                       long long* __restrict__ tbuf,  // [0]=n, [1..cap]
                       int cap, int B) {
    const float best = (float)out[0];
    const float thr = 1e-8f + 1e-8f * fabsf(best);
    const int per = (B + gridDim.x - 1) / gridDim.x;
    const int b0 = blockIdx.x * per;
    const int b1 = min(b0 + per, B);
    int cnt = 0;
    for (int i = b0 + threadIdx.x; i < b1; i += BLOCK)
        if (active[i] && fabsf(qbuf[i] - best) <= thr) {
            ++cnt;
            // unordered slots, each packing (position, q value) so the
            // host tie-break needs NO second device fetch; the host
            // sorts the <= cap entries back to ascending position
            const int slot = (int)atomicAdd(
                reinterpret_cast<unsigned long long*>(tbuf), 1ull);
            if (slot < cap)
                tbuf[1 + slot] = ((long long)i << 32)
                    | (unsigned int)__float_as_int(qbuf[i]);
        }
    __shared__ int sc[BLOCK];
    sc[threadIdx.x] = cnt;
    __syncthreads();
    for (int s = BLOCK / 2; s > 0; s >>= 1) {
        if (threadIdx.x < s) sc[threadIdx.x] += sc[threadIdx.x + s];
        __syncthreads();
    }
    if (threadIdx.x == 0 && sc[0] > 0)
        atomicAdd(out + 2, (double)sc[0]);
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

torch::Tensor pair_dsum_es(torch::Tensor delta16, torch::Tensor dall,
                           torch::Tensor pair_c, torch::Tensor pair_neg,
                           torch::Tensor seg_off, torch::Tensor seg_h) {
    TORCH_CHECK(delta16.is_cuda() && delta16.dtype() == torch::kFloat16);
    TORCH_CHECK(delta16.size(-1) == P_POINTS);
    TORCH_CHECK(dall.dtype() == torch::kFloat32);
    const int H = delta16.size(1);
    const int K = pair_c.size(0);
    auto a16 = torch::empty({K, P_POINTS},
                            delta16.options().dtype(torch::kBFloat16));
    auto stream = c10::hip::getCurrentHIPStream();
    dim3 grid((K + 15) / 16);
    hipLaunchKernelGGL(pairops::pair_dsum_es_kernel, grid, dim3(BLOCK), 0,
                       stream.stream(),
                       reinterpret_cast<const _Float16*>(
                           delta16.data_ptr()),
                       dall.data_ptr<float>(),
                       pair_c.data_ptr<int>(), pair_neg.data_ptr<int>(),
                       seg_off.data_ptr<int>(), seg_h.data_ptr<int>(),
                       reinterpret_cast<hip_bfloat16*>(a16.data_ptr()),
                       K, H);
    return a16;
}

torch::Tensor pair_gemm_entropy(torch::Tensor a16, torch::Tensor egw,
                                torch::Tensor vmask, torch::Tensor pair_c,
                                torch::Tensor pi_hat,
                                torch::Tensor pbest_before,
                                torch::Tensor mixture0, int64_t tile,
                                int64_t ablate, torch::Tensor grp) {
    TORCH_CHECK(a16.is_cuda() && a16.dtype() == torch::kBFloat16);
    TORCH_CHECK(egw.dtype() == torch::kBFloat16);
    const int K = a16.size(0);
    const int H = mixture0.size(0);
    TORCH_CHECK(K % tile == 0, "pair count must be tile-padded");
    TORCH_CHECK(egw.size(1) == 2 * H);
    const int mstride = 2 * H + 4;  // LDS row pad against bank conflicts
    auto h_after = torch::empty({K}, pi_hat.options());
    auto stream = c10::hip::getCurrentHIPStream();
    const auto ab16 = reinterpret_cast<const hip_bfloat16*>(
        a16.data_ptr());
    const auto eb16 = reinterpret_cast<const hip_bfloat16*>(
        egw.data_ptr());
    if (tile == 128 && 2 * H > 272) {
        // wide-H split pipeline: M to global, entropy second pass
        TORCH_CHECK(2 * H <= 4096, "pair engine caps at H = 2048");
        const int twoH = 2 * H;
        auto mout = torch::empty({(long)K, (long)twoH},
                                 a16.options());
        const size_t shmem = 4 * 128 * WSTRIDE * sizeof(hip_bfloat16);
        dim3 grid(K / 128, (twoH + 127) / 128);
        hipLaunchKernelGGL(pairops::pair_gemm_wide_kernel, grid,
                           dim3(BLOCK), shmem, stream.stream(), ab16,
                           eb16, pair_c.data_ptr<int>(),
                           reinterpret_cast<hip_bfloat16*>(
                               mout.data_ptr()),
                           twoH);
        hipLaunchKernelGGL(pairops::pair_entropy_wide_kernel,
                           dim3((K + 3) / 4), dim3(BLOCK), 0,
                           stream.stream(),
                           reinterpret_cast<const hip_bfloat16*>(
                               mout.data_ptr()),
                           reinterpret_cast<const unsigned*>(
                               vmask.data_ptr<int>()),
                           pair_c.data_ptr<int>(), (int)vmask.size(1),
                           pi_hat.data_ptr<float>(),
                           pbest_before.data_ptr<float>(),
                           mixture0.data_ptr<float>(),
                           h_after.data_ptr<float>(), K, H);
    } else if (tile == 128) {
        TORCH_CHECK(2 * H <= 272, "128-pair tile needs 2H <= 272");
        const size_t shmem = (size_t)2 * H * BSTRIDE
                           * sizeof(hip_bfloat16);
        const int JT = (2 * H + 15) / 16;
        // tile groups (same-class runs, capped): default 1 tile/block
        torch::Tensor grp_t = grp;
        if (grp_t.numel() == 0)
            grp_t = torch::arange(K / 128 + 1,
                                  pair_c.options().dtype(torch::kInt32));
        TORCH_CHECK(grp_t.scalar_type() == torch::kInt32
                    && grp_t.is_contiguous());
        const int ngrp = grp_t.numel() - 1;
        auto launch = [&](auto kern) {
            hipLaunchKernelGGL(kern, dim3(ngrp), dim3(BLOCK2), shmem,
                               stream.stream(), ab16, eb16,
                               reinterpret_cast<const unsigned*>(
                                   vmask.data_ptr<int>()),
                               pair_c.data_ptr<int>(),
                               (int)vmask.size(1),
                               pi_hat.data_ptr<float>(),
                               pbest_before.data_ptr<float>(),
                               mixture0.data_ptr<float>(),
                               h_after.data_ptr<float>(),
                               grp_t.data_ptr<int>(), H, mstride);
        };
        (void)ablate;  // phase-ablation diagnostics retired
        if (JT <= 4) launch(pairops::pair_gemm_entropy128_kernel<4>);
        else if (JT <= 8)
            launch(pairops::pair_gemm_entropy128_kernel<8>);
        else if (JT <= 16)
            launch(pairops::pair_gemm_entropy128_kernel<16>);
        else launch(pairops::pair_gemm_entropy128_kernel<17>);
    } else {
        TORCH_CHECK(tile == 16, "tile must be 16 or 64");
        const size_t shmem = 16 * P_POINTS * sizeof(hip_bfloat16)
                           + (size_t)16 * mstride * sizeof(float);
        TORCH_CHECK(shmem <= 160 * 1024, "H too large for the fused "
                    "pair kernel (use the table engine beyond H=1024)");
        hipLaunchKernelGGL(pairops::pair_gemm_entropy16_kernel,
                           dim3(K / 16), dim3(BLOCK), shmem,
                           stream.stream(), ab16, eb16,
                           reinterpret_cast<const unsigned*>(
                               vmask.data_ptr<int>()),
                           pair_c.data_ptr<int>(),
                           (int)vmask.size(1),
                           pi_hat.data_ptr<float>(),
                           pbest_before.data_ptr<float>(),
                           mixture0.data_ptr<float>(),
                           h_after.data_ptr<float>(), H, mstride);
    }
    return h_after;
}

torch::Tensor pair_eig_finalize(torch::Tensor h_after,
                                torch::Tensor h_base,
                                torch::Tensor pair_c,
                                torch::Tensor cand_off,
                                torch::Tensor cand_ck,
                                torch::Tensor cand_ids,
                                torch::Tensor adjusted,
                                torch::Tensor row_sums,
                                double H_before) {
    const int B = cand_ids.size(0);
    const int C = adjusted.size(1);
    TORCH_CHECK(h_base.is_contiguous() && h_base.size(0) == C);
    TORCH_CHECK(cand_ck.scalar_type() == torch::kInt64,
                "cand_ck must be the packed int64 hit table");
    TORCH_CHECK(cand_ids.is_contiguous() && cand_off.is_contiguous()
                && cand_ck.is_contiguous(),
                "finalize inputs must be contiguous");
    auto q = torch::empty({B}, adjusted.options());
    auto stream = c10::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(pairops::pair_eig_finalize_kernel,
                       dim3((B + 3) / 4), dim3(BLOCK), 0, stream.stream(),
                       h_after.data_ptr<float>(),
                       h_base.data_ptr<float>(), pair_c.data_ptr<int>(),
                       cand_off.data_ptr<int>(),
                       reinterpret_cast<const long long*>(
                           cand_ck.data_ptr<int64_t>()),
                       cand_ids.data_ptr<long>(),
                       adjusted.data_ptr<float>(),
                       row_sums.data_ptr<float>(),
                       (float)H_before, q.data_ptr<float>(), B, C);
    return q;
}

void acq_select(torch::Tensor q0, torch::Tensor h0, torch::Tensor active,
                torch::Tensor qbuf, torch::Tensor out,
                torch::Tensor ties) {
    TORCH_CHECK(q0.is_cuda() && q0.dtype() == torch::kFloat32
                && q0.is_contiguous(), "q0 must be contiguous fp32");
    TORCH_CHECK(h0.dtype() == torch::kFloat32 && h0.numel() == 1);
    TORCH_CHECK(active.dtype() == torch::kBool && active.is_contiguous()
                && active.numel() == q0.numel());
    TORCH_CHECK(qbuf.dtype() == torch::kFloat32 && qbuf.is_contiguous()
                && qbuf.numel() == q0.numel());
    TORCH_CHECK(out.dtype() == torch::kFloat64 && out.is_contiguous()
                && out.numel() == 3);
    TORCH_CHECK(ties.dtype() == torch::kInt64 && ties.is_contiguous()
                && ties.numel() >= 2, "ties buffer too small");
    const int cap = ties.numel() - 1;
    const int B = q0.numel();
    auto pmax = torch::empty({ACQ_GRID}, q0.options());
    auto pidx = torch::empty({ACQ_GRID},
                             q0.options().dtype(torch::kInt32));
    auto stream = c10::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(pairops::acq_select_part_kernel, dim3(ACQ_GRID),
                       dim3(BLOCK), 0, stream.stream(),
                       q0.data_ptr<float>(), h0.data_ptr<float>(),
                       active.data_ptr<bool>(), qbuf.data_ptr<float>(),
                       pmax.data_ptr<float>(), pidx.data_ptr<int>(), B);
    hipLaunchKernelGGL(pairops::acq_select_combine_kernel, dim3(1),
                       dim3(ACQ_GRID), 0, stream.stream(),
                       pmax.data_ptr<float>(), pidx.data_ptr<int>(),
                       out.data_ptr<double>(),
                       reinterpret_cast<long long*>(
                           ties.data_ptr<int64_t>()));
    hipLaunchKernelGGL(pairops::acq_select_ties_kernel, dim3(ACQ_GRID),
                       dim3(BLOCK), 0, stream.stream(),
                       qbuf.data_ptr<float>(), active.data_ptr<bool>(),
                       out.data_ptr<double>(),
                       reinterpret_cast<long long*>(
                           ties.data_ptr<int64_t>()),
                       cap, B);
    C10_HIP_CHECK(hipGetLastError());
}

torch::Tensor pair_gemm_entropy_cls(torch::Tensor a16, torch::Tensor egw,
                                    torch::Tensor pair_b,
                                    torch::Tensor pair_c,
                                    torch::Tensor cls,
                                    torch::Tensor pi_hat,
                                    torch::Tensor pbest_before,
                                    torch::Tensor mixture0) {
    // wide-H pools (no v-select bitmask): 128x128 M-writing GEMM + the
    // cls-based entropy pass; any 2H (memory-bound by the (K, 2H) M
    // workspace)
    TORCH_CHECK(a16.is_cuda() && a16.dtype() == torch::kBFloat16);
    const int K = a16.size(0);
    const int H = mixture0.size(0);
    TORCH_CHECK(K % 128 == 0, "cls route needs 128-pair tiles");
    const int twoH = 2 * H;
    auto mout = torch::empty({(long)K, (long)twoH}, a16.options());
    auto h_after = torch::empty({K}, pi_hat.options());
    auto stream = c10::hip::getCurrentHIPStream();
    const size_t shmem = 4 * 128 * WSTRIDE * sizeof(hip_bfloat16);
    dim3 grid(K / 128, (twoH + 127) / 128);
    hipLaunchKernelGGL(pairops::pair_gemm_wide_kernel, grid, dim3(BLOCK),
                       shmem, stream.stream(),
                       reinterpret_cast<const hip_bfloat16*>(
                           a16.data_ptr()),
                       reinterpret_cast<const hip_bfloat16*>(
                           egw.data_ptr()),
                       pair_c.data_ptr<int>(),
                       reinterpret_cast<hip_bfloat16*>(mout.data_ptr()),
                       twoH);
    hipLaunchKernelGGL(pairops::pair_entropy_wide_cls_kernel,
                       dim3((K + 3) / 4), dim3(BLOCK), 0, stream.stream(),
                       reinterpret_cast<const hip_bfloat16*>(
                           mout.data_ptr()),
                       cls.data_ptr<int>(), pair_b.data_ptr<int>(),
                       pair_c.data_ptr<int>(), pi_hat.data_ptr<float>(),
                       pbest_before.data_ptr<float>(),
                       mixture0.data_ptr<float>(),
                       h_after.data_ptr<float>(), K, H);
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
          "v3 fused pairing MFMA GEMM + entropy epilogue -> (K,)",
          pybind11::arg("a16"), pybind11::arg("egw"),
          pybind11::arg("vmask"), pybind11::arg("pair_c"),
          pybind11::arg("pi_hat"),
          pybind11::arg("pbest_before"), pybind11::arg("mixture0"),
          pybind11::arg("tile"), pybind11::arg("ablate") = 0,
          pybind11::arg("grp") = torch::Tensor());
    m.def("pair_gemm_entropy_cls", &pair_gemm_entropy_cls,
          "v3 wide-H pairing GEMM + cls-based entropy -> (K,)");
    m.def("pair_eig_finalize", &pair_eig_finalize,
          "v3 per-candidate EIG assembly (deterministic) -> (B,)");
    m.def("acq_select", &acq_select,
          "fused acquisition epilogue: qbuf = active ? h0+q0 : -inf; "
          "out(f64[3]) = [masked max, first argmax, isclose tie count]");
    m.def("mfma_probe", &mfma_probe,
          "16x16x32 bf16 MFMA fragment-layout probe");
}
