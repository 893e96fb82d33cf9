// bf16 MFMA GEMM with fused epilogue (bias + activation + residual add).
//
// Computes C[M,N] = act(A[M,K] @ B[K,N] + bias) [+ residual], all bf16
// row-major with f32 accumulation. B is stored [K,N] (the framework's
// GEMM-ready weight layout; see pipeedge_amd/ops/torch_ref.py docstring).
//
// Structure (cdna_hip_programming.md section 5 canonical anatomy): 64x64x64
// block tile, 4 waves in 2x2, each wave one 32x32 sub-tile of 2x2 MFMA
// fragments; A and B^T tiles staged in padded LDS; fragment reads are
// 16-B ds_read_b128. Replaces what the reference ran through HF Linear
// modules on CPU (reference vit.py:45-53).
#include <torch/extension.h>

#include "common.h"

namespace {

constexpr int BM = 64;
constexpr int BN = 64;
constexpr int BK = 64;
constexpr int PAD = 8;  // LDS row pad: breaks the 128-B-row bank conflict

__global__ __launch_bounds__(256) void gemm_bf16_kernel(
        const __bf16* __restrict__ A, const __bf16* __restrict__ B,
        const __bf16* __restrict__ bias, const __bf16* __restrict__ residual,
        __bf16* __restrict__ C, int M, int N, int K, int act) {
    __shared__ __bf16 As[BM][BK + PAD];
    __shared__ __bf16 Bs[BN][BK + PAD];  // B tile transposed: Bs[n][k]

    const int bm = blockIdx.x * BM;
    const int bn = blockIdx.y * BN;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int wm = (wid >> 1) * 32;  // wave row offset in tile
    const int wn = (wid & 1) * 32;   // wave col offset in tile

    f32x4 acc[2][2] = {};

    for (int k0 = 0; k0 < K; k0 += BK) {
        // ---- stage A tile: each thread 2x 16-B pieces ----
        #pragma unroll
        for (int it = 0; it < 2; ++it) {
            const int r = (tid >> 3) + it * 32;
            const int c = (tid & 7) * 8;
            const int gr = bm + r;
            const int gc = k0 + c;
            bf16x8 v = {};
            if (gr < M) {
                if (gc + 7 < K) {
                    v = *reinterpret_cast<const bf16x8*>(
                        A + (long)gr * K + gc);
                } else {
                    for (int j = 0; j < 8; ++j)
                        if (gc + j < K) v[j] = A[(long)gr * K + gc + j];
                }
            }
            *reinterpret_cast<bf16x8*>(&As[r][c]) = v;
        }
        // ---- stage B tile transposed ----
        #pragma unroll
        for (int it = 0; it < 2; ++it) {
            const int kr = (tid >> 3) + it * 32;
            const int nc = (tid & 7) * 8;
            const int gk = k0 + kr;
            const int gn = bn + nc;
            bf16x8 v = {};
            if (gk < K) {
                if (gn + 7 < N) {
                    v = *reinterpret_cast<const bf16x8*>(
                        B + (long)gk * N + gn);
                } else {
                    for (int j = 0; j < 8; ++j)
                        if (gn + j < N) v[j] = B[(long)gk * N + gn + j];
                }
            }
            #pragma unroll
            for (int j = 0; j < 8; ++j) Bs[nc + j][kr] = v[j];
        }
        __syncthreads();

        // ---- MFMA inner loop over the K-tile ----
        #pragma unroll
        for (int kk = 0; kk < BK; kk += 32) {
            const int kbase = kk + (lane >> 4) * 8;
            #pragma unroll
            for (int mi = 0; mi < 2; ++mi) {
                const bf16x8 a = *reinterpret_cast<const bf16x8*>(
                    &As[wm + mi * 16 + (lane & 15)][kbase]);
                #pragma unroll
                for (int ni = 0; ni < 2; ++ni) {
                    const bf16x8 b = *reinterpret_cast<const bf16x8*>(
                        &Bs[wn + ni * 16 + (lane & 15)][kbase]);
                    acc[mi][ni] = mfma_16x16x32_bf16(a, b, acc[mi][ni]);
                }
            }
        }
        __syncthreads();
    }

    // ---- epilogue: bias + activation + residual, bf16 store ----
    #pragma unroll
    for (int mi = 0; mi < 2; ++mi) {
        #pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
            const int col = bn + wn + ni * 16 + (lane & 15);
            if (col >= N) continue;
            const float bv = bias != nullptr ? bf2f(bias[col]) : 0.0f;
            #pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int row = bm + wm + mi * 16 + (lane >> 4) * 4 + r;
                if (row >= M) continue;
                float v = acc[mi][ni][r] + bv;
                if (act == ACT_GELU) v = gelu_f(v);
                else if (act == ACT_TANH) v = tanhf(v);
                if (residual != nullptr)
                    v += bf2f(residual[(long)row * N + col]);
                C[(long)row * N + col] = f2bf(v);
            }
        }
    }
}

}  // namespace

torch::Tensor gemm_bf16(torch::Tensor a2d, torch::Tensor b,
                        c10::optional<torch::Tensor> bias, int64_t act,
                        c10::optional<torch::Tensor> residual2d) {
    TORCH_CHECK(a2d.is_cuda() && a2d.dtype() == torch::kBFloat16,
                "gemm: A must be bf16 on GPU");
    TORCH_CHECK(b.dtype() == torch::kBFloat16, "gemm: B must be bf16");
    TORCH_CHECK(a2d.dim() == 2 && b.dim() == 2);
    const int M = a2d.size(0), K = a2d.size(1), N = b.size(1);
    TORCH_CHECK(b.size(0) == K, "gemm: inner dims mismatch");
    auto a_c = a2d.contiguous();
    auto b_c = b.contiguous();
    auto out = torch::empty({M, N}, a2d.options());
    const __bf16* bias_p = nullptr;
    torch::Tensor bias_c;
    if (bias.has_value()) {
        bias_c = bias->contiguous();
        TORCH_CHECK(bias_c.numel() == N, "gemm: bias size mismatch");
        bias_p = reinterpret_cast<const __bf16*>(bias_c.data_ptr());
    }
    const __bf16* res_p = nullptr;
    torch::Tensor res_c;
    if (residual2d.has_value()) {
        res_c = residual2d->contiguous();
        TORCH_CHECK(res_c.size(0) == M && res_c.size(1) == N,
                    "gemm: residual shape mismatch");
        res_p = reinterpret_cast<const __bf16*>(res_c.data_ptr());
    }
    dim3 grid((M + BM - 1) / BM, (N + BN - 1) / BN);
    hipLaunchKernelGGL(gemm_bf16_kernel, grid, dim3(256), 0,
                       pe_current_stream(),
                       reinterpret_cast<const __bf16*>(a_c.data_ptr()),
                       reinterpret_cast<const __bf16*>(b_c.data_ptr()),
                       bias_p, res_p,
                       reinterpret_cast<__bf16*>(out.data_ptr()),
                       M, N, K, (int)act);
    HIP_CHECK_LAST();
    return out;
}

// debug helper: one 16x16x32 MFMA tile with the documented fragment layout -
// used by the GPU tests to pin the lane->element mapping on real hardware
torch::Tensor debug_mfma_tile(torch::Tensor a, torch::Tensor b);

namespace {
__global__ void debug_mfma_kernel(const __bf16* __restrict__ A,
                                  const __bf16* __restrict__ B,
                                  float* __restrict__ D) {
    const int lane = threadIdx.x & 63;
    bf16x8 a, b;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
        a[j] = A[(lane & 15) * 32 + (lane >> 4) * 8 + j];   // A[row][k]
        b[j] = B[((lane >> 4) * 8 + j) * 16 + (lane & 15)]; // B[k][col]
    }
    f32x4 c = {};
    c = mfma_16x16x32_bf16(a, b, c);
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
        const int row = (lane >> 4) * 4 + r;
        const int col = lane & 15;
        D[row * 16 + col] = c[r];
    }
}
}  // namespace

torch::Tensor debug_mfma_tile(torch::Tensor a, torch::Tensor b) {
    TORCH_CHECK(a.sizes() == torch::IntArrayRef({16, 32}));
    TORCH_CHECK(b.sizes() == torch::IntArrayRef({32, 16}));
    auto a_c = a.to(torch::kBFloat16).contiguous();
    auto b_c = b.to(torch::kBFloat16).contiguous();
    auto out = torch::empty({16, 16},
                            a.options().dtype(torch::kFloat32));
    hipLaunchKernelGGL(debug_mfma_kernel, dim3(1), dim3(64), 0,
                       pe_current_stream(),
                       reinterpret_cast<const __bf16*>(a_c.data_ptr()),
                       reinterpret_cast<const __bf16*>(b_c.data_ptr()),
                       out.data_ptr<float>());
    HIP_CHECK_LAST();
    return out;
}
