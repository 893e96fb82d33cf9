// Fused LayerNorm kernels (bf16 in/out, f32 statistics).
//
// One wave per row (seq position): vectorized bf16x8 loads, wave-level shfl
// reduction (no LDS, no barriers - cdna_hip_programming.md Appendix B
// "Reduction"). `add_layernorm` fuses the residual add of BERT's post-LN
// blocks (reference bert.py:41-52) into the same pass.
#include <torch/extension.h>

#include "common.h"

namespace {

template <bool WITH_RESIDUAL>
__global__ __launch_bounds__(256) void layernorm_kernel(
        const __bf16* __restrict__ x, const __bf16* __restrict__ residual,
        const __bf16* __restrict__ weight, const __bf16* __restrict__ bias,
        __bf16* __restrict__ out, long rows, int D, float eps) {
    // 4 waves per block, one row per wave (wave-local shfl reduction only)
    const long row = blockIdx.x * 4 + (threadIdx.x >> 6);
    if (row >= rows) return;
    const int lane = threadIdx.x & 63;
    const __bf16* xr = x + row * D;
    const __bf16* rr = WITH_RESIDUAL ? residual + row * D : nullptr;

    // strided vec8 pass: sum and sumsq
    float sum = 0.0f, sumsq = 0.0f;
    for (int i = lane * 8; i < D; i += 64 * 8) {
        bf16x8 v = *reinterpret_cast<const bf16x8*>(xr + i);
        if (WITH_RESIDUAL) {
            bf16x8 r = *reinterpret_cast<const bf16x8*>(rr + i);
            #pragma unroll
            for (int j = 0; j < 8; ++j) {
                const float f = bf2f(v[j]) + bf2f(r[j]);
                sum += f;
                sumsq += f * f;
            }
        } else {
            #pragma unroll
            for (int j = 0; j < 8; ++j) {
                const float f = bf2f(v[j]);
                sum += f;
                sumsq += f * f;
            }
        }
    }
    #pragma unroll
    for (int m = 32; m >= 1; m >>= 1) {
        sum += __shfl_xor(sum, m);
        sumsq += __shfl_xor(sumsq, m);
    }
    const float mean = sum / D;
    const float var = sumsq / D - mean * mean;
    const float rstd = rsqrtf(var + eps);

    for (int i = lane * 8; i < D; i += 64 * 8) {
        bf16x8 v = *reinterpret_cast<const bf16x8*>(xr + i);
        bf16x8 w = *reinterpret_cast<const bf16x8*>(weight + i);
        bf16x8 b = *reinterpret_cast<const bf16x8*>(bias + i);
        bf16x8 o;
        if (WITH_RESIDUAL) {
            bf16x8 r = *reinterpret_cast<const bf16x8*>(rr + i);
            #pragma unroll
            for (int j = 0; j < 8; ++j) {
                const float f = bf2f(v[j]) + bf2f(r[j]);
                o[j] = f2bf((f - mean) * rstd * bf2f(w[j]) + bf2f(b[j]));
            }
        } else {
            #pragma unroll
            for (int j = 0; j < 8; ++j) {
                const float f = bf2f(v[j]);
                o[j] = f2bf((f - mean) * rstd * bf2f(w[j]) + bf2f(b[j]));
            }
        }
        *reinterpret_cast<bf16x8*>(out + row * D + i) = o;
    }
}

}  // namespace

static void check_ln_args(const torch::Tensor& x, const torch::Tensor& w,
                          const torch::Tensor& b) {
    TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16,
                "layernorm: bf16 GPU tensors required");
    TORCH_CHECK(x.size(-1) % 8 == 0,
                "layernorm: hidden dim must be a multiple of 8");
    TORCH_CHECK(w.numel() == x.size(-1) && b.numel() == x.size(-1));
}

torch::Tensor layernorm_bf16(torch::Tensor x, torch::Tensor weight,
                             torch::Tensor bias, double eps) {
    check_ln_args(x, weight, bias);
    auto x_c = x.contiguous();
    auto w_c = weight.contiguous();
    auto b_c = bias.contiguous();
    auto out = torch::empty_like(x_c);
    const int D = x.size(-1);
    const long rows = x.numel() / D;
    hipLaunchKernelGGL((layernorm_kernel<false>), dim3((rows + 3) / 4),
                       dim3(256), 0, pe_current_stream(),
                       reinterpret_cast<const __bf16*>(x_c.data_ptr()),
                       nullptr,
                       reinterpret_cast<const __bf16*>(w_c.data_ptr()),
                       reinterpret_cast<const __bf16*>(b_c.data_ptr()),
                       reinterpret_cast<__bf16*>(out.data_ptr()),
                       rows, D, (float)eps);
    HIP_CHECK_LAST();
    return out;
}

torch::Tensor add_layernorm_bf16(torch::Tensor x, torch::Tensor residual,
                                 torch::Tensor weight, torch::Tensor bias,
                                 double eps) {
    check_ln_args(x, weight, bias);
    TORCH_CHECK(residual.sizes() == x.sizes());
    auto x_c = x.contiguous();
    auto r_c = residual.contiguous();
    auto w_c = weight.contiguous();
    auto b_c = bias.contiguous();
    auto out = torch::empty_like(x_c);
    const int D = x.size(-1);
    const long rows = x.numel() / D;
    hipLaunchKernelGGL((layernorm_kernel<true>), dim3((rows + 3) / 4),
                       dim3(256), 0, pe_current_stream(),
                       reinterpret_cast<const __bf16*>(x_c.data_ptr()),
                       reinterpret_cast<const __bf16*>(r_c.data_ptr()),
                       reinterpret_cast<const __bf16*>(w_c.data_ptr()),
                       reinterpret_cast<const __bf16*>(b_c.data_ptr()),
                       reinterpret_cast<__bf16*>(out.data_ptr()),
                       rows, D, (float)eps);
    HIP_CHECK_LAST();
    return out;
}
