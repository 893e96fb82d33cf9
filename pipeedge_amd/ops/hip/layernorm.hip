// Fused LayerNorm kernels (bf16 in/out, f32 statistics).
//
// Single-pass design: one wave per row holds the whole row in registers
// (NV x bf16x8 per lane), computes mean/var by wave-level shfl reduction (no
// LDS, no barriers - cdna_hip_programming.md Appendix B "Reduction"), and
// normalizes from registers - one global read + one write per element.
// 4 rows per 256-thread block. Rows wider than 8*64*NV_MAX fall back to a
// two-pass strided loop. `add_layernorm` fuses the residual add of BERT's
// post-LN blocks (reference bert.py:41-52) into the same pass.
#include <torch/extension.h>

#include "common.h"

namespace {

// single-pass: D <= NV * 512
template <bool WITH_RESIDUAL, int NV>
__global__ __launch_bounds__(256) void layernorm_reg_kernel(
        const __bf16* __restrict__ x, const __bf16* __restrict__ residual,
        const __bf16* __restrict__ weight, const __bf16* __restrict__ bias,
        __bf16* __restrict__ out, long rows, int D, float eps) {
    const long row = blockIdx.x * 4 + (threadIdx.x >> 6);
    if (row >= rows) return;
    const int lane = threadIdx.x & 63;
    const __bf16* xr = x + row * D;
    const __bf16* rr = WITH_RESIDUAL ? residual + row * D : nullptr;

    float vals[NV][8];
    float sum = 0.0f, sumsq = 0.0f;
    #pragma unroll
    for (int j = 0; j < NV; ++j) {
        const int i = (j * 64 + lane) * 8;
        if (i < D) {
            bf16x8 v = *reinterpret_cast<const bf16x8*>(xr + i);
            if (WITH_RESIDUAL) {
                bf16x8 r = *reinterpret_cast<const bf16x8*>(rr + i);
                #pragma unroll
                for (int k = 0; k < 8; ++k)
                    vals[j][k] = bf2f(v[k]) + bf2f(r[k]);
            } else {
                #pragma unroll
                for (int k = 0; k < 8; ++k) vals[j][k] = bf2f(v[k]);
            }
            #pragma unroll
            for (int k = 0; k < 8; ++k) {
                sum += vals[j][k];
                sumsq += vals[j][k] * vals[j][k];
            }
        }
    }
    #pragma unroll
    for (int m = 32; m >= 1; m >>= 1) {
        sum += __shfl_xor(sum, m);
        sumsq += __shfl_xor(sumsq, m);
    }
    const float mean = sum / D;
    const float rstd = rsqrtf(sumsq / D - mean * mean + eps);

    #pragma unroll
    for (int j = 0; j < NV; ++j) {
        const int i = (j * 64 + lane) * 8;
        if (i < D) {
            bf16x8 w = *reinterpret_cast<const bf16x8*>(weight + i);
            bf16x8 b = *reinterpret_cast<const bf16x8*>(bias + i);
            bf16x8 o;
            #pragma unroll
            for (int k = 0; k < 8; ++k)
                o[k] = f2bf((vals[j][k] - mean) * rstd * bf2f(w[k]) +
                            bf2f(b[k]));
            *reinterpret_cast<bf16x8*>(out + row * D + i) = o;
        }
    }
}

// two-pass fallback for very wide rows
template <bool WITH_RESIDUAL>
__global__ __launch_bounds__(256) void layernorm_wide_kernel(
        const __bf16* __restrict__ x, const __bf16* __restrict__ residual,
        const __bf16* __restrict__ weight, const __bf16* __restrict__ bias,
        __bf16* __restrict__ out, long rows, int D, float eps) {
    const long row = blockIdx.x * 4 + (threadIdx.x >> 6);
    if (row >= rows) return;
    const int lane = threadIdx.x & 63;
    const __bf16* xr = x + row * D;
    const __bf16* rr = WITH_RESIDUAL ? residual + row * D : nullptr;
    float sum = 0.0f, sumsq = 0.0f;
    for (int i = lane * 8; i < D; i += 64 * 8) {
        bf16x8 v = *reinterpret_cast<const bf16x8*>(xr + i);
        #pragma unroll
        for (int k = 0; k < 8; ++k) {
            float f = bf2f(v[k]);
            if (WITH_RESIDUAL)
                f += bf2f(rr[i + k]);
            sum += f;
            sumsq += f * f;
        }
    }
    #pragma unroll
    for (int m = 32; m >= 1; m >>= 1) {
        sum += __shfl_xor(sum, m);
        sumsq += __shfl_xor(sumsq, m);
    }
    const float mean = sum / D;
    const float rstd = rsqrtf(sumsq / D - mean * mean + eps);
    for (int i = lane * 8; i < D; i += 64 * 8) {
        bf16x8 v = *reinterpret_cast<const bf16x8*>(xr + i);
        bf16x8 w = *reinterpret_cast<const bf16x8*>(weight + i);
        bf16x8 b = *reinterpret_cast<const bf16x8*>(bias + i);
        bf16x8 o;
        #pragma unroll
        for (int k = 0; k < 8; ++k) {
            float f = bf2f(v[k]);
            if (WITH_RESIDUAL)
                f += bf2f(rr[i + k]);
            o[k] = f2bf((f - mean) * rstd * bf2f(w[k]) + bf2f(b[k]));
        }
        *reinterpret_cast<bf16x8*>(out + row * D + i) = o;
    }
}

template <bool WITH_RESIDUAL>
void launch_ln(const __bf16* x, const __bf16* res, const __bf16* w,
               const __bf16* b, __bf16* out, long rows, int D, float eps) {
    const dim3 grid((rows + 3) / 4);
    const dim3 block(256);
    auto stream = pe_current_stream();
    if (D <= 512) {
        hipLaunchKernelGGL((layernorm_reg_kernel<WITH_RESIDUAL, 1>), grid,
                           block, 0, stream, x, res, w, b, out, rows, D, eps);
    } else if (D <= 1024) {
        hipLaunchKernelGGL((layernorm_reg_kernel<WITH_RESIDUAL, 2>), grid,
                           block, 0, stream, x, res, w, b, out, rows, D, eps);
    } else if (D <= 1536) {
        hipLaunchKernelGGL((layernorm_reg_kernel<WITH_RESIDUAL, 3>), grid,
                           block, 0, stream, x, res, w, b, out, rows, D, eps);
    } else {
        hipLaunchKernelGGL((layernorm_wide_kernel<WITH_RESIDUAL>), grid,
                           block, 0, stream, x, res, w, b, out, rows, D, eps);
    }
    HIP_CHECK_LAST();
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
    launch_ln<false>(reinterpret_cast<const __bf16*>(x_c.data_ptr()), nullptr,
                     reinterpret_cast<const __bf16*>(w_c.data_ptr()),
                     reinterpret_cast<const __bf16*>(b_c.data_ptr()),
                     reinterpret_cast<__bf16*>(out.data_ptr()), rows, D,
                     (float)eps);
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
    launch_ln<true>(reinterpret_cast<const __bf16*>(x_c.data_ptr()),
                    reinterpret_cast<const __bf16*>(r_c.data_ptr()),
                    reinterpret_cast<const __bf16*>(w_c.data_ptr()),
                    reinterpret_cast<const __bf16*>(b_c.data_ptr()),
                    reinterpret_cast<__bf16*>(out.data_ptr()), rows, D,
                    (float)eps);
    return out;
}
