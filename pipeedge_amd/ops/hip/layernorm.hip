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

#include <cstdlib>

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
    sum = wave64_reduce(sum, SumOp());
    sumsq = wave64_reduce(sumsq, SumOp());
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

// split-row variant: TWO waves per row (doubles thread-level parallelism
// for the latency-bound small-row regime), cross-wave stats via LDS
template <bool WITH_RESIDUAL>
__global__ __launch_bounds__(256) void layernorm_split_kernel(
        const __bf16* __restrict__ x, const __bf16* __restrict__ residual,
        const __bf16* __restrict__ weight, const __bf16* __restrict__ bias,
        __bf16* __restrict__ out, long rows, int D, float eps) {
    __shared__ float stats[2][2][2];  // [row-in-block][wave-half][sum/sumsq]
    const long row = blockIdx.x * 2 + (threadIdx.x >> 7);
    if (row >= rows) return;
    const int half = (threadIdx.x >> 6) & 1;   // which half of the row
    const int lane = threadIdx.x & 63;
    const int rb = threadIdx.x >> 7;           // row slot in block
    const int base = half * (D / 2);
    const __bf16* xr = x + row * D;
    const __bf16* rr = WITH_RESIDUAL ? residual + row * D : nullptr;

    float vals[2][8];
    float sum = 0.0f, sumsq = 0.0f;
    #pragma unroll
    for (int j = 0; j < 2; ++j) {
        const int i = base + (j * 64 + lane) * 8;
        if (i < base + D / 2) {
            bf16x8 v = *reinterpret_cast<const bf16x8*>(xr + i);
            #pragma unroll
            for (int k = 0; k < 8; ++k) {
                float f = bf2f(v[k]);
                if (WITH_RESIDUAL) f += bf2f(rr[i + k]);
                vals[j][k] = f;
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
    if (lane == 0) {
        stats[rb][half][0] = sum;
        stats[rb][half][1] = sumsq;
    }
    __syncthreads();
    sum = stats[rb][0][0] + stats[rb][1][0];
    sumsq = stats[rb][0][1] + stats[rb][1][1];
    const float mean = sum / D;
    const float rstd = rsqrtf(sumsq / D - mean * mean + eps);
    #pragma unroll
    for (int j = 0; j < 2; ++j) {
        const int i = base + (j * 64 + lane) * 8;
        if (i < base + D / 2) {
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

// 8-rows-per-block variant: one 32-lane half-wave per row. Halves the
// number of dispatch rounds at transformer row counts (1576 rows: 394
// blocks -> 197, i.e. two HBM-latency rounds -> one); D <= NV * 256.
template <bool WITH_RESIDUAL, int NV>
__global__ __launch_bounds__(256) void layernorm_reg8_kernel(
        const __bf16* __restrict__ x, const __bf16* __restrict__ residual,
        const __bf16* __restrict__ weight, const __bf16* __restrict__ bias,
        __bf16* __restrict__ out, long rows, int D, float eps) {
    const long row = blockIdx.x * 8 + (threadIdx.x >> 5);
    if (row >= rows) return;
    const int lane = threadIdx.x & 31;
    const __bf16* xr = x + row * D;
    const __bf16* rr = WITH_RESIDUAL ? residual + row * D : nullptr;

    float vals[NV][8];
    float sum = 0.0f, sumsq = 0.0f;
    #pragma unroll
    for (int j = 0; j < NV; ++j) {
        const int i = (j * 32 + lane) * 8;
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
    sum = half32_reduce(sum, SumOp());
    sumsq = half32_reduce(sumsq, SumOp());
    const float mean = sum / D;
    const float rstd = rsqrtf(sumsq / D - mean * mean + eps);

    #pragma unroll
    for (int j = 0; j < NV; ++j) {
        const int i = (j * 32 + lane) * 8;
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
    static const bool split = []() {
        const char* e = std::getenv("PIPEEDGE_LN_SPLIT");
        return e != nullptr && std::atoi(e) != 0;  // default OFF (HW sweep:
        // the 1-wave register path wins at transformer row counts)
    }();
    static const int rows_per_block = []() {
        const char* e = std::getenv("PIPEEDGE_LN_R");
        return e ? std::atoi(e) : 4;
    }();
    if (rows_per_block == 8 && D <= 1024) {
        const dim3 grid8((rows + 7) / 8);
        if (D <= 256)
            hipLaunchKernelGGL((layernorm_reg8_kernel<WITH_RESIDUAL, 1>),
                               grid8, block, 0, stream, x, res, w, b, out,
                               rows, D, eps);
        else if (D <= 512)
            hipLaunchKernelGGL((layernorm_reg8_kernel<WITH_RESIDUAL, 2>),
                               grid8, block, 0, stream, x, res, w, b, out,
                               rows, D, eps);
        else if (D <= 768)
            hipLaunchKernelGGL((layernorm_reg8_kernel<WITH_RESIDUAL, 3>),
                               grid8, block, 0, stream, x, res, w, b, out,
                               rows, D, eps);
        else
            hipLaunchKernelGGL((layernorm_reg8_kernel<WITH_RESIDUAL, 4>),
                               grid8, block, 0, stream, x, res, w, b, out,
                               rows, D, eps);
        HIP_CHECK_LAST();
        return;
    }
    if (split && D % 16 == 0 && D / 2 <= 1024) {
        // 2 waves per row, 2 rows per block
        hipLaunchKernelGGL((layernorm_split_kernel<WITH_RESIDUAL>),
                           dim3((rows + 1) / 2), block, 0, stream, x, res, w,
                           b, out, rows, D, eps);
    } else if (D <= 512) {
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

// -------------------------------------------------------------------------
// ln_stats: per-row (sum, sumsq) in f32 - seeds the folded-LN GEMM epilogue
// (ops/__init__.py ln_linear) at stage entry, where no upstream GEMM
// emitted stats.
// -------------------------------------------------------------------------
namespace {
__global__ __launch_bounds__(256) void ln_stats_kernel(
        const __bf16* __restrict__ x, float* __restrict__ stats, long rows,
        int D) {
    const long row = blockIdx.x * 4 + (threadIdx.x >> 6);
    if (row >= rows) return;
    const int lane = threadIdx.x & 63;
    const __bf16* xr = x + row * D;
    float sum = 0.0f, sumsq = 0.0f;
    for (int i = lane * 8; i < D; i += 64 * 8) {
        bf16x8 v = *reinterpret_cast<const bf16x8*>(xr + i);
        #pragma unroll
        for (int k = 0; k < 8; ++k) {
            const float f = bf2f(v[k]);
            sum += f;
            sumsq += f * f;
        }
    }
    #pragma unroll
    for (int m = 32; m >= 1; m >>= 1) {
        sum += __shfl_xor(sum, m);
        sumsq += __shfl_xor(sumsq, m);
    }
    if (lane == 0) {
        stats[row * 2] = sum;
        stats[row * 2 + 1] = sumsq;
    }
}
}  // namespace

torch::Tensor ln_stats_bf16(torch::Tensor x) {
    TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16);
    TORCH_CHECK(x.size(-1) % 8 == 0);
    auto x_c = x.contiguous();
    const int D = x.size(-1);
    const long rows = x.numel() / D;
    auto stats = torch::empty({rows, 1, 2},
                              x.options().dtype(torch::kFloat32));
    hipLaunchKernelGGL(ln_stats_kernel, dim3((rows + 3) / 4), dim3(256), 0,
                       pe_current_stream(),
                       reinterpret_cast<const __bf16*>(x_c.data_ptr()),
                       stats.data_ptr<float>(), rows, D);
    HIP_CHECK_LAST();
    return stats;
}
