// Patch extraction for ViT/DeiT patch embedding.
//
// [B, C, H, W] -> [B, (H/P)*(W/P), C*P*P] with per-patch element order
// (c, ph, pw). This turns the reference's 16x16/s16 Conv2d
// (HF ViTEmbeddings, reference vit.py:96) into pure data movement + the MFMA
// GEMM in gemm.hip. Vectorized 16-B path when P is a multiple of 8.
#include <torch/extension.h>

#include "common.h"

namespace {

// vector path: each thread moves 8 contiguous pixels of one patch row piece
__global__ void patchify_vec8_kernel(const __bf16* __restrict__ in,
                                     __bf16* __restrict__ out,
                                     int B, int C, int H, int W, int P) {
    const int np_w = W / P;
    const int np = (H / P) * np_w;
    const int pdim = C * P * P;
    const long total = (long)B * np * (pdim / 8);
    const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= total) return;
    const int vec_per_patch = pdim / 8;
    const long bp = idx / vec_per_patch;     // (b, patch)
    const int e8 = idx % vec_per_patch;      // vec index within patch
    const int b = bp / np;
    const int p = bp % np;
    const int e = e8 * 8;                    // element index: c*P*P + ph*P + pw
    const int c = e / (P * P);
    const int ph = (e % (P * P)) / P;
    const int pw = e % P;
    const int py = p / np_w, px = p % np_w;
    const long src = (((long)b * C + c) * H + py * P + ph) * W + px * P + pw;
    *reinterpret_cast<bf16x8*>(out + bp * pdim + e) =
        *reinterpret_cast<const bf16x8*>(in + src);
}

// scalar path for P not a multiple of 8 (e.g. ViT-Huge P=14)
__global__ void patchify_scalar_kernel(const __bf16* __restrict__ in,
                                       __bf16* __restrict__ out,
                                       int B, int C, int H, int W, int P) {
    const int np_w = W / P;
    const int np = (H / P) * np_w;
    const int pdim = C * P * P;
    const long total = (long)B * np * pdim;
    const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= total) return;
    const long bp = idx / pdim;
    const int e = idx % pdim;
    const int b = bp / np;
    const int p = bp % np;
    const int c = e / (P * P);
    const int ph = (e % (P * P)) / P;
    const int pw = e % P;
    const int py = p / np_w, px = p % np_w;
    const long src = (((long)b * C + c) * H + py * P + ph) * W + px * P + pw;
    out[idx] = in[src];
}

// Fused embedding tail: out[b, s, :] = (s < T ? prefix[s] : x[b, s-T]) +
// pos[s]. Replaces the torch cat(cls/dist, patches) + pos-add + expand-copy
// trio (3 launch-bound kernels, ~19us per microbatch at ViT-B shape) with
// one pass (reference HF ViTEmbeddings.forward, reference vit.py:96).
__global__ void embed_finish_kernel(const __bf16* __restrict__ x,
                                    const __bf16* __restrict__ prefix,
                                    const __bf16* __restrict__ pos,
                                    __bf16* __restrict__ out,
                                    int B, int S, int T, int D) {
    const long total = (long)B * S * (D / 8);
    const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= total) return;
    const int vd = D / 8;
    const long row = idx / vd;
    const int d = (idx % vd) * 8;
    const int s = row % S;
    const long b = row / S;
    bf16x8 v;
    if (s < T) {
        v = *reinterpret_cast<const bf16x8*>(prefix + (long)s * D + d);
    } else {
        v = *reinterpret_cast<const bf16x8*>(
            x + ((long)b * (S - T) + (s - T)) * D + d);
    }
    const bf16x8 p = *reinterpret_cast<const bf16x8*>(pos + (long)s * D + d);
    bf16x8 o;
    #pragma unroll
    for (int k = 0; k < 8; ++k) o[k] = f2bf(bf2f(v[k]) + bf2f(p[k]));
    *reinterpret_cast<bf16x8*>(out + row * D + d) = o;
}

}  // namespace

torch::Tensor embed_finish_bf16(torch::Tensor x, torch::Tensor prefix,
                                torch::Tensor pos) {
    TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16,
                "embed_finish: bf16 GPU tensors required");
    TORCH_CHECK(x.dim() == 3 && prefix.dim() == 2 && pos.dim() == 2);
    auto x_c = x.contiguous();
    auto pre_c = prefix.contiguous();
    auto pos_c = pos.contiguous();
    const int B = x_c.size(0), NP = x_c.size(1), D = x_c.size(2);
    const int T = pre_c.size(0);
    const int S = T + NP;
    TORCH_CHECK(pos_c.size(0) == S && pos_c.size(1) == D &&
                pre_c.size(1) == D && D % 8 == 0);
    auto out = torch::empty({B, S, D}, x_c.options());
    const long total = (long)B * S * (D / 8);
    const int block = 256;
    hipLaunchKernelGGL(embed_finish_kernel,
                       dim3((total + block - 1) / block), dim3(block), 0,
                       pe_current_stream(),
                       reinterpret_cast<const __bf16*>(x_c.data_ptr()),
                       reinterpret_cast<const __bf16*>(pre_c.data_ptr()),
                       reinterpret_cast<const __bf16*>(pos_c.data_ptr()),
                       reinterpret_cast<__bf16*>(out.data_ptr()),
                       B, S, T, D);
    HIP_CHECK_LAST();
    return out;
}

torch::Tensor patchify_bf16(torch::Tensor pixels, int64_t patch_size) {
    TORCH_CHECK(pixels.is_cuda() && pixels.dtype() == torch::kBFloat16,
                "patchify: bf16 GPU tensor required");
    TORCH_CHECK(pixels.dim() == 4);
    auto x = pixels.contiguous();
    const int B = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
    const int P = (int)patch_size;
    TORCH_CHECK(H % P == 0 && W % P == 0);
    const int np = (H / P) * (W / P);
    const int pdim = C * P * P;
    auto out = torch::empty({B, np, pdim}, x.options());
    if (P % 8 == 0) {
        const long total = (long)B * np * (pdim / 8);
        const int block = 256;
        const long grid = (total + block - 1) / block;
        hipLaunchKernelGGL(patchify_vec8_kernel, dim3(grid), dim3(block), 0,
                           pe_current_stream(),
                           reinterpret_cast<const __bf16*>(x.data_ptr()),
                           reinterpret_cast<__bf16*>(out.data_ptr()),
                           B, C, H, W, P);
    } else {
        const long total = (long)B * np * pdim;
        const int block = 256;
        const long grid = (total + block - 1) / block;
        hipLaunchKernelGGL(patchify_scalar_kernel, dim3(grid), dim3(block), 0,
                           pe_current_stream(),
                           reinterpret_cast<const __bf16*>(x.data_ptr()),
                           reinterpret_cast<__bf16*>(out.data_ptr()),
                           B, C, H, W, P);
    }
    HIP_CHECK_LAST();
    return out;
}
