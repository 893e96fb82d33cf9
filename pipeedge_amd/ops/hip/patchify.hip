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

}  // namespace

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
