// Hardware layout probes for the primitives the attention v2 kernel builds
// on (guide section 5.4 rule 16: pin lane->element mappings on real hardware
// with asymmetric inputs before writing kernels against them):
//   - v_mfma_f32_32x32x16_bf16 fragment layouts
//   - ds_read_b64_tr_b16 lane/element mapping
//   - v_permlane32_swap_b32 exchange semantics
#include <torch/extension.h>

#include "common.h"

namespace {

typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef unsigned uint2_ __attribute__((ext_vector_type(2)));
typedef __bf16 v4bf __attribute__((ext_vector_type(4)));

// raw dump: lane l supplies a[l*8..l*8+7] as the A frag and b[...] as B;
// output is the raw per-lane C registers d[lane][0..15]
__global__ void mfma32_raw_kernel(const __bf16* __restrict__ a,
                                  const __bf16* __restrict__ b,
                                  float* __restrict__ d) {
    const int lane = threadIdx.x;
    bf16x8 af, bf;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
        af[j] = a[lane * 8 + j];
        bf[j] = b[lane * 8 + j];
    }
    f32x16 c = {};
    c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, c, 0, 0, 0);
    #pragma unroll
    for (int i = 0; i < 16; ++i) d[lane * 16 + i] = c[i];
}

// tr-read dump: LDS[e] = e (as bf16-convertible small ints); lane l reads
// 64 bits at byte offset l*8; output out[lane][0..3] = returned elements
__global__ void tr16_raw_kernel(float* __restrict__ out) {
    __shared__ __bf16 lds[512];
    const int lane = threadIdx.x;
    for (int i = lane; i < 512; i += 64) lds[i] = (__bf16)(float)i;
    __syncthreads();
    auto* p = reinterpret_cast<__attribute__((address_space(3))) v4bf*>(
        reinterpret_cast<uintptr_t>(&lds[0]) + lane * 8);
    v4bf r = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p);
    #pragma unroll
    for (int j = 0; j < 4; ++j) out[lane * 4 + j] = (float)r[j];
}

// permlane32_swap dump: vdst = lane, src = 1000 + lane; out = both results
__global__ void permlane32_raw_kernel(int* __restrict__ out) {
    const int lane = threadIdx.x;
    uint2_ r = __builtin_amdgcn_permlane32_swap((unsigned)lane,
                                                (unsigned)(1000 + lane),
                                                false, false);
    out[lane * 2] = (int)r.x;
    out[lane * 2 + 1] = (int)r.y;
}

}  // namespace

torch::Tensor debug_mfma32_raw(torch::Tensor a, torch::Tensor b) {
    TORCH_CHECK(a.numel() == 512 && b.numel() == 512);
    auto a_c = a.to(torch::kBFloat16).cuda().contiguous();
    auto b_c = b.to(torch::kBFloat16).cuda().contiguous();
    auto d = torch::zeros({64, 16},
                          torch::dtype(torch::kFloat32).device(torch::kCUDA));
    hipLaunchKernelGGL(mfma32_raw_kernel, dim3(1), dim3(64), 0,
                       pe_current_stream(),
                       reinterpret_cast<const __bf16*>(a_c.data_ptr()),
                       reinterpret_cast<const __bf16*>(b_c.data_ptr()),
                       d.data_ptr<float>());
    HIP_CHECK_LAST();
    return d;
}

torch::Tensor debug_tr16_raw() {
    auto d = torch::zeros({64, 4},
                          torch::dtype(torch::kFloat32).device(torch::kCUDA));
    hipLaunchKernelGGL(tr16_raw_kernel, dim3(1), dim3(64), 0,
                       pe_current_stream(), d.data_ptr<float>());
    HIP_CHECK_LAST();
    return d;
}

torch::Tensor debug_permlane32_raw() {
    auto d = torch::zeros({64, 2},
                          torch::dtype(torch::kInt32).device(torch::kCUDA));
    hipLaunchKernelGGL(permlane32_raw_kernel, dim3(1), dim3(64), 0,
                       pe_current_stream(), d.data_ptr<int>());
    HIP_CHECK_LAST();
    return d;
}
