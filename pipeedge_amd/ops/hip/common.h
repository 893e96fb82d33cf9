// Shared helpers for the pipeedge-amd CDNA4 (gfx950) kernels.
//
// Conventions:
// - compute dtype bf16, accumulate f32 via MFMA (v_mfma_f32_16x16x32_bf16);
// - wave = 64 lanes; blocks are multiples of 64 threads;
// - fragment layout for mfma_f32_16x16x32_bf16 (verified on HW by
//   tests/test_gpu_ops.py::test_mfma_fragment_layout):
//     A[16r][32k]: lane l holds 8 contiguous k: row=l&15, k=(l>>4)*8+j
//     B[32k][16c]: lane l holds 8 contiguous k: col=l&15, k=(l>>4)*8+j
//     C/D[16r][16c]: lane l, reg r: col=l&15, row=(l>>4)*4+r
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#ifdef __HIPCC__
#include <c10/hip/HIPStream.h>
inline hipStream_t pe_current_stream() {
    return c10::hip::getCurrentHIPStream().stream();
}
#endif

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(8))) unsigned short ushort8;

__device__ __forceinline__ f32x4 mfma_16x16x32_bf16(bf16x8 a, bf16x8 b, f32x4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

__device__ __forceinline__ float bf2f(__bf16 v) {
    return static_cast<float>(v);
}

__device__ __forceinline__ __bf16 f2bf(float v) {
    return static_cast<__bf16>(v);
}

// exact GeLU (erf form) matching torch.nn.functional.gelu default
__device__ __forceinline__ float gelu_f(float x) {
    return 0.5f * x * (1.0f + erff(x * 0.70710678118654752f));
}

// activation codes shared with pipeedge_amd/ops/__init__.py (_ACT_CODE)
#define ACT_NONE 0
#define ACT_GELU 1
#define ACT_TANH 2

// reduce over the 16-lane groups that share a C-fragment row set.
// Implemented with DPP row rotations (one VALU op per step: a DPP "row" is
// exactly 16 lanes) - __shfl_xor compiles to ds_bpermute (LDS-path, ~10x
// the cost) on gfx950. Rotations suffice for commutative reductions.
template <typename Op>
__device__ __forceinline__ float group16_reduce(float v, Op op) {
    #define PE_DPP_STEP(ctrl)                                               \
        {                                                                   \
            const int t_ = __builtin_amdgcn_update_dpp(                     \
                0, __builtin_bit_cast(int, v), (ctrl), 0xf, 0xf, true);     \
            v = op(v, __builtin_bit_cast(float, t_));                       \
        }
    PE_DPP_STEP(0x128)  // row_ror:8
    PE_DPP_STEP(0x124)  // row_ror:4
    PE_DPP_STEP(0x122)  // row_ror:2
    PE_DPP_STEP(0x121)  // row_ror:1
    #undef PE_DPP_STEP
    return v;
}

struct MaxOp {
    __device__ float operator()(float a, float b) const { return fmaxf(a, b); }
};
struct SumOp {
    __device__ float operator()(float a, float b) const { return a + b; }
};

// 32-lane (half-wave) reduction: DPP within the 16-lane rows, one
// cross-row exchange. xor-16 stays inside each 32-lane half.
template <typename Op>
__device__ __forceinline__ float half32_reduce(float v, Op op) {
    v = group16_reduce(v, op);
    return op(v, __shfl_xor(v, 16));
}

// full 64-lane reduction: 4 DPP steps + 2 bpermute exchanges (vs 6
// bpermute exchanges for a pure __shfl_xor ladder)
template <typename Op>
__device__ __forceinline__ float wave64_reduce(float v, Op op) {
    v = group16_reduce(v, op);
    v = op(v, __shfl_xor(v, 16));
    return op(v, __shfl_xor(v, 32));
}

#define HIP_CHECK_LAST()                                                     \
    do {                                                                     \
        hipError_t e_ = hipGetLastError();                                   \
        if (e_ != hipSuccess) {                                              \
            TORCH_CHECK(false, "HIP kernel launch failed: ",                 \
                        hipGetErrorString(e_));                              \
        }                                                                    \
    } while (0)
