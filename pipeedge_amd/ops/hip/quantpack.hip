// QuantPipe activation quantization kernels (bf16 <-> bit-packed uint32).
//
// GPU implementation of the reference's numpy encode/decode
// (reference quantization/basic_op.py; SURVEY.md section 2.6 item 9 - the
// reference runs this per-microbatch on the HOST): per-image shift/scale to
// [0,1], round to (1<<bit)-1 levels, pack floor(32/bit) values per 32-bit
// word (value i at bit offset (i % ratio) * bit), transported as
// little-endian uint8. Wire format unchanged:
//   [comm u8, shape i32, scale f32, shift f32, bit i8] per image.
//
// Three kernels: per-image min/max (grid-stride + LDS reduce), pack
// (one thread per output word), unpack (one thread per word).
#include <torch/extension.h>

#include "common.h"

namespace {

__global__ __launch_bounds__(256) void minmax_kernel(
        const __bf16* __restrict__ x, float* __restrict__ mn,
        float* __restrict__ mx, long per_image) {
    __shared__ float red[2][256];
    const long base = (long)blockIdx.x * per_image;
    float lo = 1e30f, hi = -1e30f;
    for (long i = threadIdx.x * 8; i + 7 < per_image; i += 256 * 8) {
        bf16x8 v = *reinterpret_cast<const bf16x8*>(x + base + i);
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
            const float f = bf2f(v[j]);
            lo = fminf(lo, f);
            hi = fmaxf(hi, f);
        }
    }
    // tail (per_image not a multiple of 8)
    for (long i = (per_image / 8) * 8 + threadIdx.x; i < per_image; i += 256) {
        const float f = bf2f(x[base + i]);
        lo = fminf(lo, f);
        hi = fmaxf(hi, f);
    }
    red[0][threadIdx.x] = lo;
    red[1][threadIdx.x] = hi;
    __syncthreads();
    for (int s = 128; s > 0; s >>= 1) {
        if (threadIdx.x < s) {
            red[0][threadIdx.x] = fminf(red[0][threadIdx.x],
                                        red[0][threadIdx.x + s]);
            red[1][threadIdx.x] = fmaxf(red[1][threadIdx.x],
                                        red[1][threadIdx.x + s]);
        }
        __syncthreads();
    }
    if (threadIdx.x == 0) {
        mn[blockIdx.x] = red[0][0];
        mx[blockIdx.x] = red[1][0];
    }
}

// one thread per packed uint32 word
__global__ __launch_bounds__(256) void pack_kernel(
        const __bf16* __restrict__ x, unsigned int* __restrict__ words,
        const float* __restrict__ mn, const float* __restrict__ mx,
        long n_total, long per_image, long words_per_image, long pad_words,
        int bit) {
    const long gid = (long)blockIdx.x * blockDim.x + threadIdx.x;
    if (gid >= n_total) return;  // grid rounds up to blockDim multiples
    const long img = gid / pad_words;
    const long w = gid % pad_words;
    if (w >= words_per_image) {  // padding for the static wire schema
        words[gid] = 0u;
        return;
    }
    const int ratio = 32 / bit;
    const float shift = mn[img];
    const float scale_raw = mx[img] - shift;
    const float scale = scale_raw != 0.0f ? scale_raw : 1.0f;
    const double levels = (double)((1ull << bit) - 1ull);
    unsigned int word = 0;
    for (int i = 0; i < ratio; ++i) {
        const long idx = w * (long)ratio + i;
        if (idx < per_image) {
            const float f = (bf2f(x[img * per_image + idx]) - shift) / scale;
            // rint = round-half-to-even, matching np.around / torch.round;
            // double keeps levels*f exact up to 2^32-1
            const unsigned long long q =
                (unsigned long long)llrint(levels * (double)f);
            word |= (unsigned int)(q << (i * bit));
        }
    }
    words[gid] = word;
}

// one thread per packed word; writes ratio bf16 values
__global__ __launch_bounds__(256) void unpack_kernel(
        const unsigned int* __restrict__ words, __bf16* __restrict__ out,
        const float* __restrict__ scale, const float* __restrict__ shift,
        long n_total, long per_image, long pad_words, int bit) {
    const long gid = (long)blockIdx.x * blockDim.x + threadIdx.x;
    if (gid >= n_total) return;  // grid rounds up to blockDim multiples
    const long img = gid / pad_words;
    const long w = gid % pad_words;
    const int ratio = 32 / bit;
    if (w * (long)ratio >= per_image) return;
    const float sc = scale[img];
    const float sh = shift[img];
    const float inv_levels = 1.0f / (float)((1ull << bit) - 1ull);
    const unsigned int word = words[gid];
    const unsigned long long mask = (1ull << bit) - 1ull;
    for (int i = 0; i < ratio; ++i) {
        const long idx = w * (long)ratio + i;
        if (idx < per_image)
            out[img * per_image + idx] = f2bf(
                (float)((word >> (i * bit)) & mask) * inv_levels * sc + sh);
    }
}

}  // namespace

// returns [comm u8 [U, pad_words*4], scale f32 [U], shift f32 [U]]
std::vector<torch::Tensor> quant_encode_bf16(torch::Tensor x, int64_t bit,
                                             int64_t pad_words) {
    TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16,
                "quant_encode: bf16 GPU tensor required");
    TORCH_CHECK(bit >= 1 && bit <= 32);
    auto x_c = x.contiguous();
    const long u = x.size(0);
    const long per_image = x.numel() / u;
    const int ratio = 32 / (int)bit;
    const long words_per_image = (per_image + ratio - 1) / ratio;
    const long padw = pad_words > 0 ? pad_words : words_per_image;
    TORCH_CHECK(padw >= words_per_image, "pad_words too small");
    auto opts_f = x.options().dtype(torch::kFloat32);
    auto mn = torch::empty({u}, opts_f);
    auto mx = torch::empty({u}, opts_f);
    auto stream = pe_current_stream();
    hipLaunchKernelGGL(minmax_kernel, dim3(u), dim3(256), 0, stream,
                       reinterpret_cast<const __bf16*>(x_c.data_ptr()),
                       mn.data_ptr<float>(), mx.data_ptr<float>(), per_image);
    auto comm = torch::empty({u, padw * 4}, x.options().dtype(torch::kUInt8));
    const long total = u * padw;
    hipLaunchKernelGGL(pack_kernel, dim3((total + 255) / 256), dim3(256), 0,
                       stream,
                       reinterpret_cast<const __bf16*>(x_c.data_ptr()),
                       reinterpret_cast<unsigned int*>(comm.data_ptr()),
                       mn.data_ptr<float>(), mx.data_ptr<float>(), total,
                       per_image, words_per_image, padw, (int)bit);
    HIP_CHECK_LAST();
    auto scale = (mx - mn);
    return {comm, scale, mn};
}

torch::Tensor quant_decode_bf16(torch::Tensor comm, torch::Tensor scale,
                                torch::Tensor shift, int64_t bit,
                                std::vector<int64_t> image_shape) {
    TORCH_CHECK(comm.is_cuda() && comm.dtype() == torch::kUInt8);
    auto comm_c = comm.contiguous();
    const long u = comm.size(0);
    long per_image = 1;
    for (auto s : image_shape) per_image *= s;
    const long padw = comm.size(1) / 4;
    std::vector<int64_t> out_shape{u};
    out_shape.insert(out_shape.end(), image_shape.begin(), image_shape.end());
    auto out = torch::empty(out_shape, comm.options().dtype(torch::kBFloat16));
    auto scale_f = scale.to(torch::kFloat32).contiguous();
    auto shift_f = shift.to(torch::kFloat32).contiguous();
    const long total = u * padw;
    hipLaunchKernelGGL(unpack_kernel, dim3((total + 255) / 256), dim3(256),
                       0, pe_current_stream(),
                       reinterpret_cast<const unsigned int*>(comm_c.data_ptr()),
                       reinterpret_cast<__bf16*>(out.data_ptr()),
                       scale_f.data_ptr<float>(), shift_f.data_ptr<float>(),
                       total, per_image, padw, (int)bit);
    HIP_CHECK_LAST();
    return out;
}
