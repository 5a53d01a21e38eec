// Fused bias + exact-erf GELU forward/backward for MI355X (gfx950).
//
// The bias+activation fusion point of the reference's LinearActivation
// (src/modeling.py:141-185, jit bias_gelu at :126-139). Elementwise,
// memory-bound: grid-stride waves, 16 B/lane vector IO. Backward also
// produces dbias = column-sum(dx) via per-block LDS accumulation + a
// deterministic column-reduce pass.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../common.h"

namespace bpa {

// declared in layernorm.hip
__global__ void col_reduce_kernel(const float* __restrict__ parts, int nparts,
                                  int H, float* __restrict__ out);

template <typename T, int VEC>
__global__ void bias_gelu_fwd_kernel(const T* __restrict__ x,
                                     const float* __restrict__ bias,
                                     T* __restrict__ y, int64_t rows, int H) {
  const int64_t total_vec = rows * (H / VEC);
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total_vec;
       i += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    const int64_t e = i * VEC;
    const int c = static_cast<int>(e % H);
    T v[VEC], o[VEC];
    *reinterpret_cast<uint4*>(v) = *reinterpret_cast<const uint4*>(x + e);
#pragma unroll
    for (int k = 0; k < VEC; ++k) {
      o[k] = DTraits<T>::from_f32(
          gelu_fwd(DTraits<T>::to_f32(v[k]) + bias[c + k]));
    }
    *reinterpret_cast<uint4*>(y + e) = *reinterpret_cast<const uint4*>(o);
  }
}

template <typename T, int VEC>
__global__ void bias_gelu_bwd_kernel(const T* __restrict__ dy,
                                     const T* __restrict__ x,
                                     const float* __restrict__ bias,
                                     T* __restrict__ dx, int64_t rows, int H) {
  const int64_t total_vec = rows * (H / VEC);
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total_vec;
       i += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    const int64_t e = i * VEC;
    const int c = static_cast<int>(e % H);
    T dv[VEC], xv[VEC], o[VEC];
    *reinterpret_cast<uint4*>(dv) = *reinterpret_cast<const uint4*>(dy + e);
    *reinterpret_cast<uint4*>(xv) = *reinterpret_cast<const uint4*>(x + e);
#pragma unroll
    for (int k = 0; k < VEC; ++k) {
      o[k] = DTraits<T>::from_f32(
          DTraits<T>::to_f32(dv[k]) *
          gelu_bwd(DTraits<T>::to_f32(xv[k]) + bias[c + k]));
    }
    *reinterpret_cast<uint4*>(dx + e) = *reinterpret_cast<const uint4*>(o);
  }
}

// column sum of a [rows, H] tensor into fp32 out[H] (atomic per chunk).
// grid (ceil(H/256), ceil(rows/kColSumRows)); out must be zero-filled.
constexpr int kColSumRows = 64;
template <typename T>
__global__ void col_sum_kernel(const T* __restrict__ src, int64_t rows, int H,
                               float* __restrict__ out) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= H) return;
  const int64_t r0 = static_cast<int64_t>(blockIdx.y) * kColSumRows;
  const int64_t r1 = tmin<int64_t>(r0 + kColSumRows, rows);
  float acc = 0.f;
  for (int64_t r = r0; r < r1; ++r)
    acc += DTraits<T>::to_f32(src[r * H + c]);
  atomicAdd(&out[c], acc);
}

#define DISPATCH_FLOATING2(TYPE, NAME, ...)                                  \
  [&] {                                                                      \
    if (TYPE == at::kBFloat16) {                                             \
      using scalar_t = __hip_bfloat16;                                       \
      constexpr int kVec = 8;                                                \
      return __VA_ARGS__();                                                  \
    } else if (TYPE == at::kHalf) {                                          \
      using scalar_t = __half;                                               \
      constexpr int kVec = 8;                                                \
      return __VA_ARGS__();                                                  \
    } else if (TYPE == at::kFloat) {                                         \
      using scalar_t = float;                                                \
      constexpr int kVec = 4;                                                \
      return __VA_ARGS__();                                                  \
    } else {                                                                 \
      TORCH_CHECK(false, NAME, ": unsupported dtype");                       \
    }                                                                        \
  }()

torch::Tensor bias_gelu_fwd(torch::Tensor x, torch::Tensor bias) {
  TORCH_CHECK(x.dim() == 2 && x.is_contiguous(), "bias_gelu_fwd: bad x");
  const int64_t rows = x.size(0);
  const int H = x.size(1);
  auto bias_f = bias.contiguous().to(torch::kFloat32);
  auto y = torch::empty_like(x);
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_FLOATING2(x.scalar_type(), "bias_gelu_fwd", [&] {
    TORCH_CHECK(H % kVec == 0, "bias_gelu_fwd: H % ", kVec, " != 0");
    const int64_t total = rows * (H / kVec);
    const int blocks =
        static_cast<int>(tmin<int64_t>((total + 255) / 256, 2048));
    hipLaunchKernelGGL((bias_gelu_fwd_kernel<scalar_t, kVec>), dim3(blocks),
                       dim3(256), 0, stream,
                       reinterpret_cast<const scalar_t*>(x.data_ptr()),
                       bias_f.data_ptr<float>(),
                       reinterpret_cast<scalar_t*>(y.data_ptr()), rows, H);
  });
  return y;
}

std::vector<torch::Tensor> bias_gelu_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor bias) {
  const int64_t rows = x.size(0);
  const int H = x.size(1);
  auto bias_f = bias.contiguous().to(torch::kFloat32);
  auto dx = torch::empty_like(x);
  auto opts = x.options().dtype(torch::kFloat32);
  auto dbias = torch::zeros({H}, opts);
  auto stream = at::hip::getCurrentHIPStream();
  auto dy_c = dy.contiguous();
  DISPATCH_FLOATING2(x.scalar_type(), "bias_gelu_bwd", [&] {
    TORCH_CHECK(H % kVec == 0, "bias_gelu_bwd: H % ", kVec, " != 0");
    const int64_t total = rows * (H / kVec);
    const int blocks =
        static_cast<int>(tmin<int64_t>((total + 255) / 256, 2048));
    hipLaunchKernelGGL((bias_gelu_bwd_kernel<scalar_t, kVec>), dim3(blocks),
                       dim3(256), 0, stream,
                       reinterpret_cast<const scalar_t*>(dy_c.data_ptr()),
                       reinterpret_cast<const scalar_t*>(x.data_ptr()),
                       bias_f.data_ptr<float>(),
                       reinterpret_cast<scalar_t*>(dx.data_ptr()), rows, H);
    dim3 sgrid((H + 255) / 256,
               static_cast<unsigned>((rows + kColSumRows - 1) / kColSumRows));
    hipLaunchKernelGGL((col_sum_kernel<scalar_t>), sgrid, dim3(256), 0,
                       stream,
                       reinterpret_cast<const scalar_t*>(dx.data_ptr()), rows,
                       H, dbias.data_ptr<float>());
  });
  if (bias.scalar_type() != torch::kFloat32) {
    return {dx, dbias.to(bias.scalar_type())};
  }
  return {dx, dbias};
}

}  // namespace bpa
