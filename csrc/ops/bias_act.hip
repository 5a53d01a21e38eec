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

template <typename T, int VEC, int NW>
__global__ void bias_gelu_bwd_kernel(const T* __restrict__ dy,
                                     const T* __restrict__ x,
                                     const float* __restrict__ bias,
                                     T* __restrict__ dx,
                                     float* __restrict__ part_dbias,
                                     int64_t rows, int H, int rows_per_block) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* lb = reinterpret_cast<float*>(smem_raw) + wave * H;  // [NW][H]
  for (int c = lane; c < H; c += WAVE_SIZE) lb[c] = 0.f;

  const int64_t row0 = static_cast<int64_t>(blockIdx.x) * rows_per_block;
  const int64_t row_end = min(row0 + rows_per_block, rows);
  for (int64_t r = row0 + wave; r < row_end; r += NW) {
    const T* dyr = dy + r * H;
    const T* xr = x + r * H;
    T* dxr = dx + r * H;
    for (int c = lane * VEC; c < H; c += WAVE_SIZE * VEC) {
      T dv[VEC], xv[VEC], o[VEC];
      *reinterpret_cast<uint4*>(dv) = *reinterpret_cast<const uint4*>(dyr + c);
      *reinterpret_cast<uint4*>(xv) = *reinterpret_cast<const uint4*>(xr + c);
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        float g = DTraits<T>::to_f32(dv[k]) *
                  gelu_bwd(DTraits<T>::to_f32(xv[k]) + bias[c + k]);
        o[k] = DTraits<T>::from_f32(g);
        lb[c + k] += g;  // lane-owned columns of the wave's slab
      }
      *reinterpret_cast<uint4*>(dxr + c) = *reinterpret_cast<const uint4*>(o);
    }
  }
  __syncthreads();
  float* slab0 = reinterpret_cast<float*>(smem_raw);
  float* pb = part_dbias + static_cast<int64_t>(blockIdx.x) * H;
  for (int c = threadIdx.x; c < H; c += blockDim.x) {
    float acc = 0.f;
#pragma unroll
    for (int w = 0; w < NW; ++w) acc += slab0[w * H + c];
    pb[c] = acc;
  }
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
  constexpr int NW = 4;
  const int rows_per_block = 16;
  const int nblocks = static_cast<int>((rows + rows_per_block - 1) / rows_per_block);
  auto opts = x.options().dtype(torch::kFloat32);
  auto part_b = torch::empty({nblocks, H}, opts);
  auto dbias = torch::zeros({H}, opts);
  auto stream = at::hip::getCurrentHIPStream();
  const size_t lds = NW * static_cast<size_t>(H) * sizeof(float);
  TORCH_CHECK(lds <= 160 * 1024, "bias_gelu_bwd: H too large");
  auto dy_c = dy.contiguous();
  DISPATCH_FLOATING2(x.scalar_type(), "bias_gelu_bwd", [&] {
    TORCH_CHECK(H % kVec == 0, "bias_gelu_bwd: H % ", kVec, " != 0");
    if (lds > 48 * 1024) {
      HIP_CHECK(hipFuncSetAttribute(
          reinterpret_cast<const void*>(&bias_gelu_bwd_kernel<scalar_t, kVec, NW>),
          hipFuncAttributeMaxDynamicSharedMemorySize, lds));
    }
    hipLaunchKernelGGL((bias_gelu_bwd_kernel<scalar_t, kVec, NW>),
                       dim3(nblocks), dim3(NW * WAVE_SIZE), lds, stream,
                       reinterpret_cast<const scalar_t*>(dy_c.data_ptr()),
                       reinterpret_cast<const scalar_t*>(x.data_ptr()),
                       bias_f.data_ptr<float>(),
                       reinterpret_cast<scalar_t*>(dx.data_ptr()),
                       part_b.data_ptr<float>(), rows, H, rows_per_block);
  });
  dim3 rgrid((H + 255) / 256, (nblocks + kColChunk - 1) / kColChunk), rblock(256);
  hipLaunchKernelGGL(col_reduce_kernel, rgrid, rblock, 0, stream,
                     part_b.data_ptr<float>(), nblocks, H,
                     dbias.data_ptr<float>());
  if (bias.scalar_type() != torch::kFloat32) {
    return {dx, dbias.to(bias.scalar_type())};
  }
  return {dx, dbias};
}

}  // namespace bpa
