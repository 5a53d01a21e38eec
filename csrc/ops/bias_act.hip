// Fused bias + exact-erf GELU forward/backward for MI355X (gfx950).
//
// The bias+activation fusion point of the reference's LinearActivation
// (src/modeling.py:141-185, jit bias_gelu at :126-139). Elementwise,
// memory-bound: 2D grid (column-block, row) with no per-element integer
// division, 16 B/lane vector IO, enough resident waves to hide HBM3E
// latency (grid is NOT capped — [12288,4096] launches ~24k workgroups
// across the 256 CUs / 8 XCDs).
//
// Backward fuses the dbias column reduction into the dx pass: each wave
// keeps a per-lane fp32 accumulator over its row slice (registers, no
// LDS traffic in the hot loop), waves combine through an 8 KiB LDS slab,
// blocks write [grid.y, H] partials reduced by col_reduce_kernel. This
// removes the old second full read of dx (100 MB at phase-1 FFN shapes).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../common.h"

namespace bpa {

torch::Tensor col_reduce_full(torch::Tensor parts);  // layernorm.hip

template <typename T, int VEC>
__global__ void bias_gelu_fwd_kernel(const T* __restrict__ x,
                                     const float* __restrict__ bias,
                                     T* __restrict__ y, int rows, int H) {
  const int vcol = blockIdx.x * blockDim.x + threadIdx.x;
  const int c = vcol * VEC;
  if (c >= H) return;
  float bv[VEC];
#pragma unroll
  for (int k = 0; k < VEC; ++k) bv[k] = bias[c + k];
  for (int r = blockIdx.y; r < rows; r += gridDim.y) {
    const int64_t e = static_cast<int64_t>(r) * H + c;
    T v[VEC], o[VEC];
    *reinterpret_cast<uint4*>(v) = *reinterpret_cast<const uint4*>(x + e);
#pragma unroll
    for (int k = 0; k < VEC; ++k) {
      o[k] = DTraits<T>::from_f32(gelu_fwd(DTraits<T>::to_f32(v[k]) + bv[k]));
    }
    *reinterpret_cast<uint4*>(y + e) = *reinterpret_cast<const uint4*>(o);
  }
}

// Backward: dx = dy * gelu'(x + bias); per-column dbias partials folded in.
// Block: NW waves; wave w covers rows {r0+w, r0+w+NW, ...} of a
// ROWS_PER_BLOCK row stripe and a 256*VEC-wide column window; per-lane
// fp32 accumulators combine across waves through LDS at stripe end.
template <typename T, int VEC, int NW>
__global__ void bias_gelu_bwd_kernel(const T* __restrict__ dy,
                                     const T* __restrict__ x,
                                     const float* __restrict__ bias,
                                     T* __restrict__ dx,
                                     float* __restrict__ part_dbias, int rows,
                                     int H, int rows_per_block) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;
  // every wave covers the same 64-lane-wide column window (so the LDS
  // combine lines up lane-for-lane) and a disjoint subset of the rows
  const int c = (blockIdx.x * WAVE_SIZE + lane) * VEC;

  float acc[VEC];
#pragma unroll
  for (int k = 0; k < VEC; ++k) acc[k] = 0.f;
  float bv[VEC];
  if (c < H) {
#pragma unroll
    for (int k = 0; k < VEC; ++k) bv[k] = bias[c + k];
  }

  const int r0 = blockIdx.y * rows_per_block;
  const int r1 = min(r0 + rows_per_block, rows);
  if (c < H) {
    for (int r = r0 + wave; r < r1; r += NW) {
      const int64_t e = static_cast<int64_t>(r) * H + c;
      T dv[VEC], xv[VEC], o[VEC];
      *reinterpret_cast<uint4*>(dv) = *reinterpret_cast<const uint4*>(dy + e);
      *reinterpret_cast<uint4*>(xv) = *reinterpret_cast<const uint4*>(x + e);
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        const float g = DTraits<T>::to_f32(dv[k]) *
                        gelu_bwd(DTraits<T>::to_f32(xv[k]) + bv[k]);
        o[k] = DTraits<T>::from_f32(g);
        acc[k] += g;
      }
      *reinterpret_cast<uint4*>(dx + e) = *reinterpret_cast<const uint4*>(o);
    }
  }

  // cross-wave combine: [NW][64*VEC] fp32 slab (NW=4, VEC=8 -> 8 KiB)
  extern __shared__ __attribute__((aligned(16))) float slab[];
  float* mine = slab + (wave * WAVE_SIZE + lane) * VEC;
#pragma unroll
  for (int k = 0; k < VEC; ++k) mine[k] = acc[k];
  __syncthreads();
  if (wave == 0 && c < H) {
    float tot[VEC];
#pragma unroll
    for (int k = 0; k < VEC; ++k) tot[k] = 0.f;
#pragma unroll
    for (int w = 0; w < NW; ++w) {
      const float* src = slab + (w * WAVE_SIZE + lane) * VEC;
#pragma unroll
      for (int k = 0; k < VEC; ++k) tot[k] += src[k];
    }
    float* dst = part_dbias + static_cast<int64_t>(blockIdx.y) * H + c;
#pragma unroll
    for (int k = 0; k < VEC; ++k) dst[k] = tot[k];
  }
}

// Generic column sum [rows, H] -> fp32 [H] (two-stage, deterministic).
// Used for the packed-QKV bias gradient (replacing ATen's bf16 reduce
// at ~2.5 TB/s with a streaming partial pass + col_reduce).
template <typename T, int VEC>
__global__ void col_sum_partial_kernel(const T* __restrict__ x,
                                       float* __restrict__ part, int rows,
                                       int H, int rows_per_chunk) {
  const int c = (blockIdx.x * blockDim.x + threadIdx.x) * VEC;
  if (c >= H) return;
  const int r0 = blockIdx.y * rows_per_chunk;
  const int r1 = min(r0 + rows_per_chunk, rows);
  float acc[VEC] = {};
  for (int r = r0; r < r1; ++r) {
    T v[VEC];
    *reinterpret_cast<uint4*>(v) =
        *reinterpret_cast<const uint4*>(x + static_cast<int64_t>(r) * H + c);
#pragma unroll
    for (int k = 0; k < VEC; ++k) acc[k] += DTraits<T>::to_f32(v[k]);
  }
  float* p = part + static_cast<int64_t>(blockIdx.y) * H + c;
#pragma unroll
  for (int k = 0; k < VEC; ++k) p[k] = acc[k];
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
    const int vec_per_row = H / kVec;
    const int threads = tmin(256, ((vec_per_row + 63) / 64) * 64);
    dim3 grid((vec_per_row + threads - 1) / threads,
              static_cast<unsigned>(tmin<int64_t>(rows, 65535)));
    hipLaunchKernelGGL((bias_gelu_fwd_kernel<scalar_t, kVec>), grid,
                       dim3(threads), 0, stream,
                       reinterpret_cast<const scalar_t*>(x.data_ptr()),
                       bias_f.data_ptr<float>(),
                       reinterpret_cast<scalar_t*>(y.data_ptr()),
                       static_cast<int>(rows), H);
  });
  return y;
}

std::vector<torch::Tensor> bias_gelu_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor bias) {
  const int64_t rows = x.size(0);
  const int H = x.size(1);
  auto bias_f = bias.contiguous().to(torch::kFloat32);
  auto dx = torch::empty_like(x);
  auto fopts = x.options().dtype(torch::kFloat32);
  auto stream = at::hip::getCurrentHIPStream();
  constexpr int NW = 4;
  const int rows_per_block = 16;
  const int stripe_count =
      static_cast<int>((rows + rows_per_block - 1) / rows_per_block);
  torch::Tensor dbias;
  auto dy_c = dy.contiguous();
  DISPATCH_FLOATING2(x.scalar_type(), "bias_gelu_bwd", [&] {
    TORCH_CHECK(H % kVec == 0, "bias_gelu_bwd: H % ", kVec, " != 0");
    auto part = torch::empty({stripe_count, H}, fopts);
    const int lanes_per_row = WAVE_SIZE;  // one wave per row slice
    dim3 grid((H / kVec + lanes_per_row - 1) / lanes_per_row, stripe_count);
    const size_t lds = NW * WAVE_SIZE * kVec * sizeof(float);
    hipLaunchKernelGGL((bias_gelu_bwd_kernel<scalar_t, kVec, NW>), grid,
                       dim3(NW * WAVE_SIZE), lds, stream,
                       reinterpret_cast<const scalar_t*>(dy_c.data_ptr()),
                       reinterpret_cast<const scalar_t*>(x.data_ptr()),
                       bias_f.data_ptr<float>(),
                       reinterpret_cast<scalar_t*>(dx.data_ptr()),
                       part.data_ptr<float>(), static_cast<int>(rows), H,
                       rows_per_block);
    dbias = col_reduce_full(part);
  });
  if (bias.scalar_type() != torch::kFloat32) {
    return {dx, dbias.to(bias.scalar_type())};
  }
  return {dx, dbias};
}

torch::Tensor col_sum(torch::Tensor x) {
  TORCH_CHECK(x.dim() == 2 && x.is_contiguous(), "col_sum: bad x");
  const int rows = x.size(0);
  const int H = x.size(1);
  const int rows_per_chunk = 32;
  const int n_chunks = (rows + rows_per_chunk - 1) / rows_per_chunk;
  auto fopts = x.options().dtype(torch::kFloat32);
  torch::Tensor out;
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_FLOATING2(x.scalar_type(), "col_sum", [&] {
    TORCH_CHECK(H % kVec == 0, "col_sum: H % ", kVec, " != 0");
    auto part = torch::empty({n_chunks, H}, fopts);
    const int slices = H / kVec;
    const int threads = tmin(256, ((slices + 63) / 64) * 64);
    dim3 grid((slices + threads - 1) / threads, n_chunks);
    hipLaunchKernelGGL((col_sum_partial_kernel<scalar_t, kVec>), grid,
                       dim3(threads), 0, stream,
                       reinterpret_cast<const scalar_t*>(x.data_ptr()),
                       part.data_ptr<float>(), rows, H, rows_per_chunk);
    out = col_reduce_full(part);
  });
  return out;
}

}  // namespace bpa
