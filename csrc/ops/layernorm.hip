// Fused LayerNorm forward/backward for MI355X (gfx950).
//
// Replaces Apex FusedLayerNormAffineFunction (reference call sites:
// src/modeling.py:299-335). Shapes in this framework: rows up to ~12k
// (B*S), H in {64..4096}, eps 1e-12. Memory-bound: one wave per row,
// bf16x8 (16 B/lane) vector loads, mean/var by 64-lane shuffle
// reduction; backward dgamma/dbeta use a deterministic two-stage
// partial-sum reduction (no atomics).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../common.h"

namespace bpa {

template <typename T, int VEC>
__global__ void ln_fwd_kernel(const T* __restrict__ x,
                              const float* __restrict__ gamma,
                              const float* __restrict__ beta,
                              T* __restrict__ y, float* __restrict__ mean,
                              float* __restrict__ rstd, int rows, int H,
                              float eps) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;
  const int row = blockIdx.x * (blockDim.x / WAVE_SIZE) + wave;
  if (row >= rows) return;
  const T* xr = x + static_cast<int64_t>(row) * H;
  T* yr = y + static_cast<int64_t>(row) * H;

  float sum = 0.f, sumsq = 0.f;
  for (int c = lane * VEC; c < H; c += WAVE_SIZE * VEC) {
    T v[VEC];
    *reinterpret_cast<uint4*>(v) = *reinterpret_cast<const uint4*>(xr + c);
#pragma unroll
    for (int k = 0; k < VEC; ++k) {
      float f = DTraits<T>::to_f32(v[k]);
      sum += f;
      sumsq += f * f;
    }
  }
  sum = wave_reduce_sum(sum);
  sumsq = wave_reduce_sum(sumsq);
  const float mu = sum / H;
  const float var = fmaxf(sumsq / H - mu * mu, 0.f);
  const float rs = rsqrtf(var + eps);
  if (lane == 0) {
    mean[row] = mu;
    rstd[row] = rs;
  }
  for (int c = lane * VEC; c < H; c += WAVE_SIZE * VEC) {
    T v[VEC];
    *reinterpret_cast<uint4*>(v) = *reinterpret_cast<const uint4*>(xr + c);
    T o[VEC];
#pragma unroll
    for (int k = 0; k < VEC; ++k) {
      float f = DTraits<T>::to_f32(v[k]);
      o[k] = DTraits<T>::from_f32((f - mu) * rs * gamma[c + k] + beta[c + k]);
    }
    *reinterpret_cast<uint4*>(yr + c) = *reinterpret_cast<const uint4*>(o);
  }
}

// backward: dx in one pass; dgamma/dbeta accumulate in LDS (fast LDS
// f32 atomics) per block, one global write per column per block, then a
// deterministic column-reduce kernel over the per-block partials.
template <typename T, int VEC, int NW>
__global__ void ln_bwd_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                              const float* __restrict__ gamma,
                              const float* __restrict__ mean,
                              const float* __restrict__ rstd,
                              T* __restrict__ dx,
                              float* __restrict__ part_dgamma,
                              float* __restrict__ part_dbeta, int rows, int H,
                              int rows_per_block) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  // per-wave private slabs: no atomics, no cross-wave contention
  float* lg = reinterpret_cast<float*>(smem_raw) + wave * 2 * H;  // [NW][2H]
  float* lb = lg + H;
  for (int c = lane; c < 2 * H; c += WAVE_SIZE) lg[c] = 0.f;

  const int row0 = blockIdx.x * rows_per_block;
  const int row_end = min(row0 + rows_per_block, rows);
  for (int r = row0 + wave; r < row_end; r += NW) {
    const T* dyr = dy + static_cast<int64_t>(r) * H;
    const T* xr = x + static_cast<int64_t>(r) * H;
    T* dxr = dx + static_cast<int64_t>(r) * H;
    const float mu = mean[r], rs = rstd[r];

    float s1 = 0.f, s2 = 0.f;
    for (int c = lane * VEC; c < H; c += WAVE_SIZE * VEC) {
      T dv[VEC], xv[VEC];
      *reinterpret_cast<uint4*>(dv) = *reinterpret_cast<const uint4*>(dyr + c);
      *reinterpret_cast<uint4*>(xv) = *reinterpret_cast<const uint4*>(xr + c);
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        float d = DTraits<T>::to_f32(dv[k]);
        float xh = (DTraits<T>::to_f32(xv[k]) - mu) * rs;
        float dw = d * gamma[c + k];
        s1 += dw * xh;
        s2 += dw;
      }
    }
    s1 = wave_reduce_sum(s1) / H;
    s2 = wave_reduce_sum(s2) / H;

    for (int c = lane * VEC; c < H; c += WAVE_SIZE * VEC) {
      T dv[VEC], xv[VEC], o[VEC];
      *reinterpret_cast<uint4*>(dv) = *reinterpret_cast<const uint4*>(dyr + c);
      *reinterpret_cast<uint4*>(xv) = *reinterpret_cast<const uint4*>(xr + c);
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        float d = DTraits<T>::to_f32(dv[k]);
        float xh = (DTraits<T>::to_f32(xv[k]) - mu) * rs;
        float dw = d * gamma[c + k];
        o[k] = DTraits<T>::from_f32(rs * (dw - s2 - xh * s1));
        lg[c + k] += d * xh;  // this lane owns these columns in its slab
        lb[c + k] += d;
      }
      *reinterpret_cast<uint4*>(dxr + c) = *reinterpret_cast<const uint4*>(o);
    }
  }
  __syncthreads();
  float* slab0 = reinterpret_cast<float*>(smem_raw);
  float* pg = part_dgamma + static_cast<int64_t>(blockIdx.x) * H;
  float* pb = part_dbeta + static_cast<int64_t>(blockIdx.x) * H;
  for (int c = threadIdx.x; c < H; c += blockDim.x) {
    float ag = 0.f, ab = 0.f;
#pragma unroll
    for (int w = 0; w < NW; ++w) {
      ag += slab0[w * 2 * H + c];
      ab += slab0[w * 2 * H + H + c];
    }
    pg[c] = ag;
    pb[c] = ab;
  }
}

// grid (ceil(H/256), ceil(nparts/kColChunk)); out must be zero-filled.
__global__ void col_reduce_kernel(const float* __restrict__ parts, int nparts,
                                  int H, float* __restrict__ out) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= H) return;
  const int p0 = blockIdx.y * kColChunk;
  const int p1 = min(p0 + kColChunk, nparts);
  float acc = 0.f;
  for (int p = p0; p < p1; ++p)
    acc += parts[static_cast<int64_t>(p) * H + c];
  atomicAdd(&out[c], acc);
}

#define DISPATCH_FLOATING(TYPE, NAME, ...)                                   \
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

std::vector<torch::Tensor> ln_fwd(torch::Tensor x, torch::Tensor gamma,
                                  torch::Tensor beta, double eps) {
  TORCH_CHECK(x.dim() == 2 && x.is_contiguous(), "ln_fwd: x must be 2D contiguous");
  const int rows = x.size(0), H = x.size(1);
  auto gamma_f = gamma.contiguous().to(torch::kFloat32);
  auto beta_f = beta.contiguous().to(torch::kFloat32);
  auto y = torch::empty_like(x);
  auto opts = x.options().dtype(torch::kFloat32);
  auto mean = torch::empty({rows}, opts);
  auto rstd = torch::empty({rows}, opts);
  auto stream = at::hip::getCurrentHIPStream();
  constexpr int WAVES = 4;
  dim3 grid((rows + WAVES - 1) / WAVES), block(WAVES * WAVE_SIZE);
  DISPATCH_FLOATING(x.scalar_type(), "ln_fwd", [&] {
    TORCH_CHECK(H % kVec == 0, "ln_fwd: H must be a multiple of ", kVec);
    hipLaunchKernelGGL((ln_fwd_kernel<scalar_t, kVec>), grid, block, 0, stream,
                       reinterpret_cast<const scalar_t*>(x.data_ptr()),
                       gamma_f.data_ptr<float>(), beta_f.data_ptr<float>(),
                       reinterpret_cast<scalar_t*>(y.data_ptr()),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(), rows, H,
                       static_cast<float>(eps));
  });
  return {y, mean, rstd};
}

std::vector<torch::Tensor> ln_bwd(torch::Tensor dy, torch::Tensor x,
                                  torch::Tensor gamma, torch::Tensor mean,
                                  torch::Tensor rstd) {
  TORCH_CHECK(dy.sizes() == x.sizes(), "ln_bwd: shape mismatch");
  const int rows = x.size(0), H = x.size(1);
  auto gamma_f = gamma.contiguous().to(torch::kFloat32);
  auto dx = torch::empty_like(x);
  constexpr int NW = 4;                    // waves per block
  const int rows_per_block = 16;           // 4 rows per wave
  const int nblocks = (rows + rows_per_block - 1) / rows_per_block;
  auto opts = x.options().dtype(torch::kFloat32);
  auto part_g = torch::empty({nblocks, H}, opts);
  auto part_b = torch::empty({nblocks, H}, opts);
  auto dgamma = torch::zeros({H}, opts);
  auto dbeta = torch::zeros({H}, opts);
  auto stream = at::hip::getCurrentHIPStream();
  dim3 grid(nblocks), block(NW * WAVE_SIZE);
  const size_t lds = NW * 2 * static_cast<size_t>(H) * sizeof(float);
  TORCH_CHECK(lds <= 160 * 1024, "ln_bwd: H too large for LDS accumulation");
  auto dy_c = dy.contiguous();
  DISPATCH_FLOATING(x.scalar_type(), "ln_bwd", [&] {
    TORCH_CHECK(H % kVec == 0, "ln_bwd: H must be a multiple of ", kVec);
    hipLaunchKernelGGL((ln_bwd_kernel<scalar_t, kVec, NW>), grid,
                       block, lds, stream,
                       reinterpret_cast<const scalar_t*>(dy_c.data_ptr()),
                       reinterpret_cast<const scalar_t*>(x.data_ptr()),
                       gamma_f.data_ptr<float>(), mean.data_ptr<float>(),
                       rstd.data_ptr<float>(),
                       reinterpret_cast<scalar_t*>(dx.data_ptr()),
                       part_g.data_ptr<float>(), part_b.data_ptr<float>(), rows,
                       H, rows_per_block);
  });
  dim3 rgrid((H + 255) / 256, (nblocks + kColChunk - 1) / kColChunk), rblock(256);
  hipLaunchKernelGGL(col_reduce_kernel, rgrid, rblock, 0, stream,
                     part_g.data_ptr<float>(), nblocks, H,
                     dgamma.data_ptr<float>());
  hipLaunchKernelGGL(col_reduce_kernel, rgrid, rblock, 0, stream,
                     part_b.data_ptr<float>(), nblocks, H,
                     dbeta.data_ptr<float>());
  if (gamma.scalar_type() != torch::kFloat32) {
    return {dx, dgamma.to(gamma.scalar_type()), dbeta.to(gamma.scalar_type())};
  }
  return {dx, dgamma, dbeta};
}

}  // namespace bpa
