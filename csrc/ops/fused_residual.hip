// Fused bias + dropout + residual-add + LayerNorm for MI355X (gfx950).
//
// One kernel for the residual junction the reference runs as 4 ops
// (BertSelfOutput/BertOutput: src/modeling.py:432-443, 468-479):
//   z = dropout(x + bias) + residual ; y = LN(z) * gamma + beta
// Dropout uses Philox4x32-10 (seed, offset) and stores the keep-mask as
// bytes so backward is exact. One wave per row, 16 B/lane vector IO.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../common.h"

namespace bpa {

__global__ void col_reduce_kernel(const float* __restrict__ parts, int nparts,
                                  int H, float* __restrict__ out);

template <typename T, int VEC, bool HAS_BIAS, bool TRAIN_DROP>
__global__ void bdrl_fwd_kernel(
    const T* __restrict__ x, const float* __restrict__ bias,
    const T* __restrict__ residual, const float* __restrict__ gamma,
    const float* __restrict__ beta, T* __restrict__ y, T* __restrict__ z,
    uint8_t* __restrict__ mask, float* __restrict__ mean,
    float* __restrict__ rstd, int rows, int H, float p, float eps,
    uint64_t seed, uint64_t offset) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;
  const int row = blockIdx.x * (blockDim.x / WAVE_SIZE) + wave;
  if (row >= rows) return;
  const int64_t base = static_cast<int64_t>(row) * H;
  const float keep_scale = TRAIN_DROP ? 1.0f / (1.0f - p) : 1.0f;
  Philox philox(seed);

  float sum = 0.f, sumsq = 0.f;
  for (int c = lane * VEC; c < H; c += WAVE_SIZE * VEC) {
    T xv[VEC], rv[VEC], zv[VEC];
    *reinterpret_cast<uint4*>(xv) = *reinterpret_cast<const uint4*>(x + base + c);
    *reinterpret_cast<uint4*>(rv) =
        *reinterpret_cast<const uint4*>(residual + base + c);
    uint8_t mv[VEC];
    if (TRAIN_DROP) {
      // 4 uniforms per philox call; VEC=8 -> 2 calls, VEC=4 -> 1 call
#pragma unroll
      for (int q = 0; q < VEC / 4; ++q) {
        uint32_t r4[4];
        philox(offset + (base + c) / 4 + q, r4);
#pragma unroll
        for (int j = 0; j < 4; ++j)
          mv[q * 4 + j] = u32_to_uniform(r4[j]) >= p ? 1 : 0;
      }
    }
#pragma unroll
    for (int k = 0; k < VEC; ++k) {
      float t = DTraits<T>::to_f32(xv[k]);
      if (HAS_BIAS) t += bias[c + k];
      if (TRAIN_DROP) t = mv[k] ? t * keep_scale : 0.f;
      t += DTraits<T>::to_f32(rv[k]);
      zv[k] = DTraits<T>::from_f32(t);
      sum += t;
      sumsq += t * t;
    }
    *reinterpret_cast<uint4*>(z + base + c) = *reinterpret_cast<const uint4*>(zv);
    if (TRAIN_DROP) {
      if (VEC == 8)
        *reinterpret_cast<uint2*>(mask + base + c) =
            *reinterpret_cast<const uint2*>(mv);
      else
        *reinterpret_cast<uint32_t*>(mask + base + c) =
            *reinterpret_cast<const uint32_t*>(mv);
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
    T zv[VEC], ov[VEC];
    *reinterpret_cast<uint4*>(zv) = *reinterpret_cast<const uint4*>(z + base + c);
#pragma unroll
    for (int k = 0; k < VEC; ++k) {
      float t = DTraits<T>::to_f32(zv[k]);
      ov[k] = DTraits<T>::from_f32((t - mu) * rs * gamma[c + k] + beta[c + k]);
    }
    *reinterpret_cast<uint4*>(y + base + c) = *reinterpret_cast<const uint4*>(ov);
  }
}

template <typename T, int VEC, int NW, bool HAS_BIAS, bool TRAIN_DROP>
__global__ void bdrl_bwd_kernel(
    const T* __restrict__ dy, const T* __restrict__ z,
    const uint8_t* __restrict__ mask, const float* __restrict__ gamma,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    T* __restrict__ dx, T* __restrict__ dz_res,
    float* __restrict__ part_dgamma, float* __restrict__ part_dbeta,
    float* __restrict__ part_dbias, int rows, int H, float p,
    int rows_per_block) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  const int n_slabs = HAS_BIAS ? 3 : 2;
  // per-wave private slabs [NW][n_slabs][H]: no atomics, no contention
  float* lg = reinterpret_cast<float*>(smem_raw) + wave * n_slabs * H;
  float* lb = lg + H;
  float* lbias = lb + H;
  for (int c = lane; c < n_slabs * H; c += WAVE_SIZE) lg[c] = 0.f;
  const float keep_scale = TRAIN_DROP ? 1.0f / (1.0f - p) : 1.0f;

  const int row0 = blockIdx.x * rows_per_block;
  const int row_end = min(row0 + rows_per_block, rows);
  for (int r = row0 + wave; r < row_end; r += NW) {
    const int64_t base = static_cast<int64_t>(r) * H;
    const float mu = mean[r], rs = rstd[r];
    float s1 = 0.f, s2 = 0.f;
    for (int c = lane * VEC; c < H; c += WAVE_SIZE * VEC) {
      T dv[VEC], zv[VEC];
      *reinterpret_cast<uint4*>(dv) = *reinterpret_cast<const uint4*>(dy + base + c);
      *reinterpret_cast<uint4*>(zv) = *reinterpret_cast<const uint4*>(z + base + c);
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        float d = DTraits<T>::to_f32(dv[k]);
        float zh = (DTraits<T>::to_f32(zv[k]) - mu) * rs;
        float dw = d * gamma[c + k];
        s1 += dw * zh;
        s2 += dw;
        lg[c + k] += d * zh;
        lb[c + k] += d;
      }
    }
    s1 = wave_reduce_sum(s1) / H;
    s2 = wave_reduce_sum(s2) / H;

    for (int c = lane * VEC; c < H; c += WAVE_SIZE * VEC) {
      T dv[VEC], zv[VEC], dzo[VEC], dxo[VEC];
      *reinterpret_cast<uint4*>(dv) = *reinterpret_cast<const uint4*>(dy + base + c);
      *reinterpret_cast<uint4*>(zv) = *reinterpret_cast<const uint4*>(z + base + c);
      uint8_t mv[VEC];
      if (TRAIN_DROP) {
        if (VEC == 8)
          *reinterpret_cast<uint2*>(mv) =
              *reinterpret_cast<const uint2*>(mask + base + c);
        else
          *reinterpret_cast<uint32_t*>(mv) =
              *reinterpret_cast<const uint32_t*>(mask + base + c);
      }
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        float d = DTraits<T>::to_f32(dv[k]);
        float zh = (DTraits<T>::to_f32(zv[k]) - mu) * rs;
        float dzk = rs * (d * gamma[c + k] - s2 - zh * s1);
        dzo[k] = DTraits<T>::from_f32(dzk);  // grad to residual input
        float dxk = TRAIN_DROP ? (mv[k] ? dzk * keep_scale : 0.f) : dzk;
        dxo[k] = DTraits<T>::from_f32(dxk);
        if (HAS_BIAS) lbias[c + k] += dxk;
      }
      *reinterpret_cast<uint4*>(dz_res + base + c) =
          *reinterpret_cast<const uint4*>(dzo);
      *reinterpret_cast<uint4*>(dx + base + c) =
          *reinterpret_cast<const uint4*>(dxo);
    }
  }
  __syncthreads();
  float* slab0 = reinterpret_cast<float*>(smem_raw);
  for (int c = threadIdx.x; c < H; c += blockDim.x) {
    float ag = 0.f, ab = 0.f, abias = 0.f;
#pragma unroll
    for (int w = 0; w < NW; ++w) {
      ag += slab0[w * n_slabs * H + c];
      ab += slab0[w * n_slabs * H + H + c];
      if (HAS_BIAS) abias += slab0[w * n_slabs * H + 2 * H + c];
    }
    part_dgamma[static_cast<int64_t>(blockIdx.x) * H + c] = ag;
    part_dbeta[static_cast<int64_t>(blockIdx.x) * H + c] = ab;
    if (HAS_BIAS)
      part_dbias[static_cast<int64_t>(blockIdx.x) * H + c] = abias;
  }
}

#define DISPATCH_T(TYPE, NAME, ...)                                          \
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

std::vector<torch::Tensor> bias_dropout_residual_ln_fwd(
    torch::Tensor x, c10::optional<torch::Tensor> bias, torch::Tensor residual,
    torch::Tensor gamma, torch::Tensor beta, double p, double eps,
    int64_t seed, int64_t offset) {
  TORCH_CHECK(x.dim() == 2 && x.is_contiguous(), "bdrl_fwd: bad x");
  const int rows = x.size(0), H = x.size(1);
  auto res_c = residual.contiguous();
  TORCH_CHECK(res_c.scalar_type() == x.scalar_type(),
              "bdrl_fwd: residual dtype must match x");
  auto gamma_f = gamma.contiguous().to(torch::kFloat32);
  auto beta_f = beta.contiguous().to(torch::kFloat32);
  torch::Tensor bias_f;
  const bool has_bias = bias.has_value();
  if (has_bias) bias_f = bias->contiguous().to(torch::kFloat32);
  const bool train_drop = p > 0.0;

  auto y = torch::empty_like(x);
  auto z = torch::empty_like(x);
  auto fopts = x.options().dtype(torch::kFloat32);
  auto mask = train_drop
                  ? torch::empty({rows, H}, x.options().dtype(torch::kUInt8))
                  : torch::empty({0}, x.options().dtype(torch::kUInt8));
  auto mean = torch::empty({rows}, fopts);
  auto rstd = torch::empty({rows}, fopts);
  auto stream = at::hip::getCurrentHIPStream();
  constexpr int WAVES = 4;
  dim3 grid((rows + WAVES - 1) / WAVES), block(WAVES * WAVE_SIZE);
  DISPATCH_T(x.scalar_type(), "bdrl_fwd", [&] {
    TORCH_CHECK(H % kVec == 0, "bdrl_fwd: H % ", kVec, " != 0");
    auto launch = [&](auto has_bias_c, auto train_c) {
      hipLaunchKernelGGL(
          (bdrl_fwd_kernel<scalar_t, kVec, decltype(has_bias_c)::value,
                           decltype(train_c)::value>),
          grid, block, 0, stream,
          reinterpret_cast<const scalar_t*>(x.data_ptr()),
          has_bias ? bias_f.data_ptr<float>() : nullptr,
          reinterpret_cast<const scalar_t*>(res_c.data_ptr()),
          gamma_f.data_ptr<float>(), beta_f.data_ptr<float>(),
          reinterpret_cast<scalar_t*>(y.data_ptr()),
          reinterpret_cast<scalar_t*>(z.data_ptr()),
          train_drop ? mask.data_ptr<uint8_t>() : nullptr,
          mean.data_ptr<float>(), rstd.data_ptr<float>(), rows, H,
          static_cast<float>(p), static_cast<float>(eps),
          static_cast<uint64_t>(seed), static_cast<uint64_t>(offset));
    };
    if (has_bias && train_drop)
      launch(std::true_type{}, std::true_type{});
    else if (has_bias)
      launch(std::true_type{}, std::false_type{});
    else if (train_drop)
      launch(std::false_type{}, std::true_type{});
    else
      launch(std::false_type{}, std::false_type{});
  });
  return {y, z, mask, mean, rstd};
}

std::vector<torch::Tensor> bias_dropout_residual_ln_bwd(
    torch::Tensor dy, torch::Tensor z, torch::Tensor mask, torch::Tensor gamma,
    torch::Tensor mean, torch::Tensor rstd, double p, bool has_bias) {
  const int rows = z.size(0), H = z.size(1);
  auto gamma_f = gamma.contiguous().to(torch::kFloat32);
  auto dy_c = dy.contiguous();
  auto dx = torch::empty_like(z);
  auto dz_res = torch::empty_like(z);
  constexpr int NW = 4;
  const int rows_per_block = 16;
  const int nblocks = (rows + rows_per_block - 1) / rows_per_block;
  auto fopts = z.options().dtype(torch::kFloat32);
  auto part_g = torch::empty({nblocks, H}, fopts);
  auto part_b = torch::empty({nblocks, H}, fopts);
  auto part_bias =
      has_bias ? torch::empty({nblocks, H}, fopts) : torch::empty({0}, fopts);
  const bool train_drop = p > 0.0;
  auto stream = at::hip::getCurrentHIPStream();
  const size_t lds =
      NW * (has_bias ? 3 : 2) * static_cast<size_t>(H) * sizeof(float);
  TORCH_CHECK(lds <= 160 * 1024, "bdrl_bwd: H too large");
  DISPATCH_T(z.scalar_type(), "bdrl_bwd", [&] {
    TORCH_CHECK(H % kVec == 0, "bdrl_bwd: H % ", kVec, " != 0");
    auto launch = [&](auto has_bias_c, auto train_c) {
      if (lds > 48 * 1024) {
        HIP_CHECK(hipFuncSetAttribute(
            reinterpret_cast<const void*>(
                &bdrl_bwd_kernel<scalar_t, kVec, NW,
                                 decltype(has_bias_c)::value,
                                 decltype(train_c)::value>),
            hipFuncAttributeMaxDynamicSharedMemorySize, lds));
      }
      hipLaunchKernelGGL(
          (bdrl_bwd_kernel<scalar_t, kVec, NW, decltype(has_bias_c)::value,
                           decltype(train_c)::value>),
          dim3(nblocks), dim3(NW * WAVE_SIZE), lds, stream,
          reinterpret_cast<const scalar_t*>(dy_c.data_ptr()),
          reinterpret_cast<const scalar_t*>(z.data_ptr()),
          train_drop ? mask.data_ptr<uint8_t>() : nullptr,
          gamma_f.data_ptr<float>(), mean.data_ptr<float>(),
          rstd.data_ptr<float>(),
          reinterpret_cast<scalar_t*>(dx.data_ptr()),
          reinterpret_cast<scalar_t*>(dz_res.data_ptr()),
          part_g.data_ptr<float>(), part_b.data_ptr<float>(),
          has_bias ? part_bias.data_ptr<float>() : nullptr, rows, H,
          static_cast<float>(p), rows_per_block);
    };
    if (has_bias && train_drop)
      launch(std::true_type{}, std::true_type{});
    else if (has_bias)
      launch(std::true_type{}, std::false_type{});
    else if (train_drop)
      launch(std::false_type{}, std::true_type{});
    else
      launch(std::false_type{}, std::false_type{});
  });
  auto dgamma = torch::zeros({H}, fopts);
  auto dbeta = torch::zeros({H}, fopts);
  auto dbias = has_bias ? torch::zeros({H}, fopts) : torch::empty({0}, fopts);
  dim3 rgrid((H + 255) / 256, (nblocks + kColChunk - 1) / kColChunk), rblock(256);
  hipLaunchKernelGGL(col_reduce_kernel, rgrid, rblock, 0, stream,
                     part_g.data_ptr<float>(), nblocks, H,
                     dgamma.data_ptr<float>());
  hipLaunchKernelGGL(col_reduce_kernel, rgrid, rblock, 0, stream,
                     part_b.data_ptr<float>(), nblocks, H,
                     dbeta.data_ptr<float>());
  if (has_bias) {
    hipLaunchKernelGGL(col_reduce_kernel, rgrid, rblock, 0, stream,
                       part_bias.data_ptr<float>(), nblocks, H,
                       dbias.data_ptr<float>());
  }
  if (gamma.scalar_type() != torch::kFloat32) {
    dgamma = dgamma.to(gamma.scalar_type());
    dbeta = dbeta.to(gamma.scalar_type());
    if (has_bias) dbias = dbias.to(gamma.scalar_type());
  }
  return {dx, dbias, dz_res, dgamma, dbeta};
}

}  // namespace bpa
