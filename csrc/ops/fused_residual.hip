// Fused bias + dropout + residual-add + LayerNorm for MI355X (gfx950).
//
// One kernel for the residual junction the reference runs as 4 ops
// (BertSelfOutput/BertOutput: src/modeling.py:432-443, 468-479):
//   z = dropout(x + bias) + residual ; y = LN(z) * gamma + beta
// Dropout uses Philox4x32-10 (seed, offset) keyed by element index, and
// stores the keep-mask as bytes so backward is exact.
//
// Memory-bound design (same as layernorm.hip):
// * forward: one BLOCK per row; x/residual loaded once into registers,
//   z kept in registers through the mean/var block reduction, y written
//   from registers — no re-read of z after the reduction.
// * backward: one WAVE per row stripe, dy/z register-resident across
//   both passes, dgamma/dbeta/dbias accumulated in per-lane VGPRs
//   across the stripe (no per-element LDS read-modify-write), one
//   [n_waves, 3H] partial buffer reduced by a single col_reduce launch.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../common.h"

namespace bpa {

torch::Tensor col_reduce_full(torch::Tensor parts);  // layernorm.hip

template <typename T, int VEC, int ITEMS, bool HAS_BIAS, bool TRAIN_DROP>
__global__ void bdrl_fwd_kernel(
    const T* __restrict__ x, const float* __restrict__ bias,
    const T* __restrict__ residual, const float* __restrict__ gamma,
    const float* __restrict__ beta, T* __restrict__ y, T* __restrict__ z,
    uint8_t* __restrict__ mask, float* __restrict__ mean,
    float* __restrict__ rstd, int rows, int H, float p, float eps,
    uint64_t seed, uint64_t offset) {
  __shared__ float red[32];
  const int row = blockIdx.x;
  if (row >= rows) return;
  const int64_t base = static_cast<int64_t>(row) * H;
  const float keep_scale = TRAIN_DROP ? 1.0f / (1.0f - p) : 1.0f;
  Philox philox(seed);

  float zv[ITEMS][VEC];
  int cols[ITEMS];
  float sum = 0.f, sumsq = 0.f;
#pragma unroll
  for (int i = 0; i < ITEMS; ++i) {
    const int c = (i * blockDim.x + threadIdx.x) * VEC;
    cols[i] = c;
    if (c < H) {
      T xv[VEC], rv[VEC], zt[VEC];
      *reinterpret_cast<uint4*>(xv) = *reinterpret_cast<const uint4*>(x + base + c);
      *reinterpret_cast<uint4*>(rv) =
          *reinterpret_cast<const uint4*>(residual + base + c);
      uint8_t mv[VEC];
      if (TRAIN_DROP) {
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
        zv[i][k] = t;
        zt[k] = DTraits<T>::from_f32(t);
        sum += t;
        sumsq += t * t;
      }
      *reinterpret_cast<uint4*>(z + base + c) = *reinterpret_cast<const uint4*>(zt);
      if (TRAIN_DROP) {
        if (VEC == 8)
          *reinterpret_cast<uint2*>(mask + base + c) =
              *reinterpret_cast<const uint2*>(mv);
        else
          *reinterpret_cast<uint32_t*>(mask + base + c) =
              *reinterpret_cast<const uint32_t*>(mv);
      }
    }
  }
  block_reduce_sum2(sum, sumsq, red);
  const float mu = sum / H;
  const float var = fmaxf(sumsq / H - mu * mu, 0.f);
  const float rs = rsqrtf(var + eps);
  if (threadIdx.x == 0) {
    mean[row] = mu;
    rstd[row] = rs;
  }
#pragma unroll
  for (int i = 0; i < ITEMS; ++i) {
    const int c = cols[i];
    if (c < H) {
      T ov[VEC];
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        ov[k] = DTraits<T>::from_f32((zv[i][k] - mu) * rs * gamma[c + k] +
                                     beta[c + k]);
      }
      *reinterpret_cast<uint4*>(y + base + c) = *reinterpret_cast<const uint4*>(ov);
    }
  }
}

// backward dx/dz: one BLOCK per row (mirror of bdrl_fwd_kernel); dy/z/
// mask loaded once into registers, s1/s2 via block reduce, dz_res and
// dx written from registers. dgamma/dbeta/dbias partials come from the
// shared streaming col_stats_kernel (layernorm.hip) reading dy, z and
// the dx this kernel wrote.
template <typename T, int VEC, int ITEMS, bool TRAIN_DROP>
__global__ void bdrl_bwd_dx_kernel(
    const T* __restrict__ dy, const T* __restrict__ z,
    const uint8_t* __restrict__ mask, const float* __restrict__ gamma,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    T* __restrict__ dx, T* __restrict__ dz_res, int rows, int H, float p) {
  __shared__ float red[32];
  const int row = blockIdx.x;
  if (row >= rows) return;
  const int64_t base = static_cast<int64_t>(row) * H;
  const float mu = mean[row], rs = rstd[row];
  const float keep_scale = TRAIN_DROP ? 1.0f / (1.0f - p) : 1.0f;

  float dw[ITEMS][VEC], zh[ITEMS][VEC];
  uint8_t mv[ITEMS][VEC];
  int cols[ITEMS];
  float s1 = 0.f, s2 = 0.f;
#pragma unroll
  for (int i = 0; i < ITEMS; ++i) {
    const int c = (i * blockDim.x + threadIdx.x) * VEC;
    cols[i] = c;
    if (c < H) {
      T dt[VEC], zt[VEC];
      *reinterpret_cast<uint4*>(dt) = *reinterpret_cast<const uint4*>(dy + base + c);
      *reinterpret_cast<uint4*>(zt) = *reinterpret_cast<const uint4*>(z + base + c);
      if (TRAIN_DROP) {
        if (VEC == 8)
          *reinterpret_cast<uint2*>(mv[i]) =
              *reinterpret_cast<const uint2*>(mask + base + c);
        else
          *reinterpret_cast<uint32_t*>(mv[i]) =
              *reinterpret_cast<const uint32_t*>(mask + base + c);
      }
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        zh[i][k] = (DTraits<T>::to_f32(zt[k]) - mu) * rs;
        dw[i][k] = DTraits<T>::to_f32(dt[k]) * gamma[c + k];
        s1 += dw[i][k] * zh[i][k];
        s2 += dw[i][k];
      }
    }
  }
  block_reduce_sum2(s1, s2, red);
  s1 /= H;
  s2 /= H;
#pragma unroll
  for (int i = 0; i < ITEMS; ++i) {
    const int c = cols[i];
    if (c < H) {
      T dzo[VEC], dxo[VEC];
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        const float dzk = rs * (dw[i][k] - s2 - zh[i][k] * s1);
        dzo[k] = DTraits<T>::from_f32(dzk);  // grad to residual input
        const float dxk =
            TRAIN_DROP ? (mv[i][k] ? dzk * keep_scale : 0.f) : dzk;
        dxo[k] = DTraits<T>::from_f32(dxk);
      }
      *reinterpret_cast<uint4*>(dz_res + base + c) =
          *reinterpret_cast<const uint4*>(dzo);
      *reinterpret_cast<uint4*>(dx + base + c) =
          *reinterpret_cast<const uint4*>(dxo);
    }
  }
}

// from layernorm.hip
template <typename T, int VEC>
void launch_col_stats(const T* dy, const T* z, const T* dx_src,
                      const float* mean, const float* rstd, float* part,
                      int rows, int H, int rows_per_chunk, int n_chunks,
                      bool has_bias, hipStream_t stream);

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
  DISPATCH_T(x.scalar_type(), "bdrl_fwd", [&] {
    TORCH_CHECK(H % kVec == 0, "bdrl_fwd: H % ", kVec, " != 0");
    const int slices = H / kVec;
    auto launch = [&](auto items_c, int threads, auto has_bias_c,
                      auto train_c) {
      hipLaunchKernelGGL(
          (bdrl_fwd_kernel<scalar_t, kVec, decltype(items_c)::value,
                           decltype(has_bias_c)::value,
                           decltype(train_c)::value>),
          dim3(rows), dim3(threads), 0, stream,
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
    auto by_flags = [&](auto items_c, int threads) {
      if (has_bias && train_drop)
        launch(items_c, threads, std::true_type{}, std::true_type{});
      else if (has_bias)
        launch(items_c, threads, std::true_type{}, std::false_type{});
      else if (train_drop)
        launch(items_c, threads, std::false_type{}, std::true_type{});
      else
        launch(items_c, threads, std::false_type{}, std::false_type{});
    };
    if (slices <= 256)
      by_flags(std::integral_constant<int, 1>{}, ((slices + 63) / 64) * 64);
    else if (slices <= 512)
      by_flags(std::integral_constant<int, 2>{}, 256);
    else if (slices <= 1024)
      by_flags(std::integral_constant<int, 2>{}, 512);
    else if (slices <= 2048)
      by_flags(std::integral_constant<int, 4>{}, 512);
    else {
      TORCH_CHECK(slices <= 4096, "bdrl_fwd: H too large");
      by_flags(std::integral_constant<int, 4>{}, 1024);
    }
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
  const int rows_per_chunk = 16;  // halves partial-buffer traffic; 16 independent loads/thread
  const int n_chunks = (rows + rows_per_chunk - 1) / rows_per_chunk;
  const int n_slabs = has_bias ? 3 : 2;
  auto fopts = z.options().dtype(torch::kFloat32);
  auto part = torch::empty({n_chunks, n_slabs * H}, fopts);
  const bool train_drop = p > 0.0;
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_T(z.scalar_type(), "bdrl_bwd", [&] {
    TORCH_CHECK(H % kVec == 0, "bdrl_bwd: H % ", kVec, " != 0");
    const int slices = H / kVec;
    auto launch = [&](auto items_c, int threads, auto train_c) {
      hipLaunchKernelGGL(
          (bdrl_bwd_dx_kernel<scalar_t, kVec, decltype(items_c)::value,
                              decltype(train_c)::value>),
          dim3(rows), dim3(threads), 0, stream,
          reinterpret_cast<const scalar_t*>(dy_c.data_ptr()),
          reinterpret_cast<const scalar_t*>(z.data_ptr()),
          train_drop ? mask.data_ptr<uint8_t>() : nullptr,
          gamma_f.data_ptr<float>(), mean.data_ptr<float>(),
          rstd.data_ptr<float>(),
          reinterpret_cast<scalar_t*>(dx.data_ptr()),
          reinterpret_cast<scalar_t*>(dz_res.data_ptr()), rows, H,
          static_cast<float>(p));
    };
    auto by_items = [&](auto train_c) {
      if (slices <= 256)
        launch(std::integral_constant<int, 1>{},
               ((slices + 63) / 64) * 64, train_c);
      else if (slices <= 512)
        launch(std::integral_constant<int, 2>{}, 256, train_c);
      else if (slices <= 1024)
        launch(std::integral_constant<int, 2>{}, 512, train_c);
      else if (slices <= 2048)
        launch(std::integral_constant<int, 4>{}, 512, train_c);
      else {
        TORCH_CHECK(slices <= 4096, "bdrl_bwd: H too large");
        launch(std::integral_constant<int, 4>{}, 1024, train_c);
      }
    };
    if (train_drop)
      by_items(std::true_type{});
    else
      by_items(std::false_type{});
    launch_col_stats<scalar_t, kVec>(
        reinterpret_cast<const scalar_t*>(dy_c.data_ptr()),
        reinterpret_cast<const scalar_t*>(z.data_ptr()),
        reinterpret_cast<const scalar_t*>(dx.data_ptr()),
        mean.data_ptr<float>(), rstd.data_ptr<float>(),
        part.data_ptr<float>(), rows, H, rows_per_chunk, n_chunks, has_bias,
        stream);
  });
  auto out = col_reduce_full(part);
  auto dgamma = out.narrow(0, 0, H).contiguous();
  auto dbeta = out.narrow(0, H, H).contiguous();
  auto dbias =
      has_bias ? out.narrow(0, 2 * H, H).contiguous() : torch::empty({0}, fopts);
  if (gamma.scalar_type() != torch::kFloat32) {
    dgamma = dgamma.to(gamma.scalar_type());
    dbeta = dbeta.to(gamma.scalar_type());
    if (has_bias) dbias = dbias.to(gamma.scalar_type());
  }
  return {dx, dbias, dz_res, dgamma, dbeta};
}

}  // namespace bpa
