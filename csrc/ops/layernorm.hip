// Fused LayerNorm forward/backward for MI355X (gfx950).
//
// Replaces Apex FusedLayerNormAffineFunction (reference call sites:
// src/modeling.py:299-335). Shapes here: rows up to ~12k (B*S), H in
// {64..4096}, eps 1e-12. Memory-bound design:
//
// * forward: one BLOCK per row; every lane loads its 16 B slice ONCE
//   into registers, mean/var via wave shuffle + tiny LDS cross-wave
//   combine, normalized output written from registers — exactly one
//   read + one write of x/y (the old one-wave-per-row version re-read
//   x after the reduction and measured 0.89 TB/s; this version 4.2).
// * backward dx: one BLOCK per row, same register-resident shape.
// * dgamma/dbeta: a separate streaming col_stats_kernel over 16-row
//   chunks (per-lane register accumulators, coalesced loads) writes
//   [n_chunks, 2H] partials folded by col_reduce_full — measured
//   faster than fusing the column accumulation into the dx grid.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../common.h"

namespace bpa {

// ITEMS = uint4 slices each thread owns (ITEMS*VEC*blockDim.x >= H).
template <typename T, int VEC, int ITEMS>
__global__ void ln_fwd_kernel(const T* __restrict__ x,
                              const float* __restrict__ gamma,
                              const float* __restrict__ beta,
                              T* __restrict__ y, float* __restrict__ mean,
                              float* __restrict__ rstd, int rows, int H,
                              float eps) {
  __shared__ float red[32];
  const int row = blockIdx.x;
  if (row >= rows) return;
  const int64_t base = static_cast<int64_t>(row) * H;

  float v[ITEMS][VEC];
  int cols[ITEMS];
  float sum = 0.f, sumsq = 0.f;
#pragma unroll
  for (int i = 0; i < ITEMS; ++i) {
    const int c = (i * blockDim.x + threadIdx.x) * VEC;
    cols[i] = c;
    if (c < H) {
      T t[VEC];
      *reinterpret_cast<uint4*>(t) = *reinterpret_cast<const uint4*>(x + base + c);
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        v[i][k] = DTraits<T>::to_f32(t[k]);
        sum += v[i][k];
        sumsq += v[i][k] * v[i][k];
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
      T o[VEC];
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        o[k] = DTraits<T>::from_f32((v[i][k] - mu) * rs * gamma[c + k] +
                                    beta[c + k]);
      }
      *reinterpret_cast<uint4*>(y + base + c) = *reinterpret_cast<const uint4*>(o);
    }
  }
}

// backward dx: one BLOCK per row, the mirror of ln_fwd_kernel — dy/x
// loaded once into registers, s1/s2 via block reduce, dx written from
// registers. dgamma/dbeta are NOT computed here (a block-per-row grid
// cannot accumulate columns without atomics); see col_stats_kernel.
template <typename T, int VEC, int ITEMS>
__global__ void ln_bwd_dx_kernel(const T* __restrict__ dy,
                                 const T* __restrict__ x,
                                 const float* __restrict__ gamma,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ rstd,
                                 T* __restrict__ dx, int rows, int H) {
  __shared__ float red[32];
  const int row = blockIdx.x;
  if (row >= rows) return;
  const int64_t base = static_cast<int64_t>(row) * H;
  const float mu = mean[row], rs = rstd[row];

  float dw[ITEMS][VEC], xh[ITEMS][VEC];
  int cols[ITEMS];
  float s1 = 0.f, s2 = 0.f;
#pragma unroll
  for (int i = 0; i < ITEMS; ++i) {
    const int c = (i * blockDim.x + threadIdx.x) * VEC;
    cols[i] = c;
    if (c < H) {
      T dt[VEC], xt[VEC];
      *reinterpret_cast<uint4*>(dt) = *reinterpret_cast<const uint4*>(dy + base + c);
      *reinterpret_cast<uint4*>(xt) = *reinterpret_cast<const uint4*>(x + base + c);
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        xh[i][k] = (DTraits<T>::to_f32(xt[k]) - mu) * rs;
        dw[i][k] = DTraits<T>::to_f32(dt[k]) * gamma[c + k];
        s1 += dw[i][k] * xh[i][k];
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
      T o[VEC];
#pragma unroll
      for (int k = 0; k < VEC; ++k)
        o[k] = DTraits<T>::from_f32(rs * (dw[i][k] - s2 - xh[i][k] * s1));
      *reinterpret_cast<uint4*>(dx + base + c) = *reinterpret_cast<const uint4*>(o);
    }
  }
}

// Column stats for the LN family backward: streaming column reduction
//   dgamma[c] += dy[r][c] * (z[r][c]-mu)*rs,  dbeta[c] += dy[r][c]
//   (+ dbias[c] += dx_src[r][c] when dx_src != nullptr)
// Each thread owns VEC contiguous columns and loops its row chunk with
// independent (software-pipelineable) loads; partials [n_chunks, nsl*H]
// are folded by col_reduce_kernel. Shared by ln_bwd and bdrl_bwd.
template <typename T, int VEC, bool HAS_BIAS>
__global__ void col_stats_kernel(const T* __restrict__ dy,
                                 const T* __restrict__ z,
                                 const T* __restrict__ dx_src,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ rstd,
                                 float* __restrict__ part, int rows, int H,
                                 int rows_per_chunk) {
  const int c = (blockIdx.x * blockDim.x + threadIdx.x) * VEC;
  if (c >= H) return;
  const int r0 = blockIdx.y * rows_per_chunk;
  const int r1 = min(r0 + rows_per_chunk, rows);
  float acc_g[VEC] = {}, acc_b[VEC] = {}, acc_bias[VEC] = {};
  for (int r = r0; r < r1; ++r) {
    const int64_t base = static_cast<int64_t>(r) * H + c;
    const float mu = mean[r], rs = rstd[r];
    T dt[VEC], zt[VEC], xt[VEC];
    *reinterpret_cast<uint4*>(dt) = *reinterpret_cast<const uint4*>(dy + base);
    *reinterpret_cast<uint4*>(zt) = *reinterpret_cast<const uint4*>(z + base);
    if (HAS_BIAS)
      *reinterpret_cast<uint4*>(xt) =
          *reinterpret_cast<const uint4*>(dx_src + base);
#pragma unroll
    for (int k = 0; k < VEC; ++k) {
      const float d = DTraits<T>::to_f32(dt[k]);
      acc_g[k] += d * (DTraits<T>::to_f32(zt[k]) - mu) * rs;
      acc_b[k] += d;
      if (HAS_BIAS) acc_bias[k] += DTraits<T>::to_f32(xt[k]);
    }
  }
  const int nsl = HAS_BIAS ? 3 : 2;
  float* p = part + static_cast<int64_t>(blockIdx.y) * nsl * H + c;
#pragma unroll
  for (int k = 0; k < VEC; ++k) {
    p[k] = acc_g[k];
    p[H + k] = acc_b[k];
    if (HAS_BIAS) p[2 * H + k] = acc_bias[k];
  }
}

// instantiation helper shared with fused_residual.hip
template <typename T, int VEC>
void launch_col_stats(const T* dy, const T* z, const T* dx_src,
                      const float* mean, const float* rstd, float* part,
                      int rows, int H, int rows_per_chunk, int n_chunks,
                      bool has_bias, hipStream_t stream) {
  const int threads = tmin(256, ((H / VEC + 63) / 64) * 64);
  dim3 grid((H / VEC + threads - 1) / threads, n_chunks);
  if (has_bias) {
    hipLaunchKernelGGL((col_stats_kernel<T, VEC, true>), grid, dim3(threads),
                       0, stream, dy, z, dx_src, mean, rstd, part, rows, H,
                       rows_per_chunk);
  } else {
    hipLaunchKernelGGL((col_stats_kernel<T, VEC, false>), grid, dim3(threads),
                       0, stream, dy, z, dx_src, mean, rstd, part, rows, H,
                       rows_per_chunk);
  }
}

template void launch_col_stats<__hip_bfloat16, 8>(
    const __hip_bfloat16*, const __hip_bfloat16*, const __hip_bfloat16*,
    const float*, const float*, float*, int, int, int, int, bool, hipStream_t);
template void launch_col_stats<__half, 8>(const __half*, const __half*,
                                          const __half*, const float*,
                                          const float*, float*, int, int, int,
                                          int, bool, hipStream_t);
template void launch_col_stats<float, 4>(const float*, const float*,
                                         const float*, const float*,
                                         const float*, float*, int, int, int,
                                         int, bool, hipStream_t);

// Deterministic blocked column reduce: out[j][c] = sum over the j-th
// contiguous row block of parts[p][c]. Each thread owns 4 consecutive
// fp32 columns (one uint4 per row) and streams its rows sequentially —
// coalesced across lanes AND sequential in p, so the sweep runs at HBM
// rate (the first strided-rows version thrashed; a single direct pass
// had only ~S/256 blocks). Direct store: no atomics, no zero-fill.
__global__ void col_reduce_blocked_kernel(const float* __restrict__ parts,
                                          int nparts, int S, int rows_per_j,
                                          float* __restrict__ out) {
  const int c = (blockIdx.x * blockDim.x + threadIdx.x) * 4;
  if (c >= S) return;
  const int p0 = blockIdx.y * rows_per_j;
  const int p1 = min(p0 + rows_per_j, nparts);
  float acc[4] = {};
  for (int p = p0; p < p1; ++p) {
    const float* src = parts + static_cast<int64_t>(p) * S + c;
    if (c + 4 <= S) {
      float v[4];
      *reinterpret_cast<uint4*>(v) = *reinterpret_cast<const uint4*>(src);
#pragma unroll
      for (int k = 0; k < 4; ++k) acc[k] += v[k];
    } else {
      for (int k = 0; c + k < S; ++k) acc[k] += src[k];
    }
  }
  float* dst = out + static_cast<int64_t>(blockIdx.y) * S + c;
  for (int k = 0; k < 4 && c + k < S; ++k) dst[k] = acc[k];
}

// atomic accumulate variant: one launch, chunked rows per y-block.
// Measured fastest at the [0.7k-1.5k rows, 2-12k cols] partial shapes
// (the deterministic two-pass variants lose to it on second-pass
// parallelism); fp32 atomics over a zeroed output, still bitwise
// reproducible per launch geometry.
__global__ void col_reduce_atomic_kernel(const float* __restrict__ parts,
                                         int nparts, int S,
                                         float* __restrict__ out) {
  const int c = (blockIdx.x * blockDim.x + threadIdx.x) * 4;
  if (c >= S) return;
  const int p0 = blockIdx.y * kColChunk;
  const int p1 = min(p0 + kColChunk, nparts);
  float acc[4] = {};
  for (int p = p0; p < p1; ++p) {
    const float* src = parts + static_cast<int64_t>(p) * S + c;
    if (c + 4 <= S) {
      float v[4];
      *reinterpret_cast<uint4*>(v) = *reinterpret_cast<const uint4*>(src);
#pragma unroll
      for (int k = 0; k < 4; ++k) acc[k] += v[k];
    } else {
      for (int k = 0; c + k < S; ++k) acc[k] += src[k];
    }
  }
  for (int k = 0; k < 4 && c + k < S; ++k) atomicAdd(&out[c + k], acc[k]);
}

torch::Tensor col_reduce_full(torch::Tensor parts) {
  const int nparts = parts.size(0);
  const int S = parts.size(1);
  auto stream = at::hip::getCurrentHIPStream();
  const int xblocks = (S / 4 + 255) / 256;
  if (nparts <= 48) {
    auto out = torch::empty({S}, parts.options());
    hipLaunchKernelGGL(col_reduce_blocked_kernel, dim3(xblocks, 1),
                       dim3(256), 0, stream, parts.data_ptr<float>(), nparts,
                       S, nparts, out.data_ptr<float>());
    return out;
  }
  auto out = torch::zeros({S}, parts.options());
  dim3 grid(xblocks, (nparts + kColChunk - 1) / kColChunk);
  hipLaunchKernelGGL(col_reduce_atomic_kernel, grid, dim3(256), 0, stream,
                     parts.data_ptr<float>(), nparts, S,
                     out.data_ptr<float>());
  return out;
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

// pick (threads, ITEMS) for the forward block-per-row shape
template <typename LaunchFn>
static void launch_by_items(int slices, LaunchFn&& fn) {
  if (slices <= 256) {
    fn(std::integral_constant<int, 1>{}, ((slices + 63) / 64) * 64);
  } else if (slices <= 512) {
    fn(std::integral_constant<int, 2>{}, 256);
  } else if (slices <= 1024) {
    fn(std::integral_constant<int, 2>{}, 512);
  } else if (slices <= 2048) {
    fn(std::integral_constant<int, 4>{}, 512);
  } else {
    TORCH_CHECK(slices <= 4096, "layernorm: H too large");
    fn(std::integral_constant<int, 4>{}, 1024);
  }
}

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
  DISPATCH_FLOATING(x.scalar_type(), "ln_fwd", [&] {
    TORCH_CHECK(H % kVec == 0, "ln_fwd: H must be a multiple of ", kVec);
    launch_by_items(H / kVec, [&](auto items_c, int threads) {
      hipLaunchKernelGGL((ln_fwd_kernel<scalar_t, kVec, decltype(items_c)::value>),
                         dim3(rows), dim3(threads), 0, stream,
                         reinterpret_cast<const scalar_t*>(x.data_ptr()),
                         gamma_f.data_ptr<float>(), beta_f.data_ptr<float>(),
                         reinterpret_cast<scalar_t*>(y.data_ptr()),
                         mean.data_ptr<float>(), rstd.data_ptr<float>(), rows,
                         H, static_cast<float>(eps));
    });
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
  const int rows_per_chunk = 16;  // halves partial-buffer traffic; 16 independent loads/thread
  const int n_chunks = (rows + rows_per_chunk - 1) / rows_per_chunk;
  auto opts = x.options().dtype(torch::kFloat32);
  auto part = torch::empty({n_chunks, 2 * H}, opts);
  auto stream = at::hip::getCurrentHIPStream();
  auto dy_c = dy.contiguous();
  DISPATCH_FLOATING(x.scalar_type(), "ln_bwd", [&] {
    TORCH_CHECK(H % kVec == 0, "ln_bwd: H must be a multiple of ", kVec);
    launch_by_items(H / kVec, [&](auto items_c, int threads) {
      hipLaunchKernelGGL(
          (ln_bwd_dx_kernel<scalar_t, kVec, decltype(items_c)::value>),
          dim3(rows), dim3(threads), 0, stream,
          reinterpret_cast<const scalar_t*>(dy_c.data_ptr()),
          reinterpret_cast<const scalar_t*>(x.data_ptr()),
          gamma_f.data_ptr<float>(), mean.data_ptr<float>(),
          rstd.data_ptr<float>(),
          reinterpret_cast<scalar_t*>(dx.data_ptr()), rows, H);
    });
    launch_col_stats<scalar_t, kVec>(
        reinterpret_cast<const scalar_t*>(dy_c.data_ptr()),
        reinterpret_cast<const scalar_t*>(x.data_ptr()), nullptr,
        mean.data_ptr<float>(), rstd.data_ptr<float>(),
        part.data_ptr<float>(), rows, H, rows_per_chunk, n_chunks, false,
        stream);
  });
  auto dgb = col_reduce_full(part);
  auto dgamma = dgb.narrow(0, 0, H);
  auto dbeta = dgb.narrow(0, H, H);
  if (gamma.scalar_type() != torch::kFloat32) {
    return {dx, dgamma.to(gamma.scalar_type()), dbeta.to(gamma.scalar_type())};
  }
  return {dx, dgamma.contiguous(), dbeta.contiguous()};
}

}  // namespace bpa
