// Fused log-softmax + NLL cross-entropy (ignore_index) for MI355X.
//
// The MLM loss (reference: run_pretraining.py:58-72, CrossEntropyLoss
// ignore_index=-1 over vocab ~30528). Forward never materializes the
// [N, V] log-softmax: one 256-thread block per row does an online
// max/sum-exp reduction and stores only the per-row logsumexp.
// Backward recomputes softmax on the fly.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../common.h"

namespace bpa {

template <typename T, int VEC>
__global__ void ce_fwd_kernel(const T* __restrict__ logits,
                              const int64_t* __restrict__ labels,
                              float* __restrict__ row_loss,  // [2, rows]
                              float* __restrict__ lse_out, int64_t rows, int V,
                              int64_t ignore_index) {
  const int row = blockIdx.x;
  if (row >= rows) return;
  const int64_t label = labels[row];
  const T* xr = logits + static_cast<int64_t>(row) * V;

  float m = -INFINITY, s = 0.f;
  for (int c = threadIdx.x * VEC; c < V; c += blockDim.x * VEC) {
    T v[VEC];
    *reinterpret_cast<uint4*>(v) = *reinterpret_cast<const uint4*>(xr + c);
#pragma unroll
    for (int k = 0; k < VEC; ++k) {
      float f = DTraits<T>::to_f32(v[k]);
      if (f > m) {
        s *= __expf(m - f);
        m = f;
      }
      s += __expf(f - m);
    }
  }
  __shared__ float sm[256], ss[256];
  sm[threadIdx.x] = m;
  ss[threadIdx.x] = s;
  __syncthreads();
  for (int stride = blockDim.x / 2; stride > 0; stride >>= 1) {
    if (threadIdx.x < stride) {
      float m2 = sm[threadIdx.x + stride], s2 = ss[threadIdx.x + stride];
      float m1 = sm[threadIdx.x], s1 = ss[threadIdx.x];
      float mn = fmaxf(m1, m2);
      // exp(-inf - -inf) guard: empty partials have s == 0
      float sn = (s1 == 0.f ? 0.f : s1 * __expf(m1 - mn)) +
                 (s2 == 0.f ? 0.f : s2 * __expf(m2 - mn));
      sm[threadIdx.x] = mn;
      ss[threadIdx.x] = sn;
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    const float lse = sm[0] + __logf(ss[0]);
    lse_out[row] = lse;
    // per-row loss/valid instead of atomicAdd into one address: the
    // 2 x rows same-line L2 atomics serialized (~20 us at rows=1920);
    // a one-block reduce kernel folds these in ~2 us
    const bool valid = label != ignore_index;
    row_loss[row] = valid ? lse - DTraits<T>::to_f32(xr[label]) : 0.f;
    row_loss[rows + row] = valid ? 1.f : 0.f;
  }
}

// fold [2, rows] (loss, valid) -> sums [2] = {loss_sum, count}
__global__ void ce_reduce_kernel(const float* __restrict__ row_loss,
                                 float* __restrict__ sums, int64_t rows) {
  float l = 0.f, c = 0.f;
  for (int64_t i = threadIdx.x; i < rows; i += blockDim.x) {
    l += row_loss[i];
    c += row_loss[rows + i];
  }
  __shared__ float sl[256], sc[256];
  sl[threadIdx.x] = l;
  sc[threadIdx.x] = c;
  __syncthreads();
  for (int stride = blockDim.x / 2; stride > 0; stride >>= 1) {
    if (threadIdx.x < stride) {
      sl[threadIdx.x] += sl[threadIdx.x + stride];
      sc[threadIdx.x] += sc[threadIdx.x + stride];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    sums[0] = sl[0];
    sums[1] = sc[0];
  }
}

template <typename T, int VEC>
__global__ void ce_bwd_kernel(const float* __restrict__ dloss,
                              const T* __restrict__ logits,
                              const int64_t* __restrict__ labels,
                              const float* __restrict__ lse,
                              const int* __restrict__ count,
                              T* __restrict__ dlogits, int64_t rows, int V,
                              int64_t ignore_index) {
  const int row = blockIdx.x;
  if (row >= rows) return;
  const int64_t label = labels[row];
  const T* xr = logits + static_cast<int64_t>(row) * V;
  T* dr = dlogits + static_cast<int64_t>(row) * V;
  const int n_valid = max(*count, 1);
  const float scale = (label == ignore_index) ? 0.f : dloss[0] / n_valid;
  const float row_lse = lse[row];
  for (int c = threadIdx.x * VEC; c < V; c += blockDim.x * VEC) {
    T v[VEC], o[VEC];
    *reinterpret_cast<uint4*>(v) = *reinterpret_cast<const uint4*>(xr + c);
#pragma unroll
    for (int k = 0; k < VEC; ++k) {
      float p = __expf(DTraits<T>::to_f32(v[k]) - row_lse);
      float g = scale * (p - ((c + k) == label ? 1.f : 0.f));
      o[k] = DTraits<T>::from_f32(g);
    }
    *reinterpret_cast<uint4*>(dr + c) = *reinterpret_cast<const uint4*>(o);
  }
}

#define DISPATCH_CE(TYPE, NAME, ...)                                         \
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

std::vector<torch::Tensor> ce_fwd(torch::Tensor logits, torch::Tensor labels,
                                  int64_t ignore_index) {
  TORCH_CHECK(logits.dim() == 2 && logits.is_contiguous(), "ce_fwd: bad logits");
  const int64_t rows = logits.size(0);
  const int V = logits.size(1);
  auto labels_c = labels.contiguous();
  auto fopts = logits.options().dtype(torch::kFloat32);
  auto row_loss = torch::empty({2, rows}, fopts);
  auto sums = torch::empty({2}, fopts);  // {loss_sum, count}
  auto lse = torch::empty({rows}, fopts);
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_CE(logits.scalar_type(), "ce_fwd", [&] {
    TORCH_CHECK(V % kVec == 0, "ce_fwd: V % ", kVec, " != 0");
    hipLaunchKernelGGL((ce_fwd_kernel<scalar_t, kVec>), dim3(rows), dim3(256),
                       0, stream,
                       reinterpret_cast<const scalar_t*>(logits.data_ptr()),
                       labels_c.data_ptr<int64_t>(),
                       row_loss.data_ptr<float>(),
                       lse.data_ptr<float>(), rows, V, ignore_index);
  });
  hipLaunchKernelGGL(ce_reduce_kernel, dim3(1), dim3(256), 0, stream,
                     row_loss.data_ptr<float>(), sums.data_ptr<float>(),
                     rows);
  return {sums.select(0, 0), sums.select(0, 1), lse};
}

torch::Tensor ce_bwd(torch::Tensor dloss, torch::Tensor logits,
                     torch::Tensor labels, torch::Tensor lse,
                     torch::Tensor count, int64_t ignore_index) {
  const int64_t rows = logits.size(0);
  const int V = logits.size(1);
  auto labels_c = labels.contiguous();
  auto dlogits = torch::empty_like(logits);
  auto dloss_f = dloss.to(torch::kFloat32).reshape({1}).contiguous();
  auto count_i = count.to(torch::kInt32).reshape({1}).contiguous();
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_CE(logits.scalar_type(), "ce_bwd", [&] {
    hipLaunchKernelGGL((ce_bwd_kernel<scalar_t, kVec>), dim3(rows), dim3(256),
                       0, stream, dloss_f.data_ptr<float>(),
                       reinterpret_cast<const scalar_t*>(logits.data_ptr()),
                       labels_c.data_ptr<int64_t>(), lse.data_ptr<float>(),
                       count_i.data_ptr<int>(),
                       reinterpret_cast<scalar_t*>(dlogits.data_ptr()), rows,
                       V, ignore_index);
  });
  return dlogits;
}

}  // namespace bpa
