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
                              float* __restrict__ loss_sum,
                              int* __restrict__ count,
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
    if (label != ignore_index) {
      const float xl = DTraits<T>::to_f32(xr[label]);
      atomicAdd(loss_sum, lse - xl);
      atomicAdd(count, 1);
    }
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
  auto loss_sum = torch::zeros({1}, fopts);
  auto count = torch::zeros({1}, logits.options().dtype(torch::kInt32));
  auto lse = torch::empty({rows}, fopts);
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_CE(logits.scalar_type(), "ce_fwd", [&] {
    TORCH_CHECK(V % kVec == 0, "ce_fwd: V % ", kVec, " != 0");
    hipLaunchKernelGGL((ce_fwd_kernel<scalar_t, kVec>), dim3(rows), dim3(256),
                       0, stream,
                       reinterpret_cast<const scalar_t*>(logits.data_ptr()),
                       labels_c.data_ptr<int64_t>(),
                       loss_sum.data_ptr<float>(), count.data_ptr<int>(),
                       lse.data_ptr<float>(), rows, V, ignore_index);
  });
  return {loss_sum.squeeze(0), count.squeeze(0).to(torch::kFloat32), lse};
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
