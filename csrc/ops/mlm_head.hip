// Hand-written MFMA MLM-decoder GEMM with fused bias + cross-entropy
// forward for MI355X (gfx950).
//
// logits[P,V] = h[P,K] @ W[V,K]^T + bias[V], V = vocab (~30528),
// K = hidden (1024), P = gathered masked rows (B * max_preds) — the
// single biggest GEMM of the reference's MLM head
// (src/modeling.py:570-578 decoder matmul + run_pretraining.py:58-72
// CrossEntropyLoss). One kernel computes the GEMM, adds the bias, and
// produces the cross-entropy forward statistics in the epilogue:
// per-(row, 64-column-stripe) online (max, sum-exp) fp32 partials
// folded by a tiny second kernel into the per-row logsumexp + NLL
// loss. The separate full [P,V] read of a standalone CE-forward pass
// disappears; backward reuses ce_bwd (csrc/ops/cross_entropy.hip) on
// the bf16 logits this kernel stores.
//
// Tile choice (measured, see docs/KERNELS.md): 128x128 and
// stripe-persistent (W-streamed-once) structures both landed 210-300us
// against hipBLASLt's ~92us — PMC showed waves parked 5.7x their busy
// cycles; this shape is bound by on-chip operand re-reads and MFMA
// work per barrier, not HBM (the 62.5-MB W matrix is L3-resident,
// Infinity Cache = 256 MB). The library's own pick is a 256x256
// macro-tile, which quadruples FLOPs per staged byte; this kernel uses
// the same: 512 threads = 8 waves as 2(M)x4(N), each wave a 128x64
// sub-tile (acc 128 regs), BK=64 double-buffered global_load_lds
// (128 KB LDS, one block per CU), one __syncthreads per k-step.
//
// LDS images are lane-linear (glds writes wave-uniform base +
// lane*16), so the SOURCE address carries a 16-B-granule XOR swizzle:
//   LDS[row r][granule p] = global[row r][granule p ^ (r & 7)]
// A row is one 128-B cache line and the permutation stays inside it,
// so coalescing is untouched; fragment b64 reads spread each 16-lane
// group over 8 granules (2-way conflict), noise under the MFMAs.
//
// Numerics: fp32 MFMA accumulation; the CE statistics are computed
// from the bf16-ROUNDED logits (exactly the values backward re-reads),
// so forward lse and backward softmax see identical inputs.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../common.h"

namespace bpa {

namespace mh {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((__vector_size__(4 * sizeof(__bf16)))) __bf16 bf16x4v;

#define MFMA16(A, B, C) \
  __builtin_amdgcn_mfma_f32_16x16x32_bf16((A), (B), (C), 0, 0, 0)

constexpr int kBM = 256;  // h-rows per block
constexpr int kBN = 256;  // vocab columns per block (4 x 64 stripes)
constexpr int kBK = 64;   // K step (two MFMA depths per staged tile)
constexpr int kATileB = kBM * kBK * 2;    // h image bytes (32 KB)
constexpr int kWTileB = kBN * kBK * 2;    // W image bytes (32 KB)
constexpr int kBufB = kATileB + kWTileB;  // one pipeline buffer (64 KB)

// element byte offset of (row r, k) inside one swizzled image
__device__ __forceinline__ int swz_off(int r, int k) {
  return r * (kBK * 2) + ((((k >> 3) ^ (r & 7)) << 4) | ((k & 7) << 1));
}

// fragment for MFMA depth base ks (0 or 32): lane (g = lane>>4,
// li = lane&15) holds row rb+li, k = ks+4g..+3 and ks+16+4g..+3 (the
// probe-verified gfx950 16x16x32 layout).
__device__ __forceinline__ bf16x8 frag_k(const char* img, int rb, int ks) {
  const int lane = threadIdx.x & 63;
  const int g = (lane >> 4) & 3, li = lane & 15;
  const int r = rb + li;
  union {
    bf16x8 v;
    bf16x4v h[2];
  } f;
  f.h[0] = *reinterpret_cast<const bf16x4v*>(img + swz_off(r, ks + 4 * g));
  f.h[1] =
      *reinterpret_cast<const bf16x4v*>(img + swz_off(r, ks + 16 + 4 * g));
  return f.v;
}

}  // namespace mh

using mh::bf16x8;
using mh::f32x4;

// grid: (P/256, ceil(V/256)); 512 threads = 8 waves as 2(M) x 4(N),
// each wave a 128x64 sub-tile. The wave's 64-col stripe is also the
// CE-partial granule, so row statistics never cross waves.
__global__ __launch_bounds__(512) void mlm_fwd_kernel(
    const __bf16* __restrict__ h,     // [P, K]
    const __bf16* __restrict__ w,     // [V, K]
    const float* __restrict__ bias,   // [V]
    __bf16* __restrict__ logits,      // [P, V]
    float* __restrict__ part,         // [P, nStripes, 2] (max, sumexp)
    int P, int V, int K, int nStripes) {
  const int m0 = blockIdx.x * mh::kBM;
  const int tid = threadIdx.x;
  const int lane = tid & 63, wave = tid >> 6;
  const int wi = wave >> 2, wj = wave & 3;  // 2 x 4 wave grid
  const int g = (lane >> 4), li = lane & 15;
  const int stripe = blockIdx.y * 4 + wj;   // this wave's 64-col stripe
  const int n0 = blockIdx.y * mh::kBN;
  const int nw0 = n0 + wj * 64;

  extern __shared__ __attribute__((aligned(16))) char smem[];

  // per-wave column stripe is fixed: bias once, in registers
  const bool any = nw0 < V;                  // wave has valid columns
  const bool full = (nw0 + 64) <= V;
  float bv[4];
  bool cv[4];
#pragma unroll
  for (int tj = 0; tj < 4; ++tj) {
    const int nj = nw0 + tj * 16 + li;
    cv[tj] = nj < V;
    bv[tj] = cv[tj] ? bias[nj] : 0.f;
  }

  // global_load_lds staging: per wave 4 A + 4 W calls x 8 rows x 128 B
  const int st_sub = (lane >> 3);                // row within 8-row piece
  const int st_swz = ((lane & 7) ^ st_sub) * 8;  // swizzled source granule
  auto stage = [&](int k0, int b) {
    char* base = smem + b * mh::kBufB;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int r = wave * 32 + c * 8 + st_sub;  // 0..255
      const __bf16* ga =
          h + static_cast<int64_t>(m0 + r) * K + k0 + st_swz;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)ga,
          (__attribute__((address_space(3))) void*)(base +
                                                    (wave * 32 + c * 8) *
                                                        128),
          16, 0, 0);
      // vocab tail: clamp W row (always-legal; epilogue masks)
      const __bf16* gw =
          w + static_cast<int64_t>(min(n0 + r, V - 1)) * K + k0 + st_swz;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)gw,
          (__attribute__((address_space(3))) void*)(base + mh::kATileB +
                                                    (wave * 32 + c * 8) *
                                                        128),
          16, 0, 0);
    }
  };

  f32x4 acc[8][4] = {};
  stage(0, 0);
  __syncthreads();

  int buf = 0;
  for (int k0 = 0; k0 < K; k0 += mh::kBK) {
    if (k0 + mh::kBK < K) stage(k0 + mh::kBK, buf ^ 1);  // DMA under MFMA
    const char* at = smem + buf * mh::kBufB;
    const char* wt = at + mh::kATileB;
#pragma unroll
    for (int kd = 0; kd < 2; ++kd) {  // two MFMA depths per staged tile
      bf16x8 bfr[4];
#pragma unroll
      for (int t = 0; t < 4; ++t)
        bfr[t] = mh::frag_k(wt, wj * 64 + t * 16, kd * 32);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ti = 0; ti < 8; ++ti) {
        const bf16x8 af = mh::frag_k(at, wi * 128 + ti * 16, kd * 32);
#pragma unroll
        for (int tj = 0; tj < 4; ++tj)
          acc[ti][tj] = MFMA16(af, bfr[tj], acc[ti][tj]);
      }
      __builtin_amdgcn_s_setprio(0);
    }
    __syncthreads();
    buf ^= 1;
  }

  // ---- epilogue: bias, bf16 logits, per-stripe CE stats (no LDS) ----
  if (!any) return;
#pragma unroll
  for (int ti = 0; ti < 8; ++ti) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int mi = m0 + wi * 128 + ti * 16 + g * 4 + r;
      float mx = -INFINITY, sx = 0.f;
      if (full) {
#pragma unroll
        for (int tj = 0; tj < 4; ++tj) {
          const int nj = nw0 + tj * 16 + li;
          const __bf16 ob = __bf16(acc[ti][tj][r] + bv[tj]);
          logits[static_cast<int64_t>(mi) * V + nj] = ob;
          const float f = static_cast<float>(ob);
          if (f > mx) {
            sx *= __expf(mx - f);
            mx = f;
          }
          sx += __expf(f - mx);
        }
      } else {
#pragma unroll
        for (int tj = 0; tj < 4; ++tj) {
          const int nj = nw0 + tj * 16 + li;
          const __bf16 ob = __bf16(acc[ti][tj][r] + bv[tj]);
          if (cv[tj]) {
            logits[static_cast<int64_t>(mi) * V + nj] = ob;
            const float f = static_cast<float>(ob);
            if (f > mx) {
              sx *= __expf(mx - f);
              mx = f;
            }
            sx += __expf(f - mx);
          }
        }
      }
      // butterfly over the 16-lane li-group: every lane gets the row's
      // (max, sumexp) over this wave's 64-col stripe
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) {
        const float m2 = __shfl_xor(mx, off, 64);
        const float s2 = __shfl_xor(sx, off, 64);
        const float mn = fmaxf(mx, m2);
        sx = (sx == 0.f ? 0.f : sx * __expf(mx - mn)) +
             (s2 == 0.f ? 0.f : s2 * __expf(m2 - mn));
        mx = mn;
      }
      if (li == 0) {
        float* p =
            part + (static_cast<int64_t>(mi) * nStripes + stripe) * 2;
        p[0] = mx;
        p[1] = sx;
      }
    }
  }
}

// fold per-row partials -> lse, loss: one wave per row (4 rows per
// block), coalesced 8-B lane reads over the stripe partials.
__global__ void mlm_fold_kernel(const float* __restrict__ part,
                                const __bf16* __restrict__ logits,
                                const int64_t* __restrict__ labels,
                                float* __restrict__ row_out,  // [2, rows]
                                float* __restrict__ lse_out, int rows,
                                int nStripes, int V, int64_t ignore_index) {
  const int row = blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= rows) return;
  const int lane = threadIdx.x & 63;
  const float* pr = part + static_cast<int64_t>(row) * nStripes * 2;
  float m = -INFINITY, s = 0.f;
  for (int t = lane; t < nStripes; t += 64) {
    const float m2 = pr[t * 2], s2 = pr[t * 2 + 1];
    const float mn = fmaxf(m, m2);
    s = (s == 0.f ? 0.f : s * __expf(m - mn)) +
        (s2 == 0.f ? 0.f : s2 * __expf(m2 - mn));
    m = mn;
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const float m2 = __shfl_xor(m, off, 64);
    const float s2 = __shfl_xor(s, off, 64);
    const float mn = fmaxf(m, m2);
    s = (s == 0.f ? 0.f : s * __expf(m - mn)) +
        (s2 == 0.f ? 0.f : s2 * __expf(m2 - mn));
    m = mn;
  }
  if (lane == 0) {
    const float lse = m + __logf(s);
    lse_out[row] = lse;
    const int64_t label = labels[row];
    // per-row (loss, valid) instead of same-address atomics (2 x rows
    // serialized L2 atomics); mlm_sum_kernel folds them
    const bool valid = label != ignore_index;
    const float xl = valid ? static_cast<float>(
                                 logits[static_cast<int64_t>(row) * V +
                                        (valid ? label : 0)])
                           : 0.f;
    row_out[row] = valid ? lse - xl : 0.f;
    row_out[rows + row] = valid ? 1.f : 0.f;
  }
}

// fold [2, rows] (loss, valid) -> sums [2] = {loss_sum, count}
__global__ void mlm_sum_kernel(const float* __restrict__ row_out,
                               float* __restrict__ sums, int rows) {
  float l = 0.f, c = 0.f;
  for (int i = threadIdx.x; i < rows; i += blockDim.x) {
    l += row_out[i];
    c += row_out[rows + i];
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

bool mlm_head_supported(int64_t P, int64_t V, int64_t K) {
  return P % mh::kBM == 0 && K % mh::kBK == 0 && K >= 64 && V >= 2;
}

// h [P,K] bf16, w [V,K] bf16, bias [V] fp32, labels [P] int64 ->
// {logits [P,V] bf16, loss_sum f32, count f32, lse [P] f32}
std::vector<torch::Tensor> mlm_head_fwd(torch::Tensor h, torch::Tensor w,
                                        torch::Tensor bias,
                                        torch::Tensor labels,
                                        int64_t ignore_index) {
  TORCH_CHECK(h.is_cuda() && h.dim() == 2 && h.is_contiguous() &&
                  h.scalar_type() == torch::kBFloat16,
              "mlm_head_fwd: h must be contiguous 2D bf16");
  TORCH_CHECK(w.is_cuda() && w.dim() == 2 && w.is_contiguous() &&
                  w.scalar_type() == torch::kBFloat16,
              "mlm_head_fwd: w must be contiguous 2D bf16");
  TORCH_CHECK(bias.is_cuda() && bias.dim() == 1 &&
                  bias.scalar_type() == torch::kFloat32,
              "mlm_head_fwd: bias must be 1D fp32");
  const int P = h.size(0), K = h.size(1), V = w.size(0);
  TORCH_CHECK(w.size(1) == K && bias.size(0) == V,
              "mlm_head_fwd: shape mismatch");
  TORCH_CHECK(mlm_head_supported(P, V, K), "mlm_head_fwd: unsupported shape");
  auto labels_c = labels.contiguous();
  TORCH_CHECK(labels_c.size(0) == P, "mlm_head_fwd: labels/P mismatch");

  const int nStripes = (V + 63) / 64;
  auto fopts = h.options().dtype(torch::kFloat32);
  auto logits = torch::empty({P, V}, h.options());
  auto part = torch::empty({P, nStripes, 2}, fopts);
  auto row_out = torch::empty({2, P}, fopts);
  auto sums = torch::empty({2}, fopts);  // {loss_sum, count}
  auto lse = torch::empty({P}, fopts);
  auto stream = at::hip::getCurrentHIPStream();

  const size_t lds = 2 * mh::kBufB;
  HIP_CHECK(hipFuncSetAttribute(
      reinterpret_cast<const void*>(&mlm_fwd_kernel),
      hipFuncAttributeMaxDynamicSharedMemorySize, lds));
  hipLaunchKernelGGL(mlm_fwd_kernel,
                     dim3(P / mh::kBM, (V + mh::kBN - 1) / mh::kBN),
                     dim3(512), lds, stream,
                     reinterpret_cast<const __bf16*>(h.data_ptr()),
                     reinterpret_cast<const __bf16*>(w.data_ptr()),
                     bias.data_ptr<float>(),
                     reinterpret_cast<__bf16*>(logits.data_ptr()),
                     part.data_ptr<float>(), P, V, K, nStripes);
  hipLaunchKernelGGL(mlm_fold_kernel, dim3((P + 3) / 4), dim3(256), 0,
                     stream, part.data_ptr<float>(),
                     reinterpret_cast<const __bf16*>(logits.data_ptr()),
                     labels_c.data_ptr<int64_t>(), row_out.data_ptr<float>(),
                     lse.data_ptr<float>(), P, nStripes, V, ignore_index);
  hipLaunchKernelGGL(mlm_sum_kernel, dim3(1), dim3(256), 0, stream,
                     row_out.data_ptr<float>(), sums.data_ptr<float>(), P);
  return {logits, sums.select(0, 0), sums.select(0, 1), lse};
}

}  // namespace bpa
