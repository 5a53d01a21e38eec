// Hand-written MFMA MLM-decoder GEMM with fused bias + cross-entropy
// forward for MI355X (gfx950).
//
// logits[P,V] = h[P,K] @ W[V,K]^T + bias[V], V = vocab (~30528),
// K = hidden (1024), P = gathered masked rows (B * max_preds) — the
// single biggest GEMM of the reference's MLM head
// (src/modeling.py:570-578 decoder matmul + run_pretraining.py:58-72
// CrossEntropyLoss). One kernel computes the GEMM, adds the bias, and
// produces the cross-entropy forward statistics in the epilogue:
// per-(row, column-stripe) online (max, sum-exp) fp32 partials folded
// by a tiny second kernel into the per-row logsumexp + NLL loss. The
// separate full [P,V] read of a standalone CE-forward pass disappears;
// backward reuses ce_bwd (csrc/ops/cross_entropy.hip) on the bf16
// logits this kernel stores.
//
// Decomposition — this shape is "M tiny, N huge" (the guide's
// M=256 projection-GEMM regime): each block owns a 64-column vocab
// stripe and loops over P in 128-row chunks with the k-loop innermost,
// so a W stripe is fetched once and re-read from cache, and the small
// h operand (2.6-5 MB) is L2/L3-resident. Two fill levers on top:
//   * P-split: V/64 stripes alone are ~477 blocks = 23% of the chip
//     (PMC: waves parked 5.7x their busy cycles). The chunk loop is
//     split nSplit ways across blocks; the same-stripe splits are
//     placed on the SAME XCD (observed placement: XCD = block % 8,
//     MI355X_MICROARCH "Workgroup dispatch") so the stripe's W tile is
//     fetched into one L2 once.
//   * 3-buffer global_load_lds pipeline with counted s_waitcnt
//     vmcnt(N) and a raw s_barrier, keeping one 24-KB tile in flight
//     ACROSS each barrier (a plain __syncthreads drains the DMA queue
//     with vmcnt(0) every step).
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

constexpr int kBN = 64;      // W-rows (vocab stripe) per block
constexpr int kChunk = 128;  // h-rows per chunk of the in-block P loop
constexpr int kBK = 64;      // K step (two MFMA depths per staged tile)
constexpr int kATileB = kChunk * kBK * 2;  // h image bytes (16 KB)
constexpr int kWTileB = kBN * kBK * 2;     // W image bytes (8 KB)
constexpr int kBufB = kATileB + kWTileB;   // one pipeline buffer (24 KB)
constexpr int kGldsPerTile = 6;  // per-wave glds calls per staged tile

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

#define WAIT_VM(n) asm volatile("s_waitcnt vmcnt(" #n ")" ::: "memory")
#define RAW_BARRIER()                                    \
  do {                                                   \
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");   \
    __builtin_amdgcn_s_barrier();                        \
  } while (0)

}  // namespace mh

using mh::bf16x8;
using mh::f32x4;

// grid: 8 * ceil((V/64)/8) * nSplit linear blocks of 256 threads
// (4 waves); block -> (stripe, split) keeps same-stripe splits on one
// XCD. Each wave owns a 32-row x 64-col sub-tile of the current chunk.
__global__ __launch_bounds__(256) void mlm_fwd_kernel(
    const __bf16* __restrict__ h,     // [P, K]
    const __bf16* __restrict__ w,     // [V, K]
    const float* __restrict__ bias,   // [V]
    __bf16* __restrict__ logits,      // [P, V]
    float* __restrict__ part,         // [P, nTiles, 2] (max, sumexp)
    int P, int V, int K, int nTiles, int nSplit) {
  const int L = blockIdx.x;
  const int q = L >> 3;
  const int split = q % nSplit;
  const int stripe = (L & 7) + 8 * (q / nSplit);
  if (stripe >= nTiles) return;  // ragged tail of the XCD mapping
  const int n0 = stripe * mh::kBN;
  const int tid = threadIdx.x;
  const int lane = tid & 63, wave = tid >> 6;
  const int g = (lane >> 4), li = lane & 15;
  const int nChunks = P / mh::kChunk;
  const int kSteps = K / mh::kBK;
  const int myChunks = (nChunks - split + nSplit - 1) / nSplit;
  if (myChunks <= 0) return;
  const int total = myChunks * kSteps;

  extern __shared__ __attribute__((aligned(16))) char smem[];

  // column stripe is fixed for the whole block: bias once, in registers
  const bool full = (n0 + mh::kBN) <= V;
  float bv[4];
  bool cv[4];
#pragma unroll
  for (int tj = 0; tj < 4; ++tj) {
    const int nj = n0 + tj * 16 + li;
    cv[tj] = nj < V;
    bv[tj] = cv[tj] ? bias[nj] : 0.f;
  }

  // global_load_lds staging of one (h-chunk, W-stripe) k-slice pair:
  // per wave 4 A calls + 2 W calls x 8 rows x 128 B, 16 B per lane.
  const int st_sub = (lane >> 3);                // row within 8-row piece
  const int st_swz = ((lane & 7) ^ st_sub) * 8;  // swizzled source granule
  auto stage = [&](int mc, int k0, int b) {
    char* base = smem + b * mh::kBufB;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int r = wave * 32 + c * 8 + st_sub;
      const __bf16* ga =
          h + static_cast<int64_t>(mc * mh::kChunk + r) * K + k0 + st_swz;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)ga,
          (__attribute__((address_space(3))) void*)(base +
                                                    (wave * 32 + c * 8) *
                                                        128),
          16, 0, 0);
    }
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      const int r = wave * 16 + c * 8 + st_sub;
      // vocab tail stripe: clamp W row (always-legal; epilogue masks)
      const __bf16* gw =
          w + static_cast<int64_t>(min(n0 + r, V - 1)) * K + k0 + st_swz;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)gw,
          (__attribute__((address_space(3))) void*)(base + mh::kATileB +
                                                    (wave * 16 + c * 8) *
                                                        128),
          16, 0, 0);
    }
  };

  // stage-pointer counters (chunk index is split-strided through P)
  int st_mc = split, st_kk = 0, staged = 0;
  auto advance_st = [&]() {
    if (++st_kk == kSteps) {
      st_kk = 0;
      st_mc += nSplit;
    }
    ++staged;
  };
  stage(st_mc, 0, 0);
  advance_st();
  if (staged < total) {
    stage(st_mc, st_kk * mh::kBK, 1);
    advance_st();
  }

  f32x4 acc[2][4] = {};
  int mc = split, kk = 0;
  for (int s = 0; s < total; ++s) {
    // tile s must have landed; tile s+1 (6 glds) may stay in flight
    // across the barrier — a raw barrier, NOT __syncthreads, which
    // would emit vmcnt(0) and drain the whole DMA queue
    if (s + 1 < total) {
      WAIT_VM(6);
    } else {
      WAIT_VM(0);
    }
    RAW_BARRIER();
    if (staged < total) {  // refill the buffer freed two steps ago
      stage(st_mc, st_kk * mh::kBK, staged % 3);
      advance_st();
    }

    const char* at = smem + (s % 3) * mh::kBufB;
    const char* wt = at + mh::kATileB;
#pragma unroll
    for (int kd = 0; kd < 2; ++kd) {  // two MFMA depths per staged tile
      bf16x8 af[2], bfr[4];
#pragma unroll
      for (int t = 0; t < 2; ++t)
        af[t] = mh::frag_k(at, wave * 32 + t * 16, kd * 32);
#pragma unroll
      for (int t = 0; t < 4; ++t) bfr[t] = mh::frag_k(wt, t * 16, kd * 32);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ti = 0; ti < 2; ++ti)
#pragma unroll
        for (int tj = 0; tj < 4; ++tj)
          acc[ti][tj] = MFMA16(af[ti], bfr[tj], acc[ti][tj]);
      __builtin_amdgcn_s_setprio(0);
    }

    if (++kk == kSteps) {
      // ---- chunk epilogue: bias, bf16 logits, CE stats (no LDS) ----
      const int m0 = mc * mh::kChunk;
#pragma unroll
      for (int ti = 0; ti < 2; ++ti) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int mi = m0 + wave * 32 + ti * 16 + g * 4 + r;
          float mx = -INFINITY, sx = 0.f;
          if (full) {
#pragma unroll
            for (int tj = 0; tj < 4; ++tj) {
              const int nj = n0 + tj * 16 + li;
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
              const int nj = n0 + tj * 16 + li;
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
          // butterfly over the 16-lane li-group: every lane gets the
          // row's (max, sumexp) over this 64-col stripe
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
                part +
                (static_cast<int64_t>(mi) * nTiles + stripe) * 2;
            p[0] = mx;
            p[1] = sx;
          }
        }
      }
#pragma unroll
      for (int ti = 0; ti < 2; ++ti)
#pragma unroll
        for (int tj = 0; tj < 4; ++tj)
          acc[ti][tj] = f32x4{0.f, 0.f, 0.f, 0.f};
      kk = 0;
      mc += nSplit;
    }
  }
}

// fold per-row partials -> lse, loss: one wave per row (4 rows per
// block), coalesced 8-B lane reads over the stripe partials.
__global__ void mlm_fold_kernel(const float* __restrict__ part,
                                const __bf16* __restrict__ logits,
                                const int64_t* __restrict__ labels,
                                float* __restrict__ sums,  // {loss, count}
                                float* __restrict__ lse_out, int rows,
                                int nTiles, int V, int64_t ignore_index) {
  const int row = blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= rows) return;
  const int lane = threadIdx.x & 63;
  const float* pr = part + static_cast<int64_t>(row) * nTiles * 2;
  float m = -INFINITY, s = 0.f;
  for (int t = lane; t < nTiles; t += 64) {
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
    if (label != ignore_index) {
      const float xl = static_cast<float>(
          logits[static_cast<int64_t>(row) * V + label]);
      atomicAdd(&sums[0], lse - xl);
      atomicAdd(&sums[1], 1.f);
    }
  }
}

bool mlm_head_supported(int64_t P, int64_t V, int64_t K) {
  return P % mh::kChunk == 0 && K % mh::kBK == 0 && K >= 64 && V >= 2;
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

  const int nTiles = (V + mh::kBN - 1) / mh::kBN;
  const int nChunks = P / mh::kChunk;
  // fill the chip: ~477 stripes alone are <2 blocks/CU; split the
  // chunk loop until ~1536 blocks, bounded by the chunk count
  int nSplit = 1;
  while (nSplit < nChunks && nTiles * (nSplit + 1) <= 1920) ++nSplit;
  const int grid = 8 * ((nTiles + 7) / 8) * nSplit;

  auto fopts = h.options().dtype(torch::kFloat32);
  auto logits = torch::empty({P, V}, h.options());
  auto part = torch::empty({P, nTiles, 2}, fopts);
  auto sums = torch::zeros({2}, fopts);  // {loss_sum, count}
  auto lse = torch::empty({P}, fopts);
  auto stream = at::hip::getCurrentHIPStream();

  const size_t lds = 3 * mh::kBufB;
  HIP_CHECK(hipFuncSetAttribute(
      reinterpret_cast<const void*>(&mlm_fwd_kernel),
      hipFuncAttributeMaxDynamicSharedMemorySize, lds));
  hipLaunchKernelGGL(mlm_fwd_kernel, dim3(grid), dim3(256), lds, stream,
                     reinterpret_cast<const __bf16*>(h.data_ptr()),
                     reinterpret_cast<const __bf16*>(w.data_ptr()),
                     bias.data_ptr<float>(),
                     reinterpret_cast<__bf16*>(logits.data_ptr()),
                     part.data_ptr<float>(), P, V, K, nTiles, nSplit);
  hipLaunchKernelGGL(mlm_fold_kernel, dim3((P + 3) / 4), dim3(256), 0,
                     stream, part.data_ptr<float>(),
                     reinterpret_cast<const __bf16*>(logits.data_ptr()),
                     labels_c.data_ptr<int64_t>(), sums.data_ptr<float>(),
                     lse.data_ptr<float>(), P, nTiles, V, ignore_index);
  return {logits, sums.select(0, 0), sums.select(0, 1), lse};
}

}  // namespace bpa
