// Hand-written split-K MFMA weight-gradient GEMM for MI355X (gfx950).
//
// dW[M,N] = dy^T @ x with dy [K,M] and x [K,N] row-major bf16 (the
// "TN" wgrad form of every encoder linear: QKV [3072,1024], attn-out
// [1024,1024], FFN1 [4096,1024], FFN2 [1024,4096] at K = B*S tokens).
// hipBLASLt/rocBLAS top out at ~580-790 TF/s on these shapes even
// under an exhaustive TunableOp search because the output is small:
// a 128^2-tiled launch is only 64-256 workgroups on a 256-CU chip.
// Splitting K across gridDim.z restores occupancy; fp32 partial tiles
// are summed and cast by a tiny combine kernel (fp32 accumulation
// end-to-end - tighter than the library's bf16 epilogue path).
//
// Structure per block (256 threads = 4 waves as 2x2): C tile 128x128,
// K loop in 64-deep steps; both operands staged [64][128] natural
// row-major in LDS (padded stride 144 elems = 72 dwords: the
// transpose-read's {72r*... = 8r + 2c} bank pattern covers every even
// bank exactly once - conflict-free) and their k-strided MFMA
// fragments produced by ds_read_b64_tr_b16 (same frag_tr recipe as
// csrc/ops/attention.hip; semantics probe-verified by tr16_probe).
// Staging is split issue-early/write-late (T14): the next k-step's
// global loads issue before this step's 32 MFMAs per wave, so HBM
// latency hides under the matrix work; double-buffered LDS, one
// barrier pair per k-step.
//
// Reference op being replaced: the implicit wgrad GEMMs of
// src/modeling.py's nn.Linear calls (SURVEY.md §2.4 "Backward of all
// of the above").

#include <torch/extension.h>
#include <cstdlib>
#include <ATen/hip/HIPContext.h>

#include "../common.h"

namespace bpa {

namespace wg {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((__vector_size__(4 * sizeof(__bf16)))) __bf16 bf16x4v;
typedef __attribute__((address_space(3))) bf16x4v* lds_bf16x4p;

#define MFMA16(A, B, C) \
  __builtin_amdgcn_mfma_f32_16x16x32_bf16((A), (B), (C), 0, 0, 0)

constexpr int kBM = 128;   // C tile rows (M)
constexpr int kBN = 128;   // C tile cols (N)
constexpr int kBK = 32;    // K step (32-deep: 36.8 KB LDS -> 4 blocks/CU)
constexpr int kStride = 144;  // LDS row stride (elems), tr-conflict-free

// Transposed fragment from a natural [kBK][cols] row-major LDS image at
// stride kStride: element e = img[rb + g*4 + (e&3) + 16*(e>>2)][cb+li].
__device__ __forceinline__ bf16x8 frag_tr(const __bf16* img, int rb,
                                          int cb) {
  const int lane = threadIdx.x & 63;
  const int g = (lane >> 4) & 3, t = lane & 15;
  const __bf16* p0 =
      img + (rb + g * 4 + (t >> 2)) * kStride + cb + (t & 3) * 4;
  union {
    bf16x8 v;
    bf16x4v h[2];
  } r;
  r.h[0] = __builtin_amdgcn_ds_read_tr16_b64_v4bf16((lds_bf16x4p)(p0));
  r.h[1] = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (lds_bf16x4p)(p0 + 16 * kStride));
  return r.v;
}

}  // namespace wg

using wg::bf16x8;
using wg::f32x4;

// grid: (M/128, N/128, splitk); block 256. BK = 32 or 64: 64 halves
// the barrier count and doubles the MFMA run per staged tile (36.8 ->
// 73.7 KB LDS, 4 -> 2 blocks/CU); which wins is per-shape (sweep in
// benchmarks/wgrad_sweep.py) and picked by the host wrapper.
template <int BK>
__global__ __launch_bounds__(256) void wgrad_tn_kernel(
    const __bf16* __restrict__ dy,  // [K, M]
    const __bf16* __restrict__ x,   // [K, N]
    float* __restrict__ part,       // [splitk, M, N]
    int K, int M, int N, int k_slice) {
  const int m0 = blockIdx.x * wg::kBM;
  const int n0 = blockIdx.y * wg::kBN;
  const int kz0 = blockIdx.z * k_slice;
  const int kz1 = min(K, kz0 + k_slice);
  const int tid = threadIdx.x;
  const int lane = tid & 63, wave = tid >> 6;
  const int wi = wave >> 1, wj = wave & 1;  // 2x2 wave grid, 64x64 each
  const int g = (lane >> 4), li = lane & 15;

  // single LDS base + offset arithmetic: a runtime-indexed POINTER
  // ARRAY here makes hipcc lose the LDS address space and emit
  // flat_store for the staging writes (observed in the .s; 2-3x the
  // ds_write cost plus per-store 64-bit compares)
  extern __shared__ __attribute__((aligned(16))) char smem[];
  __bf16* lds = reinterpret_cast<__bf16*>(smem);
  constexpr int kTile = BK * wg::kStride;
  constexpr int kTPR = 256 / BK;      // staging threads per k-row
  constexpr int kSlots = BK / 16;     // uint4 slots per thread per matrix

  // staging map: thread -> (row r, kSlots 8-elem slots 8*kTPR apart);
  // at BK=32 each ds_write_b128's 8-lane group covers one 128-B row
  // (conflict-free); at BK=64 it covers two half-rows (2-way on half
  // the banks - measured cheaper than the extra barriers it replaces)
  const int st_r = tid / kTPR;
  const int st_c = (tid % kTPR) * 8;

  // Prefetch loads are UNCONDITIONAL with the row clamped to the last
  // valid one (a conditional load/zero branch inside the loop-carried
  // lambda segfaults hipcc/ROCm 7.2's gfx950 optimizer at -O2/-O3);
  // rows past the slice end are zeroed at LDS-write time instead.
  uint4 pf_dy[kSlots], pf_x[kSlots];
  auto issue_loads = [&](int k0) {
    const int krow = min(k0 + st_r, K - 1);  // clamp: always legal memory
    const uint4* ds = reinterpret_cast<const uint4*>(
        dy + static_cast<int64_t>(krow) * M + m0 + st_c);
    const uint4* xs = reinterpret_cast<const uint4*>(
        x + static_cast<int64_t>(krow) * N + n0 + st_c);
#pragma unroll
    for (int q = 0; q < kSlots; ++q) {
      pf_dy[q] = ds[q * kTPR];
      pf_x[q] = xs[q * kTPR];
    }
  };

  f32x4 acc[4][4] = {};
  issue_loads(kz0);

  int buf = 0;
  for (int k0 = kz0; k0 < kz1; k0 += BK) {
    // write the in-flight k-step into buf (T14 write-late); zero rows
    // past the slice end so the clamped prefetch cannot contaminate
    if (k0 + st_r >= kz1) {
#pragma unroll
      for (int q = 0; q < kSlots; ++q) pf_dy[q] = pf_x[q] = uint4{0, 0, 0, 0};
    }
    const int boff = buf * 2 * kTile;
#pragma unroll
    for (int q = 0; q < kSlots; ++q) {
      *reinterpret_cast<uint4*>(
          &lds[boff + st_r * wg::kStride + st_c + 8 * kTPR * q]) = pf_dy[q];
      *reinterpret_cast<uint4*>(
          &lds[boff + kTile + st_r * wg::kStride + st_c + 8 * kTPR * q]) =
          pf_x[q];
    }
    __syncthreads();
    issue_loads(k0 + BK < kz1 ? k0 + BK : k0);  // T14 issue-early

    const __bf16* dyt = lds + boff;
    const __bf16* xt = lds + boff + kTile;
#pragma unroll
    for (int kd = 0; kd < BK / 32; ++kd) {
      bf16x8 af[4], bfr[4];
#pragma unroll
      for (int t = 0; t < 4; ++t)
        af[t] = wg::frag_tr(dyt, kd * 32, wi * 64 + t * 16);
#pragma unroll
      for (int t = 0; t < 4; ++t)
        bfr[t] = wg::frag_tr(xt, kd * 32, wj * 64 + t * 16);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ti = 0; ti < 4; ++ti)
#pragma unroll
        for (int tj = 0; tj < 4; ++tj)
          acc[ti][tj] = MFMA16(af[ti], bfr[tj], acc[ti][tj]);
      __builtin_amdgcn_s_setprio(0);
    }
    // no trailing barrier: the next iteration writes the OTHER buffer,
    // whose last readers finished before this iteration's barrier
    buf ^= 1;
  }

  // epilogue: fp32 partial tile [M,N] for this k-slice
  float* out = part + static_cast<int64_t>(blockIdx.z) * M * N;
#pragma unroll
  for (int ti = 0; ti < 4; ++ti) {
#pragma unroll
    for (int tj = 0; tj < 4; ++tj) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int mi = m0 + wi * 64 + ti * 16 + g * 4 + r;
        const int nj = n0 + wj * 64 + tj * 16 + li;
        out[static_cast<int64_t>(mi) * N + nj] = acc[ti][tj][r];
      }
    }
  }
}

// 256x256-tile variant: 512 threads = 8 waves as 2(M)x4(N), each wave
// a 128x64 C tile (acc[8][4] f32x4 = 128 AGPRs). Halves the operand
// re-read amplification of the 128^2 tile (per-tile strips are shared
// by half as many blocks), trading occupancy (2 waves/SIMD, 1 block/CU)
// for traffic. Selected by the host wrapper when M and N tile by 256.
namespace wg {
constexpr int kStride256 = 272;  // 136 dw: 8r+2c covers even banks once
}

__global__ __launch_bounds__(512) void wgrad_tn256_kernel(
    const __bf16* __restrict__ dy,  // [K, M]
    const __bf16* __restrict__ x,   // [K, N]
    float* __restrict__ part,       // [splitk, M, N]
    int K, int M, int N, int k_slice) {
  const int m0 = blockIdx.x * 256;
  const int n0 = blockIdx.y * 256;
  const int kz0 = blockIdx.z * k_slice;
  const int kz1 = min(K, kz0 + k_slice);
  const int tid = threadIdx.x;
  const int lane = tid & 63, wave = tid >> 6;
  const int wi = wave >> 2, wj = wave & 3;  // 2x4 wave grid: 128x64 each
  const int g = (lane >> 4), li = lane & 15;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  __bf16* lds = reinterpret_cast<__bf16*>(smem);
  constexpr int kTile = wg::kBK * wg::kStride256;

  // staging: 32 rows x 256 cols = 16 bf16 per thread per matrix
  const int st_r = tid >> 4;
  const int st_c = (tid & 15) * 16;

  uint4 pf_dy[2], pf_x[2];
  auto issue_loads = [&](int k0) {
    const int krow = min(k0 + st_r, K - 1);
    const uint4* ds = reinterpret_cast<const uint4*>(
        dy + static_cast<int64_t>(krow) * M + m0 + st_c);
    const uint4* xs = reinterpret_cast<const uint4*>(
        x + static_cast<int64_t>(krow) * N + n0 + st_c);
#pragma unroll
    for (int q = 0; q < 2; ++q) {
      pf_dy[q] = ds[q];
      pf_x[q] = xs[q];
    }
  };

  f32x4 acc[8][4] = {};
  issue_loads(kz0);

  int buf = 0;
  for (int k0 = kz0; k0 < kz1; k0 += wg::kBK) {
    if (k0 + st_r >= kz1) {
#pragma unroll
      for (int q = 0; q < 2; ++q) pf_dy[q] = pf_x[q] = uint4{0, 0, 0, 0};
    }
    const int boff = buf * 2 * kTile;
#pragma unroll
    for (int q = 0; q < 2; ++q) {
      *reinterpret_cast<uint4*>(
          &lds[boff + st_r * wg::kStride256 + st_c + 8 * q]) = pf_dy[q];
      *reinterpret_cast<uint4*>(
          &lds[boff + kTile + st_r * wg::kStride256 + st_c + 8 * q]) =
          pf_x[q];
    }
    __syncthreads();
    issue_loads(k0 + wg::kBK < kz1 ? k0 + wg::kBK : k0);

    const __bf16* dyt = lds + boff;
    const __bf16* xt = lds + boff + kTile;
    {
      bf16x8 bfr[4];
#pragma unroll
      for (int t = 0; t < 4; ++t) {
        const int lane2 = threadIdx.x & 63;
        const int gg = (lane2 >> 4) & 3, tt = lane2 & 15;
        const __bf16* p0 = xt +
            (gg * 4 + (tt >> 2)) * wg::kStride256 + wj * 64 + t * 16 +
            (tt & 3) * 4;
        union {
          bf16x8 v;
          wg::bf16x4v h[2];
        } r;
        r.h[0] =
            __builtin_amdgcn_ds_read_tr16_b64_v4bf16((wg::lds_bf16x4p)(p0));
        r.h[1] = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
            (wg::lds_bf16x4p)(p0 + 16 * wg::kStride256));
        bfr[t] = r.v;
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ti = 0; ti < 8; ++ti) {
        const int lane2 = threadIdx.x & 63;
        const int gg = (lane2 >> 4) & 3, tt = lane2 & 15;
        const __bf16* p0 = dyt +
            (gg * 4 + (tt >> 2)) * wg::kStride256 + wi * 128 + ti * 16 +
            (tt & 3) * 4;
        union {
          bf16x8 v;
          wg::bf16x4v h[2];
        } r;
        r.h[0] =
            __builtin_amdgcn_ds_read_tr16_b64_v4bf16((wg::lds_bf16x4p)(p0));
        r.h[1] = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
            (wg::lds_bf16x4p)(p0 + 16 * wg::kStride256));
        const bf16x8 af = r.v;
#pragma unroll
        for (int tj = 0; tj < 4; ++tj)
          acc[ti][tj] = MFMA16(af, bfr[tj], acc[ti][tj]);
      }
      __builtin_amdgcn_s_setprio(0);
    }
    buf ^= 1;
  }

  float* out = part + static_cast<int64_t>(blockIdx.z) * M * N;
#pragma unroll
  for (int ti = 0; ti < 8; ++ti) {
#pragma unroll
    for (int tj = 0; tj < 4; ++tj) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int mi = m0 + wi * 128 + ti * 16 + g * 4 + r;
        const int nj = n0 + wj * 64 + tj * 16 + li;
        out[static_cast<int64_t>(mi) * N + nj] = acc[ti][tj][r];
      }
    }
  }
}

// part [S, M, N] fp32 -> out [M, N] bf16 (sum over S)
__global__ void wgrad_combine_kernel(const float* __restrict__ part,
                                     __bf16* __restrict__ out, int splitk,
                                     int64_t mn) {
  const int64_t i =
      (static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x) * 4;
  if (i >= mn) return;
  f32x4 s = *reinterpret_cast<const f32x4*>(part + i);
  for (int z = 1; z < splitk; ++z) {
    f32x4 p = *reinterpret_cast<const f32x4*>(
        part + static_cast<int64_t>(z) * mn + i);
#pragma unroll
    for (int k = 0; k < 4; ++k) s[k] += p[k];
  }
  __bf16 o[4];
#pragma unroll
  for (int k = 0; k < 4; ++k) o[k] = __bf16(s[k]);
  *reinterpret_cast<uint2*>(out + i) = *reinterpret_cast<uint2*>(o);
}

bool wgrad_tn_supported(int64_t K, int64_t M, int64_t N) {
  // staging reads whole uint4s per row: K rows need no alignment, but
  // M/N must tile exactly and K must be large enough to amortize
  return M % wg::kBM == 0 && N % wg::kBN == 0 && K % 4 == 0 && K >= 256;
}

bool wgrad_tn_profitable(int64_t K, int64_t M, int64_t N) {
  // where this kernel MEASURED >= hipBLASLt/rocBLAS (gemm_shapes.py):
  // tall-M, N=1024 shapes (QKV/FFN1/attn-out wgrad). N > 1024 (FFN2)
  // and very tall M (MLM decoder) stay on the library.
  return wgrad_tn_supported(K, M, N) && N <= 1024 && M >= 1024 &&
         M <= 8192 && K >= 4096;
}

// dW = dy^T @ x; dy [K,M] bf16 row-major contiguous, x [K,N] likewise.
torch::Tensor wgrad_tn(torch::Tensor dy, torch::Tensor x,
                       int64_t splitk_override) {
  TORCH_CHECK(dy.is_cuda() && dy.dim() == 2 && dy.is_contiguous() &&
                  dy.scalar_type() == torch::kBFloat16,
              "wgrad_tn: dy must be contiguous 2D bf16");
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous() &&
                  x.scalar_type() == torch::kBFloat16,
              "wgrad_tn: x must be contiguous 2D bf16");
  const int K = dy.size(0), M = dy.size(1), N = x.size(1);
  TORCH_CHECK(x.size(0) == K, "wgrad_tn: K mismatch");
  TORCH_CHECK(wgrad_tn_supported(K, M, N), "wgrad_tn: unsupported shape");
  const bool big = false;  // 256-tile variant measured SLOWER (1 block/CU cannot hide latency); kept for reference
  const int bm = big ? 256 : wg::kBM, bn = big ? 256 : wg::kBN;
  const int tiles = (M / bm) * (N / bn);
  // k-step depth: 64 halves barriers / doubles the MFMA run per tile
  // at 2 blocks/CU; override with BPA_WGRAD_BK for sweeps
  const char* env_bk = getenv("BPA_WGRAD_BK");  // per-call: sweepable
  int bk = env_bk ? atoi(env_bk) : 32;
  TORCH_CHECK(bk == 32 || bk == 64, "wgrad_tn: BK must be 32 or 64");
  // split-K sweep on MI355X (benchmarks/wgrad_sweep.py): best wall time
  // lands at tiles*splitk ~ 512-768 with splitk <= 8 (beyond that the
  // fp32 slab traffic of the combine outweighs the occupancy gain):
  // QKV 127->99us (splitk 4), attn-out 56->42 (8), FFN1 ~tie (3)
  int splitk = 1;
  while (splitk < 8 && tiles * (splitk + 1) <= 768 &&
         K / (splitk + 1) >= 8 * wg::kBK)
    ++splitk;
  if (splitk_override > 0) splitk = static_cast<int>(splitk_override);
  int k_slice = ((K + splitk - 1) / splitk + bk - 1) / bk * bk;
  splitk = (K + k_slice - 1) / k_slice;  // drop empty tail slices

  auto part = torch::empty({splitk, static_cast<int64_t>(M), N},
                           dy.options().dtype(torch::kFloat32));
  auto out = torch::empty({static_cast<int64_t>(M), N}, dy.options());
  auto stream = at::hip::getCurrentHIPStream();
  dim3 grid(M / bm, N / bn, splitk), block(big ? 512 : 256);
  const size_t lds =
      4 * bk * (big ? wg::kStride256 : wg::kStride) * sizeof(__bf16);
  if (big) {
    HIP_CHECK(hipFuncSetAttribute(
        reinterpret_cast<const void*>(&wgrad_tn256_kernel),
        hipFuncAttributeMaxDynamicSharedMemorySize, lds));
    hipLaunchKernelGGL(wgrad_tn256_kernel, grid, block, lds, stream,
                       reinterpret_cast<const __bf16*>(dy.data_ptr()),
                       reinterpret_cast<const __bf16*>(x.data_ptr()),
                       part.data_ptr<float>(), K, M, N, k_slice);
  } else if (bk == 64) {
    HIP_CHECK(hipFuncSetAttribute(
        reinterpret_cast<const void*>(&wgrad_tn_kernel<64>),
        hipFuncAttributeMaxDynamicSharedMemorySize, lds));
    hipLaunchKernelGGL(wgrad_tn_kernel<64>, grid, block, lds, stream,
                       reinterpret_cast<const __bf16*>(dy.data_ptr()),
                       reinterpret_cast<const __bf16*>(x.data_ptr()),
                       part.data_ptr<float>(), K, M, N, k_slice);
  } else {
    HIP_CHECK(hipFuncSetAttribute(
        reinterpret_cast<const void*>(&wgrad_tn_kernel<32>),
        hipFuncAttributeMaxDynamicSharedMemorySize, lds));
    hipLaunchKernelGGL(wgrad_tn_kernel<32>, grid, block, lds, stream,
                       reinterpret_cast<const __bf16*>(dy.data_ptr()),
                       reinterpret_cast<const __bf16*>(x.data_ptr()),
                       part.data_ptr<float>(), K, M, N, k_slice);
  }
  const int64_t mn = static_cast<int64_t>(M) * N;
  hipLaunchKernelGGL(wgrad_combine_kernel, dim3((mn / 4 + 255) / 256),
                     dim3(256), 0, stream, part.data_ptr<float>(),
                     reinterpret_cast<__bf16*>(out.data_ptr()), splitk, mn);
  return out;
}

}  // namespace bpa
