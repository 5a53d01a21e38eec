// Fused scaled-dot-product attention (flash-style) for MI355X (gfx950).
//
// Replaces the reference's 5-op attention path (QK^T, +mask, softmax,
// dropout, PV — src/modeling.py:403-429) with MFMA kernels built on
// v_mfma_f32_16x16x32_bf16:
//
// Forward: grid (S/128 q-tiles, B*heads); 4 waves/block, 2x16 q-rows
// per wave. K and V staged NATURALLY in LDS (K at a 72-elem stride for
// conflict-free b64 row reads; V at an 80-elem stride for conflict-free
// ds_read_b64_tr_b16 hardware-transpose fragment reads - no transposed
// image is ever built); online softmax with the swapped-QK^T trick
// (S^T = mfma(K, Q)) so each lane's P scores chain directly into the
// PV A-fragment; the padding mask enters as per-sequence valid
// lengths. Dropout keep-masks
// are pre-generated once per forward by dropout_mask_kernel (Philox
// 4x32-10, one BIT per score, rows padded to whole 32-bit words) and
// the [B*NH, S, ceil(S/32)] word tensor is read by forward and both
// backward kernels — measurably cheaper than regenerating the RNG
// stream in each of the three consumers, and 8x less mask traffic
// than the byte-mask variant.
//
// Backward (flash-2 style, atomic-free): two MFMA kernels.
// * attn_bwd_kernel: grid (S/64 kv-tiles, B*heads); each block owns one
//   K/V tile, loops over q-tiles; recomputes P from the saved
//   logsumexp; accumulates dK/dV in registers, written once,
//   exclusively, straight into dqkv.
// * attn_dq_kernel: grid (S/64 q-tiles, B*heads); mirrors the forward's
//   wave-owns-q structure, recomputes S^T/dP^T per K/V tile and chains
//   dS^T fragments into dQ += dS.K MFMAs against K^T staged in LDS —
//   dQ lands in registers and is written once (the previous design
//   pushed ~1k fp32 atomicAdds per wave per q-iter into an fp32
//   workspace plus a pack kernel; that was the backward bottleneck).
//
// Fragment convention (consistent A/B slot mapping):
//   A[i][k]/B[k][j]: i|j = lane&15, k = (lane>>4)*4 + (e&3) + 16*(e>>2)
//   C/D[i][j]:       j = lane&15,   i = (lane>>4)*4 + reg
// Only A/B consistency matters for correctness (the contraction is
// permutation-invariant when both sides agree); C/D is the documented
// gfx950 layout.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <cmath>
#include <utility>

#include "../common.h"

namespace bpa {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __hip_bfloat16 bf16_t;

#define MFMA16(A, B, C) __builtin_amdgcn_mfma_f32_16x16x32_bf16((A), (B), (C), 0, 0, 0)

constexpr int kPad = 8;        // bf16 row pad: 72-element stride, bank-clean
constexpr int kStride = 64 + kPad;
// Stride for LDS images read with ds_read_b64_tr_b16: 80 elements (40
// dwords) makes the hardware transpose read conflict-free within its
// 32-lane groups ({40r+2c} for r 0..7, c 0..3 covers every even bank
// exactly once). Row-fragment (b64) reads of the same image are 2-way
// (rows r and r+8 share a bank) - acceptable where it saves a whole
// second image (no stride satisfies both patterns: rows need
// gcd(stride_dw,64)=4, tr needs gcd=8).
constexpr int kTrStride = 80;

// Build an A/B fragment from a row-major bf16 row pointer: elements at
// k = base + g*4 + (e&3) + 16*(e>>2), loaded as two 8-byte chunks.
__device__ __forceinline__ bf16x8 frag_row(const __bf16* row, int base,
                                           int g) {
  union {
    bf16x8 v;
    uint2 u[2];
  } r;
  r.u[0] = *reinterpret_cast<const uint2*>(row + base + g * 4);
  r.u[1] = *reinterpret_cast<const uint2*>(row + base + 16 + g * 4);
  return r.v;
}

typedef __attribute__((__vector_size__(4 * sizeof(__bf16)))) __bf16 bf16x4v;
typedef __attribute__((address_space(3))) bf16x4v* lds_bf16x4p;

// Transposed A/B fragment via gfx950's hardware transpose read, straight
// from a NATURAL row-major [rows][kTrStride] bf16 LDS image: element
// e = M[rb + g*4 + (e&3) + 16*(e>>2)][cb + li] - i.e. the fragment the
// old code read from a separately-built transposed image. Semantics
// probe-verified on hardware (tr16_probe): within each 16-lane group,
// out[lane i][j] = mem[addr_of_lane(4j + (i>>2)) + (i&3)], so lane
// (g, t) points at 4 contiguous elements of row rb + g*4 + (t>>2) and
// the instruction delivers the group-transposed fragment. Replaces the
// per-thread 16x ds_write_b16 transpose scatter at staging time.
__device__ __forceinline__ bf16x8 frag_tr(const __bf16* img, int rb,
                                          int cb) {
  const int lane = threadIdx.x & 63;
  const int g = (lane >> 4) & 3, t = lane & 15;
  const __bf16* p0 =
      img + (rb + g * 4 + (t >> 2)) * kTrStride + cb + (t & 3) * 4;
  union {
    bf16x8 v;
    bf16x4v h[2];
  } r;
  r.h[0] = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (lds_bf16x4p)(p0));
  r.h[1] = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (lds_bf16x4p)(p0 + 16 * kTrStride));
  return r.v;
}

// Dropout keep-mask generation, BIT-packed: one keep decision per BIT
// (LSB-first within each uint32 word), one thread per word. Each of
// the eight 32-bit Philox outputs (two calls) yields four decisions by
// comparing its bytes against an 8-bit threshold t = round(p*256).
// The host quantizes p to t/256 first (quantize_drop_p) and every
// consumer's 1/(1-p) uses the SAME quantized p, so dropout stays
// exactly unbiased; the ~2^-8 quantization of the drop probability
// itself (0.1 -> 0.1016) is training-irrelevant. History: the original
// one-Philox-per-4-BYTES version was VALU-bound at ~1.4 TB/s and 2.9%
// of phase-2 GPU time; bytes->bits also cuts the mask footprint and
// the three consumers' read traffic 8x (24 layers x [B*NH,S,S] masks
// alive per micro-batch = 1.6 GB at phase 2 as bytes, 200 MB as bits).
// Rows are padded to whole words (stride Sw = ceil(S/32)) so no quad
// ever straddles a word. Counter for word i is offset + 2i (+1),
// deterministic in (seed, offset). Running this as its own elementwise
// kernel keeps the MFMA kernels free of RNG VALU work.
__global__ void dropout_mask_kernel(uint32_t* __restrict__ mask,
                                    int64_t words, uint32_t thresh,
                                    uint64_t seed, uint64_t offset) {
  const int64_t i =
      static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x;
  if (i >= words) return;
  Philox philox(seed);
  uint32_t r[8];
  philox(offset + 2 * i, r);
  philox(offset + 2 * i + 1, r + 4);
  uint32_t bits = 0;
#pragma unroll
  for (int j = 0; j < 32; ++j)
    bits |= (((r[j >> 2] >> (8 * (j & 3))) & 0xFFu) >= thresh ? 1u : 0u)
            << j;
  mask[i] = bits;
}

// Quantize a dropout probability to the 8-bit threshold grid the mask
// kernel samples on. Both the threshold and every keep-scale are
// derived from the returned pair so the expectation is exact.
static inline std::pair<uint32_t, float> quantize_drop_p(double p) {
  uint32_t t = static_cast<uint32_t>(std::lround(p * 256.0));
  if (t > 255) t = 255;
  return {t, static_cast<float>(t) / 256.0f};
}

// ---------------------------------------------------------------------------
// forward
// ---------------------------------------------------------------------------
template <bool TRAIN_DROP>
__global__ __launch_bounds__(256) void attn_fwd_kernel(
    const __bf16* __restrict__ qkv, const int* __restrict__ seqlens,
    __bf16* __restrict__ out, float* __restrict__ lse_out,
    const uint32_t* __restrict__ dmask, int B, int S, int NH, float p,
    float scale, uint64_t seed, uint64_t offset) {
  // 4 waves x 32 q-rows (two 16-row subtiles per wave): the K/V tile is
  // staged once per 128 q-rows, each K/V fragment read feeds TWO
  // independent MFMA chains, and the softmax work of the two subtiles
  // interleaves - targeting the measured issue-stall/parked time.
  const int bh = blockIdx.y;
  const int b = bh / NH, h = bh % NH;
  const int q0 = blockIdx.x * 128;
  const int tid = threadIdx.x;
  const int lane = tid & 63, wave = tid >> 6;
  const int g = (lane >> 4), li = lane & 15;
  const int H = NH * 64;
  const int rs3 = 3 * H;  // qkv row stride
  const __bf16* qbase = qkv + static_cast<int64_t>(b) * S * rs3 + h * 64;
  const __bf16* kbase = qbase + H;
  const __bf16* vbase = qbase + 2 * H;
  const int slen = seqlens[b];
  const float inv_keep = TRAIN_DROP ? 1.f / (1.f - p) : 1.f;
  const int Sw = (S + 31) >> 5;  // mask row stride in words
  const uint32_t* mask_base =
      TRAIN_DROP ? dmask + static_cast<int64_t>(bh) * S * Sw : nullptr;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  __bf16* K_lds = reinterpret_cast<__bf16*>(smem);             // [64][72]
  __bf16* V_lds = K_lds + 64 * kStride;   // [64][80] natural, tr-read
  float* alpha_lds = reinterpret_cast<float*>(V_lds + 64 * kTrStride);  // [4][2][16]
  float* stat_lds = alpha_lds + 4 * 2 * 16;                    // [4][2][16]
  // keep-mask words for the current kv-tile, [128 q][2 words]: staged
  // coalesced (one word per thread) instead of per-lane scattered 4-byte
  // global reads in the softmax loop
  uint32_t* mk_lds = reinterpret_cast<uint32_t*>(stat_lds + 4 * 2 * 16);

  // this wave's two q-subtiles: rows q0 + wave*32 + sub*16 + li
  int q_row[2];
  bf16x8 qfrag[2][2];
#pragma unroll
  for (int sub = 0; sub < 2; ++sub) {
    q_row[sub] = q0 + wave * 32 + sub * 16 + li;
    const int q_ld = min(q_row[sub], S - 1);
#pragma unroll
    for (int c = 0; c < 2; ++c)
      qfrag[sub][c] =
          frag_row(qbase + static_cast<int64_t>(q_ld) * rs3, 32 * c, g);
  }

  float m_run[2] = {-1e30f, -1e30f}, l_run[2] = {0.f, 0.f};
  f32x4 acc_o[2][4] = {};

  const int n_kv = (S + 63) / 64;
  for (int kt = 0; kt < n_kv; ++kt) {
    const int k0 = kt * 64;
    __syncthreads();
    // stage K natural + V transposed; 4 threads per row, 16 cols each
    {
      const int row = tid >> 2, colc = (tid & 3) * 16;
      const int krow = k0 + row;
      if (krow < S) {
        const uint4* src =
            reinterpret_cast<const uint4*>(kbase + static_cast<int64_t>(krow) * rs3 + colc);
        *reinterpret_cast<uint4*>(&K_lds[row * kStride + colc]) = src[0];
        *reinterpret_cast<uint4*>(&K_lds[row * kStride + colc + 8]) = src[1];
        const uint4* vsrc =
            reinterpret_cast<const uint4*>(vbase + static_cast<int64_t>(krow) * rs3 + colc);
        *reinterpret_cast<uint4*>(&V_lds[row * kTrStride + colc]) = vsrc[0];
        *reinterpret_cast<uint4*>(&V_lds[row * kTrStride + colc + 8]) = vsrc[1];
      } else {
        uint4 zero{0, 0, 0, 0};
        *reinterpret_cast<uint4*>(&K_lds[row * kStride + colc]) = zero;
        *reinterpret_cast<uint4*>(&K_lds[row * kStride + colc + 8]) = zero;
        *reinterpret_cast<uint4*>(&V_lds[row * kTrStride + colc]) = zero;
        *reinterpret_cast<uint4*>(&V_lds[row * kTrStride + colc + 8]) = zero;
      }
      if (TRAIN_DROP) {
        const int qrow_m = q0 + (tid >> 1);
        const int kw = (k0 >> 5) + (tid & 1);  // 2 words cover 64 keys
        uint32_t bits = 0xFFFFFFFFu;
        if (qrow_m < S && kw < Sw)
          bits = mask_base[static_cast<int64_t>(qrow_m) * Sw + kw];
        mk_lds[tid] = bits;
      }
    }
    __syncthreads();

    // S^T tiles: one K-fragment pair feeds both subtiles' MFMA chains
    f32x4 s_acc[2][4];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int t = 0; t < 4; ++t) {
      const __bf16* krow = &K_lds[(t * 16 + li) * kStride];
      const bf16x8 kf0 = frag_row(krow, 0, g);
      const bf16x8 kf1 = frag_row(krow, 32, g);
      f32x4 a0 = {}, a1 = {};
      a0 = MFMA16(kf0, qfrag[0][0], a0);
      a1 = MFMA16(kf0, qfrag[1][0], a1);
      a0 = MFMA16(kf1, qfrag[0][1], a0);
      a1 = MFMA16(kf1, qfrag[1][1], a1);
      s_acc[0][t] = a0;
      s_acc[1][t] = a1;
    }
    __builtin_amdgcn_s_setprio(0);

    float sv[2][4][4];
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      float tmax = -1e30f;
#pragma unroll
      for (int t = 0; t < 4; ++t) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int key = k0 + t * 16 + g * 4 + r;
          const bool valid = key < slen && key < S;
          sv[sub][t][r] = valid ? s_acc[sub][t][r] * scale : -1e30f;
          tmax = fmaxf(tmax, sv[sub][t][r]);
        }
      }
      tmax = fmaxf(tmax, __shfl_xor(tmax, 16, 64));
      tmax = fmaxf(tmax, __shfl_xor(tmax, 32, 64));
      const float m_new = fmaxf(m_run[sub], tmax);
      const float alpha = __expf(m_run[sub] - m_new);
      float rowsum = 0.f;
#pragma unroll
      for (int t = 0; t < 4; ++t)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          sv[sub][t][r] = __expf(sv[sub][t][r] - m_new);
          rowsum += sv[sub][t][r];
        }
      rowsum += __shfl_xor(rowsum, 16, 64);
      rowsum += __shfl_xor(rowsum, 32, 64);
      l_run[sub] = l_run[sub] * alpha + rowsum;
      m_run[sub] = m_new;
      if (g == 0) alpha_lds[(wave * 2 + sub) * 16 + li] = alpha;
    }
    __builtin_amdgcn_s_waitcnt(0);  // lgkmcnt(0): same-wave LDS visibility
#pragma unroll
    for (int sub = 0; sub < 2; ++sub)
#pragma unroll
      for (int n = 0; n < 4; ++n)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          acc_o[sub][n][r] *= alpha_lds[(wave * 2 + sub) * 16 + g * 4 + r];

    if (TRAIN_DROP) {
      // keep-mask pre-generated by dropout_mask_kernel (no RNG VALU
      // work in the MFMA kernels). k0 is a multiple of 64, so TWO
      // 32-bit words cover this kv-tile's keys for the whole t-loop -
      // one b64 LDS read per subtile replaces per-lane global loads.
#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
        const int q_loc = wave * 32 + sub * 16 + li;
        uint32_t mw[2];
        *reinterpret_cast<uint2*>(mw) =
            *reinterpret_cast<const uint2*>(&mk_lds[q_loc * 2]);
#pragma unroll
        for (int t = 0; t < 4; ++t) {
          const int kk = t * 16 + g * 4;  // key offset within the tile
          const uint32_t nib = (mw[kk >> 5] >> (kk & 31)) & 0xF;
#pragma unroll
          for (int j = 0; j < 4; ++j)
            sv[sub][t][j] = (nib >> j) & 1 ? sv[sub][t][j] * inv_keep : 0.f;
        }
      }
    }

    // P fragments chain from sv; one V^T fragment feeds both subtiles
    bf16x8 pa[2][2];
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        union {
          bf16x8 v;
          __bf16 e[8];
        } pk;
#pragma unroll
        for (int e = 0; e < 8; ++e)
          pk.e[e] = __bf16(sv[sub][2 * c + (e >> 2)][e & 3]);
        pa[sub][c] = pk.v;
      }
    }
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      const bf16x8 vf0 = frag_tr(V_lds, 0, n * 16);
      const bf16x8 vf1 = frag_tr(V_lds, 32, n * 16);
      acc_o[0][n] = MFMA16(pa[0][0], vf0, acc_o[0][n]);
      acc_o[1][n] = MFMA16(pa[1][0], vf0, acc_o[1][n]);
      acc_o[0][n] = MFMA16(pa[0][1], vf1, acc_o[0][n]);
      acc_o[1][n] = MFMA16(pa[1][1], vf1, acc_o[1][n]);
    }
    __builtin_amdgcn_s_setprio(0);
  }

  // epilogue: normalize rows by l (broadcast per-wave through LDS)
#pragma unroll
  for (int sub = 0; sub < 2; ++sub)
    if (g == 0) stat_lds[(wave * 2 + sub) * 16 + li] = l_run[sub];
  __builtin_amdgcn_s_waitcnt(0);
#pragma unroll
  for (int sub = 0; sub < 2; ++sub) {
#pragma unroll
    for (int n = 0; n < 4; ++n) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qr = q0 + wave * 32 + sub * 16 + g * 4 + r;
        if (qr < S) {
          const float l = stat_lds[(wave * 2 + sub) * 16 + g * 4 + r];
          const float o = acc_o[sub][n][r] / (l > 0.f ? l : 1.f);
          out[(static_cast<int64_t>(b) * S + qr) * H + h * 64 + n * 16 + li] =
              __bf16(o);
        }
      }
    }
    if (g == 0 && q_row[sub] < S)
      lse_out[static_cast<int64_t>(bh) * S + q_row[sub]] =
          m_run[sub] + __logf(l_run[sub] > 0.f ? l_run[sub] : 1.f);
  }
}

// ---------------------------------------------------------------------------
// backward
// ---------------------------------------------------------------------------
// delta[bh][q] = rowsum(dO * O) over the 64-wide head dim.
// 8 rows per wave: lane l covers row l/8, elements (l%8)*8..+7 as one
// 16-byte load from each of dO and O — coalesced (the 8 lanes of a row
// touch 128 contiguous bytes), and the row sum needs only 3 xor-shuffle
// rounds within the 8-lane group (the old wave-per-row version was
// reduce-latency-bound at 0.17 TB/s).
__global__ void attn_delta_kernel(const __bf16* __restrict__ dout,
                                  const __bf16* __restrict__ out,
                                  float* __restrict__ delta, int B, int S,
                                  int NH) {
  const int H = NH * 64;
  const int64_t rows = static_cast<int64_t>(B) * NH * S;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int sub = lane >> 3;          // 8-row group index in wave
  const int part = lane & 7;          // 8-element slice within the row
  const int64_t row =
      (static_cast<int64_t>(blockIdx.x) * (blockDim.x >> 6) + wave) * 8 + sub;
  if (row >= rows) return;
  const int64_t bh = row / S;
  const int q = static_cast<int>(row % S);
  const int b = static_cast<int>(bh) / NH, h = static_cast<int>(bh) % NH;
  const int64_t base =
      (static_cast<int64_t>(b) * S + q) * H + h * 64 + part * 8;
  __bf16 dv[8], ov[8];
  *reinterpret_cast<uint4*>(dv) = *reinterpret_cast<const uint4*>(dout + base);
  *reinterpret_cast<uint4*>(ov) = *reinterpret_cast<const uint4*>(out + base);
  float acc = 0.f;
#pragma unroll
  for (int k = 0; k < 8; ++k)
    acc += __bfloat162float(dv[k]) * __bfloat162float(ov[k]);
#pragma unroll
  for (int off = 4; off > 0; off >>= 1) acc += __shfl_xor(acc, off, WAVE_SIZE);
  if (part == 0) delta[row] = acc;
}

// dQ kernel: wave-owns-q structure copied from attn_fwd_kernel.
// Per K/V tile: recompute S^T = K.Q^T and dP^T = V.dO^T via MFMA,
// P = exp(S*scale - lse), dS^T = P*(dP - delta)*scale, then chain dS^T
// lane registers into A-fragments for dQ += dS.K against K^T in LDS
// (exactly how the forward chains P into the PV product). dQ stays in
// registers until the single epilogue store into dqkv's Q slots.
template <bool TRAIN_DROP>
__global__ __launch_bounds__(256) void attn_dq_kernel(
    const __bf16* __restrict__ dout, const __bf16* __restrict__ qkv,
    const int* __restrict__ seqlens, const float* __restrict__ lse,
    const float* __restrict__ delta, const uint32_t* __restrict__ dmask,
    __bf16* __restrict__ dqkv, int B, int S, int NH, float p, float scale,
    uint64_t seed, uint64_t offset) {
  // 4 waves x 32 q-rows (two 16-row subtiles per wave), mirroring
  // attn_fwd_kernel: K/V/K^T staged once per 128 q-rows, every staged
  // fragment feeds two independent MFMA chains.
  const int bh = blockIdx.y;
  const int b = bh / NH, h = bh % NH;
  const int q0 = blockIdx.x * 128;
  const int tid = threadIdx.x;
  const int lane = tid & 63, wave = tid >> 6;
  const int g = (lane >> 4), li = lane & 15;
  const int H = NH * 64;
  const int rs3 = 3 * H;
  const __bf16* qbase = qkv + static_cast<int64_t>(b) * S * rs3 + h * 64;
  const __bf16* kbase = qbase + H;
  const __bf16* vbase = qbase + 2 * H;
  const __bf16* dobase = dout + static_cast<int64_t>(b) * S * H + h * 64;
  const int slen = seqlens[b];
  const float inv_keep = TRAIN_DROP ? 1.f / (1.f - p) : 1.f;
  const int Sw = (S + 31) >> 5;  // mask row stride in words
  const uint32_t* mask_base =
      TRAIN_DROP ? dmask + static_cast<int64_t>(bh) * S * Sw : nullptr;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  // K natural at kTrStride: serves BOTH the S^T-recompute row fragments
  // (2-way bank conflicts, tolerated) and the dQ chain's K^T fragments
  // via hardware transpose reads - no separate transposed image.
  __bf16* K_lds = reinterpret_cast<__bf16*>(smem);   // [64][80] natural
  __bf16* V_lds = K_lds + 64 * kTrStride;            // [64][72] natural
  uint32_t* mk_lds = reinterpret_cast<uint32_t*>(V_lds + 64 * kStride);

  // this wave's two q-subtiles: Q/dO fragments + lse/delta in registers
  int q_row[2];
  bf16x8 qfrag[2][2], dofrag[2][2];
  float lse_q[2], dlt_q[2];
#pragma unroll
  for (int sub = 0; sub < 2; ++sub) {
    q_row[sub] = q0 + wave * 32 + sub * 16 + li;
    const int q_ld = min(q_row[sub], S - 1);
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      qfrag[sub][c] =
          frag_row(qbase + static_cast<int64_t>(q_ld) * rs3, 32 * c, g);
      dofrag[sub][c] =
          frag_row(dobase + static_cast<int64_t>(q_ld) * H, 32 * c, g);
    }
    lse_q[sub] =
        (q_row[sub] < S) ? lse[static_cast<int64_t>(bh) * S + q_row[sub]] : 0.f;
    dlt_q[sub] =
        (q_row[sub] < S) ? delta[static_cast<int64_t>(bh) * S + q_row[sub]] : 0.f;
  }

  f32x4 acc_dq[2][4] = {};

  const int n_kv = (S + 63) / 64;
  for (int kt = 0; kt < n_kv; ++kt) {
    const int k0 = kt * 64;
    __syncthreads();
    {
      const int row = tid >> 2, colc = (tid & 3) * 16;
      const int krow = k0 + row;
      if (krow < S) {
        const uint4* ks = reinterpret_cast<const uint4*>(
            kbase + static_cast<int64_t>(krow) * rs3 + colc);
        *reinterpret_cast<uint4*>(&K_lds[row * kTrStride + colc]) = ks[0];
        *reinterpret_cast<uint4*>(&K_lds[row * kTrStride + colc + 8]) = ks[1];
        const uint4* vs = reinterpret_cast<const uint4*>(
            vbase + static_cast<int64_t>(krow) * rs3 + colc);
        *reinterpret_cast<uint4*>(&V_lds[row * kStride + colc]) = vs[0];
        *reinterpret_cast<uint4*>(&V_lds[row * kStride + colc + 8]) = vs[1];
      } else {
        uint4 zero{0, 0, 0, 0};
        *reinterpret_cast<uint4*>(&K_lds[row * kTrStride + colc]) = zero;
        *reinterpret_cast<uint4*>(&K_lds[row * kTrStride + colc + 8]) = zero;
        *reinterpret_cast<uint4*>(&V_lds[row * kStride + colc]) = zero;
        *reinterpret_cast<uint4*>(&V_lds[row * kStride + colc + 8]) = zero;
      }
      if (TRAIN_DROP) {
        const int qrow_m = q0 + (tid >> 1);
        const int kw = (k0 >> 5) + (tid & 1);
        uint32_t bits = 0xFFFFFFFFu;
        if (qrow_m < S && kw < Sw)
          bits = mask_base[static_cast<int64_t>(qrow_m) * Sw + kw];
        mk_lds[tid] = bits;
      }
    }
    __syncthreads();

    // S^T and dP^T tiles (C[key][q], q = li); K/V fragments shared.
    // dS packs straight into its A-fragment slots (element
    // e = (t&1)*4 + r of chunk t>>1) - no [2][4][4] staging array.
    union FB {
      bf16x8 v;
      __bf16 e[8];
    };
    FB dsfrag[2][2];
    // (setprio brackets are inside the loops below)
    // keep-mask words for this kv-tile (two words cover keys
    // k0..k0+63 for each q-subtile; k0 is a multiple of 64)
    uint32_t mw[2][2] = {{0xFFFFFFFFu, 0xFFFFFFFFu},
                         {0xFFFFFFFFu, 0xFFFFFFFFu}};
    if (TRAIN_DROP) {
#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
        const int q_loc = wave * 32 + sub * 16 + li;
        *reinterpret_cast<uint2*>(mw[sub]) =
            *reinterpret_cast<const uint2*>(&mk_lds[q_loc * 2]);
      }
    }
#pragma unroll
    for (int t = 0; t < 4; ++t) {
      const __bf16* krow = &K_lds[(t * 16 + li) * kTrStride];
      const __bf16* vrow = &V_lds[(t * 16 + li) * kStride];
      const bf16x8 kf0 = frag_row(krow, 0, g), kf1 = frag_row(krow, 32, g);
      const bf16x8 vf0 = frag_row(vrow, 0, g), vf1 = frag_row(vrow, 32, g);
      f32x4 sacc[2] = {}, dpacc[2] = {};
#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
        sacc[sub] = MFMA16(kf0, qfrag[sub][0], sacc[sub]);
        sacc[sub] = MFMA16(kf1, qfrag[sub][1], sacc[sub]);
        dpacc[sub] = MFMA16(vf0, dofrag[sub][0], dpacc[sub]);
        dpacc[sub] = MFMA16(vf1, dofrag[sub][1], dpacc[sub]);
      }
#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
        const int kk = t * 16 + g * 4;  // key offset within the tile
        const uint32_t nib = (mw[sub][kk >> 5] >> (kk & 31)) & 0xF;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int key = k0 + t * 16 + g * 4 + r;
          const bool valid = key < slen && key < S && q_row[sub] < S;
          const float pr =
              valid ? __expf(sacc[sub][r] * scale - lse_q[sub]) : 0.f;
          float dpd = dpacc[sub][r];
          if (TRAIN_DROP) {
            dpd = (nib >> r) & 1 ? dpd * inv_keep : 0.f;
          }
          dsfrag[sub][t >> 1].e[(t & 1) * 4 + r] =
              __bf16(pr * (dpd - dlt_q[sub]) * scale);
        }
      }
    }
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      const bf16x8 ktf0 = frag_tr(K_lds, 0, n * 16);
      const bf16x8 ktf1 = frag_tr(K_lds, 32, n * 16);
      acc_dq[0][n] = MFMA16(dsfrag[0][0].v, ktf0, acc_dq[0][n]);
      acc_dq[1][n] = MFMA16(dsfrag[1][0].v, ktf0, acc_dq[1][n]);
      acc_dq[0][n] = MFMA16(dsfrag[0][1].v, ktf1, acc_dq[0][n]);
      acc_dq[1][n] = MFMA16(dsfrag[1][1].v, ktf1, acc_dq[1][n]);
    }
    __builtin_amdgcn_s_setprio(0);
  }

  // epilogue: one store per element into dqkv Q slots
#pragma unroll
  for (int sub = 0; sub < 2; ++sub) {
#pragma unroll
    for (int n = 0; n < 4; ++n) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qr = q0 + wave * 32 + sub * 16 + g * 4 + r;
        if (qr < S) {
          dqkv[(static_cast<int64_t>(b) * S + qr) * rs3 + h * 64 + n * 16 +
               li] = __bf16(acc_dq[sub][n][r]);
        }
      }
    }
  }
}

template <bool TRAIN_DROP>
__global__ __launch_bounds__(256) void attn_bwd_kernel(
    const __bf16* __restrict__ dout, const __bf16* __restrict__ qkv,
    const int* __restrict__ seqlens, const float* __restrict__ lse,
    const float* __restrict__ delta, const uint32_t* __restrict__ dmask,
    __bf16* __restrict__ dqkv, int B, int S, int NH, float p, float scale,
    uint64_t seed, uint64_t offset) {
  // 4 waves x 32 keys (two 16-key subtiles per wave): the heavy per-
  // q-tile staging of Q/Q^T/dO/dO^T is amortized over 128 keys, and
  // every staged Q/dO/Q^T/dO^T fragment feeds two MFMA chains.
  const int bh = blockIdx.y;
  const int b = bh / NH, h = bh % NH;
  const int k0 = blockIdx.x * 128;
  const int tid = threadIdx.x;
  const int lane = tid & 63, wave = tid >> 6;
  const int g = (lane >> 4), li = lane & 15;
  const int H = NH * 64;
  const int rs3 = 3 * H;
  const __bf16* qbase = qkv + static_cast<int64_t>(b) * S * rs3 + h * 64;
  const __bf16* kbase = qbase + H;
  const __bf16* vbase = qbase + 2 * H;
  const __bf16* dobase = dout + static_cast<int64_t>(b) * S * H + h * 64;
  const int slen = seqlens[b];
  const float inv_keep = TRAIN_DROP ? 1.f / (1.f - p) : 1.f;
  const int Sw = (S + 31) >> 5;  // mask row stride in words
  const uint32_t* mask_base =
      TRAIN_DROP ? dmask + static_cast<int64_t>(bh) * S * Sw : nullptr;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  // Q and dO at kTrStride serve BOTH row fragments (S/dP recompute;
  // 2-way conflicts) and the dK/dV chains' transposed fragments via
  // hardware transpose reads - the Qt/dOt images and their 16-scalar-
  // write-per-thread transpose scatters are gone.
  __bf16* K_lds = reinterpret_cast<__bf16*>(smem);   // [128][72] natural
  __bf16* V_lds = K_lds + 128 * kStride;             // [128][72] natural
  __bf16* Q_lds = V_lds + 128 * kStride;             // [64][80] natural
  __bf16* dO_lds = Q_lds + 64 * kTrStride;           // [64][80] natural
  float* lse_lds = reinterpret_cast<float*>(dO_lds + 64 * kTrStride);  // [64]
  float* dlt_lds = lse_lds + 64;                                      // [64]
  // keep-mask tile [64 q][128 keys] as bits, [4 words][64 q], staged per
  // q-tile (raw global reads were byte columns - latency-bound at this
  // kernel's occupancy; 16 lanes share each word via LDS broadcast)
  uint32_t* mk_lds = reinterpret_cast<uint32_t*>(dlt_lds + 64);

  // stage K and V (natural) once: two 64-row passes
#pragma unroll
  for (int pass = 0; pass < 2; ++pass) {
    const int row = pass * 64 + (tid >> 2), colc = (tid & 3) * 16;
    const int krow = k0 + row;
    __bf16 kv[16], vv[16];
    if (krow < S) {
      const uint4* ks =
          reinterpret_cast<const uint4*>(kbase + static_cast<int64_t>(krow) * rs3 + colc);
      *reinterpret_cast<uint4*>(&kv[0]) = ks[0];
      *reinterpret_cast<uint4*>(&kv[8]) = ks[1];
      const uint4* vs =
          reinterpret_cast<const uint4*>(vbase + static_cast<int64_t>(krow) * rs3 + colc);
      *reinterpret_cast<uint4*>(&vv[0]) = vs[0];
      *reinterpret_cast<uint4*>(&vv[8]) = vs[1];
    } else {
#pragma unroll
      for (int j = 0; j < 16; ++j) kv[j] = vv[j] = __bf16(0.f);
    }
    *reinterpret_cast<uint4*>(&K_lds[row * kStride + colc]) =
        *reinterpret_cast<uint4*>(&kv[0]);
    *reinterpret_cast<uint4*>(&K_lds[row * kStride + colc + 8]) =
        *reinterpret_cast<uint4*>(&kv[8]);
    *reinterpret_cast<uint4*>(&V_lds[row * kStride + colc]) =
        *reinterpret_cast<uint4*>(&vv[0]);
    *reinterpret_cast<uint4*>(&V_lds[row * kStride + colc + 8]) =
        *reinterpret_cast<uint4*>(&vv[8]);
  }
  __syncthreads();

  f32x4 dv_acc[2][4] = {};  // dV^T[sub][dh-tile] for the wave's 2x16 keys
  f32x4 dk_acc[2][4] = {};

  // async-STAGE split (T14): each q-tile's Q/dO/lse/delta/mask global
  // loads are ISSUED one iteration early, so their HBM latency hides
  // under the previous tile's 64-MFMA recompute instead of sitting
  // serialized between the two staging barriers. The +16 VGPRs of
  // in-flight data keep this kernel at its LDS-bound 2 waves/SIMD.
  const int st_row = tid >> 2, st_colc = (tid & 3) * 16;
  uint4 pf_q0, pf_q1, pf_d0, pf_d1;
  auto issue_loads = [&](int q0t) {
    const int qrow = q0t + st_row;
    if (qrow < S) {
      const uint4* qs = reinterpret_cast<const uint4*>(
          qbase + static_cast<int64_t>(qrow) * rs3 + st_colc);
      pf_q0 = qs[0];
      pf_q1 = qs[1];
      const uint4* ds = reinterpret_cast<const uint4*>(
          dobase + static_cast<int64_t>(qrow) * H + st_colc);
      pf_d0 = ds[0];
      pf_d1 = ds[1];
    } else {
      pf_q0 = pf_q1 = pf_d0 = pf_d1 = uint4{0, 0, 0, 0};
    }
  };
  issue_loads(0);

  const int n_q = (S + 63) / 64;
  for (int qt = 0; qt < n_q; ++qt) {
    const int q0 = qt * 64;
    __syncthreads();  // all waves done reading the previous tile's Q/dO
    // stage this q-tile from the prefetched registers
    {
      *reinterpret_cast<uint4*>(&Q_lds[st_row * kTrStride + st_colc]) = pf_q0;
      *reinterpret_cast<uint4*>(&Q_lds[st_row * kTrStride + st_colc + 8]) =
          pf_q1;
      *reinterpret_cast<uint4*>(&dO_lds[st_row * kTrStride + st_colc]) =
          pf_d0;
      *reinterpret_cast<uint4*>(&dO_lds[st_row * kTrStride + st_colc + 8]) =
          pf_d1;
      if (tid < 64) {
        const int qr = q0 + tid;
        lse_lds[tid] = (qr < S) ? lse[static_cast<int64_t>(bh) * S + qr] : 0.f;
        dlt_lds[tid] = (qr < S) ? delta[static_cast<int64_t>(bh) * S + qr] : 0.f;
      }
      if (TRAIN_DROP) {
        // keep-mask tile [64 q][128 keys] bits, transposed [word][row]
        // so a q-subtile's 4 row-words are one aligned b128 read below;
        // words past the row end are ones (those keys are zeroed by the
        // kvalid/slen guards)
        const int kw = (k0 >> 5) + (tid & 3);  // k0 is a multiple of 128
        const int qrow_m = q0 + st_row;
        uint32_t bits = 0xFFFFFFFFu;
        if (qrow_m < S && kw < Sw)
          bits = mask_base[static_cast<int64_t>(qrow_m) * Sw + kw];
        mk_lds[(tid & 3) * 64 + st_row] = bits;
      }
    }
    __syncthreads();
    if (qt + 1 < n_q) issue_loads(q0 + 64);

    // recompute S tiles for the wave's 2x16 keys x 64 q-rows:
    // S[q][key]: A = Q rows, B = K^T (natural K rows); C col = key = li.
    // Q/dO fragments are shared across the two key subtiles.
    const int key_local0 = wave * 32 + li;
    const int key_local1 = wave * 32 + 16 + li;
    // P and dS go STRAIGHT into their B-fragment slots as each q-tile
    // is produced (element e = (mq&1)*4 + r of chunk mq>>1) - no
    // [2][4][4] staging arrays (they cost ~64 VGPRs and halved
    // occupancy at 324 VGPRs/wave)
    union FB {
      bf16x8 v;
      __bf16 e[8];
    };
    FB pfrag[2][2], dsfrag[2][2];
#pragma unroll
    for (int mq = 0; mq < 4; ++mq) {
      const __bf16* qrow_n = &Q_lds[(mq * 16 + li) * kTrStride];
      const __bf16* dorow = &dO_lds[(mq * 16 + li) * kTrStride];
      const bf16x8 qa0 = frag_row(qrow_n, 0, g), qa1 = frag_row(qrow_n, 32, g);
      const bf16x8 da0 = frag_row(dorow, 0, g), da1 = frag_row(dorow, 32, g);
      f32x4 acc[2] = {}, dp[2] = {};
#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
        const int key_local = sub ? key_local1 : key_local0;
        const __bf16* krow_n = &K_lds[key_local * kStride];
        const __bf16* vrow = &V_lds[key_local * kStride];
        acc[sub] = MFMA16(qa0, frag_row(krow_n, 0, g), acc[sub]);
        acc[sub] = MFMA16(qa1, frag_row(krow_n, 32, g), acc[sub]);
        dp[sub] = MFMA16(da0, frag_row(vrow, 0, g), dp[sub]);
        dp[sub] = MFMA16(da1, frag_row(vrow, 32, g), dp[sub]);
      }
      uint32_t mrow4[4];  // keep-mask words for rows mq*16+g*4+0..3
      if (TRAIN_DROP)     // both key subtiles live in word column `wave`
        *reinterpret_cast<uint4*>(mrow4) = *reinterpret_cast<const uint4*>(
            &mk_lds[wave * 64 + mq * 16 + g * 4]);
#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
        const int key_abs = k0 + (sub ? key_local1 : key_local0);
        const bool kvalid = key_abs < slen && key_abs < S;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int q_abs = q0 + mq * 16 + g * 4 + r;
          const float row_lse = lse_lds[mq * 16 + g * 4 + r];
          const float d_row = dlt_lds[mq * 16 + g * 4 + r];
          float pr = (kvalid && q_abs < S)
                         ? __expf(acc[sub][r] * scale - row_lse)
                         : 0.f;
          float dpd = dp[sub][r];
          float pkeep = pr;
          if (TRAIN_DROP) {
            const int key_l = sub ? key_local1 : key_local0;
            const bool keep = (mrow4[r] >> (key_l & 31)) & 1;
            dpd = keep ? dpd * inv_keep : 0.f;
            pkeep = keep ? pr * inv_keep : 0.f;  // dropped P feeds dV
          }
          pfrag[sub][mq >> 1].e[(mq & 1) * 4 + r] = __bf16(pkeep);
          dsfrag[sub][mq >> 1].e[(mq & 1) * 4 + r] =
              __bf16(pr * (dpd - d_row) * scale);
        }
      }
    }

    // dV^T += dO^T x P ; dK^T += Q^T x dS — dO^T/Q^T fragments shared
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int m = 0; m < 4; ++m) {
      const bf16x8 dt0 = frag_tr(dO_lds, 0, m * 16);
      const bf16x8 dt1 = frag_tr(dO_lds, 32, m * 16);
      const bf16x8 qt0 = frag_tr(Q_lds, 0, m * 16);
      const bf16x8 qt1 = frag_tr(Q_lds, 32, m * 16);
      dv_acc[0][m] = MFMA16(dt0, pfrag[0][0].v, dv_acc[0][m]);
      dv_acc[1][m] = MFMA16(dt0, pfrag[1][0].v, dv_acc[1][m]);
      dv_acc[0][m] = MFMA16(dt1, pfrag[0][1].v, dv_acc[0][m]);
      dv_acc[1][m] = MFMA16(dt1, pfrag[1][1].v, dv_acc[1][m]);
      dk_acc[0][m] = MFMA16(qt0, dsfrag[0][0].v, dk_acc[0][m]);
      dk_acc[1][m] = MFMA16(qt0, dsfrag[1][0].v, dk_acc[1][m]);
      dk_acc[0][m] = MFMA16(qt1, dsfrag[0][1].v, dk_acc[0][m]);
      dk_acc[1][m] = MFMA16(qt1, dsfrag[1][1].v, dk_acc[1][m]);
    }
    __builtin_amdgcn_s_setprio(0);
  }

  // write dK/dV straight into dqkv (this block owns keys k0..k0+127)
#pragma unroll
  for (int sub = 0; sub < 2; ++sub) {
#pragma unroll
    for (int m = 0; m < 4; ++m) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int dh = m * 16 + g * 4 + r;  // C row = dh/d index
        const int key_abs = k0 + wave * 32 + sub * 16 + li;
        if (key_abs < S) {
          const int64_t rowb = static_cast<int64_t>(b) * S + key_abs;
          dqkv[rowb * rs3 + H + h * 64 + dh] = __bf16(dk_acc[sub][m][r]);
          dqkv[rowb * rs3 + 2 * H + h * 64 + dh] = __bf16(dv_acc[sub][m][r]);
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// ds_read_b64_tr_b16 semantics probe (debug utility; used once per ROCm
// bump to verify the lane->element mapping the PV/dK/dV fragment reads
// rely on). LDS is filled with the identity pattern (element i = i) and
// each lane reads at its own 8-byte-aligned address (8*lane bytes);
// out[l][j] then IS the element index lane l's slot j received.
// ---------------------------------------------------------------------------
__global__ void tr16_probe_kernel(float* __restrict__ out, int addr_mode) {
  __shared__ __bf16 lds[1024];
  const int t = threadIdx.x;
  for (int i = t; i < 1024; i += 64) lds[i] = __bf16(static_cast<float>(i));
  __syncthreads();
  typedef __attribute__((__vector_size__(4 * sizeof(__bf16)))) __bf16 bf16x4v;
  int elem_off;
  switch (addr_mode) {
    case 0: elem_off = t * 4; break;           // 8 B per lane, contiguous
    case 1: elem_off = 0; break;               // uniform base
    case 2: elem_off = (t & 15) * 4; break;    // repeats per 16-lane group
    default: elem_off = (t >> 4) * 4; break;   // group-constant
  }
  auto p = (__attribute__((address_space(3))) bf16x4v*)(&lds[elem_off]);
  bf16x4v v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p);
#pragma unroll
  for (int j = 0; j < 4; ++j) out[t * 4 + j] = static_cast<float>(v[j]);
}

torch::Tensor tr16_probe(int64_t addr_mode) {
  auto out = torch::empty({64, 4}, torch::TensorOptions()
                                       .dtype(torch::kFloat32)
                                       .device(torch::kCUDA));
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(tr16_probe_kernel, dim3(1), dim3(64), 0, stream,
                     out.data_ptr<float>(), static_cast<int>(addr_mode));
  return out;
}

// ---------------------------------------------------------------------------
// host wrappers
// ---------------------------------------------------------------------------
std::vector<torch::Tensor> attention_fwd(torch::Tensor qkv,
                                         torch::Tensor seqlens,
                                         int64_t num_heads, double p,
                                         int64_t seed, int64_t offset) {
  TORCH_CHECK(qkv.dim() == 3 && qkv.is_contiguous(), "attn_fwd: bad qkv");
  TORCH_CHECK(qkv.scalar_type() == torch::kBFloat16,
              "attn_fwd: bf16 only (autocast path)");
  const int B = qkv.size(0), S = qkv.size(1);
  const int H = qkv.size(2) / 3;
  const int NH = static_cast<int>(num_heads);
  TORCH_CHECK(H == NH * 64, "attn_fwd: head_dim must be 64");
  TORCH_CHECK(S % 16 == 0, "attn_fwd: S must be a multiple of 16");
  auto seql = seqlens.to(qkv.device(), torch::kInt32).contiguous();
  auto out = torch::empty({B, S, H}, qkv.options());
  auto lse = torch::empty({B * NH, S}, qkv.options().dtype(torch::kFloat32));
  const bool train_drop = p > 0.0;
  // keep-mask BITS [B*NH, S, ceil(S/32)] int32 words: written once
  // here, read (not regenerated) by both backward kernels
  const int Sw = (S + 31) / 32;
  auto dmask = train_drop
                   ? torch::empty({static_cast<int64_t>(B) * NH, S, Sw},
                                  qkv.options().dtype(torch::kInt32))
                   : torch::empty({0}, qkv.options().dtype(torch::kInt32));
  const float scale = 1.0f / sqrtf(64.f);
  auto stream = at::hip::getCurrentHIPStream();
  dim3 grid((S + 127) / 128, B * NH), block(256);  // 128 q-rows per block
  const size_t lds = (64 * kStride + 64 * kTrStride) * sizeof(__bf16) +
                     4 * 64 * sizeof(float) + 128 * 2 * sizeof(uint32_t);
  if (train_drop) {
    const auto [thresh, p_q] = quantize_drop_p(p);
    const int64_t words = static_cast<int64_t>(B) * NH * S * Sw;
    hipLaunchKernelGGL(dropout_mask_kernel,
                       dim3((words + 255) / 256), dim3(256), 0, stream,
                       reinterpret_cast<uint32_t*>(dmask.data_ptr<int>()),
                       words, thresh, static_cast<uint64_t>(seed),
                       static_cast<uint64_t>(offset));
    hipLaunchKernelGGL((attn_fwd_kernel<true>), grid, block, lds, stream,
                       reinterpret_cast<const __bf16*>(qkv.data_ptr()),
                       seql.data_ptr<int>(),
                       reinterpret_cast<__bf16*>(out.data_ptr()),
                       lse.data_ptr<float>(),
                       reinterpret_cast<const uint32_t*>(
                           dmask.data_ptr<int>()),
                       B, S, NH, p_q, scale,
                       static_cast<uint64_t>(seed),
                       static_cast<uint64_t>(offset));
  } else {
    hipLaunchKernelGGL((attn_fwd_kernel<false>), grid, block, lds, stream,
                       reinterpret_cast<const __bf16*>(qkv.data_ptr()),
                       seql.data_ptr<int>(),
                       reinterpret_cast<__bf16*>(out.data_ptr()),
                       lse.data_ptr<float>(), nullptr, B, S, NH,
                       static_cast<float>(p), scale,
                       static_cast<uint64_t>(seed),
                       static_cast<uint64_t>(offset));
  }
  return {out, lse, dmask};
}

torch::Tensor attention_bwd(torch::Tensor dout, torch::Tensor qkv,
                            torch::Tensor seqlens, torch::Tensor out,
                            torch::Tensor lse, torch::Tensor dmask,
                            int64_t num_heads, double p, int64_t seed,
                            int64_t offset) {
  const int B = qkv.size(0), S = qkv.size(1);
  const int H = qkv.size(2) / 3;
  const int NH = static_cast<int>(num_heads);
  auto seql = seqlens.to(qkv.device(), torch::kInt32).contiguous();
  auto dqkv = torch::empty_like(qkv);
  auto fopts = qkv.options().dtype(torch::kFloat32);
  auto delta = torch::empty({static_cast<int64_t>(B) * NH * S}, fopts);
  auto stream = at::hip::getCurrentHIPStream();
  auto dout_c = dout.contiguous();

  const int64_t rows = static_cast<int64_t>(B) * NH * S;
  const int64_t delta_waves = (rows + 7) / 8;  // 8 rows per wave
  hipLaunchKernelGGL(attn_delta_kernel, dim3((delta_waves + 3) / 4),
                     dim3(256), 0, stream,
                     reinterpret_cast<const __bf16*>(dout_c.data_ptr()),
                     reinterpret_cast<const __bf16*>(out.data_ptr()),
                     delta.data_ptr<float>(), B, S, NH);

  dim3 grid((S + 127) / 128, B * NH), block(256);  // 128 keys per bwd block
  dim3 grid_dq((S + 127) / 128, B * NH);           // 128 q-rows per dq block
  // K/V hold 128 rows at kStride; Q/dO 64 rows at kTrStride (tr-read);
  // + 1 KB bit-mask tile -> ~59 KB (over the 64 KB default dynamic-LDS
  // cap; MI355X has 160 KB per CU)
  const size_t lds = (4 * 64 * kStride + 2 * 64 * kTrStride) *
                         sizeof(__bf16) +
                     2 * 64 * sizeof(float) + 64 * 4 * sizeof(uint32_t);
  const size_t lds_dq = (64 * kStride + 64 * kTrStride) * sizeof(__bf16) +
                        128 * 2 * sizeof(uint32_t);
  HIP_CHECK(hipFuncSetAttribute(
      reinterpret_cast<const void*>(&attn_bwd_kernel<true>),
      hipFuncAttributeMaxDynamicSharedMemorySize, lds));
  HIP_CHECK(hipFuncSetAttribute(
      reinterpret_cast<const void*>(&attn_bwd_kernel<false>),
      hipFuncAttributeMaxDynamicSharedMemorySize, lds));
  const float scale = 1.0f / sqrtf(64.f);
  const bool train_drop = p > 0.0;
  // same 8-bit quantization as the forward's mask generation, so the
  // backward keep-scale matches the stored mask's statistics exactly
  const float p_q = quantize_drop_p(p).second;
  const uint32_t* mask_ptr =
      train_drop ? reinterpret_cast<const uint32_t*>(dmask.data_ptr<int>())
                 : nullptr;
  auto args = [&](auto kernel, dim3 g, size_t lds_bytes) {
    hipLaunchKernelGGL(kernel, g, block, lds_bytes, stream,
                       reinterpret_cast<const __bf16*>(dout_c.data_ptr()),
                       reinterpret_cast<const __bf16*>(qkv.data_ptr()),
                       seql.data_ptr<int>(), lse.data_ptr<float>(),
                       delta.data_ptr<float>(), mask_ptr,
                       reinterpret_cast<__bf16*>(dqkv.data_ptr()), B, S, NH,
                       p_q, scale,
                       static_cast<uint64_t>(seed),
                       static_cast<uint64_t>(offset));
  };
  if (train_drop) {
    args(attn_bwd_kernel<true>, grid, lds);
    args(attn_dq_kernel<true>, grid_dq, lds_dq);
  } else {
    args(attn_bwd_kernel<false>, grid, lds);
    args(attn_dq_kernel<false>, grid_dq, lds_dq);
  }
  return dqkv;
}

}  // namespace bpa
