// Fused embedding gather + add + LayerNorm + dropout for MI355X (gfx950).
//
// BertEmbeddings in one kernel (reference: src/modeling.py:338-373):
//   z = word[ids] + pos[s] (+ tok[tt]) ; y = dropout(LN(z))
// Tables may be fp32 (autocast mode) or bf16/fp16 (pure-bf16 master-
// weight mode); the output is emitted in the requested dtype. Backward:
// dz from LN-backward (stored to a fp32 buffer), then deterministic
// pos/token-type reductions and an atomic scatter-add into a dense fp32
// word-embedding gradient (cast to the table dtype by the Python
// wrapper when the tables are low-precision).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../common.h"

namespace bpa {

torch::Tensor col_reduce_full(torch::Tensor parts);  // layernorm.hip

template <typename T, typename TW, int VEC, bool HAS_TOK, bool TRAIN_DROP>
__global__ void embed_fwd_kernel(
    const int64_t* __restrict__ ids, const int64_t* __restrict__ tt,
    const TW* __restrict__ word, const TW* __restrict__ pos,
    const TW* __restrict__ tok, const float* __restrict__ gamma,
    const float* __restrict__ beta, T* __restrict__ y, float* __restrict__ z,
    uint8_t* __restrict__ mask, float* __restrict__ mean,
    float* __restrict__ rstd, int rows, int seq_len, int H, float p, float eps,
    uint64_t seed, uint64_t offset) {
  // one BLOCK per row; ITEMS elements per thread held in registers
  // through the mean/var reduction (see layernorm.hip ln_fwd_kernel)
  __shared__ float red[32];
  const int row = blockIdx.x;
  if (row >= rows) return;
  const int64_t base = static_cast<int64_t>(row) * H;
  const int s = row % seq_len;
  const int64_t wid = ids[row];
  const int64_t tid_ = HAS_TOK ? tt[row] : 0;
  const TW* wrow = word + wid * H;
  const TW* prow = pos + static_cast<int64_t>(s) * H;
  const TW* trow = HAS_TOK ? tok + tid_ * H : nullptr;
  const float keep_scale = TRAIN_DROP ? 1.0f / (1.0f - p) : 1.0f;
  Philox philox(seed);

  constexpr int ITEMS = 2;  // covers H <= 2*blockDim*VEC
  float zv[ITEMS][VEC];
  int cols[ITEMS];
  float sum = 0.f, sumsq = 0.f;
#pragma unroll
  for (int i = 0; i < ITEMS; ++i) {
    const int c = (i * blockDim.x + threadIdx.x) * VEC;
    cols[i] = c;
    if (c < H) {
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        float t = DTraits<TW>::to_f32(wrow[c + k]) +
                  DTraits<TW>::to_f32(prow[c + k]);
        if (HAS_TOK) t += DTraits<TW>::to_f32(trow[c + k]);
        zv[i][k] = t;
        sum += t;
        sumsq += t * t;
      }
      if (sizeof(float) * VEC == 16) {
        *reinterpret_cast<uint4*>(z + base + c) =
            *reinterpret_cast<const uint4*>(zv[i]);
      } else {
#pragma unroll
        for (int k = 0; k < VEC; ++k) z[base + c + k] = zv[i][k];
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
        float t = (zv[i][k] - mu) * rs * gamma[c + k] + beta[c + k];
        if (TRAIN_DROP) t = mv[k] ? t * keep_scale : 0.f;
        ov[k] = DTraits<T>::from_f32(t);
      }
      if (sizeof(T) * VEC == 16) {
        *reinterpret_cast<uint4*>(y + base + c) =
            *reinterpret_cast<const uint4*>(ov);
      } else {
#pragma unroll
        for (int k = 0; k < VEC; ++k) y[base + c + k] = ov[k];
      }
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
}

// backward stage 1: dz = LN_bwd(dropout_bwd(dy)); per-block dgamma/dbeta
template <typename T, int VEC, int NW, bool TRAIN_DROP>
__global__ void embed_bwd_dz_kernel(
    const T* __restrict__ dy, const float* __restrict__ z,
    const uint8_t* __restrict__ mask, const float* __restrict__ gamma,
    const float* __restrict__ beta, const float* __restrict__ mean,
    const float* __restrict__ rstd, float* __restrict__ dz,
    float* __restrict__ part_dgamma, float* __restrict__ part_dbeta, int rows,
    int H, float p, int rows_per_block) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* lg = reinterpret_cast<float*>(smem_raw) + wave * 2 * H;  // [NW][2H]
  float* lb = lg + H;
  for (int c = lane; c < 2 * H; c += WAVE_SIZE) lg[c] = 0.f;
  const float keep_scale = TRAIN_DROP ? 1.0f / (1.0f - p) : 1.0f;

  const int row0 = blockIdx.x * rows_per_block;
  const int row_end = min(row0 + rows_per_block, rows);
  for (int r = row0 + wave; r < row_end; r += NW) {
    const int64_t base = static_cast<int64_t>(r) * H;
    const float mu = mean[r], rs = rstd[r];
    float s1 = 0.f, s2 = 0.f;
    for (int c = lane * VEC; c < H; c += WAVE_SIZE * VEC) {
      T dv[VEC];
      *reinterpret_cast<uint4*>(dv) = *reinterpret_cast<const uint4*>(dy + base + c);
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
        if (TRAIN_DROP) d = mv[k] ? d * keep_scale : 0.f;
        float zh = (z[base + c + k] - mu) * rs;
        float dw = d * gamma[c + k];
        s1 += dw * zh;
        s2 += dw;
        lg[c + k] += d * zh;
        lb[c + k] += d;
        dz[base + c + k] = dw;  // stash dw; finalized below after row sums
      }
    }
    s1 = wave_reduce_sum(s1) / H;
    s2 = wave_reduce_sum(s2) / H;
    for (int c = lane * VEC; c < H; c += WAVE_SIZE * VEC) {
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        float zh = (z[base + c + k] - mu) * rs;
        dz[base + c + k] = rs * (dz[base + c + k] - s2 - zh * s1);
      }
    }
  }
  __syncthreads();
  float* slab0 = reinterpret_cast<float*>(smem_raw);
  for (int c = threadIdx.x; c < H; c += blockDim.x) {
    float ag = 0.f, ab = 0.f;
#pragma unroll
    for (int w = 0; w < NW; ++w) {
      ag += slab0[w * 2 * H + c];
      ab += slab0[w * 2 * H + H + c];
    }
    part_dgamma[static_cast<int64_t>(blockIdx.x) * H + c] = ag;
    part_dbeta[static_cast<int64_t>(blockIdx.x) * H + c] = ab;
  }
}

// backward stage 2a: word-embedding scatter-add (atomic; duplicate ids rare)
__global__ void embed_bwd_word_kernel(const float* __restrict__ dz,
                                      const int64_t* __restrict__ ids,
                                      float* __restrict__ d_word, int64_t rows,
                                      int H) {
  const int64_t total = rows * H;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    const int64_t r = i / H;
    const int c = static_cast<int>(i % H);
    atomicAdd(&d_word[ids[r] * H + c], dz[i]);
  }
}

// backward stage 2b: position-embedding reduce over batch (deterministic)
__global__ void embed_bwd_pos_kernel(const float* __restrict__ dz,
                                     float* __restrict__ d_pos, int batch,
                                     int seq_len, int H) {
  const int64_t total = static_cast<int64_t>(seq_len) * H;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    const int s = static_cast<int>(i / H);
    const int c = static_cast<int>(i % H);
    float acc = 0.f;
    for (int b = 0; b < batch; ++b)
      acc += dz[(static_cast<int64_t>(b) * seq_len + s) * H + c];
    d_pos[static_cast<int64_t>(s) * H + c] = acc;
  }
}

// backward stage 2c: token-type reduce (few types; per-block LDS partials)
template <int NW>
__global__ void embed_bwd_tok_kernel(const float* __restrict__ dz,
                                     const int64_t* __restrict__ tt,
                                     float* __restrict__ parts, int rows, int H,
                                     int n_types, int rows_per_block) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* lt = reinterpret_cast<float*>(smem_raw);  // [n_types][H]
  for (int c = threadIdx.x; c < n_types * H; c += blockDim.x) lt[c] = 0.f;
  __syncthreads();
  const int row0 = blockIdx.x * rows_per_block;
  const int row_end = min(row0 + rows_per_block, rows);
  for (int r = row0 + wave; r < row_end; r += NW) {
    const int t = static_cast<int>(tt[r]);
    const int64_t base = static_cast<int64_t>(r) * H;
    for (int c = lane; c < H; c += WAVE_SIZE)
      atomicAdd(&lt[t * H + c], dz[base + c]);
  }
  __syncthreads();
  float* out = parts + static_cast<int64_t>(blockIdx.x) * n_types * H;
  for (int c = threadIdx.x; c < n_types * H; c += blockDim.x) out[c] = lt[c];
}

#define DISPATCH_OUT(TYPE, NAME, ...)                                        \
  [&] {                                                                      \
    if (TYPE == at::kBFloat16) {                                             \
      using out_t = __hip_bfloat16;                                          \
      constexpr int kVec = 8;                                                \
      return __VA_ARGS__();                                                  \
    } else if (TYPE == at::kHalf) {                                          \
      using out_t = __half;                                                  \
      constexpr int kVec = 8;                                                \
      return __VA_ARGS__();                                                  \
    } else if (TYPE == at::kFloat) {                                         \
      using out_t = float;                                                   \
      constexpr int kVec = 4;                                                \
      return __VA_ARGS__();                                                  \
    } else {                                                                 \
      TORCH_CHECK(false, NAME, ": unsupported dtype");                       \
    }                                                                        \
  }()

std::vector<torch::Tensor> embedding_ln_dropout_fwd(
    torch::Tensor ids, c10::optional<torch::Tensor> tt, torch::Tensor word,
    torch::Tensor pos, c10::optional<torch::Tensor> tok, torch::Tensor gamma,
    torch::Tensor beta, double p, double eps, int64_t seed, int64_t offset,
    torch::ScalarType out_dtype) {
  TORCH_CHECK(ids.dim() == 2, "embed_fwd: ids must be [B, S]");
  TORCH_CHECK(pos.scalar_type() == word.scalar_type() &&
                  (!tok.has_value() ||
                   tok->scalar_type() == word.scalar_type()),
              "embed_fwd: all tables must share a dtype");
  const int batch = ids.size(0), seq_len = ids.size(1);
  const int rows = batch * seq_len;
  const int H = word.size(1);
  auto ids_c = ids.contiguous();
  const bool has_tok = tok.has_value() && tt.has_value();
  torch::Tensor tt_c;
  if (has_tok) tt_c = tt->contiguous();
  auto gamma_f = gamma.contiguous().to(torch::kFloat32);
  auto beta_f = beta.contiguous().to(torch::kFloat32);
  const bool train_drop = p > 0.0;

  auto y = torch::empty({batch, seq_len, H},
                        word.options().dtype(out_dtype));
  auto fopts = word.options().dtype(torch::kFloat32);
  auto z = torch::empty({rows, H}, fopts);
  auto mask = train_drop
                  ? torch::empty({rows, H}, word.options().dtype(torch::kUInt8))
                  : torch::empty({0}, word.options().dtype(torch::kUInt8));
  auto mean = torch::empty({rows}, fopts);
  auto rstd = torch::empty({rows}, fopts);
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_OUT(out_dtype, "embed_fwd", [&] {
    using scalar_out_t = out_t;
    constexpr int kVecOuter = kVec;
    TORCH_CHECK(H % kVec == 0, "embed_fwd: H % ", kVec, " != 0");
    const int slices = H / kVec;
    const int threads =
        tmin(1024, (((slices + 1) / 2 + WAVE_SIZE - 1) / WAVE_SIZE) *
                       WAVE_SIZE);
    TORCH_CHECK(slices <= 2 * threads, "embed_fwd: H too large");
    DISPATCH_OUT(word.scalar_type(), "embed_fwd_table", [&] {
      using tw_t = out_t;  // inner dispatch alias
      auto launch = [&](auto tok_flag, auto train_c) {
        hipLaunchKernelGGL(
            (embed_fwd_kernel<scalar_out_t, tw_t, kVecOuter,
                              decltype(tok_flag)::value,
                              decltype(train_c)::value>),
            dim3(rows), dim3(threads), 0, stream, ids_c.data_ptr<int64_t>(),
            has_tok ? tt_c.data_ptr<int64_t>() : nullptr,
            reinterpret_cast<const tw_t*>(word.data_ptr()),
            reinterpret_cast<const tw_t*>(pos.data_ptr()),
            has_tok ? reinterpret_cast<const tw_t*>(tok->data_ptr())
                    : nullptr,
            gamma_f.data_ptr<float>(), beta_f.data_ptr<float>(),
            reinterpret_cast<scalar_out_t*>(y.data_ptr()),
            z.data_ptr<float>(),
            train_drop ? mask.data_ptr<uint8_t>() : nullptr,
            mean.data_ptr<float>(), rstd.data_ptr<float>(), rows, seq_len,
            H, static_cast<float>(p), static_cast<float>(eps),
            static_cast<uint64_t>(seed), static_cast<uint64_t>(offset));
      };
      if (has_tok && train_drop) launch(std::true_type{}, std::true_type{});
      else if (has_tok) launch(std::true_type{}, std::false_type{});
      else if (train_drop) launch(std::false_type{}, std::true_type{});
      else launch(std::false_type{}, std::false_type{});
    });
  });
  return {y, z, mask, mean, rstd};
}

std::vector<torch::Tensor> embedding_ln_dropout_bwd(
    torch::Tensor dy, torch::Tensor ids, c10::optional<torch::Tensor> tt,
    torch::Tensor z, torch::Tensor mask, torch::Tensor gamma,
    torch::Tensor mean, torch::Tensor rstd, double p, int64_t vocab,
    int64_t max_pos, int64_t n_types) {
  const int batch = ids.size(0), seq_len = ids.size(1);
  const int rows = batch * seq_len;
  const int H = z.size(1);
  auto ids_c = ids.contiguous();
  auto gamma_f = gamma.contiguous().to(torch::kFloat32);
  auto dy2 = dy.contiguous().view({rows, H});
  const bool train_drop = p > 0.0;
  const bool has_tok = tt.has_value() && n_types > 0;

  auto fopts = z.options();
  auto dz = torch::empty({rows, H}, fopts);
  constexpr int NW = 4;
  const int rows_per_block = 16;
  const int nblocks = (rows + rows_per_block - 1) / rows_per_block;
  auto part_g = torch::empty({nblocks, H}, fopts);
  auto part_b = torch::empty({nblocks, H}, fopts);
  auto stream = at::hip::getCurrentHIPStream();
  const size_t lds = NW * 2 * static_cast<size_t>(H) * sizeof(float);
  DISPATCH_OUT(dy.scalar_type(), "embed_bwd", [&] {
    auto launch = [&](auto train_c) {
      if (lds > 48 * 1024) {
        HIP_CHECK(hipFuncSetAttribute(
            reinterpret_cast<const void*>(
                &embed_bwd_dz_kernel<out_t, kVec, NW,
                                     decltype(train_c)::value>),
            hipFuncAttributeMaxDynamicSharedMemorySize, lds));
      }
      hipLaunchKernelGGL(
          (embed_bwd_dz_kernel<out_t, kVec, NW, decltype(train_c)::value>),
          dim3(nblocks), dim3(NW * WAVE_SIZE), lds, stream,
          reinterpret_cast<const out_t*>(dy2.data_ptr()), z.data_ptr<float>(),
          train_drop ? mask.data_ptr<uint8_t>() : nullptr,
          gamma_f.data_ptr<float>(), nullptr, mean.data_ptr<float>(),
          rstd.data_ptr<float>(), dz.data_ptr<float>(),
          part_g.data_ptr<float>(), part_b.data_ptr<float>(), rows, H,
          static_cast<float>(p), rows_per_block);
    };
    if (train_drop) launch(std::true_type{});
    else launch(std::false_type{});
  });

  auto d_word = torch::zeros({vocab, H}, fopts);
  auto d_pos = torch::empty({max_pos, H}, fopts);
  auto d_tok = has_tok ? torch::empty({n_types, H}, fopts)
                       : torch::empty({0}, fopts);
  auto dgamma = col_reduce_full(part_g);
  auto dbeta = col_reduce_full(part_b);

  const int64_t total = static_cast<int64_t>(rows) * H;
  const int sblocks = static_cast<int>(std::min<int64_t>((total + 255) / 256, 2048));
  hipLaunchKernelGGL(embed_bwd_word_kernel, dim3(sblocks), dim3(256), 0,
                     stream, dz.data_ptr<float>(), ids_c.data_ptr<int64_t>(),
                     d_word.data_ptr<float>(), rows, H);
  const int64_t ptotal = static_cast<int64_t>(seq_len) * H;
  const int pblocks = static_cast<int>(std::min<int64_t>((ptotal + 255) / 256, 2048));
  hipLaunchKernelGGL(embed_bwd_pos_kernel, dim3(pblocks), dim3(256), 0,
                     stream, dz.data_ptr<float>(), d_pos.data_ptr<float>(),
                     batch, seq_len, H);
  if (has_tok) {
    auto tt_c = tt->contiguous();
    auto tparts = torch::empty({nblocks, n_types * H}, fopts);
    const size_t tlds = static_cast<size_t>(n_types) * H * sizeof(float);
    hipLaunchKernelGGL((embed_bwd_tok_kernel<NW>), dim3(nblocks),
                       dim3(NW * WAVE_SIZE), tlds, stream,
                       dz.data_ptr<float>(), tt_c.data_ptr<int64_t>(),
                       tparts.data_ptr<float>(), rows, H,
                       static_cast<int>(n_types), rows_per_block);
    d_tok = col_reduce_full(tparts).view({n_types, H});
  }
  return {d_word, d_pos, d_tok, dgamma, dbeta};
}

}  // namespace bpa
