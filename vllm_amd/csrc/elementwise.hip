// Elementwise / normalization / cache-write kernels for MI355X (gfx950).
//
// All memory-bound: the design rule is HBM3E bandwidth (G13 — vectorize
// bf16 as short8, 16B/lane), one pass, fused where the producer allows.
//
// Reference roles: csrc/libtorch_stable/layernorm_kernels.cu (:15,:107),
// pos_encoding_kernels.cu (:77), activation_kernels.cu (:102),
// cache_kernels.cu reshape_and_cache_flash (:264) — re-designed for
// CDNA4, not ported.

#include <torch/all.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace vllm_amd {

// ---------------------------------------------------------------------------
// RMSNorm: one workgroup per token row; vectorized short8 loads; two-level
// (wave, block) reduction of sum of squares.
template <typename Tag, int BLOCK>
__global__ void rms_norm_kernel(short* __restrict__ out,
                                const short* __restrict__ in,
                                const short* __restrict__ weight,
                                const float eps, const int hidden) {
  __shared__ float red[BLOCK / WAVE_SIZE];
  const int64_t row = blockIdx.x;
  const short* x = in + row * hidden;
  short* o = out + row * hidden;

  float ss = 0.f;
  const int nvec = hidden / 8;
  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    s16x8 v = reinterpret_cast<const s16x8*>(x)[i];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = to_f32<Tag>(v[j]);
      ss += f * f;
    }
  }
  ss = block_reduce_sum<BLOCK / WAVE_SIZE>(ss, red);
  const float inv = rsqrtf(ss / hidden + eps);

  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    s16x8 v = reinterpret_cast<const s16x8*>(x)[i];
    s16x8 w = reinterpret_cast<const s16x8*>(weight)[i];
    s16x8 r;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      r[j] = from_f32<Tag>(to_f32<Tag>(v[j]) * inv * to_f32<Tag>(w[j]));
    reinterpret_cast<s16x8*>(o)[i] = r;
  }
}

// Fused residual-add + RMSNorm: residual += x (written back), out = norm.
template <typename Tag, int BLOCK>
__global__ void fused_add_rms_norm_kernel(short* __restrict__ x,
                                          short* __restrict__ residual,
                                          const short* __restrict__ weight,
                                          const float eps, const int hidden) {
  __shared__ float red[BLOCK / WAVE_SIZE];
  const int64_t row = blockIdx.x;
  short* xr = x + row * hidden;
  short* rr = residual + row * hidden;

  float ss = 0.f;
  const int nvec = hidden / 8;
  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    s16x8 xv = reinterpret_cast<const s16x8*>(xr)[i];
    s16x8 rv = reinterpret_cast<const s16x8*>(rr)[i];
    s16x8 sum;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = to_f32<Tag>(xv[j]) + to_f32<Tag>(rv[j]);
      sum[j] = from_f32<Tag>(f);
      float g = to_f32<Tag>(sum[j]);  // variance of the ROUNDED sum
      ss += g * g;
    }
    reinterpret_cast<s16x8*>(rr)[i] = sum;  // new residual
  }
  ss = block_reduce_sum<BLOCK / WAVE_SIZE>(ss, red);
  const float inv = rsqrtf(ss / hidden + eps);

  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    s16x8 sv = reinterpret_cast<const s16x8*>(rr)[i];
    s16x8 wv = reinterpret_cast<const s16x8*>(weight)[i];
    s16x8 r;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      r[j] = from_f32<Tag>(to_f32<Tag>(sv[j]) * inv * to_f32<Tag>(wv[j]));
    reinterpret_cast<s16x8*>(xr)[i] = r;
  }
}

// ---------------------------------------------------------------------------
// SiLU-and-mul: out[t, i] = silu(x[t, i]) * x[t, d+i].
// 2D grid (row, column chunk): one short8 vector per thread per launch so
// small decode batches still put >>256 workgroups in flight (8 XCDs).
template <typename Tag>
__global__ void silu_and_mul_kernel(short* __restrict__ out,
                                    const short* __restrict__ in,
                                    const int d) {
  const int64_t row = blockIdx.x;
  const short* gate = in + row * 2 * d;
  const short* up = gate + d;
  short* o = out + row * d;
  const int nvec = d / 8;
  {
    const int i = blockIdx.y * blockDim.x + threadIdx.x;
    if (i >= nvec) return;
    s16x8 g = reinterpret_cast<const s16x8*>(gate)[i];
    s16x8 u = reinterpret_cast<const s16x8*>(up)[i];
    s16x8 r;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = to_f32<Tag>(g[j]);
      float s = gf / (1.f + __expf(-gf));
      r[j] = from_f32<Tag>(s * to_f32<Tag>(u[j]));
    }
    reinterpret_cast<s16x8*>(o)[i] = r;
  }
}

// GELU(tanh)-and-mul. Same 2D grid as silu_and_mul_kernel.
template <typename Tag>
__global__ void gelu_and_mul_kernel(short* __restrict__ out,
                                    const short* __restrict__ in,
                                    const int d) {
  const int64_t row = blockIdx.x;
  const short* gate = in + row * 2 * d;
  const short* up = gate + d;
  short* o = out + row * d;
  const int nvec = d / 8;
  {
    const int i = blockIdx.y * blockDim.x + threadIdx.x;
    if (i >= nvec) return;
    s16x8 g = reinterpret_cast<const s16x8*>(gate)[i];
    s16x8 u = reinterpret_cast<const s16x8*>(up)[i];
    s16x8 r;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float xf = to_f32<Tag>(g[j]);
      float c = 0.7978845608028654f * (xf + 0.044715f * xf * xf * xf);
      float gelu = 0.5f * xf * (1.f + tanhf(c));
      r[j] = from_f32<Tag>(gelu * to_f32<Tag>(u[j]));
    }
    reinterpret_cast<s16x8*>(o)[i] = r;
  }
}

// ---------------------------------------------------------------------------
// Rotary embedding (neox style), in-place on q and k.
// cos_sin_cache: [max_pos, rot] fp32 = [cos(rot/2) | sin(rot/2)].
// q: [T, Hq, D], k: [T, Hkv, D] (contiguous); rotates first `rot` dims.
// Grid: (T); block: 256. Threads cover (head, pair) space.
template <typename Tag>
__global__ void rope_kernel(short* __restrict__ q, short* __restrict__ k,
                            const int64_t* __restrict__ positions,
                            const float* __restrict__ cos_sin,
                            const int rot, const int head_dim,
                            const int num_q_heads, const int num_kv_heads,
                            const int64_t q_stride, const int64_t k_stride) {
  const int64_t t = blockIdx.x;
  const int64_t pos = positions[t];
  const float* cs = cos_sin + pos * rot;
  const int half = rot / 2;

  const int total = (num_q_heads + num_kv_heads) * half;
  for (int idx = threadIdx.x; idx < total; idx += blockDim.x) {
    const int h = idx / half;
    const int p = idx % half;
    short* base;
    if (h < num_q_heads) {
      base = q + t * q_stride + (int64_t)h * head_dim;
    } else {
      base = k + t * k_stride + (int64_t)(h - num_q_heads) * head_dim;
    }
    const float c = cs[p];
    const float s = cs[half + p];
    const float x1 = to_f32<Tag>(base[p]);
    const float x2 = to_f32<Tag>(base[half + p]);
    base[p] = from_f32<Tag>(x1 * c - x2 * s);
    base[half + p] = from_f32<Tag>(x2 * c + x1 * s);
  }
}

// ---------------------------------------------------------------------------
// reshape_and_cache: scatter new K/V rows into the paged cache.
// key/value: [T, Hkv, D]; cache: [2, num_blocks, Hkv, block_size, D]
// (head-major inside the block so each (block, head) tile is contiguous);
// slot_mapping: [T] int64 = block_id * block_size + offset.
// One workgroup per token; vectorized 16B copies.
template <typename Tag, typename CTag>
__global__ void reshape_and_cache_kernel(
    const short* __restrict__ key, const short* __restrict__ value,
    typename CacheTraits<CTag>::elem* __restrict__ kv_cache,
    const int64_t* __restrict__ slot_mapping,
    const int64_t kv_stride,  // elements between K and V planes
    const int64_t k_row_stride, const int64_t v_row_stride,
    const int num_kv_heads, const int head_dim, const int block_size) {
  using CT = CacheTraits<CTag>;
  using cvec = typename CT::vec8;
  const int64_t t = blockIdx.x;
  const int64_t slot = slot_mapping[t];
  if (slot < 0) return;  // padding token
  const int64_t blk = slot / block_size;
  const int off = (int)(slot % block_size);
  const short* krow = key + t * k_row_stride;
  const short* vrow = value + t * v_row_stride;
  const int dvec = head_dim / 8;
  const int nvec = (num_kv_heads * head_dim) / 8;
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    const int h = i / dvec;
    const int dv = i % dvec;
    const int64_t dst =
        (((blk * num_kv_heads + h) * block_size + off) * head_dim) / 8 + dv;
    s16x8 kv8 = reinterpret_cast<const s16x8*>(krow)[i];
    s16x8 vv8 = reinterpret_cast<const s16x8*>(vrow)[i];
    cvec kc, vc;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      kc[j] = CT::put(to_f32<Tag>(kv8[j]));
      vc[j] = CT::put(to_f32<Tag>(vv8[j]));
    }
    reinterpret_cast<cvec*>(kv_cache)[dst] = kc;
    reinterpret_cast<cvec*>(kv_cache + kv_stride)[dst] = vc;
  }
}

// ===========================================================================
// Host-side launchers
// ===========================================================================

static inline void check_16b(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(
      t.scalar_type() == torch::kBFloat16 || t.scalar_type() == torch::kHalf,
      name, " must be bf16 or fp16");
}

// [T, H, D] view whose heads are contiguous within a row but whose rows may
// be strided (a head slice of the fused QKV projection output). Rows must
// stay 16-byte aligned for the short8 loads.
static inline void check_rows_16b(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.dim() == 3, name, " must be [T, H, D]");
  TORCH_CHECK(t.stride(2) == 1 && t.stride(1) == t.size(2), name,
              " heads must be contiguous within a row");
  TORCH_CHECK(t.stride(0) % 8 == 0, name, " row stride must be 16B-aligned");
  TORCH_CHECK(
      t.scalar_type() == torch::kBFloat16 || t.scalar_type() == torch::kHalf,
      name, " must be bf16 or fp16");
}

#define DISPATCH_16B(TENSOR, FN)                       \
  if ((TENSOR).scalar_type() == torch::kBFloat16) {    \
    FN(BF16Tag);                                       \
  } else {                                             \
    FN(FP16Tag);                                       \
  }

torch::Tensor rms_norm(torch::Tensor x, torch::Tensor weight, double eps) {
  check_16b(x, "x");
  auto out = torch::empty_like(x);
  const int hidden = x.size(-1);
  TORCH_CHECK(hidden % 8 == 0, "hidden must be a multiple of 8");
  const int64_t rows = x.numel() / hidden;
  constexpr int BLOCK = 256;
  auto stream = c10::hip::getCurrentHIPStream().stream();
#define LAUNCH_RMS(TAG)                                                  \
  hipLaunchKernelGGL((rms_norm_kernel<TAG, BLOCK>), dim3(rows),          \
                     dim3(BLOCK), 0, stream,                             \
                     (short*)out.data_ptr(), (const short*)x.data_ptr(), \
                     (const short*)weight.data_ptr(), (float)eps, hidden)
  DISPATCH_16B(x, LAUNCH_RMS);
#undef LAUNCH_RMS
  HIP_CHECK_KERNEL();
  return out;
}

void fused_add_rms_norm(torch::Tensor x, torch::Tensor residual,
                        torch::Tensor weight, double eps) {
  check_16b(x, "x");
  check_16b(residual, "residual");
  const int hidden = x.size(-1);
  TORCH_CHECK(hidden % 8 == 0);
  const int64_t rows = x.numel() / hidden;
  constexpr int BLOCK = 256;
  auto stream = c10::hip::getCurrentHIPStream().stream();
#define LAUNCH_FARN(TAG)                                            \
  hipLaunchKernelGGL((fused_add_rms_norm_kernel<TAG, BLOCK>),       \
                     dim3(rows), dim3(BLOCK), 0, stream,            \
                     (short*)x.data_ptr(), (short*)residual.data_ptr(), \
                     (const short*)weight.data_ptr(), (float)eps, hidden)
  DISPATCH_16B(x, LAUNCH_FARN);
#undef LAUNCH_FARN
  HIP_CHECK_KERNEL();
}

torch::Tensor silu_and_mul(torch::Tensor x) {
  check_16b(x, "x");
  const int d = x.size(-1) / 2;
  TORCH_CHECK(d % 8 == 0);
  auto sizes = x.sizes().vec();
  sizes.back() = d;
  auto out = torch::empty(sizes, x.options());
  const int64_t rows = x.numel() / (2 * d);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const int chunks = (d / 8 + 255) / 256;
#define LAUNCH_SILU(TAG)                                              \
  hipLaunchKernelGGL((silu_and_mul_kernel<TAG>), dim3(rows, chunks),  \
                     dim3(256), 0, stream, (short*)out.data_ptr(),    \
                     (const short*)x.data_ptr(), d)
  DISPATCH_16B(x, LAUNCH_SILU);
#undef LAUNCH_SILU
  HIP_CHECK_KERNEL();
  return out;
}

torch::Tensor gelu_and_mul(torch::Tensor x) {
  check_16b(x, "x");
  const int d = x.size(-1) / 2;
  TORCH_CHECK(d % 8 == 0);
  auto sizes = x.sizes().vec();
  sizes.back() = d;
  auto out = torch::empty(sizes, x.options());
  const int64_t rows = x.numel() / (2 * d);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const int chunks = (d / 8 + 255) / 256;
#define LAUNCH_GELU(TAG)                                              \
  hipLaunchKernelGGL((gelu_and_mul_kernel<TAG>), dim3(rows, chunks),  \
                     dim3(256), 0, stream, (short*)out.data_ptr(),    \
                     (const short*)x.data_ptr(), d)
  DISPATCH_16B(x, LAUNCH_GELU);
#undef LAUNCH_GELU
  HIP_CHECK_KERNEL();
  return out;
}

void rotary_embedding(torch::Tensor positions, torch::Tensor q,
                      torch::Tensor k, torch::Tensor cos_sin_cache,
                      int64_t rot) {
  check_rows_16b(q, "q");
  check_rows_16b(k, "k");
  TORCH_CHECK(cos_sin_cache.scalar_type() == torch::kFloat32);
  const int T = q.size(0);
  const int num_q_heads = q.size(1);
  const int num_kv_heads = k.size(1);
  const int head_dim = q.size(2);
  auto stream = c10::hip::getCurrentHIPStream().stream();
#define LAUNCH_ROPE(TAG)                                                   \
  hipLaunchKernelGGL((rope_kernel<TAG>), dim3(T), dim3(256), 0, stream,    \
                     (short*)q.data_ptr(), (short*)k.data_ptr(),           \
                     positions.data_ptr<int64_t>(),                        \
                     cos_sin_cache.data_ptr<float>(), (int)rot, head_dim,  \
                     num_q_heads, num_kv_heads, q.stride(0), k.stride(0))
  DISPATCH_16B(q, LAUNCH_ROPE);
#undef LAUNCH_ROPE
  HIP_CHECK_KERNEL();
}

void reshape_and_cache(torch::Tensor key, torch::Tensor value,
                       torch::Tensor kv_cache, torch::Tensor slot_mapping) {
  check_rows_16b(key, "key");
  check_rows_16b(value, "value");
  TORCH_CHECK(kv_cache.is_contiguous(), "kv_cache must be contiguous");
  const int T = key.size(0);
  if (T == 0) return;
  const int num_kv_heads = key.size(1);
  const int head_dim = key.size(2);
  const int block_size = kv_cache.size(3);
  TORCH_CHECK(head_dim % 8 == 0);
  TORCH_CHECK(kv_cache.size(2) == num_kv_heads);
  const int64_t kv_stride = kv_cache.stride(0);
  auto stream = c10::hip::getCurrentHIPStream().stream();
#define LAUNCH_RC2(TAG, CTAG)                                          \
  hipLaunchKernelGGL((reshape_and_cache_kernel<TAG, CTAG>), dim3(T),   \
                     dim3(128), 0, stream,                             \
                     (const short*)key.data_ptr(),                     \
                     (const short*)value.data_ptr(),                   \
                     (CacheTraits<CTAG>::elem*)kv_cache.data_ptr(),    \
                     slot_mapping.data_ptr<int64_t>(), kv_stride,      \
                     key.stride(0), value.stride(0),                    \
                     num_kv_heads, head_dim, block_size)
  const bool fp8c = kv_cache.scalar_type() == torch::kFloat8_e4m3fn;
  if (key.scalar_type() == torch::kBFloat16) {
    if (fp8c) { LAUNCH_RC2(BF16Tag, FP8CacheTag); }
    else      { LAUNCH_RC2(BF16Tag, BF16Tag); }
  } else {
    if (fp8c) { LAUNCH_RC2(FP16Tag, FP8CacheTag); }
    else      { LAUNCH_RC2(FP16Tag, FP16Tag); }
  }
#undef LAUNCH_RC2
  HIP_CHECK_KERNEL();
}

}  // namespace vllm_amd
