// MLA decode attention for CDNA4 (gfx950).
//
// Absorbed-MQA decode over the compressed latent KV cache (blueprint:
// reference vllm/model_executor/layers/attention/mla_attention.py:40-190
// "data-movement friendly" decode path): queries are pre-absorbed
// through W_UK so every head attends with dim L+R (kv_lora 512 + rope
// 64 = 576) against ONE shared latent stream, and the value is the
// first L dims of the same stream. That makes decode a GEMM-shaped
// problem — Q[32 heads, 576] x K^T[576, 32 tok] and P[32, 32] x
// V[32 tok, 512] per tile — so this kernel runs on MFMA 16x16x32-bf16,
// unlike the GQA decode kernel (attention_decode.hip) which is
// VALU-bound at 1-16 queries/KV head.
//
// Block = 4 waves = 32 heads x 32-token KV subtiles:
//   QK:  wave(mtile=w&1, ntile=w>>1): 18 MFMAs over the 576 dim.
//   softmax: per-head online max/sum shared through LDS (wave-redundant
//   recompute from LDS keeps the update single-barrier).
//   PV:  wave w owns V columns [w*128, (w+1)*128); P staged [h][t] bf16.
// The KV subtile is staged once: [tok][576] (QK reads rows) plus a
// scatter-transposed copy of the first 512 dims [dim][tok] (PV reads
// rows) — ~70 KB LDS, 2 blocks/CU.
//
// Grid (num_decodes, num_parts, ceil(Hq/32)); partitions of 1024 tokens
// produce fp32 partials + (m, l) merged by mla_reduce_kernel — same
// flash-partition scheme as the GQA decode kernel.

#include <torch/all.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include "common.h"

namespace vllm_amd {

namespace mla {

constexpr int LORA = 512;
constexpr int ROPE = 64;
constexpr int DK = LORA + ROPE;  // 576
constexpr int HG = 32;           // heads per block
constexpr int SUB = 32;          // KV tokens per subtile
constexpr int PART = 1024;       // tokens per grid partition
constexpr int THREADS = 256;
constexpr float NEG_INF = -1e30f;

DEVINL int swz(int row, int byte_off) { return byte_off ^ ((row & 7) << 4); }

template <typename Tag>
__launch_bounds__(THREADS) __global__
void mla_decode_kernel(short* __restrict__ out,           // [T, Hq, LORA]
                       const short* __restrict__ q_nope,  // [T, Hq, LORA]
                       const short* __restrict__ q_pe,    // [T, Hq, ROPE]
                       const short* __restrict__ kv,      // [NB, BS, DK]
                       const int* __restrict__ block_table, int bt_stride,
                       const int* __restrict__ seq_lens, float scale,
                       int num_heads, int block_size,
                       float* __restrict__ tmp_out,  // [T, Hq, parts, LORA]
                       float* __restrict__ tmp_lse,  // [T, Hq, parts, 2]
                       int max_parts) {
  const int seq = blockIdx.x;
  const int part = blockIdx.y;
  const int hg0 = blockIdx.z * HG;
  const int ctx = seq_lens[seq];
  const int t_begin = part * PART;
  if (t_begin >= ctx) return;
  const int t_end = min(t_begin + PART, ctx);
  const bool single = gridDim.y == 1;

  __shared__ short k_lds[SUB * DK];       // [tok][576], swizzled 16B slots
  __shared__ short vt_lds[LORA * SUB];    // [dim][tok]
  __shared__ short p_lds[HG * SUB];       // [head][tok] bf16
  __shared__ float m_st[HG], l_st[HG], f_st[HG];
  __shared__ float wmax[2][HG], wsum[2][HG];

  const int tid = threadIdx.x;
  const int wave = tid / WAVE_SIZE;
  const int lane = tid % WAVE_SIZE;
  const int l16 = lane & 15;
  const int l4 = lane >> 4;
  const int mtile = wave & 1;
  const int ntile = wave >> 1;

  if (tid < HG) {
    m_st[tid] = NEG_INF;
    l_st[tid] = 0.f;
  }
  // Running max per owned head row, kept REDUNDANTLY in registers by
  // the two waves sharing an mtile (identical deterministic updates
  // from LDS wmax) — m_st is written only by owner lanes and read only
  // across barriers, so there is no read/write race in the softmax
  // phase.
  float m_reg[4] = {NEG_INF, NEG_INF, NEG_INF, NEG_INF};

  // Q fragments: wave's rows are heads mtile*16 + l16 of this group.
  const int head = hg0 + mtile * 16 + l16;
  const bool head_ok = head < num_heads;
  const long qn_row = ((long)seq * num_heads + (head_ok ? head : 0)) * LORA;
  const long qp_row = ((long)seq * num_heads + (head_ok ? head : 0)) * ROPE;
  s16x8 qfrag[18];
#pragma unroll
  for (int kf = 0; kf < 18; ++kf) {
    const int f = kf * 32 + l4 * 8;
    s16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
    if (head_ok)
      v = (f < LORA)
              ? *reinterpret_cast<const s16x8*>(q_nope + qn_row + f)
              : *reinterpret_cast<const s16x8*>(q_pe + qp_row + (f - LORA));
    qfrag[kf] = v;
  }

  // PV accumulators: wave owns V cols [wave*128, wave*128+128) for both
  // head mtiles: acc[mt][nf] covers heads mt*16.. x cols nf*16..
  f32x4 o_acc[2][8];
#pragma unroll
  for (int mt = 0; mt < 2; ++mt)
#pragma unroll
    for (int nf = 0; nf < 8; ++nf) o_acc[mt][nf] = f32x4{0.f, 0.f, 0.f, 0.f};

  const float sc2 = scale * 1.4426950408889634f;  // log2(e)
  __syncthreads();

  for (int sub0 = t_begin; sub0 < t_end; sub0 += SUB) {
    // ---- stage: 32 tokens x 1152B; 2304 16B chunks over 256 threads.
    {
      for (int c = tid; c < SUB * (DK / 8); c += THREADS) {
        const int tok = c / (DK / 8);
        const int f8 = c % (DK / 8);      // 8-elem feature chunk
        const int t_glob = sub0 + tok;
        s16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
        if (t_glob < t_end) {
          const int blk = block_table[seq * bt_stride + t_glob / block_size];
          v = *reinterpret_cast<const s16x8*>(
              kv + ((long)blk * block_size + t_glob % block_size) * DK +
              f8 * 8);
        }
        *reinterpret_cast<s16x8*>(reinterpret_cast<char*>(k_lds) +
                                  tok * (DK * 2) + swz(tok, f8 * 16)) = v;
        if (f8 * 8 < LORA) {
#pragma unroll
          for (int j = 0; j < 8; ++j)
            vt_lds[(f8 * 8 + j) * SUB + tok] = v[j];
        }
      }
    }
    __syncthreads();

    // ---- QK: S[16h x 16t] for (mtile, ntile).
    f32x4 s_frag = {0.f, 0.f, 0.f, 0.f};
    {
      const int tok = ntile * 16 + l16;  // B-frag col
#pragma unroll
      for (int kf = 0; kf < 18; ++kf) {
        s16x8 bfrag = *reinterpret_cast<const s16x8*>(
            reinterpret_cast<const char*>(k_lds) + tok * (DK * 2) +
            swz(tok, kf * 64 + l4 * 16));
        s_frag = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[kf], bfrag,
                                                         s_frag, 0, 0, 0);
      }
    }
    // C-frag: col=lane&15=token, row=l4*4+r=head-in-mtile.
    const int tok_g = sub0 + ntile * 16 + l16;
    const bool tok_ok = tok_g < t_end;
    float s_val[4];
    float wm = NEG_INF;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      s_val[r] = tok_ok ? s_frag[r] * sc2 : NEG_INF;
      // per-head max needs a reduce over tokens (lanes l16); do all 4
      // head rows: each lane carries 4 heads, reduce each row below.
    }
    // Reduce max over the 16 token lanes for each of the 4 head rows.
    float rowmax[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float v = s_val[r];
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        v = fmaxf(v, __shfl_xor(v, off, 64));
      rowmax[r] = v;
    }
    if (l16 == 0) {
#pragma unroll
      for (int r = 0; r < 4; ++r)
        wmax[ntile][mtile * 16 + l4 * 4 + r] = rowmax[r];
    }
    __syncthreads();

    // ---- softmax update (wave-redundant from LDS).
    float factor[4], m_new[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int h = mtile * 16 + l4 * 4 + r;
      const float mo = m_reg[r];
      m_new[r] = fmaxf(mo, fmaxf(wmax[0][h], wmax[1][h]));
      factor[r] = exp2f(mo - m_new[r]);  // exp2(-inf - x) = 0 first time
      m_reg[r] = m_new[r];
    }
    float rowsum[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const float p = tok_ok ? exp2f(s_val[r] - m_new[r]) : 0.f;
      p_lds[(mtile * 16 + l4 * 4 + r) * SUB + ntile * 16 + l16] =
          from_f32<Tag>(p);
      float v = p;
#pragma unroll
      for (int off = 8; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
      rowsum[r] = v;
    }
    if (l16 == 0) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int h = mtile * 16 + l4 * 4 + r;
        wsum[ntile][h] = rowsum[r];
        if (ntile == 0) {
          m_st[h] = m_new[r];
          f_st[h] = factor[r];
        }
      }
    }
    __syncthreads();

    // ---- PV: o_acc[mt] covers heads mt*16 + (l4*4 + r) rows.
#pragma unroll
    for (int mt = 0; mt < 2; ++mt) {
      // rescale accumulators by this subtile's factor per head row
#pragma unroll
      for (int nf = 0; nf < 8; ++nf) {
#pragma unroll
        for (int r = 0; r < 4; ++r)
          o_acc[mt][nf][r] *= f_st[mt * 16 + l4 * 4 + r];
      }
      const int arow = mt * 16 + l16;  // A-frag head row
      s16x8 afrag = *reinterpret_cast<const s16x8*>(
          p_lds + arow * SUB + l4 * 8);
#pragma unroll
      for (int nf = 0; nf < 8; ++nf) {
        const int col = wave * 128 + nf * 16 + l16;
        s16x8 bfrag = *reinterpret_cast<const s16x8*>(
            vt_lds + col * SUB + l4 * 8);
        o_acc[mt][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag, bfrag, o_acc[mt][nf], 0, 0, 0);
      }
    }
    // l state update (one lane per head).
    if (wave == 0 && tid < HG)
      l_st[tid] = l_st[tid] * f_st[tid] + wsum[0][tid] + wsum[1][tid];
    __syncthreads();
  }

  // ---- write out. o_acc rows: head = mt*16 + l4*4 + r; col = wave*128
  // + nf*16 + l16.
#pragma unroll
  for (int mt = 0; mt < 2; ++mt) {
#pragma unroll
    for (int nf = 0; nf < 8; ++nf) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int h = hg0 + mt * 16 + l4 * 4 + r;
        if (h >= num_heads) continue;
        const int col = wave * 128 + nf * 16 + l16;
        const float l = l_st[mt * 16 + l4 * 4 + r];
        if (single) {
          out[((long)seq * num_heads + h) * LORA + col] =
              from_f32<Tag>(o_acc[mt][nf][r] / l);
        } else {
          tmp_out[(((long)seq * num_heads + h) * max_parts + part) * LORA +
                  col] = o_acc[mt][nf][r];
        }
      }
    }
  }
  if (!single && wave == 0 && tid < HG) {
    const int h = hg0 + tid;
    if (h < num_heads) {
      const long base = (((long)seq * num_heads + h) * max_parts + part) * 2;
      tmp_lse[base] = m_st[tid];      // log2-domain max
      tmp_lse[base + 1] = l_st[tid];  // sum
    }
  }
}

// Merge partitions: out[s,h,:] = sum_p w_p * tmp[s,h,p,:] / sum_p w_p l_p
// with w_p = exp2(m_p - m_max). Grid (T, Hq), 256 threads over 512 cols.
template <typename Tag>
__global__ void mla_reduce_kernel(short* __restrict__ out,
                                  const float* __restrict__ tmp_out,
                                  const float* __restrict__ tmp_lse,
                                  const int* __restrict__ seq_lens,
                                  int num_heads, int max_parts) {
  const int seq = blockIdx.x;
  const int h = blockIdx.y;
  const int parts = min((seq_lens[seq] + PART - 1) / PART, max_parts);
  const long lse_base = (((long)seq * num_heads + h) * max_parts) * 2;
  float m = NEG_INF;
  for (int p = 0; p < parts; ++p) m = fmaxf(m, tmp_lse[lse_base + p * 2]);
  float l = 0.f;
  for (int p = 0; p < parts; ++p)
    l += exp2f(tmp_lse[lse_base + p * 2] - m) * tmp_lse[lse_base + p * 2 + 1];
  const float inv_l = 1.f / l;
  const long o_base = ((long)seq * num_heads + h) * (long)max_parts * LORA;
  for (int c = threadIdx.x; c < LORA; c += blockDim.x) {
    float acc = 0.f;
    for (int p = 0; p < parts; ++p)
      acc += exp2f(tmp_lse[lse_base + p * 2] - m) *
             tmp_out[o_base + (long)p * LORA + c];
    out[((long)seq * num_heads + h) * LORA + c] = from_f32<Tag>(acc * inv_l);
  }
}

}  // namespace mla

void mla_decode(torch::Tensor out, torch::Tensor q_nope, torch::Tensor q_pe,
                torch::Tensor kv_cache, torch::Tensor block_table,
                torch::Tensor seq_lens, double scale, int64_t max_seq_len,
                torch::Tensor tmp_out, torch::Tensor tmp_lse) {
  using namespace mla;
  TORCH_CHECK(q_nope.dim() == 3 && q_nope.size(2) == LORA,
              "mla_decode expects kv_lora_rank=512");
  TORCH_CHECK(q_pe.size(2) == ROPE, "mla_decode expects rope dim 64");
  TORCH_CHECK(kv_cache.size(2) == DK);
  TORCH_CHECK(q_nope.is_contiguous() && q_pe.is_contiguous() &&
              out.is_contiguous() && kv_cache.is_contiguous());
  TORCH_CHECK(out.scalar_type() == q_nope.scalar_type());
  const int T = (int)q_nope.size(0);
  const int Hq = (int)q_nope.size(1);
  const int bs = (int)kv_cache.size(1);
  const int parts = std::max<int>(1, (int)((max_seq_len + PART - 1) / PART));
  if (parts > 1)
    TORCH_CHECK(tmp_out.numel() >= (long)T * Hq * parts * LORA &&
                tmp_lse.numel() >= (long)T * Hq * parts * 2);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  dim3 grid(T, parts, ceil_div(Hq, HG));
  auto launch = [&](auto tag) {
    using Tag = decltype(tag);
    hipLaunchKernelGGL(
        (mla_decode_kernel<Tag>), grid, dim3(THREADS), 0, stream,
        (short*)out.data_ptr(), (const short*)q_nope.data_ptr(),
        (const short*)q_pe.data_ptr(), (const short*)kv_cache.data_ptr(),
        block_table.data_ptr<int>(), (int)block_table.size(1),
        seq_lens.data_ptr<int>(), (float)scale, Hq, bs,
        parts > 1 ? tmp_out.data_ptr<float>() : nullptr,
        parts > 1 ? tmp_lse.data_ptr<float>() : nullptr, parts);
    if (parts > 1) {
      hipLaunchKernelGGL((mla_reduce_kernel<Tag>), dim3(T, Hq), dim3(256),
                         0, stream, (short*)out.data_ptr(),
                         tmp_out.data_ptr<float>(),
                         tmp_lse.data_ptr<float>(),
                         seq_lens.data_ptr<int>(), Hq, parts);
    }
  };
  if (q_nope.scalar_type() == torch::kBFloat16)
    launch(BF16Tag{});
  else if (q_nope.scalar_type() == torch::kHalf)
    launch(FP16Tag{});
  else
    TORCH_CHECK(false, "mla_decode: bf16/fp16 only");
  HIP_CHECK_KERNEL();
}

}  // namespace vllm_amd
