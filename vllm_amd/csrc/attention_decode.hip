// Paged decode attention for MI355X (gfx950, CDNA4).
//
// Design (MI355X-first, not a port):
//   Decode attention is HBM-bandwidth bound: per context token and KV head
//   the kernel reads 2*D*2 bytes of K+V and does GQA_group*4*D flops —
//   arithmetic intensity = group flops/byte (≈4-8), far below the
//   157 TF / 6.3 TB/s = 25 flops/byte VALU roofline. So this kernel is a
//   pure-VALU streaming kernel tuned for the load path, not MFMA:
//   16B/lane short8 K/V loads (G13), 16-lane thread groups per token,
//   flash-decoding partitions of PART tokens with a separate LSE-merge
//   reduce kernel for long contexts.
//
//   The KV cache layout [2, blocks, kv_heads, block_size=64, D] keeps each
//   (block, head) tile contiguous (16 KB at D=128), so a partition is a
//   handful of fully-coalesced streams.
//
// Role of the reference's paged_attention (csrc/rocm/attention.cu:321,1420),
// re-designed: the reference packs GQA into MFMA16 tiles; on MI355X the
// memory-bound regime makes VALU dot products simpler and just as fast.

#include <torch/all.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace vllm_amd {

// Tunables. PART must be a multiple of the token-iteration width
// (BLOCK/16 tokens per iteration) and of the cache block size (64).
constexpr int DEC_BLOCK = 256;          // threads per workgroup (4 waves)
constexpr int DEC_PART = 512;           // context tokens per partition
constexpr int MAX_GROUP = 8;            // max GQA ratio handled in registers

// One workgroup: (decode seq, kv head, partition). Computes the partial
// attention output for GROUP query heads over PART context tokens with an
// unnormalized softmax (running max m, exp-sum l), writing either the
// final bf16 output (single partition) or fp32 partials + (m, l) scratch.
//
// Thread layout: 16 lanes per token (lane d covers dims 8d..8d+7 as one
// short8 = 16 B load), so a 256-thread block streams 16 tokens per
// iteration; DEC_PART/16 = 32 iterations.
// HD = head_dim (64/128/256). LPT = HD/8 lanes cover one token's dims
// (8 dims per lane, one 16 B load); SLOTS = DEC_BLOCK/LPT token slots
// stream in parallel, two tokens per slot-iteration.
template <typename Tag, typename CTag, int GROUP, bool FINAL, int HD>
__global__ __launch_bounds__(DEC_BLOCK) void paged_decode_kernel(
    short* __restrict__ out,            // FINAL: [Tdec, Hq, D] (16-bit)
    float* __restrict__ tmp_out,        // else: [Tdec, Hq, parts, D]
    float* __restrict__ tmp_lse,        // else: [Tdec, Hq, parts, 2]
    const short* __restrict__ q,        // [Tdec, Hq, D]
    const typename CacheTraits<CTag>::elem* __restrict__ kv_cache,
                                        // [2, blocks, Hkv, 64, D]
    const int* __restrict__ block_table,   // [num_reqs, max_blocks]
    const int* __restrict__ seq_lens,      // [num_reqs]
    const float scale, const int num_q_heads, const int num_kv_heads,
    const int head_dim, const int max_blocks_per_req, const int num_parts,
    const int64_t kv_plane_stride,      // elements between K and V planes
    const int64_t q_stride,             // elements between q token rows
    const int sliding_window) {
  const int seq = blockIdx.x;       // decode row == request row
  const int kvh = blockIdx.y;
  const int part = blockIdx.z;
  const int ctx = seq_lens[seq];

  int t_begin = part * DEC_PART;
  int t_end = min(ctx, t_begin + DEC_PART);
  if (sliding_window > 0) {
    // Query position is ctx-1; keys < ctx - sliding_window are masked.
    t_begin = max(t_begin, ctx - sliding_window);
  }
  const int hq0 = kvh * GROUP;  // first query head of this group

  if (t_begin >= t_end) {
    if (!FINAL && threadIdx.x < GROUP * 2) {
      const int g = threadIdx.x / 2;
      float* lse = tmp_lse +
          (((int64_t)seq * num_q_heads + hq0 + g) * num_parts + part) * 2;
      lse[threadIdx.x % 2] = (threadIdx.x % 2 == 0) ? -3.0e38f : 0.f;
    }
    return;
  }

  constexpr int LPT = HD / 8;                   // lanes per token
  constexpr int SLOTS_PER_WAVE = WAVE_SIZE / LPT;
  constexpr int SLOTS = DEC_BLOCK / LPT;
  const int lane_d = threadIdx.x & (LPT - 1);   // dim-slice owner
  const int slot = (threadIdx.x / LPT) % SLOTS_PER_WAVE;
  const int wave = threadIdx.x / WAVE_SIZE;     // 0..3

  // LDS: physical block ids + cross-wave merge scratch.
  __shared__ int blk_ids[DEC_PART / 64];
  __shared__ float ml_red[4][GROUP][2];       // per-wave (m, l)
  __shared__ float o_red[4][GROUP][HD];       // per-wave accumulators

  if (threadIdx.x < DEC_PART / 64) {
    const int cache_blk = (part * DEC_PART) / 64 + threadIdx.x;
    blk_ids[threadIdx.x] =
        block_table[(int64_t)seq * max_blocks_per_req + cache_blk];
  }
  __syncthreads();

  // Q fragment: each lane16 holds its 8-dim slice for all GROUP heads.
  float qf[GROUP][8];
  {
    const short* qbase =
        q + (int64_t)seq * q_stride + hq0 * head_dim + lane_d * 8;
#pragma unroll
    for (int g = 0; g < GROUP; ++g) {
      s16x8 v = *reinterpret_cast<const s16x8*>(qbase + g * head_dim);
#pragma unroll
      for (int j = 0; j < 8; ++j) qf[g][j] = to_f32<Tag>(v[j]) * scale;
    }
  }

  // ---- single pass: online softmax per (wave, slot) ----------------------
  // Each slot streams tokens t = t_begin + wave*4 + slot + 16*i. K and V
  // rows are read once, 16 B per lane (G13); no syncthreads in the loop, so
  // the 4 waves keep the HBM pipeline full independently.
  const int64_t head_tile_stride = (int64_t)64 * head_dim;  // one (blk,head)
  float m_s[GROUP], l_s[GROUP], acc[GROUP][8];
#pragma unroll
  for (int g = 0; g < GROUP; ++g) {
    m_s[g] = -3.0e38f;
    l_s[g] = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[g][j] = 0.f;
  }

  // Two tokens per (wave, slot) iteration: doubles the K/V loads in
  // flight per lane and halves the online-softmax update chain.
  const int t0_base = t_begin + wave * SLOTS_PER_WAVE + slot;
  for (int t = t0_base; t < t_end; t += 2 * SLOTS) {
    const int t1 = t + SLOTS;
    const bool has1 = t1 < t_end;
    using CT = CacheTraits<CTag>;
    using cvec = typename CT::vec8;
    const int local0 = t - part * DEC_PART;
    const typename CT::elem* base0 = kv_cache +
        ((int64_t)blk_ids[local0 / 64] * num_kv_heads + kvh) *
            head_tile_stride +
        (int64_t)(t % 64) * head_dim + lane_d * 8;
    const int local1 = has1 ? t1 - part * DEC_PART : local0;
    const typename CT::elem* base1 = has1 ? kv_cache +
        ((int64_t)blk_ids[local1 / 64] * num_kv_heads + kvh) *
            head_tile_stride +
        (int64_t)(t1 % 64) * head_dim + lane_d * 8 : base0;
    cvec k0 = *reinterpret_cast<const cvec*>(base0);
    cvec v0 = *reinterpret_cast<const cvec*>(base0 + kv_plane_stride);
    cvec k1 = *reinterpret_cast<const cvec*>(base1);
    cvec v1 = *reinterpret_cast<const cvec*>(base1 + kv_plane_stride);
    float kf0[8], vf0[8], kf1[8], vf1[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      kf0[j] = CT::get(k0, j);
      vf0[j] = CT::get(v0, j);
      kf1[j] = CT::get(k1, j);
      vf1[j] = CT::get(v1, j);
    }
#pragma unroll
    for (int g = 0; g < GROUP; ++g) {
      float sc0 = 0.f, sc1 = 0.f;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        sc0 += qf[g][j] * kf0[j];
        sc1 += qf[g][j] * kf1[j];
      }
      // Reduce over the LPT dim lanes of this slot (consecutive lanes).
#pragma unroll
      for (int off = 1; off < LPT; off <<= 1) {
        sc0 += __shfl_xor(sc0, off, 64);
        sc1 += __shfl_xor(sc1, off, 64);
      }
      if (!has1) sc1 = -3.0e38f;
      const float m_new = fmaxf(m_s[g], fmaxf(sc0, sc1));
      const float p0 = __expf(sc0 - m_new);
      const float p1 = has1 ? __expf(sc1 - m_new) : 0.f;
      if (m_new > m_s[g]) {
        const float corr = __expf(m_s[g] - m_new);
        l_s[g] = l_s[g] * corr + p0 + p1;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          acc[g][j] = acc[g][j] * corr + p0 * vf0[j] + p1 * vf1[j];
        m_s[g] = m_new;
      } else {
        l_s[g] += p0 + p1;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          acc[g][j] += p0 * vf0[j] + p1 * vf1[j];
      }
    }
  }

  // ---- merge the slots within each wave (butterfly over lanes LPT..32) ---
#pragma unroll
  for (int off = LPT; off <= 32; off <<= 1) {
#pragma unroll
    for (int g = 0; g < GROUP; ++g) {
      const float m_o = __shfl_xor(m_s[g], off, 64);
      const float l_o = __shfl_xor(l_s[g], off, 64);
      const float M = fmaxf(m_s[g], m_o);
      const float c1 = __expf(m_s[g] - M);
      const float c2 = __expf(m_o - M);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float a_o = __shfl_xor(acc[g][j], off, 64);
        acc[g][j] = acc[g][j] * c1 + a_o * c2;
      }
      l_s[g] = l_s[g] * c1 + l_o * c2;
      m_s[g] = M;
    }
  }

  // ---- merge the 4 waves via LDS ------------------------------------------
  if ((threadIdx.x & 63) < LPT) {
#pragma unroll
    for (int g = 0; g < GROUP; ++g) {
#pragma unroll
      for (int j = 0; j < 8; ++j) o_red[wave][g][lane_d * 8 + j] = acc[g][j];
      if (lane_d == 0) {
        ml_red[wave][g][0] = m_s[g];
        ml_red[wave][g][1] = l_s[g];
      }
    }
  }
  __syncthreads();
  // HD threads: one per dim; every thread recomputes the scalar merge.
  if (threadIdx.x < HD) {
    const int d = threadIdx.x;
#pragma unroll
    for (int g = 0; g < GROUP; ++g) {
      float M = fmaxf(fmaxf(ml_red[0][g][0], ml_red[1][g][0]),
                      fmaxf(ml_red[2][g][0], ml_red[3][g][0]));
      float L = 0.f, o = 0.f;
#pragma unroll
      for (int w = 0; w < 4; ++w) {
        const float c = __expf(ml_red[w][g][0] - M);
        L += ml_red[w][g][1] * c;
        o += o_red[w][g][d] * c;
      }
      if (FINAL) {
        const float inv_l = L > 0.f ? 1.f / L : 0.f;
        out[((int64_t)seq * num_q_heads + hq0 + g) * head_dim + d] =
            from_f32<Tag>(o * inv_l);
      } else {
        tmp_out[((((int64_t)seq * num_q_heads + hq0 + g) * num_parts + part) *
                 head_dim) + d] = o;
        if (d == 0) {
          float* lse = tmp_lse +
              (((int64_t)seq * num_q_heads + hq0 + g) * num_parts + part) * 2;
          lse[0] = M;
          lse[1] = L;
        }
      }
    }
  }
}

// Merge per-partition partials: out = sum_p o_p * exp(m_p - M) / sum_p
// l_p * exp(m_p - M). Grid (Tdec, Hq), block = D threads (one per dim).
template <typename Tag>
__global__ void paged_decode_reduce_kernel(
    short* __restrict__ out,           // [Tdec, Hq, D]
    const float* __restrict__ tmp_out, // [Tdec, Hq, parts, D]
    const float* __restrict__ tmp_lse, // [Tdec, Hq, parts, 2]
    const int* __restrict__ seq_lens, const int num_parts,
    const int head_dim, const int num_q_heads) {
  const int seq = blockIdx.x;
  const int h = blockIdx.y;
  const int d = threadIdx.x;
  const int used_parts = min(num_parts, (seq_lens[seq] + DEC_PART - 1) /
                                            DEC_PART);
  const float* lse =
      tmp_lse + (((int64_t)seq * num_q_heads + h) * num_parts) * 2;
  float M = -3.0e38f;
  for (int p = 0; p < used_parts; ++p) M = fmaxf(M, lse[p * 2]);
  float L = 0.f;
  for (int p = 0; p < used_parts; ++p)
    L += lse[p * 2 + 1] * __expf(lse[p * 2] - M);
  const float inv_l = L > 0.f ? 1.f / L : 0.f;
  const float* o_base =
      tmp_out + (((int64_t)seq * num_q_heads + h) * num_parts) * head_dim;
  float o = 0.f;
  for (int p = 0; p < used_parts; ++p)
    o += o_base[p * head_dim + d] * __expf(lse[p * 2] - M);
  out[((int64_t)seq * num_q_heads + h) * head_dim + d] =
      from_f32<Tag>(o * inv_l);
}

// ---------------------------------------------------------------------------
// Host launcher. q: [Tdec, Hq, D] (contiguous slice of the step's q);
// out: [Tdec, Hq, D]; kv_cache: [2, blocks, Hkv, 64, D];
// block_table/seq_lens cover the decode rows (rows 0..Tdec-1).
void paged_decode_attention(torch::Tensor out, torch::Tensor q,
                            torch::Tensor kv_cache,
                            torch::Tensor block_table,
                            torch::Tensor seq_lens, double scale,
                            int64_t max_seq_len, int64_t sliding_window,
                            torch::Tensor tmp_out, torch::Tensor tmp_lse) {
  const int num_seqs = q.size(0);
  if (num_seqs == 0) return;
  const int num_q_heads = q.size(1);
  const int head_dim = q.size(2);
  const int num_kv_heads = kv_cache.size(2);
  const int group = num_q_heads / num_kv_heads;
  TORCH_CHECK(head_dim == 64 || head_dim == 128 || head_dim == 256,
              "decode kernel supports head_dim 64/128/256, got ", head_dim);
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == head_dim &&
              q.stride(0) % 8 == 0, "q must be head-contiguous [T,H,D]");
  TORCH_CHECK(out.is_contiguous(), "out must be contiguous");
  TORCH_CHECK(kv_cache.size(3) == 64, "cache block_size must be 64");
  TORCH_CHECK(group >= 1 && group <= MAX_GROUP && num_q_heads ==
              group * num_kv_heads, "GQA group must be 1..8, got ", group);
  const int num_parts =
      (int)((max_seq_len + DEC_PART - 1) / DEC_PART);
  const int max_blocks = block_table.size(1);
  auto stream = c10::hip::getCurrentHIPStream().stream();

  dim3 grid(num_seqs, num_kv_heads, num_parts);
  dim3 block(DEC_BLOCK);

#define LAUNCH_DEC_HD(TAG, CTAG, G, FIN, HD)                                 \
  hipLaunchKernelGGL((paged_decode_kernel<TAG, CTAG, G, FIN, HD>), grid,     \
                     block,                                                  \
                     0, stream, (short*)out.data_ptr(),                      \
                     FIN ? nullptr : tmp_out.data_ptr<float>(),              \
                     FIN ? nullptr : tmp_lse.data_ptr<float>(),              \
                     (const short*)q.data_ptr(),                             \
                     (const CacheTraits<CTAG>::elem*)kv_cache.data_ptr(),    \
                     block_table.data_ptr<int>(), seq_lens.data_ptr<int>(),  \
                     (float)scale, num_q_heads, num_kv_heads, head_dim,      \
                     max_blocks, num_parts, kv_cache.stride(0),              \
                     q.stride(0), (int)sliding_window)

#define LAUNCH_DEC(TAG, CTAG, G, FIN)                                        \
  switch (head_dim) {                                                        \
    case 64: LAUNCH_DEC_HD(TAG, CTAG, G, FIN, 64); break;                    \
    case 256: LAUNCH_DEC_HD(TAG, CTAG, G, FIN, 256); break;                  \
    default: LAUNCH_DEC_HD(TAG, CTAG, G, FIN, 128); break;                   \
  }

#define DISPATCH_GROUP(TAG, CTAG, FIN)                                       \
  switch (group) {                                                           \
    case 1: LAUNCH_DEC(TAG, CTAG, 1, FIN); break;                            \
    case 2: LAUNCH_DEC(TAG, CTAG, 2, FIN); break;                            \
    case 4: LAUNCH_DEC(TAG, CTAG, 4, FIN); break;                            \
    case 5: LAUNCH_DEC(TAG, CTAG, 5, FIN); break;                            \
    case 6: LAUNCH_DEC(TAG, CTAG, 6, FIN); break;                            \
    case 8: LAUNCH_DEC(TAG, CTAG, 8, FIN); break;                            \
    case 3: LAUNCH_DEC(TAG, CTAG, 3, FIN); break;                            \
    case 7: LAUNCH_DEC(TAG, CTAG, 7, FIN); break;                            \
    default: TORCH_CHECK(false, "unsupported GQA group ", group);            \
  }

#define DISPATCH_CACHE(TAG, FIN)                                             \
  if (kv_cache.scalar_type() == torch::kFloat8_e4m3fn) {                     \
    DISPATCH_GROUP(TAG, FP8CacheTag, FIN);                                   \
  } else {                                                                   \
    DISPATCH_GROUP(TAG, TAG, FIN);                                           \
  }

  const bool is_bf16 = q.scalar_type() == torch::kBFloat16;
  if (num_parts == 1) {
    if (is_bf16) { DISPATCH_CACHE(BF16Tag, true); }
    else         { DISPATCH_CACHE(FP16Tag, true); }
  } else {
    if (is_bf16) { DISPATCH_CACHE(BF16Tag, false); }
    else         { DISPATCH_CACHE(FP16Tag, false); }
    HIP_CHECK_KERNEL();
    dim3 rgrid(num_seqs, num_q_heads);
    if (is_bf16) {
      hipLaunchKernelGGL((paged_decode_reduce_kernel<BF16Tag>), rgrid,
                         dim3(head_dim), 0, stream, (short*)out.data_ptr(),
                         tmp_out.data_ptr<float>(),
                         tmp_lse.data_ptr<float>(), seq_lens.data_ptr<int>(),
                         num_parts, head_dim, num_q_heads);
    } else {
      hipLaunchKernelGGL((paged_decode_reduce_kernel<FP16Tag>), rgrid,
                         dim3(head_dim), 0, stream, (short*)out.data_ptr(),
                         tmp_out.data_ptr<float>(),
                         tmp_lse.data_ptr<float>(), seq_lens.data_ptr<int>(),
                         num_parts, head_dim, num_q_heads);
    }
  }
  HIP_CHECK_KERNEL();
#undef DISPATCH_CACHE
#undef DISPATCH_GROUP
#undef LAUNCH_DEC
#undef LAUNCH_DEC_HD
}

}  // namespace vllm_amd
