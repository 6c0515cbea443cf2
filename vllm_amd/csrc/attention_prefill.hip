// Flash-style varlen prefill attention over the paged KV cache for
// MI355X (gfx950, CDNA4).
//
// Design (MI355X-first):
//   - MFMA 16x16x32 bf16 tiles; 4 waves per workgroup, each wave owns a
//     16-row Q subtile of a 64-row Q tile; KV tiles of 64 tokens = exactly
//     one (cache block, head) contiguous 16 KB chunk of the head-major
//     cache layout [2, blocks, Hkv, 64, D].
//   - K staged row-major [64 tok][128 d] in LDS with the st-style XOR
//     swizzle (byte ^= (tok&7)<<4) so the B-fragment ds_read_b128 across 16
//     token-rows is bank-conflict-free (guide §6 G4: row-major D=128 is a
//     32-way conflict otherwise).
//   - V staged TRANSPOSED [128 d][64 tok] (same XOR swizzle on d) so the
//     PV step computes O^T = V^T · P^T with per-lane-contiguous token runs
//     for both operands; P round-trips through a small per-wave LDS tile.
//   - Online softmax per q row in exp2 domain with the scale folded in;
//     no S matrix ever materialized.
//
// Role of the reference's triton_unified_attention.py:179 (prefill side)
// and prefix_prefill.py — re-designed for CDNA4 wave64 MFMA, not ported.

#include <torch/all.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace vllm_amd {

constexpr int PF_BLOCK = 256;  // 4 waves
constexpr int QTILE = 64;      // q rows per workgroup (16 per wave)
constexpr int KVTILE = 64;     // keys per inner tile == cache block_size
constexpr float LOG2E = 1.4426950408889634f;

// LDS byte-offset XOR swizzle: spread the 16 same-column rows of a
// row-major tile across 8 distinct 16B slots (guide §6 Guideline 4).
DEVINL int swz(int row, int col_byte) {
  return (col_byte ^ ((row & 7) << 4));
}

template <typename Tag, typename CTag, int HEAD_DIM>
__global__ __launch_bounds__(PF_BLOCK) void prefill_attention_kernel(
    short* __restrict__ out,             // [T, Hq, D]
    const short* __restrict__ q,         // [T, Hq, D]
    const typename CacheTraits<CTag>::elem* __restrict__ kv_cache,
                                         // [2, blocks, Hkv, 64, D]
    const int* __restrict__ block_table, // [num_reqs, max_blocks]
    const int* __restrict__ query_start_loc,  // [num_reqs+1]
    const int* __restrict__ seq_lens,         // [num_reqs]
    const float scale, const int num_q_heads, const int num_kv_heads,
    const int max_blocks_per_req, const int num_decodes,
    const int64_t kv_plane_stride, const int64_t q_stride,
    const int sliding_window) {
  const int req = num_decodes + blockIdx.z;
  const int h = blockIdx.y;
  const int kvh = h / (num_q_heads / num_kv_heads);
  const int q_start = query_start_loc[req];
  const int ql = query_start_loc[req + 1] - q_start;
  const int qtile0 = blockIdx.x * QTILE;
  if (qtile0 >= ql) return;
  const int ctx = seq_lens[req];

  const int wave = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;
  const int l16 = lane & 15;
  const int l4 = lane >> 4;  // 0..3

  // LDS tiles (bf16 stored as short).
  __shared__ short k_lds[KVTILE * HEAD_DIM];        // [tok][d], swizzled
  __shared__ short vt_lds[HEAD_DIM * KVTILE];       // [d][tok], swizzled
  __shared__ short p_lds[4][16 * 80];               // per wave [q][80] pad
  __shared__ float f_lds[4][16];                    // per wave factor/l swap

  // --- Q fragments: wave owns q rows qtile0 + wave*16 + [0,16) ------------
  // A-frag layout (16x16x32): lane&15 = row, elems k = (lane>>4)*8 + j.
  const int q_row_frag = qtile0 + wave * 16 + l16;           // for A-frags
  const int q_row_safe = min(q_row_frag, ql - 1);
  s16x8 qfrag[HEAD_DIM / 32];
  {
    const short* qbase =
        q + (int64_t)(q_start + q_row_safe) * q_stride + (int64_t)h * HEAD_DIM;
#pragma unroll
    for (int kk = 0; kk < HEAD_DIM / 32; ++kk)
      qfrag[kk] =
          *reinterpret_cast<const s16x8*>(qbase + kk * 32 + l4 * 8);
  }

  // Online-softmax state, in the exp2 domain with scale folded in.
  // m/l indexed by the S C-fragment rows this lane reduces:
  // q row (within wave tile) = l4*4 + reg.
  float m_st[4], l_st[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_st[r] = -3.0e38f; l_st[r] = 0.f; }
  // O^T accumulator: 8 d-subtiles; C layout col(lane&15)=q row,
  // row=(lane>>4)*4+reg = d within subtile.
  f32x4 o_acc[HEAD_DIM / 16];
#pragma unroll
  for (int i = 0; i < HEAD_DIM / 16; ++i)
    o_acc[i] = f32x4{0.f, 0.f, 0.f, 0.f};

  const float scale_log2e = scale * LOG2E;
  const int pos0 = ctx - ql;  // global position of local q row 0
  const int rows_here = min(QTILE, ql - qtile0);
  const int kv_end = pos0 + qtile0 + rows_here;  // last row's pos + 1
  int kt0 = 0;
  if (sliding_window > 0) {
    const int first_key = max(0, pos0 + qtile0 - sliding_window + 1);
    kt0 = first_key / KVTILE;
  }
  const int64_t head_tile = (int64_t)KVTILE * HEAD_DIM;

  for (int kt = kt0; kt * KVTILE < kv_end; ++kt) {
    // --- stage K [64][128] and V^T [128][64] (all 4 waves cooperate) ----
    using CT = CacheTraits<CTag>;
    using cvec = typename CT::vec8;
    const int phys = block_table[(int64_t)req * max_blocks_per_req + kt];
    const typename CT::elem* ksrc =
        kv_cache + ((int64_t)phys * num_kv_heads + kvh) * head_tile;
    const typename CT::elem* vsrc = ksrc + kv_plane_stride;
    // K: 256 threads x (KVTILE*D/8/256) iters x 8 elements, swizzled
    // ds_write_b128. (fp8 cache: converted to the compute dtype while
    // staging to LDS — the MFMA tiles always run bf16/fp16.)
    constexpr int CPR = HEAD_DIM / 8;  // 8-elem chunks per row
#pragma unroll
    for (int it = 0; it < KVTILE * CPR / PF_BLOCK; ++it) {
      const int vec = it * PF_BLOCK + threadIdx.x;  // 8-elem chunk index
      const int tok = vec / CPR;
      const int cb = (vec % CPR) * 16;              // col byte
      cvec kraw = *reinterpret_cast<const cvec*>(ksrc + vec * 8);
      s16x8 kv8;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        kv8[j] = from_f32<Tag>(CT::get(kraw, j));
      *reinterpret_cast<s16x8*>(
          reinterpret_cast<char*>(k_lds) + tok * (HEAD_DIM * 2) +
          swz(tok, cb)) = kv8;
      // V: read the same shape, scatter-transpose into vt_lds.
      cvec vraw = *reinterpret_cast<const cvec*>(vsrc + vec * 8);
      const int d0 = (vec % CPR) * 8;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int d = d0 + j;
        *reinterpret_cast<short*>(
            reinterpret_cast<char*>(vt_lds) + d * 128 +
            swz(d, tok * 2)) = from_f32<Tag>(CT::get(vraw, j));
      }
    }
    __syncthreads();

    // --- S = Q K^T for this wave's 16 q rows, 64 keys -------------------
    f32x4 s_frag[4];
#pragma unroll
    for (int n = 0; n < 4; ++n) s_frag[n] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int n = 0; n < 4; ++n) {       // key subtile
      const int tok = n * 16 + l16;     // B-frag col = token
#pragma unroll
      for (int kk = 0; kk < HEAD_DIM / 32; ++kk) {  // d chunk of 32
        const int cb = (kk * 32 + l4 * 8) * 2;
        s16x8 bfrag = *reinterpret_cast<const s16x8*>(
            reinterpret_cast<const char*>(k_lds) + tok * (HEAD_DIM * 2) +
            swz(tok, cb));
        s_frag[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            qfrag[kk], bfrag, s_frag[n], 0, 0, 0);
      }
    }

    // --- mask + online softmax (exp2 domain) ----------------------------
    // S C-frag: col = lane&15 = token-in-subtile; row = l4*4 + reg = q row.
    float p[4][4];   // [key subtile][reg]
    float rmax[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) rmax[r] = -3.0e38f;
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      const int kpos = kt * KVTILE + n * 16 + l16;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow = qtile0 + wave * 16 + l4 * 4 + r;
        const int qpos = pos0 + qrow;
        float s = s_frag[n][r] * scale_log2e;
        const bool masked = (kpos > qpos) || (qrow >= ql) ||
            (sliding_window > 0 && kpos <= qpos - sliding_window);
        p[n][r] = masked ? -3.0e38f : s;
        rmax[r] = fmaxf(rmax[r], p[n][r]);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      rmax[r] = group16_reduce_max(rmax[r]);
      const float m_new = fmaxf(m_st[r], rmax[r]);
      const float factor = exp2f(m_st[r] - m_new);
      l_st[r] *= factor;
      m_st[r] = m_new;
      // stash rescale factor for the O^T lanes (col=q layout)
      if (l16 == 0) f_lds[wave][l4 * 4 + r] = factor;
    }
    float rsum[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int n = 0; n < 4; ++n)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float pv = exp2f(p[n][r] - m_st[r]);
        p[n][r] = pv;
        rsum[r] += pv;
        // P^T staged for PV: p_lds[q row][token], rows padded to 80.
        p_lds[wave][(l4 * 4 + r) * 80 + n * 16 + l16] = from_f32<Tag>(pv);
      }
#pragma unroll
    for (int r = 0; r < 4; ++r)
      l_st[r] += group16_reduce_sum(rsum[r]);

    __syncthreads();  // f_lds/p_lds visible; also guards k/vt re-stage

    // --- rescale O^T and accumulate PV -----------------------------------
    const float fac = f_lds[wave][l16];  // this lane's q column factor
#pragma unroll
    for (int i = 0; i < HEAD_DIM / 16; ++i)
#pragma unroll
      for (int r = 0; r < 4; ++r) o_acc[i][r] *= fac;

    // O^T[d, q] += V^T[d, tok] * P^T[tok, q]:
    //   A-frag from vt_lds: row = d = msub*16 + l16, k = kc*32 + l4*8 + j
    //   B-frag from p_lds:  col = q = l16,          k = kc*32 + l4*8 + j
#pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      const int tb = (kc * 32 + l4 * 8) * 2;  // token byte offset
      s16x8 pfrag = *reinterpret_cast<const s16x8*>(
          &p_lds[wave][l16 * 80] + (kc * 32 + l4 * 8));
#pragma unroll
      for (int msub = 0; msub < HEAD_DIM / 16; ++msub) {
        const int d = msub * 16 + l16;
        s16x8 afrag = *reinterpret_cast<const s16x8*>(
            reinterpret_cast<const char*>(vt_lds) + d * 128 + swz(d, tb));
        o_acc[msub] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag, pfrag, o_acc[msub], 0, 0, 0);
      }
    }
    __syncthreads();  // done with k/vt/p for this tile
  }

  // --- epilogue: normalize and write O --------------------------------
  // Redistribute l to the O^T col=q layout.
#pragma unroll
  for (int r = 0; r < 4; ++r)
    if (l16 == 0) f_lds[wave][l4 * 4 + r] = l_st[r];
  __syncthreads();
  const int qrow_o = qtile0 + wave * 16 + l16;
  if (qrow_o < ql) {
    const float l = f_lds[wave][l16];
    const float inv_l = l > 0.f ? 1.f / l : 0.f;
    short* obase =
        out + ((int64_t)(q_start + qrow_o) * num_q_heads + h) * HEAD_DIM;
#pragma unroll
    for (int msub = 0; msub < HEAD_DIM / 16; ++msub)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        obase[msub * 16 + l4 * 4 + r] = from_f32<Tag>(o_acc[msub][r] * inv_l);
  }
}

// ---------------------------------------------------------------------------
// Host launcher. q/out: [T, Hq, D] covering the WHOLE step batch;
// rows for prefill requests are query_start_loc[num_decodes..num_reqs].
void prefill_attention(torch::Tensor out, torch::Tensor q,
                       torch::Tensor kv_cache, torch::Tensor block_table,
                       torch::Tensor query_start_loc, torch::Tensor seq_lens,
                       double scale, int64_t num_decodes,
                       int64_t max_query_len, int64_t sliding_window) {
  const int num_reqs = seq_lens.size(0);
  const int num_prefills = num_reqs - (int)num_decodes;
  if (num_prefills == 0) return;
  const int num_q_heads = q.size(1);
  const int head_dim = q.size(2);
  const int num_kv_heads = kv_cache.size(2);
  TORCH_CHECK(head_dim == 64 || head_dim == 128 || head_dim == 256,
              "prefill kernel supports head_dim 64/128/256");
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == head_dim &&
              q.stride(0) % 8 == 0, "q must be head-contiguous [T,H,D]");
  TORCH_CHECK(out.is_contiguous(), "out must be contiguous");
  TORCH_CHECK(kv_cache.size(3) == KVTILE, "cache block_size must be 64");
  TORCH_CHECK(num_q_heads % num_kv_heads == 0);
  const int max_tiles = (int)((max_query_len + QTILE - 1) / QTILE);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  dim3 grid(max_tiles, num_q_heads, num_prefills);

#define LAUNCH_PF(TAG, CTAG)                                                 \
  hipLaunchKernelGGL((prefill_attention_kernel<TAG, CTAG, D>), grid,         \
                     dim3(PF_BLOCK),                                         \
                     0, stream, (short*)out.data_ptr(),                      \
                     (const short*)q.data_ptr(),                             \
                     (const CacheTraits<CTAG>::elem*)kv_cache.data_ptr(),    \
                     block_table.data_ptr<int>(),                            \
                     query_start_loc.data_ptr<int>(),                        \
                     seq_lens.data_ptr<int>(), (float)scale, num_q_heads,    \
                     num_kv_heads, (int)block_table.size(1),                 \
                     (int)num_decodes, kv_cache.stride(0),                   \
                     q.stride(0), (int)sliding_window)

  const bool fp8c = kv_cache.scalar_type() == torch::kFloat8_e4m3fn;
  auto dispatch = [&](auto dtag) {
    constexpr int D = decltype(dtag)::value;
    if (q.scalar_type() == torch::kBFloat16) {
      if (fp8c) { LAUNCH_PF(BF16Tag, FP8CacheTag); }
      else      { LAUNCH_PF(BF16Tag, BF16Tag); }
    } else {
      if (fp8c) { LAUNCH_PF(FP16Tag, FP8CacheTag); }
      else      { LAUNCH_PF(FP16Tag, FP16Tag); }
    }
  };
  if (head_dim == 64) dispatch(std::integral_constant<int, 64>{});
  else if (head_dim == 128) dispatch(std::integral_constant<int, 128>{});
  else dispatch(std::integral_constant<int, 256>{});
#undef LAUNCH_PF
  HIP_CHECK_KERNEL();
}

}  // namespace vllm_amd
