// Flash-style attention forward for gfx950 (head_dim 64, bf16).
//
// Replaces the aotriton (Triton-compiled) SDPA on the BERT/Llama hot paths
// (VERDICT round-1 item 4) with hand-written CDNA4:
//
//   grid = (q_tiles, batch*heads); block = 4 waves; each wave OWNS 16 q
//   rows (no cross-wave softmax reduction). Per 64-key tile:
//     stage K,V tile (16 KB LDS, coalesced d-contiguous loads)
//     S = Q·Kᵀ            8 × mfma_16x16x32_bf16 per wave
//     online softmax      rowmax/rowsum via 16-lane shfl_xor reduce,
//                         exp2f with log2e folded into the scale
//     P → LDS             (C-layout scatter, bf16)
//     O += P·V            A-frags from the P image, B-frags = Vᵀ via
//                         ds_read_b64_tr_b16 (the attention-V recipe)
//   epilogue: O /= rowsum, bf16 store; optional logsumexp out (bwd).
//
// Causal masking for decoder models skips whole future tiles and masks
// the diagonal one; GQA maps kv_head = head / (H/Hkv) outside the kernel
// (the caller passes strides so K/V may have fewer heads).
#include "common.h"

typedef __bf16 abf16x8_t __attribute__((ext_vector_type(8)));
typedef __attribute__((ext_vector_type(4))) __bf16 abf16x4_t;
typedef __attribute__((address_space(3))) unsigned int a_lds_uint;
typedef __attribute__((address_space(1))) const unsigned int a_global_uint;

__device__ __forceinline__ abf16x4_t at_tr_read_b64(const char* addr) {
  unsigned a32 =
      (unsigned)(__SIZE_TYPE__)(__attribute__((address_space(3))) const char*)addr;
  abf16x4_t out;
  asm volatile("ds_read_b64_tr_b16 %0, %1" : "=v"(out) : "v"(a32) : "memory");
  return out;
}

// K tile image: [64 keys][64 d] bf16, linear (d-contiguous rows of 128 B);
// B-frag read row = key (lane&15 within 16-key group) — 16 lanes hit 16
// different 128-B rows at the same 16-B slot → XOR-swizzle by (row>>1)&7
// exactly like gemm.hip. V tile image: same linear layout, read via
// tr_b16 subtiles [4 key][16 d] → needs the wgrad-style [k/4][d/16][4][16]
// subtile order instead; we store V in that order during staging.
extern "C" __global__ void __launch_bounds__(256)
attn_fwd_d64_kernel(const short* __restrict__ Q, const short* __restrict__ K,
                    const short* __restrict__ V, short* __restrict__ O,
                    float* __restrict__ Lse, int S, int Skv, int nheads,
                    int kv_heads, float scale, int causal) {
  __shared__ short lds[(64 * 64) * 2 + (64 * 64)];  // K img + V img + P img
  short* ldsK = lds;                 // [64][64] swizzled 16-B slots
  short* ldsV = lds + 64 * 64;       // [16 kgrp][4 dgrp][4][16] subtiles
  short* ldsP = lds + 2 * 64 * 64;   // [64 q][64 key] linear
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;         // wave owns q rows [wave*16, wave*16+16)
  const int bh = blockIdx.y;
  const int q0 = blockIdx.x * 64;
  // GQA: kv head = head / (H / Hkv)
  const int b = bh / nheads, head = bh - b * nheads;
  const int kvbh = b * kv_heads + head / (nheads / kv_heads);
  const short* Qb = Q + (long long)bh * S * 64;
  const short* Kb = K + (long long)kvbh * Skv * 64;
  const short* Vb = V + (long long)kvbh * Skv * 64;

  // ---- load this wave's Q fragments (persist in regs) -------------------
  // A-frag: lane holds 8 d-elems at q-row (lane&15), d-group (lane>>4)*8,
  // kk in {0,1} → d offset kk*32 + (lane>>4)*8
  abf16x8_t q_frag[2];
  {
    int row = q0 + wave * 16 + (lane & 15);
    int rr = row < S ? row : S - 1;
#pragma unroll
    for (int kkq = 0; kkq < 2; ++kkq)
      q_frag[kkq] = *(const abf16x8_t*)(Qb + (long long)rr * 64 + kkq * 32 +
                                        (lane >> 4) * 8);
  }
  const bool qrow_ok = q0 + wave * 16 + (lane & 15) < S;

  // accumulators: O rows (lane>>4)*4+r of the wave's 16, cols ni*16+(lane&15)
  f32x4 o_acc[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) o_acc[i] = f32x4{0.f, 0.f, 0.f, 0.f};
  // per-lane row state for its 4 rows (same 4 rows in every 16-lane group)
  float row_max[4], row_sum[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    row_max[r] = -1e30f;
    row_sum[r] = 0.f;
  }
  const float l2e_scale = scale * 1.44269504f;  // exp → exp2

  const int ntiles = (Skv + 63) / 64;
  const int last_tile =
      causal ? (q0 + 63) / 64 : ntiles - 1;  // skip fully-masked tiles
  for (int kt = 0; kt <= last_tile && kt < ntiles; ++kt) {
    const int k0 = kt * 64;
    // ---- stage K (swizzled slots) and V (tr subtiles) ------------------
    // 256 threads × 16 B = 4 KB per pass; 2 passes each for K and V
#pragma unroll
    for (int pass = 0; pass < 2; ++pass) {
      int r = pass * 32 + (tid >> 3);      // key row 0..63
      int slot = tid & 7;                  // 16-B slot in the 128-B row
      int krow = k0 + r;
      int src_slot = slot ^ ((r >> 1) & 7);
      const short* src = Kb + (long long)(krow < Skv ? krow : Skv - 1) * 64 +
                         src_slot * 8;
      *(s16x8*)(ldsK + r * 64 + slot * 8) =
          krow < Skv ? *(const s16x8*)src : s16x8{0, 0, 0, 0, 0, 0, 0, 0};
      // V: subtile image [kgrp=r>>2][dgrp=slot>>1][krow&3][16 d]
      const short* vsrc = Vb + (long long)(krow < Skv ? krow : Skv - 1) * 64 +
                          (slot) * 8;
      short* vdst = ldsV + ((r >> 2) * 4 + (slot >> 1)) * 64 + (r & 3) * 16 +
                    (slot & 1) * 8;
      *(s16x8*)vdst =
          krow < Skv ? *(const s16x8*)vsrc : s16x8{0, 0, 0, 0, 0, 0, 0, 0};
    }
    __syncthreads();

    // ---- S = Q·Kᵀ (this wave's 16 rows × 64 keys) ----------------------
    f32x4 s_acc[4];
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) s_acc[ni] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      int row = ni * 16 + (lane & 15);  // key row
#pragma unroll
      for (int kkq = 0; kkq < 2; ++kkq) {
        int ks = kkq * 4 + (lane >> 4);
        abf16x8_t k_frag = *(const abf16x8_t*)(
            (const char*)ldsK + row * 128 + ((ks ^ (row >> 1)) & 7) * 16);
        s_acc[ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            q_frag[kkq], k_frag, s_acc[ni], 0, 0, 0);
      }
    }
    // ---- online softmax over this tile's 64 keys ----------------------
    // masks: tail keys (k0+col >= Skv) and causal diagonal
    float tile_max[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) tile_max[r] = -1e30f;
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      int col = k0 + ni * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int qrow = q0 + wave * 16 + (lane >> 4) * 4 + r;
        float v = s_acc[ni][r] * l2e_scale;
        if (col >= Skv || (causal && col > qrow)) v = -1e30f;
        s_acc[ni][r] = v;
        tile_max[r] = fmaxf(tile_max[r], v);
      }
    }
    // reduce max across the 16-lane column group
#pragma unroll
    for (int off = 1; off < 16; off <<= 1)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        tile_max[r] = fmaxf(tile_max[r],
                            __shfl_xor(tile_max[r], off, 64));
    float corr[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float m_new = fmaxf(row_max[r], tile_max[r]);
      corr[r] = exp2f(row_max[r] - m_new);
      row_max[r] = m_new;
    }
    // P = exp2(S - m); accumulate row sums
    float tile_sum[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float p = exp2f(s_acc[ni][r] - row_max[r]);
        s_acc[ni][r] = p;
        tile_sum[r] += p;
      }
#pragma unroll
    for (int off = 1; off < 16; off <<= 1)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        tile_sum[r] += __shfl_xor(tile_sum[r], off, 64);
#pragma unroll
    for (int r = 0; r < 4; ++r)
      row_sum[r] = row_sum[r] * corr[r] + tile_sum[r];
    // rescale O accumulator
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
#pragma unroll
      for (int r = 0; r < 4; ++r) o_acc[ni][r] *= corr[r];

    // ---- P → LDS (wave's 16×64 block, bf16, C-layout scatter) ----------
    {
      short* base = ldsP + (wave * 16) * 64;
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          base[((lane >> 4) * 4 + r) * 64 + ni * 16 + (lane & 15)] =
              f32_to_bf16(s_acc[ni][r]);
    }
    // wave-local P image: only THIS wave reads it back, so no cross-wave
    // barrier is needed (each wave owns its 16 rows)
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    // ---- O += P·V ------------------------------------------------------
#pragma unroll
    for (int kkp = 0; kkp < 2; ++kkp) {
      abf16x8_t p_frag = *(const abf16x8_t*)(
          ldsP + ((wave * 16) + (lane & 15)) * 64 + kkp * 32 +
          (lane >> 4) * 8);
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        // Vᵀ frag: rows = d (ni*16 + lane&15? no — tr_b16 gives us
        // [key][d] → [d][key]): subtile (kgrp, dgrp=ni), 16-lane group
        // reads [4 key][16 d] transposed
        abf16x4_t lo, hi;
        int g0 = kkp * 8 + (lane >> 4) * 2;
        const char* pv = (const char*)ldsV + ((g0 * 4 + ni) * 64) * 2 +
                         (lane & 15) * 8;
        lo = at_tr_read_b64(pv);
        const char* pv2 = (const char*)ldsV + (((g0 + 1) * 4 + ni) * 64) * 2 +
                          (lane & 15) * 8;
        hi = at_tr_read_b64(pv2);
        abf16x8_t v_frag;
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          v_frag[e] = lo[e];
          v_frag[e + 4] = hi[e];
        }
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        o_acc[ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            p_frag, v_frag, o_acc[ni], 0, 0, 0);
      }
    }
    __syncthreads();  // K/V/P LDS reuse next tile
  }

  // ---- epilogue: O /= rowsum; store; logsumexp -------------------------
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int qrow = q0 + wave * 16 + (lane >> 4) * 4 + r;
    if (qrow >= S) continue;
    float inv = row_sum[r] > 0.f ? 1.f / row_sum[r] : 0.f;
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
      O[(long long)bh * S * 64 + (long long)qrow * 64 + ni * 16 +
        (lane & 15)] = f32_to_bf16(o_acc[ni][r] * inv);
    if (Lse && (lane & 15) == 0)
      Lse[(long long)bh * S + qrow] =
          row_max[r] * 0.69314718f + logf(row_sum[r]);
  }
}
