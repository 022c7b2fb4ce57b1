// bf16 MFMA GEMM for gfx950 — C[M,N] = A[M,K] · B[N,K]^T  ("NT": both
// operands K-contiguous), fp32 accumulate, bf16 output.
//
// Structure (cdna_hip_programming.md §5 ladder, step-2/3 class):
//   128×128 tile, BK=64, 4 waves (2×2), 64×64 output per wave as 4×4
//   fragments of v_mfma_f32_16x16x32_bf16; LDS double-buffered with
//   +16B-padded rows (row stride 72 bf16 = 144 B → the 16-lane groups of
//   ds_read_b128 hit 16 distinct 16-B slots: conflict-free); register
//   staging global→LDS with 16-B vector accesses.
//
// Fragment layout (verified against rocm CK gfx950 attributes):
//   A/B: lane l holds 8 contiguous K elems at row/col (l&15), k-group
//        (l>>4)*8.   C/D: col = lane&15, row = (lane>>4)*4 + reg.
//
// The other Linear-layer products (dgrad/wgrad) are composed NT +
// transpose_bf16 (transpose cost ~3-8% of GEMM; a tr_b16-read variant can
// remove it later).
#include "common.h"

#define BM 128
#define BN 128
#define BK 64
#define LDS_STRIDE 72  // 64 + 8 bf16 pad
#define NTH 256

typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));

extern "C" __global__ void __launch_bounds__(NTH)
gemm_nt_bf16_kernel(const short* __restrict__ A, const short* __restrict__ B,
                    short* __restrict__ C, int M, int N, int K,
                    int lda, int ldb, int ldc) {
  __shared__ short lds[2 * 2 * BM * LDS_STRIDE];  // [buf][op][row][LDS_STRIDE]
  auto As = [&](int buf) -> short* { return lds + buf * 2 * BM * LDS_STRIDE; };
  auto Bs = [&](int buf) -> short* {
    return lds + (buf * 2 + 1) * BM * LDS_STRIDE;
  };

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = (wave >> 1) * 64;  // wave row offset in tile
  const int wc = (wave & 1) * 64;   // wave col offset in tile

  const int row0 = blockIdx.y * BM;
  const int col0 = blockIdx.x * BN;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int nk = (K + BK - 1) / BK;

  // ---- staging helper: load [128 rows][64 cols] K-slab of src into dst LDS
  // chunk c in [0,1024): row = c>>3, col16 = (c&7)*8 ; thread t does
  // c = t + 256*i (8 consecutive threads cover one 128-B row: coalesced).
  auto stage = [&](const short* __restrict__ src, int ld, int base_row,
                   int max_row, int k0, short* dst) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int c = tid + NTH * i;
      int r = c >> 3;
      int col = (c & 7) * 8;
      int gr = base_row + r;
      int gk = k0 + col;
      s16x8 v;
      if (gr < max_row && gk + 8 <= K) {
        v = *(const s16x8*)(src + (long long)gr * ld + gk);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int kk = gk + j;
          v[j] = (gr < max_row && kk < K) ? src[(long long)gr * ld + kk] : (short)0;
        }
      }
      *(s16x8*)(dst + r * LDS_STRIDE + col) = v;
    }
  };

  stage(A, lda, row0, M, 0, As(0));
  stage(B, ldb, col0, N, 0, Bs(0));
  __syncthreads();

  for (int kt = 0; kt < nk; ++kt) {
    int cur = kt & 1, nxt = cur ^ 1;
    if (kt + 1 < nk) {  // prefetch next K-slab while computing this one
      stage(A, lda, row0, M, (kt + 1) * BK, As(nxt));
      stage(B, ldb, col0, N, (kt + 1) * BK, Bs(nxt));
    }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {  // two K=32 sub-steps per BK=64
      const int koff = kk * 32 + (lane >> 4) * 8;
      bf16x8_t a_frag[4], b_frag[4];
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        a_frag[mi] = *(const bf16x8_t*)(As(cur) + (wr + mi * 16 + (lane & 15)) * LDS_STRIDE + koff);
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        b_frag[ni] = *(const bf16x8_t*)(Bs(cur) + (wc + ni * 16 + (lane & 15)) * LDS_STRIDE + koff);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
    }
    __syncthreads();
  }

  // ---- epilogue: C[row][col] ; row = row0+wr+mi*16+(lane>>4)*4+r,
  //                col = col0+wc+ni*16+(lane&15)
  const int crow_base = row0 + wr + (lane >> 4) * 4;
  const int ccol_base = col0 + wc + (lane & 15);
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = crow_base + mi * 16 + r;
      if (row >= M) continue;
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        int col = ccol_base + ni * 16;
        if (col < N) C[(long long)row * ldc + col] = f32_to_bf16(acc[mi][ni][r]);
      }
    }
  }
}

// ---------------------------------------------------------------- transpose
// out[N,M] = in[M,N]^T, bf16. 64×64 LDS tiles, padded stride.
#define TDIM 64
#define TSTRIDE 72
extern "C" __global__ void __launch_bounds__(256)
transpose_bf16_kernel(const short* __restrict__ in, short* __restrict__ out,
                      int M, int N) {
  __shared__ short tile[TDIM * TSTRIDE];
  const int tid = threadIdx.x;
  const int tile_r = blockIdx.y * TDIM;  // row offset in `in`
  const int tile_c = blockIdx.x * TDIM;  // col offset in `in`
  // load: each thread 2 chunks of 8 (64*64/8/256 = 2)
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    int c = tid + 256 * i;          // chunk id in [0,512)
    int r = c >> 3;                 // row in tile
    int col = (c & 7) * 8;
    int gr = tile_r + r, gc = tile_c + col;
    s16x8 v;
    if (gr < M && gc + 8 <= N) {
      v = *(const s16x8*)(in + (long long)gr * N + gc);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        v[j] = (gr < M && gc + j < N) ? in[(long long)gr * N + gc + j] : (short)0;
    }
    *(s16x8*)(tile + r * TSTRIDE + col) = v;
  }
  __syncthreads();
  // store: out row = in col. thread t writes chunks of out rows.
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    int c = tid + 256 * i;
    int r = c >> 3;                 // row in OUT tile (= col in in)
    int col = (c & 7) * 8;          // col in OUT tile (= row in in)
    int gr = tile_c + r, gc = tile_r + col;
    if (gr >= N) continue;
    s16x8 v;
#pragma unroll
    for (int j = 0; j < 8; ++j) v[j] = tile[(col + j) * TSTRIDE + r];
    if (gc + 8 <= M) {
      *(s16x8*)(out + (long long)gr * M + gc) = v;
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        if (gc + j < M) out[(long long)gr * M + gc + j] = v[j];
    }
  }
}

// ===========================================================================
// GEMM v2 — glds-staged, K-major LDS image (full-tile shapes only:
// M%128==0, N%128==0, K%64==0; the stride-general v1 kernel handles the rest)
//
// Design (cdna_hip_programming.md §5 step-3 class + T1 swizzle):
//  * LDS image per operand = [8 kslots][128 rows] of 16-B slots (a "kslot"
//    is 8 contiguous K elements). global_load_lds writes lane-linear, so a
//    wave stages 64 rows of one kslot per instruction — and the MFMA
//    fragment read (lane l: row (l&15), kslot (l>>4)) then reads 16
//    CONSECUTIVE slots per 16-lane group: bank-conflict-free by
//    construction, no XOR swizzle needed.
//  * double-buffered; one __shared__ object (compiler-waitcnt trap);
//    __syncthreads() drains the in-flight glds of the next tile.
//  * XCD-aware bijective blockIdx remap (8 XCDs with private L2s).
// ===========================================================================
#define V2_SLOTS_PER_OP 1024  // 8 kslots * 128 rows
#define V2_BUF_SLOTS 2048     // A + B

typedef __attribute__((address_space(3))) unsigned int lds_uint;
typedef __attribute__((address_space(1))) const unsigned int global_uint;

extern "C" __global__ void __launch_bounds__(NTH)
gemm_nt_bf16_v2_kernel(const short* __restrict__ A, const short* __restrict__ B,
                       short* __restrict__ C, int M, int N, int K) {
  __shared__ short lds[2 * V2_BUF_SLOTS * 8];  // 64 KB
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = (wave >> 1) * 64;
  const int wc = (wave & 1) * 64;

  // bijective XCD remap: contiguous grid chunk per XCD
  const int nwg = gridDim.x;
  int wgid = blockIdx.x;
  {
    int q = nwg >> 3, r = nwg & 7;
    int xcd = wgid & 7, idx = wgid >> 3;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int ntile_n = N / BN;
  const int row0 = (wgid / ntile_n) * BM;
  const int col0 = (wgid % ntile_n) * BN;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int nk = K / BK;

  // wave stages 4 (kslot,half) pieces of A and of B per K-step
  auto stage = [&](int buf, int kt) {
    const long long k0 = (long long)kt * BK;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int idx = wave * 4 + i;
      int kslot = idx >> 1, half = idx & 1;
      const short* srcA = A + (long long)(row0 + half * 64 + lane) * K + k0 + kslot * 8;
      int dstA = (buf * V2_BUF_SLOTS + kslot * 128 + half * 64) * 16;  // bytes
      __builtin_amdgcn_global_load_lds(
          (global_uint*)srcA, (lds_uint*)((char*)lds + dstA), 16, 0, 0);
      const short* srcB = B + (long long)(col0 + half * 64 + lane) * K + k0 + kslot * 8;
      int dstB = (buf * V2_BUF_SLOTS + V2_SLOTS_PER_OP + kslot * 128 + half * 64) * 16;
      __builtin_amdgcn_global_load_lds(
          (global_uint*)srcB, (lds_uint*)((char*)lds + dstB), 16, 0, 0);
    }
  };

  stage(0, 0);
  __syncthreads();  // vmcnt(0) drain + barrier (glds in flight)

  for (int kt = 0; kt < nk; ++kt) {
    int cur = kt & 1, nxt = cur ^ 1;
    if (kt + 1 < nk) stage(nxt, kt + 1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int ks = kk * 4 + (lane >> 4);
      bf16x8_t a_frag[4], b_frag[4];
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        a_frag[mi] = *(const bf16x8_t*)(
            lds + (cur * V2_BUF_SLOTS + ks * 128 + wr + mi * 16 + (lane & 15)) * 8);
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        b_frag[ni] = *(const bf16x8_t*)(
            lds + (cur * V2_BUF_SLOTS + V2_SLOTS_PER_OP + ks * 128 + wc + ni * 16 +
                   (lane & 15)) * 8);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
    }
    __syncthreads();  // drains next tile's glds; releases cur for overwrite
  }

  const int crow_base = row0 + wr + (lane >> 4) * 4;
  const int ccol_base = col0 + wc + (lane & 15);
#pragma unroll
  for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = crow_base + mi * 16 + r;
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        C[(long long)row * N + ccol_base + ni * 16] = f32_to_bf16(acc[mi][ni][r]);
    }
}


// v3 — single-buffer, two barriers per K-step: 32 KB LDS → ~5 blocks/CU;
// inter-block overlap hides the glds drain (the ladder's step-3 shape).
extern "C" __global__ void __launch_bounds__(NTH)
gemm_nt_bf16_v3_kernel(const short* __restrict__ A, const short* __restrict__ B,
                       short* __restrict__ C, int M, int N, int K) {
  __shared__ short lds[V2_BUF_SLOTS * 8];  // 32 KB
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = (wave >> 1) * 64;
  const int wc = (wave & 1) * 64;
  const int nwg = gridDim.x;
  int wgid = blockIdx.x;
  {
    int q = nwg >> 3, r = nwg & 7;
    int xcd = wgid & 7, idx = wgid >> 3;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int ntile_n = N / BN;
  const int row0 = (wgid / ntile_n) * BM;
  const int col0 = (wgid % ntile_n) * BN;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int nk = K / BK;
  for (int kt = 0; kt < nk; ++kt) {
    const long long k0 = (long long)kt * BK;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int idx = wave * 4 + i;
      int kslot = idx >> 1, half = idx & 1;
      const short* srcA = A + (long long)(row0 + half * 64 + lane) * K + k0 + kslot * 8;
      __builtin_amdgcn_global_load_lds(
          (global_uint*)srcA,
          (lds_uint*)((char*)lds + (kslot * 128 + half * 64) * 16), 16, 0, 0);
      const short* srcB = B + (long long)(col0 + half * 64 + lane) * K + k0 + kslot * 8;
      __builtin_amdgcn_global_load_lds(
          (global_uint*)srcB,
          (lds_uint*)((char*)lds + (V2_SLOTS_PER_OP + kslot * 128 + half * 64) * 16),
          16, 0, 0);
    }
    __syncthreads();  // drain glds + barrier
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int ks = kk * 4 + (lane >> 4);
      bf16x8_t a_frag[4], b_frag[4];
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        a_frag[mi] = *(const bf16x8_t*)(
            lds + (ks * 128 + wr + mi * 16 + (lane & 15)) * 8);
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        b_frag[ni] = *(const bf16x8_t*)(
            lds + (V2_SLOTS_PER_OP + ks * 128 + wc + ni * 16 + (lane & 15)) * 8);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
    }
    __syncthreads();  // everyone done reading before next overwrite
  }
  const int crow_base = row0 + wr + (lane >> 4) * 4;
  const int ccol_base = col0 + wc + (lane & 15);
#pragma unroll
  for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = crow_base + mi * 16 + r;
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        C[(long long)row * N + ccol_base + ni * 16] = f32_to_bf16(acc[mi][ni][r]);
    }
}

// v4 — full-cacheline glds + source-XOR swizzle (rule 21).
// Each glds instruction covers 8 consecutive rows × the full 128-B K-slab
// (coalesced: 8 whole lines per instruction), with the kslot index of each
// lane's 16-B piece XOR-permuted by ((row>>1)&7) on the SOURCE address
// (bijective over each 16-row fragment: parity bit picks the 128-B half of
// the 256-B bank line, (row>>1)&7 covers its 8 slots exactly once); the
// fragment read applies the same XOR. Residual 2-way conflict (rows r,r+8)
// is below the MFMA-issue bound. Single LDS buffer (32 KB → 3 waves/SIMD).
extern "C" __global__ void __launch_bounds__(NTH)
gemm_nt_bf16_v4_kernel(const short* __restrict__ A, const short* __restrict__ B,
                       short* __restrict__ C, int M, int N, int K) {
  __shared__ short lds[V2_BUF_SLOTS * 8];  // 32 KB: A tile + B tile
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = (wave >> 1) * 64;
  const int wc = (wave & 1) * 64;
  const int nwg = gridDim.x;
  int wgid = blockIdx.x;
  {
    int q = nwg >> 3, r = nwg & 7;
    int xcd = wgid & 7, idx = wgid >> 3;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int ntile_n = N / BN;
  const int row0 = (wgid / ntile_n) * BM;
  const int col0 = (wgid % ntile_n) * BN;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  // per-lane source column offset (elements): XOR-permuted kslot within row
  const int row_in_piece = lane >> 3;

  const int nk = K / BK;
  for (int kt = 0; kt < nk; ++kt) {
    const long long k0 = (long long)kt * BK;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int pa = wave * 4 + i;  // A piece: rows pa*8 .. pa*8+7
      const int src_koff =
          (((lane & 7) ^ (pa * 4 + (lane >> 4))) & 7) * 8;
      const short* srcA =
          A + (long long)(row0 + pa * 8 + row_in_piece) * K + k0 + src_koff;
      __builtin_amdgcn_global_load_lds(
          (global_uint*)srcA, (lds_uint*)((char*)lds + pa * 1024), 16, 0, 0);
      const short* srcB =
          B + (long long)(col0 + pa * 8 + row_in_piece) * K + k0 + src_koff;
      __builtin_amdgcn_global_load_lds(
          (global_uint*)srcB, (lds_uint*)((char*)lds + 16384 + pa * 1024), 16, 0,
          0);
    }
    __syncthreads();  // drain glds + barrier
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int ks = kk * 4 + (lane >> 4);
      bf16x8_t a_frag[4], b_frag[4];
#pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        int row = wr + mi * 16 + (lane & 15);
        a_frag[mi] = *(const bf16x8_t*)((char*)lds + row * 128 +
                                        ((ks ^ (row >> 1)) & 7) * 16);
      }
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        int row = wc + ni * 16 + (lane & 15);
        b_frag[ni] = *(const bf16x8_t*)((char*)lds + 16384 + row * 128 +
                                        ((ks ^ (row >> 1)) & 7) * 16);
      }
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
    }
    __syncthreads();
  }
  const int crow_base = row0 + wr + (lane >> 4) * 4;
  const int ccol_base = col0 + wc + (lane & 15);
#pragma unroll
  for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = crow_base + mi * 16 + r;
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        C[(long long)row * N + ccol_base + ni * 16] = f32_to_bf16(acc[mi][ni][r]);
    }
}

// v5 — v4 + double-buffered glds (overlap staging of tile k+1 with compute).
// Each glds instruction covers 8 consecutive rows × the full 128-B K-slab
// (coalesced: 8 whole lines per instruction), with the kslot index of each
// lane's 16-B piece XOR-permuted by ((row>>1)&7) on the SOURCE address
// (bijective over each 16-row fragment: parity bit picks the 128-B half of
// the 256-B bank line, (row>>1)&7 covers its 8 slots exactly once); the
// fragment read applies the same XOR. Residual 2-way conflict (rows r,r+8)
// is below the MFMA-issue bound. Single LDS buffer (32 KB → 3 waves/SIMD).
extern "C" __global__ void __launch_bounds__(NTH)
gemm_nt_bf16_v5_kernel(const short* __restrict__ A, const short* __restrict__ B,
                       short* __restrict__ C, int M, int N, int K) {
  __shared__ short lds[2 * V2_BUF_SLOTS * 8];  // 64 KB double-buffered
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = (wave >> 1) * 64;
  const int wc = (wave & 1) * 64;
  const int nwg = gridDim.x;
  int wgid = blockIdx.x;
  {
    int q = nwg >> 3, r = nwg & 7;
    int xcd = wgid & 7, idx = wgid >> 3;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int ntile_n = N / BN;
  const int row0 = (wgid / ntile_n) * BM;
  const int col0 = (wgid % ntile_n) * BN;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  // per-lane source column offset (elements): XOR-permuted kslot within row
  const int row_in_piece = lane >> 3;

  const int nk = K / BK;
  auto stage = [&](int buf, int kt) {
    const long long k0 = (long long)kt * BK;
    char* base = (char*)lds + buf * 32768;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int pa = wave * 4 + i;
      const int src_koff =
          (((lane & 7) ^ (pa * 4 + (lane >> 4))) & 7) * 8;
      const short* srcA =
          A + (long long)(row0 + pa * 8 + row_in_piece) * K + k0 + src_koff;
      __builtin_amdgcn_global_load_lds(
          (global_uint*)srcA, (lds_uint*)(base + pa * 1024), 16, 0, 0);
      const short* srcB =
          B + (long long)(col0 + pa * 8 + row_in_piece) * K + k0 + src_koff;
      __builtin_amdgcn_global_load_lds(
          (global_uint*)srcB, (lds_uint*)(base + 16384 + pa * 1024), 16, 0, 0);
    }
  };
  stage(0, 0);
  __syncthreads();
  for (int kt = 0; kt < nk; ++kt) {
    int cur = kt & 1;
    if (kt + 1 < nk) stage(cur ^ 1, kt + 1);
    char* base = (char*)lds + cur * 32768;
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int ks = kk * 4 + (lane >> 4);
      bf16x8_t a_frag[4], b_frag[4];
#pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        int row = wr + mi * 16 + (lane & 15);
        a_frag[mi] = *(const bf16x8_t*)(base + row * 128 +
                                        ((ks ^ (row >> 1)) & 7) * 16);
      }
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        int row = wc + ni * 16 + (lane & 15);
        b_frag[ni] = *(const bf16x8_t*)(base + 16384 + row * 128 +
                                        ((ks ^ (row >> 1)) & 7) * 16);
      }
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
    }
    __syncthreads();
  }
  const int crow_base = row0 + wr + (lane >> 4) * 4;
  const int ccol_base = col0 + wc + (lane & 15);
#pragma unroll
  for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = crow_base + mi * 16 + r;
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        C[(long long)row * N + ccol_base + ni * 16] = f32_to_bf16(acc[mi][ni][r]);
    }
}

// v5b — v5's 128² tile with the v7 barrier discipline: counted
// s_waitcnt vmcnt(8) + raw s_barrier instead of __syncthreads' vmcnt(0)
// drain. The drain is a fixed per-K-step cost, so this targets SHALLOW-K
// shapes (BERT's K=768 classes) where it is the largest fraction.
extern "C" __global__ void __launch_bounds__(NTH)
gemm_nt_bf16_v5b_kernel(const short* __restrict__ A, const short* __restrict__ B,
                        short* __restrict__ C, int M, int N, int K) {
  __shared__ short lds[2 * V2_BUF_SLOTS * 8];
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = (wave >> 1) * 64;
  const int wc = (wave & 1) * 64;
  const int nwg = gridDim.x;
  int wgid = blockIdx.x;
  {
    int q = nwg >> 3, r = nwg & 7;
    int xcd = wgid & 7, idx = wgid >> 3;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int ntile_n = N / BN;
  const int row0 = (wgid / ntile_n) * BM;
  const int col0 = (wgid % ntile_n) * BN;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int row_in_piece = lane >> 3;
  const int nk = K / BK;
  auto stage = [&](int buf, int kt) {
    const long long k0 = (long long)kt * BK;
    char* base = (char*)lds + buf * 32768;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int pa = wave * 4 + i;
      const int src_koff = (((lane & 7) ^ (pa * 4 + (lane >> 4))) & 7) * 8;
      const short* srcA =
          A + (long long)(row0 + pa * 8 + row_in_piece) * K + k0 + src_koff;
      __builtin_amdgcn_global_load_lds(
          (global_uint*)srcA, (lds_uint*)(base + pa * 1024), 16, 0, 0);
      const short* srcB =
          B + (long long)(col0 + pa * 8 + row_in_piece) * K + k0 + src_koff;
      __builtin_amdgcn_global_load_lds(
          (global_uint*)srcB, (lds_uint*)(base + 16384 + pa * 1024), 16, 0, 0);
    }
  };
  stage(0, 0);
  for (int kt = 0; kt < nk; ++kt) {
    int cur = kt & 1;
    if (kt + 1 < nk) {
      stage(cur ^ 1, kt + 1);
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    char* base = (char*)lds + cur * 32768;
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int ks = kk * 4 + (lane >> 4);
      bf16x8_t a_frag[4], b_frag[4];
#pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        int row = wr + mi * 16 + (lane & 15);
        a_frag[mi] = *(const bf16x8_t*)(base + row * 128 +
                                        ((ks ^ (row >> 1)) & 7) * 16);
      }
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        int row = wc + ni * 16 + (lane & 15);
        b_frag[ni] = *(const bf16x8_t*)(base + 16384 + row * 128 +
                                        ((ks ^ (row >> 1)) & 7) * 16);
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    __builtin_amdgcn_s_barrier();
  }
  const int crow_base = row0 + wr + (lane >> 4) * 4;
  const int ccol_base = col0 + wc + (lane & 15);
#pragma unroll
  for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = crow_base + mi * 16 + r;
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        C[(long long)row * N + ccol_base + ni * 16] = f32_to_bf16(acc[mi][ni][r]);
    }
}

// v6 — 256×256 tile, 8 waves, BK=64, v4-style full-line glds + src XOR
// swizzle, single LDS buffer (64 KB → 2 blocks/CU). Twice the
// compute-per-staged-byte of the 128² tiles.
extern "C" __global__ void __launch_bounds__(512)
gemm_nt_bf16_v6_kernel(const short* __restrict__ A, const short* __restrict__ B,
                       short* __restrict__ C, int M, int N, int K) {
  __shared__ short lds[2 * 2048 * 8];  // A[256][64] + B[256][64] bf16 = 64 KB
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;           // 0..7
  const int wr = (wave >> 2) * 128;    // 2 M-groups of 128
  const int wc = (wave & 3) * 64;      // 4 N-groups of 64
  const int nwg = gridDim.x;
  int wgid = blockIdx.x;
  {
    int q = nwg >> 3, r = nwg & 7;
    int xcd = wgid & 7, idx = wgid >> 3;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int ntile_n = N / 256;
  const int row0 = (wgid / ntile_n) * 256;
  const int col0 = (wgid % ntile_n) * 256;

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int row_in_piece = lane >> 3;

  const int nk = K / BK;
  for (int kt = 0; kt < nk; ++kt) {
    const long long k0 = (long long)kt * BK;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int pa = wave * 4 + i;  // 32 pieces of 8 rows cover 256 rows
      const int src_koff =
          (((lane & 7) ^ (pa * 4 + (lane >> 4))) & 7) * 8;
      const short* srcA =
          A + (long long)(row0 + pa * 8 + row_in_piece) * K + k0 + src_koff;
      __builtin_amdgcn_global_load_lds(
          (global_uint*)srcA, (lds_uint*)((char*)lds + pa * 1024), 16, 0, 0);
      const short* srcB =
          B + (long long)(col0 + pa * 8 + row_in_piece) * K + k0 + src_koff;
      __builtin_amdgcn_global_load_lds(
          (global_uint*)srcB, (lds_uint*)((char*)lds + 32768 + pa * 1024), 16, 0,
          0);
    }
    __syncthreads();  // drain glds + barrier
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int ks = kk * 4 + (lane >> 4);
      bf16x8_t b_frag[4];
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        int row = wc + ni * 16 + (lane & 15);
        b_frag[ni] = *(const bf16x8_t*)((char*)lds + 32768 + row * 128 +
                                        ((ks ^ (row >> 1)) & 7) * 16);
      }
#pragma unroll
      for (int mi = 0; mi < 8; ++mi) {
        int row = wr + mi * 16 + (lane & 15);
        bf16x8_t a_frag = *(const bf16x8_t*)((char*)lds + row * 128 +
                                             ((ks ^ (row >> 1)) & 7) * 16);
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag, b_frag[ni], acc[mi][ni], 0, 0, 0);
      }
    }
    __syncthreads();
  }
  const int crow_base = row0 + wr + (lane >> 4) * 4;
  const int ccol_base = col0 + wc + (lane & 15);
#pragma unroll
  for (int mi = 0; mi < 8; ++mi)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = crow_base + mi * 16 + r;
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        C[(long long)row * N + ccol_base + ni * 16] = f32_to_bf16(acc[mi][ni][r]);
    }
}

// v7 — 256² tile, double-buffered glds with COUNTED vmcnt + raw barriers.
// Per K-tile: issue the 8 prefetch glds for tile t+1 into buf^1, then
// `s_waitcnt vmcnt(8)` (= everything older, i.e. tile t's stages, has
// landed), one barrier, free-running fragment compute, one barrier before
// the next tile may overwrite. No vmcnt(0) drain in the loop (the
// structural stall of the __syncthreads() variants — guide T3/T4).
extern "C" __global__ void __launch_bounds__(512)
gemm_nt_bf16_v7_kernel(const short* __restrict__ A, const short* __restrict__ B,
                       short* __restrict__ C, int M, int N, int K) {
  __shared__ short lds[2 * 2 * 2048 * 8];  // 2 buf × (A+B)[256][64] = 128 KB
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = (wave >> 2) * 128;
  const int wc = (wave & 3) * 64;
  const int nwg = gridDim.x;
  int wgid = blockIdx.x;
  {
    int q = nwg >> 3, r = nwg & 7;
    int xcd = wgid & 7, idx = wgid >> 3;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int ntile_n = N / 256;
  const int row0 = (wgid / ntile_n) * 256;
  const int col0 = (wgid % ntile_n) * 256;

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int row_in_piece = lane >> 3;

  auto stage = [&](int buf, int kt) {
    const long long k0 = (long long)kt * BK;
    char* base = (char*)lds + buf * 65536;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int pa = wave * 4 + i;
      const int src_koff =
          (((lane & 7) ^ (pa * 4 + (lane >> 4))) & 7) * 8;
      const short* srcA =
          A + (long long)(row0 + pa * 8 + row_in_piece) * K + k0 + src_koff;
      __builtin_amdgcn_global_load_lds(
          (global_uint*)srcA, (lds_uint*)(base + pa * 1024), 16, 0, 0);
      const short* srcB =
          B + (long long)(col0 + pa * 8 + row_in_piece) * K + k0 + src_koff;
      __builtin_amdgcn_global_load_lds(
          (global_uint*)srcB, (lds_uint*)(base + 32768 + pa * 1024), 16, 0, 0);
    }
  };

  const int nk = K / BK;
  stage(0, 0);
  for (int kt = 0; kt < nk; ++kt) {
    int cur = kt & 1;
    if (kt + 1 < nk) {
      stage(cur ^ 1, kt + 1);
      // tile t's 8 stages are everything older than the 8 just issued
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();  // all waves' tile-t data visible
    char* base = (char*)lds + cur * 65536;
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int ks = kk * 4 + (lane >> 4);
      bf16x8_t b_frag[4];
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        int row = wc + ni * 16 + (lane & 15);
        b_frag[ni] = *(const bf16x8_t*)(base + 32768 + row * 128 +
                                        ((ks ^ (row >> 1)) & 7) * 16);
      }
#pragma unroll
      for (int mi = 0; mi < 8; ++mi) {
        int row = wr + mi * 16 + (lane & 15);
        bf16x8_t a_frag = *(const bf16x8_t*)(base + row * 128 +
                                             ((ks ^ (row >> 1)) & 7) * 16);
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag, b_frag[ni], acc[mi][ni], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
      }
    }
    __builtin_amdgcn_s_barrier();  // reads done before t+1 overwrites buf
  }
  const int crow_base = row0 + wr + (lane >> 4) * 4;
  const int ccol_base = col0 + wc + (lane & 15);
#pragma unroll
  for (int mi = 0; mi < 8; ++mi)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = crow_base + mi * 16 + r;
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        C[(long long)row * N + ccol_base + ni * 16] = f32_to_bf16(acc[mi][ni][r]);
    }
}

// v8 — v7 geometry + the guide's phase interleave (T3/T4/T5): 4 phases per
// K-tile, each = {stage 1 piece of tile t+1 ∥ ds_read one fragment
// quadrant → raw barrier → MFMA×16 under setprio}; counted vmcnt(2) once
// per K-tile (at phase 0, after issuing that phase's 2 glds). B fragments
// are re-used across the two phases that share an N-half.
extern "C" __global__ void __launch_bounds__(512)
gemm_nt_bf16_v8_kernel(const short* __restrict__ A, const short* __restrict__ B,
                       short* __restrict__ C, int M, int N, int K) {
  __shared__ short lds[2 * 2 * 2048 * 8];  // 2 buf × (A+B)[256][64] = 128 KB
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = (wave >> 2) * 128;
  const int wc = (wave & 3) * 64;
  const int nwg = gridDim.x;
  int wgid = blockIdx.x;
  {
    int q = nwg >> 3, r = nwg & 7;
    int xcd = wgid & 7, idx = wgid >> 3;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int ntile_n = N / 256;
  const int row0 = (wgid / ntile_n) * 256;
  const int col0 = (wgid % ntile_n) * 256;

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int row_in_piece = lane >> 3;

  // stage ONE (A,B) piece pair: piece index p in [0,4) per wave
  auto stage_piece = [&](int buf, int kt, int p) {
    const long long k0 = (long long)kt * BK;
    char* base = (char*)lds + buf * 65536;
    int pa = wave * 4 + p;
    const int src_koff =
        (((lane & 7) ^ (pa * 4 + (lane >> 4))) & 7) * 8;
    const short* srcA =
        A + (long long)(row0 + pa * 8 + row_in_piece) * K + k0 + src_koff;
    __builtin_amdgcn_global_load_lds(
        (global_uint*)srcA, (lds_uint*)(base + pa * 1024), 16, 0, 0);
    const short* srcB =
        B + (long long)(col0 + pa * 8 + row_in_piece) * K + k0 + src_koff;
    __builtin_amdgcn_global_load_lds(
        (global_uint*)srcB, (lds_uint*)(base + 32768 + pa * 1024), 16, 0, 0);
  };

  const int nk = K / BK;
#pragma unroll
  for (int p = 0; p < 4; ++p) stage_piece(0, 0, p);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (int kt = 0; kt < nk; ++kt) {
    int cur = kt & 1;
    char* base = (char*)lds + cur * 65536;
    bf16x8_t b_frag[2][2];  // [ni within half][kk]
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      const int mh = p & 1, nh = p >> 1;
      if (kt + 1 < nk) stage_piece(cur ^ 1, kt + 1, p);
      if (p == 0 && kt > 0) {
        if (kt + 1 < nk) {
          // tile t's 8 stages are everything older than this phase's 2
          asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
        } else {
          // last tile: NOTHING was issued this phase, so vmcnt(2) would
          // leave the newest 2 of tile t's own stages in flight — drain
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        }
        __builtin_amdgcn_s_barrier();
      }
      // B quadrant: read on even phases, reuse on odd
      if (mh == 0) {
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
#pragma unroll
          for (int kk = 0; kk < 2; ++kk) {
            int row = wc + (nh * 2 + ni) * 16 + (lane & 15);
            int ks = kk * 4 + (lane >> 4);
            b_frag[ni][kk] = *(const bf16x8_t*)(
                base + 32768 + row * 128 + ((ks ^ (row >> 1)) & 7) * 16);
          }
      }
      bf16x8_t a_frag[4][2];
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
          int row = wr + (mh * 4 + mi) * 16 + (lane & 15);
          int ks = kk * 4 + (lane >> 4);
          a_frag[mi][kk] = *(const bf16x8_t*)(
              base + row * 128 + ((ks ^ (row >> 1)) & 7) * 16);
        }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
#pragma unroll
          for (int kk = 0; kk < 2; ++kk)
            acc[mh * 4 + mi][nh * 2 + ni] =
                __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a_frag[mi][kk], b_frag[ni][kk],
                    acc[mh * 4 + mi][nh * 2 + ni], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_s_barrier();
    }
    if (kt + 1 == nk) asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  const int crow_base = row0 + wr + (lane >> 4) * 4;
  const int ccol_base = col0 + wc + (lane & 15);
#pragma unroll
  for (int mi = 0; mi < 8; ++mi)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = crow_base + mi * 16 + r;
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        C[(long long)row * N + ccol_base + ni * 16] = f32_to_bf16(acc[mi][ni][r]);
    }
}

// ===========================================================================
// v9 — wgrad kernel: C[Na,Nb] = A^T·B for A[M,Na], B[M,Nb] (both row-major,
// reduction = M). Both operands are reduction-STRIDED in memory; tiles are
// staged in [m/4][out/16][4][16] bf16 subtiles (glds full-line coalesced,
// lane-linear) and fragments are gathered with the gfx950 hardware
// transpose read ds_read_b64_tr_b16 (guide T10): lane l, elem j reads
// lds[(l&15) + j*16 + (l>>4)*64] relative to a per-lane base — one read
// yields 4 consecutive reduction elements of the lane's output column.
// Replaces the transpose_bf16(dY) + transpose_bf16(X) pre-passes.
// Tile: 128×128 out, 64 reduction/step, 4 waves, double-buffered.
// ===========================================================================
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4_t;

DEVINL bf16x4_t tr_read_b64_issue(const char* addr) {
  // issue WITHOUT a wait — caller batches one lgkmcnt(0) before the first
  // consumer (guide §5.7 form iii; a wait per read serializes the 16-read
  // fragment gather)
  unsigned a32 =
      (unsigned)(__SIZE_TYPE__)(__attribute__((address_space(3))) const char*)addr;
  bf16x4_t out;
  asm volatile("ds_read_b64_tr_b16 %0, %1" : "=v"(out) : "v"(a32) : "memory");
  return out;
}

DEVINL void tr_read_drain() {
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_sched_barrier(0);  // rule 18: MFMAs must not hoist past
}

extern "C" __global__ void __launch_bounds__(NTH)
gemm_wgrad_bf16_kernel(const short* __restrict__ A, const short* __restrict__ B,
                       short* __restrict__ C, float* __restrict__ Cws,
                       int M, int Na, int Nb, int nsplit) {
  // LDS: 2 buf × 2 ops × [64][128] bf16 subtiled = 64 KB
  __shared__ short lds[2 * 2 * 8192];
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = (wave >> 1) * 64;  // output rows (A columns)
  const int wc = (wave & 1) * 64;   // output cols (B columns)
  const int ntile_n = Nb / BN;
  int wgid = blockIdx.x;
  {
    int nwg = gridDim.x;
    int q = nwg >> 3, r = nwg & 7;
    int xcd = wgid & 7, idx = wgid >> 3;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int split = wgid % nsplit;  // split-K slice over the reduction
  wgid /= nsplit;
  const int row0 = (wgid / ntile_n) * BM;  // Na offset
  const int col0 = (wgid % ntile_n) * BN;  // Nb offset

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  // glds source pattern: one 1-KB instruction = one 4-row m-group × 128
  // out cols in subtile order; lane l fetches (m = mb + ((l&7)>>1),
  // col = (l>>3)*16 + (l&1)*8), 16 B each → 4 full 256-B rows/instr.
  const int src_m_in_grp = (lane & 7) >> 1;
  const int src_col = ((lane >> 3) * 16 + (lane & 1) * 8);

  auto stage = [&](int buf, int mt) {
    const long long m0 = (long long)mt * BK;
    char* base = (char*)lds + buf * 32768;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int grp = wave * 4 + i;  // 16 m-groups of 4 cover 64 reduction rows
      const short* srcA =
          A + (m0 + grp * 4 + src_m_in_grp) * (long long)Na + row0 + src_col;
      __builtin_amdgcn_global_load_lds(
          (global_uint*)srcA, (lds_uint*)(base + grp * 1024), 16, 0, 0);
      const short* srcB =
          B + (m0 + grp * 4 + src_m_in_grp) * (long long)Nb + col0 + src_col;
      __builtin_amdgcn_global_load_lds(
          (global_uint*)srcB, (lds_uint*)(base + 16384 + grp * 1024), 16, 0, 0);
    }
  };

  const int nk_total = M / BK;
  const int per = (nk_total + nsplit - 1) / nsplit;
  const int mt0 = split * per;
  const int mt1 = (mt0 + per < nk_total) ? mt0 + per : nk_total;
  const int nk = mt1 - mt0;
  if (nk <= 0) {  // empty slice: still must contribute zeros
    if (nsplit > 1) {
      const int crow_base0 = row0 + wr + (lane >> 4) * 4;
      const int ccol_base0 = col0 + wc + (lane & 15);
      float* W = Cws + (long long)split * Na * Nb;
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int r = 0; r < 4; ++r)
#pragma unroll
          for (int ni = 0; ni < 4; ++ni)
            W[(long long)(crow_base0 + mi * 16 + r) * Nb + ccol_base0 + ni * 16] =
                0.f;
    }
    return;
  }
  stage(0, mt0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (int mti = 0; mti < nk; ++mti) {
    int mt = mt0 + mti;
    int cur = mti & 1;
    if (mti + 1 < nk) {
      stage(cur ^ 1, mt + 1);
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    char* base = (char*)lds + cur * 32768;
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      // within-buffer per-lane tr-read base for reduction group
      // m-group index g = kk*8 + (lane>>4)*2 + off4 ; subtile (g, out>>4)
      bf16x8_t a_frag[4], b_frag[4];
#pragma unroll
      for (int fi = 0; fi < 4; ++fi) {
        int colA = wr + fi * 16;  // output-row block (Na side)
        bf16x4_t lo, hi;
        {
          // per-lane addr = subtile base + (lane&15)*8B: the 16-lane group
          // collectively loads the [4][16] subtile; HW returns lane l elem j
          // = data[l + 16j] = (row j, col l) — the transpose we need
          int g0 = kk * 8 + (lane >> 4) * 2;
          const char* pa = base + ((g0 * 8 + (colA >> 4)) * 64) * 2 +
                           (lane & 15) * 8;
          lo = tr_read_b64_issue(pa);
          const char* pa2 = base + (((g0 + 1) * 8 + (colA >> 4)) * 64) * 2 +
                            (lane & 15) * 8;
          hi = tr_read_b64_issue(pa2);
        }
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          a_frag[fi][e] = lo[e];
          a_frag[fi][e + 4] = hi[e];
        }
        int colB = wc + fi * 16;
        {
          int g0 = kk * 8 + (lane >> 4) * 2;
          const char* pb = base + 16384 + ((g0 * 8 + (colB >> 4)) * 64) * 2 +
                           (lane & 15) * 8;
          lo = tr_read_b64_issue(pb);
          const char* pb2 = base + 16384 +
                            (((g0 + 1) * 8 + (colB >> 4)) * 64) * 2 +
                            (lane & 15) * 8;
          hi = tr_read_b64_issue(pb2);
        }
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          b_frag[fi][e] = lo[e];
          b_frag[fi][e + 4] = hi[e];
        }
      }
      tr_read_drain();
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
    }
    __builtin_amdgcn_s_barrier();
  }
  const int crow_base = row0 + wr + (lane >> 4) * 4;
  const int ccol_base = col0 + wc + (lane & 15);
  if (nsplit == 1) {
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = crow_base + mi * 16 + r;
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          C[(long long)row * Nb + ccol_base + ni * 16] =
              f32_to_bf16(acc[mi][ni][r]);
      }
  } else {
    float* W = Cws + (long long)split * Na * Nb;
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = crow_base + mi * 16 + r;
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          W[(long long)row * Nb + ccol_base + ni * 16] = acc[mi][ni][r];
      }
  }
}

// reduce split-K fp32 slabs -> bf16
extern "C" __global__ void __launch_bounds__(256)
splitk_reduce_kernel(const float* __restrict__ ws, short* __restrict__ out,
                     long long n, int nsplit) {
  // 4-wide: f32x4 slab loads (all slabs' loads in flight) + short4 store;
  // the scalar version measured 6.7 ms/round on the llama config
  long long i4 = ((long long)blockIdx.x * 256 + threadIdx.x) * 4;
  long long stride = (long long)gridDim.x * 256 * 4;
  for (; i4 + 4 <= n; i4 += stride) {
    f32x4 acc = *(const f32x4*)(ws + i4);
    for (int s = 1; s < nsplit; ++s) {
      f32x4 v = *(const f32x4*)(ws + (long long)s * n + i4);
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[j] += v[j];
    }
    short o4[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) o4[j] = f32_to_bf16(acc[j]);
    *(unsigned long long*)(out + i4) = *(unsigned long long*)o4;
  }
  if (i4 < n) {
    for (; i4 < n; ++i4) {
      float acc = 0.f;
      for (int s = 0; s < nsplit; ++s) acc += ws[(long long)s * n + i4];
      out[i4] = f32_to_bf16(acc);
    }
  }
}

// v10 — dgrad kernel: C[M,K] = A[M,R]·B[R,K] (A reduction-contiguous like
// the NT kernels; B reduction-strided like wgrad). A staged v4-style
// (full-line glds + src XOR swizzle, b128 fragment reads); B staged in
// [r/4][k/16][4][16] subtiles and gathered with tr_b16. Replaces the
// transpose_bf16(W) pre-pass of every Linear dgrad.
extern "C" __global__ void __launch_bounds__(NTH)
gemm_dgrad_bf16_kernel(const short* __restrict__ A, const short* __restrict__ B,
                       short* __restrict__ C, float* __restrict__ Cws,
                       int M, int R, int K, int nsplit) {
  __shared__ short lds[2 * 2 * 8192];  // 2 buf × (A 16KB + B 16KB)
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = (wave >> 1) * 64;
  const int wc = (wave & 1) * 64;
  const int ntile_n = K / BN;
  int wgid = blockIdx.x;
  {
    int nwg = gridDim.x;
    int q = nwg >> 3, r = nwg & 7;
    int xcd = wgid & 7, idx = wgid >> 3;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int split = wgid % nsplit;
  wgid /= nsplit;
  const int row0 = (wgid / ntile_n) * BM;
  const int col0 = (wgid % ntile_n) * BN;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  // A source (v4 pattern): rows of 128-B red-slabs, XOR-permuted kslot
  const int a_row_in_piece = lane >> 3;
  // B source (wgrad pattern): 4-row red groups × subtile order
  const int b_m_in_grp = (lane & 7) >> 1;
  const int b_col = ((lane >> 3) * 16 + (lane & 1) * 8);

  auto stage = [&](int buf, int rt) {
    const long long r0 = (long long)rt * BK;
    char* base = (char*)lds + buf * 32768;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int pa = wave * 4 + i;  // A: 16 pieces of 8 rows
      const int a_koff =
          (((lane & 7) ^ (pa * 4 + (lane >> 4))) & 7) * 8;
      const short* srcA =
          A + (long long)(row0 + pa * 8 + a_row_in_piece) * R + r0 + a_koff;
      __builtin_amdgcn_global_load_lds(
          (global_uint*)srcA, (lds_uint*)(base + pa * 1024), 16, 0, 0);
      // B: 16 red-groups of 4 rows
      const short* srcB =
          B + (r0 + pa * 4 + b_m_in_grp) * (long long)K + col0 + b_col;
      __builtin_amdgcn_global_load_lds(
          (global_uint*)srcB, (lds_uint*)(base + 16384 + pa * 1024), 16, 0, 0);
    }
  };

  const int nk_total = R / BK;
  const int per = (nk_total + nsplit - 1) / nsplit;
  const int rt0 = split * per;
  const int rt1 = (rt0 + per < nk_total) ? rt0 + per : nk_total;
  const int nk = rt1 - rt0;
  if (nk <= 0) {
    if (nsplit > 1) {
      const int crow0 = row0 + wr + (lane >> 4) * 4;
      const int ccol0 = col0 + wc + (lane & 15);
      float* W = Cws + (long long)split * M * K;
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int r = 0; r < 4; ++r)
#pragma unroll
          for (int ni = 0; ni < 4; ++ni)
            W[(long long)(crow0 + mi * 16 + r) * K + ccol0 + ni * 16] = 0.f;
    }
    return;
  }
  stage(0, rt0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (int rti = 0; rti < nk; ++rti) {
    int rt = rt0 + rti;
    int cur = rti & 1;
    if (rti + 1 < nk) {
      stage(cur ^ 1, rt + 1);
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    char* base = (char*)lds + cur * 32768;
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int ks = kk * 4 + (lane >> 4);
      bf16x8_t a_frag[4], b_frag[4];
#pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        int row = wr + mi * 16 + (lane & 15);
        a_frag[mi] = *(const bf16x8_t*)(base + row * 128 +
                                        ((ks ^ (row >> 1)) & 7) * 16);
      }
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        int colB = wc + ni * 16;
        int g0 = kk * 8 + (lane >> 4) * 2;
        bf16x4_t lo = tr_read_b64_issue(
            base + 16384 + ((g0 * 8 + (colB >> 4)) * 64) * 2 + (lane & 15) * 8);
        bf16x4_t hi = tr_read_b64_issue(
            base + 16384 + (((g0 + 1) * 8 + (colB >> 4)) * 64) * 2 +
            (lane & 15) * 8);
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          b_frag[ni][e] = lo[e];
          b_frag[ni][e + 4] = hi[e];
        }
      }
      tr_read_drain();
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
    }
    __builtin_amdgcn_s_barrier();
  }
  const int crow_base = row0 + wr + (lane >> 4) * 4;
  const int ccol_base = col0 + wc + (lane & 15);
  if (nsplit == 1) {
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = crow_base + mi * 16 + r;
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          C[(long long)row * K + ccol_base + ni * 16] =
              f32_to_bf16(acc[mi][ni][r]);
      }
  } else {
    float* W = Cws + (long long)split * M * K;
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = crow_base + mi * 16 + r;
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          W[(long long)row * K + ccol_base + ni * 16] = acc[mi][ni][r];
      }
  }
}

// ===========================================================================
// v9/v10 — the guide's two-barrier phase discipline on the 256² tile
// (cdna_hip_programming.md §5 "256² 8-phase template"), with optional
// split-K so starved grids (few 256² tiles) still fill 256 CUs. Each phase:
//
//   { staged glds piece ∥ fragment ds_reads } → s_barrier →
//   lgkmcnt(0) → setprio(1) → 16 MFMA → setprio(0) → s_barrier
//
// The mid-phase barrier separates the memory phase from the MFMA phase
// chip-wide (T5 stagger); vmcnt stays counted (never 0 mid-loop).
//
// STAGE_MODE 0 (v9): v8's per-wave piece schedule — phase p stages piece p
//   of tile t+1 into buf^1; boundary wait vmcnt(2) after p0's own issue.
// STAGE_MODE 1 (v10): group-aligned liveness schedule — strips are issued
//   in the order their LDS slots die, which legalizes staging two tiles
//   ahead and a boundary wait of vmcnt(4) with issue→consume distance of
//   4-6 phases:
//     p0: A strips a2,a3 of t+1 (G1: rows 64-127/192-255, buf^1)
//     p1: B strips h2,h3 of t+1 (H1: cols (c&63)>=32, buf^1)
//     p2: B strips h0,h1 of t+2 (H0 of THIS buf — dead since p0)
//     p3: A strips a0,a1 of t+2 (G0 of THIS buf — dead since p2)
// ===========================================================================
template <int STAGE_MODE, int SPLITK>
__device__ __forceinline__ void gemm_nt_v9_body(
    const short* __restrict__ A, const short* __restrict__ B,
    short* __restrict__ C, float* __restrict__ Cws, int M, int N, int K,
    int nsplit) {
  if (!SPLITK) nsplit = 1;  // compile-time: no workspace path at all
  __shared__ short lds[2 * 2 * 2048 * 8];  // 2 buf × (A+B)[256][64] = 128 KB
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = (wave >> 2) * 128;
  const int wc = (wave & 3) * 64;
  const int nwg = gridDim.x;
  int wgid = blockIdx.x;
  {
    int q = nwg >> 3, r = nwg & 7;
    int xcd = wgid & 7, idx = wgid >> 3;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int split = SPLITK ? wgid % nsplit : 0;
  if (SPLITK) wgid /= nsplit;
  const int ntile_n = N / 256;
  const int row0 = (wgid / ntile_n) * 256;
  const int col0 = (wgid % ntile_n) * 256;

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int row_in_piece = lane >> 3;

  // stage ONE 8-row strip of operand `op` (0=A at strip*1024, 1=B at
  // 32768+strip*1024) for K-tile kt into buffer buf. The XOR source key
  // matches the ds_read side's ((ks ^ (row>>1)) & 7) for any strip index.
  auto stage_strip = [&](int buf, int kt, int strip, int op) {
    const long long k0 = (long long)kt * BK;
    char* base = (char*)lds + buf * 65536 + op * 32768;
    const int src_koff = (((lane & 7) ^ (strip * 4 + (lane >> 4))) & 7) * 8;
    const short* src = (op ? B : A) +
        (long long)((op ? col0 : row0) + strip * 8 + row_in_piece) * K + k0 +
        src_koff;
    __builtin_amdgcn_global_load_lds((global_uint*)src,
                                     (lds_uint*)(base + strip * 1024), 16, 0, 0);
  };

  const int nk_total = K / BK;
  const int per = (nk_total + nsplit - 1) / nsplit;
  const int kt0 = split * per;
  const int kt1 = (kt0 + per < nk_total) ? kt0 + per : nk_total;
  const int nk = kt1 - kt0;
  const int crow_base = row0 + wr + (lane >> 4) * 4;
  const int ccol_base = col0 + wc + (lane & 15);
  // dispatch guarantees nk_total % nsplit == 0, so no slice is empty

  // per-wave strip sets (STAGE_MODE 1): a0,a1 ∈ G0 (rows 0-63,128-191),
  // a2,a3 = +8 ∈ G1; h0,h1 ∈ H0 (strips with (s&7)<4), h2,h3 = +4 ∈ H1.
  const int aG0 = (wave >> 2) * 16 + (wave & 3) * 2;
  const int hH0 = (wave >> 1) * 8 + (wave & 1) * 2;

  if (STAGE_MODE == 0) {
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      stage_strip(0, kt0, wave * 4 + p, 0);
      stage_strip(0, kt0, wave * 4 + p, 1);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  } else {
    // prologue mirrors the steady-state issue order; tile kt0+1's H0/G0
    // ride across the first boundary in flight (vmcnt(4))
    stage_strip(0, kt0, hH0, 1);
    stage_strip(0, kt0, hH0 + 1, 1);
    stage_strip(0, kt0, aG0, 0);
    stage_strip(0, kt0, aG0 + 1, 0);
    stage_strip(0, kt0, aG0 + 8, 0);
    stage_strip(0, kt0, aG0 + 9, 0);
    stage_strip(0, kt0, hH0 + 4, 1);
    stage_strip(0, kt0, hH0 + 5, 1);
    if (nk > 1) {
      stage_strip(1, kt0 + 1, hH0, 1);
      stage_strip(1, kt0 + 1, hH0 + 1, 1);
      stage_strip(1, kt0 + 1, aG0, 0);
      stage_strip(1, kt0 + 1, aG0 + 1, 0);
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
  }
  __builtin_amdgcn_s_barrier();

  for (int kti = 0; kti < nk; ++kti) {
    const int kt = kt0 + kti;
    const int cur = kti & 1;
    char* base = (char*)lds + cur * 65536;
    bf16x8_t b_frag[2][2];
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      const int mh = p & 1, nh = p >> 1;
      // ---- memory phase: staged glds piece, then this phase's ds_reads
      if (STAGE_MODE == 0) {
        if (kti + 1 < nk) {
          stage_strip(cur ^ 1, kt + 1, wave * 4 + p, 0);
          stage_strip(cur ^ 1, kt + 1, wave * 4 + p, 1);
        }
        if (p == 0 && kti > 0) {
          // everything except p0's own 2 strips must have landed
          if (kti + 1 < nk)
            asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
          else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
          __builtin_amdgcn_s_barrier();
        }
      } else {
        if (p == 0 && kti > 0) {
          // in flight: the 4 strips issued at p2/p3 of the previous tile
          // (tile kt+1's H0/G0); everything of tile kt has landed
          if (kti + 1 < nk)
            asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
          else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
          __builtin_amdgcn_s_barrier();
        }
        switch (p) {
          case 0:
            if (kti + 1 < nk) {
              stage_strip(cur ^ 1, kt + 1, aG0 + 8, 0);
              stage_strip(cur ^ 1, kt + 1, aG0 + 9, 0);
            }
            break;
          case 1:
            if (kti + 1 < nk) {
              stage_strip(cur ^ 1, kt + 1, hH0 + 4, 1);
              stage_strip(cur ^ 1, kt + 1, hH0 + 5, 1);
            }
            break;
          case 2:
            if (kti + 2 < nk) {
              stage_strip(cur, kt + 2, hH0, 1);
              stage_strip(cur, kt + 2, hH0 + 1, 1);
            }
            break;
          case 3:
            if (kti + 2 < nk) {
              stage_strip(cur, kt + 2, aG0, 0);
              stage_strip(cur, kt + 2, aG0 + 1, 0);
            }
            break;
        }
      }
      if (mh == 0) {
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
#pragma unroll
          for (int kk = 0; kk < 2; ++kk) {
            int row = wc + (nh * 2 + ni) * 16 + (lane & 15);
            int ks = kk * 4 + (lane >> 4);
            b_frag[ni][kk] = *(const bf16x8_t*)(
                base + 32768 + row * 128 + ((ks ^ (row >> 1)) & 7) * 16);
          }
      }
      bf16x8_t a_frag[4][2];
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
          int row = wr + (mh * 4 + mi) * 16 + (lane & 15);
          int ks = kk * 4 + (lane >> 4);
          a_frag[mi][kk] = *(const bf16x8_t*)(
              base + row * 128 + ((ks ^ (row >> 1)) & 7) * 16);
        }
      // ---- mid-phase barrier: memory phase ends, MFMA phase begins ----
      __builtin_amdgcn_sched_barrier(0);
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
#pragma unroll
          for (int kk = 0; kk < 2; ++kk)
            acc[mh * 4 + mi][nh * 2 + ni] =
                __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a_frag[mi][kk], b_frag[ni][kk],
                    acc[mh * 4 + mi][nh * 2 + ni], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_s_barrier();
    }
    if (kti + 1 == nk) asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  if (!SPLITK) {
#pragma unroll
    for (int mi = 0; mi < 8; ++mi)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = crow_base + mi * 16 + r;
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          C[(long long)row * N + ccol_base + ni * 16] =
              f32_to_bf16(acc[mi][ni][r]);
      }
  } else {
    float* W = Cws + (long long)split * M * N;
#pragma unroll
    for (int mi = 0; mi < 8; ++mi)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = crow_base + mi * 16 + r;
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          W[(long long)row * N + ccol_base + ni * 16] = acc[mi][ni][r];
      }
  }
}

extern "C" __global__ void __launch_bounds__(512)
gemm_nt_bf16_v9_kernel(const short* __restrict__ A, const short* __restrict__ B,
                       short* __restrict__ C, float* __restrict__ Cws, int M,
                       int N, int K, int nsplit) {
  gemm_nt_v9_body<0, 0>(A, B, C, nullptr, M, N, K, 1);
}

extern "C" __global__ void __launch_bounds__(512)
gemm_nt_bf16_v9sk_kernel(const short* __restrict__ A,
                         const short* __restrict__ B, short* __restrict__ C,
                         float* __restrict__ Cws, int M, int N, int K,
                         int nsplit) {
  gemm_nt_v9_body<0, 1>(A, B, C, Cws, M, N, K, nsplit);
}

extern "C" __global__ void __launch_bounds__(512)
gemm_nt_bf16_v10_kernel(const short* __restrict__ A,
                        const short* __restrict__ B, short* __restrict__ C,
                        float* __restrict__ Cws, int M, int N, int K,
                        int nsplit) {
  gemm_nt_v9_body<1, 0>(A, B, C, nullptr, M, N, K, 1);
}

extern "C" __global__ void __launch_bounds__(512)
gemm_nt_bf16_v10sk_kernel(const short* __restrict__ A,
                          const short* __restrict__ B, short* __restrict__ C,
                          float* __restrict__ Cws, int M, int N, int K,
                          int nsplit) {
  gemm_nt_v9_body<1, 1>(A, B, C, Cws, M, N, K, nsplit);
}

// ===========================================================================
// wgrad v2 — 256² output tile (8 waves as 2M×4N, wave = 128×64 out), BK=64
// reduction rows, double-buffered glds staging in [m-grp][colsub][4][16]
// subtile order, tr_b16 fragment gathers. Twice the compute-per-staged-byte
// of the 128² wgrad; requires Na%256 && Nb%256. Split-K as in wgrad v1.
// ===========================================================================
extern "C" __global__ void __launch_bounds__(512)
gemm_wgrad2_bf16_kernel(const short* __restrict__ A, const short* __restrict__ B,
                        short* __restrict__ C, float* __restrict__ Cws,
                        int M, int Na, int Nb, int nsplit) {
  __shared__ short lds[2 * 2 * 16384];  // 2 buf × (A+B)[64][256] = 128 KB
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = (wave >> 2) * 128;  // output rows (Na side)
  const int wc = (wave & 3) * 64;    // output cols (Nb side)
  const int ntile_n = Nb / 256;
  int wgid = blockIdx.x;
  {
    int nwg = gridDim.x;
    int q = nwg >> 3, r = nwg & 7;
    int xcd = wgid & 7, idx = wgid >> 3;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int split = wgid % nsplit;
  wgid /= nsplit;
  const int row0 = (wgid / ntile_n) * 256;
  const int col0 = (wgid % ntile_n) * 256;

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int src_m_in_grp = (lane & 7) >> 1;
  const int src_col = ((lane >> 3) * 16 + (lane & 1) * 8);

  // one glds = 4 m-rows × 128 out cols; per wave: 2 (grp,half) pairs × 2 ops
  auto stage = [&](int buf, int mt) {
    const long long m0 = (long long)mt * BK;
    char* base = (char*)lds + buf * 65536;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int grp = wave * 2 + (i >> 1);  // m-group 0..15
      int half = i & 1;               // 128-col half
      const short* srcA = A + (m0 + grp * 4 + src_m_in_grp) * (long long)Na +
                          row0 + half * 128 + src_col;
      __builtin_amdgcn_global_load_lds(
          (global_uint*)srcA,
          (lds_uint*)(base + grp * 2048 + half * 1024), 16, 0, 0);
      const short* srcB = B + (m0 + grp * 4 + src_m_in_grp) * (long long)Nb +
                          col0 + half * 128 + src_col;
      __builtin_amdgcn_global_load_lds(
          (global_uint*)srcB,
          (lds_uint*)(base + 32768 + grp * 2048 + half * 1024), 16, 0, 0);
    }
  };

  const int nk_total = M / BK;
  const int per = (nk_total + nsplit - 1) / nsplit;
  const int mt0 = split * per;
  const int mt1 = (mt0 + per < nk_total) ? mt0 + per : nk_total;
  const int nk = mt1 - mt0;
  const int crow_base = row0 + wr + (lane >> 4) * 4;
  const int ccol_base = col0 + wc + (lane & 15);
  if (nk <= 0) {
    if (nsplit > 1) {
      float* W = Cws + (long long)split * Na * Nb;
#pragma unroll
      for (int mi = 0; mi < 8; ++mi)
#pragma unroll
        for (int r = 0; r < 4; ++r)
#pragma unroll
          for (int ni = 0; ni < 4; ++ni)
            W[(long long)(crow_base + mi * 16 + r) * Nb + ccol_base + ni * 16] =
                0.f;
    }
    return;
  }
  stage(0, mt0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (int mti = 0; mti < nk; ++mti) {
    int mt = mt0 + mti;
    int cur = mti & 1;
    if (mti + 1 < nk) {
      stage(cur ^ 1, mt + 1);
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    char* base = (char*)lds + cur * 65536;
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8_t a_frag[8], b_frag[4];
      int g0 = kk * 8 + (lane >> 4) * 2;
#pragma unroll
      for (int fi = 0; fi < 8; ++fi) {
        int colA = wr + fi * 16;
        bf16x4_t lo = tr_read_b64_issue(
            base + ((g0 * 16 + (colA >> 4)) * 64) * 2 + (lane & 15) * 8);
        bf16x4_t hi = tr_read_b64_issue(
            base + (((g0 + 1) * 16 + (colA >> 4)) * 64) * 2 + (lane & 15) * 8);
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          a_frag[fi][e] = lo[e];
          a_frag[fi][e + 4] = hi[e];
        }
      }
#pragma unroll
      for (int fi = 0; fi < 4; ++fi) {
        int colB = wc + fi * 16;
        bf16x4_t lo = tr_read_b64_issue(
            base + 32768 + ((g0 * 16 + (colB >> 4)) * 64) * 2 +
            (lane & 15) * 8);
        bf16x4_t hi = tr_read_b64_issue(
            base + 32768 + (((g0 + 1) * 16 + (colB >> 4)) * 64) * 2 +
            (lane & 15) * 8);
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          b_frag[fi][e] = lo[e];
          b_frag[fi][e + 4] = hi[e];
        }
      }
      tr_read_drain();
#pragma unroll
      for (int mi = 0; mi < 8; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
    }
    __builtin_amdgcn_s_barrier();
  }
  if (nsplit == 1) {
#pragma unroll
    for (int mi = 0; mi < 8; ++mi)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = crow_base + mi * 16 + r;
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          C[(long long)row * Nb + ccol_base + ni * 16] =
              f32_to_bf16(acc[mi][ni][r]);
      }
  } else {
    float* W = Cws + (long long)split * Na * Nb;
#pragma unroll
    for (int mi = 0; mi < 8; ++mi)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = crow_base + mi * 16 + r;
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          W[(long long)row * Nb + ccol_base + ni * 16] = acc[mi][ni][r];
      }
  }
}
