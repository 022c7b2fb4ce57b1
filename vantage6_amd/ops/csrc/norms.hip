// LayerNorm / RMSNorm forward + backward (gfx950).
//
// One 256-thread workgroup per row-group (grid-stride over rows), bf16 I/O,
// fp32 accumulation, 16-B vector accesses. A row (<= 16 KB bf16 at H=8192)
// is L1-resident between the two passes of forward. Backward accumulates
// per-block dweight/dbias partials into an fp32 workspace; a small second
// kernel column-reduces the workspace (deterministic — no atomics).
#include "common.h"

#define NTHREADS 256
// max row chunks a thread tracks in registers: H <= NTHREADS*8*MAXCH
#define MAXCH 8

DEVINL float block_reduce_sum(float v, float* red) {
  // wave-level butterfly (no barriers), then one cross-wave LDS step —
  // the 8-level barrier tree cost ~16 barriers per row (88 us/call LN bwd
  // at H=768: mostly barrier latency, 160 of 256 threads idle)
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v += __shfl_xor(v, off, 64);
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = v;
  __syncthreads();
  float out = red[0] + red[1] + red[2] + red[3];
  __syncthreads();  // red[] reusable by the next reduction
  return out;
}

// ---------------------------------------------------------------- LN fwd
extern "C" __global__ void __launch_bounds__(NTHREADS)
layernorm_fwd_kernel(const short* __restrict__ x, const short* __restrict__ w,
                     const short* __restrict__ b, short* __restrict__ y,
                     float* __restrict__ mean_out, float* __restrict__ rstd_out,
                     long long nrows, int H, float eps) {
  __shared__ float red[NTHREADS];
  for (long long row = blockIdx.x; row < nrows; row += gridDim.x) {
    const short* xr = x + row * (long long)H;
    short* yr = y + row * (long long)H;
    float s1 = 0.f, s2 = 0.f;
    for (int i = threadIdx.x * 8; i < H; i += NTHREADS * 8) {
      if (i + 8 <= H) {
        s16x8 v = *(const s16x8*)(xr + i);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = bf16_to_f32(v[j]);
          s1 += f; s2 = fmaf(f, f, s2);
        }
      } else {
        for (int k = i; k < H; ++k) { float f = bf16_to_f32(xr[k]); s1 += f; s2 = fmaf(f, f, s2); }
      }
    }
    s1 = block_reduce_sum(s1, red);
    s2 = block_reduce_sum(s2, red);
    float mean = s1 / H;
    float var = s2 / H - mean * mean;
    float rstd = rsqrtf(var + eps);
    if (threadIdx.x == 0) { mean_out[row] = mean; rstd_out[row] = rstd; }
    for (int i = threadIdx.x * 8; i < H; i += NTHREADS * 8) {
      if (i + 8 <= H) {
        s16x8 v = *(const s16x8*)(xr + i);
        s16x8 wv = *(const s16x8*)(w + i);
        s16x8 bv = *(const s16x8*)(b + i);
        s16x8 o;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float xh = (bf16_to_f32(v[j]) - mean) * rstd;
          o[j] = f32_to_bf16(fmaf(xh, bf16_to_f32(wv[j]), bf16_to_f32(bv[j])));
        }
        *(s16x8*)(yr + i) = o;
      } else {
        for (int k = i; k < H; ++k) {
          float xh = (bf16_to_f32(xr[k]) - mean) * rstd;
          yr[k] = f32_to_bf16(fmaf(xh, bf16_to_f32(w[k]), bf16_to_f32(b[k])));
        }
      }
    }
  }
}

// ---------------------------------------------------------------- LN bwd
extern "C" __global__ void __launch_bounds__(NTHREADS)
layernorm_bwd_kernel(const short* __restrict__ dy, const short* __restrict__ x,
                     const short* __restrict__ w, const float* __restrict__ mean,
                     const float* __restrict__ rstd, short* __restrict__ dx,
                     float* __restrict__ dw_ws, float* __restrict__ db_ws,
                     long long nrows, int H) {
  __shared__ float red[NTHREADS];
  float dw_acc[MAXCH * 8];
  float db_acc[MAXCH * 8];
  const int nch = (H + NTHREADS * 8 - 1) / (NTHREADS * 8);
#pragma unroll
  for (int c = 0; c < MAXCH * 8; ++c) { dw_acc[c] = 0.f; db_acc[c] = 0.f; }
  for (long long row = blockIdx.x; row < nrows; row += gridDim.x) {
    const short* xr = x + row * (long long)H;
    const short* dyr = dy + row * (long long)H;
    short* dxr = dx + row * (long long)H;
    float mu = mean[row], rs = rstd[row];
    float c1 = 0.f, c2 = 0.f;  // mean(wdy), mean(wdy*xhat)
    for (int i = threadIdx.x * 8, c = 0; i < H; i += NTHREADS * 8, ++c) {
      for (int j = 0; j < 8; ++j) {
        int k = i + j;
        if (k >= H) break;
        float xh = (bf16_to_f32(xr[k]) - mu) * rs;
        float d = bf16_to_f32(dyr[k]);
        float wdy = bf16_to_f32(w[k]) * d;
        c1 += wdy; c2 = fmaf(wdy, xh, c2);
      }
    }
    c1 = block_reduce_sum(c1, red) / H;
    c2 = block_reduce_sum(c2, red) / H;
#pragma unroll MAXCH
    for (int c = 0; c < MAXCH; ++c) {
      int i = threadIdx.x * 8 + c * NTHREADS * 8;
      if (i >= H) break;
      if (i + 8 <= H) {
        s16x8 xv = *(const s16x8*)(xr + i);
        s16x8 dv = *(const s16x8*)(dyr + i);
        s16x8 wv = *(const s16x8*)(w + i);
        s16x8 o;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float xh = (bf16_to_f32(xv[j]) - mu) * rs;
          float d = bf16_to_f32(dv[j]);
          float wdy = bf16_to_f32(wv[j]) * d;
          o[j] = f32_to_bf16((wdy - c1 - xh * c2) * rs);
          dw_acc[c * 8 + j] = fmaf(d, xh, dw_acc[c * 8 + j]);
          db_acc[c * 8 + j] += d;
        }
        *(s16x8*)(dxr + i) = o;
      } else {
        for (int j = 0; i + j < H && j < 8; ++j) {
          int k = i + j;
          float xh = (bf16_to_f32(xr[k]) - mu) * rs;
          float d = bf16_to_f32(dyr[k]);
          float wdy = bf16_to_f32(w[k]) * d;
          dxr[k] = f32_to_bf16((wdy - c1 - xh * c2) * rs);
          dw_acc[c * 8 + j] = fmaf(d, xh, dw_acc[c * 8 + j]);
          db_acc[c * 8 + j] += d;
        }
      }
    }
  }
  // accumulate into 16 contention slices [16][H] (serial column sums over
  // [grid][H] workspaces cost 234 us/call — cross-XCD reads; same fix as BN)
  float* dwr = dw_ws + (long long)(blockIdx.x & 15) * H;
  float* dbr = db_ws + (long long)(blockIdx.x & 15) * H;
#pragma unroll MAXCH
  for (int c = 0; c < MAXCH; ++c) {
    int i = threadIdx.x * 8 + c * NTHREADS * 8;
    if (i >= H) break;
    for (int j = 0; j < 8 && i + j < H; ++j) {
      atomicAdd(&dwr[i + j], dw_acc[c * 8 + j]);
      atomicAdd(&dbr[i + j], db_acc[c * 8 + j]);
    }
  }
}

// -------------------------------------------------------------- RMS fwd
extern "C" __global__ void __launch_bounds__(NTHREADS)
rmsnorm_fwd_kernel(const short* __restrict__ x, const short* __restrict__ w,
                   short* __restrict__ y, float* __restrict__ rstd_out,
                   long long nrows, int H, float eps) {
  __shared__ float red[NTHREADS];
  for (long long row = blockIdx.x; row < nrows; row += gridDim.x) {
    const short* xr = x + row * (long long)H;
    short* yr = y + row * (long long)H;
    float s2 = 0.f;
    for (int i = threadIdx.x * 8; i < H; i += NTHREADS * 8) {
      if (i + 8 <= H) {
        s16x8 v = *(const s16x8*)(xr + i);
#pragma unroll
        for (int j = 0; j < 8; ++j) { float f = bf16_to_f32(v[j]); s2 = fmaf(f, f, s2); }
      } else {
        for (int k = i; k < H; ++k) { float f = bf16_to_f32(xr[k]); s2 = fmaf(f, f, s2); }
      }
    }
    s2 = block_reduce_sum(s2, red);
    float rstd = rsqrtf(s2 / H + eps);
    if (threadIdx.x == 0) rstd_out[row] = rstd;
    for (int i = threadIdx.x * 8; i < H; i += NTHREADS * 8) {
      if (i + 8 <= H) {
        s16x8 v = *(const s16x8*)(xr + i);
        s16x8 wv = *(const s16x8*)(w + i);
        s16x8 o;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          o[j] = f32_to_bf16(bf16_to_f32(v[j]) * rstd * bf16_to_f32(wv[j]));
        *(s16x8*)(yr + i) = o;
      } else {
        for (int k = i; k < H; ++k)
          yr[k] = f32_to_bf16(bf16_to_f32(xr[k]) * rstd * bf16_to_f32(w[k]));
      }
    }
  }
}

// -------------------------------------------------------------- RMS bwd
extern "C" __global__ void __launch_bounds__(NTHREADS)
rmsnorm_bwd_kernel(const short* __restrict__ dy, const short* __restrict__ x,
                   const short* __restrict__ w, const float* __restrict__ rstd,
                   short* __restrict__ dx, float* __restrict__ dw_ws,
                   long long nrows, int H) {
  __shared__ float red[NTHREADS];
  float dw_acc[MAXCH * 8];
#pragma unroll
  for (int c = 0; c < MAXCH * 8; ++c) dw_acc[c] = 0.f;
  for (long long row = blockIdx.x; row < nrows; row += gridDim.x) {
    const short* xr = x + row * (long long)H;
    const short* dyr = dy + row * (long long)H;
    short* dxr = dx + row * (long long)H;
    float rs = rstd[row];
    float c2 = 0.f;  // mean(wdy * xhat)
    for (int i = threadIdx.x * 8; i < H; i += NTHREADS * 8) {
      for (int j = 0; j < 8; ++j) {
        int k = i + j;
        if (k >= H) break;
        float xh = bf16_to_f32(xr[k]) * rs;
        float wdy = bf16_to_f32(w[k]) * bf16_to_f32(dyr[k]);
        c2 = fmaf(wdy, xh, c2);
      }
    }
    c2 = block_reduce_sum(c2, red) / H;
#pragma unroll MAXCH
    for (int c = 0; c < MAXCH; ++c) {
      int i = threadIdx.x * 8 + c * NTHREADS * 8;
      if (i >= H) break;
      for (int j = 0; j < 8 && i + j < H; ++j) {
        int k = i + j;
        float xh = bf16_to_f32(xr[k]) * rs;
        float d = bf16_to_f32(dyr[k]);
        float wdy = bf16_to_f32(w[k]) * d;
        dxr[k] = f32_to_bf16((wdy - xh * c2) * rs);
        dw_acc[c * 8 + j] = fmaf(d, xh, dw_acc[c * 8 + j]);
      }
    }
  }
  float* dwr = dw_ws + (long long)(blockIdx.x & 15) * H;
#pragma unroll MAXCH
  for (int c = 0; c < MAXCH; ++c) {
    int i = threadIdx.x * 8 + c * NTHREADS * 8;
    if (i >= H) break;
    for (int j = 0; j < 8 && i + j < H; ++j)
      atomicAdd(&dwr[i + j], dw_acc[c * 8 + j]);
  }
}

// column-sum of an fp32 workspace [nblocks][H] -> bf16 [H]
// (nblocks is 16 with the atomic-slice layout)
extern "C" __global__ void __launch_bounds__(NTHREADS)
colsum_ws_kernel(const float* __restrict__ ws, short* __restrict__ out,
                 int nblocks, int H) {
  int h = blockIdx.x * NTHREADS + threadIdx.x;
  if (h >= H) return;
  float acc = 0.f;
  for (int b = 0; b < nblocks; ++b) acc += ws[(long long)b * H + h];
  out[h] = f32_to_bf16(acc);
}

extern "C" __global__ void __launch_bounds__(NTHREADS)
zero_f32_kernel(float* __restrict__ p, long long n) {
  long long i = (long long)blockIdx.x * NTHREADS + threadIdx.x;
  long long stride = (long long)gridDim.x * NTHREADS;
  for (; i < n; i += stride) p[i] = 0.f;
}

// ---------------------------------------------------------------------------
// Row-per-WAVE backward variants (H <= 2048): 4 rows in flight per block,
// all reductions are wave-local __shfl_xor butterflies — no barriers at
// all. The block-per-row kernels above are latency-bound at small H
// (83 us/call at H=768: 96 of 256 threads active, serial rows, barrier
// trees); these run the practical transformer widths (768/1024/2048).
// ---------------------------------------------------------------------------
#define WMAXCH 4  // H <= 64*8*WMAXCH = 2048

DEVINL float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

template <int WCH>
__device__ __forceinline__ void layernorm_bwd_wave_body(
    const short* __restrict__ dy, const short* __restrict__ x,
    const short* __restrict__ w, const float* __restrict__ mean,
    const float* __restrict__ rstd, short* __restrict__ dx,
    float* __restrict__ dw_ws, float* __restrict__ db_ws, long long nrows,
    int H) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  float dw_acc[WCH * 8];
  float db_acc[WCH * 8];
#pragma unroll
  for (int c = 0; c < WCH * 8; ++c) { dw_acc[c] = 0.f; db_acc[c] = 0.f; }
  // w is row-invariant: load it ONCE; x/dy are register-cached across the
  // two passes; WCH is a compile-time chunk count so H=768 doesn't pay
  // the H=2048 register footprint
  s16x8 wreg[WCH];
#pragma unroll
  for (int c = 0; c < WCH; ++c) {
    int i = lane * 8 + c * 512;
    wreg[c] = i < H ? *(const s16x8*)(w + i) : s16x8{0, 0, 0, 0, 0, 0, 0, 0};
  }
  for (long long row = (long long)blockIdx.x * 4 + wave; row < nrows;
       row += (long long)gridDim.x * 4) {
    const short* xr = x + row * (long long)H;
    const short* dyr = dy + row * (long long)H;
    short* dxr = dx + row * (long long)H;
    float mu = mean[row], rs = rstd[row];
    s16x8 xv[WCH], dv[WCH];
#pragma unroll
    for (int c = 0; c < WCH; ++c) {  // all loads in flight together
      int i = lane * 8 + c * 512;
      if (i < H) {
        xv[c] = *(const s16x8*)(xr + i);
        dv[c] = *(const s16x8*)(dyr + i);
      }
    }
    float c1 = 0.f, c2 = 0.f;
#pragma unroll WCH
    for (int c = 0; c < WCH; ++c) {
      int i = lane * 8 + c * 512;
      if (i >= H) break;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xh = (bf16_to_f32(xv[c][j]) - mu) * rs;
        float wdy = bf16_to_f32(wreg[c][j]) * bf16_to_f32(dv[c][j]);
        c1 += wdy;
        c2 = fmaf(wdy, xh, c2);
      }
    }
    c1 = wave_reduce_sum(c1) / H;
    c2 = wave_reduce_sum(c2) / H;
#pragma unroll WCH
    for (int c = 0; c < WCH; ++c) {
      int i = lane * 8 + c * 512;
      if (i >= H) break;
      s16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xh = (bf16_to_f32(xv[c][j]) - mu) * rs;
        float d = bf16_to_f32(dv[c][j]);
        float wdy = bf16_to_f32(wreg[c][j]) * d;
        o[j] = f32_to_bf16((wdy - c1 - xh * c2) * rs);
        dw_acc[c * 8 + j] = fmaf(d, xh, dw_acc[c * 8 + j]);
        db_acc[c * 8 + j] += d;
      }
      *(s16x8*)(dxr + i) = o;
    }
  }
  // block-level LDS combine first: one global atomic set per BLOCK
  __shared__ float comb[2 * 2048];
  if (wave == 0) {
#pragma unroll WCH
    for (int c = 0; c < WCH; ++c) {
      int i = lane * 8 + c * 512;
      if (i >= H) break;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        comb[i + j] = dw_acc[c * 8 + j];
        comb[2048 + i + j] = db_acc[c * 8 + j];
      }
    }
  }
  __syncthreads();
#pragma unroll
  for (int wv = 1; wv < 4; ++wv) {
    if (wave == wv) {
#pragma unroll WCH
      for (int c = 0; c < WCH; ++c) {
        int i = lane * 8 + c * 512;
        if (i >= H) break;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          comb[i + j] += dw_acc[c * 8 + j];
          comb[2048 + i + j] += db_acc[c * 8 + j];
        }
      }
    }
    __syncthreads();
  }
  float* dwr = dw_ws + (long long)(blockIdx.x & 15) * H;
  float* dbr = db_ws + (long long)(blockIdx.x & 15) * H;
  if (wave == 0) {
#pragma unroll WCH
    for (int c = 0; c < WCH; ++c) {
      int i = lane * 8 + c * 512;
      if (i >= H) break;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        atomicAdd(&dwr[i + j], comb[i + j]);
        atomicAdd(&dbr[i + j], comb[2048 + i + j]);
      }
    }
  }
}

extern "C" __global__ void __launch_bounds__(NTHREADS)
layernorm_bwd_wave_kernel(const short* __restrict__ dy,
                          const short* __restrict__ x,
                          const short* __restrict__ w,
                          const float* __restrict__ mean,
                          const float* __restrict__ rstd,
                          short* __restrict__ dx, float* __restrict__ dw_ws,
                          float* __restrict__ db_ws, long long nrows, int H) {
  layernorm_bwd_wave_body<4>(dy, x, w, mean, rstd, dx, dw_ws, db_ws, nrows, H);
}

extern "C" __global__ void __launch_bounds__(NTHREADS)
layernorm_bwd_wave2_kernel(const short* __restrict__ dy,
                           const short* __restrict__ x,
                           const short* __restrict__ w,
                           const float* __restrict__ mean,
                           const float* __restrict__ rstd,
                           short* __restrict__ dx, float* __restrict__ dw_ws,
                           float* __restrict__ db_ws, long long nrows, int H) {
  layernorm_bwd_wave_body<2>(dy, x, w, mean, rstd, dx, dw_ws, db_ws, nrows, H);
}

extern "C" __global__ void __launch_bounds__(NTHREADS)
rmsnorm_bwd_wave_kernel(const short* __restrict__ dy,
                        const short* __restrict__ x,
                        const short* __restrict__ w,
                        const float* __restrict__ rstd, short* __restrict__ dx,
                        float* __restrict__ dw_ws, long long nrows, int H) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  float dw_acc[WMAXCH * 8];
#pragma unroll
  for (int c = 0; c < WMAXCH * 8; ++c) dw_acc[c] = 0.f;
  s16x8 wreg[WMAXCH];  // row-invariant; x/dy register-cached across passes
#pragma unroll
  for (int c = 0; c < WMAXCH; ++c) {
    int i = lane * 8 + c * 512;
    wreg[c] = i < H ? *(const s16x8*)(w + i) : s16x8{0, 0, 0, 0, 0, 0, 0, 0};
  }
  for (long long row = (long long)blockIdx.x * 4 + wave; row < nrows;
       row += (long long)gridDim.x * 4) {
    const short* xr = x + row * (long long)H;
    const short* dyr = dy + row * (long long)H;
    short* dxr = dx + row * (long long)H;
    float rs = rstd[row];
    s16x8 xv[WMAXCH], dv[WMAXCH];
#pragma unroll
    for (int c = 0; c < WMAXCH; ++c) {
      int i = lane * 8 + c * 512;
      if (i < H) {
        xv[c] = *(const s16x8*)(xr + i);
        dv[c] = *(const s16x8*)(dyr + i);
      }
    }
    float c2 = 0.f;
#pragma unroll WMAXCH
    for (int c = 0; c < WMAXCH; ++c) {
      int i = lane * 8 + c * 512;
      if (i >= H) break;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xh = bf16_to_f32(xv[c][j]) * rs;
        float wdy = bf16_to_f32(wreg[c][j]) * bf16_to_f32(dv[c][j]);
        c2 = fmaf(wdy, xh, c2);
      }
    }
    c2 = wave_reduce_sum(c2) / H;
#pragma unroll WMAXCH
    for (int c = 0; c < WMAXCH; ++c) {
      int i = lane * 8 + c * 512;
      if (i >= H) break;
      s16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xh = bf16_to_f32(xv[c][j]) * rs;
        float d = bf16_to_f32(dv[c][j]);
        float wdy = bf16_to_f32(wreg[c][j]) * d;
        o[j] = f32_to_bf16((wdy - xh * c2) * rs);
        dw_acc[c * 8 + j] = fmaf(d, xh, dw_acc[c * 8 + j]);
      }
      *(s16x8*)(dxr + i) = o;
    }
  }
  __shared__ float comb[2048];
  if (wave == 0) {
#pragma unroll WMAXCH
    for (int c = 0; c < WMAXCH; ++c) {
      int i = lane * 8 + c * 512;
      if (i >= H) break;
#pragma unroll
      for (int j = 0; j < 8; ++j) comb[i + j] = dw_acc[c * 8 + j];
    }
  }
  __syncthreads();
#pragma unroll
  for (int wv = 1; wv < 4; ++wv) {
    if (wave == wv) {
#pragma unroll WMAXCH
      for (int c = 0; c < WMAXCH; ++c) {
        int i = lane * 8 + c * 512;
        if (i >= H) break;
#pragma unroll
        for (int j = 0; j < 8; ++j) comb[i + j] += dw_acc[c * 8 + j];
      }
    }
    __syncthreads();
  }
  float* dwr = dw_ws + (long long)(blockIdx.x & 15) * H;
  if (wave == 0) {
#pragma unroll WMAXCH
    for (int c = 0; c < WMAXCH; ++c) {
      int i = lane * 8 + c * 512;
      if (i >= H) break;
#pragma unroll
      for (int j = 0; j < 8; ++j) atomicAdd(&dwr[i + j], comb[i + j]);
    }
  }
}

// ------------------------------------------------------------ bf16 colsum
// db[N] = sum over M rows of dy[M,N] (Linear bias grad). Row-major bf16
// reads are fully coalesced (vs torch's generic reduce, which is also NOT
// hipGraph-replay-safe: its semaphore workspace is memset at allocation
// time only). 16-slice fp32 workspace + colsum_ws finalize, all kernels —
// zeroing included — so every byte of state is re-written on each replay.
extern "C" __global__ void __launch_bounds__(256)
colsum_bf16_kernel(const short* __restrict__ dy, float* __restrict__ ws,
                   long long M, int N, int rchunk) {
  const int col0 = blockIdx.x * 2048 + threadIdx.x * 8;
  if (col0 + 8 > N) return;
  const long long r0 = (long long)blockIdx.y * rchunk;
  const long long r1 = (r0 + rchunk < M) ? r0 + rchunk : M;
  float acc[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) acc[j] = 0.f;
  long long r = r0;
  for (; r + 3 < r1; r += 4) {  // 4 independent loads in flight
    s16x8 v0 = *(const s16x8*)(dy + r * N + col0);
    s16x8 v1 = *(const s16x8*)(dy + (r + 1) * N + col0);
    s16x8 v2 = *(const s16x8*)(dy + (r + 2) * N + col0);
    s16x8 v3 = *(const s16x8*)(dy + (r + 3) * N + col0);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      acc[j] += (bf16_to_f32(v0[j]) + bf16_to_f32(v1[j])) +
                (bf16_to_f32(v2[j]) + bf16_to_f32(v3[j]));
  }
  for (; r < r1; ++r) {
    s16x8 v = *(const s16x8*)(dy + r * N + col0);
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] += bf16_to_f32(v[j]);
  }
  float* slice = ws + (long long)(blockIdx.y & 15) * N;
#pragma unroll
  for (int j = 0; j < 8; ++j) atomicAdd(&slice[col0 + j], acc[j]);
}

// -------------------------------------------- fused cross-entropy (LM head)
// One block per row, two HBM-light passes (the second re-reads the row from
// L2 — a 256 KB bf16 row stays resident). Writes lse[r] and
// loss[r] = lse - x[target] (0 for ignored rows). Replaces the torch
// fp32 softmax+nll chain on a 128k-vocab LM head, whose fp32 logits
// materialisation alone moves 2x the row's bf16 bytes (profile: ~5 ms of
// cunn_SoftMax* + bfloat16_copy per llama round). V must be % 8.
extern "C" __global__ void __launch_bounds__(256)
ce_fwd_kernel(const short* __restrict__ x, const long long* __restrict__ t,
              float* __restrict__ lse, float* __restrict__ loss,
              long long V, long long ignore_index) {
  __shared__ float red[4];
  int row = blockIdx.x;
  const short* xr = x + (long long)row * V;
  float m = -1e30f;
  for (long long i = threadIdx.x * 8LL; i < V; i += 256 * 8) {
    s16x8 v8 = *(const s16x8*)(xr + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) m = fmaxf(m, bf16_to_f32(v8[j]));
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) m = fmaxf(m, __shfl_xor(m, off, 64));
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = m;
  __syncthreads();
  m = fmaxf(fmaxf(red[0], red[1]), fmaxf(red[2], red[3]));
  __syncthreads();
  float s = 0.f;
  for (long long i = threadIdx.x * 8LL; i < V; i += 256 * 8) {
    s16x8 v8 = *(const s16x8*)(xr + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) s += __expf(bf16_to_f32(v8[j]) - m);
  }
  s = block_reduce_sum(s, red);
  if (threadIdx.x == 0) {
    float l = m + __logf(s);
    lse[row] = l;
    long long tr = t[row];
    loss[row] = (tr == ignore_index) ? 0.f : l - bf16_to_f32(xr[tr]);
  }
}

// dX[r][v] = dloss[r] * (exp(x - lse) - 1[v==target]); bf16 out, one pass.
// dloss[r] is the autograd grad of the per-row loss vector (0 for ignored
// rows, so those rows get exact zero grads). Grid (v-chunks, R).
extern "C" __global__ void __launch_bounds__(256)
ce_bwd_kernel(const short* __restrict__ x, const long long* __restrict__ t,
              const float* __restrict__ lse, const float* __restrict__ dlv,
              short* __restrict__ dx, long long V, long long ignore_index) {
  int row = blockIdx.y;
  float g = dlv[row], l = lse[row];
  long long tr = t[row];
  // an ignored row's loss is a constant 0 — its grad is exactly 0 even
  // though autograd hands every row d(sum)/drow = 1
  if (tr == ignore_index) g = 0.f;
  const short* xr = x + (long long)row * V;
  short* dr = dx + (long long)row * V;
  long long stride = (long long)gridDim.x * 256 * 8;
  for (long long i = ((long long)blockIdx.x * 256 + threadIdx.x) * 8;
       i < V; i += stride) {
    s16x8 v8 = *(const s16x8*)(xr + i);
    s16x8 o8;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float p = __expf(bf16_to_f32(v8[j]) - l);
      if (i + j == tr) p -= 1.f;
      o8[j] = f32_to_bf16(p * g);
    }
    *(s16x8*)(dr + i) = o8;
  }
}
