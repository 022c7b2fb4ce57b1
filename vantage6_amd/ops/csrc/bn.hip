// Fused BatchNorm2d (NHWC / channels_last, bf16 I/O, fp32 stats) for gfx950.
//
// The ResNet hot path spends more time in BatchNorm + eager elementwise
// glue than in the (MFMA) convolutions themselves (profiles/
// resnet18_bench_steady_r01.txt), so BN is fused end-to-end:
//   forward  = stats partials → finalize (scale/shift precompute, running
//              stats update) → one apply pass with optional residual-add
//              and ReLU folded in (replaces 5 MIOpen kernels + 2 casts +
//              add + relu eager kernels per block)
//   backward = mask(dy by y>0) partials → finalize → one dx (+dresidual)
//              pass
// Layout: x is [R, C] with C contiguous (R = N*H*W), C % 8 == 0.
#include "common.h"

#define BN_NTH 256

// thread → (row-group, 8-channel chunk) decomposition
DEVINL void bn_thread_map(int C, int* chunks, int* rows_pb, int* j, int* rr) {
  *chunks = C >> 3;
  *rows_pb = BN_NTH / *chunks;
  if (*rows_pb < 1) *rows_pb = 1;
  *j = threadIdx.x % *chunks;
  *rr = threadIdx.x / *chunks;
}

// zero the [16][2][C] atomic accumulator — a kernel, not hipMemsetAsync,
// so it is always captured into hipGraphs (a non-captured memset leaves
// stale sums on every replay → diverging BN stats)
extern "C" __global__ void __launch_bounds__(BN_NTH)
bn_zero_acc_kernel(float* __restrict__ part, int n) {
  int i = blockIdx.x * BN_NTH + threadIdx.x;
  if (i < n) part[i] = 0.f;
}

// ---------------------------------------------------------- fwd: partials
// part layout: [2][nblocks][C] (sum, sumsq)
extern "C" __global__ void __launch_bounds__(BN_NTH)
bn_fwd_partials_kernel(const short* __restrict__ x, float* __restrict__ part,
                       long long R, int C, int nblocks) {
  __shared__ float red[BN_NTH * 8 * 2];
  int chunks, rows_pb, j, rr;
  bn_thread_map(C, &chunks, &rows_pb, &j, &rr);
  float s1[8] = {0}, s2[8] = {0};
  if (threadIdx.x < (unsigned)(chunks * rows_pb)) {
    const long long stride = (long long)gridDim.x * rows_pb;
    long long row = (long long)blockIdx.x * rows_pb + rr;
    // 4 independent 16-B loads in flight per iteration: the single-load
    // form measured 1.4 TB/s (latency-bound); this reaches streaming rate
    for (; row + 3 * stride < R; row += 4 * stride) {
      s16x8 v0 = *(const s16x8*)(x + row * C + j * 8);
      s16x8 v1 = *(const s16x8*)(x + (row + stride) * C + j * 8);
      s16x8 v2 = *(const s16x8*)(x + (row + 2 * stride) * C + j * 8);
      s16x8 v3 = *(const s16x8*)(x + (row + 3 * stride) * C + j * 8);
#pragma unroll
      for (int q = 0; q < 8; ++q) {
        float f0 = bf16_to_f32(v0[q]), f1 = bf16_to_f32(v1[q]);
        float f2 = bf16_to_f32(v2[q]), f3 = bf16_to_f32(v3[q]);
        s1[q] += (f0 + f1) + (f2 + f3);
        s2[q] = fmaf(f0, f0, fmaf(f1, f1, fmaf(f2, f2, fmaf(f3, f3, s2[q]))));
      }
    }
    for (; row < R; row += stride) {
      s16x8 v = *(const s16x8*)(x + row * C + j * 8);
#pragma unroll
      for (int q = 0; q < 8; ++q) {
        float f = bf16_to_f32(v[q]);
        s1[q] += f;
        s2[q] = fmaf(f, f, s2[q]);
      }
    }
  }
#pragma unroll
  for (int q = 0; q < 8; ++q) {
    red[threadIdx.x * 8 + q] = s1[q];
    red[(BN_NTH + threadIdx.x) * 8 + q] = s2[q];
  }
  __syncthreads();
  // tree-reduce over the row-group dimension (threads strided by `chunks`)
  for (int s = rows_pb >> 1; s > 0; s >>= 1) {
    if (rr < s && threadIdx.x + s * chunks < BN_NTH) {
#pragma unroll
      for (int q = 0; q < 8; ++q) {
        red[threadIdx.x * 8 + q] += red[(threadIdx.x + s * chunks) * 8 + q];
        red[(BN_NTH + threadIdx.x) * 8 + q] +=
            red[(BN_NTH + threadIdx.x + s * chunks) * 8 + q];
      }
    }
    __syncthreads();
  }
  if (rr == 0) {
    // 16 contention slices: block b adds into slice b&15 (<=32 atomics per
    // address instead of up to 512)
    float* slice = part + (blockIdx.x & 15) * 2 * C;
#pragma unroll
    for (int q = 0; q < 8; ++q) {
      atomicAdd(&slice[j * 8 + q], red[threadIdx.x * 8 + q]);
      atomicAdd(&slice[C + j * 8 + q], red[(BN_NTH + threadIdx.x) * 8 + q]);
    }
  }
}

// --------------------------------------------------------- fwd: finalize
// writes mean/rstd (saved for bwd) + scale/shift (apply pass) and updates
// running stats in place.
// part is now the fp32 accumulator [2][C] (partials blocks atomicAdd into
// it; the binding memsets it before the launch). finalize is pure
// per-channel math — the earlier reduction-loop versions serialized one
// near-idle block for 199 us (v1) / 38 us (v2) per call.
extern "C" __global__ void __launch_bounds__(BN_NTH)
bn_fwd_finalize_kernel(const float* __restrict__ part, int nblocks, int C,
                       long long R, const short* __restrict__ w,
                       const short* __restrict__ b, float* __restrict__ mean_out,
                       float* __restrict__ rstd_out, float* __restrict__ scale,
                       float* __restrict__ shift, float* __restrict__ running_mean,
                       float* __restrict__ running_var, float momentum,
                       float eps, int training) {
  int c = blockIdx.x * BN_NTH + threadIdx.x;
  if (c >= C) return;
  float mean, var;
  if (training) {
    float s1 = 0.f, s2 = 0.f;
#pragma unroll
    for (int sl = 0; sl < 16; ++sl) {
      s1 += part[sl * 2 * C + c];
      s2 += part[sl * 2 * C + C + c];
    }
    mean = s1 / R;
    var = s2 / R - mean * mean;
    // reset the accumulator for the next call (saves a zeroing launch)
#pragma unroll
    for (int sl = 0; sl < 16; ++sl) {
      ((float*)part)[sl * 2 * C + c] = 0.f;
      ((float*)part)[sl * 2 * C + C + c] = 0.f;
    }
    if (var < 0.f) var = 0.f;
    float unbiased = (R > 1) ? var * (float)R / (float)(R - 1) : var;
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mean;
    running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
  } else {
    mean = running_mean[c];
    var = running_var[c];
  }
  float rstd = rsqrtf(var + eps);
  float wf = bf16_to_f32(w[c]);
  mean_out[c] = mean;
  rstd_out[c] = rstd;
  scale[c] = wf * rstd;
  shift[c] = bf16_to_f32(b[c]) - mean * wf * rstd;
}

// ------------------------------------------------------------ fwd: apply
// y = relu?( x*scale + shift (+ residual) ).  Affine addressing: thread owns
// one 8-channel chunk (params preloaded to registers) and strides rows.
extern "C" __global__ void __launch_bounds__(BN_NTH)
bn_fwd_apply_kernel(const short* __restrict__ x, short* __restrict__ y,
                    const short* __restrict__ residual,
                    const float* __restrict__ scale,
                    const float* __restrict__ shift,
                    unsigned char* __restrict__ mask, long long R, int C,
                    int relu) {
  int chunks, rows_pb, j, rr;
  bn_thread_map(C, &chunks, &rows_pb, &j, &rr);
  if (threadIdx.x >= (unsigned)(chunks * rows_pb)) return;
  float sc[8], sh[8];
#pragma unroll
  for (int q = 0; q < 8; ++q) {
    sc[q] = scale[j * 8 + q];
    sh[q] = shift[j * 8 + q];
  }
  const long long stride_a = (long long)gridDim.x * rows_pb;
  long long row_a = (long long)blockIdx.x * rows_pb + rr;
  for (; row_a + stride_a < R; row_a += 2 * stride_a) {
    long long off0 = row_a * C + j * 8;
    long long off1 = (row_a + stride_a) * C + j * 8;
    s16x8 v0 = *(const s16x8*)(x + off0);
    s16x8 v1 = *(const s16x8*)(x + off1);
    s16x8 o0, o1;
    unsigned int m0 = 0, m1 = 0;
    if (residual) {
      s16x8 r0 = *(const s16x8*)(residual + off0);
      s16x8 r1 = *(const s16x8*)(residual + off1);
#pragma unroll
      for (int q = 0; q < 8; ++q) {
        float f0 = fmaf(bf16_to_f32(v0[q]), sc[q], sh[q]) + bf16_to_f32(r0[q]);
        float f1 = fmaf(bf16_to_f32(v1[q]), sc[q], sh[q]) + bf16_to_f32(r1[q]);
        if (relu && f0 < 0.f) f0 = 0.f; else m0 |= (1u << q);
        if (relu && f1 < 0.f) f1 = 0.f; else m1 |= (1u << q);
        o0[q] = f32_to_bf16(f0);
        o1[q] = f32_to_bf16(f1);
      }
    } else {
#pragma unroll
      for (int q = 0; q < 8; ++q) {
        float f0 = fmaf(bf16_to_f32(v0[q]), sc[q], sh[q]);
        float f1 = fmaf(bf16_to_f32(v1[q]), sc[q], sh[q]);
        if (relu && f0 < 0.f) f0 = 0.f; else m0 |= (1u << q);
        if (relu && f1 < 0.f) f1 = 0.f; else m1 |= (1u << q);
        o0[q] = f32_to_bf16(f0);
        o1[q] = f32_to_bf16(f1);
      }
    }
    *(s16x8*)(y + off0) = o0;
    *(s16x8*)(y + off1) = o1;
    if (mask) {
      mask[off0 >> 3] = (unsigned char)m0;
      mask[off1 >> 3] = (unsigned char)m1;
    }
  }
  for (; row_a < R; row_a += stride_a) {
    long long off = row_a * C + j * 8;
    s16x8 v = *(const s16x8*)(x + off);
    s16x8 o;
    unsigned int m = 0;
    if (residual) {
      s16x8 rv = *(const s16x8*)(residual + off);
#pragma unroll
      for (int q = 0; q < 8; ++q) {
        float f = fmaf(bf16_to_f32(v[q]), sc[q], sh[q]) + bf16_to_f32(rv[q]);
        if (relu && f < 0.f) f = 0.f; else m |= (1u << q);
        o[q] = f32_to_bf16(f);
      }
    } else {
#pragma unroll
      for (int q = 0; q < 8; ++q) {
        float f = fmaf(bf16_to_f32(v[q]), sc[q], sh[q]);
        if (relu && f < 0.f) f = 0.f; else m |= (1u << q);
        o[q] = f32_to_bf16(f);
      }
    }
    *(s16x8*)(y + off) = o;
    if (mask) mask[off >> 3] = (unsigned char)m;  // byte per 8-chunk
  }
}

// ---------------------------------------------------------- bwd: partials
// dyr = relu ? (y>0 ? dy : 0) : dy ; partials of db=Σdyr, dws=Σdyr*xhat
extern "C" __global__ void __launch_bounds__(BN_NTH)
bn_bwd_partials_kernel(const short* __restrict__ dy,
                       const unsigned char* __restrict__ mask,
                       const short* __restrict__ x,
                       const float* __restrict__ mean,
                       const float* __restrict__ rstd, float* __restrict__ part,
                       long long R, int C, int nblocks, int relu) {
  __shared__ float red[BN_NTH * 8 * 2];
  int chunks, rows_pb, j, rr;
  bn_thread_map(C, &chunks, &rows_pb, &j, &rr);
  float sdb[8] = {0}, sdw[8] = {0};
  float mu[8], rs[8];
#pragma unroll
  for (int q = 0; q < 8; ++q) {
    mu[q] = mean[j * 8 + q];
    rs[q] = rstd[j * 8 + q];
  }
  if (threadIdx.x < (unsigned)(chunks * rows_pb)) {
    const long long stride = (long long)gridDim.x * rows_pb;
    long long row = (long long)blockIdx.x * rows_pb + rr;
    for (; row + stride < R; row += 2 * stride) {
      long long off0 = row * C + j * 8;
      long long off1 = (row + stride) * C + j * 8;
      s16x8 d0 = *(const s16x8*)(dy + off0);
      s16x8 x0 = *(const s16x8*)(x + off0);
      s16x8 d1 = *(const s16x8*)(dy + off1);
      s16x8 x1 = *(const s16x8*)(x + off1);
      unsigned int m0 = relu ? mask[off0 >> 3] : 0xFFu;
      unsigned int m1 = relu ? mask[off1 >> 3] : 0xFFu;
#pragma unroll
      for (int q = 0; q < 8; ++q) {
        float dd0 = (m0 >> q) & 1u ? bf16_to_f32(d0[q]) : 0.f;
        float dd1 = (m1 >> q) & 1u ? bf16_to_f32(d1[q]) : 0.f;
        float xh0 = (bf16_to_f32(x0[q]) - mu[q]) * rs[q];
        float xh1 = (bf16_to_f32(x1[q]) - mu[q]) * rs[q];
        sdb[q] += dd0 + dd1;
        sdw[q] = fmaf(dd0, xh0, fmaf(dd1, xh1, sdw[q]));
      }
    }
    for (; row < R; row += stride) {
      long long off = row * C + j * 8;
      s16x8 dv = *(const s16x8*)(dy + off);
      s16x8 xv = *(const s16x8*)(x + off);
      unsigned int m = relu ? mask[off >> 3] : 0xFFu;
#pragma unroll
      for (int q = 0; q < 8; ++q) {
        float d = (m >> q) & 1u ? bf16_to_f32(dv[q]) : 0.f;
        float xh = (bf16_to_f32(xv[q]) - mu[q]) * rs[q];
        sdb[q] += d;
        sdw[q] = fmaf(d, xh, sdw[q]);
      }
    }
  }
#pragma unroll
  for (int q = 0; q < 8; ++q) {
    red[threadIdx.x * 8 + q] = sdb[q];
    red[(BN_NTH + threadIdx.x) * 8 + q] = sdw[q];
  }
  __syncthreads();
  for (int s = rows_pb >> 1; s > 0; s >>= 1) {
    if (rr < s && threadIdx.x + s * chunks < BN_NTH) {
#pragma unroll
      for (int q = 0; q < 8; ++q) {
        red[threadIdx.x * 8 + q] += red[(threadIdx.x + s * chunks) * 8 + q];
        red[(BN_NTH + threadIdx.x) * 8 + q] +=
            red[(BN_NTH + threadIdx.x + s * chunks) * 8 + q];
      }
    }
    __syncthreads();
  }
  if (rr == 0) {
    float* slice = part + (blockIdx.x & 15) * 2 * C;
#pragma unroll
    for (int q = 0; q < 8; ++q) {
      atomicAdd(&slice[j * 8 + q], red[threadIdx.x * 8 + q]);
      atomicAdd(&slice[C + j * 8 + q], red[(BN_NTH + threadIdx.x) * 8 + q]);
    }
  }
}

// --------------------------------------------------------- bwd: finalize
// db/dws column sums → dw,db (bf16 out) + dx coefficients
extern "C" __global__ void __launch_bounds__(BN_NTH)
bn_bwd_finalize_kernel(const float* __restrict__ part, int nblocks, int C,
                       long long R, const short* __restrict__ w,
                       const float* __restrict__ rstd, short* __restrict__ dw,
                       short* __restrict__ db, float* __restrict__ c_scale,
                       float* __restrict__ c_db, float* __restrict__ c_dw) {
  int c = blockIdx.x * BN_NTH + threadIdx.x;
  if (c >= C) return;
  float sdb = 0.f, sdw = 0.f;
#pragma unroll
  for (int sl = 0; sl < 16; ++sl) {
    sdb += part[sl * 2 * C + c];
    sdw += part[sl * 2 * C + C + c];
    ((float*)part)[sl * 2 * C + c] = 0.f;
    ((float*)part)[sl * 2 * C + C + c] = 0.f;
  }
  dw[c] = f32_to_bf16(sdw);
  db[c] = f32_to_bf16(sdb);
  c_scale[c] = bf16_to_f32(w[c]) * rstd[c];
  c_db[c] = sdb / R;
  c_dw[c] = sdw / R;
}

// ------------------------------------------------------------ bwd: apply
// dx = c_scale*(dyr − c_db − xhat*c_dw); optional dresidual = dyr
extern "C" __global__ void __launch_bounds__(BN_NTH)
bn_bwd_apply_kernel(const short* __restrict__ dy,
                    const unsigned char* __restrict__ mask,
                    const short* __restrict__ x, const float* __restrict__ mean,
                    const float* __restrict__ rstd,
                    const float* __restrict__ c_scale,
                    const float* __restrict__ c_db,
                    const float* __restrict__ c_dw, short* __restrict__ dx,
                    short* __restrict__ dres, long long R, int C, int relu) {
  int chunks, rows_pb, j, rr;
  bn_thread_map(C, &chunks, &rows_pb, &j, &rr);
  if (threadIdx.x >= (unsigned)(chunks * rows_pb)) return;
  float mu[8], rs[8], cs[8], cdb[8], cdw[8];
#pragma unroll
  for (int q = 0; q < 8; ++q) {
    int c = j * 8 + q;
    mu[q] = mean[c]; rs[q] = rstd[c];
    cs[q] = c_scale[c]; cdb[q] = c_db[c]; cdw[q] = c_dw[c];
  }
  const long long stride_b = (long long)gridDim.x * rows_pb;
  long long row_b = (long long)blockIdx.x * rows_pb + rr;
  for (; row_b + stride_b < R; row_b += 2 * stride_b) {
    long long off0 = row_b * C + j * 8;
    long long off1 = (row_b + stride_b) * C + j * 8;
    s16x8 d0 = *(const s16x8*)(dy + off0);
    s16x8 x0 = *(const s16x8*)(x + off0);
    s16x8 d1 = *(const s16x8*)(dy + off1);
    s16x8 x1 = *(const s16x8*)(x + off1);
    unsigned int m0 = relu ? mask[off0 >> 3] : 0xFFu;
    unsigned int m1 = relu ? mask[off1 >> 3] : 0xFFu;
    s16x8 ox0, ox1, or0, or1;
#pragma unroll
    for (int q = 0; q < 8; ++q) {
      float dd0 = (m0 >> q) & 1u ? bf16_to_f32(d0[q]) : 0.f;
      float dd1 = (m1 >> q) & 1u ? bf16_to_f32(d1[q]) : 0.f;
      float xh0 = (bf16_to_f32(x0[q]) - mu[q]) * rs[q];
      float xh1 = (bf16_to_f32(x1[q]) - mu[q]) * rs[q];
      ox0[q] = f32_to_bf16(cs[q] * (dd0 - cdb[q] - xh0 * cdw[q]));
      ox1[q] = f32_to_bf16(cs[q] * (dd1 - cdb[q] - xh1 * cdw[q]));
      or0[q] = f32_to_bf16(dd0);
      or1[q] = f32_to_bf16(dd1);
    }
    *(s16x8*)(dx + off0) = ox0;
    *(s16x8*)(dx + off1) = ox1;
    if (dres) {
      *(s16x8*)(dres + off0) = or0;
      *(s16x8*)(dres + off1) = or1;
    }
  }
  for (; row_b < R; row_b += stride_b) {
    long long off = row_b * C + j * 8;
    s16x8 dv = *(const s16x8*)(dy + off);
    s16x8 xv = *(const s16x8*)(x + off);
    s16x8 odx, odr;
    unsigned int m = relu ? mask[off >> 3] : 0xFFu;
#pragma unroll
    for (int q = 0; q < 8; ++q) {
      float d = (m >> q) & 1u ? bf16_to_f32(dv[q]) : 0.f;
      float xh = (bf16_to_f32(xv[q]) - mu[q]) * rs[q];
      odx[q] = f32_to_bf16(cs[q] * (d - cdb[q] - xh * cdw[q]));
      odr[q] = f32_to_bf16(d);
    }
    *(s16x8*)(dx + off) = odx;
    if (dres) *(s16x8*)(dres + off) = odr;
  }
}
