// Probe ds_read_b64_tr_b16 semantics: fill LDS with identifiable values,
// read with (a) uniform address, (b) per-lane addr = base + lane*8 bytes,
// dump per-lane results.
#include <hip/hip_runtime.h>
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4_t;

extern "C" __global__ void tr_probe(float* out_uniform, float* out_lane) {
  __shared__ unsigned short lds[1024];
  int l = threadIdx.x;
  // element value = its bf16 index (exactly representable < 256... use idx)
  for (int i = l; i < 1024; i += 64) {
    // encode index as bf16: value = i (integers < 256 exact; up to 1024 ok in bf16? 1024 exact)
    float f = (float)i;
    unsigned int u = __float_as_uint(f);
    lds[i] = (unsigned short)(u >> 16);
  }
  __syncthreads();
  bf16x4_t a, b;
  unsigned base = (unsigned)(__SIZE_TYPE__)(__attribute__((address_space(3))) unsigned short*)&lds[0];
  unsigned addr_u = base;            // uniform
  unsigned addr_l = base + l * 8;    // per-lane 8B
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)" : "=v"(a) : "v"(addr_u) : "memory");
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)" : "=v"(b) : "v"(addr_l) : "memory");
  for (int j = 0; j < 4; ++j) {
    out_uniform[l * 4 + j] = (float)a[j];
    out_lane[l * 4 + j] = (float)b[j];
  }
}
