// Python bindings (torch extension) for the vantage6_amd CDNA4 kernels.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

#include "common.h"

// kernels defined in sibling TUs
extern "C" __global__ void fused_sgd_kernel(short*, float*, const short*, float*,
                                            long long, float, float, float, int);
struct SgdChunk { const short* grad; long long offset; long long n; };
extern "C" __global__ void fused_sgd_mt_kernel(const SgdChunk*, int, short*,
                                               float*, float*, float, float,
                                               float, int);
extern "C" __global__ void adamw_tick_kernel(int*);
extern "C" __global__ void fused_adamw_mt_kernel(const SgdChunk*, int, short*,
                                                 float*, float*, float*, float,
                                                 float, float, float, float,
                                                 float, float, const int*);
extern "C" __global__ void fused_adamw_kernel(short*, float*, const short*, float*,
                                              float*, long long, float, float, float,
                                              float, float, float, float, const int*);
extern "C" __global__ void secagg_quantize_mask_kernel(long long*, const float*,
                                                       long long, double,
                                                       unsigned long long, int, int);
extern "C" __global__ void dp_scale_noise_kernel(short*, long long, float, float,
                                                 unsigned long long,
                                                 const int*);
extern "C" __global__ void norm2_bf16_kernel(const short*, long long, float*);
extern "C" __global__ void swiglu_fwd_kernel(const short*, const short*, short*,
                                             long long);
extern "C" __global__ void rope_kernel(const short*, short*, const float*,
                                       const float*, long long, int, int, int);
extern "C" __global__ void swiglu_bwd_kernel(const short*, const short*,
                                             const short*, short*, short*,
                                             long long);
extern "C" __global__ void layernorm_fwd_kernel(const short*, const short*,
                                                const short*, short*, float*, float*,
                                                long long, int, float);
extern "C" __global__ void layernorm_bwd_kernel(const short*, const short*,
                                                const short*, const float*,
                                                const float*, short*, float*, float*,
                                                long long, int);
extern "C" __global__ void rmsnorm_fwd_kernel(const short*, const short*, short*,
                                              float*, long long, int, float);
extern "C" __global__ void rmsnorm_bwd_kernel(const short*, const short*,
                                              const short*, const float*, short*,
                                              float*, long long, int);
extern "C" __global__ void colsum_ws_kernel(const float*, short*, int, int);
extern "C" __global__ void layernorm_bwd_wave2_kernel(
    const short*, const short*, const short*, const float*, const float*,
    short*, float*, float*, long long, int);
extern "C" __global__ void layernorm_bwd_wave_kernel(const short*, const short*,
                                                     const short*, const float*,
                                                     const float*, short*,
                                                     float*, float*, long long,
                                                     int);
extern "C" __global__ void ce_fwd_kernel(const short*, const long long*,
                                         float*, float*, long long, long long);
extern "C" __global__ void ce_bwd_kernel(const short*, const long long*,
                                         const float*, const float*, short*,
                                         long long, long long);
extern "C" __global__ void rmsnorm_bwd_wave_kernel(const short*, const short*,
                                                   const short*, const float*,
                                                   short*, float*, long long,
                                                   int);
extern "C" __global__ void zero_f32_kernel(float*, long long);
extern "C" __global__ void emb_scatter_kernel(const long long*, const short*,
                                              float*, long long, int);
extern "C" __global__ void f32_to_bf16_kernel(const float*, short*, long long);
extern "C" __global__ void colsum_bf16_kernel(const short*, float*, long long,
                                              int, int);
extern "C" __global__ void gemm_nt_bf16_kernel(const short*, const short*, short*,
                                               int, int, int, int, int, int);
extern "C" __global__ void gemm_nt_bf16_v2_kernel(const short*, const short*,
                                                  short*, int, int, int);
extern "C" __global__ void gemm_nt_bf16_v3_kernel(const short*, const short*,
                                                  short*, int, int, int);
extern "C" __global__ void gemm_nt_bf16_v4_kernel(const short*, const short*,
                                                  short*, int, int, int);
extern "C" __global__ void gemm_nt_bf16_v5_kernel(const short*, const short*,
                                                  short*, int, int, int);
extern "C" __global__ void gemm_nt_bf16_v5b_kernel(const short*, const short*,
                                                   short*, int, int, int);
extern "C" __global__ void gemm_nt_bf16_v6_kernel(const short*, const short*,
                                                  short*, int, int, int);
extern "C" __global__ void gemm_nt_bf16_v7_kernel(const short*, const short*,
                                                  short*, int, int, int);
extern "C" __global__ void gemm_nt_bf16_v8_kernel(const short*, const short*,
                                                  short*, int, int, int);
extern "C" __global__ void gemm_nt_bf16_v9_kernel(const short*, const short*,
                                                  short*, float*, int, int, int,
                                                  int);
extern "C" __global__ void gemm_nt_bf16_v9sk_kernel(const short*, const short*,
                                                    short*, float*, int, int,
                                                    int, int);
extern "C" __global__ void gemm_nt_bf16_v10_kernel(const short*, const short*,
                                                   short*, float*, int, int,
                                                   int, int);
extern "C" __global__ void gemm_nt_bf16_v10sk_kernel(const short*, const short*,
                                                     short*, float*, int, int,
                                                     int, int);
extern "C" __global__ void gemm_wgrad_bf16_kernel(const short*, const short*,
                                                  short*, float*, int, int, int,
                                                  int);
extern "C" __global__ void gemm_wgrad2_bf16_kernel(const short*, const short*,
                                                   short*, float*, int, int,
                                                   int, int);
extern "C" __global__ void splitk_reduce_kernel(const float*, short*, long long,
                                                int);
extern "C" __global__ void gemm_dgrad_bf16_kernel(const short*, const short*,
                                                  short*, float*, int, int, int,
                                                  int);
extern "C" __global__ void transpose_bf16_kernel(const short*, short*, int, int);
extern "C" __global__ void conv_nhwc_fwd64_kernel(const short*, const short*,
                                                  short*, const short*, int,
                                                  int, int, int, int, int, int,
                                                  int, int, int, int);
extern "C" __global__ void conv_nhwc_fwd128_kernel(const short*, const short*,
                                                   short*, const short*, int,
                                                   int, int, int, int, int,
                                                   int, int, int, int, int);
extern "C" __global__ void conv_flip_weights_kernel(const short*, short*, int,
                                                    int, int, int);
extern "C" __global__ void maxpool_nhwc_fwd_kernel(const short*, short*,
                                                   unsigned char*, int, int,
                                                   int, int, int, int);
extern "C" __global__ void maxpool_nhwc_bwd_kernel(const short*,
                                                   const unsigned char*,
                                                   short*, int, int, int, int,
                                                   int, int);
extern "C" __global__ void attn_fwd_d64_kernel(const short*, const short*,
                                               const short*, short*, float*,
                                               int, int, int, int, float, int,
                                               long long, long long, long long,
                                               long long, long long,
                                               long long);
extern "C" __global__ void attn_fwd_d64_plin_kernel(
    const short*, const short*, const short*, short*, float*, int, int, int,
    int, float, int, long long, long long, long long, long long, long long,
    long long);
extern "C" __global__ void attn_bwd_d_kernel(const short*, const short*,
                                             float*, int, int, long long,
                                             long long, long long, long long);
extern "C" __global__ void attn_bwd_dq_d64_kernel(
    const short*, const short*, const short*, const short*, const float*,
    const float*, short*, int, int, int, int, float, int, long long,
    long long, long long, long long, long long, long long);
extern "C" __global__ void attn_bwd_dkdv_d64_kernel(
    const short*, const short*, const short*, const short*, const float*,
    const float*, short*, short*, float*, float*, int, int, int, int, float,
    int, long long, long long, long long, long long, long long, long long);
extern "C" __global__ void conv_nhwc_wgrad_kernel(
    const short*, const short*, float*, short*, const short*, int, int, int,
    int, int, int, int, int, int, int, int, int, int, unsigned long long,
    unsigned long long);
extern "C" __global__ void conv_nhwc_fwd64st_kernel(const short*, const short*,
                                                    short*, const short*,
                                                    float*, int, int, int, int,
                                                    int, int, int, int, int,
                                                    int, int);
extern "C" __global__ void conv_nhwc_fwd128st_kernel(const short*, const short*,
                                                     short*, const short*,
                                                     float*, int, int, int,
                                                     int, int, int, int, int,
                                                     int, int, int);
extern "C" __global__ void conv_nhwc_fwd64sk_kernel(const short*, const short*,
                                                    float*, const short*, int,
                                                    int, int, int, int, int,
                                                    int, int, int, int, int,
                                                    int);
extern "C" __global__ void conv_nhwc_fwd128sk_kernel(const short*, const short*,
                                                     float*, const short*, int,
                                                     int, int, int, int, int,
                                                     int, int, int, int, int,
                                                     int);
extern "C" __global__ void bn_fwd_partials_kernel(const short*, float*, long long,
                                                  int, int);
extern "C" __global__ void bn_zero_acc_kernel(float*, int);
extern "C" __global__ void bn_fwd_finalize_kernel(const float*, int, int, long long,
                                                  const short*, const short*, float*,
                                                  float*, float*, float*, float*,
                                                  float*, float, float, int);
extern "C" __global__ void bn_fwd_apply_kernel(const short*, short*, const short*,
                                               const float*, const float*,
                                               unsigned char*, long long, int, int);
extern "C" __global__ void bn_bwd_partials_kernel(const short*,
                                                  const unsigned char*,
                                                  const short*, const float*,
                                                  const float*, float*, long long,
                                                  int, int, int);
extern "C" __global__ void bn_bwd_finalize_kernel(const float*, int, int, long long,
                                                  const short*, const float*, short*,
                                                  short*, float*, float*, float*);
extern "C" __global__ void bn_bwd_apply_kernel(const short*,
                                               const unsigned char*,
                                               const short*, const float*,
                                               const float*, const float*,
                                               const float*, const float*, short*,
                                               short*, long long, int, int);

#define CHECK_HIP(t, d)                                              \
  TORCH_CHECK((t).is_cuda(), #t " must be a HIP tensor");            \
  TORCH_CHECK((t).is_contiguous(), #t " must be contiguous");        \
  TORCH_CHECK((t).scalar_type() == (d), #t " has wrong dtype")

static inline hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

static inline int ew_grid(long long n_items) {
  long long blocks = (n_items + 255) / 256;
  return (int)std::min<long long>(blocks, 2048);
}

// ----------------------------------------------------------------- optim
void fused_sgd(torch::Tensor param, torch::Tensor master, torch::Tensor grad,
               torch::Tensor mom, double lr, double momentum,
               double weight_decay, bool nesterov) {
  CHECK_HIP(param, torch::kBFloat16);
  CHECK_HIP(master, torch::kFloat32);
  CHECK_HIP(grad, torch::kBFloat16);
  CHECK_HIP(mom, torch::kFloat32);
  long long n = param.numel();
  hipLaunchKernelGGL(fused_sgd_kernel, dim3(ew_grid((n + 7) / 8)), dim3(256), 0,
                     cur_stream(), (short*)param.data_ptr(),
                     master.data_ptr<float>(), (const short*)grad.data_ptr(),
                     mom.data_ptr<float>(), n, (float)lr, (float)momentum,
                     (float)weight_decay, nesterov ? 1 : 0);
}

void fused_adamw(torch::Tensor param, torch::Tensor master, torch::Tensor grad,
                 torch::Tensor exp_avg, torch::Tensor exp_avg_sq, double lr,
                 double beta1, double beta2, double eps, double weight_decay,
                 int64_t step, c10::optional<torch::Tensor> step_dev) {
  int* sd = nullptr;
  if (step_dev.has_value()) {
    TORCH_CHECK(step_dev->scalar_type() == torch::kInt32, "step_dev must be int32");
    sd = step_dev->data_ptr<int>();
    hipLaunchKernelGGL(adamw_tick_kernel, dim3(1), dim3(1), 0, cur_stream(), sd);
  }
  CHECK_HIP(param, torch::kBFloat16);
  CHECK_HIP(master, torch::kFloat32);
  CHECK_HIP(grad, torch::kBFloat16);
  long long n = param.numel();
  float bc1 = 1.0f - powf((float)beta1, (float)step);
  float bc2 = 1.0f - powf((float)beta2, (float)step);
  hipLaunchKernelGGL(fused_adamw_kernel, dim3(ew_grid((n + 7) / 8)), dim3(256), 0,
                     cur_stream(), (short*)param.data_ptr(),
                     master.data_ptr<float>(), (const short*)grad.data_ptr(),
                     exp_avg.data_ptr<float>(), exp_avg_sq.data_ptr<float>(), n,
                     (float)lr, (float)beta1, (float)beta2, (float)eps,
                     (float)weight_decay, bc1, bc2, sd);
}

// table: int64 [n_chunks, 3] = {grad ptr, flat offset, numel} (device)
void fused_sgd_mt(torch::Tensor table, torch::Tensor param, torch::Tensor master,
                  torch::Tensor mom, double lr, double momentum,
                  double weight_decay, bool nesterov) {
  CHECK_HIP(table, torch::kInt64);
  CHECK_HIP(param, torch::kBFloat16);
  int n_chunks = (int)table.size(0);
  int grid = std::min(n_chunks, 2048);
  hipLaunchKernelGGL(fused_sgd_mt_kernel, dim3(grid), dim3(256), 0,
                     cur_stream(), (const SgdChunk*)table.data_ptr(), n_chunks,
                     (short*)param.data_ptr(), master.data_ptr<float>(),
                     mom.data_ptr<float>(), (float)lr, (float)momentum,
                     (float)weight_decay, nesterov ? 1 : 0);
}

void fused_adamw_mt(torch::Tensor table, torch::Tensor param,
                    torch::Tensor master, torch::Tensor exp_avg,
                    torch::Tensor exp_avg_sq, double lr, double beta1,
                    double beta2, double eps, double weight_decay,
                    int64_t step, c10::optional<torch::Tensor> step_dev) {
  int* sd = nullptr;
  if (step_dev.has_value()) {
    TORCH_CHECK(step_dev->scalar_type() == torch::kInt32, "step_dev must be int32");
    sd = step_dev->data_ptr<int>();
    hipLaunchKernelGGL(adamw_tick_kernel, dim3(1), dim3(1), 0, cur_stream(), sd);
  }
  CHECK_HIP(table, torch::kInt64);
  int n_chunks = (int)table.size(0);
  int grid = std::min(n_chunks, 2048);
  float bc1 = 1.0f - powf((float)beta1, (float)step);
  float bc2 = 1.0f - powf((float)beta2, (float)step);
  hipLaunchKernelGGL(fused_adamw_mt_kernel, dim3(grid), dim3(256), 0,
                     cur_stream(), (const SgdChunk*)table.data_ptr(), n_chunks,
                     (short*)param.data_ptr(), master.data_ptr<float>(),
                     exp_avg.data_ptr<float>(), exp_avg_sq.data_ptr<float>(),
                     (float)lr, (float)beta1, (float)beta2, (float)eps,
                     (float)weight_decay, bc1, bc2, sd);
}

torch::Tensor colsum_bf16(torch::Tensor dy) {
  CHECK_HIP(dy, torch::kBFloat16);
  TORCH_CHECK(dy.dim() == 2 && dy.is_contiguous(), "colsum_bf16: 2-D contig");
  long long M = dy.size(0);
  int N = (int)dy.size(1);
  TORCH_CHECK(N % 8 == 0, "colsum_bf16: N must be a multiple of 8");
  auto ws = torch::empty({16, (long long)N}, dy.options().dtype(torch::kFloat32));
  auto db = torch::empty({(long long)N}, dy.options());
  int zgrid = (int)((16LL * N + 2047) / 2048);
  hipLaunchKernelGGL(zero_f32_kernel, dim3(zgrid), dim3(256), 0, cur_stream(),
                     ws.data_ptr<float>(), 16LL * N);
  int stripes = (N + 2047) / 2048;
  // fill the chip (~256 workgroups) without piling atomics onto the
  // 16-slice workspace: more stripes -> wider row-chunks
  int target_gy = 256 / stripes > 0 ? 256 / stripes : 1;
  int rchunk = (int)((M + target_gy - 1) / target_gy);
  rchunk = ((rchunk + 15) / 16) * 16;
  if (rchunk < 16) rchunk = 16;
  int gy = (int)((M + rchunk - 1) / rchunk);
  hipLaunchKernelGGL(colsum_bf16_kernel, dim3(stripes, gy), dim3(256), 0,
                     cur_stream(), (const short*)dy.data_ptr(),
                     ws.data_ptr<float>(), M, N, rchunk);
  int cgrid = (N + 255) / 256;
  hipLaunchKernelGGL(colsum_ws_kernel, dim3(cgrid), dim3(256), 0, cur_stream(),
                     ws.data_ptr<float>(), (short*)db.data_ptr(), 16, N);
  return db;
}

torch::Tensor emb_grad_bf16(torch::Tensor ids, torch::Tensor dy,
                            int64_t vocab) {
  TORCH_CHECK(ids.scalar_type() == torch::kInt64 && ids.is_cuda() &&
              ids.is_contiguous(), "emb_grad_bf16: ids int64 contig cuda");
  CHECK_HIP(dy, torch::kBFloat16);
  TORCH_CHECK(dy.dim() == 2 && dy.is_contiguous(), "emb_grad_bf16: dy 2-D");
  long long n_idx = ids.numel();
  int H = (int)dy.size(1);
  TORCH_CHECK(dy.size(0) == n_idx, "ids/dy row mismatch");
  auto ws = torch::empty({vocab, (long long)H},
                         dy.options().dtype(torch::kFloat32));
  auto dw = torch::empty({vocab, (long long)H}, dy.options());
  long long n = (long long)vocab * H;
  hipLaunchKernelGGL(zero_f32_kernel, dim3(ew_grid(n)), dim3(256), 0,
                     cur_stream(), ws.data_ptr<float>(), n);
  int grid = (int)std::min<long long>(n_idx, 8192);
  hipLaunchKernelGGL(emb_scatter_kernel, dim3(grid), dim3(256), 0,
                     cur_stream(), (const long long*)ids.data_ptr<int64_t>(),
                     (const short*)dy.data_ptr(), ws.data_ptr<float>(), n_idx,
                     H);
  hipLaunchKernelGGL(f32_to_bf16_kernel, dim3(ew_grid(n)), dim3(256), 0,
                     cur_stream(), ws.data_ptr<float>(), (short*)dw.data_ptr(),
                     n);
  return dw;
}

// ---------------------------------------------------------------- secagg
void secagg_quantize_mask(torch::Tensor q, torch::Tensor master, double weight,
                          double scale, int64_t seed, int64_t rank,
                          int64_t world_size) {
  CHECK_HIP(q, torch::kInt64);
  CHECK_HIP(master, torch::kFloat32);
  long long n = master.numel();
  hipLaunchKernelGGL(secagg_quantize_mask_kernel, dim3(ew_grid(n)), dim3(256), 0,
                     cur_stream(), (long long*)q.data_ptr(),
                     master.data_ptr<float>(), n, weight * scale,
                     (unsigned long long)seed, (int)rank, (int)world_size);
}

// -------------------------------------------------------------------- dp
void dp_scale_noise(torch::Tensor grad, double scale, double noise_std,
                    int64_t seed, c10::optional<torch::Tensor> step_dev) {
  CHECK_HIP(grad, torch::kBFloat16);
  long long n = grad.numel();
  const int* sd = nullptr;
  if (step_dev.has_value()) {
    TORCH_CHECK(step_dev->scalar_type() == torch::kInt32, "step_dev int32");
    sd = step_dev->data_ptr<int>();
    // pre-increment (captured) so each graph replay draws fresh noise
    hipLaunchKernelGGL(adamw_tick_kernel, dim3(1), dim3(1), 0, cur_stream(),
                       const_cast<int*>(sd));
  }
  hipLaunchKernelGGL(dp_scale_noise_kernel, dim3(ew_grid((n + 1) / 2)), dim3(256),
                     0, cur_stream(), (short*)grad.data_ptr(), n, (float)scale,
                     (float)noise_std, (unsigned long long)seed, sd);
}

torch::Tensor grad_norm2_partials(torch::Tensor grad) {
  CHECK_HIP(grad, torch::kBFloat16);
  long long n = grad.numel();
  int blocks = ew_grid((n + 7) / 8);
  auto out = torch::empty({blocks}, grad.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(norm2_bf16_kernel, dim3(blocks), dim3(256), 0, cur_stream(),
                     (const short*)grad.data_ptr(), n, out.data_ptr<float>());
  return out;
}

torch::Tensor swiglu_fwd(torch::Tensor g, torch::Tensor u) {
  CHECK_HIP(g, torch::kBFloat16);
  auto out = torch::empty_like(g);
  long long n = g.numel();
  hipLaunchKernelGGL(swiglu_fwd_kernel, dim3(ew_grid((n + 7) / 8)), dim3(256),
                     0, cur_stream(), (const short*)g.data_ptr(),
                     (const short*)u.data_ptr(), (short*)out.data_ptr(), n);
  return out;
}

std::vector<torch::Tensor> swiglu_bwd(torch::Tensor dout, torch::Tensor g,
                                      torch::Tensor u) {
  CHECK_HIP(dout, torch::kBFloat16);
  auto dg = torch::empty_like(g);
  auto du = torch::empty_like(u);
  long long n = g.numel();
  hipLaunchKernelGGL(swiglu_bwd_kernel, dim3(ew_grid((n + 7) / 8)), dim3(256),
                     0, cur_stream(), (const short*)dout.data_ptr(),
                     (const short*)g.data_ptr(), (const short*)u.data_ptr(),
                     (short*)dg.data_ptr(), (short*)du.data_ptr(), n);
  return {dg, du};
}

torch::Tensor rope_apply(torch::Tensor x, torch::Tensor coss, torch::Tensor sins,
                         int64_t dir) {
  CHECK_HIP(x, torch::kBFloat16);
  int D = (int)x.size(-1);
  int S = (int)x.size(-2);
  long long rows = x.numel() / D;
  auto out = torch::empty_like(x);
  long long total = rows * (D / 2);
  hipLaunchKernelGGL(rope_kernel, dim3(ew_grid(total)), dim3(256), 0,
                     cur_stream(), (const short*)x.data_ptr(),
                     (short*)out.data_ptr(), coss.data_ptr<float>(),
                     sins.data_ptr<float>(), rows, S, D, (int)dir);
  return out;
}

// ----------------------------------------------------------------- norms
static inline std::pair<long long, int> rows_cols(const torch::Tensor& x) {
  int H = (int)x.size(-1);
  long long rows = x.numel() / H;
  return {rows, H};
}

std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor b, double eps) {
  CHECK_HIP(x, torch::kBFloat16);
  auto [rows, H] = rows_cols(x);
  TORCH_CHECK(H % 2 == 0 && H <= 16384, "H must be even and <= 16384");
  auto y = torch::empty_like(x);
  auto mean = torch::empty({rows}, x.options().dtype(torch::kFloat32));
  auto rstd = torch::empty({rows}, x.options().dtype(torch::kFloat32));
  int grid = (int)std::min<long long>(rows, 2048);
  hipLaunchKernelGGL(layernorm_fwd_kernel, dim3(grid), dim3(256), 0, cur_stream(),
                     (const short*)x.data_ptr(), (const short*)w.data_ptr(),
                     (const short*)b.data_ptr(), (short*)y.data_ptr(),
                     mean.data_ptr<float>(), rstd.data_ptr<float>(), rows, H,
                     (float)eps);
  return {y, mean, rstd};
}

std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor w, torch::Tensor mean,
                                         torch::Tensor rstd) {
  CHECK_HIP(dy, torch::kBFloat16);
  auto [rows, H] = rows_cols(x);
  auto dx = torch::empty_like(x);
  int grid = (int)std::min<long long>(rows, 1024);
  auto dw_ws = torch::empty({16, H}, x.options().dtype(torch::kFloat32));
  auto db_ws = torch::empty({16, H}, x.options().dtype(torch::kFloat32));
  int zgrid = (16 * H + 255) / 256;
  hipLaunchKernelGGL(zero_f32_kernel, dim3(zgrid), dim3(256), 0, cur_stream(),
                     dw_ws.data_ptr<float>(), 16LL * H);
  hipLaunchKernelGGL(zero_f32_kernel, dim3(zgrid), dim3(256), 0, cur_stream(),
                     db_ws.data_ptr<float>(), 16LL * H);
  // wave-parallel rows + block LDS combine before the global atomics:
  // 68 vs 93 us at 4096x768 (the first variant without the combine lost
  // 2.4x to its per-wave atomic tail)
  static bool use_wave = [] {
    const char* v = getenv("V6AMD_LN_WAVE");
    return v ? atoi(v) != 0 : true;
  }();
  if (use_wave && H <= 2048) {  // row-per-wave, barrier-free reductions
    int wgrid = (int)std::min<long long>((rows + 3) / 4, 2048);
    auto kw = H <= 1024 ? layernorm_bwd_wave2_kernel
                        : layernorm_bwd_wave_kernel;
    hipLaunchKernelGGL(kw, dim3(wgrid), dim3(256), 0,
                       cur_stream(), (const short*)dy.data_ptr(),
                       (const short*)x.data_ptr(), (const short*)w.data_ptr(),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       (short*)dx.data_ptr(), dw_ws.data_ptr<float>(),
                       db_ws.data_ptr<float>(), rows, H);
  } else {
    hipLaunchKernelGGL(layernorm_bwd_kernel, dim3(grid), dim3(256), 0,
                       cur_stream(), (const short*)dy.data_ptr(),
                       (const short*)x.data_ptr(), (const short*)w.data_ptr(),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       (short*)dx.data_ptr(), dw_ws.data_ptr<float>(),
                       db_ws.data_ptr<float>(), rows, H);
  }
  auto dw = torch::empty({H}, x.options());
  auto db = torch::empty({H}, x.options());
  int cgrid = (H + 255) / 256;
  hipLaunchKernelGGL(colsum_ws_kernel, dim3(cgrid), dim3(256), 0, cur_stream(),
                     dw_ws.data_ptr<float>(), (short*)dw.data_ptr(), 16, H);
  hipLaunchKernelGGL(colsum_ws_kernel, dim3(cgrid), dim3(256), 0, cur_stream(),
                     db_ws.data_ptr<float>(), (short*)db.data_ptr(), 16, H);
  return {dx, dw, db};
}

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w,
                                       double eps) {
  CHECK_HIP(x, torch::kBFloat16);
  auto [rows, H] = rows_cols(x);
  TORCH_CHECK(H % 2 == 0 && H <= 16384, "H must be even and <= 16384");
  auto y = torch::empty_like(x);
  auto rstd = torch::empty({rows}, x.options().dtype(torch::kFloat32));
  int grid = (int)std::min<long long>(rows, 2048);
  hipLaunchKernelGGL(rmsnorm_fwd_kernel, dim3(grid), dim3(256), 0, cur_stream(),
                     (const short*)x.data_ptr(), (const short*)w.data_ptr(),
                     (short*)y.data_ptr(), rstd.data_ptr<float>(), rows, H,
                     (float)eps);
  return {y, rstd};
}

static bool use_wave_rms() {
  static bool v = [] {
    const char* e = getenv("V6AMD_LN_WAVE");
    return e ? atoi(e) != 0 : true;
  }();
  return v;
}

// fused LM-head cross-entropy: fwd returns {loss[R] fp32, lse[R] fp32}.
std::vector<torch::Tensor> ce_fwd(torch::Tensor x, torch::Tensor t,
                                  long long ignore_index) {
  CHECK_HIP(x, torch::kBFloat16);
  CHECK_HIP(t, torch::kLong);
  TORCH_CHECK(x.dim() == 2, "ce_fwd: x must be [R, V]");
  long long R = x.size(0), V = x.size(1);
  TORCH_CHECK(V % 8 == 0, "ce_fwd: V must be a multiple of 8");
  TORCH_CHECK(t.numel() == R, "ce_fwd: targets/rows mismatch");
  auto lse = torch::empty({R}, x.options().dtype(torch::kFloat32));
  auto loss = torch::empty({R}, x.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(ce_fwd_kernel, dim3((unsigned)R), dim3(256), 0,
                     cur_stream(), (const short*)x.data_ptr(),
                     (const long long*)t.data_ptr(), lse.data_ptr<float>(),
                     loss.data_ptr<float>(), V, ignore_index);
  return {loss, lse};
}

torch::Tensor ce_bwd(torch::Tensor x, torch::Tensor t, torch::Tensor lse,
                     torch::Tensor dlv, long long ignore_index) {
  CHECK_HIP(x, torch::kBFloat16);
  CHECK_HIP(dlv, torch::kFloat32);
  long long R = x.size(0), V = x.size(1);
  auto dx = torch::empty_like(x);
  int gx = (int)std::min<long long>((V / 8 + 255) / 256, 4096);
  hipLaunchKernelGGL(ce_bwd_kernel, dim3(gx, (unsigned)R), dim3(256), 0,
                     cur_stream(), (const short*)x.data_ptr(),
                     (const long long*)t.data_ptr(), lse.data_ptr<float>(),
                     dlv.data_ptr<float>(), (short*)dx.data_ptr(), V,
                     ignore_index);
  return dx;
}

std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor w, torch::Tensor rstd) {
  CHECK_HIP(dy, torch::kBFloat16);
  auto [rows, H] = rows_cols(x);
  auto dx = torch::empty_like(x);
  int grid = (int)std::min<long long>(rows, 1024);
  auto dw_ws = torch::empty({16, H}, x.options().dtype(torch::kFloat32));
  int zgrid = (16 * H + 255) / 256;
  hipLaunchKernelGGL(zero_f32_kernel, dim3(zgrid), dim3(256), 0, cur_stream(),
                     dw_ws.data_ptr<float>(), 16LL * H);
  if (use_wave_rms() && H <= 2048) {
    int wgrid = (int)std::min<long long>((rows + 3) / 4, 2048);
    hipLaunchKernelGGL(rmsnorm_bwd_wave_kernel, dim3(wgrid), dim3(256), 0,
                       cur_stream(), (const short*)dy.data_ptr(),
                       (const short*)x.data_ptr(), (const short*)w.data_ptr(),
                       rstd.data_ptr<float>(), (short*)dx.data_ptr(),
                       dw_ws.data_ptr<float>(), rows, H);
  } else {
    hipLaunchKernelGGL(rmsnorm_bwd_kernel, dim3(grid), dim3(256), 0,
                       cur_stream(), (const short*)dy.data_ptr(),
                       (const short*)x.data_ptr(), (const short*)w.data_ptr(),
                       rstd.data_ptr<float>(), (short*)dx.data_ptr(),
                       dw_ws.data_ptr<float>(), rows, H);
  }
  auto dw = torch::empty({H}, x.options());
  int cgrid = (H + 255) / 256;
  hipLaunchKernelGGL(colsum_ws_kernel, dim3(cgrid), dim3(256), 0, cur_stream(),
                     dw_ws.data_ptr<float>(), (short*)dw.data_ptr(), 16, H);
  return {dx, dw};
}

// -------------------------------------------------------------------- bn
static inline int bn_rows_pb(int C) {
  int chunks = C / 8;
  int r = 256 / chunks;
  return r < 1 ? 1 : r;
}

std::vector<torch::Tensor> bn_fwd(torch::Tensor x, torch::Tensor w,
                                  torch::Tensor b, torch::Tensor running_mean,
                                  torch::Tensor running_var, bool training,
                                  double momentum, double eps, bool relu,
                                  c10::optional<torch::Tensor> residual,
                                  c10::optional<torch::Tensor> acc,
                                  bool stats_precomputed) {
  CHECK_HIP(x, torch::kBFloat16);
  int C = (int)x.size(-1);
  TORCH_CHECK(C % 8 == 0 && C <= 2048, "BN channels must be %8==0 and <=2048");
  long long R = x.numel() / C;
  auto fopts = x.options().dtype(torch::kFloat32);
  auto mean = torch::empty({C}, fopts);
  auto rstd = torch::empty({C}, fopts);
  auto scale = torch::empty({C}, fopts);
  auto shift = torch::empty({C}, fopts);
  int rows_pb = bn_rows_pb(C);
  // grid sized so each thread runs ~8 row-iterations: the unrolled loads
  // need that depth to hide latency (single-iteration grids measured
  // 1.4 TB/s on the partials pass)
  long long cover = (R + rows_pb - 1) / rows_pb;
  int nblocks = (int)std::min<long long>(
      cover, std::max<long long>(128, cover / 16));
  nblocks = std::min(nblocks, 384);
  torch::Tensor part;
  if (acc.has_value()) {
    part = acc.value();  // persistent, zeroed at creation; finalize re-zeroes
  } else {
    part = torch::empty({16, 2, C}, fopts);
    if (training)
      hipLaunchKernelGGL(bn_zero_acc_kernel, dim3((32 * C + 255) / 256),
                         dim3(256), 0, cur_stream(), part.data_ptr<float>(),
                         32 * C);
  }
  if (training && !stats_precomputed) {
    hipLaunchKernelGGL(bn_fwd_partials_kernel, dim3(nblocks), dim3(256), 0,
                       cur_stream(), (const short*)x.data_ptr(),
                       part.data_ptr<float>(), R, C, nblocks);
  } else if (stats_precomputed) {
    // the producing conv's epilogue already accumulated sum/sumsq
    TORCH_CHECK(acc.has_value(),
                "bn_fwd: stats_precomputed requires the shared accumulator");
  }
  hipLaunchKernelGGL(bn_fwd_finalize_kernel, dim3((C + 255) / 256), dim3(256), 0,
                     cur_stream(), part.data_ptr<float>(), nblocks, C, R,
                     (const short*)w.data_ptr(), (const short*)b.data_ptr(),
                     mean.data_ptr<float>(), rstd.data_ptr<float>(),
                     scale.data_ptr<float>(), shift.data_ptr<float>(),
                     running_mean.data_ptr<float>(),
                     running_var.data_ptr<float>(), (float)momentum, (float)eps,
                     training ? 1 : 0);
  auto y = torch::empty_like(x);
  const short* res_ptr = nullptr;
  if (residual.has_value()) {
    CHECK_HIP(residual.value(), torch::kBFloat16);
    res_ptr = (const short*)residual->data_ptr();
  }
  torch::Tensor msk;
  unsigned char* mask_ptr = nullptr;
  if (relu) {  // relu bitmask: backward never re-reads y
    msk = torch::empty({R * C / 8}, x.options().dtype(torch::kUInt8));
    mask_ptr = (unsigned char*)msk.data_ptr();
  } else {
    msk = torch::empty({0}, x.options().dtype(torch::kUInt8));
  }
  int agrid = (int)std::min<long long>(
      cover, std::max<long long>(256, cover / 4));
  hipLaunchKernelGGL(bn_fwd_apply_kernel, dim3(agrid), dim3(256), 0,
                     cur_stream(), (const short*)x.data_ptr(),
                     (short*)y.data_ptr(), res_ptr, scale.data_ptr<float>(),
                     shift.data_ptr<float>(), mask_ptr, R, C, relu ? 1 : 0);
  return {y, mean, rstd, msk};
}

std::vector<torch::Tensor> bn_bwd(torch::Tensor dy, torch::Tensor mask,
                                  torch::Tensor x, torch::Tensor w,
                                  torch::Tensor mean, torch::Tensor rstd,
                                  bool relu, bool want_dres,
                                  c10::optional<torch::Tensor> acc) {
  CHECK_HIP(dy, torch::kBFloat16);
  int C = (int)x.size(-1);
  long long R = x.numel() / C;
  auto fopts = x.options().dtype(torch::kFloat32);
  int rows_pb = bn_rows_pb(C);
  long long cover = (R + rows_pb - 1) / rows_pb;
  int nblocks = (int)std::min<long long>(
      cover, std::max<long long>(128, cover / 16));
  nblocks = std::min(nblocks, 384);
  torch::Tensor part;
  if (acc.has_value()) {
    part = acc.value();
  } else {
    part = torch::empty({16, 2, C}, fopts);
    hipLaunchKernelGGL(bn_zero_acc_kernel, dim3((32 * C + 255) / 256), dim3(256),
                       0, cur_stream(), part.data_ptr<float>(), 32 * C);
  }
  hipLaunchKernelGGL(bn_bwd_partials_kernel, dim3(nblocks), dim3(256), 0,
                     cur_stream(), (const short*)dy.data_ptr(),
                     (const unsigned char*)mask.data_ptr(),
                     (const short*)x.data_ptr(),
                     mean.data_ptr<float>(), rstd.data_ptr<float>(),
                     part.data_ptr<float>(), R, C, nblocks, relu ? 1 : 0);
  auto dw = torch::empty({C}, x.options());
  auto db = torch::empty({C}, x.options());
  auto c_scale = torch::empty({C}, fopts);
  auto c_db = torch::empty({C}, fopts);
  auto c_dw = torch::empty({C}, fopts);
  hipLaunchKernelGGL(bn_bwd_finalize_kernel, dim3((C + 255) / 256), dim3(256), 0,
                     cur_stream(), part.data_ptr<float>(), nblocks, C, R,
                     (const short*)w.data_ptr(), rstd.data_ptr<float>(),
                     (short*)dw.data_ptr(), (short*)db.data_ptr(),
                     c_scale.data_ptr<float>(), c_db.data_ptr<float>(),
                     c_dw.data_ptr<float>());
  auto dx = torch::empty_like(x);
  torch::Tensor dres;
  short* dres_ptr = nullptr;
  if (want_dres) {
    dres = torch::empty_like(x);
    dres_ptr = (short*)dres.data_ptr();
  }
  int agrid = (int)std::min<long long>(
      cover, std::max<long long>(256, cover / 4));
  hipLaunchKernelGGL(bn_bwd_apply_kernel, dim3(agrid), dim3(256), 0,
                     cur_stream(), (const short*)dy.data_ptr(),
                     (const unsigned char*)mask.data_ptr(),
                     (const short*)x.data_ptr(),
                     mean.data_ptr<float>(), rstd.data_ptr<float>(),
                     c_scale.data_ptr<float>(), c_db.data_ptr<float>(),
                     c_dw.data_ptr<float>(), (short*)dx.data_ptr(), dres_ptr, R,
                     C, relu ? 1 : 0);
  if (want_dres) return {dx, dw, db, dres};
  return {dx, dw, db};
}

// ------------------------------------------------------------------ gemm
torch::Tensor transpose_bf16(torch::Tensor a) {
  CHECK_HIP(a, torch::kBFloat16);
  TORCH_CHECK(a.dim() == 2, "transpose_bf16 expects 2-D");
  int M = (int)a.size(0), N = (int)a.size(1);
  auto out = torch::empty({N, M}, a.options());
  dim3 grid((N + 63) / 64, (M + 63) / 64);
  hipLaunchKernelGGL(transpose_bf16_kernel, grid, dim3(256), 0, cur_stream(),
                     (const short*)a.data_ptr(), (short*)out.data_ptr(), M, N);
  return out;
}

torch::Tensor gemm_wgrad(torch::Tensor a, torch::Tensor b) {
  // C[Na,Nb] = A^T · B for A[M,Na], B[M,Nb] (reduction-strided operands,
  // tr_b16 fragment reads — no transpose pre-passes)
  CHECK_HIP(a, torch::kBFloat16);
  CHECK_HIP(b, torch::kBFloat16);
  int M = (int)a.size(0), Na = (int)a.size(1), Nb = (int)b.size(1);
  TORCH_CHECK((int)b.size(0) == M, "reduction dims mismatch");
  TORCH_CHECK(M % 64 == 0 && Na % 128 == 0 && Nb % 128 == 0,
              "gemm_wgrad needs M%64==0, Na/Nb%128==0");
  auto c = torch::empty({Na, Nb}, a.options());
  // 256² tile (v2): measured SLOWER than the 128² tile on every training
  // shape (tools/wgrad2_ab.py — e.g. 59 vs 26 µs at 2048²/M=1024: the
  // tr_b16 gather latency doubles with the tile while hiding no more of
  // it, and the grid shrinks 4×). Kept behind V6AMD_WGRAD2=1.
  static int use_v2 = [] {
    const char* v = getenv("V6AMD_WGRAD2");
    return v ? atoi(v) : 0;
  }();
  bool v2 = use_v2 && Na % 256 == 0 && Nb % 256 == 0;
  int tiles = v2 ? (Na / 256) * (Nb / 256) : (Na / 128) * (Nb / 128);
  auto kfn = v2 ? gemm_wgrad2_bf16_kernel : gemm_wgrad_bf16_kernel;
  int nth = v2 ? 512 : 256;
  // split the long reduction when the tile grid starves the 256 CUs
  int nsplit = 1;
  while (tiles * nsplit < 384 && nsplit < 8 && (M / 64) / (nsplit * 2) >= 2)
    nsplit *= 2;
  if (nsplit == 1) {
    hipLaunchKernelGGL(kfn, dim3(tiles), dim3(nth), 0,
                       cur_stream(), (const short*)a.data_ptr(),
                       (const short*)b.data_ptr(), (short*)c.data_ptr(),
                       nullptr, M, Na, Nb, 1);
    return c;
  }
  auto ws = torch::empty({nsplit, Na, Nb},
                         a.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(kfn, dim3(tiles * nsplit), dim3(nth), 0,
                     cur_stream(), (const short*)a.data_ptr(),
                     (const short*)b.data_ptr(), nullptr,
                     ws.data_ptr<float>(), M, Na, Nb, nsplit);
  long long n = (long long)Na * Nb;
  hipLaunchKernelGGL(splitk_reduce_kernel, dim3(ew_grid((n + 3) / 4)), dim3(256), 0,
                     cur_stream(), ws.data_ptr<float>(), (short*)c.data_ptr(),
                     n, nsplit);
  return c;
}

torch::Tensor gemm_dgrad(torch::Tensor a, torch::Tensor b) {
  // C[M,K] = A[M,R]·B[R,K]: B reduction-strided via tr_b16 (no transpose)
  CHECK_HIP(a, torch::kBFloat16);
  CHECK_HIP(b, torch::kBFloat16);
  int M = (int)a.size(0), R = (int)a.size(1), K = (int)b.size(1);
  TORCH_CHECK((int)b.size(0) == R, "inner dims mismatch");
  TORCH_CHECK(M % 128 == 0 && K % 128 == 0 && R % 64 == 0,
              "gemm_dgrad needs M/K%128==0, R%64==0");
  auto c = torch::empty({M, K}, a.options());
  int tiles = (M / 128) * (K / 128);
  int nsplit = 1;
  while (tiles * nsplit < 384 && nsplit < 8 && (R / 64) / (nsplit * 2) >= 2)
    nsplit *= 2;
  if (nsplit == 1) {
    hipLaunchKernelGGL(gemm_dgrad_bf16_kernel, dim3(tiles), dim3(256), 0,
                       cur_stream(), (const short*)a.data_ptr(),
                       (const short*)b.data_ptr(), (short*)c.data_ptr(),
                       nullptr, M, R, K, 1);
    return c;
  }
  auto ws = torch::empty({nsplit, M, K}, a.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(gemm_dgrad_bf16_kernel, dim3(tiles * nsplit), dim3(256), 0,
                     cur_stream(), (const short*)a.data_ptr(),
                     (const short*)b.data_ptr(), nullptr, ws.data_ptr<float>(),
                     M, R, K, nsplit);
  long long n = (long long)M * K;
  hipLaunchKernelGGL(splitk_reduce_kernel, dim3(ew_grid((n + 3) / 4)), dim3(256), 0,
                     cur_stream(), ws.data_ptr<float>(), (short*)c.data_ptr(),
                     n, nsplit);
  return c;
}

// 256²-tile NT with split-K: fills the chip when the tile grid is starved
// (e.g. llama hidden GEMMs: 1024×2048 = 32 tiles over 256 CUs).
static torch::Tensor gemm_nt_splitk(torch::Tensor A, torch::Tensor B, int M,
                                    int N, int K, int nsplit, int variant) {
  auto c = torch::empty({M, N}, A.options());
  int tiles = (M / 256) * (N / 256);
  auto kfn = variant == 10 ? gemm_nt_bf16_v10sk_kernel : gemm_nt_bf16_v9sk_kernel;
  if (nsplit == 1) {
    auto k1 = variant == 10 ? gemm_nt_bf16_v10_kernel : gemm_nt_bf16_v9_kernel;
    hipLaunchKernelGGL(k1, dim3(tiles), dim3(512), 0, cur_stream(),
                       (const short*)A.data_ptr(), (const short*)B.data_ptr(),
                       (short*)c.data_ptr(), nullptr, M, N, K, 1);
    return c;
  }
  auto ws = torch::empty({nsplit, M, N}, A.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(kfn, dim3(tiles * nsplit), dim3(512), 0, cur_stream(),
                     (const short*)A.data_ptr(), (const short*)B.data_ptr(),
                     nullptr, ws.data_ptr<float>(), M, N, K, nsplit);
  long long n = (long long)M * N;
  hipLaunchKernelGGL(splitk_reduce_kernel, dim3(ew_grid((n + 3) / 4)), dim3(256), 0,
                     cur_stream(), ws.data_ptr<float>(), (short*)c.data_ptr(),
                     n, nsplit);
  return c;
}

// direct variant launcher for A/B probes (no dispatch heuristics):
// variant ∈ {5,6,7,8,9,10}; nsplit only for 9/10.
torch::Tensor gemm_nt_probe(torch::Tensor a, torch::Tensor b, int variant,
                            int nsplit) {
  CHECK_HIP(a, torch::kBFloat16);
  CHECK_HIP(b, torch::kBFloat16);
  int M = (int)a.size(0), K = (int)a.size(1), N = (int)b.size(0);
  TORCH_CHECK((int)b.size(1) == K, "inner dims mismatch");
  if (variant >= 9) {
    TORCH_CHECK(M % 256 == 0 && N % 256 == 0 && K % 64 == 0, "shape");
    TORCH_CHECK((K / 64) % nsplit == 0, "nsplit must divide K-tiles");
    return gemm_nt_splitk(a, b, M, N, K, nsplit, variant);
  }
  auto c = torch::empty({M, N}, a.options());
  if (variant >= 6) {
    TORCH_CHECK(M % 256 == 0 && N % 256 == 0 && K % 64 == 0, "shape");
    auto k6 = variant == 8   ? gemm_nt_bf16_v8_kernel
              : variant == 7 ? gemm_nt_bf16_v7_kernel
                             : gemm_nt_bf16_v6_kernel;
    hipLaunchKernelGGL(k6, dim3((M / 256) * (N / 256)), dim3(512), 0,
                       cur_stream(), (const short*)a.data_ptr(),
                       (const short*)b.data_ptr(), (short*)c.data_ptr(), M, N,
                       K);
    return c;
  }
  TORCH_CHECK(M % 128 == 0 && N % 128 == 0 && K % 64 == 0, "shape");
  auto k5 = variant == 55 ? gemm_nt_bf16_v5b_kernel : gemm_nt_bf16_v5_kernel;
  hipLaunchKernelGGL(k5, dim3((M / 128) * (N / 128)),
                     dim3(256), 0, cur_stream(), (const short*)a.data_ptr(),
                     (const short*)b.data_ptr(), (short*)c.data_ptr(), M, N, K);
  return c;
}

// Direct NHWC conv forward (conv.hip): x [N,C,H,W] channels_last bf16,
// w [Cout,Cin,KH,KW] channels_last bf16 → out [N,Cout,OH,OW] channels_last.
std::vector<torch::Tensor> conv_nhwc_fwd_stats(torch::Tensor x,
                                               torch::Tensor w,
                                               int64_t stride, int64_t pad,
                                               torch::Tensor bn_acc);

torch::Tensor conv_nhwc_fwd(torch::Tensor x, torch::Tensor w, int64_t stride,
                            int64_t pad) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast), "x NHWC");
  TORCH_CHECK(w.is_contiguous(at::MemoryFormat::ChannelsLast), "w NHWC");
  int Nb = (int)x.size(0), Cin = (int)x.size(1), H = (int)x.size(2),
      Wd = (int)x.size(3);
  int Cout = (int)w.size(0), KH = (int)w.size(2), KW = (int)w.size(3);
  TORCH_CHECK((int)w.size(1) == Cin, "channel mismatch");
  TORCH_CHECK(Cin % 64 == 0, "conv_nhwc_fwd needs Cin%64==0");
  TORCH_CHECK(Cout % 64 == 0, "conv_nhwc_fwd needs Cout%64==0");
  int OH = (H + 2 * (int)pad - KH) / (int)stride + 1;
  int OW = (Wd + 2 * (int)pad - KW) / (int)stride + 1;
  auto out = torch::empty({Nb, Cout, OH, OW},
                          x.options().memory_format(at::MemoryFormat::ChannelsLast));
  // 128-short zero page for out-of-bounds gather lanes (halo)
  static torch::Tensor zero_page;
  if (!zero_page.defined() || zero_page.device() != x.device())
    zero_page = torch::zeros({128}, x.options());
  int M = Nb * OH * OW;
  int mtiles = (M + 127) / 128;
  bool wide = Cout % 128 == 0 && Cout > 64;
  int tbn = wide ? 128 : 64;
  int tiles = mtiles * (Cout / tbn);
  int nk = KH * KW * (Cin / 64);
  // split the reduction when the tile grid starves the 256 CUs (deep
  // layers: 7×7×512 outputs = 52 tiles); no empty slices allowed
  int ns = 1;
  // only truly starved grids: measured in-graph, the fp32-workspace +
  // reduce overhead beats the fill gain above ~80 tiles (l2/l3 ResNet
  // shapes regressed; l4 at 52 tiles gains 30-55%)
  if (tiles < 80) {
    for (int cand = 8; cand > 1; cand >>= 1) {
      int per = (nk + cand - 1) / cand;
      bool no_empty_slice = (cand - 1) * per < nk;
      // each slice needs >= 4 K-tiles or the double-buffer pipeline is
      // all prologue (measured: 1x1 convs regress at per-slice nk 1)
      if (no_empty_slice && nk / cand >= 4 && tiles * cand >= 200 &&
          tiles * cand <= 1024) {
        ns = cand;
        break;
      }
    }
  }
  if (ns > 1) {
    auto ws = torch::empty({ns, M, Cout}, x.options().dtype(torch::kFloat32));
    auto kfn = wide ? conv_nhwc_fwd128sk_kernel : conv_nhwc_fwd64sk_kernel;
    hipLaunchKernelGGL(kfn, dim3(tiles * ns), dim3(256), 0, cur_stream(),
                       (const short*)x.data_ptr(), (const short*)w.data_ptr(),
                       ws.data_ptr<float>(),
                       (const short*)zero_page.data_ptr(), Nb, H, Wd, Cin,
                       Cout, OH, OW, KH, KW, (int)stride, (int)pad, ns);
    long long n = (long long)M * Cout;
    hipLaunchKernelGGL(splitk_reduce_kernel, dim3(ew_grid((n + 3) / 4)), dim3(256), 0,
                       cur_stream(), ws.data_ptr<float>(),
                       (short*)out.data_ptr(), n, ns);
    return out;
  }
  auto kfn = wide ? conv_nhwc_fwd128_kernel : conv_nhwc_fwd64_kernel;
  hipLaunchKernelGGL(kfn, dim3(tiles), dim3(256), 0, cur_stream(),
                     (const short*)x.data_ptr(), (const short*)w.data_ptr(),
                     (short*)out.data_ptr(), (const short*)zero_page.data_ptr(),
                     Nb, H, Wd, Cin, Cout, OH, OW, KH, KW, (int)stride,
                     (int)pad);
  return out;
}

// flash-style attention forward, head_dim 64 (attention.hip).
// q [B,H,S,64], k/v [B,Hkv,Skv,64] contiguous bf16 → (o, logsumexp)
std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, bool causal,
                                    double scale, bool want_lse) {
  for (auto& t : {q, k, v}) {
    TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kBFloat16 &&
                t.stride(3) == 1, "attn_fwd: bf16 cuda, head-dim contiguous");
  }
  TORCH_CHECK(k.strides() == v.strides(), "attn_fwd: k/v layouts differ");
  int B = (int)q.size(0), H = (int)q.size(1), S = (int)q.size(2);
  int Hkv = (int)k.size(1), Skv = (int)k.size(2);
  TORCH_CHECK(q.size(3) == 64 && k.size(3) == 64 && v.size(3) == 64,
              "attn_fwd: head_dim must be 64");
  TORCH_CHECK(H % Hkv == 0, "attn_fwd: H % Hkv");
  // match q's exact layout (slice views of a fused qkv tensor included):
  // empty_strided allocates the full addressed span, so strided kernel
  // writes stay in bounds; the gaps are never read back
  auto o = torch::empty_strided(q.sizes(), q.strides(), q.options());
  torch::Tensor lse;
  float* lse_ptr = nullptr;
  if (want_lse) {
    lse = torch::empty({B, H, S}, q.options().dtype(torch::kFloat32));
    lse_ptr = lse.data_ptr<float>();
  } else {
    lse = torch::empty({0}, q.options().dtype(torch::kFloat32));
  }
  dim3 grid((S + 127) / 128, B * H);
  static int plin = [] {
    const char* v = getenv("V6AMD_ATTN_PLIN");
    return v ? atoi(v) : 0;
  }();
  auto kfn = plin ? attn_fwd_d64_plin_kernel : attn_fwd_d64_kernel;
  hipLaunchKernelGGL(kfn, grid, dim3(512), 0, cur_stream(),
                     (const short*)q.data_ptr(), (const short*)k.data_ptr(),
                     (const short*)v.data_ptr(), (short*)o.data_ptr(), lse_ptr,
                     S, Skv, H, Hkv, (float)scale, causal ? 1 : 0,
                     q.stride(0), q.stride(1), q.stride(2), k.stride(0),
                     k.stride(1), k.stride(2));
  return {o, lse};
}

// attention backward (attention.hip): recompute-from-lse flash backward.
std::vector<torch::Tensor> attn_bwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, torch::Tensor o,
                                    torch::Tensor dout, torch::Tensor lse,
                                    bool causal, double scale) {
  int B = (int)q.size(0), H = (int)q.size(1), S = (int)q.size(2);
  int Hkv = (int)k.size(1), Skv = (int)k.size(2);
  auto dq = torch::empty_strided(q.sizes(), q.strides(), q.options());
  auto dk = torch::empty_strided(k.sizes(), k.strides(), k.options());
  auto dv = torch::empty_strided(v.sizes(), v.strides(), v.options());
  auto dsum = torch::empty({B, H, S}, q.options().dtype(torch::kFloat32));
  long long nrows = (long long)B * H * S;
  // the kernels take ONE stride set for the q-side tensors (q, dO, dq):
  // relayout dO into q's layout when it differs (one copy, rare)
  torch::Tensor dco = dout;
  if (dout.strides() != q.strides()) {
    dco = torch::empty_strided(q.sizes(), q.strides(), q.options());
    dco.copy_(dout);
  }
  // D = rowsum(dO ∘ O) via the strided kernel (the torch-expression form
  // materialised ~100 MB of fp32 temporaries per call)
  hipLaunchKernelGGL(attn_bwd_d_kernel, dim3((int)((nrows + 255) / 256)),
                     dim3(256), 0, cur_stream(),
                     (const short*)dco.data_ptr(), (const short*)o.data_ptr(),
                     dsum.data_ptr<float>(), H, S, nrows, q.stride(0),
                     q.stride(1), q.stride(2));
  hipLaunchKernelGGL(attn_bwd_dq_d64_kernel, dim3((S + 127) / 128, B * H),
                     dim3(512), 0, cur_stream(), (const short*)q.data_ptr(),
                     (const short*)k.data_ptr(), (const short*)v.data_ptr(),
                     (const short*)dco.data_ptr(), lse.data_ptr<float>(),
                     dsum.data_ptr<float>(), (short*)dq.data_ptr(), S, Skv, H,
                     Hkv, (float)scale, causal ? 1 : 0, q.stride(0),
                     q.stride(1), q.stride(2), k.stride(0), k.stride(1),
                     k.stride(2));
  int group = H / Hkv;
  if (group == 1) {
    hipLaunchKernelGGL(attn_bwd_dkdv_d64_kernel,
                       dim3((Skv + 127) / 128, B * Hkv, 1), dim3(512), 0,
                       cur_stream(), (const short*)q.data_ptr(),
                       (const short*)k.data_ptr(), (const short*)v.data_ptr(),
                       (const short*)dco.data_ptr(), lse.data_ptr<float>(),
                       dsum.data_ptr<float>(), (short*)dk.data_ptr(),
                       (short*)dv.data_ptr(), nullptr, nullptr, S, Skv, H,
                       Hkv, (float)scale, causal ? 1 : 0, q.stride(0),
                       q.stride(1), q.stride(2), k.stride(0), k.stride(1),
                       k.stride(2));
    return {dq, dk, dv};
  }
  // GQA: one z-slice per group q-head, fp32 slabs, then a reduce — the
  // slabs (and dk/dv) use the contiguous [B,Hkv,Skv,64] layout
  dk = torch::empty({B, Hkv, Skv, 64}, k.options());
  dv = torch::empty({B, Hkv, Skv, 64}, v.options());
  long long nkv = (long long)B * Hkv * Skv * 64;
  auto wsk = torch::empty({group, nkv}, q.options().dtype(torch::kFloat32));
  auto wsv = torch::empty({group, nkv}, q.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(attn_bwd_dkdv_d64_kernel,
                     dim3((Skv + 127) / 128, B * Hkv, group), dim3(512), 0,
                     cur_stream(), (const short*)q.data_ptr(),
                     (const short*)k.data_ptr(), (const short*)v.data_ptr(),
                     (const short*)dco.data_ptr(), lse.data_ptr<float>(),
                     dsum.data_ptr<float>(), nullptr, nullptr,
                     wsk.data_ptr<float>(), wsv.data_ptr<float>(), S, Skv, H,
                     Hkv, (float)scale, causal ? 1 : 0, q.stride(0),
                     q.stride(1), q.stride(2), k.stride(0), k.stride(1),
                     k.stride(2));
  hipLaunchKernelGGL(splitk_reduce_kernel, dim3(ew_grid((nkv + 3) / 4)), dim3(256), 0,
                     cur_stream(), wsk.data_ptr<float>(),
                     (short*)dk.data_ptr(), nkv, group);
  hipLaunchKernelGGL(splitk_reduce_kernel, dim3(ew_grid((nkv + 3) / 4)), dim3(256), 0,
                     cur_stream(), wsv.data_ptr<float>(),
                     (short*)dv.data_ptr(), nkv, group);
  return {dq, dk, dv};
}

// conv fwd + fused BN-stat accumulation (acc = the downstream BN's
// [16][2][Cout] atomic accumulator). Returns (out, stats_done): the
// split-K path does not produce stats (the caller then runs the separate
// BN partials pass).
std::vector<torch::Tensor> conv_nhwc_fwd_stats(torch::Tensor x,
                                               torch::Tensor w,
                                               int64_t stride, int64_t pad,
                                               torch::Tensor bn_acc) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast), "x NHWC");
  TORCH_CHECK(w.is_contiguous(at::MemoryFormat::ChannelsLast), "w NHWC");
  int Nb = (int)x.size(0), Cin = (int)x.size(1), H = (int)x.size(2),
      Wd = (int)x.size(3);
  int Cout = (int)w.size(0), KH = (int)w.size(2), KW = (int)w.size(3);
  TORCH_CHECK(Cin % 64 == 0 && Cout % 64 == 0);
  TORCH_CHECK(bn_acc.numel() == 32LL * Cout, "bn_acc must be [16][2][Cout]");
  int OH = (H + 2 * (int)pad - KH) / (int)stride + 1;
  int OW = (Wd + 2 * (int)pad - KW) / (int)stride + 1;
  int M = Nb * OH * OW;
  int mtiles = (M + 127) / 128;
  bool wide = Cout % 128 == 0 && Cout > 64;
  int tbn = wide ? 128 : 64;
  int tiles = mtiles * (Cout / tbn);
  int nk = KH * KW * (Cin / 64);
  bool starved = tiles < 80;
  int ns = 1;
  if (starved) {
    for (int cand = 8; cand > 1; cand >>= 1) {
      int per = (nk + cand - 1) / cand;
      if ((cand - 1) * per < nk && nk / cand >= 4 && tiles * cand >= 200 &&
          tiles * cand <= 1024) {
        ns = cand;
        break;
      }
    }
  }
  auto done = torch::zeros({1}, torch::TensorOptions().dtype(torch::kUInt8));
  if (ns > 1) {
    // split-K path has no stats — delegate and report stats_done = 0
    auto out = conv_nhwc_fwd(x, w, stride, pad);
    return {out, done};
  }
  auto out = torch::empty({Nb, Cout, OH, OW},
                          x.options().memory_format(at::MemoryFormat::ChannelsLast));
  static torch::Tensor zero_page2;
  if (!zero_page2.defined() || zero_page2.device() != x.device())
    zero_page2 = torch::zeros({128}, x.options());
  auto kfn = wide ? conv_nhwc_fwd128st_kernel : conv_nhwc_fwd64st_kernel;
  hipLaunchKernelGGL(kfn, dim3(tiles), dim3(256), 0, cur_stream(),
                     (const short*)x.data_ptr(), (const short*)w.data_ptr(),
                     (short*)out.data_ptr(),
                     (const short*)zero_page2.data_ptr(),
                     bn_acc.data_ptr<float>(), Nb, H, Wd, Cin, Cout, OH, OW,
                     KH, KW, (int)stride, (int)pad);
  done.fill_(1);
  return {out, done};
}

// conv wgrad: dW = per-tap dy^T · gathered-x (conv.hip); returns the
// channels-last weight-grad tensor. Split-K over output pixels.
torch::Tensor conv_nhwc_wgrad(torch::Tensor dy, torch::Tensor x, int64_t kh,
                              int64_t kw, int64_t stride, int64_t pad) {
  TORCH_CHECK(dy.is_cuda() && dy.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(dy.is_contiguous(at::MemoryFormat::ChannelsLast), "dy NHWC");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast), "x NHWC");
  int Nb = (int)x.size(0), Cin = (int)x.size(1), H = (int)x.size(2),
      Wd = (int)x.size(3);
  int Cout = (int)dy.size(1), OH = (int)dy.size(2), OW = (int)dy.size(3);
  TORCH_CHECK(Cout % 64 == 0 && Cin >= 64 && (Cin & (Cin - 1)) == 0,
              "conv_nhwc_wgrad needs Cout%64==0 and power-of-2 Cin>=64");
  int cin_log2 = 0;
  while ((1 << cin_log2) < Cin) ++cin_log2;
  int M = Nb * OH * OW;
  int taps = (int)(kh * kw);
  int NbCols = taps * Cin;
  auto dw = torch::empty({Cout, Cin, (long)kh, (long)kw},
                         x.options().memory_format(at::MemoryFormat::ChannelsLast));
  static torch::Tensor zero_page;
  if (!zero_page.defined() || zero_page.device() != x.device())
    zero_page = torch::zeros({128}, x.options());
  int tiles = ((Cout + 127) / 128) * ((NbCols + 127) / 128);
  int nk = (M + 63) / 64;
  unsigned long long magic_ohow = ((1ULL << 40) / (OH * OW)) + 1;
  unsigned long long magic_ow = ((1ULL << 40) / OW) + 1;
  int ns = 1;
  for (int cand = 64; cand > 1; cand >>= 1) {
    int per = (nk + cand - 1) / cand;
    if ((cand - 1) * per < nk && nk / cand >= 8 && tiles * cand <= 768) {
      ns = cand;
      break;
    }
  }
  if (ns == 1) {
    hipLaunchKernelGGL(conv_nhwc_wgrad_kernel, dim3(tiles), dim3(256), 0,
                       cur_stream(), (const short*)dy.data_ptr(),
                       (const short*)x.data_ptr(), nullptr,
                       (short*)dw.data_ptr(),
                       (const short*)zero_page.data_ptr(), M, NbCols, H, Wd,
                       cin_log2, Cout, OH, OW, (int)kh, (int)kw, (int)stride,
                       (int)pad, 1, magic_ohow, magic_ow);
    return dw;
  }
  auto ws = torch::empty({ns, (long)Cout * NbCols},
                         x.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(conv_nhwc_wgrad_kernel, dim3(tiles * ns), dim3(256), 0,
                     cur_stream(), (const short*)dy.data_ptr(),
                     (const short*)x.data_ptr(), ws.data_ptr<float>(), nullptr,
                     (const short*)zero_page.data_ptr(), M, NbCols, H, Wd,
                     cin_log2, Cout, OH, OW, (int)kh, (int)kw, (int)stride,
                     (int)pad, ns, magic_ohow, magic_ow);
  long long n = (long long)Cout * NbCols;
  hipLaunchKernelGGL(splitk_reduce_kernel, dim3(ew_grid((n + 3) / 4)), dim3(256), 0,
                     cur_stream(), ws.data_ptr<float>(), (short*)dw.data_ptr(),
                     n, ns);
  return dw;
}

// NHWC max-pool 3x3/s2/p1 forward (+argmax) and backward (conv.hip).
std::vector<torch::Tensor> maxpool_nhwc_fwd(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast), "x NHWC");
  int Nb = (int)x.size(0), C = (int)x.size(1), H = (int)x.size(2),
      Wd = (int)x.size(3);
  TORCH_CHECK(C % 8 == 0);
  int OH = (H + 1) / 2, OW = (Wd + 1) / 2;
  auto y = torch::empty({Nb, C, OH, OW},
                        x.options().memory_format(at::MemoryFormat::ChannelsLast));
  auto idx = torch::empty({(long long)Nb * OH * OW * C},
                          x.options().dtype(torch::kUInt8));
  long long total = (long long)Nb * OH * OW * (C / 8);
  hipLaunchKernelGGL(maxpool_nhwc_fwd_kernel,
                     dim3((int)((total + 255) / 256)), dim3(256), 0,
                     cur_stream(), (const short*)x.data_ptr(),
                     (short*)y.data_ptr(), (unsigned char*)idx.data_ptr(), Nb,
                     H, Wd, C, OH, OW);
  return {y, idx};
}

torch::Tensor maxpool_nhwc_bwd(torch::Tensor dy, torch::Tensor idx, int64_t H,
                               int64_t Wd) {
  TORCH_CHECK(dy.is_cuda() && dy.scalar_type() == torch::kBFloat16);
  auto dyc = dy.contiguous(at::MemoryFormat::ChannelsLast);
  int Nb = (int)dyc.size(0), C = (int)dyc.size(1), OH = (int)dyc.size(2),
      OW = (int)dyc.size(3);
  auto dx = torch::empty({Nb, C, (long)H, (long)Wd},
                         dy.options().memory_format(at::MemoryFormat::ChannelsLast));
  long long total = (long long)Nb * H * Wd * (C / 8);
  hipLaunchKernelGGL(maxpool_nhwc_bwd_kernel,
                     dim3((int)((total + 255) / 256)), dim3(256), 0,
                     cur_stream(), (const short*)dyc.data_ptr(),
                     (const unsigned char*)idx.data_ptr(),
                     (short*)dx.data_ptr(), Nb, (int)H, (int)Wd, C, OH, OW);
  return dx;
}

// Weight transform for stride-1 dgrad (one kernel: flip + channel
// transpose into channels-last layout for conv_nhwc_fwd).
torch::Tensor conv_flip_weights(torch::Tensor w) {
  TORCH_CHECK(w.is_cuda() && w.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(w.is_contiguous(at::MemoryFormat::ChannelsLast), "w NHWC");
  int Cout = (int)w.size(0), Cin = (int)w.size(1), KH = (int)w.size(2),
      KW = (int)w.size(3);
  auto wd = torch::empty({Cin, Cout, KH, KW},
                         w.options().memory_format(at::MemoryFormat::ChannelsLast));
  int blocks = KH * KW * ((Cin + 63) / 64) * ((Cout + 63) / 64);
  hipLaunchKernelGGL(conv_flip_weights_kernel, dim3(blocks), dim3(256), 0,
                     cur_stream(), (const short*)w.data_ptr(),
                     (short*)wd.data_ptr(), Cout, Cin, KH, KW);
  return wd;
}

// C[M,N] = op(A)·op(B); inputs 2-D bf16 contiguous.
torch::Tensor gemm_bf16(torch::Tensor a, torch::Tensor b, bool trans_a,
                        bool trans_b) {
  CHECK_HIP(a, torch::kBFloat16);
  CHECK_HIP(b, torch::kBFloat16);
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2, "gemm_bf16 expects 2-D");
  if (!trans_a && !trans_b) {
    // dgrad form: A·B with B reduction-strided
    int M_ = (int)a.size(0), R_ = (int)a.size(1), K_ = (int)b.size(1);
    static bool use_dgrad = [] {
      const char* v = getenv("V6AMD_DGRAD_KERNEL");
      return v ? atoi(v) != 0 : true;
    }();
    if (use_dgrad && M_ % 128 == 0 && K_ % 128 == 0 && R_ % 64 == 0)
      return gemm_dgrad(a, b);
  }
  if (trans_a && !trans_b) {
    // wgrad form: A^T·B with both operands reduction-strided
    int M_ = (int)a.size(0), Na_ = (int)a.size(1), Nb_ = (int)b.size(1);
    static bool use_wgrad = [] {
      const char* v = getenv("V6AMD_WGRAD_KERNEL");
      return v ? atoi(v) != 0 : true;
    }();
    if (use_wgrad && M_ % 64 == 0 && Na_ % 128 == 0 && Nb_ % 128 == 0 &&
        (long long)M_ * Na_ > 65536)
      return gemm_wgrad(a, b);
  }
  // normalize to NT: C[M,N] = A'[M,K] · B'[N,K]^T
  torch::Tensor A = trans_a ? transpose_bf16(a) : a;
  torch::Tensor B = trans_b ? b : transpose_bf16(b);
  int M = (int)A.size(0), K = (int)A.size(1), N = (int)B.size(0);
  TORCH_CHECK((int)B.size(1) == K, "inner dims mismatch");
  // pad to tile multiples so the fast kernels always run (zero rows/cols
  // are exact under GEMM); slice the result back
  int M0 = M, N0 = N;
  {
    int Kp = (K + 63) / 64 * 64;
    int align = (M >= 256 || N >= 256) ? 256 : 128;
    int Mp = (M + align - 1) / align * align;
    int Np = (N + align - 1) / align * align;
    namespace F = torch::nn::functional;
    if (Kp != K || Mp != M) {
      A = torch::constant_pad_nd(A, {0, Kp - K, 0, Mp - M});
    }
    if (Kp != K || Np != N) {
      B = torch::constant_pad_nd(B, {0, Kp - K, 0, Np - N});
    }
    M = Mp; N = Np; K = Kp;
  }
  auto c = torch::empty({M, N}, a.options());
  if (M % 128 == 0 && N % 128 == 0 && K % 64 == 0) {
    int nwg = (M / 128) * (N / 128);
    static int variant = [] {
      const char* v = getenv("V6AMD_GEMM_KERNEL");
      return v ? atoi(v) : 0;  // 0 = measured shape-aware dispatch below
    }();
    int nwg6_pre = (M / 256) * (N / 256);
    bool div256 = M % 256 == 0 && N % 256 == 0;
    // measured dispatch (tools/gemm_v9_ab.py, profiles/kernel_perf_r02.md):
    //  big grid + deep K        → v8   (1054@4k³, 1116@8k³)
    //  huge grid, mid K         → v10  (lm_head 1024×128256×2048: 978)
    //  huge grid, shallow K     → v7   (bert_head 4096×30720×768: 807)
    //  starved grid + deep K    → v10 split-K (llama down 1024×2048×8192:
    //                             480 vs 340 on 128² tiles)
    //  starved grid, shallow K  → v5 128² (llama gate 1024×8192×2048: 625
    //                             beats every 256² variant)
    if (variant == 0 && div256) {
      if (nwg6_pre >= 128 && K >= 2560) {
        hipLaunchKernelGGL(gemm_nt_bf16_v8_kernel, dim3(nwg6_pre), dim3(512),
                           0, cur_stream(), (const short*)A.data_ptr(),
                           (const short*)B.data_ptr(), (short*)c.data_ptr(), M,
                           N, K);
        if (M0 != M || N0 != N)
          return c.narrow(0, 0, M0).narrow(1, 0, N0).contiguous();
        return c;
      }
      if (nwg6_pre >= 384) {
        if (K >= 1536) {
          auto out = gemm_nt_splitk(A, B, M, N, K, 1, 10);
          if (M0 != M || N0 != N)
            return out.narrow(0, 0, M0).narrow(1, 0, N0).contiguous();
          return out;
        }
        hipLaunchKernelGGL(gemm_nt_bf16_v7_kernel, dim3(nwg6_pre), dim3(512),
                           0, cur_stream(), (const short*)A.data_ptr(),
                           (const short*)B.data_ptr(), (short*)c.data_ptr(), M,
                           N, K);
        if (M0 != M || N0 != N)
          return c.narrow(0, 0, M0).narrow(1, 0, N0).contiguous();
        return c;
      }
      if (K >= 4096) {
        int nk_tiles = K / 64;
        int ns = 8;
        while (ns > 1 && (nk_tiles % ns != 0 || nk_tiles / ns < 4)) ns >>= 1;
        auto out = gemm_nt_splitk(A, B, M, N, K, ns, 10);
        if (M0 != M || N0 != N)
          return out.narrow(0, 0, M0).narrow(1, 0, N0).contiguous();
        return out;
      }
      // fall through to v5 128² tiles (starved grid, shallow K)
    }
    if ((variant >= 6 || variant == 9 || variant == 10) && div256) {
      if (variant >= 9) {
        auto out = gemm_nt_splitk(A, B, M, N, K, 1, variant);
        if (M0 != M || N0 != N)
          return out.narrow(0, 0, M0).narrow(1, 0, N0).contiguous();
        return out;
      }
      auto k6 = variant == 8   ? gemm_nt_bf16_v8_kernel
                : variant == 7 ? gemm_nt_bf16_v7_kernel
                               : gemm_nt_bf16_v6_kernel;
      hipLaunchKernelGGL(k6, dim3(nwg6_pre), dim3(512), 0, cur_stream(),
                         (const short*)A.data_ptr(), (const short*)B.data_ptr(),
                         (short*)c.data_ptr(), M, N, K);
      if (M0 != M || N0 != N)
        return c.narrow(0, 0, M0).narrow(1, 0, N0).contiguous();
      return c;
    }
    auto kfn = gemm_nt_bf16_v4_kernel;
    if (variant == 2) kfn = gemm_nt_bf16_v2_kernel;
    if (variant == 3) kfn = gemm_nt_bf16_v3_kernel;
    if (variant == 0 || variant == 5 || variant == 6 || variant == 7 ||
        variant >= 9)
      kfn = gemm_nt_bf16_v5_kernel;
    hipLaunchKernelGGL(kfn, dim3(nwg), dim3(256), 0, cur_stream(),
                       (const short*)A.data_ptr(), (const short*)B.data_ptr(),
                       (short*)c.data_ptr(), M, N, K);
  } else {
    dim3 grid((N + 127) / 128, (M + 127) / 128);  // BN=BM=128 (gemm.hip)
    hipLaunchKernelGGL(gemm_nt_bf16_kernel, grid, dim3(256), 0, cur_stream(),
                       (const short*)A.data_ptr(), (const short*)B.data_ptr(),
                       (short*)c.data_ptr(), M, N, K, K, K, N);
  }
  if (M0 != M || N0 != N)
    return c.narrow(0, 0, M0).narrow(1, 0, N0).contiguous();
  return c;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("fused_sgd", &fused_sgd, "fused SGD-momentum on flat buffers");
  m.def("fused_adamw", &fused_adamw, "fused AdamW on flat buffers",
        py::arg("param"), py::arg("master"), py::arg("grad"), py::arg("exp_avg"),
        py::arg("exp_avg_sq"), py::arg("lr"), py::arg("beta1"), py::arg("beta2"),
        py::arg("eps"), py::arg("weight_decay"), py::arg("step"),
        py::arg("step_dev") = py::none());
  m.def("fused_sgd_mt", &fused_sgd_mt, "multi-tensor SGD, scattered grads");
  m.def("fused_adamw_mt", &fused_adamw_mt, "multi-tensor AdamW, scattered grads",
        py::arg("table"), py::arg("param"), py::arg("master"), py::arg("exp_avg"),
        py::arg("exp_avg_sq"), py::arg("lr"), py::arg("beta1"), py::arg("beta2"),
        py::arg("eps"), py::arg("weight_decay"), py::arg("step"),
        py::arg("step_dev") = py::none());
  m.def("secagg_quantize_mask", &secagg_quantize_mask,
        "fixed-point quantize + pairwise Philox masks");
  m.def("dp_scale_noise", &dp_scale_noise, "DP clip-scale + gaussian noise",
        py::arg("grad"), py::arg("scale"), py::arg("noise_std"), py::arg("seed"),
        py::arg("step_dev") = py::none());
  m.def("grad_norm2_partials", &grad_norm2_partials, "block partial sum of g^2");
  m.def("rope_apply", &rope_apply);
  m.def("colsum_bf16", &colsum_bf16, "column sum of a bf16 matrix (bias grad)");
  m.def("ce_fwd", &ce_fwd, "fused bf16 cross-entropy fwd: {loss[R], lse[R]}");
  m.def("ce_bwd", &ce_bwd, "fused bf16 cross-entropy bwd: dlogits");
  m.def("emb_grad_bf16", &emb_grad_bf16, "graph-safe embedding backward");
  m.def("swiglu_fwd", &swiglu_fwd);
  m.def("swiglu_bwd", &swiglu_bwd);
  m.def("layernorm_fwd", &layernorm_fwd);
  m.def("layernorm_bwd", &layernorm_bwd);
  m.def("rmsnorm_fwd", &rmsnorm_fwd);
  m.def("rmsnorm_bwd", &rmsnorm_bwd);
  m.def("bn_fwd", &bn_fwd, "fused NHWC BatchNorm fwd (+relu/+residual)",
        py::arg("x"), py::arg("w"), py::arg("b"), py::arg("running_mean"),
        py::arg("running_var"), py::arg("training"), py::arg("momentum"),
        py::arg("eps"), py::arg("relu"), py::arg("residual") = py::none(),
        py::arg("acc") = py::none(), py::arg("stats_precomputed") = false);
  m.def("bn_bwd", &bn_bwd, "fused NHWC BatchNorm bwd");
  m.def("gemm_bf16", &gemm_bf16, "bf16 MFMA GEMM, fp32 accumulate");
  m.def("attn_bwd", &attn_bwd, "flash-style attention bwd (d=64)",
        py::arg("q"), py::arg("k"), py::arg("v"), py::arg("o"), py::arg("dout"),
        py::arg("lse"), py::arg("causal") = false, py::arg("scale") = 0.125);
  m.def("attn_fwd", &attn_fwd, "flash-style attention fwd (d=64)",
        py::arg("q"), py::arg("k"), py::arg("v"), py::arg("causal") = false,
        py::arg("scale") = 0.125, py::arg("want_lse") = true);
  m.def("conv_nhwc_wgrad", &conv_nhwc_wgrad, "direct NHWC conv weight grad",
        py::arg("dy"), py::arg("x"), py::arg("kh"), py::arg("kw"),
        py::arg("stride") = 1, py::arg("pad") = 1);
  m.def("maxpool_nhwc_fwd", &maxpool_nhwc_fwd, "NHWC maxpool 3x3/s2/p1 + argmax");
  m.def("maxpool_nhwc_bwd", &maxpool_nhwc_bwd, "NHWC maxpool backward (gather)");
  m.def("conv_flip_weights", &conv_flip_weights,
        "flip+transpose conv weights for stride-1 dgrad");
  m.def("conv_nhwc_fwd_stats", &conv_nhwc_fwd_stats,
        "conv forward with fused BN-stat accumulation",
        py::arg("x"), py::arg("w"), py::arg("stride"), py::arg("pad"),
        py::arg("bn_acc"));
  m.def("conv_nhwc_fwd", &conv_nhwc_fwd, "direct NHWC conv forward",
        py::arg("x"), py::arg("w"), py::arg("stride") = 1, py::arg("pad") = 1);
  m.def("gemm_nt_probe", &gemm_nt_probe, "direct NT kernel-variant launcher",
        py::arg("a"), py::arg("b"), py::arg("variant"), py::arg("nsplit") = 1);
  m.def("gemm_wgrad", &gemm_wgrad, "A^T*B via tr_b16 reads (no transposes)");
  m.def("gemm_dgrad", &gemm_dgrad, "A*B with strided B via tr_b16 reads");
  m.def("transpose_bf16", &transpose_bf16);
}
