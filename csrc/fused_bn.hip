// Fused NHWC BatchNorm (+ReLU, +residual) training kernels for MI355X.
//
// Motivation (profiles/resnet50_b256_steady_r01.txt): MIOpen's spatial BN
// splits fwd into MeanVariance+Norm and bwd into DScaleDBias+DX, and eager
// torch adds separate residual-add and ReLU kernels — together ~50% of a
// ResNet-50 bf16 step on MI355X while convs are ~30%.  These kernels fuse:
//   forward : one stats pass over x, one apply pass producing
//             y = relu(scale*x + shift [+ residual])
//   backward: one reduce pass (dgamma/dbeta with the ReLU mask folded in),
//             one apply pass producing dx [and d_residual]
// All passes are HBM-bound streaming over an [M, C] view (NHWC, channels
// innermost): 16 B/lane vectorized bf16x8 access (guide Guideline 13),
// fp32 accumulation, per-channel coefficients staged in LDS.
//
// Layout contract: x is channels-last (NHWC) contiguous, C % 8 == 0.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#define BN_BLOCK 256

namespace {

typedef __hip_bfloat16 bf16;
typedef float floatv4 __attribute__((ext_vector_type(4)));

union bf16x8 {
  uint4 u4;            // one 16-B load/store
  unsigned short h[8];
};

__device__ __forceinline__ float b2f(unsigned short v) {
  __hip_bfloat16_raw r;
  r.x = v;
  return __bfloat162float(*reinterpret_cast<bf16*>(&r));
}

__device__ __forceinline__ unsigned short f2b(float f) {
  bf16 h = __float2bfloat16(f);
  return *reinterpret_cast<unsigned short*>(&h);
}

// ---------------------------------------------------------------------------
// Forward stats: per-channel sum and sum-of-squares over the M rows.
// Each block owns a row-tile for ALL channels; threads own fixed channel
// octets so partial sums live in registers; one LDS reduce per block, then
// the block writes its per-channel partials to ``part`` ([nblocks, 2C]:
// sums first, then sum-of-squares).  NO global atomics: ~2000 blocks
// atomically RMW-ing the same C addresses serialize per address (~measured
// 400+us/launch on the 14x14 layers); per-block partials + a coalesced
// second-stage reduce in the finalize kernel run at streaming speed.
// Grid is capped at BN_NBLK blocks (512 x 4 waves = 8 waves/CU on 256 CUs,
// enough to saturate HBM for a pure streaming kernel).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(BN_BLOCK) void bn_fwd_stats_kernel(
    const bf16x8* __restrict__ x, float* __restrict__ part, long M, int c8,
    int rows_per_block) {
  const int slots = c8;                      // channel octets per row
  const int rows_par = BN_BLOCK / slots;     // rows processed concurrently
  const int slot = threadIdx.x % slots;      // this thread's channel octet
  const int rsub = threadIdx.x / slots;
  long row0 = (long)blockIdx.x * rows_per_block;
  long row_end = min(row0 + rows_per_block, M);
  float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  float acc2[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  if (rsub < rows_par) {
    #pragma unroll 2
    for (long r = row0 + rsub; r < row_end; r += rows_par) {
      bf16x8 v = x[r * slots + slot];
      #pragma unroll
      for (int k = 0; k < 8; ++k) {
        float f = b2f(v.h[k]);
        acc[k] += f;
        acc2[k] += f * f;
      }
    }
  }
  // Reduce across the rows_par threads sharing a slot: LDS tree.
  __shared__ float smem[BN_BLOCK * 8];
  const int C = c8 * 8;
  float* out = part + (long)blockIdx.x * 2 * C;
  #pragma unroll
  for (int k = 0; k < 8; ++k) smem[threadIdx.x * 8 + k] = acc[k];
  __syncthreads();
  if (rsub == 0) {
    #pragma unroll 1
    for (int rr = 1; rr < rows_par; ++rr)
      #pragma unroll
      for (int k = 0; k < 8; ++k) acc[k] += smem[(rr * slots + slot) * 8 + k];
    #pragma unroll
    for (int k = 0; k < 8; ++k) out[slot * 8 + k] = acc[k];
  }
  __syncthreads();
  #pragma unroll
  for (int k = 0; k < 8; ++k) smem[threadIdx.x * 8 + k] = acc2[k];
  __syncthreads();
  if (rsub == 0) {
    #pragma unroll 1
    for (int rr = 1; rr < rows_par; ++rr)
      #pragma unroll
      for (int k = 0; k < 8; ++k) acc2[k] += smem[(rr * slots + slot) * 8 + k];
    #pragma unroll
    for (int k = 0; k < 8; ++k) out[C + slot * 8 + k] = acc2[k];
  }
}

// Finalize: reduce block partials, mean/var -> apply coefficients +
// running-stat update.  1024 threads laid out as 16 partial-lanes x 64
// channel-lanes per block: each wave reads 64 consecutive channels of one
// partial row (fully coalesced) and 16 rows are in flight, so the
// [nblocks, 2C] sweep is latency-hidden even when C is small (a
// one-thread-per-channel version measured 42 us/call — 4.4 ms/step —
// because 1-8 blocks of serial loads ran on a handful of CUs).
#define FIN_PPAR 16
#define FIN_CH 64

__device__ __forceinline__ void finalize_reduce_pair(
    const float* __restrict__ part, int nblocks, int C, int c, bool valid,
    float* s_out, float* s2_out) {
  const int pp = threadIdx.x / FIN_CH;   // partial lane
  const int cc = threadIdx.x % FIN_CH;
  float s = 0.f, s2 = 0.f;
  if (valid) {
    for (int p = pp; p < nblocks; p += FIN_PPAR) {
      s += part[(long)p * 2 * C + c];
      s2 += part[(long)p * 2 * C + C + c];
    }
  }
  __shared__ float smem[2 * FIN_PPAR * FIN_CH];
  smem[pp * FIN_CH + cc] = s;
  smem[FIN_PPAR * FIN_CH + pp * FIN_CH + cc] = s2;
  __syncthreads();
  if (pp == 0) {
    #pragma unroll
    for (int r = 1; r < FIN_PPAR; ++r) {
      s += smem[r * FIN_CH + cc];
      s2 += smem[FIN_PPAR * FIN_CH + r * FIN_CH + cc];
    }
    *s_out = s;
    *s2_out = s2;
  }
}

__global__ __launch_bounds__(FIN_PPAR * FIN_CH) void bn_fwd_finalize_kernel(
    const float* __restrict__ part, int nblocks,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    float* __restrict__ mean_out, float* __restrict__ invstd_out,
    float* __restrict__ scale_out, float* __restrict__ shift_out,
    float* __restrict__ running_mean, float* __restrict__ running_var,
    long M, int C, float eps, float momentum) {
  const int c = blockIdx.x * FIN_CH + threadIdx.x % FIN_CH;
  float s, s2;
  finalize_reduce_pair(part, nblocks, C, c, c < C, &s, &s2);
  if (threadIdx.x >= FIN_CH || c >= C) return;
  float m = s / (float)M;
  float var = s2 / (float)M - m * m;
  var = fmaxf(var, 0.f);
  float invstd = rsqrtf(var + eps);
  float sc = gamma[c] * invstd;
  mean_out[c] = m;
  invstd_out[c] = invstd;
  scale_out[c] = sc;
  shift_out[c] = beta[c] - m * sc;
  if (running_mean != nullptr) {
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * m;
    float unbiased = var * (float)M / (float)max(M - 1, 1L);
    running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
  }
}

// ---------------------------------------------------------------------------
// Forward apply: y = act(scale*x + shift [+ residual]).  Pure streaming.
// ---------------------------------------------------------------------------
template <bool RELU, bool RES>
__global__ __launch_bounds__(BN_BLOCK) void bn_fwd_apply_kernel(
    const bf16x8* __restrict__ x, const bf16x8* __restrict__ res,
    bf16x8* __restrict__ y, const float* __restrict__ scale,
    const float* __restrict__ shift, long M, int c8, int rows_per_block) {
  // Row-tile geometry (same as the stats kernel): each thread owns a fixed
  // channel octet, so its scale/shift coefficients live in 16 registers and
  // there is no per-element modulo; a wave still covers 64 consecutive
  // octets of one row pair = fully coalesced 16 B/lane streaming.
  const int slots = c8;
  const int rows_par = BN_BLOCK / slots;
  const int slot = threadIdx.x % slots;
  const int rsub = threadIdx.x / slots;
  float sc[8], sh[8];
  #pragma unroll
  for (int k = 0; k < 8; ++k) {
    sc[k] = scale[slot * 8 + k];
    sh[k] = shift[slot * 8 + k];
  }
  long row0 = (long)blockIdx.x * rows_per_block;
  long row_end = min(row0 + rows_per_block, M);
  if (rsub >= rows_par) return;
  #pragma unroll 2
  for (long r = row0 + rsub; r < row_end; r += rows_par) {
    long i = r * slots + slot;
    bf16x8 v = x[i];
    bf16x8 rr;
    if (RES) rr = res[i];
    bf16x8 o;
    #pragma unroll
    for (int k = 0; k < 8; ++k) {
      float f = b2f(v.h[k]) * sc[k] + sh[k];
      if (RES) f += b2f(rr.h[k]);
      if (RELU) f = fmaxf(f, 0.f);
      o.h[k] = f2b(f);
    }
    y[i] = o;
  }
}

// ---------------------------------------------------------------------------
// Backward reduce: dbeta = sum(dy_eff), dgamma = sum(dy_eff * xhat) with the
// ReLU mask (y > 0) folded in; optionally writes d_residual = dy_eff.
// Same block geometry as the stats kernel.
// ---------------------------------------------------------------------------
template <bool RELU, bool RES>
__global__ __launch_bounds__(BN_BLOCK) void bn_bwd_reduce_kernel(
    const bf16x8* __restrict__ x, const bf16x8* __restrict__ dy,
    const bf16x8* __restrict__ y, bf16x8* __restrict__ dres,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    float* __restrict__ part, long M, int c8, int rows_per_block) {
  const int slots = c8;
  const int rows_par = BN_BLOCK / slots;
  const int slot = threadIdx.x % slots;
  const int rsub = threadIdx.x / slots;
  long row0 = (long)blockIdx.x * rows_per_block;
  long row_end = min(row0 + rows_per_block, M);
  float m[8], is[8];
  #pragma unroll
  for (int k = 0; k < 8; ++k) {
    m[k] = mean[slot * 8 + k];
    is[k] = invstd[slot * 8 + k];
  }
  float db[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  float dg[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  if (rsub < rows_par) {
    #pragma unroll 2
    for (long r = row0 + rsub; r < row_end; r += rows_par) {
      long i = r * slots + slot;
      bf16x8 vdy = dy[i];
      bf16x8 vx = x[i];
      bf16x8 vy;
      if (RELU) vy = y[i];
      bf16x8 vdr;
      #pragma unroll
      for (int k = 0; k < 8; ++k) {
        float g = b2f(vdy.h[k]);
        if (RELU && b2f(vy.h[k]) <= 0.f) g = 0.f;
        if (RES) vdr.h[k] = f2b(g);
        float xh = (b2f(vx.h[k]) - m[k]) * is[k];
        db[k] += g;
        dg[k] += g * xh;
      }
      if (RES) dres[i] = vdr;
    }
  }
  __shared__ float smem[BN_BLOCK * 8];
  const int C = c8 * 8;
  float* out = part + (long)blockIdx.x * 2 * C;
  #pragma unroll
  for (int k = 0; k < 8; ++k) smem[threadIdx.x * 8 + k] = db[k];
  __syncthreads();
  if (rsub == 0) {
    #pragma unroll 1
    for (int rr = 1; rr < rows_par; ++rr)
      #pragma unroll
      for (int k = 0; k < 8; ++k) db[k] += smem[(rr * slots + slot) * 8 + k];
    #pragma unroll
    for (int k = 0; k < 8; ++k) out[slot * 8 + k] = db[k];
  }
  __syncthreads();
  #pragma unroll
  for (int k = 0; k < 8; ++k) smem[threadIdx.x * 8 + k] = dg[k];
  __syncthreads();
  if (rsub == 0) {
    #pragma unroll 1
    for (int rr = 1; rr < rows_par; ++rr)
      #pragma unroll
      for (int k = 0; k < 8; ++k) dg[k] += smem[(rr * slots + slot) * 8 + k];
    #pragma unroll
    for (int k = 0; k < 8; ++k) out[C + slot * 8 + k] = dg[k];
  }
}

// Reduce bwd partials ([nblocks, 2C]: dbeta rows then dgamma) -> dbeta,
// dgamma.  Same 16x64 geometry as bn_fwd_finalize_kernel.
__global__ __launch_bounds__(FIN_PPAR * FIN_CH) void bn_bwd_finalize_kernel(
    const float* __restrict__ part, int nblocks, float* __restrict__ dbeta,
    float* __restrict__ dgamma, int C) {
  const int c = blockIdx.x * FIN_CH + threadIdx.x % FIN_CH;
  float db, dg;
  finalize_reduce_pair(part, nblocks, C, c, c < C, &db, &dg);
  if (threadIdx.x >= FIN_CH || c >= C) return;
  dbeta[c] = db;
  dgamma[c] = dg;
}

// ---------------------------------------------------------------------------
// Backward apply: dx = (gamma*invstd) * (dy_eff - dbeta/M - xhat*dgamma/M)
// ---------------------------------------------------------------------------
template <bool RELU>
__global__ __launch_bounds__(BN_BLOCK) void bn_bwd_apply_kernel(
    const bf16x8* __restrict__ x, const bf16x8* __restrict__ dy,
    const bf16x8* __restrict__ y, bf16x8* __restrict__ dx,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ gamma, const float* __restrict__ dbeta,
    const float* __restrict__ dgamma, long M, int c8, float invM,
    int rows_per_block) {
  // Row-tile geometry: per-thread fixed channel octet, coefficients in
  // registers (see bn_fwd_apply_kernel).
  const int slots = c8;
  const int rows_par = BN_BLOCK / slots;
  const int slot = threadIdx.x % slots;
  const int rsub = threadIdx.x / slots;
  float a[8], m[8], is[8], db[8], dg[8];
  #pragma unroll
  for (int k = 0; k < 8; ++k) {
    const int c = slot * 8 + k;
    is[k] = invstd[c];
    a[k] = gamma[c] * is[k];
    m[k] = mean[c];
    db[k] = dbeta[c] * invM;
    dg[k] = dgamma[c] * invM;
  }
  long row0 = (long)blockIdx.x * rows_per_block;
  long row_end = min(row0 + rows_per_block, M);
  if (rsub >= rows_par) return;
  #pragma unroll 2
  for (long r = row0 + rsub; r < row_end; r += rows_par) {
    long i = r * slots + slot;
    bf16x8 vdy = dy[i];
    bf16x8 vx = x[i];
    bf16x8 vy;
    if (RELU) vy = y[i];
    bf16x8 o;
    #pragma unroll
    for (int k = 0; k < 8; ++k) {
      float g = b2f(vdy.h[k]);
      if (RELU && b2f(vy.h[k]) <= 0.f) g = 0.f;
      float xh = (b2f(vx.h[k]) - m[k]) * is[k];
      float d = a[k] * (g - db[k] - xh * dg[k]);
      o.h[k] = f2b(d);
    }
    dx[i] = o;
  }
}

int pick_row_tiles(long M, int c8) {
  // BN_NBLK blocks of 4 waves = 8 waves/CU on 256 CUs: saturates HBM for a
  // streaming reduction while keeping the partial buffer / second-stage
  // reduce small.  More blocks would only lengthen the finalize loop.
  long target_blocks = 512;
  long rows_per_block = (M + target_blocks - 1) / target_blocks;
  if (rows_per_block < 16) rows_per_block = 16;
  return (int)rows_per_block;
}

}  // namespace

// ---------------------------------------------------------------------------
// Entry points.  x: [M, C] view of NHWC bf16; params fp32.
// ---------------------------------------------------------------------------
std::vector<at::Tensor> bn_fwd_train(
    at::Tensor x, c10::optional<at::Tensor> residual, at::Tensor gamma,
    at::Tensor beta, c10::optional<at::Tensor> running_mean,
    c10::optional<at::Tensor> running_var, double eps, double momentum,
    bool relu) {
  TORCH_CHECK(x.scalar_type() == at::kBFloat16, "bn_fwd_train: bf16 only");
  const long C = x.size(-1);
  TORCH_CHECK(C % 8 == 0 && C <= 4096, "bn: C must be %8==0 and <=4096");
  const long M = x.numel() / C;
  const int c8 = (int)(C / 8);
  TORCH_CHECK(BN_BLOCK % c8 == 0 || c8 >= BN_BLOCK,
              "bn: C/8 must divide 256 or exceed it");
  auto stream = at::hip::getCurrentHIPStream().stream();
  auto opts = gamma.options();
  auto mean = at::empty({C}, opts);
  auto invstd = at::empty({C}, opts);
  auto scale = at::empty({C}, opts);
  auto shift = at::empty({C}, opts);
  auto y = at::empty_like(x);
  const int slots = c8 <= BN_BLOCK ? c8 : BN_BLOCK;
  TORCH_CHECK(slots == c8, "bn: C too large for single-block-span layout");
  int rpb = pick_row_tiles(M, c8);
  int nblocks = (int)((M + rpb - 1) / rpb);
  auto part = at::empty({(long)nblocks * 2 * C}, opts);
  hipLaunchKernelGGL(bn_fwd_stats_kernel, dim3(nblocks), dim3(BN_BLOCK), 0,
                     stream, (const bf16x8*)x.data_ptr(),
                     part.data_ptr<float>(), M, c8, rpb);
  hipLaunchKernelGGL(bn_fwd_finalize_kernel, dim3((C + FIN_CH - 1) / FIN_CH),
                     dim3(FIN_PPAR * FIN_CH),
                     0, stream, part.data_ptr<float>(), nblocks,
                     gamma.data_ptr<float>(), beta.data_ptr<float>(),
                     mean.data_ptr<float>(), invstd.data_ptr<float>(),
                     scale.data_ptr<float>(), shift.data_ptr<float>(),
                     running_mean ? running_mean->data_ptr<float>() : nullptr,
                     running_var ? running_var->data_ptr<float>() : nullptr,
                     M, (int)C, (float)eps, (float)momentum);
  const bf16x8* resp =
      residual ? (const bf16x8*)residual->data_ptr() : nullptr;
  auto launch_apply = [&](auto relu_t, auto res_t) {
    hipLaunchKernelGGL((bn_fwd_apply_kernel<decltype(relu_t)::value,
                                            decltype(res_t)::value>),
                       dim3(nblocks), dim3(BN_BLOCK), 0, stream,
                       (const bf16x8*)x.data_ptr(), resp,
                       (bf16x8*)y.data_ptr(), scale.data_ptr<float>(),
                       shift.data_ptr<float>(), M, c8, rpb);
  };
  if (relu && residual) launch_apply(std::true_type{}, std::true_type{});
  else if (relu) launch_apply(std::true_type{}, std::false_type{});
  else if (residual) launch_apply(std::false_type{}, std::true_type{});
  else launch_apply(std::false_type{}, std::false_type{});
  return {y, mean, invstd};
}

at::Tensor bn_fwd_eval(at::Tensor x, c10::optional<at::Tensor> residual,
                       at::Tensor gamma, at::Tensor beta,
                       at::Tensor running_mean, at::Tensor running_var,
                       double eps, bool relu) {
  const long C = x.size(-1);
  const long M = x.numel() / C;
  const int c8 = (int)(C / 8);
  auto stream = at::hip::getCurrentHIPStream().stream();
  auto invstd = (running_var + eps).rsqrt();
  auto scale = (gamma * invstd).contiguous();
  auto shift = (beta - running_mean * scale).contiguous();
  auto y = at::empty_like(x);
  int rpb = pick_row_tiles(M, c8);
  int nblocks = (int)((M + rpb - 1) / rpb);
  const bf16x8* resp =
      residual ? (const bf16x8*)residual->data_ptr() : nullptr;
  auto launch_apply = [&](auto relu_t, auto res_t) {
    hipLaunchKernelGGL((bn_fwd_apply_kernel<decltype(relu_t)::value,
                                            decltype(res_t)::value>),
                       dim3(nblocks), dim3(BN_BLOCK), 0, stream,
                       (const bf16x8*)x.data_ptr(), resp,
                       (bf16x8*)y.data_ptr(), scale.data_ptr<float>(),
                       shift.data_ptr<float>(), M, c8, rpb);
  };
  if (relu && residual) launch_apply(std::true_type{}, std::true_type{});
  else if (relu) launch_apply(std::true_type{}, std::false_type{});
  else if (residual) launch_apply(std::false_type{}, std::true_type{});
  else launch_apply(std::false_type{}, std::false_type{});
  return y;
}

std::vector<at::Tensor> bn_bwd(at::Tensor x, at::Tensor dy, at::Tensor y,
                               at::Tensor mean, at::Tensor invstd,
                               at::Tensor gamma, bool relu, bool needs_dres) {
  const long C = x.size(-1);
  const long M = x.numel() / C;
  const int c8 = (int)(C / 8);
  auto stream = at::hip::getCurrentHIPStream().stream();
  auto opts = gamma.options();
  auto dbeta = at::empty({C}, opts);
  auto dgamma = at::empty({C}, opts);
  auto dx = at::empty_like(x);
  at::Tensor dres;
  bf16x8* dresp = nullptr;
  if (needs_dres) {
    dres = at::empty_like(x);
    dresp = (bf16x8*)dres.data_ptr();
  }
  int rpb = pick_row_tiles(M, c8);
  int nblocks = (int)((M + rpb - 1) / rpb);
  auto part = at::empty({(long)nblocks * 2 * C}, opts);
  auto launch_reduce = [&](auto relu_t, auto res_t) {
    hipLaunchKernelGGL((bn_bwd_reduce_kernel<decltype(relu_t)::value,
                                             decltype(res_t)::value>),
                       dim3(nblocks), dim3(BN_BLOCK), 0, stream,
                       (const bf16x8*)x.data_ptr(),
                       (const bf16x8*)dy.data_ptr(),
                       (const bf16x8*)y.data_ptr(), dresp,
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       part.data_ptr<float>(), M, c8, rpb);
  };
  if (relu && needs_dres) launch_reduce(std::true_type{}, std::true_type{});
  else if (relu) launch_reduce(std::true_type{}, std::false_type{});
  else if (needs_dres) launch_reduce(std::false_type{}, std::true_type{});
  else launch_reduce(std::false_type{}, std::false_type{});
  hipLaunchKernelGGL(bn_bwd_finalize_kernel, dim3((C + FIN_CH - 1) / FIN_CH),
                     dim3(FIN_PPAR * FIN_CH),
                     0, stream, part.data_ptr<float>(), nblocks,
                     dbeta.data_ptr<float>(), dgamma.data_ptr<float>(),
                     (int)C);
  auto launch_apply = [&](auto relu_t) {
    hipLaunchKernelGGL((bn_bwd_apply_kernel<decltype(relu_t)::value>),
                       dim3(nblocks), dim3(BN_BLOCK), 0, stream,
                       (const bf16x8*)x.data_ptr(),
                       (const bf16x8*)dy.data_ptr(),
                       (const bf16x8*)y.data_ptr(), (bf16x8*)dx.data_ptr(),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       gamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                       dgamma.data_ptr<float>(), M, c8, 1.f / (float)M,
                       rpb);
  };
  if (relu) launch_apply(std::true_type{});
  else launch_apply(std::false_type{});
  if (!needs_dres) dres = at::Tensor();
  return {dx, dgamma, dbeta, dres};
}
