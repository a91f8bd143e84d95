// Flash-attention backward (FA-2 split) for MI355X — ROUND-2 WORK IN
// PROGRESS, correctness-first v0.  Same status and gating as fa_fwd.hip:
// compiles for gfx950, exercised only by STOKE_FA_TEST=1 tests, nothing in
// the framework calls it.  Uses the identical MFMA fragment-layout
// assumptions (validated first by the mfma_probe test).
//
// Standard FA-2 decomposition with saved logsumexp L and
// delta = rowsum(dO * O):
//   P    = exp(S*scale - L)
//   dV  += P^T dO
//   dP   = dO V^T;   dS = P * (dP - delta) * scale
//   dK  += dS^T Q;   dQ += dS K
// Two kernels: fa_bwd_dkv (grid over KV 16-row waves, streams Q/dO tiles)
// and fa_bwd_dq (grid over Q 16-row waves, streams K/V tiles); plus a tiny
// delta kernel.  No atomics: each wave owns its dK/dV (resp. dQ) rows.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

typedef __hip_bfloat16 bf16;
typedef __attribute__((ext_vector_type(8))) short bf16x8v;
typedef __attribute__((ext_vector_type(4))) float f32x4v;

__device__ __forceinline__ bf16 fb_f2bf(float f) {
  return __float2bfloat16(f);
}

__device__ __forceinline__ float fb_b2f(const bf16& h) {
  return __bfloat162float(h);
}

// delta[b,h,s] = sum_d dO[b,h,s,d] * O[b,h,s,d]  (fp32)
__global__ __launch_bounds__(256) void fa_delta_kernel(
    const bf16* __restrict__ dout, const bf16* __restrict__ out,
    float* __restrict__ delta, long rows, int DH) {
  // one wave per row, lanes stride the head dim
  const long row = (long)blockIdx.x * 4 + (threadIdx.x >> 6);
  const int l = threadIdx.x & 63;
  if (row >= rows) return;
  float s = 0.f;
  for (int d = l; d < DH; d += 64)
    s += fb_b2f(dout[row * DH + d]) * fb_b2f(out[row * DH + d]);
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1) s += __shfl_down(s, off);
  if (l == 0) delta[row] = s;
}

// ---------------------------------------------------------------------------
// dK/dV kernel: wave owns 16 KV rows; iterates q tiles of 32 (from the
// causal diagonal onward).  Works in the transposed frame: C rows = kv,
// C cols = q.
// ---------------------------------------------------------------------------
template <int DH>
__global__ __launch_bounds__(256) void fa_bwd_dkv_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, const bf16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    float* __restrict__ dk, float* __restrict__ dv,  // fp32 accum buffers
    int S, int H, int HKV, int causal) {
  constexpr int DC = DH / 32;
  const int wave = threadIdx.x >> 6;
  const int l = threadIdx.x & 63;
  const int krow0 = blockIdx.x * 64 + wave * 16;  // first KV row
  const int bh = blockIdx.y;                       // over B*H (q heads!)
  const int h = bh % H;
  const int hkv = h / (H / HKV);
  const int b = bh / H;
  const bool wave_active = krow0 < S;
  const bf16* qp = q + (((long)b * H + h) * S) * DH;
  const bf16* kp = k + (((long)b * HKV + hkv) * S) * DH;
  const bf16* vp = v + (((long)b * HKV + hkv) * S) * DH;
  const bf16* dop = dout + (((long)b * H + h) * S) * DH;
  const float* lp = lse + ((long)b * H + h) * S;
  const float* dp = delta + ((long)b * H + h) * S;
  // dK/dV accumulate per (b, q-head): summed over q-head groups on the
  // host for GQA.  Layout: [B, H, S, DH] fp32.
  float* dkp = dk + (((long)b * H + h) * S) * DH;
  float* dvp = dv + (((long)b * H + h) * S) * DH;

  // K and V fragments for this wave's 16 kv rows (A-layout: row = l&15)
  const int krow = min(krow0 + (l & 15), S - 1);
  bf16x8v kfrag[DC], vfrag[DC];
  #pragma unroll
  for (int c = 0; c < DC; ++c) {
    kfrag[c] = *reinterpret_cast<const bf16x8v*>(
        kp + (long)krow * DH + c * 32 + (l >> 4) * 8);
    vfrag[c] = *reinterpret_cast<const bf16x8v*>(
        vp + (long)krow * DH + c * 32 + (l >> 4) * 8);
  }
  f32x4v dkacc[DH / 16], dvacc[DH / 16];
  #pragma unroll
  for (int f = 0; f < DH / 16; ++f) {
    dkacc[f] = {0.f, 0.f, 0.f, 0.f};
    dvacc[f] = {0.f, 0.f, 0.f, 0.f};
  }
  __shared__ bf16 st_lds[4][16][32];  // staged P^T / dS^T tiles per wave

  const float scale = rsqrtf((float)DH);
  // causal: q tiles start at the block's first kv row
  const int q0 = causal ? ((blockIdx.x * 64) & ~31) : 0;
  for (int qb = q0; qb < S; qb += 32) {
    // ---- S^T quadrants: rows = kv (this wave's 16), cols = q tile half
    float pt[2][4];   // P^T values for this lane's 4 C rows
    float dpt[2][4];  // dP^T values
    #pragma unroll
    for (int qd = 0; qd < 2; ++qd) {
      const int qcol = min(qb + qd * 16 + (l & 15), S - 1);
      f32x4v sacc = {0.f, 0.f, 0.f, 0.f};
      f32x4v dpacc = {0.f, 0.f, 0.f, 0.f};
      #pragma unroll
      for (int c = 0; c < DC; ++c) {
        // B-frag from Q rows (S^T = K Q^T) and dO rows (dP^T = V dO^T)
        bf16x8v qb_frag = *reinterpret_cast<const bf16x8v*>(
            qp + (long)qcol * DH + c * 32 + (l >> 4) * 8);
        bf16x8v dob_frag = *reinterpret_cast<const bf16x8v*>(
            dop + (long)qcol * DH + c * 32 + (l >> 4) * 8);
        sacc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(kfrag[c], qb_frag,
                                                       sacc, 0, 0, 0);
        dpacc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(vfrag[c], dob_frag,
                                                        dpacc, 0, 0, 0);
      }
      const float L = lp[min(qb + qd * 16 + (l & 15), S - 1)];
      const float dl = dp[min(qb + qd * 16 + (l & 15), S - 1)];
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int kr = krow0 + (l >> 4) * 4 + r;  // kv row of this C elem
        const int qc = qb + qd * 16 + (l & 15);   // q col
        float p = 0.f;
        if (qc < S && kr < S && (!causal || kr <= qc))
          p = __expf(sacc[r] * scale - L);
        pt[qd][r] = p;
        dpt[qd][r] = p * (dpacc[r] - dl) * scale;  // = dS^T element
      }
    }
    // ---- dV += P^T x dO : stage P^T, MFMA against dO rows
    #pragma unroll
    for (int qd = 0; qd < 2; ++qd)
      #pragma unroll
      for (int r = 0; r < 4; ++r)
        st_lds[wave][(l >> 4) * 4 + r][qd * 16 + (l & 15)] =
            fb_f2bf(pt[qd][r]);
    __syncthreads();
    bf16x8v afrag;
    #pragma unroll
    for (int e = 0; e < 8; ++e)
      afrag[e] = (short)*reinterpret_cast<unsigned short*>(
          &st_lds[wave][l & 15][(l >> 4) * 8 + e]);
    #pragma unroll
    for (int f = 0; f < DH / 16; ++f) {
      bf16x8v bfrag;
      #pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int qr = min(qb + (l >> 4) * 8 + e, S - 1);
        bfrag[e] = (short)*reinterpret_cast<const unsigned short*>(
            &dop[(long)qr * DH + f * 16 + (l & 15)]);
      }
      dvacc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag,
                                                         dvacc[f], 0, 0, 0);
    }
    __syncthreads();
    // ---- dK += dS^T x Q : stage dS^T, MFMA against Q rows
    #pragma unroll
    for (int qd = 0; qd < 2; ++qd)
      #pragma unroll
      for (int r = 0; r < 4; ++r)
        st_lds[wave][(l >> 4) * 4 + r][qd * 16 + (l & 15)] =
            fb_f2bf(dpt[qd][r]);
    __syncthreads();
    #pragma unroll
    for (int e = 0; e < 8; ++e)
      afrag[e] = (short)*reinterpret_cast<unsigned short*>(
          &st_lds[wave][l & 15][(l >> 4) * 8 + e]);
    #pragma unroll
    for (int f = 0; f < DH / 16; ++f) {
      bf16x8v bfrag;
      #pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int qr = min(qb + (l >> 4) * 8 + e, S - 1);
        bfrag[e] = (short)*reinterpret_cast<const unsigned short*>(
            &qp[(long)qr * DH + f * 16 + (l & 15)]);
      }
      dkacc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag,
                                                         dkacc[f], 0, 0, 0);
    }
    __syncthreads();
  }
  // epilogue: C rows = kv rows
  #pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int kr = krow0 + (l >> 4) * 4 + r;
    if (!wave_active || kr >= S) continue;
    #pragma unroll
    for (int f = 0; f < DH / 16; ++f) {
      dkp[(long)kr * DH + f * 16 + (l & 15)] = dkacc[f][r];
      dvp[(long)kr * DH + f * 16 + (l & 15)] = dvacc[f][r];
    }
  }
}

// ---------------------------------------------------------------------------
// dQ kernel: wave owns 16 q rows; iterates KV tiles of 32 up to the causal
// bound.  Mirrors the forward's structure with the dS x K product.
// ---------------------------------------------------------------------------
template <int DH>
__global__ __launch_bounds__(256) void fa_bwd_dq_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, const bf16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    bf16* __restrict__ dq, int S, int H, int HKV, int causal) {
  constexpr int DC = DH / 32;
  const int wave = threadIdx.x >> 6;
  const int l = threadIdx.x & 63;
  const int row0 = blockIdx.x * 64 + wave * 16;
  const int bh = blockIdx.y;
  const int h = bh % H;
  const int hkv = h / (H / HKV);
  const int b = bh / H;
  const bool wave_active = row0 < S;
  const bf16* qp = q + (((long)b * H + h) * S) * DH;
  const bf16* kp = k + (((long)b * HKV + hkv) * S) * DH;
  const bf16* vp = v + (((long)b * HKV + hkv) * S) * DH;
  const bf16* dop = dout + (((long)b * H + h) * S) * DH;
  const float* lp = lse + ((long)b * H + h) * S;
  const float* dp = delta + ((long)b * H + h) * S;

  const int qrow = min(row0 + (l & 15), S - 1);
  bf16x8v qfrag[DC], dofrag[DC];
  #pragma unroll
  for (int c = 0; c < DC; ++c) {
    qfrag[c] = *reinterpret_cast<const bf16x8v*>(
        qp + (long)qrow * DH + c * 32 + (l >> 4) * 8);
    dofrag[c] = *reinterpret_cast<const bf16x8v*>(
        dop + (long)qrow * DH + c * 32 + (l >> 4) * 8);
  }
  f32x4v dqacc[DH / 16];
  #pragma unroll
  for (int f = 0; f < DH / 16; ++f) dqacc[f] = {0.f, 0.f, 0.f, 0.f};
  __shared__ bf16 ds_lds[4][16][32];

  const float scale = rsqrtf((float)DH);
  const int block_last_row = min(blockIdx.x * 64 + 63, S - 1);
  const int kv_end = causal ? (block_last_row + 1) : S;
  for (int kb = 0; kb < kv_end; kb += 32) {
    float dsv[2][4];
    #pragma unroll
    for (int qd = 0; qd < 2; ++qd) {
      const int kcol = min(kb + qd * 16 + (l & 15), S - 1);
      f32x4v sacc = {0.f, 0.f, 0.f, 0.f};
      f32x4v dpacc = {0.f, 0.f, 0.f, 0.f};
      #pragma unroll
      for (int c = 0; c < DC; ++c) {
        bf16x8v kb_frag = *reinterpret_cast<const bf16x8v*>(
            kp + (long)kcol * DH + c * 32 + (l >> 4) * 8);
        bf16x8v vb_frag = *reinterpret_cast<const bf16x8v*>(
            vp + (long)kcol * DH + c * 32 + (l >> 4) * 8);
        sacc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[c], kb_frag,
                                                       sacc, 0, 0, 0);
        dpacc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dofrag[c], vb_frag,
                                                        dpacc, 0, 0, 0);
      }
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int rr = row0 + (l >> 4) * 4 + r;
        const int cc = kb + qd * 16 + (l & 15);
        float p = 0.f;
        if (cc < S && rr < S && (!causal || cc <= rr))
          p = __expf(sacc[r] * scale - lp[min(rr, S - 1)]);
        dsv[qd][r] = p * (dpacc[r] - dp[min(rr, S - 1)]) * scale;
      }
    }
    // stage dS, then dQ += dS x K
    #pragma unroll
    for (int qd = 0; qd < 2; ++qd)
      #pragma unroll
      for (int r = 0; r < 4; ++r)
        ds_lds[wave][(l >> 4) * 4 + r][qd * 16 + (l & 15)] =
            fb_f2bf(dsv[qd][r]);
    __syncthreads();
    bf16x8v afrag;
    #pragma unroll
    for (int e = 0; e < 8; ++e)
      afrag[e] = (short)*reinterpret_cast<unsigned short*>(
          &ds_lds[wave][l & 15][(l >> 4) * 8 + e]);
    #pragma unroll
    for (int f = 0; f < DH / 16; ++f) {
      bf16x8v bfrag;
      #pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int kr = min(kb + (l >> 4) * 8 + e, S - 1);
        bfrag[e] = (short)*reinterpret_cast<const unsigned short*>(
            &kp[(long)kr * DH + f * 16 + (l & 15)]);
      }
      dqacc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag,
                                                         dqacc[f], 0, 0, 0);
    }
    __syncthreads();
  }
  #pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int rr = row0 + (l >> 4) * 4 + r;
    if (!wave_active || rr >= S) continue;
    #pragma unroll
    for (int f = 0; f < DH / 16; ++f)
      dq[(((long)b * H + h) * S + rr) * DH + f * 16 + (l & 15)] =
          fb_f2bf(dqacc[f][r]);
  }
}

}  // namespace

std::vector<at::Tensor> fa_bwd(at::Tensor q, at::Tensor k, at::Tensor v,
                               at::Tensor out, at::Tensor dout,
                               at::Tensor lse, bool causal) {
  const int B = q.size(0), H = q.size(1), S = q.size(2), DH = q.size(3);
  const int HKV = k.size(1);
  TORCH_CHECK(DH == 64 || DH == 128, "fa_bwd v0: head dim 64 or 128");
  auto stream = at::hip::getCurrentHIPStream().stream();
  auto delta = at::empty({B, H, S}, q.options().dtype(at::kFloat));
  const long rows = (long)B * H * S;
  hipLaunchKernelGGL(fa_delta_kernel, dim3((unsigned)((rows + 3) / 4)),
                     dim3(256), 0, stream, (const bf16*)dout.data_ptr(),
                     (const bf16*)out.data_ptr(), delta.data_ptr<float>(),
                     rows, DH);
  auto dq = at::empty_like(q);
  // per-q-head fp32 dk/dv; GQA groups summed below
  auto dk_full = at::empty({B, H, S, DH}, q.options().dtype(at::kFloat));
  auto dv_full = at::empty({B, H, S, DH}, q.options().dtype(at::kFloat));
  dim3 grid((S + 63) / 64, B * H);
  if (DH == 128) {
    hipLaunchKernelGGL((fa_bwd_dkv_kernel<128>), grid, dim3(256), 0, stream,
                       (const bf16*)q.data_ptr(), (const bf16*)k.data_ptr(),
                       (const bf16*)v.data_ptr(), (const bf16*)dout.data_ptr(),
                       lse.data_ptr<float>(), delta.data_ptr<float>(),
                       dk_full.data_ptr<float>(), dv_full.data_ptr<float>(),
                       S, H, HKV, causal ? 1 : 0);
    hipLaunchKernelGGL((fa_bwd_dq_kernel<128>), grid, dim3(256), 0, stream,
                       (const bf16*)q.data_ptr(), (const bf16*)k.data_ptr(),
                       (const bf16*)v.data_ptr(), (const bf16*)dout.data_ptr(),
                       lse.data_ptr<float>(), delta.data_ptr<float>(),
                       (bf16*)dq.data_ptr(), S, H, HKV, causal ? 1 : 0);
  } else {
    hipLaunchKernelGGL((fa_bwd_dkv_kernel<64>), grid, dim3(256), 0, stream,
                       (const bf16*)q.data_ptr(), (const bf16*)k.data_ptr(),
                       (const bf16*)v.data_ptr(), (const bf16*)dout.data_ptr(),
                       lse.data_ptr<float>(), delta.data_ptr<float>(),
                       dk_full.data_ptr<float>(), dv_full.data_ptr<float>(),
                       S, H, HKV, causal ? 1 : 0);
    hipLaunchKernelGGL((fa_bwd_dq_kernel<64>), grid, dim3(256), 0, stream,
                       (const bf16*)q.data_ptr(), (const bf16*)k.data_ptr(),
                       (const bf16*)v.data_ptr(), (const bf16*)dout.data_ptr(),
                       lse.data_ptr<float>(), delta.data_ptr<float>(),
                       (bf16*)dq.data_ptr(), S, H, HKV, causal ? 1 : 0);
  }
  // GQA: sum q-head groups back to the kv heads (torch op; v0 simplicity)
  auto dk = dk_full.view({B, HKV, H / HKV, S, DH}).sum(2).to(q.dtype());
  auto dv = dv_full.view({B, HKV, H / HKV, S, DH}).sum(2).to(q.dtype());
  return {dq, dk, dv};
}
