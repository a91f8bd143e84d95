// Flash-attention forward (FA-2 style) for MI355X — ROUND-2 WORK IN
// PROGRESS, correctness-first v0.
//
// Status: compiles for gfx950 and is exercised ONLY by the env-gated tests
// in tests/test_fa_wip.py (STOKE_FA_TEST=1); nothing in the framework or
// benchmarks calls it yet.  The production path remains
// F.scaled_dot_product_attention (AOTriton).  See NOTES.md for the design
// and the round-2 plan (backward, LDS double-buffering, larger tiles).
//
// v0 design (one wave = 16 query rows; 4 independent waves per block):
//   * S = Q K^T via __builtin_amdgcn_mfma_f32_16x16x32_bf16 over head-dim
//     chunks of 32; KV walked in 32-column tiles (two 16x16 S quadrants).
//   * online softmax: per-lane running rowmax/rowsum for the wave's 4
//     C-rows, cross-lane reduced over the 16 lanes of each row group.
//   * P staged through LDS to convert the C-fragment layout into the
//     A-fragment layout for the P x V MFMAs.
//   * causal masking per element; GQA by head-index mapping.
//
// MFMA fragment-layout assumptions (validated by the mfma_probe_* tests
// before anything else — the guide defers exact A/B maps to the ISA doc):
//   A[16x32]: lane l holds row (l & 15), k = (l >> 4)*8 + e,  e = 0..7
//   B[32x16]: lane l holds col (l & 15), k = (l >> 4)*8 + e
//   C[16x16]: lane l holds col (l & 15), row (l >> 4)*4 + r,  r = 0..3

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

typedef __hip_bfloat16 bf16;
typedef __attribute__((ext_vector_type(8))) short bf16x8v;  // MFMA A/B frag
typedef __attribute__((ext_vector_type(4))) float f32x4v;   // MFMA C/D frag

__device__ __forceinline__ float fa_b2f(short v) {
  __hip_bfloat16_raw r;
  r.x = (unsigned short)v;
  return __bfloat162float(*reinterpret_cast<bf16*>(&r));
}

__device__ __forceinline__ short fa_f2b(float f) {
  bf16 h = __float2bfloat16(f);
  return (short)*reinterpret_cast<unsigned short*>(&h);
}

__device__ __forceinline__ bf16 fa_f2bf(float f) {
  return __float2bfloat16(f);
}

// ---------------------------------------------------------------------------
// Layout probe: D = A x B for one 16x16x32 MFMA with fragments loaded per the
// assumed lane maps.  The host test checks against a torch fp32 matmul with
// ASYMMETRIC operands, so a wrong map fails loudly (and tells round 2 what
// to fix before any attention debugging).
// ---------------------------------------------------------------------------
__global__ void mfma_probe_16x16x32(const bf16* __restrict__ A,  // [16,32] rm
                                    const bf16* __restrict__ B,  // [32,16] rm
                                    float* __restrict__ D) {     // [16,16] rm
  const int l = threadIdx.x;  // 64 lanes
  bf16x8v a, b;
  #pragma unroll
  for (int e = 0; e < 8; ++e) {
    const int k = (l >> 4) * 8 + e;
    a[e] = (short)*reinterpret_cast<const unsigned short*>(&A[(l & 15) * 32 + k]);
    b[e] = (short)*reinterpret_cast<const unsigned short*>(&B[k * 16 + (l & 15)]);
  }
  f32x4v c = {0.f, 0.f, 0.f, 0.f};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
  #pragma unroll
  for (int r = 0; r < 4; ++r)
    D[((l >> 4) * 4 + r) * 16 + (l & 15)] = c[r];
}

// ---------------------------------------------------------------------------
// FA forward v0.  q: [B,H,S,Dh], k/v: [B,Hkv,S,Dh] bf16 contiguous,
// Dh in {64, 128}; out: [B,H,S,Dh] bf16; lse: [B,H,S] fp32.
// Grid: (ceil(S/64), B*H); block 256 = 4 waves, wave w owns q rows
// [blk*64 + w*16, +16).
// ---------------------------------------------------------------------------
template <int DH>
__global__ __launch_bounds__(256) void fa_fwd_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, bf16* __restrict__ out,
    float* __restrict__ lse, int S, int H, int HKV, int causal) {
  constexpr int DC = DH / 32;  // head-dim chunks per MFMA K
  const int wave = threadIdx.x >> 6;
  const int l = threadIdx.x & 63;
  const int row0 = blockIdx.x * 64 + wave * 16;  // first q row of this wave
  const int bh = blockIdx.y;
  const int h = bh % H;
  const int hkv = h / (H / HKV);
  const int b = bh / H;
  // NOTE: no early return — every wave must reach the block barriers below;
  // out-of-range rows are clamped and their stores guarded.
  const bool wave_active = row0 < S;
  const bf16* qp = q + (((long)b * H + h) * S) * DH;
  const bf16* kp = k + (((long)b * HKV + hkv) * S) * DH;
  const bf16* vp = v + (((long)b * HKV + hkv) * S) * DH;

  // Q fragments for this wave's 16 rows, kept in registers for the whole
  // KV sweep.  Lane l -> row (l&15); rows past S replicate the last valid
  // row (their outputs are never stored).
  const int qrow = min(min(row0 + (l & 15), S - 1), S - 1);
  bf16x8v qfrag[DC];
  #pragma unroll
  for (int c = 0; c < DC; ++c) {
    const bf16* src = qp + (long)qrow * DH + c * 32 + (l >> 4) * 8;
    qfrag[c] = *reinterpret_cast<const bf16x8v*>(src);
  }

  float m[4], lsum[4];
  #pragma unroll
  for (int r = 0; r < 4; ++r) {
    m[r] = -1e30f;
    lsum[r] = 0.f;
  }
  f32x4v oacc[DH / 16];
  #pragma unroll
  for (int f = 0; f < DH / 16; ++f) oacc[f] = {0.f, 0.f, 0.f, 0.f};

  // LDS: per-wave P tile [16 rows][32 kcols] bf16
  __shared__ bf16 p_lds[4][16][32];

  // All waves in the block iterate the same KV range (to the block's last
  // row under causal masking) so __syncthreads stays uniform.
  const int block_last_row = min(blockIdx.x * 64 + 63, S - 1);
  const int kv_end = causal ? (block_last_row + 1) : S;
  for (int kb = 0; kb < kv_end; kb += 32) {
    // ---- S quadrants: Sq[16][16] for kcol halves 0 and 1
    float srows[2][4];  // [quadrant][C row] for this lane
    #pragma unroll
    for (int qd = 0; qd < 2; ++qd) {
      f32x4v acc = {0.f, 0.f, 0.f, 0.f};
      const int kcol = min(kb + qd * 16 + (l & 15), S - 1);
      #pragma unroll
      for (int c = 0; c < DC; ++c) {
        bf16x8v bfrag;
        const bf16* src = kp + (long)kcol * DH + c * 32 + (l >> 4) * 8;
        bfrag = *reinterpret_cast<const bf16x8v*>(src);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[c], bfrag, acc,
                                                      0, 0, 0);
      }
      const float scale = rsqrtf((float)DH);
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int rr = row0 + (l >> 4) * 4 + r;      // this C-elem's q row
        const int cc = kb + qd * 16 + (l & 15);      // its k column
        float s = acc[r] * scale;
        if (cc >= S || rr >= S || (causal && cc > rr)) s = -1e30f;
        srows[qd][r] = s;
      }
    }
    // ---- online softmax: row max over the 16 lanes of each row group
    float mnew[4];
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      float mx = fmaxf(srows[0][r], srows[1][r]);
      #pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        mx = fmaxf(mx, __shfl_xor(mx, off, 16));
      mnew[r] = fmaxf(m[r], mx);
    }
    float alpha[4];
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      alpha[r] = __expf(m[r] - mnew[r]);
      m[r] = mnew[r];
    }
    // ---- P = exp(S - m); row sums; stage P into LDS in A-layout order
    #pragma unroll
    for (int qd = 0; qd < 2; ++qd) {
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        float p = __expf(srows[qd][r] - m[r]);
        srows[qd][r] = p;
        p_lds[wave][(l >> 4) * 4 + r][qd * 16 + (l & 15)] = fa_f2bf(p);
      }
    }
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      float ps = srows[0][r] + srows[1][r];
      #pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        ps += __shfl_xor(ps, off, 16);
      lsum[r] = lsum[r] * alpha[r] + ps;
    }
    __syncthreads();
    // ---- O = O*alpha + P x V  (P from LDS in A layout; V B-frags global)
    bf16x8v pfrag;
    #pragma unroll
    for (int e = 0; e < 8; ++e)
      pfrag[e] = (short)*reinterpret_cast<unsigned short*>(
          &p_lds[wave][l & 15][(l >> 4) * 8 + e]);
    #pragma unroll
    for (int f = 0; f < DH / 16; ++f) {
      #pragma unroll
      for (int r = 0; r < 4; ++r) oacc[f][r] *= alpha[r];
      bf16x8v vfrag;
      #pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int krow = min(kb + (l >> 4) * 8 + e, S - 1);
        vfrag[e] = (short)*reinterpret_cast<const unsigned short*>(
            &vp[(long)krow * DH + f * 16 + (l & 15)]);
      }
      oacc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pfrag, vfrag, oacc[f],
                                                        0, 0, 0);
    }
    __syncthreads();
  }
  // ---- epilogue: O /= lsum; store O (bf16) and logsumexp (fp32)
  #pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int rr = row0 + (l >> 4) * 4 + r;
    if (!wave_active || rr >= S) continue;
    const float inv = 1.f / fmaxf(lsum[r], 1e-30f);
    #pragma unroll
    for (int f = 0; f < DH / 16; ++f) {
      out[(((long)b * H + h) * S + rr) * DH + f * 16 + (l & 15)] =
          fa_f2bf(oacc[f][r] * inv);
    }
    if ((l & 15) == 0)
      lse[((long)b * H + h) * S + rr] = m[r] + __logf(fmaxf(lsum[r], 1e-30f));
  }
}

}  // namespace

void mfma_probe(at::Tensor A, at::Tensor B, at::Tensor D) {
  auto stream = at::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(mfma_probe_16x16x32, dim3(1), dim3(64), 0, stream,
                     (const bf16*)A.data_ptr(), (const bf16*)B.data_ptr(),
                     D.data_ptr<float>());
}

std::vector<at::Tensor> fa_fwd(at::Tensor q, at::Tensor k, at::Tensor v,
                               bool causal) {
  TORCH_CHECK(q.scalar_type() == at::kBFloat16 && q.dim() == 4 &&
                  q.is_contiguous() && k.is_contiguous() && v.is_contiguous(),
              "fa_fwd: contiguous [B,H,S,D] bf16");
  const int B = q.size(0), H = q.size(1), S = q.size(2), DH = q.size(3);
  const int HKV = k.size(1);
  TORCH_CHECK(DH == 64 || DH == 128, "fa_fwd v0: head dim 64 or 128");
  TORCH_CHECK(H % HKV == 0, "fa_fwd: H must be a multiple of H_kv");
  auto out = at::empty_like(q);
  auto lse = at::empty({B, H, S}, q.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream().stream();
  dim3 grid((S + 63) / 64, B * H);
  if (DH == 128) {
    hipLaunchKernelGGL((fa_fwd_kernel<128>), grid, dim3(256), 0, stream,
                       (const bf16*)q.data_ptr(), (const bf16*)k.data_ptr(),
                       (const bf16*)v.data_ptr(), (bf16*)out.data_ptr(),
                       lse.data_ptr<float>(), S, H, HKV, causal ? 1 : 0);
  } else {
    hipLaunchKernelGGL((fa_fwd_kernel<64>), grid, dim3(256), 0, stream,
                       (const bf16*)q.data_ptr(), (const bf16*)k.data_ptr(),
                       (const bf16*)v.data_ptr(), (bf16*)out.data_ptr(),
                       lse.data_ptr<float>(), S, H, HKV, causal ? 1 : 0);
  }
  return {out, lse};
}
