// Flash-attention forward (FA-2 style) for MI355X gfx950 — v1.
//
// v1 follows the measured CDNA4 attention structure from the platform
// guide's ladder (8-wave 32x32 MFMA, swapped QK^T so softmax is lane-local,
// XOR-swizzled K image, transposed V image, in-register P->bf16):
//   * block = 8 waves x 32 q rows = 256-row Q tile; KV walked in 64-row
//     tiles staged in LDS once per block (shared by all 8 waves).
//   * S^T = K Q^T via v_mfma_f32_32x32x16_bf16: the C fragment then holds,
//     per lane, one q COLUMN and 16 kv rows -> the softmax row-reduce is
//     15 in-lane fmax + one lane<->lane+32 exchange, no LDS round trip.
//   * K tile LDS image is row-major with a 16-B-slot XOR swizzle
//     (slot ^= row&15) so the A-fragment ds_read_b128 is bank-conflict-free.
//   * V is staged TRANSPOSED ([d][kv], padded) so the PV A-fragment
//     (V^T[d][kv]) is a contiguous b128 read; O accumulates as O^T[d][q].
//   * P (f32, C layout) is repacked to bf16 A/B fragments in registers:
//     v_cvt_pk_bf16_f32 pairs + one __shfl_xor(32) half-exchange per pair
//     of dwords — no LDS staging of P.
//
// Fragment maps (validated on hardware by tests/test_fa.py probes):
//   16x16x32: A row l&15, k=(l>>4)*8+e; B col l&15 same k;
//             C col l&15, row (l>>4)*4+r.
//   32x32x16: A row l&31, k=(l>>5)*8+e; B col l&31 same k;
//             C col l&31, row (r&3)+8*(r>>2)+4*(l>>5), r=0..15.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

typedef __hip_bfloat16 bf16;
typedef __attribute__((ext_vector_type(8))) short bf16x8v;   // MFMA A/B frag
typedef __attribute__((ext_vector_type(4))) float f32x4v;    // 16x16 C frag
typedef __attribute__((ext_vector_type(16))) float f32x16v;  // 32x32 C frag

__device__ __forceinline__ bf16 fa_f2bf(float f) {
  return __float2bfloat16(f);
}

__device__ __forceinline__ float fa_bf2f(short v) {
  __hip_bfloat16_raw r;
  r.x = (unsigned short)v;
  return __bfloat162float(*reinterpret_cast<bf16*>(&r));
}

// pack two f32 -> two bf16 in one dword (one v_cvt_pk_bf16_f32 — the
// scalar cvt+or sequence costs 3-4 VALU ops per dword; guide T12)
__device__ __forceinline__ unsigned int pk_bf16(float lo, float hi) {
  unsigned int r;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

// ---------------------------------------------------------------------------
// Layout probes (host tests check against torch matmuls with ASYMMETRIC
// operands, per the guide: symmetric inputs hide row/col swaps).
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

__global__ void mfma_probe_32x32x16(const bf16* __restrict__ A,  // [32,16] rm
                                    const bf16* __restrict__ B,  // [16,32] rm
                                    float* __restrict__ D) {     // [32,32] rm
  const int l = threadIdx.x;  // 64 lanes
  bf16x8v a, b;
  #pragma unroll
  for (int e = 0; e < 8; ++e) {
    const int k = (l >> 5) * 8 + e;
    a[e] = (short)*reinterpret_cast<const unsigned short*>(&A[(l & 31) * 16 + k]);
    b[e] = (short)*reinterpret_cast<const unsigned short*>(&B[k * 32 + (l & 31)]);
  }
  f32x16v c;
  #pragma unroll
  for (int r = 0; r < 16; ++r) c[r] = 0.f;
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
  #pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5);
    D[row * 32 + (l & 31)] = c[r];
  }
}

// ---------------------------------------------------------------------------
// FA forward v1.  q: [B,H,S,Dh], k/v: [B,Hkv,S,Dh] bf16 contiguous,
// Dh in {64, 128}; out: [B,H,S,Dh] bf16; lse: [B,H,S] fp32.
// Grid: (ceil(S/256), B*H); block 512 = 8 waves, wave w owns q rows
// [blk*256 + w*32, +32).
// ---------------------------------------------------------------------------
#define FA_KVB 64  // kv rows per staged tile
#define FA_VPAD 8  // V^T row padding (elements) to spread write banks

// Strided addressing: every tensor is a 4-D view with unit stride on the
// head dim; (sb, sh, ss) are the batch/head/row strides in ELEMENTS, so
// both [B,H,S,D]-contiguous and [B,S,H,D]-contiguous (attention's natural
// projection layout — no transpose copies) run the same kernel.
template <int DH>
__global__ __launch_bounds__(512) void fa_fwd_v1(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, bf16* __restrict__ out,
    float* __restrict__ lse, int S, int H, int HKV, int causal,
    long qsb, long qsh, long qss, long ksb, long ksh, long kss,
    long vsb, long vsh, long vss, long osb, long osh, long oss) {
  constexpr int DCH = DH / 16;    // 16-wide k-chunks of the head dim
  constexpr int DT = DH / 32;     // 32-row d-tiles of O^T
  constexpr int KSLOT = DH / 8;   // 16-B slots per K row
  const int wave = threadIdx.x >> 6;
  const int l = threadIdx.x & 63;
  const int lq = l & 31;          // this lane's q column / d row
  const int h2 = l >> 5;          // lane half (0: lanes 0-31, 1: 32-63)
  const int bh = blockIdx.y;
  const int h = bh % H;
  const int hkv = h / (H / HKV);
  const int b = bh / H;

  const bf16* qp = q + (long)b * qsb + (long)h * qsh;
  const bf16* kp = k + (long)b * ksb + (long)hkv * ksh;
  const bf16* vp = v + (long)b * vsb + (long)hkv * vsh;

  // K image: row-major [KVB][DH], 16-B slot index XOR-swizzled with row&15.
  // V image: transposed [DH][KVB + pad].
  __shared__ bf16 kbuf[FA_KVB * DH];
  __shared__ bf16 vbuf[DH * (FA_KVB + FA_VPAD)];

  // Causal load balance: each block handles q strip b and its mirror
  // (cf. fa_bwd; the kv sweep of strip 0 is 16x shorter than strip max)
  const int nstrip = (S + 255) / 256;
  for (int halfi = 0; halfi < 2; ++halfi) {
  const int strip = halfi == 0 ? (int)blockIdx.x
                               : nstrip - 1 - (int)blockIdx.x;
  if (halfi == 1 && (!causal || strip <= (int)blockIdx.x)) break;
  const int qbase = strip * 256;
  const int row0 = qbase + wave * 32;          // wave's first q row
  const int qrow = min(row0 + lq, S - 1);      // this lane's q row (clamped)

  // Q as B-operand fragments, resident for the whole KV sweep:
  // frag ch: lane holds q col lq, k-elems d = ch*16 + h2*8 + e.
  bf16x8v qf[DCH];
  #pragma unroll
  for (int ch = 0; ch < DCH; ++ch)
    qf[ch] = *reinterpret_cast<const bf16x8v*>(
        qp + (long)qrow * qss + ch * 16 + h2 * 8);

  f32x16v oacc[DT];
  #pragma unroll
  for (int dt = 0; dt < DT; ++dt)
    #pragma unroll
    for (int r = 0; r < 16; ++r) oacc[dt][r] = 0.f;
  float m = -1e30f, lsum = 0.f;
  // softmax runs in the exp2 domain (v_exp_f32 IS base-2; folding log2e
  // into the scale removes one v_mul per exponential)
  const float scale = rsqrtf((float)DH) * 1.44269504089f;

  // ---- tile staging (T14 register split: global loads for tile t+1 are
  // issued during tile t's compute; the LDS writes happen after the
  // end-of-tile barrier).  Thread halves specialize: tid<256 stages K
  // (b128 loads -> swizzled b128 LDS writes), tid>=256 stages V
  // (4 kv-rows of one 8-wide d-chunk -> 8 ds_write_b64 into the
  // transposed image; 16 lanes share a d-row so write banks are distinct).
  constexpr int KPIECE = FA_KVB * KSLOT / 256;  // K b128 pieces per thread
  const int tid = threadIdx.x;
  const bool is_k = tid < 256;
  const int ka = tid & 255;
  const int v_dchunk = ka >> 4;       // V: which 8-wide d chunk
  const int v_kvq = ka & 15;          // V: which 4-row kv quad
  const bool v_active = !is_k && v_dchunk < DH / 8;
  bf16x8v stage[4];                   // K: KPIECE pieces; V: 4 kv rows

  auto load_tile = [&](int kb) {
    if (is_k) {
      #pragma unroll
      for (int i = 0; i < KPIECE; ++i) {
        const int p = ka * KPIECE + i;
        const int row = p / KSLOT;
        const int slot = p % KSLOT;
        const int grow = min(kb + row, S - 1);
        stage[i] = *reinterpret_cast<const bf16x8v*>(
            kp + (long)grow * kss + slot * 8);
      }
    } else if (v_active) {
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int grow = min(kb + v_kvq * 4 + i, S - 1);
        stage[i] = *reinterpret_cast<const bf16x8v*>(
            vp + (long)grow * vss + v_dchunk * 8);
      }
    }
  };

  auto write_tile = [&]() {
    if (is_k) {
      #pragma unroll
      for (int i = 0; i < KPIECE; ++i) {
        const int p = ka * KPIECE + i;
        const int row = p / KSLOT;
        const int slot = p % KSLOT;
        const int sslot = slot ^ (row & (KSLOT - 1));
        *reinterpret_cast<bf16x8v*>(&kbuf[row * DH + sslot * 8]) = stage[i];
      }
    } else if (v_active) {
      // stage[i] holds V[kv0+i][d0..d0+7]; emit b64 of 4 kv values per d
      union { bf16x8v v8[4]; short s[4][8]; } u;
      #pragma unroll
      for (int i = 0; i < 4; ++i) u.v8[i] = stage[i];
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        union { short s4[4]; unsigned long long d; } pack;
        #pragma unroll
        for (int i = 0; i < 4; ++i) pack.s4[i] = u.s[i][j];
        *reinterpret_cast<unsigned long long*>(
            &vbuf[(v_dchunk * 8 + j) * (FA_KVB + FA_VPAD) + v_kvq * 4]) =
            pack.d;
      }
    }
  };

  const int kv_end = causal ? min(qbase + 256, S) : S;
  load_tile(0);
  write_tile();
  __syncthreads();
  for (int kb = 0; kb < kv_end; kb += FA_KVB) {

    // ---- S^T tiles: sacc[ct] = K[ct*32..+32] x Q^T  (C: lane=q col,
    // 16 kv rows each); mask + scale in-register.
    float p32[32];  // this lane's P values, kv = ct*32 + (r&3)+8*(r>>2)+4*h2
    float tmax = -1e30f;
    #pragma unroll
    for (int ct = 0; ct < 2; ++ct) {
      f32x16v sacc;
      #pragma unroll
      for (int r = 0; r < 16; ++r) sacc[r] = 0.f;
      __builtin_amdgcn_s_setprio(1);
      #pragma unroll
      for (int ch = 0; ch < DCH; ++ch) {
        // A-frag: K row kv = ct*32 + lq, k-elems d = ch*16 + h2*8 + e
        const int krow = ct * 32 + lq;
        const int slot = (ch * 2 + h2) ^ (krow & (KSLOT - 1));
        bf16x8v kf = *reinterpret_cast<const bf16x8v*>(
            &kbuf[krow * DH + slot * 8]);
        sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[ch], sacc,
                                                       0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);
      #pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kvg = kb + ct * 32 + (r & 3) + 8 * (r >> 2) + 4 * h2;
        const int qg = row0 + lq;
        float s = sacc[r] * scale;
        if (kvg >= S || qg >= S || (causal && kvg > qg)) s = -1e30f;
        p32[ct * 16 + r] = s;
        tmax = fmaxf(tmax, s);
      }
    }
    // issue next tile's global loads now — they complete under the
    // softmax + PV compute below (T14 async-stage split)
    if (kb + FA_KVB < kv_end) load_tile(kb + FA_KVB);

    // ---- online softmax (lane-local rows; combine lane<->lane+32).
    // defer-max (guide T13): when this tile's max is within 8 of the
    // running max for EVERY row in the wave, keep the old max — P stays
    // bounded by e^8 and the whole O-rescale pass is skipped.
    tmax = fmaxf(tmax, __shfl_xor(tmax, 32));
    const bool nores = __all(tmax - m <= 8.0f) && m > -1e29f;
    const float mnew = nores ? m : fmaxf(m, tmax);
    float psum = 0.f;
    #pragma unroll
    for (int i = 0; i < 32; ++i) {
      const float p = __builtin_amdgcn_exp2f(p32[i] - mnew);
      p32[i] = p;
      psum += p;
    }
    psum += __shfl_xor(psum, 32);
    if (nores) {
      lsum += psum;
    } else {
      const float alpha = __builtin_amdgcn_exp2f(m - mnew);
      m = mnew;
      lsum = lsum * alpha + psum;
      #pragma unroll
      for (int dt = 0; dt < DT; ++dt)
        #pragma unroll
        for (int r = 0; r < 16; ++r) oacc[dt][r] *= alpha;
    }

    // ---- P -> bf16 B-fragments in registers.  Fragment kch: lane needs
    // kv = kch*16 + h2*8 + e; elements e<4 live in the h=0 lane of this q,
    // e>=4 in the h=1 lane.  Each lane packs the 4 values it owns for both
    // destination halves, then one half-exchange per dword pair.
    bf16x8v pa[FA_KVB / 16];
    #pragma unroll
    for (int kch = 0; kch < FA_KVB / 16; ++kch) {
      unsigned int d01[2], d23[2];  // prep for dest half 0 / half 1
      #pragma unroll
      for (int ht = 0; ht < 2; ++ht) {
        float pv[4];
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
          // kv = kch*16 + ht*8 + h2*4 + j; h2 is lane-dependent, so index
          // p32 with BOTH compile-time candidates and cndmask-select —
          // a runtime index would spill p32 to scratch.
          const int kv0 = kch * 16 + ht * 8 + j;       // h2 == 0
          const int kv1 = kv0 + 4;                     // h2 == 1
          const int i0 = (kv0 >> 5) * 16 + ((kv0 & 31) & 3) + 4 * ((kv0 & 31) >> 3);
          const int i1 = (kv1 >> 5) * 16 + ((kv1 & 31) & 3) + 4 * ((kv1 & 31) >> 3);
          pv[j] = h2 ? p32[i1] : p32[i0];
        }
        unsigned int* dst = ht == 0 ? d01 : d23;
        dst[0] = pk_bf16(pv[0], pv[1]);
        dst[1] = pk_bf16(pv[2], pv[3]);
      }
      const unsigned int s01_0 = __shfl_xor((int)d01[0], 32);
      const unsigned int s01_1 = __shfl_xor((int)d01[1], 32);
      const unsigned int s23_0 = __shfl_xor((int)d23[0], 32);
      const unsigned int s23_1 = __shfl_xor((int)d23[1], 32);
      unsigned int w0, w1, w2, w3;
      if (h2 == 0) {
        w0 = d01[0]; w1 = d01[1]; w2 = s01_0; w3 = s01_1;
      } else {
        w0 = s23_0; w1 = s23_1; w2 = d23[0]; w3 = d23[1];
      }
      unsigned int frag[4] = {w0, w1, w2, w3};
      pa[kch] = *reinterpret_cast<bf16x8v*>(frag);
    }

    // ---- O^T += V^T x P^T   (A: V^T[d][kv] contiguous b128 from vbuf)
    __builtin_amdgcn_s_setprio(1);
    #pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
      #pragma unroll
      for (int kch = 0; kch < FA_KVB / 16; ++kch) {
        bf16x8v vf = *reinterpret_cast<const bf16x8v*>(
            &vbuf[(dt * 32 + lq) * (FA_KVB + FA_VPAD) + kch * 16 + h2 * 8]);
        oacc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, pa[kch],
                                                           oacc[dt], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();  // every wave done reading this tile's K/V images
    if (kb + FA_KVB < kv_end) {
      write_tile();
      __syncthreads();
    }
  }

  // ---- epilogue: O^T[d][q] /= lsum; store out[q][d] (paired 4-B stores),
  // lse[q] = m + log(lsum).
  const int qg = row0 + lq;
  if (qg < S) {
    const float inv = 1.f / fmaxf(lsum, 1e-30f);
    bf16* op = out + (long)b * osb + (long)h * osh + (long)qg * oss;
    #pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
      #pragma unroll
      for (int r = 0; r < 16; r += 2) {
        const int d = dt * 32 + (r & 3) + 8 * (r >> 2) + 4 * h2;
        *reinterpret_cast<unsigned int*>(op + d) =
            pk_bf16(oacc[dt][r] * inv, oacc[dt][r + 1] * inv);
      }
    }
    if (h2 == 0)  // convert base-2 running max back to natural log for bwd
      lse[((long)b * H + h) * S + qg] =
          m * 0.69314718056f + __logf(fmaxf(lsum, 1e-30f));
  }
  __syncthreads();  // LDS tiles reused by the mirror strip
  }  // halfi
}

}  // namespace

void mfma_probe(at::Tensor A, at::Tensor B, at::Tensor D) {
  auto stream = at::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(mfma_probe_16x16x32, dim3(1), dim3(64), 0, stream,
                     (const bf16*)A.data_ptr(), (const bf16*)B.data_ptr(),
                     D.data_ptr<float>());
}

void mfma_probe32(at::Tensor A, at::Tensor B, at::Tensor D) {
  auto stream = at::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(mfma_probe_32x32x16, dim3(1), dim3(64), 0, stream,
                     (const bf16*)A.data_ptr(), (const bf16*)B.data_ptr(),
                     D.data_ptr<float>());
}

static bool fa_strides_ok(const at::Tensor& t) {
  // unit stride on the head dim, 8-element (16-B) alignment on the others
  return t.stride(3) == 1 && t.stride(0) % 8 == 0 && t.stride(1) % 8 == 0 &&
         t.stride(2) % 8 == 0;
}

std::vector<at::Tensor> fa_fwd(at::Tensor q, at::Tensor k, at::Tensor v,
                               bool causal) {
  TORCH_CHECK(q.scalar_type() == at::kBFloat16 && q.dim() == 4,
              "fa_fwd: 4-D bf16 [B,H,S,D] (any stride, unit head-dim)");
  if (!fa_strides_ok(q)) q = q.contiguous();
  if (!fa_strides_ok(k)) k = k.contiguous();
  if (!fa_strides_ok(v)) v = v.contiguous();
  const int B = q.size(0), H = q.size(1), S = q.size(2), DH = q.size(3);
  const int HKV = k.size(1);
  TORCH_CHECK(DH == 64 || DH == 128, "fa_fwd: head dim 64 or 128");
  TORCH_CHECK(H % HKV == 0, "fa_fwd: H must be a multiple of H_kv");
  TORCH_CHECK(k.size(2) == S && v.size(2) == S && k.size(3) == DH &&
                  v.size(3) == DH && v.size(1) == HKV,
              "fa_fwd: self-attention shapes only (shared S, D, H_kv)");
  // out in [B,S,H,D]-contiguous storage, returned as a [B,H,S,D] view:
  // the caller's transpose(1,2).reshape(B,S,H*D) is then a free view.
  auto out_bshd = at::empty({B, S, H, DH}, q.options());
  auto out = out_bshd.permute({0, 2, 1, 3});
  auto lse = at::empty({B, H, S}, q.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream().stream();
  const int nstrip = (S + 255) / 256;
  dim3 grid(causal ? (nstrip + 1) / 2 : nstrip, B * H);
  const auto L = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, dim3(512), 0, stream,
                       (const bf16*)q.data_ptr(), (const bf16*)k.data_ptr(),
                       (const bf16*)v.data_ptr(), (bf16*)out.data_ptr(),
                       lse.data_ptr<float>(), S, H, HKV, causal ? 1 : 0,
                       q.stride(0), q.stride(1), q.stride(2),
                       k.stride(0), k.stride(1), k.stride(2),
                       v.stride(0), v.stride(1), v.stride(2),
                       out.stride(0), out.stride(1), out.stride(2));
  };
  if (DH == 128) {
    L(fa_fwd_v1<128>);
  } else {
    L(fa_fwd_v1<64>);
  }
  return {out, lse};
}
