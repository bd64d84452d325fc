// Flash attention forward v2 for CDNA4/gfx950 — swapped-operand 32x32 MFMA
// with fully in-register softmax (guide §B attention ladder, T12).
//
// PMC on v1 measured SQ_INSTS_VALU:SQ_INSTS_MFMA = 33:1 (the P-slab LDS
// round-trip + 4-step shuffle reduces). v2 removes both:
//   * S^T = mfma_f32_32x32x16(K, Q^T): each lane's C column IS one query row
//     (col = lane&31), its 16 regs hold this tile's 32 keys split across the
//     two 32-lane halves -> row max/sum = 15 in-lane ops + ONE shfl_xor(32).
//   * P^T repack for the PV B-fragment = 8 v_cvt_pk_bf16_f32 + 4
//     v_permlane32_swap per 32-key tile (no LDS slab, no barrier).
//   * PV: O^T = mfma(V^T, P^T) accumulated in C layout (same per-lane query
//     column), so the online rescale is a per-lane scalar multiply.
// Block = 4 waves x 32 query rows = 128 q/block; K/V(+transposed V) tiles
// LDS-staged with the T14 async split; bias tile staged per k-tile.
// Same contract as v1: O (strided BSHD view) + LSE, bias/kpad/causal and
// fused attention-prob dropout.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../common.h"

namespace {

typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef _Float16 f16x8 __attribute__((ext_vector_type(8)));
typedef unsigned int uint2v __attribute__((ext_vector_type(2)));

template <typename T> struct MF32;
template <> struct MF32<__hip_bfloat16> {
  using frag = bf16x8;
  static DEVINL f32x16 mma(frag a, frag b, f32x16 c) {
    return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
  }
  // pack 2 fp32 -> one dword of 2 bf16 (compiler emits v_cvt_pk_bf16_f32)
  static DEVINL unsigned pack2(float lo, float hi) {
    union { __hip_bfloat162 h2; unsigned u; } u;
    u.h2 = __hip_bfloat162(__float2bfloat16(lo), __float2bfloat16(hi));
    return u.u;
  }
};
template <> struct MF32<__half> {
  using frag = f16x8;
  static DEVINL f32x16 mma(frag a, frag b, f32x16 c) {
    return __builtin_amdgcn_mfma_f32_32x32x16_f16(a, b, c, 0, 0, 0);
  }
  static DEVINL unsigned pack2(float lo, float hi) {
    union { __half2 h2; unsigned u; } u;
    u.h2 = __half2(__float2half(lo), __float2half(hi));
    return u.u;
  }
};

constexpr int BQ2 = 128;  // q rows per block (4 waves x 32)
constexpr int BK2 = 32;   // keys per tile
constexpr float NEG2 = -1e30f;

struct S3 { long sb, sh, ss; };

template <typename T, bool CAUSAL, bool HAS_BIAS, bool HAS_KPAD>
__global__ __launch_bounds__(256) void attn_fwd_v2_kernel(
    const T* __restrict__ q, const T* __restrict__ k, const T* __restrict__ v,
    const T* __restrict__ bias, const float* __restrict__ kpad,
    T* __restrict__ out, float* __restrict__ lse,
    float* __restrict__ opart, float* __restrict__ mpart, float* __restrict__ lpart,
    int B, int H, int Sq, int Sk,
    S3 qs, S3 ks, S3 vs, S3 os, long bias_sb, long bias_sh,
    float p_drop, const long* __restrict__ seed_ptr) {
  constexpr int D = 64;
  using FR = typename MF32<T>::frag;
  constexpr int KLD = D + 8;    // k_lds row elems (144 B rows)
  constexpr int VLD = BK2 + 8;  // vt rows (80 B)
  constexpr int BLD = BK2 + 8;  // bias rows

  __shared__ T k_lds[BK2 * KLD];
  __shared__ T vt_lds[D * VLD];
  __shared__ T bias_lds[HAS_BIAS ? BQ2 * BLD : 1];

  const int bh = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const int q0 = blockIdx.x * BQ2;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int lq = lane & 31;     // this lane's query column within the wave tile
  const int half = lane >> 5;   // key-half owner

  const T* qbase = q + (long)b * qs.sb + (long)h * qs.sh;
  const T* kbase = k + (long)b * ks.sb + (long)h * ks.sh;
  const T* vbase = v + (long)b * vs.sb + (long)h * vs.sh;
  T* obase = out + (long)b * os.sb + (long)h * os.sh;
  const T* bias_bh = HAS_BIAS ? bias + (long)b * bias_sb + (long)h * bias_sh : nullptr;
  const float* kpad_b = HAS_KPAD ? kpad + (long)b * Sk : nullptr;

  const int my_q = q0 + wid * 32 + lq;            // this lane's query row
  const int my_q_c = min(my_q, Sq - 1);

  // Q^T B-fragments, loaded once: B[d = s*16 + half*8 + j][q=lq] = Q[q][d]
  FR qfrag[4];
#pragma unroll
  for (int s = 0; s < 4; ++s)
    qfrag[s] = *reinterpret_cast<const FR*>(
        qbase + (long)my_q_c * qs.ss + s * 16 + half * 8);

  // softmax state (one query row per lane) + O^T accumulator (2 d-tiles)
  float m_i = -INFINITY, l_i = 0.f;
  f32x16 o_acc[2];
#pragma unroll
  for (int r = 0; r < 16; ++r) { o_acc[0][r] = 0.f; o_acc[1][r] = 0.f; }

  const int diag_off = Sk - Sq;
  const int k_end = CAUSAL ? min(Sk, q0 + BQ2 + diag_off) : Sk;
  const long long seed = seed_ptr ? *seed_ptr : 0;
  const float keep_inv = p_drop > 0.f ? 1.f / (1.f - p_drop) : 1.f;

  // T14 staging: K rows ([32][D] image) + V cols (vt [D][BK2] image) + bias
  const int sj = threadIdx.x / (D / 8);            // k-row (32 rows, 8 chunks)
  const int sd = (threadIdx.x % (D / 8)) * 8;
  VecT<T, 8> kreg, vreg;
  const int bj = threadIdx.x / (BK2 / 8);          // bias q-row (128 rows, 4 chunks)
  const int bc = (threadIdx.x % (BK2 / 8)) * 8;
  constexpr int BRPP = 256 / (BK2 / 8);            // 64 rows per pass
  VecT<T, 8> breg[BQ2 / BRPP];

  auto stage_load = [&](int k0) {
    const int kr = k0 + sj;
    if (kr < Sk) {
      kreg = vload<T, 8>(kbase + (long)kr * ks.ss + sd);
      vreg = vload<T, 8>(vbase + (long)kr * vs.ss + sd);
    } else {
#pragma unroll
      for (int u = 0; u < 8; ++u) { kreg.data[u] = T(0.f); vreg.data[u] = T(0.f); }
    }
    if (HAS_BIAS) {
#pragma unroll
      for (int p = 0; p < BQ2 / BRPP; ++p) {
        const int qb = min(q0 + bj + p * BRPP, Sq - 1);
        const int kb = k0 + bc;
        if (kb + 8 <= Sk) {
          breg[p] = vload<T, 8>(bias_bh + (long)qb * Sk + kb);
        } else {
#pragma unroll
          for (int u = 0; u < 8; ++u)
            breg[p].data[u] = (kb + u < Sk) ? bias_bh[(long)qb * Sk + kb + u] : T(0.f);
        }
      }
    }
  };
  auto stage_write = [&]() {
    vstore<T, 8>(&k_lds[sj * KLD + sd], kreg);
#pragma unroll
    for (int u = 0; u < 8; ++u) vt_lds[(sd + u) * VLD + sj] = vreg.data[u];
    if (HAS_BIAS) {
#pragma unroll
      for (int p = 0; p < BQ2 / BRPP; ++p)
        vstore<T, 8>(&bias_lds[(bj + p * BRPP) * BLD + bc], breg[p]);
    }
  };

  // k-split (underfilled grids): gridDim.z chunks of k-tiles; partial
  // (o_acc, m, l) per chunk are combined by attn_fwd_merge_kernel
  const int nkt_all = (Sk + BK2 - 1) / BK2;
  const int tpk = (nkt_all + gridDim.z - 1) / gridDim.z;
  const int kz0 = blockIdx.z * tpk * BK2;
  const int kz_end = min(k_end, kz0 + tpk * BK2);

  if (kz_end > kz0) stage_load(kz0);
  for (int k0 = kz0; k0 < kz_end; k0 += BK2) {
    __syncthreads();  // previous tile's LDS consumers done
    stage_write();
    __syncthreads();
    if (k0 + BK2 < kz_end) stage_load(k0 + BK2);

    // ---- S^T = K Q^T: A = K rows, B = Q^T ----
    f32x16 st;
#pragma unroll
    for (int r = 0; r < 16; ++r) st[r] = 0.f;
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      FR a = *reinterpret_cast<const FR*>(&k_lds[lq * KLD + s * 16 + half * 8]);
      st = MF32<T>::mma(a, qfrag[s], st);
    }
    __builtin_amdgcn_s_setprio(0);

    // ---- mask + bias; in-lane rows hold keys kk(r) = (r&3)+8*(r>>2)+4*half
    float pv[16];
    float tile_max = -INFINITY;
    const bool q_ok = my_q < Sq;
    VecT<T, 4> bias4[4];
    if (HAS_BIAS) {
#pragma unroll
      for (int g = 0; g < 4; ++g)  // 4 consecutive keys per reg group -> b64
        bias4[g] = vload<T, 4>(&bias_lds[(wid * 32 + lq) * BLD + 8 * g + 4 * half]);
    }
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int kk = (r & 3) + 8 * (r >> 2) + 4 * half;
      const int kc = k0 + kk;
      float sv = st[r];
      if (HAS_BIAS) sv += DT<T>::to_f32(bias4[r >> 2].data[r & 3]);
      if (HAS_KPAD && kc < Sk) sv += kpad_b[kc];
      if (kc >= Sk || !q_ok || (CAUSAL && kc > my_q + diag_off)) sv = NEG2;
      pv[r] = sv;
      tile_max = fmaxf(tile_max, sv);
    }
    tile_max = fmaxf(tile_max, __shfl_xor(tile_max, 32, WAVE));
    const float mn = fmaxf(m_i, tile_max);
    const float alpha = (m_i == -INFINITY) ? 0.f : __expf(m_i - mn);
    m_i = mn;
    float rsum = 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      float p = (pv[r] <= NEG2) ? 0.f : __expf(pv[r] - mn);
      pv[r] = p;
      rsum += p;
    }
    rsum += __shfl_xor(rsum, 32, WAVE);
    l_i = l_i * alpha + rsum;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      o_acc[0][r] *= alpha;
      o_acc[1][r] *= alpha;
    }
    if (p_drop > 0.f) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kk = (r & 3) + 8 * (r >> 2) + 4 * half;
        const long long idx = ((long long)bh * Sq + my_q) * Sk + (k0 + kk);
        pv[r] = (hash_u01(seed, idx) >= p_drop) ? pv[r] * keep_inv : 0.f;
      }
    }

    // ---- P^T repack -> B-fragments (T12: cvt_pk + permlane32_swap) ----
    // regs 0..7 cover keys 0..15 (lower half) / shifted by 4 (upper);
    // swap(word(p0,p1), word(p4,p5)) yields w0 (both halves' first dword)
    // and w2; swap(word(p2,p3), word(p6,p7)) yields w1 and w3.
    unsigned pb[8];  // two B-fragments x 4 dwords
#pragma unroll
    for (int g = 0; g < 2; ++g) {  // g=0: keys 0..15, g=1: keys 16..31
      const int o = g * 8;
      unsigned a0 = MF32<T>::pack2(pv[o + 0], pv[o + 1]);
      unsigned b0 = MF32<T>::pack2(pv[o + 4], pv[o + 5]);
      unsigned a1 = MF32<T>::pack2(pv[o + 2], pv[o + 3]);
      unsigned b1 = MF32<T>::pack2(pv[o + 6], pv[o + 7]);
      auto r0 = __builtin_amdgcn_permlane32_swap(a0, b0, false, false);
      auto r1 = __builtin_amdgcn_permlane32_swap(a1, b1, false, false);
      pb[g * 4 + 0] = r0[0];
      pb[g * 4 + 1] = r1[0];
      pb[g * 4 + 2] = r0[1];
      pb[g * 4 + 3] = r1[1];
    }

    // ---- O^T += V^T P^T : A = V^T (2 d-tiles), accumulate over 32 keys ----
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int dt = 0; dt < 2; ++dt) {
#pragma unroll
      for (int g = 0; g < 2; ++g) {
        FR a = *reinterpret_cast<const FR*>(
            &vt_lds[(dt * 32 + lq) * VLD + g * 16 + half * 8]);
        struct U4 { unsigned x[4]; } u4{pb[g * 4], pb[g * 4 + 1],
                                        pb[g * 4 + 2], pb[g * 4 + 3]};
        FR bfr = __builtin_bit_cast(FR, u4);
        o_acc[dt] = MF32<T>::mma(a, bfr, o_acc[dt]);
      }
    }
    __builtin_amdgcn_s_setprio(0);
  }

  // ---- epilogue: normalize, store O rows (4-elem packs) + LSE ----
  if (my_q < Sq) {
    if (opart == nullptr) {  // no k-split: final store
      const float invl = (l_i > 0.f) ? 1.f / l_i : 0.f;
      T* orow = obase + (long)my_q * os.ss;
#pragma unroll
      for (int dt = 0; dt < 2; ++dt) {
#pragma unroll
        for (int grp = 0; grp < 4; ++grp) {
          const int d0 = dt * 32 + grp * 8 + half * 4;
          VecT<T, 4> ov;
#pragma unroll
          for (int u = 0; u < 4; ++u)
            ov.data[u] = DT<T>::from_f32(o_acc[dt][grp * 4 + u] * invl);
          vstore<T, 4>(orow + d0, ov);
        }
      }
      if (half == 0)
        lse[(long)bh * Sq + my_q] = (l_i > 0.f) ? m_i + logf(l_i) : -INFINITY;
    } else {  // k-split: raw fp32 partial accumulator + (m, l) per chunk
      const long zrow = ((long)blockIdx.z * gridDim.y + bh) * Sq + my_q;
      float* op = opart + zrow * D;
#pragma unroll
      for (int dt = 0; dt < 2; ++dt) {
#pragma unroll
        for (int grp = 0; grp < 4; ++grp) {
          const int d0 = dt * 32 + grp * 8 + half * 4;
          VecT<float, 4> ov;
#pragma unroll
          for (int u = 0; u < 4; ++u) ov.data[u] = o_acc[dt][grp * 4 + u];
          vstore<float, 4>(op + d0, ov);
        }
      }
      if (half == 0) {
        mpart[zrow] = m_i;
        lpart[zrow] = l_i;
      }
    }
  }
}

// combine k-split partials: o = sum_c exp(m_c - M) * o_c / L,
// L = sum_c l_c * exp(m_c - M), lse = M + log L. One thread per (bh, q, 16
// d-quads); M/L recomputed per quad (tiny vs the fp32 partial traffic).
template <typename T>
__global__ void attn_fwd_merge_kernel(const float* __restrict__ opart,
                                      const float* __restrict__ mpart,
                                      const float* __restrict__ lpart,
                                      T* __restrict__ out, float* __restrict__ lse,
                                      int nkc, long BHSq, int H, int Sq,
                                      S3 os) {
  constexpr int D = 64;
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= BHSq * (D / 4)) return;
  const int dq = (int)(idx % (D / 4));
  const long row = idx / (D / 4);  // bh * Sq + q
  float M = -INFINITY;
  for (int c = 0; c < nkc; ++c) M = fmaxf(M, mpart[c * BHSq + row]);
  float L = 0.f;
  for (int c = 0; c < nkc; ++c) {
    const float mc = mpart[c * BHSq + row];
    if (mc > -INFINITY) L += lpart[c * BHSq + row] * __expf(mc - M);
  }
  const float invl = (L > 0.f) ? 1.f / L : 0.f;
  float acc[4] = {0.f, 0.f, 0.f, 0.f};
  for (int c = 0; c < nkc; ++c) {
    const float mc = mpart[c * BHSq + row];
    if (!(mc > -INFINITY)) continue;
    const float w = __expf(mc - M) * invl;
    VecT<float, 4> ov = vload<float, 4>(opart + (c * BHSq + row) * D + dq * 4);
#pragma unroll
    for (int u = 0; u < 4; ++u) acc[u] += w * ov.data[u];
  }
  const long bh = row / Sq;
  const int qr = (int)(row % Sq);
  const int b = (int)(bh / H), h = (int)(bh % H);
  T* orow = out + (long)b * os.sb + (long)h * os.sh + (long)qr * os.ss;
  VecT<T, 4> tv;
#pragma unroll
  for (int u = 0; u < 4; ++u) tv.data[u] = DT<T>::from_f32(acc[u]);
  vstore<T, 4>(orow + dq * 4, tv);
  if (dq == 0) lse[row] = (L > 0.f) ? M + logf(L) : -INFINITY;
}

template <typename T>
void launch_v2(const torch::Tensor& q, const torch::Tensor& k, const torch::Tensor& v,
               const c10::optional<torch::Tensor>& bias,
               const c10::optional<torch::Tensor>& kpad, bool causal,
               double p_drop, const c10::optional<torch::Tensor>& seed,
               torch::Tensor& out, torch::Tensor& lse, hipStream_t stream) {
  const int B = q.size(0), H = q.size(1), Sq = q.size(2), Sk = k.size(2);
  // k-split factor (flash-decoding style) for underfilled grids
  const int qblocks = (Sq + BQ2 - 1) / BQ2;
  const int nkt = (Sk + BK2 - 1) / BK2;
  int nkc = 1;
  {
    const long base_blocks = (long)qblocks * B * H;
    while (base_blocks * nkc < 384 && nkc * 2 <= nkt / 4) nkc *= 2;
    const char* env = std::getenv("MI355_ATTN_FWD_NKC");
    if (env) nkc = std::max(1, std::min(nkt, atoi(env)));
  }
  torch::Tensor opart, mpart, lpart;
  float *opart_p = nullptr, *mpart_p = nullptr, *lpart_p = nullptr;
  if (nkc > 1) {
    const long BHSq = (long)B * H * Sq;
    auto f32 = q.options().dtype(torch::kFloat);
    opart = torch::empty({(long)nkc, BHSq, 64}, f32);
    mpart = torch::empty({(long)nkc, BHSq}, f32);
    lpart = torch::empty({(long)nkc, BHSq}, f32);
    opart_p = opart.data_ptr<float>();
    mpart_p = mpart.data_ptr<float>();
    lpart_p = lpart.data_ptr<float>();
  }
  dim3 grid(qblocks, B * H, nkc);
  S3 qst{q.stride(0), q.stride(1), q.stride(2)};
  S3 kst{k.stride(0), k.stride(1), k.stride(2)};
  S3 vst{v.stride(0), v.stride(1), v.stride(2)};
  S3 ost{out.stride(0), out.stride(1), out.stride(2)};
  const bool hb = bias.has_value() && bias->defined();
  const bool hk = kpad.has_value() && kpad->defined();
  const T* bias_p = hb ? (const T*)bias->data_ptr() : nullptr;
  const float* kpad_p = hk ? kpad->data_ptr<float>() : nullptr;
  long bias_sb = 0, bias_sh = 0;
  if (hb) {
    bias_sb = bias->size(0) > 1 ? bias->stride(0) : 0;
    bias_sh = bias->size(1) > 1 ? bias->stride(1) : 0;
  }
#define LV2(C, HB, HK)                                                              \
  hipLaunchKernelGGL((attn_fwd_v2_kernel<T, C, HB, HK>), grid, dim3(256), 0,        \
                     stream, (const T*)q.data_ptr(), (const T*)k.data_ptr(),        \
                     (const T*)v.data_ptr(), bias_p, kpad_p, (T*)out.data_ptr(),    \
                     lse.data_ptr<float>(), opart_p, mpart_p, lpart_p,              \
                     B, H, Sq, Sk, qst, kst, vst, ost,                              \
                     bias_sb, bias_sh, (float)p_drop,                               \
                     (seed.has_value() && seed->defined()) ? seed->data_ptr<long>() \
                                                           : nullptr)
  if (causal) {
    if (hb && hk) LV2(true, true, true);
    else if (hb) LV2(true, true, false);
    else if (hk) LV2(true, false, true);
    else LV2(true, false, false);
  } else {
    if (hb && hk) LV2(false, true, true);
    else if (hb) LV2(false, true, false);
    else if (hk) LV2(false, false, true);
    else LV2(false, false, false);
  }
#undef LV2
  HIP_CHECK_KERNEL();
  if (nkc > 1) {
    const long BHSq = (long)B * H * Sq;
    const long total = BHSq * 16;  // one thread per d-quad
    hipLaunchKernelGGL((attn_fwd_merge_kernel<T>), dim3((total + 255) / 256),
                       dim3(256), 0, stream, opart_p, mpart_p, lpart_p,
                       (T*)out.data_ptr(), lse.data_ptr<float>(), nkc, BHSq,
                       H, Sq, ost);
    HIP_CHECK_KERNEL();
  }
}

}  // namespace

std::vector<torch::Tensor> attn_fwd_v2(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                       c10::optional<torch::Tensor> bias,
                                       c10::optional<torch::Tensor> kpad, bool causal,
                                       double p_drop, c10::optional<torch::Tensor> seed) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 4 && q.stride(3) == 1 && q.size(3) == 64,
              "attn_fwd_v2: (B,H,Sq,64), d contiguous");
  const int B = q.size(0), H = q.size(1), Sq = q.size(2);
  auto out = torch::empty({B, Sq, H, 64}, q.options()).permute({0, 2, 1, 3});
  auto lse = torch::empty({B, H, Sq}, q.options().dtype(torch::kFloat));
  auto stream = at::hip::getCurrentHIPStream().stream();
  c10::optional<torch::Tensor> bias_c;
  if (bias.has_value() && bias->defined()) bias_c = bias->to(q.scalar_type()).contiguous();
  if (q.scalar_type() == torch::kBFloat16)
    launch_v2<__hip_bfloat16>(q, k, v, bias_c, kpad, causal, p_drop, seed, out, lse, stream);
  else if (q.scalar_type() == torch::kHalf)
    launch_v2<__half>(q, k, v, bias_c, kpad, causal, p_drop, seed, out, lse, stream);
  else
    TORCH_CHECK(false, "attn_fwd_v2: bf16/f16 only");
  return {out, lse};
}
