// Fused flash-style attention forward with additive bias for CDNA4/gfx950.
//
// Computes O = softmax(Q K^T + bias + kpad [+ causal mask]) V without
// materializing the S x S score matrix (T5 self-attention
// with relative-position bias is UNSCALED — no 1/sqrt(d)).
//
// Structure (v1, correctness-first but MFMA-native):
//   * block = 4 waves = 64 query rows of one (b, h); wave w owns rows
//     [16w, 16w+16) of the tile.
//   * K/V tiles (BK=64 x D) are LDS-staged per iteration; V additionally
//     stored TRANSPOSED (Vt[D][BK]) so both the QK^T B-fragments and the
//     PV B-fragments are contiguous 16-byte ds_read_b128 lines. Rows are
//     padded +8 elements to break the 128-B-stride bank conflict
//     (guide §6 Guideline 4).
//   * MFMA: v_mfma_f32_16x16x32_{bf16,f16}. Fragment maps:
//       A: lane l -> A[l&15][(l>>4)*8 + j]   (j = 0..7)
//       B: lane l -> B[(l>>4)*8 + j][l&15]
//       C: lane l, reg r -> C[(l>>4)*4 + r][l&15]
//   * online softmax per 16-row x 64-col block: row reduce = 4 in-register
//     values + 4-step __shfl_xor over the 16-lane group.
//   * P staged through a per-wave LDS slab to re-shape C-layout -> A-layout
//     for the PV MFMAs.
// Fully-masked causal tiles are skipped before staging.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../common.h"

namespace {

typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef _Float16 f16x8 __attribute__((ext_vector_type(8)));

template <typename T> struct MFMA16;
template <> struct MFMA16<__hip_bfloat16> {
  using frag = bf16x8;
  static DEVINL f32x4 mma(frag a, frag b, f32x4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
  }
};
template <> struct MFMA16<__half> {
  using frag = f16x8;
  static DEVINL f32x4 mma(frag a, frag b, f32x4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_f16(a, b, c, 0, 0, 0);
  }
};

constexpr int BQ = 64;   // query rows per block
constexpr int BK = 64;   // key cols per iteration
constexpr float NEG = -1e30f;

// D in {32, 64}. LDS row pads (+8 elems) break 128B-stride conflicts.
struct Strides3 { long sb, sh, ss; };  // (batch, head, seq) element strides; d is contiguous

template <typename T, int D, bool CAUSAL, bool HAS_BIAS, bool HAS_KPAD>
__global__ __launch_bounds__(256) void attn_fwd_kernel(
    const T* __restrict__ q, const T* __restrict__ k, const T* __restrict__ v,
    const T* __restrict__ bias, const float* __restrict__ kpad,
    T* __restrict__ out, float* __restrict__ lse,
    int B, int H, int Sq, int Sk,
    Strides3 qs, Strides3 ks, Strides3 vs, Strides3 os,
    long bias_sb, long bias_sh /* element strides; 0 = broadcast */,
    float p_drop, const long* __restrict__ seed_ptr) {
  using FR = typename MFMA16<T>::frag;
  constexpr int KLD = D + 8;       // K tile leading dim
  constexpr int VLD = BK + 8;      // Vt tile leading dim
  constexpr int PLD = BK + 8;      // P slab leading dim
  constexpr int NQT = 1;           // 16-row tiles per wave (fixed: 4 waves x 16 = BQ)
  constexpr int NDT = D / 16;      // output col tiles
  constexpr int NKT = BK / 16;     // score col tiles

  __shared__ T k_lds[BK * KLD];
  __shared__ T vt_lds[D * VLD];
  __shared__ T p_lds[4][16 * PLD];
  __shared__ T bias_lds[HAS_BIAS ? BQ * PLD : 1];  // [q-local][k-local]

  const int bh = blockIdx.y;       // b * H + h
  const int b = bh / H, h = bh % H;
  const int q0 = blockIdx.x * BQ;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int lg = lane >> 4;        // lane group 0..3
  const int lr = lane & 15;        // lane row/col within group

  const T* qbase = q + (long)b * qs.sb + (long)h * qs.sh;
  const T* kbase = k + (long)b * ks.sb + (long)h * ks.sh;
  const T* vbase = v + (long)b * vs.sb + (long)h * vs.sh;
  T* obase = out + (long)b * os.sb + (long)h * os.sh;
  const T* bias_bh = HAS_BIAS ? bias + (long)b * bias_sb + (long)h * bias_sh : nullptr;
  const float* kpad_b = HAS_KPAD ? kpad + (long)b * Sk : nullptr;

  // ---- load this wave's Q fragments (A-layout), clamped rows ----
  const int qrow_frag = q0 + wid * 16 + lr;  // A row for this lane
  const int qrow_c = min(qrow_frag, Sq - 1);
  FR qfrag[D / 32];
#pragma unroll
  for (int s = 0; s < D / 32; ++s) {
    const T* src = qbase + (long)qrow_c * qs.ss + s * 32 + lg * 8;
    qfrag[s] = *reinterpret_cast<const FR*>(src);
  }

  // softmax state: this lane covers rows (lg*4 + r) of the wave's 16-row tile
  float m_i[4], l_i[4];
  f32x4 o_acc[NDT];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_i[r] = -INFINITY; l_i[r] = 0.f; }
#pragma unroll
  for (int t = 0; t < NDT; ++t) o_acc[t] = {0.f, 0.f, 0.f, 0.f};

  const int diag_off = Sk - Sq;  // causal: allow kc <= qr + diag_off
  const int k_end = CAUSAL ? min(Sk, q0 + BQ + diag_off) : Sk;

  // ---- async-STAGE split (guide T14): global loads for tile t+1 issue
  // BEFORE tile t's compute; the LDS write lands after the barrier, so HBM
  // latency hides under the MFMA phase. Per thread: 2 K rows + 2 V rows.
  const int stage_j = threadIdx.x / (D / 8);
  const int stage_d0 = (threadIdx.x % (D / 8)) * 8;
  constexpr int RPP = 256 / (D / 8);
  constexpr int NPASS = BK / RPP;  // row passes per tile (<= 2 for D<=64)
  VecT<T, 8> kst[NPASS], vst_r[NPASS];
  // bias tile staging: [64 q rows x 8-col chunks] split (independent of the
  // D-based K/V split — D=32 has fewer K passes but the same 64 bias rows)
  const int bias_j = threadIdx.x / (BK / 8);
  const int bias_c0 = (threadIdx.x % (BK / 8)) * 8;
  constexpr int BRPP = 256 / (BK / 8);
  constexpr int NBPASS = BQ / BRPP;
  VecT<T, 8> bst[NBPASS];

  auto stage_load = [&](int k0) {
#pragma unroll
    for (int p = 0; p < NPASS; ++p) {
      const int kr = k0 + stage_j + p * RPP;
      if (kr < Sk) {
        kst[p] = vload<T, 8>(kbase + (long)kr * ks.ss + stage_d0);
        vst_r[p] = vload<T, 8>(vbase + (long)kr * vs.ss + stage_d0);
      } else {
#pragma unroll
        for (int u = 0; u < 8; ++u) { kst[p].data[u] = T(0.f); vst_r[p].data[u] = T(0.f); }
      }
    }
    if (HAS_BIAS) {
#pragma unroll
      for (int p = 0; p < NBPASS; ++p) {
        // coalesced vec8 along k of the (q-local, k0+...) bias rows; OOB
        // rows clamp (the softmax mask discards them anyway)
        const int qb = min(q0 + bias_j + p * BRPP, Sq - 1);
        const int kb = k0 + bias_c0;
        if (kb + 8 <= Sk) {
          bst[p] = vload<T, 8>(bias_bh + (long)qb * Sk + kb);
        } else {
#pragma unroll
          for (int u = 0; u < 8; ++u)
            bst[p].data[u] = (kb + u < Sk) ? bias_bh[(long)qb * Sk + kb + u] : T(0.f);
        }
      }
    }
  };
  auto stage_write = [&]() {
#pragma unroll
    for (int p = 0; p < NPASS; ++p) {
      const int jj = stage_j + p * RPP;
      vstore<T, 8>(&k_lds[jj * KLD + stage_d0], kst[p]);
#pragma unroll
      for (int u = 0; u < 8; ++u) vt_lds[(stage_d0 + u) * VLD + jj] = vst_r[p].data[u];
    }
    if (HAS_BIAS) {
#pragma unroll
      for (int p = 0; p < NBPASS; ++p)
        vstore<T, 8>(&bias_lds[(bias_j + p * BRPP) * PLD + bias_c0], bst[p]);
    }
  };

  if (k_end > 0) stage_load(0);
  for (int k0 = 0; k0 < k_end; k0 += BK) {
    stage_write();
    __syncthreads();
    if (k0 + BK < k_end) stage_load(k0 + BK);  // in flight under the MFMAs

    // ---- S = Q K^T for this wave's 16 rows x BK cols ----
    f32x4 s_acc[NKT];
#pragma unroll
    for (int t = 0; t < NKT; ++t) s_acc[t] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int t = 0; t < NKT; ++t) {
#pragma unroll
      for (int s = 0; s < D / 32; ++s) {
        // B fragment: K^T[kk][col] = K[col][kk] -> k_lds[t*16 + lr][s*32 + lg*8 ..]
        FR bfrag = *reinterpret_cast<const FR*>(&k_lds[(t * 16 + lr) * KLD + s * 32 + lg * 8]);
        s_acc[t] = MFMA16<T>::mma(qfrag[s], bfrag, s_acc[t]);
      }
    }

    // ---- bias/mask add + online softmax ----
    float p_val[NKT][4];
    float rmax[4], rsum[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) rmax[r] = -INFINITY;
    const int qrow0 = q0 + wid * 16 + lg * 4;
#pragma unroll
    for (int t = 0; t < NKT; ++t) {
      const int kc = k0 + t * 16 + lr;
      const bool k_ok = kc < Sk;
      const float kp = (HAS_KPAD && k_ok) ? kpad_b[kc] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qr = qrow0 + r;
        float sv = s_acc[t][r];
        if (HAS_BIAS && k_ok && qr < Sq)
          sv += DT<T>::to_f32(bias_lds[(wid * 16 + lg * 4 + r) * PLD + t * 16 + lr]);
        sv += kp;
        if (!k_ok || qr >= Sq || (CAUSAL && kc > qr + diag_off)) sv = NEG;
        p_val[t][r] = sv;
        rmax[r] = fmaxf(rmax[r], sv);
      }
    }
    // reduce max across the 16 lanes of this lane group
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        rmax[r] = fmaxf(rmax[r], __shfl_xor(rmax[r], off, 16));
    }
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const float mn = fmaxf(m_i[r], rmax[r]);
      alpha[r] = (m_i[r] == -INFINITY) ? 0.f : __expf(m_i[r] - mn);
      m_i[r] = mn;
      rsum[r] = 0.f;
#pragma unroll
      for (int t = 0; t < NKT; ++t) {
        float pv = (p_val[t][r] <= NEG) ? 0.f : __expf(p_val[t][r] - mn);
        p_val[t][r] = pv;
        rsum[r] += pv;
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int off = 8; off > 0; off >>= 1) rsum[r] += __shfl_xor(rsum[r], off, 16);
      l_i[r] = l_i[r] * alpha[r] + rsum[r];
    }
    // rescale O accumulator
#pragma unroll
    for (int t = 0; t < NDT; ++t)
#pragma unroll
      for (int r = 0; r < 4; ++r) o_acc[t][r] *= alpha[r];

    // ---- stage P (C layout -> wave-local LDS slab, row-major [16][BK]) ----
    // T5 attention-prob dropout (training): mask applied to the P used for
    // the PV accumulation; the softmax normalizer l keeps the full sum.
    const long long seed = seed_ptr ? *seed_ptr : 0;
    const float keep_inv = p_drop > 0.f ? 1.f / (1.f - p_drop) : 1.f;
#pragma unroll
    for (int t = 0; t < NKT; ++t)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float pv = p_val[t][r];
        if (p_drop > 0.f) {
          const long long qr = q0 + wid * 16 + lg * 4 + r;
          const long long kc = k0 + t * 16 + lr;
          const long long idx = ((long long)bh * Sq + qr) * Sk + kc;
          pv = (hash_u01(seed, idx) >= p_drop) ? pv * keep_inv : 0.f;
        }
        p_lds[wid][(lg * 4 + r) * PLD + t * 16 + lr] = DT<T>::from_f32(pv);
      }
    // the slab is wave-local, but make the cross-LANE write->read ordering
    // explicit (every wave reaches this point each iteration)
    __syncthreads();

    // ---- O += P V : A = P fragment, B = Vt fragment ----
#pragma unroll
    for (int t = 0; t < NDT; ++t) {
#pragma unroll
      for (int s = 0; s < BK / 32; ++s) {
        FR pa = *reinterpret_cast<const FR*>(&p_lds[wid][lr * PLD + s * 32 + lg * 8]);
        // B[kk][col] = V[kk][d] = vt_lds[(t*16+lr?) ...]: B col = d-index
        FR vb = *reinterpret_cast<const FR*>(&vt_lds[(t * 16 + lr) * VLD + s * 32 + lg * 8]);
        // NOTE: B fragment wants B[(lg*8+j)][lr] = V[kk][d=lr+16t] =
        // vt_lds[(t*16+lr)*VLD + kk]; kk = s*32 + lg*8 + j -> contiguous. OK.
        o_acc[t] = MFMA16<T>::mma(pa, vb, o_acc[t]);
      }
    }
    __syncthreads();  // all waves done with k_lds/vt_lds before restage
  }

  // ---- epilogue: normalize + store O, store LSE ----
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qr = q0 + wid * 16 + lg * 4 + r;
    if (qr >= Sq) continue;
    const float inv = (l_i[r] > 0.f) ? 1.f / l_i[r] : 0.f;
#pragma unroll
    for (int t = 0; t < NDT; ++t)
      obase[(long)qr * os.ss + t * 16 + lr] = DT<T>::from_f32(o_acc[t][r] * inv);
    if (lr == 0)
      lse[(long)bh * Sq + qr] =
          (l_i[r] > 0.f) ? m_i[r] + logf(l_i[r]) : -INFINITY;
  }
}

static Strides3 strides_of(const torch::Tensor& t) {
  return Strides3{t.stride(0), t.stride(1), t.stride(2)};
}

template <typename T, int D>
void launch_attn(const torch::Tensor& q, const torch::Tensor& k, const torch::Tensor& v,
                 const c10::optional<torch::Tensor>& bias,
                 const c10::optional<torch::Tensor>& kpad, bool causal,
                 double p_drop, const c10::optional<torch::Tensor>& seed,
                 torch::Tensor& out, torch::Tensor& lse, hipStream_t stream) {
  const int B = q.size(0), H = q.size(1), Sq = q.size(2), Sk = k.size(2);
  dim3 grid((Sq + BQ - 1) / BQ, B * H);
  Strides3 qst = strides_of(q), kst = strides_of(k), vst = strides_of(v),
           ost = strides_of(out);
  const bool has_bias = bias.has_value() && bias->defined();
  const bool has_kpad = kpad.has_value() && kpad->defined();
  const T* bias_p = has_bias ? (const T*)bias->data_ptr() : nullptr;
  const float* kpad_p = has_kpad ? kpad->data_ptr<float>() : nullptr;
  long bias_sb = 0, bias_sh = 0;
  if (has_bias) {
    TORCH_CHECK(bias->dim() == 4 && bias->size(2) == Sq && bias->size(3) == Sk,
                "bias must be (B|1, H|1, Sq, Sk)");
    bias_sb = bias->size(0) > 1 ? bias->stride(0) : 0;
    bias_sh = bias->size(1) > 1 ? bias->stride(1) : 0;
  }

#define LAUNCH(CAUSAL, HB, HK)                                                      \
  hipLaunchKernelGGL((attn_fwd_kernel<T, D, CAUSAL, HB, HK>), grid, dim3(256), 0,   \
                     stream, (const T*)q.data_ptr(), (const T*)k.data_ptr(),        \
                     (const T*)v.data_ptr(), bias_p, kpad_p, (T*)out.data_ptr(),    \
                     lse.data_ptr<float>(), B, H, Sq, Sk, qst, kst, vst, ost,       \
                     bias_sb, bias_sh, (float)p_drop,                                  \
                     (seed.has_value() && seed->defined())                             \
                         ? seed->data_ptr<long>() : nullptr)
  if (causal) {
    if (has_bias && has_kpad) LAUNCH(true, true, true);
    else if (has_bias) LAUNCH(true, true, false);
    else if (has_kpad) LAUNCH(true, false, true);
    else LAUNCH(true, false, false);
  } else {
    if (has_bias && has_kpad) LAUNCH(false, true, true);
    else if (has_bias) LAUNCH(false, true, false);
    else if (has_kpad) LAUNCH(false, false, true);
    else LAUNCH(false, false, false);
  }
#undef LAUNCH
  HIP_CHECK_KERNEL();
}

}  // namespace

std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                    c10::optional<torch::Tensor> bias,
                                    c10::optional<torch::Tensor> kpad, bool causal,
                                    double p_drop, c10::optional<torch::Tensor> seed) {
  // (B,H,S,D) semantics with ARBITRARY (b,h,s) strides, d contiguous — so
  // (B,S,H,D)-physical projection views need no transpose copies.
  TORCH_CHECK(q.is_cuda() && q.dim() == 4 && q.stride(3) == 1, "q must be (B,H,Sq,D), d contiguous");
  TORCH_CHECK(k.stride(3) == 1 && v.stride(3) == 1);
  const int B = q.size(0), H = q.size(1), Sq = q.size(2);
  const int D = q.size(3);
  // output physically (B,Sq,H,D), returned as a (B,H,Sq,D) view
  auto out = torch::empty({B, Sq, H, D}, q.options()).permute({0, 2, 1, 3});
  auto lse = torch::empty({q.size(0), q.size(1), q.size(2)},
                          q.options().dtype(torch::kFloat));
  auto stream = at::hip::getCurrentHIPStream().stream();
  c10::optional<torch::Tensor> bias_c;
  if (bias.has_value() && bias->defined()) bias_c = bias->to(q.scalar_type()).contiguous();
  if (q.scalar_type() == torch::kBFloat16 && D == 64)
    launch_attn<__hip_bfloat16, 64>(q, k, v, bias_c, kpad, causal, p_drop, seed, out, lse, stream);
  else if (q.scalar_type() == torch::kHalf && D == 64)
    launch_attn<__half, 64>(q, k, v, bias_c, kpad, causal, p_drop, seed, out, lse, stream);
  else if (q.scalar_type() == torch::kBFloat16 && D == 32)
    launch_attn<__hip_bfloat16, 32>(q, k, v, bias_c, kpad, causal, p_drop, seed, out, lse, stream);
  else if (q.scalar_type() == torch::kHalf && D == 32)
    launch_attn<__half, 32>(q, k, v, bias_c, kpad, causal, p_drop, seed, out, lse, stream);
  else
    TORCH_CHECK(false, "attn_fwd: unsupported dtype/head-dim (bf16/f16, D in {32,64}); got D=", D);
  return {out, lse};
}
