// Flash-style attention BACKWARD for CDNA4/gfx950 (SURVEY.md K4 bwd,
// "hard part #1": bwd through T5's additive relative-position bias).
//
// Given dO, Q, K, V, O, LSE: recompute P tile-by-tile and produce
//   dV = P^T dO
//   dP = dO V^T ;  dS = P * (dP - rowsum(dO*O))
//   dK = dS^T Q ;  dQ = dS K  (per-k-tile fp32 slabs, plain stores, fused
//                              slab-reduce epilogue — no atomics)
//   dBias = sum_b dS  (per-batch fp32 planes + fused batch-sum/cast)
//
// Geometry: block = 4 waves owns one (b, h, k-tile of 64 keys); waves own 16
// key rows each; loop over q-tiles of 64. All MFMA v_mfma_f32_16x16x32.
// Everything is staged in DYNAMIC LDS (~83 KB/block):
//   kt[64][D+8], ktT[D][64+8], v[64][D+8], q[64][D+8], qT[D][64+8],
//   do[64][D+8], doT[D][64+8], ds_img[64][64+8], slab[4][16][64+8]
// Transposed images exist so every MFMA B-fragment is a contiguous 16-B
// ds_read_b128 (row pads break the 128-B-stride bank conflict, guide G4).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../common.h"

namespace {

typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef _Float16 f16x8 __attribute__((ext_vector_type(8)));

template <typename T> struct MFMA16B;
template <> struct MFMA16B<__hip_bfloat16> {
  using frag = bf16x8;
  static DEVINL f32x4 mma(frag a, frag b, f32x4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
  }
};
template <> struct MFMA16B<__half> {
  using frag = f16x8;
  static DEVINL f32x4 mma(frag a, frag b, f32x4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_f16(a, b, c, 0, 0, 0);
  }
};

constexpr int BKT = 64;  // keys per block
constexpr int BQT = 64;  // queries per iteration
constexpr float NEGB = -1e30f;

// Transposed-image ([D][64], no pad) XOR swizzle: element (drow, col) lives
// at drow*64 + (col ^ key). key mixes drow's low AND mid bits so BOTH access
// patterns are spread: the staging writes walk a COLUMN with lanes 8 rows
// apart (key's drow>>3 part varies per lane -> 2-way instead of ~16-way on
// one bank) and the b128 fragment reads walk 16 consecutive rows (key's
// drow&7 part varies per lane -> conflict-free). Replaces the +8 row pads
// (12.5% LDS) whose write pattern was the round-1 PMC bank-conflict hotspot
// (ROADMAP r1: SQ_LDS_BANK_CONFLICT ~12%% of bwd wave cycles).
DEVINL int timg(int drow, int col) {
  return drow * BKT + (col ^ ((((drow & 7) ^ ((drow >> 3) & 7))) << 3));
}

struct BStr { long sb, sh, ss; };

// delta[b,h,q] = sum_d dO * O (both strided BSHD views)
template <typename T>
__global__ void attn_delta_kernel(const T* __restrict__ dout, const T* __restrict__ o,
                                  float* __restrict__ delta, int H, int Sq, int D,
                                  BStr dos, BStr os) {
  const long row = (long)blockIdx.x * (blockDim.x / WAVE) + threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int b = blockIdx.y;
  if (row >= (long)H * Sq) return;
  const int h = row / Sq, qr = row % Sq;
  const T* dor = dout + b * dos.sb + h * dos.sh + (long)qr * dos.ss;
  const T* orr = o + b * os.sb + h * os.sh + (long)qr * os.ss;
  float acc = 0.f;
  for (int d = lane; d < D; d += WAVE)
    acc += DT<T>::to_f32(dor[d]) * DT<T>::to_f32(orr[d]);
  acc = wave_reduce_sum(acc);
  if (lane == 0) delta[((long)b * H + h) * Sq + qr] = acc;
}

template <typename T, int D, bool CAUSAL, bool HAS_BIAS, bool HAS_KPAD, bool NEED_DBIAS>
__global__ __launch_bounds__(256) void attn_bwd_kernel(
    const T* __restrict__ dout, const T* __restrict__ q, const T* __restrict__ k,
    const T* __restrict__ v, const float* __restrict__ lse, const float* __restrict__ delta,
    const T* __restrict__ bias, const float* __restrict__ kpad,
    float* __restrict__ dq32, T* __restrict__ dk, T* __restrict__ dv,
    float* __restrict__ dbias32, float* __restrict__ dkv32 /*q-split slabs*/,
    int B, int H, int Sq, int Sk,
    BStr qs, BStr ks, BStr vs, BStr dos, BStr dks, BStr dvs,
    long bias_sb, long bias_sh,
    float p_drop, const long* __restrict__ seed_ptr) {
  using FR = typename MFMA16B<T>::frag;
  constexpr int LD = D + 8;        // [64][D] images
  constexpr int LDT = BKT + 8;     // ds_img + slab row stride

  extern __shared__ __attribute__((aligned(16))) char smem[];
  T* kt_img = (T*)smem;                         // [BKT][LD]
  T* ktT_img = kt_img + BKT * LD;               // [D][64] XOR-swizzled (timg)
  T* v_img = ktT_img + D * BKT;                 // [BKT][LD]
  T* q_img = v_img + BKT * LD;                  // [BQT][LD]
  T* qT_img = q_img + BQT * LD;                 // [D][64] XOR-swizzled
  T* do_img = qT_img + D * BKT;                 // [BQT][LD]
  T* doT_img = do_img + BQT * LD;               // [D][64] XOR-swizzled
  T* slab = doT_img + D * BKT;                  // [4][16][LDT] per-wave P^T / dS^T
  // ds_img ALIASES q_img when the strides match (D==64): every q_img read
  // (the S^T B-fragments) completes before the P-slab barrier, and ds_img is
  // only written after it — saves 9 KB LDS -> 2 blocks/CU instead of 1.
  T* ds_img = (LD == LDT) ? q_img : slab + 4 * 16 * LDT;  // [BQT][LDT]

  const int bh = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const int k0 = blockIdx.x * BKT;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int lg = lane >> 4;
  const int lr = lane & 15;

  const T* qbase = q + (long)b * qs.sb + (long)h * qs.sh;
  const T* kbase = k + (long)b * ks.sb + (long)h * ks.sh;
  const T* vbase = v + (long)b * vs.sb + (long)h * vs.sh;
  const T* dobase = dout + (long)b * dos.sb + (long)h * dos.sh;
  T* dkbase = dk + (long)b * dks.sb + (long)h * dks.sh;
  T* dvbase = dv + (long)b * dvs.sb + (long)h * dvs.sh;
  const float* lse_bh = lse + ((long)b * H + h) * Sq;
  const float* delta_bh = delta + ((long)b * H + h) * Sq;
  const T* bias_bh = HAS_BIAS ? bias + (long)b * bias_sb + (long)h * bias_sh : nullptr;
  const float* kpad_b = HAS_KPAD ? kpad + (long)b * Sk : nullptr;
  // dq32: (NKTILES, B, H, Sq, D) per-k-tile slabs; dbias32: (B, H, Sq, Sk)
  // per-batch planes — both written with PLAIN stores (atomics made the
  // dbias variants 2-3x slower than the no-bias one) and summed afterwards.
  float* dq_bh = dq32 + (((long)blockIdx.x * B * H + (long)b * H + h) * Sq) * D;
  float* dbias_h = NEED_DBIAS
      ? dbias32 + ((long)b * H + h) * (long)Sq * Sk : nullptr;

  // ---- stage this block's K and V (+ K^T image), zero-padded OOB ----
  {
    const int j = threadIdx.x / (D / 8);
    const int d0 = (threadIdx.x % (D / 8)) * 8;
    constexpr int RPP = 256 / (D / 8);
    for (int jj = j; jj < BKT; jj += RPP) {
      const int kr = k0 + jj;
      VecT<T, 8> kv, vv;
      if (kr < Sk) {
        kv = vload<T, 8>(kbase + (long)kr * ks.ss + d0);
        vv = vload<T, 8>(vbase + (long)kr * vs.ss + d0);
      } else {
#pragma unroll
        for (int u = 0; u < 8; ++u) { kv.data[u] = T(0.f); vv.data[u] = T(0.f); }
      }
      vstore<T, 8>(&kt_img[jj * LD + d0], kv);
      vstore<T, 8>(&v_img[jj * LD + d0], vv);
#pragma unroll
      for (int u = 0; u < 8; ++u) ktT_img[timg(d0 + u, jj)] = kv.data[u];
    }
  }

  // accumulators: wave owns k rows [16*wid, 16*wid+16)
  f32x4 dv_acc[D / 16], dk_acc[D / 16];
#pragma unroll
  for (int t = 0; t < D / 16; ++t) {
    dv_acc[t] = {0.f, 0.f, 0.f, 0.f};
    dk_acc[t] = {0.f, 0.f, 0.f, 0.f};
  }

  const int diag_off = Sk - Sq;
  // q-split (underfilled grids, e.g. B=2): gridDim.z chunks of q-tiles per
  // (b,h,k-tile); dk/dv then go to per-chunk fp32 slabs reduced afterwards
  const int ntiles_q = (Sq + BQT - 1) / BQT;
  const int tpc = (ntiles_q + gridDim.z - 1) / gridDim.z;
  const int qt_begin = blockIdx.z * tpc;
  const int qt_end = min(ntiles_q, qt_begin + tpc);
  // causal: this k-tile sees queries with qr >= k0 - diag_off
  int qt_start = qt_begin;
  if (CAUSAL) qt_start = max(qt_begin, (k0 - diag_off) / BQT);

  T* my_slab = slab + wid * 16 * LDT;

  // causal-skipped q-tiles: write the zeros the reductions expect (cheaper
  // than pre-zeroing the whole 25 MB slab on the host)
  for (int qt = qt_begin; qt < min(qt_start, qt_end); ++qt) {
    const int q0z = qt * BQT;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qr = q0z + wid * 16 + lg * 4 + r;
      if (qr >= Sq) continue;
#pragma unroll
      for (int t = 0; t < D / 16; ++t) dq_bh[(long)qr * D + t * 16 + lr] = 0.f;
      if (NEED_DBIAS) {
        const int qc = q0z + wid * 16 + lg * 4 + r;  // zero this block's kr cols
#pragma unroll
        for (int t2 = 0; t2 < 4; ++t2) {
          const int kr = k0 + t2 * 16 + lr;
          if (kr < Sk) dbias_h[(long)qc * Sk + kr] = 0.f;
        }
      }
    }
  }

  for (int qt = qt_start; qt < qt_end; ++qt) {
    const int q0 = qt * BQT;
    __syncthreads();  // previous iteration's reads done before restage
    // ---- stage Q, dO (+ transposed images) ----
    {
      const int j = threadIdx.x / (D / 8);
      const int d0 = (threadIdx.x % (D / 8)) * 8;
      constexpr int RPP = 256 / (D / 8);
      for (int jj = j; jj < BQT; jj += RPP) {
        const int qr = q0 + jj;
        VecT<T, 8> qv, dov;
        if (qr < Sq) {
          qv = vload<T, 8>(qbase + (long)qr * qs.ss + d0);
          dov = vload<T, 8>(dobase + (long)qr * dos.ss + d0);
        } else {
#pragma unroll
          for (int u = 0; u < 8; ++u) { qv.data[u] = T(0.f); dov.data[u] = T(0.f); }
        }
        vstore<T, 8>(&q_img[jj * LD + d0], qv);
        vstore<T, 8>(&do_img[jj * LD + d0], dov);
#pragma unroll
        for (int u = 0; u < 8; ++u) {
          qT_img[timg(d0 + u, jj)] = qv.data[u];
          doT_img[timg(d0 + u, jj)] = dov.data[u];
        }
      }
    }
    __syncthreads();

    // ---- S^T = K Q^T : M = wave's 16 k-rows, N = 64 q-cols, acc over d ----
    f32x4 st_acc[4], dpt_acc[4];
#pragma unroll
    for (int t = 0; t < 4; ++t) {
      st_acc[t] = {0.f, 0.f, 0.f, 0.f};
      dpt_acc[t] = {0.f, 0.f, 0.f, 0.f};
    }
#pragma unroll
    for (int t = 0; t < 4; ++t) {
#pragma unroll
      for (int s = 0; s < D / 32; ++s) {
        FR a = *reinterpret_cast<const FR*>(&kt_img[(wid * 16 + lr) * LD + s * 32 + lg * 8]);
        FR bfr = *reinterpret_cast<const FR*>(&q_img[(t * 16 + lr) * LD + s * 32 + lg * 8]);
        st_acc[t] = MFMA16B<T>::mma(a, bfr, st_acc[t]);
        // dP^T = V dO^T (same geometry, different operands)
        FR av = *reinterpret_cast<const FR*>(&v_img[(wid * 16 + lr) * LD + s * 32 + lg * 8]);
        FR bdo = *reinterpret_cast<const FR*>(&do_img[(t * 16 + lr) * LD + s * 32 + lg * 8]);
        dpt_acc[t] = MFMA16B<T>::mma(av, bdo, dpt_acc[t]);
      }
    }

    // ---- recompute P^T (and the dropout mask), compute dS^T ----
    // C layout: row (k-local) = lg*4 + r, col (q-local) = t*16 + lr
    // with attention-prob dropout: O = sum_k (m_k/keep) Pn_k V_k, so
    //   dV = (m/keep * Pn)^T dO   (mask applied to the slab P)
    //   dS = Pn * ((m/keep) * (dO V^T) - delta),  delta = rowsum(dO*O)
    const long long seed = seed_ptr ? *seed_ptr : 0;
    const float keep_inv = p_drop > 0.f ? 1.f / (1.f - p_drop) : 1.f;
    float pt[4][4], dst[4][4];
    const int krow0 = k0 + wid * 16 + lg * 4;
#pragma unroll
    for (int t = 0; t < 4; ++t) {
      const int qc = q0 + t * 16 + lr;
      const bool q_ok = qc < Sq;
      const float l = q_ok ? lse_bh[qc] : -INFINITY;
      const float dl = q_ok ? delta_bh[qc] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int kr = krow0 + r;
        float sv = st_acc[t][r];
        const bool k_ok = kr < Sk;
        if (HAS_BIAS && k_ok && q_ok) sv += DT<T>::to_f32(bias_bh[(long)qc * Sk + kr]);
        if (HAS_KPAD && k_ok) sv += kpad_b[kr];
        bool masked = !k_ok || !q_ok || (CAUSAL && kr > qc + diag_off);
        float p = (masked || l == -INFINITY) ? 0.f : __expf(sv - l);
        float mfac = 1.f;
        if (p_drop > 0.f) {
          const long long idx = ((long long)bh * Sq + qc) * Sk + kr;
          mfac = (hash_u01(seed, idx) >= p_drop) ? keep_inv : 0.f;
        }
        pt[t][r] = p * mfac;  // masked P -> slab (dV path)
        dst[t][r] = p * (dpt_acc[t][r] * mfac - dl);
      }
    }

    // ---- slab <- P^T ; dV += P^T dO ----
#pragma unroll
    for (int t = 0; t < 4; ++t)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        my_slab[(lg * 4 + r) * LDT + t * 16 + lr] = DT<T>::from_f32(pt[t][r]);
    __syncthreads();  // slab visibility (uniform control flow)
#pragma unroll
    for (int t = 0; t < D / 16; ++t) {
#pragma unroll
      for (int s = 0; s < BQT / 32; ++s) {
        FR a = *reinterpret_cast<const FR*>(&my_slab[lr * LDT + s * 32 + lg * 8]);
        FR bfr = *reinterpret_cast<const FR*>(&doT_img[timg(t * 16 + lr, s * 32 + lg * 8)]);
        dv_acc[t] = MFMA16B<T>::mma(a, bfr, dv_acc[t]);
      }
    }

    // ---- slab <- dS^T ; dK += dS^T Q ; ds_img <- dS (q-major) ----
#pragma unroll
    for (int t = 0; t < 4; ++t)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const T val = DT<T>::from_f32(dst[t][r]);
        my_slab[(lg * 4 + r) * LDT + t * 16 + lr] = val;
        ds_img[(t * 16 + lr) * LDT + wid * 16 + lg * 4 + r] = val;
      }
    __syncthreads();
#pragma unroll
    for (int t = 0; t < D / 16; ++t) {
#pragma unroll
      for (int s = 0; s < BQT / 32; ++s) {
        FR a = *reinterpret_cast<const FR*>(&my_slab[lr * LDT + s * 32 + lg * 8]);
        FR bfr = *reinterpret_cast<const FR*>(&qT_img[timg(t * 16 + lr, s * 32 + lg * 8)]);
        dk_acc[t] = MFMA16B<T>::mma(a, bfr, dk_acc[t]);
      }
    }

    // ---- dBias store (per-batch plane, summed over batch afterwards) ----
    if (NEED_DBIAS) {
#pragma unroll
      for (int t = 0; t < 4; ++t) {
        const int qc = q0 + t * 16 + lr;
        if (qc >= Sq) continue;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int kr = krow0 + r;
          if (kr < Sk) dbias_h[(long)qc * Sk + kr] = dst[t][r];
        }
      }
    }

    __syncthreads();  // ds_img complete across waves

    // ---- dQ(q-tile) = dS K : wave owns q rows [16*wid,+16), acc over k ----
    f32x4 dq_acc[D / 16];
#pragma unroll
    for (int t = 0; t < D / 16; ++t) dq_acc[t] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int t = 0; t < D / 16; ++t) {
#pragma unroll
      for (int s = 0; s < BKT / 32; ++s) {
        FR a = *reinterpret_cast<const FR*>(&ds_img[(wid * 16 + lr) * LDT + s * 32 + lg * 8]);
        FR bfr = *reinterpret_cast<const FR*>(&ktT_img[timg(t * 16 + lr, s * 32 + lg * 8)]);
        dq_acc[t] = MFMA16B<T>::mma(a, bfr, dq_acc[t]);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qr = q0 + wid * 16 + lg * 4 + r;
      if (qr >= Sq) continue;
#pragma unroll
      for (int t = 0; t < D / 16; ++t)
        dq_bh[(long)qr * D + t * 16 + lr] = dq_acc[t][r];
    }
  }

  // ---- write dK, dV ----
  if (dkv32 == nullptr) {  // no q-split: block exclusively owns these k rows
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int kr = k0 + wid * 16 + lg * 4 + r;
      if (kr >= Sk) continue;
#pragma unroll
      for (int t = 0; t < D / 16; ++t) {
        dkbase[(long)kr * dks.ss + t * 16 + lr] = DT<T>::from_f32(dk_acc[t][r]);
        dvbase[(long)kr * dvs.ss + t * 16 + lr] = DT<T>::from_f32(dv_acc[t][r]);
      }
    }
  } else {  // q-split: per-chunk fp32 slabs (2, NQC, B, H, Sk, D), plain stores
    const long plane = (long)B * H * Sk * D;
    float* dk_s = dkv32 + ((long)blockIdx.z * B * H + bh) * (long)Sk * D;
    float* dv_s = dk_s + (long)gridDim.z * plane;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int kr = k0 + wid * 16 + lg * 4 + r;
      if (kr >= Sk) continue;
#pragma unroll
      for (int t = 0; t < D / 16; ++t) {
        dk_s[(long)kr * D + t * 16 + lr] = dk_acc[t][r];
        dv_s[(long)kr * D + t * 16 + lr] = dv_acc[t][r];
      }
    }
  }
}

template <typename T, int D>
void launch_bwd(const torch::Tensor& dout, const torch::Tensor& q, const torch::Tensor& k,
                const torch::Tensor& v, const torch::Tensor& o, const torch::Tensor& lse,
                const c10::optional<torch::Tensor>& bias,
                const c10::optional<torch::Tensor>& kpad, bool causal, bool need_dbias,
                double p_drop, const c10::optional<torch::Tensor>& seed,
                torch::Tensor& dq32, torch::Tensor& dk, torch::Tensor& dv,
                torch::Tensor& dbias32, torch::Tensor* dkv32, int nqc,
                torch::Tensor& delta, hipStream_t stream) {
  const int B = q.size(0), H = q.size(1), Sq = q.size(2), Sk = k.size(2);
  BStr qs{q.stride(0), q.stride(1), q.stride(2)};
  BStr ks{k.stride(0), k.stride(1), k.stride(2)};
  BStr vs{v.stride(0), v.stride(1), v.stride(2)};
  BStr dos{dout.stride(0), dout.stride(1), dout.stride(2)};
  BStr oss{o.stride(0), o.stride(1), o.stride(2)};
  BStr dks{dk.stride(0), dk.stride(1), dk.stride(2)};
  BStr dvs{dv.stride(0), dv.stride(1), dv.stride(2)};

  {  // delta = rowsum(dO * O)
    const long rows = (long)H * Sq;
    const int wpb = 4;
    dim3 grid((rows + wpb - 1) / wpb, B);
    hipLaunchKernelGGL((attn_delta_kernel<T>), grid, dim3(wpb * WAVE), 0, stream,
                       (const T*)dout.data_ptr(), (const T*)o.data_ptr(),
                       delta.data_ptr<float>(), H, Sq, D, dos, oss);
    HIP_CHECK_KERNEL();
  }

  const bool has_bias = bias.has_value() && bias->defined();
  const bool has_kpad = kpad.has_value() && kpad->defined();
  const T* bias_p = has_bias ? (const T*)bias->data_ptr() : nullptr;
  const float* kpad_p = has_kpad ? kpad->data_ptr<float>() : nullptr;
  long bias_sb = 0, bias_sh = 0;
  if (has_bias) {
    bias_sb = bias->size(0) > 1 ? bias->stride(0) : 0;
    bias_sh = bias->size(1) > 1 ? bias->stride(1) : 0;
  }
  constexpr int LD = D + 8, LDT = BKT + 8;
  const int ds_extra = (LD == LDT) ? 0 : BQT * LDT;  // ds_img aliases q_img at D==64
  const int smem = sizeof(T) * (BKT * LD * 2 + BQT * LD * 2 + D * BKT * 3 +
                                ds_extra + 4 * 16 * LDT);
  dim3 grid((Sk + BKT - 1) / BKT, B * H, nqc);
#define LB(C, HB, HK, DB)                                                             \
  hipLaunchKernelGGL((attn_bwd_kernel<T, D, C, HB, HK, DB>), grid, dim3(256), smem,   \
                     stream, (const T*)dout.data_ptr(), (const T*)q.data_ptr(),       \
                     (const T*)k.data_ptr(), (const T*)v.data_ptr(),                  \
                     lse.data_ptr<float>(), delta.data_ptr<float>(), bias_p, kpad_p,  \
                     dq32.data_ptr<float>(), (T*)dk.data_ptr(), (T*)dv.data_ptr(),    \
                     need_dbias ? dbias32.data_ptr<float>() : nullptr,                \
                     dkv32 ? dkv32->data_ptr<float>() : nullptr, B, H, Sq, Sk,        \
                     qs, ks, vs, dos, dks, dvs, bias_sb, bias_sh, (float)p_drop,       \
                     (seed.has_value() && seed->defined())                             \
                         ? seed->data_ptr<long>() : nullptr)
  // 16 template combos
  if (causal) {
    if (has_bias && need_dbias) { if (has_kpad) LB(true, true, true, true); else LB(true, true, false, true); }
    else if (has_bias) { if (has_kpad) LB(true, true, true, false); else LB(true, true, false, false); }
    else { if (has_kpad) LB(true, false, true, false); else LB(true, false, false, false); }
  } else {
    if (has_bias && need_dbias) { if (has_kpad) LB(false, true, true, true); else LB(false, true, false, true); }
    else if (has_bias) { if (has_kpad) LB(false, true, true, false); else LB(false, true, false, false); }
    else { if (has_kpad) LB(false, false, true, false); else LB(false, false, false, false); }
  }
#undef LB
  HIP_CHECK_KERNEL();
}

// epilogue kernels (replace the aten sum/copy_/cast chains — census showed
// them costing ~1 ms/step at B=2): one fused pass each.

// dq = sum over the nkt per-k-tile fp32 slabs, written as T through the
// (b,h,s) strides of the BSHD-physical dq view. VEC floats along d per thread.
template <typename T, int VEC>
__global__ void slab_reduce_kernel(const float* __restrict__ slab,
                                   T* __restrict__ dq, int nkt, long slab_elems,
                                   int H, int Sq, int D,
                                   long s0, long s1, long s2) {
  long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * VEC;
  if (i >= slab_elems) return;
  float acc[VEC];
#pragma unroll
  for (int j = 0; j < VEC; ++j) acc[j] = 0.f;
  for (int t = 0; t < nkt; ++t) {
    VecT<float, VEC> v = vload<float, VEC>(slab + (long)t * slab_elems + i);
#pragma unroll
    for (int j = 0; j < VEC; ++j) acc[j] += v.data[j];
  }
  int d = (int)(i % D);
  long r = i / D;
  int sq = (int)(r % Sq);
  r /= Sq;
  int h = (int)(r % H);
  long b = r / H;
  T* out = dq + b * s0 + h * s1 + (long)sq * s2 + d;
  VecT<T, VEC> ov;
#pragma unroll
  for (int j = 0; j < VEC; ++j) ov.data[j] = DT<T>::from_f32(acc[j]);
  vstore<T, VEC>(out, ov);
}

// dbias = sum over batch of the fp32 per-batch planes + cast, one pass
template <typename T, int VEC>
__global__ void bsum_cast_kernel(const float* __restrict__ in, T* __restrict__ out,
                                 int B, long plane) {
  long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * VEC;
  if (i >= plane) return;
  float acc[VEC];
#pragma unroll
  for (int j = 0; j < VEC; ++j) acc[j] = 0.f;
  for (int b = 0; b < B; ++b) {
    VecT<float, VEC> v = vload<float, VEC>(in + (long)b * plane + i);
#pragma unroll
    for (int j = 0; j < VEC; ++j) acc[j] += v.data[j];
  }
  VecT<T, VEC> ov;
#pragma unroll
  for (int j = 0; j < VEC; ++j) ov.data[j] = DT<T>::from_f32(acc[j]);
  vstore<T, VEC>(out + i, ov);
}

template <typename T>
void launch_epilogue(const torch::Tensor& dq32, torch::Tensor& dq,
                     const torch::Tensor* dbias32, torch::Tensor* dbias,
                     hipStream_t stream) {
  const int nkt = dq32.size(0);
  const long slab = dq32.numel() / nkt;
  const int H = dq.size(1), Sq = dq.size(2), D = dq.size(3);
  constexpr int VEC = 4;
  long blocks = (slab / VEC + 255) / 256;
  hipLaunchKernelGGL((slab_reduce_kernel<T, VEC>), dim3(blocks), dim3(256), 0,
                     stream, dq32.data_ptr<float>(), (T*)dq.data_ptr(), nkt,
                     slab, H, Sq, D, dq.stride(0), dq.stride(1), dq.stride(2));
  HIP_CHECK_KERNEL();
  if (dbias32 != nullptr) {
    const int B = dbias32->size(0);
    const long plane = dbias32->numel() / B;
    long bblocks = (plane / VEC + 255) / 256;
    hipLaunchKernelGGL((bsum_cast_kernel<T, VEC>), dim3(bblocks), dim3(256), 0,
                       stream, dbias32->data_ptr<float>(), (T*)dbias->data_ptr(),
                       B, plane);
    HIP_CHECK_KERNEL();
  }
}

}  // namespace

// shared with attention_bwd_v3.hip: fused batch-sum + cast of the fp32
// per-batch dbias planes
void attn_dbias_reduce(const torch::Tensor& dbias32, torch::Tensor& dbias,
                       hipStream_t stream) {
  const int B = dbias32.size(0);
  const long plane = dbias32.numel() / B;
  if (plane % 4 != 0) {  // ragged: fall back to aten
    dbias.copy_(dbias32.sum(0, /*keepdim=*/true));
    return;
  }
  long blocks = (plane / 4 + 255) / 256;
  if (dbias.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL((bsum_cast_kernel<__hip_bfloat16, 4>), dim3(blocks),
                       dim3(256), 0, stream, dbias32.data_ptr<float>(),
                       (__hip_bfloat16*)dbias.data_ptr(), B, plane);
  } else {
    hipLaunchKernelGGL((bsum_cast_kernel<__half, 4>), dim3(blocks), dim3(256),
                       0, stream, dbias32.data_ptr<float>(),
                       (__half*)dbias.data_ptr(), B, plane);
  }
  HIP_CHECK_KERNEL();
}

std::vector<torch::Tensor> attn_bwd(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, torch::Tensor o, torch::Tensor lse,
                                    c10::optional<torch::Tensor> bias,
                                    c10::optional<torch::Tensor> kpad,
                                    bool causal, bool need_dbias,
                                    double p_drop, c10::optional<torch::Tensor> seed,
                                    c10::optional<torch::Tensor> packed) {
  TORCH_CHECK(q.stride(3) == 1 && k.stride(3) == 1 && v.stride(3) == 1 &&
              dout.stride(3) == 1 && o.stride(3) == 1, "head dim must be contiguous");
  const int B = q.size(0), H = q.size(1), Sq = q.size(2), Sk = k.size(2), D = q.size(3);
  auto stream = at::hip::getCurrentHIPStream().stream();
  auto opts = q.options();
  const long nkt = (Sk + 63) / 64;
  auto dq32 = torch::empty({nkt, B, H, Sq, D}, opts.dtype(torch::kFloat));
  const bool has_packed = packed.has_value() && packed->defined();
  // q-split factor: fill the 256 CUs (~2 waves of blocks) when B*H*nkt alone
  // cannot; per-chunk dk/dv fp32 slabs are reduced by the shared epilogue
  const int ntq = (Sq + 63) / 64;
  int nqc = 1;
  {
    const long base_blocks = nkt * B * H;
    while (base_blocks * nqc < 384 && nqc * 2 <= ntq) nqc *= 2;
    const char* env = std::getenv("MI355_ATTN_BWD_NQC");
    if (env) nqc = std::max(1, std::min(ntq, atoi(env)));
  }
  torch::Tensor dkv32;
  if (nqc > 1)
    dkv32 = torch::empty({2, (long)nqc, B, H, Sk, D}, opts.dtype(torch::kFloat));
  const int kslot = has_packed ? packed->size(2) - 2 : 0;  // 1 for qkv, 0 for kv
  auto dk = has_packed ? packed->select(2, kslot).permute({0, 2, 1, 3})
                       : torch::empty({B, Sk, H, D}, opts).permute({0, 2, 1, 3});
  auto dv = has_packed ? packed->select(2, kslot + 1).permute({0, 2, 1, 3})
                       : torch::empty({B, Sk, H, D}, opts).permute({0, 2, 1, 3});
  auto delta = torch::empty({B, H, Sq}, opts.dtype(torch::kFloat));
  torch::Tensor dbias32;
  if (need_dbias) {
    TORCH_CHECK(bias.has_value() && bias->size(0) == 1,
                "dbias path expects batch-broadcast bias (1,H,Sq,Sk)");
    dbias32 = torch::empty({B, H, Sq, Sk}, opts.dtype(torch::kFloat));
  }
  c10::optional<torch::Tensor> bias_c;
  if (bias.has_value() && bias->defined()) bias_c = bias->to(q.scalar_type()).contiguous();

  torch::Tensor* kvs = nqc > 1 ? &dkv32 : nullptr;
  if (q.scalar_type() == torch::kBFloat16 && D == 64)
    launch_bwd<__hip_bfloat16, 64>(dout, q, k, v, o, lse, bias_c, kpad, causal,
                                   need_dbias, p_drop, seed, dq32, dk, dv, dbias32, kvs, nqc, delta, stream);
  else if (q.scalar_type() == torch::kHalf && D == 64)
    launch_bwd<__half, 64>(dout, q, k, v, o, lse, bias_c, kpad, causal, need_dbias,
                           p_drop, seed, dq32, dk, dv, dbias32, kvs, nqc, delta, stream);
  else if (q.scalar_type() == torch::kBFloat16 && D == 32)
    launch_bwd<__hip_bfloat16, 32>(dout, q, k, v, o, lse, bias_c, kpad, causal,
                                   need_dbias, p_drop, seed, dq32, dk, dv, dbias32, kvs, nqc, delta, stream);
  else if (q.scalar_type() == torch::kHalf && D == 32)
    launch_bwd<__half, 32>(dout, q, k, v, o, lse, bias_c, kpad, causal, need_dbias,
                           p_drop, seed, dq32, dk, dv, dbias32, kvs, nqc, delta, stream);
  else
    TORCH_CHECK(false, "attn_bwd: unsupported dtype/D=", D);
  if (nqc > 1) {
    // reduce the per-chunk dk/dv slabs (same fused pass as dq)
    auto dk32 = dkv32.select(0, 0);
    auto dv32 = dkv32.select(0, 1);
    if (q.scalar_type() == torch::kBFloat16) {
      launch_epilogue<__hip_bfloat16>(dk32, dk, nullptr, nullptr, stream);
      launch_epilogue<__hip_bfloat16>(dv32, dv, nullptr, nullptr, stream);
    } else {
      launch_epilogue<__half>(dk32, dk, nullptr, nullptr, stream);
      launch_epilogue<__half>(dv32, dv, nullptr, nullptr, stream);
    }
  }

  // dq: sum the per-k-tile fp32 slabs -> low-precision BSHD-physical view;
  // dbias: batch-sum + cast. Single fused pass each (no aten sum/copy/cast).
  // With `packed` ((B,Sq,3,H,D), self-attention) dq/dk/dv are views into the
  // caller's buffer, so autograd sees ONE fused qkv grad and never cats.
  // packed slots: 3 = fused qkv grad (self-attention, Sq==Sk), 2 = fused
  // kv grad (cross-attention: dq is its own tensor)
  torch::Tensor dq;
  if (has_packed && packed->size(2) == 3) {
    TORCH_CHECK(Sq == Sk, "packed qkv grads require self-attention");
    dq = packed->select(2, 0).permute({0, 2, 1, 3});
  } else
    dq = torch::empty({B, Sq, H, D}, opts).permute({0, 2, 1, 3});
  const long plane = need_dbias ? dbias32.numel() / B : 0;
  const bool fused_ok = ((long)B * H * Sq * D) % 4 == 0 && plane % 4 == 0;
  torch::Tensor dbias;
  if (need_dbias) dbias = torch::empty({1, H, Sq, Sk}, opts);
  if (fused_ok) {
    if (q.scalar_type() == torch::kBFloat16)
      launch_epilogue<__hip_bfloat16>(dq32, dq, need_dbias ? &dbias32 : nullptr,
                                      need_dbias ? &dbias : nullptr, stream);
    else
      launch_epilogue<__half>(dq32, dq, need_dbias ? &dbias32 : nullptr,
                              need_dbias ? &dbias : nullptr, stream);
  } else {
    dq.copy_(dq32.sum(0));
    if (need_dbias) dbias.copy_(dbias32.sum(0, /*keepdim=*/true));
  }
  std::vector<torch::Tensor> outs = {dq, dk, dv};
  if (need_dbias) outs.push_back(dbias);
  return outs;
}
