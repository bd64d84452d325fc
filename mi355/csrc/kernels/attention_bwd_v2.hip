// Flash attention BACKWARD v2 for CDNA4/gfx950 — swapped-operand 32x32 MFMA
// with in-register softmax recompute (companion to attention_fwd_v2.hip).
//
// Orientation: C tiles are (q rows in regs, k cols = lane&31), so each
// wave's 32 keys' K and V B-FRAGMENTS are loaded from HBM ONCE into
// registers and never restaged. Per 32-query iteration:
//   S   = Q K^T        (A = q_img,  B = kfrag)      -> P = exp(S+bias-lse)
//   dP  = dO V^T       (A = do_img, B = vfrag)
//   dS  = P (mask*dP - delta)
//   dV^T += dO^T P     (A = doT_img, B = repack(P))   repack = cvt_pk +
//   dK^T += Q^T dS     (A = qT_img,  B = repack(dS))  v_permlane32_swap
//   dQ^T(partial) = K^T dS^T (A = ktT_img, B = ds_img b128 rows)
// dQ partials go to one fp32 slab PER WAVE (summed on the host — no
// atomics); dBias is plain per-batch stores like bwd v1.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../common.h"

namespace {

typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef _Float16 f16x8 __attribute__((ext_vector_type(8)));

template <typename T> struct MB2;
template <> struct MB2<__hip_bfloat16> {
  using frag = bf16x8;
  static DEVINL f32x16 mma(frag a, frag b, f32x16 c) {
    return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
  }
  static DEVINL unsigned pack2(float lo, float hi) {
    union { __hip_bfloat162 h2; unsigned u; } u;
    u.h2 = __hip_bfloat162(__float2bfloat16(lo), __float2bfloat16(hi));
    return u.u;
  }
};
template <> struct MB2<__half> {
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

constexpr int KB = 128;   // keys per block (4 waves x 32)
constexpr float NEGB2 = -1e30f;

struct S3b { long sb, sh, ss; };

template <typename T>
__global__ void delta_kernel_v2(const T* __restrict__ dout, const T* __restrict__ o,
                                float* __restrict__ delta, int H, int Sq, int D,
                                S3b dos, S3b os) {
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

template <typename T, bool CAUSAL, bool HAS_BIAS, bool HAS_KPAD, bool NEED_DBIAS>
__global__ __launch_bounds__(256, 2) void attn_bwd_v2_kernel(
    const T* __restrict__ dout, const T* __restrict__ q, const T* __restrict__ k,
    const T* __restrict__ v, const float* __restrict__ lse, const float* __restrict__ delta,
    const T* __restrict__ bias, const float* __restrict__ kpad,
    float* __restrict__ dq32, T* __restrict__ dk, T* __restrict__ dv,
    float* __restrict__ dbias32,
    int B, int H, int Sq, int Sk,
    S3b qs, S3b ks, S3b vs, S3b dos, S3b dks, S3b dvs,
    long bias_sb, long bias_sh, float p_drop, const long* __restrict__ seed_ptr) {
  constexpr int D = 64;
  using FR = typename MB2<T>::frag;
  constexpr int QLD = D + 8;    // q/do images [32][72]
  constexpr int TLD = 32 + 8;   // transposed images [64][40], ds [32][40]

  __shared__ T q_img[32 * QLD];
  __shared__ T do_img[32 * QLD];
  __shared__ T qT_img[D * TLD];
  __shared__ T doT_img[D * TLD];
  __shared__ T ktT_img[4][D * TLD];   // per-wave K^T (staged once)
  __shared__ T ds_img[4][32 * TLD];   // per-wave dS rows (q-major)
  __shared__ T bias_img[HAS_BIAS ? 32 * (KB + 8) : 1];  // [q][k-block]
  __shared__ float lse_lds[32], delta_lds[32];

  const int bh = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const int kb0 = blockIdx.x * KB;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int lk = lane & 31;          // key column within the wave's 32
  const int half = lane >> 5;
  const int my_k = kb0 + wid * 32 + lk;   // this lane's key row
  const int my_k_c = min(my_k, Sk - 1);

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
  // one dq slab per (k-wave): slab index = global wave id
  const long wslab = (long)blockIdx.x * 4 + wid;
  float* dq_w = dq32 + ((wslab * B + b) * H + h) * (long)Sq * D;
  float* dbias_h = NEED_DBIAS ? dbias32 + ((long)b * H + h) * (long)Sq * Sk : nullptr;

  // ---- K/V B-fragments for this wave's 32 keys, loaded once ----
  FR kfrag[4], vfrag[4];
#pragma unroll
  for (int s = 0; s < 4; ++s) {
    kfrag[s] = *reinterpret_cast<const FR*>(kbase + (long)my_k_c * ks.ss + s * 16 + half * 8);
    vfrag[s] = *reinterpret_cast<const FR*>(vbase + (long)my_k_c * vs.ss + s * 16 + half * 8);
  }
  // ---- per-wave K^T image (A-operand of the dQ product), staged once ----
  {
    // lane writes its key's row transposed: ktT[d][k-local]
    const T* krow = kbase + (long)my_k_c * ks.ss;
    for (int d = half * 32; d < half * 32 + 32; ++d)
      ktT_img[wid][d * TLD + lk] = krow[d];
  }

  // accumulators: dV^T / dK^T (2 d-tiles each, cols = this lane's key)
  f32x16 dv_acc[2], dk_acc[2];
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    dv_acc[0][r] = 0.f; dv_acc[1][r] = 0.f;
    dk_acc[0][r] = 0.f; dk_acc[1][r] = 0.f;
  }

  const int diag_off = Sk - Sq;
  int qt_start = 0;
  if (CAUSAL) qt_start = max(0, (kb0 - diag_off) / 32);
  const long long seed = seed_ptr ? *seed_ptr : 0;
  const float keep_inv = p_drop > 0.f ? 1.f / (1.f - p_drop) : 1.f;

  // staging split for q/do images: 256 threads x 8 elems = one 32x64 tile
  const int sj = threadIdx.x / (D / 8);
  const int sd = (threadIdx.x % (D / 8)) * 8;
  // bias staging: [32 q rows][KB cols]
  const int bj = threadIdx.x / (KB / 8);
  const int bc = (threadIdx.x % (KB / 8)) * 8;
  constexpr int BRPP = 256 / (KB / 8);  // 16 q-rows per pass

  for (int qt = qt_start; qt * 32 < Sq; ++qt) {
    const int q0 = qt * 32;
    __syncthreads();
    {  // stage q/do (+ transposed), lse/delta, bias
      const int qr = min(q0 + sj, Sq - 1);
      VecT<T, 8> qv = vload<T, 8>(qbase + (long)qr * qs.ss + sd);
      VecT<T, 8> dov = vload<T, 8>(dobase + (long)qr * dos.ss + sd);
      vstore<T, 8>(&q_img[sj * QLD + sd], qv);
      vstore<T, 8>(&do_img[sj * QLD + sd], dov);
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        qT_img[(sd + u) * TLD + sj] = qv.data[u];
        doT_img[(sd + u) * TLD + sj] = dov.data[u];
      }
      if (threadIdx.x < 32) {
        const int qr2 = min(q0 + (int)threadIdx.x, Sq - 1);
        lse_lds[threadIdx.x] = lse_bh[qr2];
        delta_lds[threadIdx.x] = delta_bh[qr2];
      }
      if (HAS_BIAS) {
#pragma unroll
        for (int p = 0; p < 32 / BRPP; ++p) {
          const int qb = min(q0 + bj + p * BRPP, Sq - 1);
          const int kbc = kb0 + bc;
          VecT<T, 8> bv;
          if (kbc + 8 <= Sk) {
            bv = vload<T, 8>(bias_bh + (long)qb * Sk + kbc);
          } else {
#pragma unroll
            for (int u = 0; u < 8; ++u)
              bv.data[u] = (kbc + u < Sk) ? bias_bh[(long)qb * Sk + kbc + u] : T(0.f);
          }
          vstore<T, 8>(&bias_img[(bj + p * BRPP) * (KB + 8) + bc], bv);
        }
      }
    }
    __syncthreads();

    // ---- S = Q K^T and dP = dO V^T (C: q rows in regs, k col = lane) ----
    f32x16 sacc, dpacc;
#pragma unroll
    for (int r = 0; r < 16; ++r) { sacc[r] = 0.f; dpacc[r] = 0.f; }
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      FR aq = *reinterpret_cast<const FR*>(&q_img[lk * QLD + s * 16 + half * 8]);
      sacc = MB2<T>::mma(aq, kfrag[s], sacc);
      FR ad = *reinterpret_cast<const FR*>(&do_img[lk * QLD + s * 16 + half * 8]);
      dpacc = MB2<T>::mma(ad, vfrag[s], dpacc);
    }
    __builtin_amdgcn_s_setprio(0);

    // ---- recompute P, dS (q = (r&3)+8*(r>>2)+4*half in-lane) ----
    float pp[16], ds[16];
    const bool k_ok = my_k < Sk;
    const float kp = (HAS_KPAD && k_ok) ? kpad_b[my_k] : 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int qq = (r & 3) + 8 * (r >> 2) + 4 * half;
      const int qr = q0 + qq;
      const bool q_ok = qr < Sq;
      float sv = sacc[r];
      if (HAS_BIAS) sv += DT<T>::to_f32(bias_img[qq * (KB + 8) + wid * 32 + lk]);
      sv += kp;
      const float l = q_ok ? lse_lds[qq] : -INFINITY;
      bool masked = !k_ok || !q_ok || (CAUSAL && my_k > qr + diag_off);
      float p = (masked || l == -INFINITY) ? 0.f : __expf(sv - l);
      float mfac = 1.f;
      if (p_drop > 0.f) {
        const long long idx = ((long long)bh * Sq + qr) * Sk + my_k;
        mfac = (hash_u01(seed, idx) >= p_drop) ? keep_inv : 0.f;
      }
      const float dl = q_ok ? delta_lds[qq] : 0.f;
      ds[r] = p * (dpacc[r] * mfac - dl);
      pp[r] = p * mfac;  // masked P feeds dV
      if (NEED_DBIAS && q_ok && k_ok)
        dbias_h[(long)qr * Sk + my_k] = ds[r];  // masked -> ds == 0
    }

    // ---- repacks along the q axis -> B fragments ----
    unsigned pbP[8], pbS[8];
#pragma unroll
    for (int g = 0; g < 2; ++g) {
      const int o = g * 8;
      {
        unsigned a0 = MB2<T>::pack2(pp[o + 0], pp[o + 1]);
        unsigned b0 = MB2<T>::pack2(pp[o + 4], pp[o + 5]);
        unsigned a1 = MB2<T>::pack2(pp[o + 2], pp[o + 3]);
        unsigned b1 = MB2<T>::pack2(pp[o + 6], pp[o + 7]);
        auto r0 = __builtin_amdgcn_permlane32_swap(a0, b0, false, false);
        auto r1 = __builtin_amdgcn_permlane32_swap(a1, b1, false, false);
        pbP[g * 4 + 0] = r0[0]; pbP[g * 4 + 1] = r1[0];
        pbP[g * 4 + 2] = r0[1]; pbP[g * 4 + 3] = r1[1];
      }
      {
        unsigned a0 = MB2<T>::pack2(ds[o + 0], ds[o + 1]);
        unsigned b0 = MB2<T>::pack2(ds[o + 4], ds[o + 5]);
        unsigned a1 = MB2<T>::pack2(ds[o + 2], ds[o + 3]);
        unsigned b1 = MB2<T>::pack2(ds[o + 6], ds[o + 7]);
        auto r0 = __builtin_amdgcn_permlane32_swap(a0, b0, false, false);
        auto r1 = __builtin_amdgcn_permlane32_swap(a1, b1, false, false);
        pbS[g * 4 + 0] = r0[0]; pbS[g * 4 + 1] = r1[0];
        pbS[g * 4 + 2] = r0[1]; pbS[g * 4 + 3] = r1[1];
      }
    }

    // ---- dV^T += dO^T P ; dK^T += Q^T dS ----
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int dt = 0; dt < 2; ++dt) {
#pragma unroll
      for (int g = 0; g < 2; ++g) {
        FR adoT = *reinterpret_cast<const FR*>(
            &doT_img[(dt * 32 + lk) * TLD + g * 16 + half * 8]);
        struct U4 { unsigned x[4]; } up{pbP[g * 4], pbP[g * 4 + 1], pbP[g * 4 + 2], pbP[g * 4 + 3]};
        dv_acc[dt] = MB2<T>::mma(adoT, __builtin_bit_cast(FR, up), dv_acc[dt]);
        FR aqT = *reinterpret_cast<const FR*>(
            &qT_img[(dt * 32 + lk) * TLD + g * 16 + half * 8]);
        struct U4b { unsigned x[4]; } us{pbS[g * 4], pbS[g * 4 + 1], pbS[g * 4 + 2], pbS[g * 4 + 3]};
        dk_acc[dt] = MB2<T>::mma(aqT, __builtin_bit_cast(FR, us), dk_acc[dt]);
      }
    }
    __builtin_amdgcn_s_setprio(0);

    // ---- dS -> wave-local q-major image; dQ^T partial = K^T dS^T ----
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int qq = (r & 3) + 8 * (r >> 2) + 4 * half;
      ds_img[wid][qq * TLD + lk] = DT<T>::from_f32(ds[r]);
    }
    __syncthreads();  // wave-local, but uniform barrier for lane visibility
    f32x16 dq_acc[2];
#pragma unroll
    for (int r = 0; r < 16; ++r) { dq_acc[0][r] = 0.f; dq_acc[1][r] = 0.f; }
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int dt = 0; dt < 2; ++dt) {
#pragma unroll
      for (int g = 0; g < 2; ++g) {
        FR akT = *reinterpret_cast<const FR*>(
            &ktT_img[wid][(dt * 32 + lk) * TLD + g * 16 + half * 8]);
        FR bds = *reinterpret_cast<const FR*>(
            &ds_img[wid][lk * TLD + g * 16 + half * 8]);
        dq_acc[dt] = MB2<T>::mma(akT, bds, dq_acc[dt]);
      }
    }
    __builtin_amdgcn_s_setprio(0);
    // store the wave's dQ^T partial: C cols = q (lane), rows = d (regs)
    {
      const int qr = q0 + lk;
      if (qr < Sq) {
        float* dst = dq_w + (long)qr * D;
#pragma unroll
        for (int dt = 0; dt < 2; ++dt)
#pragma unroll
          for (int grp = 0; grp < 4; ++grp) {
            const int d0 = dt * 32 + grp * 8 + half * 4;
            VecT<float, 4> ov;
#pragma unroll
            for (int u = 0; u < 4; ++u) ov.data[u] = dq_acc[dt][grp * 4 + u];
            vstore<float, 4>(dst + d0, ov);
          }
      }
    }
  }
  // causal: zero the q rows this block never visited (host uses empty alloc)
  if (CAUSAL) {
    for (int qt = 0; qt < qt_start; ++qt) {
      const int qr = qt * 32 + lk;
      if (qr < Sq) {
        float* dst = dq_w + (long)qr * D;
#pragma unroll
        for (int dt = 0; dt < 2; ++dt)
#pragma unroll
          for (int grp = 0; grp < 4; ++grp) {
            const int d0 = dt * 32 + grp * 8 + half * 4;
            VecT<float, 4> zv;
#pragma unroll
            for (int u = 0; u < 4; ++u) zv.data[u] = 0.f;
            vstore<float, 4>(dst + d0, zv);
          }
        if (NEED_DBIAS && my_k < Sk) {
          // zero this lane's key column for every skipped q row of the tile
          for (int r2 = 0; r2 < 32; ++r2) {
            const int qr2 = qt * 32 + r2;
            if (qr2 < Sq) dbias_h[(long)qr2 * Sk + my_k] = 0.f;
          }
        }
      }
    }
  }

  // ---- write dK / dV (C rows = d in regs, col = this lane's key) ----
  if (my_k < Sk) {
#pragma unroll
    for (int dt = 0; dt < 2; ++dt)
#pragma unroll
      for (int grp = 0; grp < 4; ++grp) {
        const int d0 = dt * 32 + grp * 8 + half * 4;
        VecT<T, 4> kv, vv;
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          kv.data[u] = DT<T>::from_f32(dk_acc[dt][grp * 4 + u]);
          vv.data[u] = DT<T>::from_f32(dv_acc[dt][grp * 4 + u]);
        }
        vstore<T, 4>(dkbase + (long)my_k * dks.ss + d0, kv);
        vstore<T, 4>(dvbase + (long)my_k * dvs.ss + d0, vv);
      }
  }
}

template <typename T>
void launch_bwd_v2(const torch::Tensor& dout, const torch::Tensor& q, const torch::Tensor& k,
                   const torch::Tensor& v, const torch::Tensor& o, const torch::Tensor& lse,
                   const c10::optional<torch::Tensor>& bias,
                   const c10::optional<torch::Tensor>& kpad, bool causal, bool need_dbias,
                   double p_drop, const c10::optional<torch::Tensor>& seed,
                   torch::Tensor& dq32, torch::Tensor& dk, torch::Tensor& dv,
                   torch::Tensor& dbias32, torch::Tensor& delta, hipStream_t stream) {
  const int B = q.size(0), H = q.size(1), Sq = q.size(2), Sk = k.size(2);
  constexpr int D = 64;
  S3b qs{q.stride(0), q.stride(1), q.stride(2)};
  S3b ks{k.stride(0), k.stride(1), k.stride(2)};
  S3b vs{v.stride(0), v.stride(1), v.stride(2)};
  S3b dos{dout.stride(0), dout.stride(1), dout.stride(2)};
  S3b oss{o.stride(0), o.stride(1), o.stride(2)};
  S3b dks{dk.stride(0), dk.stride(1), dk.stride(2)};
  S3b dvs{dv.stride(0), dv.stride(1), dv.stride(2)};
  {
    const long rows = (long)H * Sq;
    dim3 grid((rows + 3) / 4, B);
    hipLaunchKernelGGL((delta_kernel_v2<T>), grid, dim3(256), 0, stream,
                       (const T*)dout.data_ptr(), (const T*)o.data_ptr(),
                       delta.data_ptr<float>(), H, Sq, D, dos, oss);
    HIP_CHECK_KERNEL();
  }
  const bool hb = bias.has_value() && bias->defined();
  const bool hk = kpad.has_value() && kpad->defined();
  const T* bias_p = hb ? (const T*)bias->data_ptr() : nullptr;
  const float* kpad_p = hk ? kpad->data_ptr<float>() : nullptr;
  long bias_sb = 0, bias_sh = 0;
  if (hb) {
    bias_sb = bias->size(0) > 1 ? bias->stride(0) : 0;
    bias_sh = bias->size(1) > 1 ? bias->stride(1) : 0;
  }
  dim3 grid((Sk + KB - 1) / KB, B * H);
#define LBV2(C, HB, HK, DB)                                                            \
  hipLaunchKernelGGL((attn_bwd_v2_kernel<T, C, HB, HK, DB>), grid, dim3(256), 0,       \
                     stream, (const T*)dout.data_ptr(), (const T*)q.data_ptr(),        \
                     (const T*)k.data_ptr(), (const T*)v.data_ptr(),                   \
                     lse.data_ptr<float>(), delta.data_ptr<float>(), bias_p, kpad_p,   \
                     dq32.data_ptr<float>(), (T*)dk.data_ptr(), (T*)dv.data_ptr(),     \
                     need_dbias ? dbias32.data_ptr<float>() : nullptr, B, H, Sq, Sk,   \
                     qs, ks, vs, dos, dks, dvs, bias_sb, bias_sh, (float)p_drop,       \
                     (seed.has_value() && seed->defined()) ? seed->data_ptr<long>()    \
                                                           : nullptr)
  if (causal) {
    if (hb && need_dbias) { if (hk) LBV2(true, true, true, true); else LBV2(true, true, false, true); }
    else if (hb) { if (hk) LBV2(true, true, true, false); else LBV2(true, true, false, false); }
    else { if (hk) LBV2(true, false, true, false); else LBV2(true, false, false, false); }
  } else {
    if (hb && need_dbias) { if (hk) LBV2(false, true, true, true); else LBV2(false, true, false, true); }
    else if (hb) { if (hk) LBV2(false, true, true, false); else LBV2(false, true, false, false); }
    else { if (hk) LBV2(false, false, true, false); else LBV2(false, false, false, false); }
  }
#undef LBV2
  HIP_CHECK_KERNEL();
}

}  // namespace

std::vector<torch::Tensor> attn_bwd_v2(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                                       torch::Tensor v, torch::Tensor o, torch::Tensor lse,
                                       c10::optional<torch::Tensor> bias,
                                       c10::optional<torch::Tensor> kpad,
                                       bool causal, bool need_dbias,
                                       double p_drop, c10::optional<torch::Tensor> seed) {
  TORCH_CHECK(q.stride(3) == 1 && k.stride(3) == 1 && v.stride(3) == 1 &&
              dout.stride(3) == 1 && o.stride(3) == 1 && q.size(3) == 64,
              "attn_bwd_v2: D=64, d contiguous");
  const int B = q.size(0), H = q.size(1), Sq = q.size(2), Sk = k.size(2);
  auto opts = q.options();
  const long nw = (long)((Sk + KB - 1) / KB) * 4;  // dq slabs: one per k-wave
  auto dq32 = torch::empty({nw, B, H, Sq, 64}, opts.dtype(torch::kFloat));
  auto dk = torch::empty({B, Sk, H, 64}, opts).permute({0, 2, 1, 3});
  auto dv = torch::empty({B, Sk, H, 64}, opts).permute({0, 2, 1, 3});
  auto delta = torch::empty({B, H, Sq}, opts.dtype(torch::kFloat));
  torch::Tensor dbias32;
  if (need_dbias) {
    TORCH_CHECK(bias.has_value() && bias->size(0) == 1,
                "dbias expects batch-broadcast bias (1,H,Sq,Sk)");
    dbias32 = torch::empty({B, H, Sq, Sk}, opts.dtype(torch::kFloat));
  }
  c10::optional<torch::Tensor> bias_c;
  if (bias.has_value() && bias->defined()) bias_c = bias->to(q.scalar_type()).contiguous();
  auto stream = at::hip::getCurrentHIPStream().stream();
  if (q.scalar_type() == torch::kBFloat16)
    launch_bwd_v2<__hip_bfloat16>(dout, q, k, v, o, lse, bias_c, kpad, causal,
                                  need_dbias, p_drop, seed, dq32, dk, dv, dbias32,
                                  delta, stream);
  else if (q.scalar_type() == torch::kHalf)
    launch_bwd_v2<__half>(dout, q, k, v, o, lse, bias_c, kpad, causal, need_dbias,
                          p_drop, seed, dq32, dk, dv, dbias32, delta, stream);
  else
    TORCH_CHECK(false, "attn_bwd_v2: bf16/f16 only");
  auto dq = torch::empty({B, Sq, H, 64}, opts).permute({0, 2, 1, 3});
  dq.copy_(dq32.sum(0));
  std::vector<torch::Tensor> outs = {dq, dk, dv};
  if (need_dbias)
    outs.push_back(dbias32.sum(0, /*keepdim=*/true).to(q.scalar_type()));
  return outs;
}
