// Fused upsampled pixelwise cross-entropy (reference
// Scaling_model_training.ipynb:cell-39/52 — SegFormer's loss upsamples the
// (B,150,h,w) logits to the (B,H,W) label size and takes CE with
// ignore_index).  The torch path materializes the (B,150,H,W) fp32
// upsampled tensor (~1.26 GB at the reference config) through interpolate +
// softmax + nll fwd AND bwd — 29% of the SegFormer train step.  This pair
// computes the loss straight from the low-res logits with on-the-fly
// bilinear taps (same align_corners=False index math as seg_postprocess)
// and scatters the backward through a per-tile LDS accumulator, cutting the
// loss path's HBM traffic ~100x.
//
// fwd: one thread per output pixel — online softmax over C channels of
//      4-tap interpolated logits, per-pixel LSE saved for backward,
//      block-reduced loss/valid-count atomically accumulated.
// bwd: one block per 32x32 output tile — dLow accumulated in a dynamic-LDS
//      [C][ph][pw] fp32 patch (ds atomics), then atomically added to the
//      global fp32 dLow (patch borders overlap between tiles).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../common.h"

namespace {

struct Tap {
  int y0, x0, y1, x1;
  float w00, w01, w10, w11;
};

DEVINL Tap tap_at(int y, int x, int Hi, int Wi, int Ho, int Wo) {
  // torch area_pixel_compute_source_index (align_corners=False)
  const float sy = fmaxf(((y + 0.5f) * Hi) / Ho - 0.5f, 0.f);
  const float sx = fmaxf(((x + 0.5f) * Wi) / Wo - 0.5f, 0.f);
  Tap t;
  t.y0 = min(Hi - 1, (int)sy);
  t.x0 = min(Wi - 1, (int)sx);
  t.y1 = min(Hi - 1, t.y0 + 1);
  t.x1 = min(Wi - 1, t.x0 + 1);
  const float fy = sy - t.y0, fx = sx - t.x0;
  t.w00 = (1 - fy) * (1 - fx);
  t.w01 = (1 - fy) * fx;
  t.w10 = fy * (1 - fx);
  t.w11 = fy * fx;
  return t;
}

template <typename T>
__global__ void seg_ce_fwd_kernel(const T* __restrict__ logits,
                                  const long* __restrict__ labels,
                                  float* __restrict__ lse,
                                  float* __restrict__ loss_sum,
                                  float* __restrict__ nvalid,
                                  int B, int C, int Hi, int Wi, int Ho, int Wo,
                                  long ignore_index) {
  __shared__ float red[16];
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long total = (long)B * Ho * Wo;
  float my_loss = 0.f, my_valid = 0.f;
  if (idx < total) {
    const int x = idx % Wo;
    const int y = (idx / Wo) % Ho;
    const int b = idx / ((long)Ho * Wo);
    const long lab = labels[idx];
    if (lab == ignore_index || lab < 0 || lab >= C) {
      lse[idx] = 0.f;
    } else {
      const Tap t = tap_at(y, x, Hi, Wi, Ho, Wo);
      const long plane = (long)Hi * Wi;
      const T* base = logits + (long)b * C * plane;
      const long o00 = (long)t.y0 * Wi + t.x0, o01 = (long)t.y0 * Wi + t.x1;
      const long o10 = (long)t.y1 * Wi + t.x0, o11 = (long)t.y1 * Wi + t.x1;
      float m = -INFINITY, s = 0.f, zlab = 0.f;
      for (int c = 0; c < C; ++c) {
        const T* p = base + (long)c * plane;
        const float z = t.w00 * DT<T>::to_f32(p[o00]) +
                        t.w01 * DT<T>::to_f32(p[o01]) +
                        t.w10 * DT<T>::to_f32(p[o10]) +
                        t.w11 * DT<T>::to_f32(p[o11]);
        const float mn = fmaxf(m, z);
        s = s * __expf(m - mn) + __expf(z - mn);
        m = mn;
        if (c == (int)lab) zlab = z;
      }
      const float l = m + __logf(s);
      lse[idx] = l;
      my_loss = l - zlab;
      my_valid = 1.f;
    }
  }
  const float bl = block_reduce_sum(my_loss, red);
  const float bv = block_reduce_sum(my_valid, red);
  if (threadIdx.x == 0) {
    atomicAdd(loss_sum, bl);
    atomicAdd(nvalid, bv);
  }
}

// bwd: grid (tiles_x, tiles_y, B); 256 threads each own a 2x2 sub-block of
// the 32x32 output tile.  dlow32 must be zero-initialized.
template <typename T, int OT /*16*/>
__global__ void seg_ce_bwd_kernel(const T* __restrict__ logits,
                                  const long* __restrict__ labels,
                                  const float* __restrict__ lse,
                                  const float* __restrict__ gout,
                                  const float* __restrict__ nvalid,
                                  float* __restrict__ dlow32,
                                  int B, int C, int Hi, int Wi, int Ho, int Wo,
                                  long ignore_index) {
  extern __shared__ float acc[];  // [C][ph][pw] fp32 accum + bf16 patch
  const int b = blockIdx.z;
  const int oy0 = blockIdx.y * OT, ox0 = blockIdx.x * OT;
  // low-res patch bounds covering this tile's taps
  const Tap t00 = tap_at(oy0, ox0, Hi, Wi, Ho, Wo);
  const Tap t11 = tap_at(min(oy0 + OT - 1, Ho - 1), min(ox0 + OT - 1, Wo - 1),
                         Hi, Wi, Ho, Wo);
  const int py0 = t00.y0, px0 = t00.x0;
  const int ph = t11.y1 - py0 + 1, pw = t11.x1 - px0 + 1;
  const int pelems = C * ph * pw;
  const long plane = (long)Hi * Wi;
  const T* base = logits + (long)b * C * plane;
  for (int i = threadIdx.x; i < pelems; i += blockDim.x) acc[i] = 0.f;
  __syncthreads();

  const float gs = gout[0] / fmaxf(nvalid[0], 1.f);

  // one output pixel per thread (OT=16: 256 threads = the whole tile);
  // small tiles keep the LDS patch ~6 KiB so 4+ blocks/CU hide the
  // read->exp->ds_add dependency chains that made a 32-tile version
  // (1 block/CU) latency-bound
  {
    const int y = oy0 + (threadIdx.x >> 4), x = ox0 + (threadIdx.x & 15);
    const long pidx = ((long)b * Ho + y) * Wo + x;
    const long lab = (y < Ho && x < Wo) ? labels[pidx] : ignore_index;
    if (y < Ho && x < Wo && lab != ignore_index && lab >= 0 && lab < C) {
      const Tap t = tap_at(y, x, Hi, Wi, Ho, Wo);
      const long o00 = (long)t.y0 * Wi + t.x0, o01 = (long)t.y0 * Wi + t.x1;
      const long o10 = (long)t.y1 * Wi + t.x0, o11 = (long)t.y1 * Wi + t.x1;
      const int a00 = (t.y0 - py0) * pw + (t.x0 - px0);
      const int a01 = (t.y0 - py0) * pw + (t.x1 - px0);
      const int a10 = (t.y1 - py0) * pw + (t.x0 - px0);
      const int a11 = (t.y1 - py0) * pw + (t.x1 - px0);
      const float l = lse[pidx];
      // channel start staggered per lane: in lockstep-c every lane hammers
      // the SAME channel plane (LDS same-address atomic serialize);
      // staggered, lanes touch different planes
      const int c0 = threadIdx.x % C;
      for (int cc = 0; cc < C; ++cc) {
        const int c = (cc + c0 >= C) ? cc + c0 - C : cc + c0;
        const T* p = base + (long)c * plane;
        const float z = t.w00 * DT<T>::to_f32(p[o00]) +
                        t.w01 * DT<T>::to_f32(p[o01]) +
                        t.w10 * DT<T>::to_f32(p[o10]) +
                        t.w11 * DT<T>::to_f32(p[o11]);
        const float g = (__expf(z - l) - (c == (int)lab ? 1.f : 0.f)) * gs;
        float* ac = acc + (long)c * ph * pw;
        // unsafeAtomicAdd = HW ds_add/global_add_f32 (plain atomicAdd on
        // fp32 compiles to a CAS loop on ROCm)
        unsafeAtomicAdd(&ac[a00], t.w00 * g);
        unsafeAtomicAdd(&ac[a01], t.w01 * g);
        unsafeAtomicAdd(&ac[a10], t.w10 * g);
        unsafeAtomicAdd(&ac[a11], t.w11 * g);
      }
    }
  }
  __syncthreads();
  // patch -> global fp32 dlow (borders shared with neighbor tiles: atomics)
  float* dbase = dlow32 + (long)b * C * plane;
  for (int i = threadIdx.x; i < pelems; i += blockDim.x) {
    const float v = acc[i];
    if (v != 0.f) {
      const int c = i / (ph * pw);
      const int r = (i / pw) % ph, q = i % pw;
      unsafeAtomicAdd(&dbase[(long)c * plane + (long)(py0 + r) * Wi + (px0 + q)], v);
    }
  }
}

}  // namespace

std::vector<torch::Tensor> seg_ce_fwd(torch::Tensor logits, torch::Tensor labels,
                                      long Ho, long Wo, long ignore_index) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 4, "logits (B,C,h,w)");
  TORCH_CHECK(labels.scalar_type() == torch::kLong, "labels int64");
  auto lc = logits.contiguous();
  auto lb = labels.contiguous();
  const int B = lc.size(0), C = lc.size(1), Hi = lc.size(2), Wi = lc.size(3);
  auto f32 = lc.options().dtype(torch::kFloat);
  auto lse = torch::empty({B, Ho, Wo}, f32);
  auto loss = torch::zeros({}, f32);
  auto nvalid = torch::zeros({}, f32);
  auto stream = at::hip::getCurrentHIPStream().stream();
  const long total = (long)B * Ho * Wo;
#define CASE(T)                                                                \
  hipLaunchKernelGGL((seg_ce_fwd_kernel<T>), dim3((total + 255) / 256),        \
                     dim3(256), 0, stream, (const T*)lc.data_ptr(),            \
                     lb.data_ptr<long>(), lse.data_ptr<float>(),               \
                     loss.data_ptr<float>(), nvalid.data_ptr<float>(),         \
                     B, C, Hi, Wi, (int)Ho, (int)Wo, ignore_index)
  if (lc.scalar_type() == torch::kBFloat16) CASE(__hip_bfloat16);
  else if (lc.scalar_type() == torch::kHalf) CASE(__half);
  else if (lc.scalar_type() == torch::kFloat) CASE(float);
  else TORCH_CHECK(false, "seg_ce: unsupported dtype");
#undef CASE
  HIP_CHECK_KERNEL();
  return {loss, lse, nvalid};
}

torch::Tensor seg_ce_bwd(torch::Tensor logits, torch::Tensor labels,
                         torch::Tensor lse, torch::Tensor gout,
                         torch::Tensor nvalid, long ignore_index) {
  auto lc = logits.contiguous();
  auto lb = labels.contiguous();
  const int B = lc.size(0), C = lc.size(1), Hi = lc.size(2), Wi = lc.size(3);
  const int Ho = lse.size(1), Wo = lse.size(2);
  constexpr int OT = 16;
  // patch bound check: the 32x32 tile's taps must fit the dynamic-LDS cap
  const int ph_max = (int)((OT + 0.5f) * Hi / Ho + 2.f);
  const int pw_max = (int)((OT + 0.5f) * Wi / Wo + 2.f);
  const long smem = (long)C * ph_max * pw_max * 4;
  TORCH_CHECK(smem <= 160 * 1024,
              "seg_ce_bwd: LDS patch too large (C=", C, " ph=", ph_max,
              ") — dispatcher must fall back");
  auto dlow = torch::zeros_like(lc, lc.options().dtype(torch::kFloat));
  auto stream = at::hip::getCurrentHIPStream().stream();
  dim3 grid((Wo + OT - 1) / OT, (Ho + OT - 1) / OT, B);
#define CASE(T)                                                                \
  hipLaunchKernelGGL((seg_ce_bwd_kernel<T, OT>), grid, dim3(256), smem,        \
                     stream, (const T*)lc.data_ptr(), lb.data_ptr<long>(),     \
                     lse.data_ptr<float>(), gout.data_ptr<float>(),            \
                     nvalid.data_ptr<float>(), dlow.data_ptr<float>(),         \
                     B, C, Hi, Wi, Ho, Wo, ignore_index)
  if (lc.scalar_type() == torch::kBFloat16) CASE(__hip_bfloat16);
  else if (lc.scalar_type() == torch::kHalf) CASE(__half);
  else if (lc.scalar_type() == torch::kFloat) CASE(float);
  else TORCH_CHECK(false, "seg_ce: unsupported dtype");
#undef CASE
  HIP_CHECK_KERNEL();
  return dlow;
}
