// Bilinear upsample (align_corners=False) forward + GATHER backward for the
// SegFormer decode head (reference Scaling_model_training
// .ipynb:cell-52 — each stage feature is upsampled to the 1/4-resolution
// grid before the fuse conv).  torch's NHWC upsample_bilinear2d_backward
// scatters with atomics and cost 515 us per stage at the reference config;
// the backward here is a gather: each INPUT pixel sums the output-gradient
// window that taps it, with the per-(y,x) weights computed ONCE and reused
// across a channel block (weights do not depend on c).
//
// Only integer scale factors (Ho = s*Hi, Wo = s*Wi) are dispatched here —
// that is all the decode head produces (s = 2/4/8); anything else falls
// back to torch in the Python wrapper.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../common.h"

namespace {

// torch area_pixel_compute_source_index (align_corners=False), same math as
// seg_ce.hip::tap_at.
struct UTap {
  int p0, p1;
  float w0, w1;
};

DEVINL UTap utap(int o, int Pi, int Po) {
  const float s = fmaxf(((o + 0.5f) * Pi) / Po - 0.5f, 0.f);
  UTap t;
  t.p0 = min(Pi - 1, (int)s);
  t.p1 = min(Pi - 1, t.p0 + 1);
  const float f = s - t.p0;
  t.w0 = 1.f - f;
  t.w1 = f;
  return t;
}

// fwd: one thread per output element; fp32 tap accumulate, cast on store.
template <typename T>
__global__ void upsample_fwd_kernel(const T* __restrict__ in, T* __restrict__ out,
                                    long BC, int Hi, int Wi, int Ho, int Wo) {
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long total = BC * Ho * Wo;
  if (idx >= total) return;
  const int x = (int)(idx % Wo);
  const int y = (int)((idx / Wo) % Ho);
  const long bc = idx / ((long)Ho * Wo);
  const UTap ty = utap(y, Hi, Ho), tx = utap(x, Wi, Wo);
  const T* p = in + bc * Hi * Wi;
  const float v =
      ty.w0 * (tx.w0 * DT<T>::to_f32(p[(long)ty.p0 * Wi + tx.p0]) +
               tx.w1 * DT<T>::to_f32(p[(long)ty.p0 * Wi + tx.p1])) +
      ty.w1 * (tx.w0 * DT<T>::to_f32(p[(long)ty.p1 * Wi + tx.p0]) +
               tx.w1 * DT<T>::to_f32(p[(long)ty.p1 * Wi + tx.p1]));
  out[idx] = DT<T>::from_f32(v);
}

// bwd: one thread per (b, c-block, yi, xi) input position; scans the
// conservative [s*yi - s, s*yi + 2s) x same-for-x output window, skipping
// rows/cols whose weight for this input cell is zero.  CB channels share
// each weight evaluation.
template <typename T, int CB>
__global__ void upsample_bwd_kernel(const T* __restrict__ dout, T* __restrict__ din,
                                    int B, int C, int Hi, int Wi, int Ho, int Wo) {
  const int cb = (C + CB - 1) / CB;
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long total = (long)B * cb * Hi * Wi;
  if (idx >= total) return;
  const int xi = (int)(idx % Wi);
  const int yi = (int)((idx / Wi) % Hi);
  const int c0 = (int)((idx / ((long)Hi * Wi)) % cb) * CB;
  const int b = (int)(idx / ((long)cb * Hi * Wi));
  const int sy = Ho / Hi, sx = Wo / Wi;

  float acc[CB];
#pragma unroll
  for (int j = 0; j < CB; ++j) acc[j] = 0.f;

  const int ylo = max(0, sy * yi - sy), yhi = min(Ho, sy * yi + 2 * sy);
  const int xlo = max(0, sx * xi - sx), xhi = min(Wo, sx * xi + 2 * sx);
  const long plane = (long)Ho * Wo;
  const T* base = dout + ((long)b * C + c0) * plane;
  for (int y = ylo; y < yhi; ++y) {
    const UTap ty = utap(y, Hi, Ho);
    const float wy = (ty.p0 == yi ? ty.w0 : 0.f) + (ty.p1 == yi ? ty.w1 : 0.f);
    if (wy == 0.f) continue;
    const long rowoff = (long)y * Wo;
    for (int x = xlo; x < xhi; ++x) {
      const UTap tx = utap(x, Wi, Wo);
      const float wx = (tx.p0 == xi ? tx.w0 : 0.f) + (tx.p1 == xi ? tx.w1 : 0.f);
      if (wx == 0.f) continue;
      const float w = wy * wx;
      const long o = rowoff + x;
#pragma unroll
      for (int j = 0; j < CB; ++j)
        if (c0 + j < C) acc[j] += w * DT<T>::to_f32(base[(long)j * plane + o]);
    }
  }
  T* dst = din + ((long)b * C + c0) * (long)Hi * Wi + (long)yi * Wi + xi;
#pragma unroll
  for (int j = 0; j < CB; ++j)
    if (c0 + j < C) dst[(long)j * Hi * Wi] = DT<T>::from_f32(acc[j]);
}

}  // namespace

torch::Tensor upsample_bilinear_fwd(torch::Tensor x, long Ho, long Wo) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4, "upsample: (B,C,H,W) cuda");
  auto xc = x.contiguous();
  const int B = xc.size(0), C = xc.size(1), Hi = xc.size(2), Wi = xc.size(3);
  auto out = torch::empty({B, C, Ho, Wo}, xc.options());
  const long total = (long)B * C * Ho * Wo;
  auto stream = at::hip::getCurrentHIPStream().stream();
#define UF(T)                                                                 \
  hipLaunchKernelGGL((upsample_fwd_kernel<T>), dim3((total + 255) / 256),     \
                     dim3(256), 0, stream, (const T*)xc.data_ptr(),           \
                     (T*)out.data_ptr(), (long)B * C, Hi, Wi, (int)Ho, (int)Wo)
  if (x.scalar_type() == torch::kBFloat16) UF(__hip_bfloat16);
  else if (x.scalar_type() == torch::kHalf) UF(__half);
  else if (x.scalar_type() == torch::kFloat) UF(float);
  else TORCH_CHECK(false, "upsample: unsupported dtype");
#undef UF
  HIP_CHECK_KERNEL();
  return out;
}

torch::Tensor upsample_bilinear_bwd(torch::Tensor dout, long Hi, long Wi) {
  TORCH_CHECK(dout.is_cuda() && dout.dim() == 4, "upsample bwd: (B,C,Ho,Wo)");
  auto dc = dout.contiguous();
  const int B = dc.size(0), C = dc.size(1), Ho = dc.size(2), Wo = dc.size(3);
  TORCH_CHECK(Ho % Hi == 0 && Wo % Wi == 0,
              "upsample bwd: integer scale only (wrapper guards this)");
  auto din = torch::empty({B, C, Hi, Wi}, dc.options());
  constexpr int CB = 4;
  const long total = (long)B * ((C + CB - 1) / CB) * Hi * Wi;
  auto stream = at::hip::getCurrentHIPStream().stream();
#define UB(T)                                                                 \
  hipLaunchKernelGGL((upsample_bwd_kernel<T, CB>), dim3((total + 255) / 256), \
                     dim3(256), 0, stream, (const T*)dc.data_ptr(),           \
                     (T*)din.data_ptr(), B, C, (int)Hi, (int)Wi, Ho, Wo)
  if (dout.scalar_type() == torch::kBFloat16) UB(__hip_bfloat16);
  else if (dout.scalar_type() == torch::kHalf) UB(__half);
  else if (dout.scalar_type() == torch::kFloat) UB(float);
  else TORCH_CHECK(false, "upsample bwd: unsupported dtype");
#undef UB
  HIP_CHECK_KERNEL();
  return din;
}
