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
// bwd: one block per 64x64 output tile, one thread per 4x4 pixel quad —
//      dLow accumulated in registers then a dynamic-LDS [CC][ph][pw] fp32
//      patch, then atomically added to the global fp32 dLow (patch borders
//      overlap between tiles).
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

// bwd: each thread owns a 4x4 output-pixel QUAD.  At upsample ratio >= 3
// a quad's 16 pixels' bilinear taps all fall in one 3x3 low-res cell
// window, so per channel the thread loads 9 cells once, accumulates all 16
// pixels' weighted gradients in a 3x3 REGISTER array (tap-cell choice done
// with selects — no runtime register indexing, guide rule 20), and issues
// only 9 ds_adds.  The first version did 4 LDS atomics per (pixel,
// channel): PMC measured 94% SQ_WAIT_INST_ANY — LDS atomics are per-lane
// serialized RMW, and those 1.2G of them were the whole 6.2 ms kernel.
// Block = 256 threads = 64x64 output tile; channels processed in chunks so
// the [CC][ph][pw] fp32 patch fits LDS.  dlow32 must be zero-initialized.
template <typename T, int OT /*64*/>
__global__ void seg_ce_bwd_kernel(const T* __restrict__ logits,
                                  const long* __restrict__ labels,
                                  const float* __restrict__ lse,
                                  const float* __restrict__ gout,
                                  const float* __restrict__ nvalid,
                                  float* __restrict__ dlow32,
                                  int B, int C, int Hi, int Wi, int Ho, int Wo,
                                  long ignore_index, int CC) {
  extern __shared__ float acc[];  // [CC][ph][pw]
  const int b = blockIdx.z;
  const int oy0 = blockIdx.y * OT, ox0 = blockIdx.x * OT;
  const Tap tc00 = tap_at(oy0, ox0, Hi, Wi, Ho, Wo);
  const Tap tc11 = tap_at(min(oy0 + OT - 1, Ho - 1), min(ox0 + OT - 1, Wo - 1),
                          Hi, Wi, Ho, Wo);
  const int py0 = tc00.y0, px0 = tc00.x0;
  const int ph = tc11.y1 - py0 + 1, pw = tc11.x1 - px0 + 1;
  const long plane = (long)Hi * Wi;
  const T* base = logits + (long)b * C * plane;
  const float gs = gout[0] / fmaxf(nvalid[0], 1.f);

  // quad base: 16x16 quads of 4x4 pixels
  const int qy0 = oy0 + (threadIdx.x >> 4) * 4;
  const int qx0 = ox0 + (threadIdx.x & 15) * 4;
  const Tap tq = tap_at(qy0, qx0, Hi, Wi, Ho, Wo);
  const int cby = tq.y0, cbx = tq.x0;  // 3x3 cell window base

  // per-pixel constants (arrays indexed by the UNROLLED px loop only).
  // The tap-cell indices, valid flag and label (C <= 256, host-guarded)
  // are PACKED into one int per pixel: a first cut kept 4 separate 16-int
  // index arrays and the extra ~64 VGPRs cut occupancy enough to run 20%
  // slower (3.1 ms vs 2.6 ms) than this packed form.  (A channel-pair
  // variant with 2x the loads in flight was also slower — 3.2 ms — for the
  // same register-pressure reason.)
  // bilinear weights PACKED as half2 pairs: they are [0,1] fractions, so
  // fp16 rounding (~5e-4) is far below the bf16 logits' own quantization —
  // and it frees 32 VGPRs on a kernel that is occupancy-limited (see the
  // seg_ce entries in ROADMAP.md negative results).
  __half2 wA[16], wB[16];
  float lse16[16];
  int pk16[16];
#pragma unroll
  for (int px = 0; px < 16; ++px) {
    const int y = qy0 + (px >> 2), x = qx0 + (px & 3);
    const long pidx = ((long)b * Ho + y) * Wo + x;
    const bool inb = y < Ho && x < Wo;
    const long lab = inb ? labels[pidx] : ignore_index;
    const bool ok = inb && lab != ignore_index && lab >= 0 && lab < C;
    lse16[px] = ok ? lse[pidx] : 0.f;
    const Tap t = tap_at(inb ? y : 0, inb ? x : 0, Hi, Wi, Ho, Wo);
    wA[px] = __floats2half2_rn(t.w00, t.w01);
    wB[px] = __floats2half2_rn(t.w10, t.w11);
    // window-relative tap cells; y1/x1 may EQUAL y0/x0 at the image edge
    // (clamped taps) — the packed slots below then route both weights to
    // one cell
    const int ry0 = min(max(t.y0 - cby, 0), 2);
    const int rx0 = min(max(t.x0 - cbx, 0), 2);
    const int ry1 = min(max(t.y1 - cby, 0), 2);
    const int rx1 = min(max(t.x1 - cbx, 0), 2);
    const bool interior = (ry1 == ry0 + 1) && (rx1 == rx0 + 1);
    pk16[px] = (ry0 * 3 + rx0) | ((ry0 * 3 + rx1) << 4) |
               ((ry1 * 3 + rx0) << 8) | ((ry1 * 3 + rx1) << 12) |
               (min(ry0, 1) << 16) | (min(rx0, 1) << 17) |
               ((int)ok << 18) | ((int)interior << 19) |
               (ok ? ((int)lab & 255) << 20 : 0);
  }
  const int pb = (cby - py0) * pw + (cbx - px0);  // window base in the patch

  for (int c0 = 0; c0 < C; c0 += CC) {
    const int cn = min(CC, C - c0);
    for (int i = threadIdx.x; i < cn * ph * pw; i += blockDim.x) acc[i] = 0.f;
    __syncthreads();

    for (int c = c0; c < c0 + cn; ++c) {
      const T* p = base + (long)c * plane;
      float c9[9];
#pragma unroll
      for (int dy = 0; dy < 3; ++dy)
#pragma unroll
        for (int dx = 0; dx < 3; ++dx)
          c9[dy * 3 + dx] = DT<T>::to_f32(
              p[(long)min(cby + dy, Hi - 1) * Wi + min(cbx + dx, Wi - 1)]);
      float r9[9];
#pragma unroll
      for (int k = 0; k < 9; ++k) r9[k] = 0.f;
#pragma unroll
      for (int px = 0; px < 16; ++px) {
        const int pk = pk16[px];
        if (!((pk >> 18) & 1)) continue;
        // 2x2 tap VALUES via cheap nested ternaries on the top-left cell
        // (no runtime register indexing, guide rule 20).  This is exact even
        // for edge-clamped taps: the 3x3 window is loaded with clamped input
        // indices, so the neighbouring cell replicates the edge value.
        const int by = (pk >> 16) & 1, bx = (pk >> 17) & 1;
        const float v00 = by ? (bx ? c9[4] : c9[3]) : (bx ? c9[1] : c9[0]);
        const float v01 = by ? (bx ? c9[5] : c9[4]) : (bx ? c9[2] : c9[1]);
        const float v10 = by ? (bx ? c9[7] : c9[6]) : (bx ? c9[4] : c9[3]);
        const float v11 = by ? (bx ? c9[8] : c9[7]) : (bx ? c9[5] : c9[4]);
        const float2 wa = __half22float2(wA[px]);
        const float2 wb = __half22float2(wB[px]);
        const float z = wa.x * v00 + wa.y * v01 + wb.x * v10 + wb.y * v11;
        const float g = (__expf(z - lse16[px]) -
                         (c == ((pk >> 20) & 255) ? 1.f : 0.f)) * gs;
        const float g00 = wa.x * g, g01 = wa.y * g;
        const float g10 = wb.x * g, g11 = wb.y * g;
        if ((pk >> 19) & 1) {
          // interior (no edge-clamped taps, the common case): by/bx have a
          // PROVABLE [0,1] range from the &1, so comparisons against the
          // compile-time (dy,dx) fold most of the 36 mask terms away
#pragma unroll
          for (int k = 0; k < 9; ++k) {
            const int dy = k / 3, dx = k % 3;
            r9[k] += ((by == dy && bx == dx) ? g00 : 0.f) +
                     ((by == dy && bx + 1 == dx) ? g01 : 0.f) +
                     ((by + 1 == dy && bx == dx) ? g10 : 0.f) +
                     ((by + 1 == dy && bx + 1 == dx) ? g11 : 0.f);
          }
        } else {
          // image border: edge-clamped taps collapse pairs onto one cell —
          // scatter by the ACTUAL slots (only the right/bottom edge waves
          // ever take this side)
          const int s00 = pk & 15, s01 = (pk >> 4) & 15;
          const int s10 = (pk >> 8) & 15, s11 = (pk >> 12) & 15;
#pragma unroll
          for (int k = 0; k < 9; ++k)
            r9[k] += ((k == s00) ? g00 : 0.f) + ((k == s01) ? g01 : 0.f) +
                     ((k == s10) ? g10 : 0.f) + ((k == s11) ? g11 : 0.f);
        }
      }
      float* ac = acc + (long)(c - c0) * ph * pw + pb;
#pragma unroll
      for (int dy = 0; dy < 3; ++dy)
#pragma unroll
        for (int dx = 0; dx < 3; ++dx) {
          const float v = r9[dy * 3 + dx];
          // clamped taps collapse onto interior cells; window rows/cols
          // beyond the patch carry zeros — skip them
          if (v != 0.f && cby + dy - py0 < ph && cbx + dx - px0 < pw)
            unsafeAtomicAdd(&ac[dy * pw + dx], v);
        }
    }
    __syncthreads();
    // chunk patch -> global fp32 dlow (tile borders overlap: atomics)
    float* dbase = dlow32 + ((long)b * C + c0) * plane;
    for (int i = threadIdx.x; i < cn * ph * pw; i += blockDim.x) {
      const float v = acc[i];
      if (v != 0.f) {
        const int c = i / (ph * pw);
        const int r = (i / pw) % ph, q = i % pw;
        unsafeAtomicAdd(&dbase[(long)c * plane + (long)(py0 + r) * Wi +
                               (px0 + q)], v);
      }
    }
    __syncthreads();
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
  constexpr int OT = 64;
  const int ph_max = (int)((OT + 0.5f) * Hi / Ho + 3.f);
  const int pw_max = (int)((OT + 0.5f) * Wi / Wo + 3.f);
  // channel chunk so the [CC][ph][pw] fp32 patch fits comfortably in LDS
  const int CC = std::max(1, std::min(C, (int)(48 * 1024 /
                                              (ph_max * pw_max * 4))));
  const long smem = (long)CC * ph_max * pw_max * 4;
  TORCH_CHECK(smem <= 160 * 1024, "seg_ce_bwd: LDS patch too large");
  auto dlow = torch::zeros_like(lc, lc.options().dtype(torch::kFloat));
  auto stream = at::hip::getCurrentHIPStream().stream();
  dim3 grid((Wo + OT - 1) / OT, (Ho + OT - 1) / OT, B);
#define CASE(T)                                                                \
  hipLaunchKernelGGL((seg_ce_bwd_kernel<T, OT>), grid, dim3(256), smem,        \
                     stream, (const T*)lc.data_ptr(), lb.data_ptr<long>(),     \
                     lse.data_ptr<float>(), gout.data_ptr<float>(),            \
                     nvalid.data_ptr<float>(), dlow.data_ptr<float>(),         \
                     B, C, Hi, Wi, Ho, Wo, ignore_index, CC)
  if (lc.scalar_type() == torch::kBFloat16) CASE(__hip_bfloat16);
  else if (lc.scalar_type() == torch::kHalf) CASE(__half);
  else if (lc.scalar_type() == torch::kFloat) CASE(float);
  else TORCH_CHECK(false, "seg_ce: unsupported dtype");
#undef CASE
  HIP_CHECK_KERNEL();
  return dlow;
}
