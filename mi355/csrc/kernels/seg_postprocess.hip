// Fused semantic-segmentation postprocess: bilinear
// upsample of class logits to the original image size + channel argmax in
// one pass — replaces interpolate(B,150,H,W)+argmax (reference
// post_process_semantic_segmentation, Scaling_batch_inference.ipynb:cell-42).
// Memory-bound: reads each low-res logit plane ~once (L2-resident taps),
// writes one int32 per output pixel instead of a 150-channel upsampled map.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../common.h"

namespace {

template <typename T>
__global__ void seg_post_kernel(const T* __restrict__ logits, int* __restrict__ out,
                                int B, int C, int Hi, int Wi, int Ho, int Wo) {
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long total = (long)B * Ho * Wo;
  if (idx >= total) return;
  const int x = idx % Wo;
  const int y = (idx / Wo) % Ho;
  const int b = idx / ((long)Ho * Wo);

  // align_corners=False mapping, negative coords clamped to 0 exactly like
  // torch's area_pixel_compute_source_index
  const float sy = fmaxf(((y + 0.5f) * Hi) / Ho - 0.5f, 0.f);
  const float sx = fmaxf(((x + 0.5f) * Wi) / Wo - 0.5f, 0.f);
  const int y0 = min(Hi - 1, (int)sy);
  const int x0 = min(Wi - 1, (int)sx);
  const int y1 = min(Hi - 1, y0 + 1);
  const int x1 = min(Wi - 1, x0 + 1);
  const float fy = sy - y0;
  const float fx = sx - x0;
  const float w00 = (1 - fy) * (1 - fx), w01 = (1 - fy) * fx;
  const float w10 = fy * (1 - fx), w11 = fy * fx;

  const long plane = (long)Hi * Wi;
  const T* base = logits + (long)b * C * plane;
  float best = -INFINITY;
  int best_c = 0;
  for (int c = 0; c < C; ++c) {
    const T* p = base + (long)c * plane;
    const float v = w00 * DT<T>::to_f32(p[(long)y0 * Wi + x0]) +
                    w01 * DT<T>::to_f32(p[(long)y0 * Wi + x1]) +
                    w10 * DT<T>::to_f32(p[(long)y1 * Wi + x0]) +
                    w11 * DT<T>::to_f32(p[(long)y1 * Wi + x1]);
    if (v > best) {
      best = v;
      best_c = c;
    }
  }
  out[idx] = best_c;
}

}  // namespace

torch::Tensor seg_postprocess(torch::Tensor logits, long Ho, long Wo) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 4, "logits must be (B,C,H,W)");
  auto lc = logits.contiguous();
  const int B = lc.size(0), C = lc.size(1), Hi = lc.size(2), Wi = lc.size(3);
  auto out = torch::empty({B, Ho, Wo}, lc.options().dtype(torch::kInt));
  auto stream = at::hip::getCurrentHIPStream().stream();
  const long total = (long)B * Ho * Wo;
  const long blocks = (total + 255) / 256;
#define CASE(T)                                                                    \
  hipLaunchKernelGGL((seg_post_kernel<T>), dim3(blocks), dim3(256), 0, stream,     \
                     (const T*)lc.data_ptr(), out.data_ptr<int>(), B, C, Hi, Wi,   \
                     (int)Ho, (int)Wo)
  if (lc.scalar_type() == torch::kFloat) CASE(float);
  else if (lc.scalar_type() == torch::kBFloat16) CASE(__hip_bfloat16);
  else if (lc.scalar_type() == torch::kHalf) CASE(__half);
  else TORCH_CHECK(false, "seg_postprocess: unsupported dtype");
#undef CASE
  HIP_CHECK_KERNEL();
  return out;
}
