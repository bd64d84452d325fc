// Depthwise 3x3 stride-1 pad-1 conv for CDNA4 — SegFormer Mix-FFN (SURVEY
// K15; reference Scaling_model_training.ipynb:cell-52).  MIOpen routes these
// grouped (groups==C) bf16 shapes through generic grouped-conv / naive
// kernels; a depthwise 3x3 is a 9-tap stencil that should run at HBM speed.
//
// fwd : out[b,c,y,x] = sum_{dy,dx} in[b,c,y+dy-1,x+dx-1] * w[c,dy,dx] + bias[c]
// dIn : same stencil with the weight taps flipped (template FLIP)
// dW  : per-channel 9-tap correlation + dbias, one block per channel chunk
//
// Each thread produces 8 contiguous x outputs (vectorized b128 loads/stores,
// scalar edge taps); fp32 accumulate.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../common.h"

namespace {

template <typename T, bool FLIP, bool HAS_BIAS>
__global__ void dwconv3x3_kernel(const T* __restrict__ in,
                                 const T* __restrict__ w,
                                 const T* __restrict__ bias,
                                 T* __restrict__ out,
                                 int C, long BC, int H, int W) {
  const int xchunks = W >> 3;
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= BC * H * xchunks) return;
  const int x0 = (int)(idx % xchunks) << 3;
  const int y = (int)((idx / xchunks) % H);
  const long bc = idx / ((long)xchunks * H);
  const int c = (int)(bc % C);

  float wt[3][3];
#pragma unroll
  for (int dy = 0; dy < 3; ++dy)
#pragma unroll
    for (int dx = 0; dx < 3; ++dx)
      wt[dy][dx] = DT<T>::to_f32(
          FLIP ? w[c * 9 + (2 - dy) * 3 + (2 - dx)] : w[c * 9 + dy * 3 + dx]);

  const T* plane = in + bc * H * W;
  float acc[8];
  const float b0 = HAS_BIAS ? DT<T>::to_f32(bias[c]) : 0.f;
#pragma unroll
  for (int i = 0; i < 8; ++i) acc[i] = b0;

#pragma unroll
  for (int dy = 0; dy < 3; ++dy) {
    const int yy = y + dy - 1;
    if (yy < 0 || yy >= H) continue;
    const T* row = plane + (long)yy * W;
    float v[10];
    VecT<T, 8> mid = vload<T, 8>(row + x0);
#pragma unroll
    for (int i = 0; i < 8; ++i) v[i + 1] = DT<T>::to_f32(mid.data[i]);
    v[0] = (x0 > 0) ? DT<T>::to_f32(row[x0 - 1]) : 0.f;
    v[9] = (x0 + 8 < W) ? DT<T>::to_f32(row[x0 + 8]) : 0.f;
#pragma unroll
    for (int dx = 0; dx < 3; ++dx) {
      const float wc = wt[dy][dx];
#pragma unroll
      for (int i = 0; i < 8; ++i) acc[i] += v[i + dx] * wc;
    }
  }
  VecT<T, 8> o;
#pragma unroll
  for (int i = 0; i < 8; ++i) o.data[i] = DT<T>::from_f32(acc[i]);
  vstore<T, 8>(out + bc * H * W + (long)y * W + x0, o);
}

// dW + dbias: one block per (channel, batch-slab); fp32 atomics into the
// (C, 10) staging buffer, host casts/splits.  Each thread owns 8-wide x
// CHUNKS of the channel's planes (vectorized b128 loads like the fwd
// kernel) — the first per-element version did ~10 scalar loads + 2 integer
// divides per output element and ran 66x off the bandwidth bound.
template <typename T>
__global__ void dwconv3x3_wrw_kernel(const T* __restrict__ dout,
                                     const T* __restrict__ in,
                                     float* __restrict__ dw10,  // (C, 10), zeroed
                                     int B, int C, int H, int W, int slabs) {
  const int c = blockIdx.x;
  const int xch = W >> 3;
  const long HW = (long)H * W;
  const long M = (long)B * H * xch;  // chunk count
  const long m0 = blockIdx.y * M / slabs;
  const long m1 = (blockIdx.y + 1) * M / slabs;
  float acc[9] = {0, 0, 0, 0, 0, 0, 0, 0, 0};
  float db = 0.f;
  for (long m = m0 + threadIdx.x; m < m1; m += blockDim.x) {
    const int b = (int)(m / ((long)H * xch));
    const long i = m % ((long)H * xch);
    const int y = (int)(i / xch), x0 = ((int)(i % xch)) << 3;
    const T* plane = in + ((long)b * C + c) * HW;
    float g[8];
    VecT<T, 8> gv = vload<T, 8>(dout + ((long)b * C + c) * HW +
                                (long)y * W + x0);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      g[j] = DT<T>::to_f32(gv.data[j]);
      db += g[j];
    }
#pragma unroll
    for (int dy = 0; dy < 3; ++dy) {
      const int yy = y + dy - 1;
      if (yy < 0 || yy >= H) continue;
      const T* row = plane + (long)yy * W;
      float v[10];
      VecT<T, 8> mid = vload<T, 8>(row + x0);
#pragma unroll
      for (int j = 0; j < 8; ++j) v[j + 1] = DT<T>::to_f32(mid.data[j]);
      v[0] = (x0 > 0) ? DT<T>::to_f32(row[x0 - 1]) : 0.f;
      v[9] = (x0 + 8 < W) ? DT<T>::to_f32(row[x0 + 8]) : 0.f;
#pragma unroll
      for (int dx = 0; dx < 3; ++dx) {
        float s = 0.f;
#pragma unroll
        for (int j = 0; j < 8; ++j) s += g[j] * v[j + dx];
        acc[dy * 3 + dx] += s;
      }
    }
  }
  __shared__ float smem[16];
#pragma unroll
  for (int t = 0; t < 9; ++t) {
    const float r = block_reduce_sum(acc[t], smem);
    if (threadIdx.x == 0) atomicAdd(&dw10[c * 10 + t], r);
  }
  const float r = block_reduce_sum(db, smem);
  if (threadIdx.x == 0) atomicAdd(&dw10[c * 10 + 9], r);
}

}  // namespace

torch::Tensor dwconv3x3_fwd(torch::Tensor x, torch::Tensor w,
                            c10::optional<torch::Tensor> bias, bool flip) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.stride(3) == 1);
  const int B = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  TORCH_CHECK(W % 8 == 0, "dwconv3x3: W%8==0");
  TORCH_CHECK(w.numel() == (long)C * 9);
  auto xc = x.contiguous();
  auto wc = w.contiguous();
  auto out = torch::empty_like(xc);
  const long BC = (long)B * C;
  const long total = BC * H * (W / 8);
  auto stream = at::hip::getCurrentHIPStream().stream();
  const bool hb = bias.has_value() && bias->defined();
  torch::Tensor bc_;
  if (hb) bc_ = bias->contiguous();

#define DW_LAUNCH(T, F, HB)                                                   \
  hipLaunchKernelGGL((dwconv3x3_kernel<T, F, HB>),                            \
                     dim3((total + 255) / 256), dim3(256), 0, stream,         \
                     (const T*)xc.data_ptr(), (const T*)wc.data_ptr(),        \
                     hb ? (const T*)bc_.data_ptr() : nullptr,                 \
                     (T*)out.data_ptr(), C, BC, H, W)
  AT_DISPATCH_SWITCH(x.scalar_type(), "dwconv3x3", AT_DISPATCH_CASE(
      torch::kBFloat16, [&] {
        if (flip) { if (hb) DW_LAUNCH(__hip_bfloat16, true, true); else DW_LAUNCH(__hip_bfloat16, true, false); }
        else { if (hb) DW_LAUNCH(__hip_bfloat16, false, true); else DW_LAUNCH(__hip_bfloat16, false, false); }
      }) AT_DISPATCH_CASE(torch::kHalf, [&] {
        if (flip) { if (hb) DW_LAUNCH(__half, true, true); else DW_LAUNCH(__half, true, false); }
        else { if (hb) DW_LAUNCH(__half, false, true); else DW_LAUNCH(__half, false, false); }
      }) AT_DISPATCH_CASE(torch::kFloat, [&] {
        if (flip) { if (hb) DW_LAUNCH(float, true, true); else DW_LAUNCH(float, true, false); }
        else { if (hb) DW_LAUNCH(float, false, true); else DW_LAUNCH(float, false, false); }
      }));
#undef DW_LAUNCH
  HIP_CHECK_KERNEL();
  return out;
}

std::vector<torch::Tensor> dwconv3x3_wrw(torch::Tensor dout, torch::Tensor x) {
  const int B = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  TORCH_CHECK(W % 8 == 0, "dwconv3x3_wrw: W%8==0 (the dispatcher guards this)");
  auto dc = dout.contiguous();
  auto xc = x.contiguous();
  auto dw10 = torch::zeros({C, 10}, x.options().dtype(torch::kFloat));
  // fill ~2 blocks/CU regardless of channel count
  const int slabs = std::max(1, (int)(512 / C) + 1);
  dim3 grid(C, slabs);
  auto stream = at::hip::getCurrentHIPStream().stream();
  AT_DISPATCH_SWITCH(x.scalar_type(), "dwconv3x3_wrw", AT_DISPATCH_CASE(
      torch::kBFloat16, [&] {
        hipLaunchKernelGGL((dwconv3x3_wrw_kernel<__hip_bfloat16>), grid,
                           dim3(256), 0, stream,
                           (const __hip_bfloat16*)dc.data_ptr(),
                           (const __hip_bfloat16*)xc.data_ptr(),
                           dw10.data_ptr<float>(), B, C, H, W, slabs);
      }) AT_DISPATCH_CASE(torch::kHalf, [&] {
        hipLaunchKernelGGL((dwconv3x3_wrw_kernel<__half>), grid,
                           dim3(256), 0, stream, (const __half*)dc.data_ptr(),
                           (const __half*)xc.data_ptr(),
                           dw10.data_ptr<float>(), B, C, H, W, slabs);
      }) AT_DISPATCH_CASE(torch::kFloat, [&] {
        hipLaunchKernelGGL((dwconv3x3_wrw_kernel<float>), grid,
                           dim3(256), 0, stream, (const float*)dc.data_ptr(),
                           (const float*)xc.data_ptr(),
                           dw10.data_ptr<float>(), B, C, H, W, slabs);
      }));
  HIP_CHECK_KERNEL();
  auto dw = dw10.narrow(1, 0, 9).contiguous();
  auto db = dw10.select(1, 9).contiguous();
  return {dw, db};
}
