// SegFormer stage-1 overlapping patch embed: conv2d 7x7 stride 4 pad 3,
// Cin=3 -> Cout=32 on 512x512 (Scaling_model_training.ipynb:
// cell-52).  MIOpen has no CK solver for this bf16 Cin=3 shape and falls
// back to naive_conv_* (12 ms fwd / 36 ms wrw measured) — a hand-written
// tiled kernel runs it at memory speed.
//
// fwd: 16x16 output tile per block; the 70x70x3 input window staged in LDS,
// each thread computes all 32 Cout for one output pixel (VALU f32 FMA — at
// Cin=3/K=147 an MFMA formulation would mostly multiply zeros).
// wrw: one block per (ky,kx) tap x M-slab; thread strides output pixels
// accumulating dW[c][cout] for its tap; block-reduce + atomicAdd into the
// fp32 dW.  dX is never needed (stage 1 consumes the input image).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../common.h"

namespace {

constexpr int KS = 7, STR = 4, PAD = 3, CIN = 3, COUT = 32;
constexpr int OT = 16;                       // 16x16 output tile
constexpr int WIN = (OT - 1) * STR + KS;     // 67 -> use 70 rounded
constexpr int WROW = WIN + 3;                // LDS row stride (pad to 70)

template <typename T>
__global__ __launch_bounds__(256) void patch_embed_s1_fwd_kernel(
    const T* __restrict__ in, const T* __restrict__ w,
    const T* __restrict__ bias, T* __restrict__ out,
    int B, int H, int W, int OH, int OW) {
  __shared__ T win[CIN][WIN][WROW];
  __shared__ T wlds[COUT * CIN * KS * KS];

  const int tile_x = blockIdx.x % ((OW + OT - 1) / OT);
  const int tile_y = (blockIdx.x / ((OW + OT - 1) / OT)) % ((OH + OT - 1) / OT);
  const int b = blockIdx.x / (((OW + OT - 1) / OT) * ((OH + OT - 1) / OT));
  const int oy0 = tile_y * OT, ox0 = tile_x * OT;
  const int iy0 = oy0 * STR - PAD, ix0 = ox0 * STR - PAD;

  // stage weights (32*3*49 = 4704 elems) and the input window
  for (int i = threadIdx.x; i < COUT * CIN * KS * KS; i += blockDim.x)
    wlds[i] = w[i];
  for (int i = threadIdx.x; i < CIN * WIN * WIN; i += blockDim.x) {
    const int c = i / (WIN * WIN);
    const int r = (i / WIN) % WIN;
    const int x = i % WIN;
    const int iy = iy0 + r, ix = ix0 + x;
    T v = T(0.f);
    if (iy >= 0 && iy < H && ix >= 0 && ix < W)
      v = in[((long)b * CIN + c) * H * W + (long)iy * W + ix];
    win[c][r][x] = v;
  }
  __syncthreads();

  const int ty = threadIdx.x >> 4, tx = threadIdx.x & 15;
  const int oy = oy0 + ty, ox = ox0 + tx;
  if (oy >= OH || ox >= OW) return;

  float acc[COUT];
#pragma unroll
  for (int co = 0; co < COUT; ++co)
    acc[co] = bias ? DT<T>::to_f32(bias[co]) : 0.f;

  for (int c = 0; c < CIN; ++c)
#pragma unroll
    for (int ky = 0; ky < KS; ++ky)
#pragma unroll
      for (int kx = 0; kx < KS; ++kx) {
        const float v = DT<T>::to_f32(win[c][ty * STR + ky][tx * STR + kx]);
        const int tap = (c * KS + ky) * KS + kx;
#pragma unroll
        for (int co = 0; co < COUT; ++co)
          acc[co] += v * DT<T>::to_f32(wlds[co * CIN * KS * KS + tap]);
      }

  const long obase = ((long)b * COUT) * OH * OW + (long)oy * OW + ox;
#pragma unroll
  for (int co = 0; co < COUT; ++co)
    out[obase + (long)co * OH * OW] = DT<T>::from_f32(acc[co]);
}

// dW: grid (KS*KS taps, SLABS); each block owns one (ky,kx) for all
// (c, cout); threads stride the slab's output pixels; per-thread
// acc[CIN][COUT] fp32, block-reduced, atomicAdd into dw32 (Cout,Cin,7,7).
template <typename T>
__global__ __launch_bounds__(256) void patch_embed_s1_wrw_kernel(
    const T* __restrict__ dout, const T* __restrict__ in,
    float* __restrict__ dw32, float* __restrict__ dbias32,
    int B, int H, int W, int OH, int OW, int slabs) {
  const int tap = blockIdx.x;
  const int ky = tap / KS, kx = tap % KS;
  const long M = (long)B * OH * OW;
  const long m0 = blockIdx.y * M / slabs;
  const long m1 = (blockIdx.y + 1) * M / slabs;

  float acc[CIN][COUT];
#pragma unroll
  for (int c = 0; c < CIN; ++c)
#pragma unroll
    for (int co = 0; co < COUT; ++co) acc[c][co] = 0.f;
  float dbias[COUT];
#pragma unroll
  for (int co = 0; co < COUT; ++co) dbias[co] = 0.f;

  for (long m = m0 + threadIdx.x; m < m1; m += blockDim.x) {
    const int ox = (int)(m % OW);
    const int oy = (int)((m / OW) % OH);
    const int b = (int)(m / ((long)OH * OW));
    const int iy = oy * STR - PAD + ky, ix = ox * STR - PAD + kx;
    float g[COUT];
    const long dbase = ((long)b * COUT) * OH * OW + (long)oy * OW + ox;
#pragma unroll
    for (int co = 0; co < COUT; ++co)
      g[co] = DT<T>::to_f32(dout[dbase + (long)co * OH * OW]);
    if (tap == 0) {
#pragma unroll
      for (int co = 0; co < COUT; ++co) dbias[co] += g[co];
    }
    if (iy < 0 || iy >= H || ix < 0 || ix >= W) continue;
#pragma unroll
    for (int c = 0; c < CIN; ++c) {
      const float v = DT<T>::to_f32(in[((long)b * CIN + c) * H * W +
                                       (long)iy * W + ix]);
#pragma unroll
      for (int co = 0; co < COUT; ++co) acc[c][co] += v * g[co];
    }
  }

  __shared__ float smem[16];
#pragma unroll
  for (int c = 0; c < CIN; ++c)
#pragma unroll
    for (int co = 0; co < COUT; ++co) {
      const float r = block_reduce_sum(acc[c][co], smem);
      if (threadIdx.x == 0)
        atomicAdd(&dw32[((co * CIN + c) * KS + ky) * KS + kx], r);
    }
  if (tap == 0) {
#pragma unroll
    for (int co = 0; co < COUT; ++co) {
      const float r = block_reduce_sum(dbias[co], smem);
      if (threadIdx.x == 0) atomicAdd(&dbias32[co], r);
    }
  }
}

}  // namespace

torch::Tensor patch_embed_s1_fwd(torch::Tensor x, torch::Tensor w,
                                 c10::optional<torch::Tensor> bias) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.size(1) == 3);
  TORCH_CHECK(w.size(0) == 32 && w.size(1) == 3 && w.size(2) == 7 && w.size(3) == 7);
  const int B = x.size(0), H = x.size(2), W = x.size(3);
  const int OH = (H + 2 * 3 - 7) / 4 + 1, OW = (W + 2 * 3 - 7) / 4 + 1;
  auto xc = x.contiguous();
  auto wc = w.contiguous();
  auto out = torch::empty({B, 32, OH, OW}, x.options());
  const long blocks = (long)B * ((OH + 15) / 16) * ((OW + 15) / 16);
  auto stream = at::hip::getCurrentHIPStream().stream();
  const bool hb = bias.has_value() && bias->defined();
  torch::Tensor bc;
  if (hb) bc = bias->contiguous();
  AT_DISPATCH_SWITCH(x.scalar_type(), "pe1f", AT_DISPATCH_CASE(
      torch::kBFloat16, [&] {
        hipLaunchKernelGGL((patch_embed_s1_fwd_kernel<__hip_bfloat16>),
                           dim3(blocks), dim3(256), 0, stream,
                           (const __hip_bfloat16*)xc.data_ptr(),
                           (const __hip_bfloat16*)wc.data_ptr(),
                           hb ? (const __hip_bfloat16*)bc.data_ptr() : nullptr,
                           (__hip_bfloat16*)out.data_ptr(), B, H, W, OH, OW);
      }) AT_DISPATCH_CASE(torch::kHalf, [&] {
        hipLaunchKernelGGL((patch_embed_s1_fwd_kernel<__half>), dim3(blocks),
                           dim3(256), 0, stream, (const __half*)xc.data_ptr(),
                           (const __half*)wc.data_ptr(),
                           hb ? (const __half*)bc.data_ptr() : nullptr,
                           (__half*)out.data_ptr(), B, H, W, OH, OW);
      }) AT_DISPATCH_CASE(torch::kFloat, [&] {
        hipLaunchKernelGGL((patch_embed_s1_fwd_kernel<float>), dim3(blocks),
                           dim3(256), 0, stream, (const float*)xc.data_ptr(),
                           (const float*)wc.data_ptr(),
                           hb ? (const float*)bc.data_ptr() : nullptr,
                           (float*)out.data_ptr(), B, H, W, OH, OW);
      }));
  HIP_CHECK_KERNEL();
  return out;
}

std::vector<torch::Tensor> patch_embed_s1_wrw(torch::Tensor dout, torch::Tensor x) {
  const int B = x.size(0), H = x.size(2), W = x.size(3);
  const int OH = dout.size(2), OW = dout.size(3);
  auto dc = dout.contiguous();
  auto xc = x.contiguous();
  auto dw = torch::zeros({32, 3, 7, 7}, x.options().dtype(torch::kFloat));
  auto db = torch::zeros({32}, x.options().dtype(torch::kFloat));
  const int slabs = 12;
  auto stream = at::hip::getCurrentHIPStream().stream();
  AT_DISPATCH_SWITCH(x.scalar_type(), "pe1w", AT_DISPATCH_CASE(
      torch::kBFloat16, [&] {
        hipLaunchKernelGGL((patch_embed_s1_wrw_kernel<__hip_bfloat16>),
                           dim3(49, slabs), dim3(256), 0, stream,
                           (const __hip_bfloat16*)dc.data_ptr(),
                           (const __hip_bfloat16*)xc.data_ptr(),
                           dw.data_ptr<float>(), db.data_ptr<float>(),
                           B, H, W, OH, OW, slabs);
      }) AT_DISPATCH_CASE(torch::kHalf, [&] {
        hipLaunchKernelGGL((patch_embed_s1_wrw_kernel<__half>),
                           dim3(49, slabs), dim3(256), 0, stream,
                           (const __half*)dc.data_ptr(),
                           (const __half*)xc.data_ptr(),
                           dw.data_ptr<float>(), db.data_ptr<float>(),
                           B, H, W, OH, OW, slabs);
      }) AT_DISPATCH_CASE(torch::kFloat, [&] {
        hipLaunchKernelGGL((patch_embed_s1_wrw_kernel<float>),
                           dim3(49, slabs), dim3(256), 0, stream,
                           (const float*)dc.data_ptr(),
                           (const float*)xc.data_ptr(),
                           dw.data_ptr<float>(), db.data_ptr<float>(),
                           B, H, W, OH, OW, slabs);
      }));
  HIP_CHECK_KERNEL();
  return {dw, db};
}
