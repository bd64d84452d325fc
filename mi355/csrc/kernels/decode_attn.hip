// Incremental-decode attention: one query token against a KV cache
// (SURVEY.md K11 — the generate() hot loop, batch-inference config:
// B=256, H=12, D=64, steps <= 128, fp16/bf16).
//
// One wave per (b, h): phase 1 scores 64 keys at a time (each lane owns one
// key row -> contiguous 128 B per lane), phase 2 online-softmax update,
// phase 3 rank-1 accumulate o[d=lane] += p_j * V[j][d] — coalesced across
// lanes for each j. Memory-bound by the single KV read, which is optimal.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../common.h"

namespace {

struct DStr { long sb, sh, ss; };

template <typename T, int D>
__global__ void decode_attn_kernel(const T* __restrict__ q, const T* __restrict__ kc,
                                   const T* __restrict__ vc, const float* __restrict__ bias,
                                   const float* __restrict__ kpad,
                                   T* __restrict__ out, int B, int H, int L, long t_max,
                                   DStr qs, DStr kst, DStr vst, DStr ost,
                                   int bias_b_stride /*0 when broadcast*/) {
  static_assert(D == 32 || D == 64, "decode_attn: D in {32,64}");
  const int bh = blockIdx.x * (blockDim.x / WAVE) + threadIdx.x / WAVE;
  if (bh >= B * H) return;
  const int b = bh / H, h = bh % H;
  const int lane = threadIdx.x & (WAVE - 1);

  const T* qrow = q + (long)b * qs.sb + (long)h * qs.sh;
  const T* kbase = kc + (long)b * kst.sb + (long)h * kst.sh;
  const T* vbase = vc + (long)b * vst.sb + (long)h * vst.sh;
  const float* brow = bias ? bias + (long)(bias_b_stride ? b : 0) * H * L + (long)h * L : nullptr;

  // q into registers: every lane holds the full row (D<=64 floats, L1 broadcast)
  float qreg[D];
#pragma unroll
  for (int d = 0; d < D; ++d) qreg[d] = DT<T>::to_f32(qrow[d]);

  __shared__ float p_lds[4][WAVE];  // up to 4 waves per block
  const int wid = threadIdx.x / WAVE;

  float m = -INFINITY, l = 0.f;
  float oacc[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) oacc[j] = 0.f;

  for (int j0 = 0; j0 < L; j0 += WAVE) {
    const int j = j0 + lane;
    float s = -INFINITY;
    if (j < L) {
      const T* krow = kbase + (long)j * kst.ss;
      float acc = 0.f;
      constexpr int KV = sizeof(T) == 4 ? 4 : 8;
#pragma unroll
      for (int d = 0; d < D; d += KV) {
        VecT<T, KV> kv = vload<T, KV>(krow + d);
#pragma unroll
        for (int u = 0; u < KV; ++u) acc += qreg[d + u] * DT<T>::to_f32(kv.data[u]);
      }
      s = acc + (brow ? brow[j] : 0.f) + (kpad ? kpad[(long)b * L + j] : 0.f);
    }
    const float tile_max = wave_reduce_max(s);
    const float mn = fmaxf(m, tile_max);
    const float alpha = (m == -INFINITY) ? 0.f : __expf(m - mn);
    const float p = (j < L) ? __expf(s - mn) : 0.f;
    p_lds[wid][lane] = p;
    l = l * alpha + wave_reduce_sum(p);
    // o accumulate, 8 keys x 8 dims per lane per pass: lane handles key
    // j0 + 8*pass + (lane>>3), dims 8*(lane&7)..+8 — each instruction moves
    // a full 1 KiB per wave instead of 128 B (the old d=lane layout)
    constexpr int DCH = 8;                 // dims per lane chunk
    static_assert(D % DCH == 0, "");
    const int dgrp = lane & 7;             // dim chunk id (D=64: 8 chunks)
    const int krow = lane >> 3;            // 8 keys in flight
#pragma unroll
    for (int u = 0; u < DCH; ++u) oacc[u] *= alpha;
    const int jmax = min(WAVE, L - j0);
    for (int jj = krow; jj < jmax; jj += 8) {
      const float pj = p_lds[wid][jj];
      if (dgrp * DCH < D) {
        const T* vrow = vbase + (long)(j0 + jj) * vst.ss + dgrp * DCH;
        VecT<T, 8> vv = vload<T, 8>(vrow);
#pragma unroll
        for (int u = 0; u < DCH; ++u) oacc[u] += pj * DT<T>::to_f32(vv.data[u]);
      }
    }
    m = mn;
  }
  // fold the 8 key-groups: lanes l and l^8,l^16,l^32 share the same dim chunk
#pragma unroll
  for (int u = 0; u < 8; ++u) {
#pragma unroll
    for (int off = 8; off < WAVE; off <<= 1) oacc[u] += __shfl_xor(oacc[u], off, WAVE);
  }
  const float invl = (l > 0.f) ? 1.f / l : 0.f;
  if ((lane >> 3) == 0) {  // lanes 0..7 hold the final chunks
    const int d0 = (lane & 7) * 8;
    if (d0 < D) {
      VecT<T, 8> ov;
#pragma unroll
      for (int u = 0; u < 8; ++u) ov.data[u] = DT<T>::from_f32(oacc[u] * invl);
      vstore<T, 8>(out + (long)b * ost.sb + (long)h * ost.sh + d0, ov);
    }
  }
}

}  // namespace

torch::Tensor decode_attn(torch::Tensor q, torch::Tensor kc, torch::Tensor vc,
                          long cache_len, c10::optional<torch::Tensor> bias,
                          c10::optional<torch::Tensor> kpad) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 4 && q.size(2) == 1 && q.stride(3) == 1,
              "q must be (B,H,1,D), d contiguous");
  const int B = q.size(0), H = q.size(1), D = q.size(3);
  const long t_max = kc.size(2);
  auto out = torch::empty({B, 1, H, D}, q.options()).permute({0, 2, 1, 3});
  DStr qs{q.stride(0), q.stride(1), q.stride(2)};
  DStr kst{kc.stride(0), kc.stride(1), kc.stride(2)};
  DStr vst{vc.stride(0), vc.stride(1), vc.stride(2)};
  DStr ost{out.stride(0), out.stride(1), out.stride(2)};
  auto stream = at::hip::getCurrentHIPStream().stream();
  const float* bias_ptr = nullptr;
  const float* kpad_ptr = nullptr;
  int bias_b_stride = 0;
  torch::Tensor bias_f;
  if (kpad.has_value() && kpad->defined()) {
    TORCH_CHECK(kpad->size(-1) == cache_len && kpad->scalar_type() == torch::kFloat);
    kpad_ptr = kpad->data_ptr<float>();
  }
  if (bias.has_value() && bias->defined()) {
    bias_f = bias->to(torch::kFloat).contiguous();
    TORCH_CHECK(bias_f.size(-1) == cache_len, "bias last dim must equal cache_len");
    bias_ptr = bias_f.data_ptr<float>();
    bias_b_stride = bias_f.size(0) > 1 ? 1 : 0;
  }
  const int waves_per_block = 4;
  int blocks = (B * H + waves_per_block - 1) / waves_per_block;
#define CASE(T, DD)                                                                   \
  hipLaunchKernelGGL((decode_attn_kernel<T, DD>), dim3(blocks), dim3(waves_per_block * WAVE), \
                     0, stream, (const T*)q.data_ptr(), (const T*)kc.data_ptr(),       \
                     (const T*)vc.data_ptr(), bias_ptr, kpad_ptr, (T*)out.data_ptr(),  \
                     B, H, (int)cache_len, t_max, qs, kst, vst, ost, bias_b_stride)
  if (q.scalar_type() == torch::kBFloat16 && D == 64) CASE(__hip_bfloat16, 64);
  else if (q.scalar_type() == torch::kHalf && D == 64) CASE(__half, 64);
  else if (q.scalar_type() == torch::kFloat && D == 64) CASE(float, 64);
  else if (q.scalar_type() == torch::kBFloat16 && D == 32) CASE(__hip_bfloat16, 32);
  else if (q.scalar_type() == torch::kHalf && D == 32) CASE(__half, 32);
  else if (q.scalar_type() == torch::kFloat && D == 32) CASE(float, 32);
  else TORCH_CHECK(false, "decode_attn: unsupported dtype/D=", D);
#undef CASE
  HIP_CHECK_KERNEL();
  return out;
}
