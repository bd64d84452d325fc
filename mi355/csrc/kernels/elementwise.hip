// Fused elementwise kernels: gated-GELU combine fwd/bwd.
// Reference op: mi355/ops/torch_ref.py::geglu. Memory-bound — 16 B/lane
// vectorized grid-stride loops (guide Appendix B).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../common.h"

namespace {

constexpr float K0 = 0.7978845608028654f;   // sqrt(2/pi)
constexpr float K1 = 0.044715f;

DEVINL float gelu_tanh(float x) {
  float u = K0 * (x + K1 * x * x * x);
  return 0.5f * x * (1.0f + tanhf(u));
}

DEVINL float gelu_tanh_grad(float x) {
  float x2 = x * x;
  float u = K0 * (x + K1 * x * x2);
  float t = tanhf(u);
  float sech2 = 1.0f - t * t;
  return 0.5f * (1.0f + t) + 0.5f * x * sech2 * K0 * (1.0f + 3.0f * K1 * x2);
}

template <typename T, int VEC>
__global__ void geglu_fwd_kernel(const T* __restrict__ gate, const T* __restrict__ up,
                                 T* __restrict__ out, long n) {
  long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * VEC;
  long stride = (long)gridDim.x * blockDim.x * VEC;
  for (; i < n; i += stride) {
    VecT<T, VEC> g = vload<T, VEC>(gate + i);
    VecT<T, VEC> u = vload<T, VEC>(up + i);
    VecT<T, VEC> o;
#pragma unroll
    for (int j = 0; j < VEC; ++j)
      o.data[j] = DT<T>::from_f32(gelu_tanh(DT<T>::to_f32(g.data[j])) *
                                  DT<T>::to_f32(u.data[j]));
    vstore<T, VEC>(out + i, o);
  }
}

template <typename T, int VEC>
__global__ void geglu_bwd_kernel(const T* __restrict__ dout, const T* __restrict__ gate,
                                 const T* __restrict__ up, T* __restrict__ dgate,
                                 T* __restrict__ dup, long n) {
  long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * VEC;
  long stride = (long)gridDim.x * blockDim.x * VEC;
  for (; i < n; i += stride) {
    VecT<T, VEC> dv = vload<T, VEC>(dout + i);
    VecT<T, VEC> g = vload<T, VEC>(gate + i);
    VecT<T, VEC> u = vload<T, VEC>(up + i);
    VecT<T, VEC> og, ou;
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      float d = DT<T>::to_f32(dv.data[j]);
      float gf = DT<T>::to_f32(g.data[j]);
      float uf = DT<T>::to_f32(u.data[j]);
      og.data[j] = DT<T>::from_f32(d * uf * gelu_tanh_grad(gf));
      ou.data[j] = DT<T>::from_f32(d * gelu_tanh(gf));
    }
    vstore<T, VEC>(dgate + i, og);
    vstore<T, VEC>(dup + i, ou);
  }
}

template <typename T>
void launch_geglu_fwd(const torch::Tensor& g, const torch::Tensor& u, torch::Tensor& o,
                      hipStream_t stream) {
  constexpr int VEC = sizeof(T) == 4 ? 4 : 8;
  long n = g.numel();
  TORCH_CHECK(n % VEC == 0, "geglu: numel must be a multiple of ", VEC);
  long blocks = std::min<long>((n / VEC + 255) / 256, 2048);
  hipLaunchKernelGGL((geglu_fwd_kernel<T, VEC>), dim3(blocks), dim3(256), 0, stream,
                     (const T*)g.data_ptr(), (const T*)u.data_ptr(), (T*)o.data_ptr(), n);
  HIP_CHECK_KERNEL();
}

template <typename T>
void launch_geglu_bwd(const torch::Tensor& d, const torch::Tensor& g, const torch::Tensor& u,
                      torch::Tensor& dg, torch::Tensor& du, hipStream_t stream) {
  constexpr int VEC = sizeof(T) == 4 ? 4 : 8;
  long n = g.numel();
  long blocks = std::min<long>((n / VEC + 255) / 256, 2048);
  hipLaunchKernelGGL((geglu_bwd_kernel<T, VEC>), dim3(blocks), dim3(256), 0, stream,
                     (const T*)d.data_ptr(), (const T*)g.data_ptr(), (const T*)u.data_ptr(),
                     (T*)dg.data_ptr(), (T*)du.data_ptr(), n);
  HIP_CHECK_KERNEL();
}

}  // namespace

torch::Tensor geglu_fwd(torch::Tensor gate, torch::Tensor up) {
  TORCH_CHECK(gate.is_cuda() && gate.is_contiguous() && up.is_contiguous());
  auto out = torch::empty_like(gate);
  auto stream = at::hip::getCurrentHIPStream().stream();
  if (gate.scalar_type() == torch::kBFloat16)
    launch_geglu_fwd<__hip_bfloat16>(gate, up, out, stream);
  else if (gate.scalar_type() == torch::kHalf)
    launch_geglu_fwd<__half>(gate, up, out, stream);
  else if (gate.scalar_type() == torch::kFloat)
    launch_geglu_fwd<float>(gate, up, out, stream);
  else
    TORCH_CHECK(false, "geglu: unsupported dtype");
  return out;
}

std::vector<torch::Tensor> geglu_bwd(torch::Tensor dout, torch::Tensor gate,
                                     torch::Tensor up) {
  auto dg = torch::empty_like(gate);
  auto du = torch::empty_like(up);
  auto stream = at::hip::getCurrentHIPStream().stream();
  if (gate.scalar_type() == torch::kBFloat16)
    launch_geglu_bwd<__hip_bfloat16>(dout, gate, up, dg, du, stream);
  else if (gate.scalar_type() == torch::kHalf)
    launch_geglu_bwd<__half>(dout, gate, up, dg, du, stream);
  else if (gate.scalar_type() == torch::kFloat)
    launch_geglu_bwd<float>(dout, gate, up, dg, du, stream);
  else
    TORCH_CHECK(false, "geglu: unsupported dtype");
  return {dg, du};
}
