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

// packed form: x = (rows, 2F) GEMM output with [gate | up] halves (no
// split/clone of the fused wi projection), out = (rows, F), with the FF
// residual dropout fused on the output (hash mask, graph-safe device seed —
// same convention as rmsnorm.hip). bwd writes dgate/dup into ONE (rows, 2F)
// tensor so autograd sees a single grad for the GEMM output (no cat).
template <typename T, int VEC, bool DROP>
__global__ void geglu_packed_fwd_kernel(const T* __restrict__ x, T* __restrict__ out,
                                        long n, long F, float p, float keep_inv,
                                        const long* __restrict__ seed_ptr, long salt) {
  const long long seed = (DROP ? *seed_ptr : 0) ^ salt;
  long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * VEC;
  long stride = (long)gridDim.x * blockDim.x * VEC;
  for (; i < n; i += stride) {
    const long row = i / F, col = i - row * F;
    const T* base = x + row * 2 * F + col;
    VecT<T, VEC> g = vload<T, VEC>(base);
    VecT<T, VEC> u = vload<T, VEC>(base + F);
    VecT<T, VEC> o;
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      float h = gelu_tanh(DT<T>::to_f32(g.data[j])) * DT<T>::to_f32(u.data[j]);
      if (DROP) h = (hash_u01(seed, i + j) >= p) ? h * keep_inv : 0.f;
      o.data[j] = DT<T>::from_f32(h);
    }
    vstore<T, VEC>(out + i, o);
  }
}

template <typename T, int VEC, bool DROP>
__global__ void geglu_packed_bwd_kernel(const T* __restrict__ dout,
                                        const T* __restrict__ x, T* __restrict__ dx,
                                        long n, long F, float p, float keep_inv,
                                        const long* __restrict__ seed_ptr, long salt) {
  const long long seed = (DROP ? *seed_ptr : 0) ^ salt;
  long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * VEC;
  long stride = (long)gridDim.x * blockDim.x * VEC;
  for (; i < n; i += stride) {
    const long row = i / F, col = i - row * F;
    const T* base = x + row * 2 * F + col;
    VecT<T, VEC> dv = vload<T, VEC>(dout + i);
    VecT<T, VEC> g = vload<T, VEC>(base);
    VecT<T, VEC> u = vload<T, VEC>(base + F);
    VecT<T, VEC> og, ou;
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      float d = DT<T>::to_f32(dv.data[j]);
      if (DROP) d = (hash_u01(seed, i + j) >= p) ? d * keep_inv : 0.f;
      float gf = DT<T>::to_f32(g.data[j]);
      float uf = DT<T>::to_f32(u.data[j]);
      og.data[j] = DT<T>::from_f32(d * uf * gelu_tanh_grad(gf));
      ou.data[j] = DT<T>::from_f32(d * gelu_tanh(gf));
    }
    T* dbase = dx + row * 2 * F + col;
    vstore<T, VEC>(dbase, og);
    vstore<T, VEC>(dbase + F, ou);
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

torch::Tensor geglu_packed_fwd(torch::Tensor x, double p_drop,
                               c10::optional<torch::Tensor> seed, long salt) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.size(-1) % 2 == 0);
  const long F = x.size(-1) / 2;
  const long rows = x.numel() / (2 * F);
  auto out_sizes = x.sizes().vec();
  out_sizes.back() = F;
  auto out = torch::empty(out_sizes, x.options());
  auto stream = at::hip::getCurrentHIPStream().stream();
  const long n = rows * F;
  const bool drop = p_drop > 0.0;
  const float keep_inv = drop ? 1.0f / (1.0f - (float)p_drop) : 1.0f;
  const long* sp = drop ? seed->data_ptr<long>() : nullptr;
#define GP_FWD(T, VEC)                                                            \
  do {                                                                            \
    TORCH_CHECK(F % VEC == 0, "geglu_packed: F must be a multiple of ", VEC);     \
    long blocks = std::min<long>((n / VEC + 255) / 256, 2048);                    \
    if (drop)                                                                     \
      hipLaunchKernelGGL((geglu_packed_fwd_kernel<T, VEC, true>), dim3(blocks),   \
                         dim3(256), 0, stream, (const T*)x.data_ptr(),            \
                         (T*)out.data_ptr(), n, F, (float)p_drop, keep_inv, sp,   \
                         salt);                                                   \
    else                                                                          \
      hipLaunchKernelGGL((geglu_packed_fwd_kernel<T, VEC, false>), dim3(blocks),  \
                         dim3(256), 0, stream, (const T*)x.data_ptr(),            \
                         (T*)out.data_ptr(), n, F, 0.f, 1.f, nullptr, salt);      \
    HIP_CHECK_KERNEL();                                                           \
  } while (0)
  if (x.scalar_type() == torch::kBFloat16) GP_FWD(__hip_bfloat16, 8);
  else if (x.scalar_type() == torch::kHalf) GP_FWD(__half, 8);
  else if (x.scalar_type() == torch::kFloat) GP_FWD(float, 4);
  else TORCH_CHECK(false, "geglu_packed: unsupported dtype");
#undef GP_FWD
  return out;
}

torch::Tensor geglu_packed_bwd(torch::Tensor dout, torch::Tensor x, double p_drop,
                               c10::optional<torch::Tensor> seed, long salt) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && dout.is_contiguous());
  const long F = x.size(-1) / 2;
  const long rows = x.numel() / (2 * F);
  auto dx = torch::empty_like(x);
  auto stream = at::hip::getCurrentHIPStream().stream();
  const long n = rows * F;
  const bool drop = p_drop > 0.0;
  const float keep_inv = drop ? 1.0f / (1.0f - (float)p_drop) : 1.0f;
  const long* sp = drop ? seed->data_ptr<long>() : nullptr;
#define GP_BWD(T, VEC)                                                            \
  do {                                                                            \
    TORCH_CHECK(F % VEC == 0, "geglu_packed: F must be a multiple of ", VEC);     \
    long blocks = std::min<long>((n / VEC + 255) / 256, 2048);                    \
    if (drop)                                                                     \
      hipLaunchKernelGGL((geglu_packed_bwd_kernel<T, VEC, true>), dim3(blocks),   \
                         dim3(256), 0, stream, (const T*)dout.data_ptr(),         \
                         (const T*)x.data_ptr(), (T*)dx.data_ptr(), n, F,         \
                         (float)p_drop, keep_inv, sp, salt);                      \
    else                                                                          \
      hipLaunchKernelGGL((geglu_packed_bwd_kernel<T, VEC, false>), dim3(blocks),  \
                         dim3(256), 0, stream, (const T*)dout.data_ptr(),         \
                         (const T*)x.data_ptr(), (T*)dx.data_ptr(), n, F, 0.f,    \
                         1.f, nullptr, salt);                                     \
    HIP_CHECK_KERNEL();                                                           \
  } while (0)
  if (x.scalar_type() == torch::kBFloat16) GP_BWD(__hip_bfloat16, 8);
  else if (x.scalar_type() == torch::kHalf) GP_BWD(__half, 8);
  else if (x.scalar_type() == torch::kFloat) GP_BWD(float, 4);
  else TORCH_CHECK(false, "geglu_packed: unsupported dtype");
#undef GP_BWD
  return dx;
}

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
