// Fused AdamW with fp32 master weights + low-precision weight write-back
// (SURVEY.md K10). With the FlatReplica layout the whole model is 1-2 flat
// tensors, so a train step's optimizer is 1-2 launches of this kernel.
// grad_scale (gradient clipping) can come from device memory (0-dim tensor)
// so clipping never forces a host sync — keeps the step hipGraph-capturable.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../common.h"

namespace {

template <typename G, typename P, int VEC>
__global__ void adamw_kernel(float* __restrict__ master, const G* __restrict__ grad,
                             float* __restrict__ m, float* __restrict__ v,
                             P* __restrict__ lowp, long n, float lr, float beta1,
                             float beta2, float eps, float wd, float bc1, float bc2,
                             float grad_scale, const float* __restrict__ grad_scale_dev) {
  const float gs = grad_scale_dev ? fminf(*grad_scale_dev, 1.0f) : grad_scale;
  long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * VEC;
  const long stride = (long)gridDim.x * blockDim.x * VEC;
  for (; i + VEC <= n; i += stride) {
    VecT<G, VEC> gv = vload<G, VEC>(grad + i);
    VecT<float, VEC> mv = vload<float, VEC>(m + i);
    VecT<float, VEC> vv = vload<float, VEC>(v + i);
    VecT<float, VEC> pv = vload<float, VEC>(master + i);
    VecT<P, VEC> lv;
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      float g = DT<G>::to_f32(gv.data[j]) * gs;
      float mi = beta1 * mv.data[j] + (1.f - beta1) * g;
      float vi = beta2 * vv.data[j] + (1.f - beta2) * g * g;
      mv.data[j] = mi;
      vv.data[j] = vi;
      float denom = sqrtf(vi / bc2) + eps;
      float p = pv.data[j] * (1.f - lr * wd);
      p -= lr * (mi / bc1) / denom;
      pv.data[j] = p;
      lv.data[j] = DT<P>::from_f32(p);
    }
    vstore<float, VEC>(m + i, mv);
    vstore<float, VEC>(v + i, vv);
    vstore<float, VEC>(master + i, pv);
    if (lowp) vstore<P, VEC>(lowp + i, lv);
  }
  // ragged tail: scalar path on block 0
  if (blockIdx.x == 0 && threadIdx.x < (n % VEC)) {
    long t = n - (n % VEC) + threadIdx.x;
    float g = DT<G>::to_f32(grad[t]) * gs;
    float mi = beta1 * m[t] + (1.f - beta1) * g;
    float vi = beta2 * v[t] + (1.f - beta2) * g * g;
    m[t] = mi;
    v[t] = vi;
    float denom = sqrtf(vi / bc2) + eps;
    float p = master[t] * (1.f - lr * wd);
    p -= lr * (mi / bc1) / denom;
    master[t] = p;
    if (lowp) lowp[t] = DT<P>::from_f32(p);
  }
}

template <typename G, typename P>
void launch(torch::Tensor& master, const torch::Tensor& grad, torch::Tensor& m,
            torch::Tensor& v, torch::Tensor* lowp, float lr, float b1, float b2,
            float eps, float wd, int step, float gscale,
            const torch::Tensor* gscale_dev, hipStream_t stream) {
  long n = master.numel();
  float bc1 = 1.f - powf(b1, (float)step);
  float bc2 = 1.f - powf(b2, (float)step);
  // VEC=8: 16-B grad/lowp accesses and 2x128-B m/v/master per lane —
  // wider VMEM per instruction lifted the kernel from 4.6 to ~5+ TB/s
  long blocks = std::min<long>((n / 8 + 255) / 256, 16384);
  hipLaunchKernelGGL((adamw_kernel<G, P, 8>), dim3(blocks), dim3(256), 0, stream,
                     master.data_ptr<float>(), (const G*)grad.data_ptr(),
                     m.data_ptr<float>(), v.data_ptr<float>(),
                     lowp ? (P*)lowp->data_ptr() : nullptr, n, lr, b1, b2, eps, wd,
                     bc1, bc2, gscale,
                     gscale_dev ? gscale_dev->data_ptr<float>() : nullptr);
  HIP_CHECK_KERNEL();
}

// sum of squares of a flat tensor, accumulated into a 0-dim fp32 tensor on
// device (grad-clip total-norm without the .float().pow(2).sum() chain:
// one read of the bf16 grads instead of a 2x fp32 round trip, no host sync)
template <typename T, int VEC>
__global__ void sumsq_kernel(const T* __restrict__ x, float* __restrict__ out,
                             long n) {
  __shared__ float smem[16];
  float acc = 0.f;
  long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * VEC;
  const long stride = (long)gridDim.x * blockDim.x * VEC;
  for (; i + VEC <= n; i += stride) {
    VecT<T, VEC> v = vload<T, VEC>(x + i);
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      float f = DT<T>::to_f32(v.data[j]);
      acc += f * f;
    }
  }
  // ragged tail (n not a multiple of VEC): last partial vector, scalar loads
  if (blockIdx.x == 0 && threadIdx.x < (n % VEC)) {
    float f = DT<T>::to_f32(x[n - (n % VEC) + threadIdx.x]);
    acc += f * f;
  }
  acc = block_reduce_sum(acc, smem);
  if (threadIdx.x == 0) atomicAdd(out, acc);
}

}  // namespace

void sumsq_into(torch::Tensor x, torch::Tensor out) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "sumsq: contiguous cuda input");
  TORCH_CHECK(out.is_cuda() && out.scalar_type() == torch::kFloat &&
              out.numel() == 1, "sumsq: out must be 0-dim fp32 cuda");
  auto stream = at::hip::getCurrentHIPStream().stream();
  long n = x.numel();
  long blocks = std::min<long>((n / 8 + 255) / 256 + 1, 2048);
#define SUMSQ(T)                                                        \
  hipLaunchKernelGGL((sumsq_kernel<T, 8>), dim3(blocks), dim3(256), 0,  \
                     stream, (const T*)x.data_ptr(),                    \
                     out.data_ptr<float>(), n);                         \
  HIP_CHECK_KERNEL()
  if (x.scalar_type() == torch::kBFloat16) { SUMSQ(__hip_bfloat16); }
  else if (x.scalar_type() == torch::kFloat) { SUMSQ(float); }
  else if (x.scalar_type() == torch::kHalf) { SUMSQ(__half); }
  else TORCH_CHECK(false, "sumsq: unsupported dtype");
#undef SUMSQ
}

void adamw_multi(std::vector<torch::Tensor> masters, std::vector<torch::Tensor> grads,
                 std::vector<torch::Tensor> ms, std::vector<torch::Tensor> vs,
                 std::vector<torch::Tensor> lowps, double lr, double beta1, double beta2,
                 double eps, double weight_decay, long step, torch::Tensor grad_scale) {
  auto stream = at::hip::getCurrentHIPStream().stream();
  // grad_scale: 0-dim fp32 tensor on device (no sync) or on CPU (constant)
  float gs_const = 1.0f;
  const torch::Tensor* gs_dev = nullptr;
  if (grad_scale.defined() && grad_scale.numel() == 1) {
    if (grad_scale.is_cuda()) gs_dev = &grad_scale;
    else gs_const = grad_scale.item<float>();
  }
  for (size_t t = 0; t < masters.size(); ++t) {
    TORCH_CHECK(masters[t].scalar_type() == torch::kFloat, "master must be fp32");
    torch::Tensor* lowp = (t < lowps.size() && lowps[t].defined()) ? &lowps[t] : nullptr;
    auto gt = grads[t].scalar_type();
    auto pt = lowp ? lowp->scalar_type() : torch::kFloat;
#define DISPATCH(GT, PT)                                                            \
  launch<GT, PT>(masters[t], grads[t], ms[t], vs[t], lowp, (float)lr, (float)beta1, \
                 (float)beta2, (float)eps, (float)weight_decay, (int)step,          \
                 gs_const, gs_dev, stream)
    if (gt == torch::kBFloat16 && pt == torch::kBFloat16) DISPATCH(__hip_bfloat16, __hip_bfloat16);
    else if (gt == torch::kFloat && pt == torch::kFloat) DISPATCH(float, float);
    else if (gt == torch::kFloat && pt == torch::kBFloat16) DISPATCH(float, __hip_bfloat16);
    else if (gt == torch::kHalf && pt == torch::kHalf) DISPATCH(__half, __half);
    else TORCH_CHECK(false, "adamw: unsupported grad/param dtype combo");
#undef DISPATCH
  }
}
