// Fused vocab cross-entropy: log-softmax + NLL over 32k vocab.
// Reference op: mi355/ops/torch_ref.py::cross_entropy. One block per row,
// single-pass online max/sum in fp32, then exact lse. Backward writes
// dlogits = scale * (softmax - onehot) in one elementwise pass.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../common.h"

namespace {

template <typename T, int VEC>
__global__ void ce_fwd_kernel(const T* __restrict__ logits, const long* __restrict__ labels,
                              float* __restrict__ loss_sum, float* __restrict__ lse_out,
                              int* __restrict__ nvalid, int rows, int V, long ignore_index) {
  __shared__ float red[16];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const long lab = labels[row];
    const T* lr = logits + (long)row * V;
    // online max + sum(exp(x - max)) per thread over strided vectors
    float m = -INFINITY, s = 0.f;
    for (int i = threadIdx.x * VEC; i < V; i += blockDim.x * VEC) {
      VecT<T, VEC> v = vload<T, VEC>(lr + i);
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float f = DT<T>::to_f32(v.data[j]);
        if (f > m) {
          s *= __expf(m - f);
          m = f;
        }
        s += __expf(f - m);
      }
    }
    // combine across threads: global max then rescale sums
    float gm = block_reduce_max(m, red);
    s *= __expf(m - gm);
    float gs = block_reduce_sum(s, red);
    float lse = gm + logf(gs);
    if (threadIdx.x == 0) {
      lse_out[row] = lse;
      if (lab != ignore_index) {
        float lv = DT<T>::to_f32(lr[lab]);
        atomicAdd(loss_sum, lse - lv);
        atomicAdd(nvalid, 1);
      }
    }
    __syncthreads();
  }
}

template <typename T, int VEC>
__global__ void ce_bwd_kernel(const T* __restrict__ logits, const long* __restrict__ labels,
                              const float* __restrict__ lse, T* __restrict__ dlogits,
                              int rows, int V, long ignore_index,
                              const float* __restrict__ dloss,
                              const int* __restrict__ nvalid) {
  const float scale = dloss[0] / (float)max(nvalid[0], 1);
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const long lab = labels[row];
    const T* lr = logits + (long)row * V;
    T* dr = dlogits + (long)row * V;
    const float l = lse[row];
    const bool valid = lab != ignore_index;
    for (int i = threadIdx.x * VEC; i < V; i += blockDim.x * VEC) {
      VecT<T, VEC> v = vload<T, VEC>(lr + i);
      VecT<T, VEC> o;
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        if (!valid) {
          o.data[j] = DT<T>::from_f32(0.f);
        } else {
          float p = __expf(DT<T>::to_f32(v.data[j]) - l);
          float g = p - ((i + j) == lab ? 1.f : 0.f);
          o.data[j] = DT<T>::from_f32(scale * g);
        }
      }
      vstore<T, VEC>(dr + i, o);
    }
  }
}

}  // namespace

std::vector<torch::Tensor> ce_fwd(torch::Tensor logits, torch::Tensor labels,
                                  long ignore_index) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2 && logits.is_contiguous());
  TORCH_CHECK(labels.scalar_type() == torch::kLong);
  const int rows = logits.size(0), V = logits.size(1);
  auto loss = torch::zeros({}, logits.options().dtype(torch::kFloat));
  auto lse = torch::empty({rows}, logits.options().dtype(torch::kFloat));
  auto nvalid = torch::zeros({}, logits.options().dtype(torch::kInt));
  auto stream = at::hip::getCurrentHIPStream().stream();
  int grid = std::min(rows, 2048);
#define CASE(T, VEC)                                                                  \
  hipLaunchKernelGGL((ce_fwd_kernel<T, VEC>), dim3(grid), dim3(256), 0, stream,       \
                     (const T*)logits.data_ptr(), labels.data_ptr<long>(),            \
                     loss.data_ptr<float>(), lse.data_ptr<float>(),                   \
                     nvalid.data_ptr<int>(), rows, V, ignore_index)
  if (logits.scalar_type() == torch::kBFloat16) CASE(__hip_bfloat16, 8);
  else if (logits.scalar_type() == torch::kHalf) CASE(__half, 8);
  else if (logits.scalar_type() == torch::kFloat) CASE(float, 4);
  else TORCH_CHECK(false, "ce: unsupported dtype");
#undef CASE
  HIP_CHECK_KERNEL();
  return {loss, lse, nvalid};
}

torch::Tensor ce_bwd(torch::Tensor logits, torch::Tensor labels, torch::Tensor lse,
                     long ignore_index, torch::Tensor dloss, torch::Tensor nvalid) {
  const int rows = logits.size(0), V = logits.size(1);
  auto dlogits = torch::empty_like(logits);
  auto stream = at::hip::getCurrentHIPStream().stream();
  int grid = std::min(rows, 2048);
#define CASE(T, VEC)                                                                  \
  hipLaunchKernelGGL((ce_bwd_kernel<T, VEC>), dim3(grid), dim3(256), 0, stream,       \
                     (const T*)logits.data_ptr(), labels.data_ptr<long>(),            \
                     lse.data_ptr<float>(), (T*)dlogits.data_ptr(), rows, V,          \
                     ignore_index, dloss.data_ptr<float>(), nvalid.data_ptr<int>())
  if (logits.scalar_type() == torch::kBFloat16) CASE(__hip_bfloat16, 8);
  else if (logits.scalar_type() == torch::kHalf) CASE(__half, 8);
  else if (logits.scalar_type() == torch::kFloat) CASE(float, 4);
  else TORCH_CHECK(false, "ce: unsupported dtype");
#undef CASE
  HIP_CHECK_KERNEL();
  return dlogits;
}
