// T5LayerNorm (RMSNorm, no mean-subtract, no bias) forward + backward.
// Reference op: mi355/ops/torch_ref.py::rms_norm.
// Memory-bound: vectorized 2-byte loads (8 elems = 16 B per lane).
// Backward writes per-block dw partials to scratch (deterministic, no global
// atomics — 1024-block atomic contention on a 768-float dw was 33% of the
// whole train step) and reduces them in a second tiny kernel.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../common.h"

namespace {

// one block per row; D up to 8192 with 256 threads x 8-vec grid-stride
template <typename T, int VEC>
__global__ void rmsnorm_fwd_kernel(const T* __restrict__ x, const T* __restrict__ w,
                                   T* __restrict__ y, float* __restrict__ inv_rms,
                                   int rows, int D, float eps) {
  __shared__ float red[16];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + (long)row * D;
    T* yr = y + (long)row * D;
    float ss = 0.f;
    for (int i = threadIdx.x * VEC; i < D; i += blockDim.x * VEC) {
      VecT<T, VEC> v = vload<T, VEC>(xr + i);
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float f = DT<T>::to_f32(v.data[j]);
        ss += f * f;
      }
    }
    ss = block_reduce_sum(ss, red);
    float inv = rsqrtf(ss / D + eps);
    if (threadIdx.x == 0) inv_rms[row] = inv;
    for (int i = threadIdx.x * VEC; i < D; i += blockDim.x * VEC) {
      VecT<T, VEC> v = vload<T, VEC>(xr + i);
      VecT<T, VEC> wv = vload<T, VEC>(w + i);
      VecT<T, VEC> o;
#pragma unroll
      for (int j = 0; j < VEC; ++j)
        o.data[j] = DT<T>::from_f32(DT<T>::to_f32(v.data[j]) * inv *
                                    DT<T>::to_f32(wv.data[j]));
      vstore<T, VEC>(yr + i, o);
    }
    __syncthreads();
  }
}

// fused residual-add + RMSNorm: s = x + r; y = rmsnorm(s) (pre-norm T5
// chains "resid += delta; norm(resid)" — one kernel instead of add + norm)
// optional fused dropout on the delta branch: s = x + drop(r); mask derived
// from (seed, salt, element index) so backward reproduces it (common.h)
template <typename T, int VEC>
__global__ void add_rmsnorm_fwd_kernel(const T* __restrict__ x, const T* __restrict__ r,
                                       T* __restrict__ s_out, T* __restrict__ y,
                                       const T* __restrict__ w, float* __restrict__ inv_rms,
                                       int rows, int D, float eps,
                                       float p_drop, const long* __restrict__ seed_ptr,
                                       long salt) {
  __shared__ float red[16];
  const long long seed = (seed_ptr ? *seed_ptr : 0) ^ salt;
  const float keep_inv = p_drop > 0.f ? 1.f / (1.f - p_drop) : 1.f;
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + (long)row * D;
    const T* rr = r + (long)row * D;
    T* sr = s_out + (long)row * D;
    T* yr = y + (long)row * D;
    float ss = 0.f;
    for (int i = threadIdx.x * VEC; i < D; i += blockDim.x * VEC) {
      VecT<T, VEC> xv = vload<T, VEC>(xr + i);
      VecT<T, VEC> rv = vload<T, VEC>(rr + i);
      VecT<T, VEC> sv;
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float rf = DT<T>::to_f32(rv.data[j]);
        if (p_drop > 0.f) {
          const long long idx = (long long)row * D + i + j;
          rf = (hash_u01(seed, idx) >= p_drop) ? rf * keep_inv : 0.f;
        }
        float f = DT<T>::to_f32(xv.data[j]) + rf;
        sv.data[j] = DT<T>::from_f32(f);
        ss += f * f;
      }
      vstore<T, VEC>(sr + i, sv);
    }
    ss = block_reduce_sum(ss, red);
    float inv = rsqrtf(ss / D + eps);
    if (threadIdx.x == 0) inv_rms[row] = inv;
    for (int i = threadIdx.x * VEC; i < D; i += blockDim.x * VEC) {
      VecT<T, VEC> sv = vload<T, VEC>(sr + i);
      VecT<T, VEC> wv = vload<T, VEC>(w + i);
      VecT<T, VEC> o;
#pragma unroll
      for (int j = 0; j < VEC; ++j)
        o.data[j] = DT<T>::from_f32(DT<T>::to_f32(sv.data[j]) * inv *
                                    DT<T>::to_f32(wv.data[j]));
      vstore<T, VEC>(yr + i, o);
    }
    __syncthreads();
  }
}

// Single-pass add+RMSNorm for D <= threads*VEC: each thread owns ONE vector
// chunk, s = x + drop(r) stays in REGISTERS across the sum-of-squares
// reduction (the generic kernel re-reads s from global), and rows are
// processed in pairs to overlap the two HBM load chains (same rationale as
// rmsnorm_bwd_1p_kernel above).
template <typename T, int VEC>
__global__ void add_rmsnorm_fwd_1p_kernel(const T* __restrict__ x, const T* __restrict__ r,
                                          T* __restrict__ s_out, T* __restrict__ y,
                                          const T* __restrict__ w, float* __restrict__ inv_rms,
                                          int rows, int D, float eps,
                                          float p_drop, const long* __restrict__ seed_ptr,
                                          long salt) {
  __shared__ float red[16];
  const long long seed = (seed_ptr ? *seed_ptr : 0) ^ salt;
  const float keep_inv = p_drop > 0.f ? 1.f / (1.f - p_drop) : 1.f;
  const int i = threadIdx.x * VEC;
  const bool act = i < D;
  VecT<T, VEC> wv;
  if (act) wv = vload<T, VEC>(w + i);
  const int half = (rows + 1) >> 1;
  for (int p = blockIdx.x; p < half; p += gridDim.x) {
    const int r0 = p, r1 = p + half;
    const bool has1 = r1 < rows;
    VecT<T, VEC> sv0, sv1;
    float ss0 = 0.f, ss1 = 0.f;
    if (act) {
      VecT<T, VEC> xv0 = vload<T, VEC>(x + (long)r0 * D + i);
      VecT<T, VEC> rv0 = vload<T, VEC>(r + (long)r0 * D + i);
      VecT<T, VEC> xv1, rv1;
      if (has1) {
        xv1 = vload<T, VEC>(x + (long)r1 * D + i);
        rv1 = vload<T, VEC>(r + (long)r1 * D + i);
      }
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float rf = DT<T>::to_f32(rv0.data[j]);
        if (p_drop > 0.f) {
          const long long idx = (long long)r0 * D + i + j;
          rf = (hash_u01(seed, idx) >= p_drop) ? rf * keep_inv : 0.f;
        }
        const float f = DT<T>::to_f32(xv0.data[j]) + rf;
        sv0.data[j] = DT<T>::from_f32(f);
        ss0 += f * f;
      }
      vstore<T, VEC>(s_out + (long)r0 * D + i, sv0);
      if (has1) {
#pragma unroll
        for (int j = 0; j < VEC; ++j) {
          float rf = DT<T>::to_f32(rv1.data[j]);
          if (p_drop > 0.f) {
            const long long idx = (long long)r1 * D + i + j;
            rf = (hash_u01(seed, idx) >= p_drop) ? rf * keep_inv : 0.f;
          }
          const float f = DT<T>::to_f32(xv1.data[j]) + rf;
          sv1.data[j] = DT<T>::from_f32(f);
          ss1 += f * f;
        }
        vstore<T, VEC>(s_out + (long)r1 * D + i, sv1);
      }
    }
    ss0 = block_reduce_sum(ss0, red);
    ss1 = block_reduce_sum(ss1, red);
    if (!act) continue;
    const float inv0 = rsqrtf(ss0 / D + eps);
    const float inv1 = rsqrtf(ss1 / D + eps);
    if (threadIdx.x == 0) {
      inv_rms[r0] = inv0;
      if (has1) inv_rms[r1] = inv1;
    }
#pragma unroll
    for (int half_row = 0; half_row < 2; ++half_row) {
      if (half_row == 1 && !has1) break;
      const int row = half_row ? r1 : r0;
      const float inv = half_row ? inv1 : inv0;
      const VecT<T, VEC>& sv = half_row ? sv1 : sv0;
      VecT<T, VEC> o;
#pragma unroll
      for (int j = 0; j < VEC; ++j)
        o.data[j] = DT<T>::from_f32(DT<T>::to_f32(sv.data[j]) * inv *
                                    DT<T>::to_f32(wv.data[j]));
      vstore<T, VEC>(y + (long)row * D + i, o);
    }
  }
}

// dx = inv * w * dy - x * inv^3/D * sum(dy*w*x) ; dw_part[blk] = sum_rows dy*x*inv
template <typename T, int VEC, int MAXSLOT>
__global__ void rmsnorm_bwd_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                                   const T* __restrict__ w, const float* __restrict__ inv_rms,
                                   const T* __restrict__ dextra /*nullable*/,
                                   T* __restrict__ dx, float* __restrict__ dw_part,
                                   int rows, int D,
                                   T* __restrict__ ddelta /*nullable: dropout branch*/,
                                   float p_drop, const long* __restrict__ seed_ptr,
                                   long salt) {
  __shared__ float red[16];
  const long long seed = (seed_ptr ? *seed_ptr : 0) ^ salt;
  const float keep_inv = p_drop > 0.f ? 1.f / (1.f - p_drop) : 1.f;
  float dwacc[MAXSLOT * VEC];
#pragma unroll
  for (int j = 0; j < MAXSLOT * VEC; ++j) dwacc[j] = 0.f;

  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + (long)row * D;
    const T* dyr = dy + (long)row * D;
    T* dxr = dx + (long)row * D;
    const float inv = inv_rms[row];
    float dot = 0.f;
    for (int i = threadIdx.x * VEC; i < D; i += blockDim.x * VEC) {
      VecT<T, VEC> xv = vload<T, VEC>(xr + i);
      VecT<T, VEC> dyv = vload<T, VEC>(dyr + i);
      VecT<T, VEC> wv = vload<T, VEC>(w + i);
#pragma unroll
      for (int j = 0; j < VEC; ++j)
        dot += DT<T>::to_f32(dyv.data[j]) * DT<T>::to_f32(wv.data[j]) *
               DT<T>::to_f32(xv.data[j]);
    }
    dot = block_reduce_sum(dot, red);
    const float c = inv * inv * inv / D * dot;
    int slot = 0;
    for (int i = threadIdx.x * VEC; i < D; i += blockDim.x * VEC, ++slot) {
      VecT<T, VEC> xv = vload<T, VEC>(xr + i);
      VecT<T, VEC> dyv = vload<T, VEC>(dyr + i);
      VecT<T, VEC> wv = vload<T, VEC>(w + i);
      VecT<T, VEC> o;
      VecT<T, VEC> ev;
      if (dextra) ev = vload<T, VEC>(dextra + (long)row * D + i);
      VecT<T, VEC> od;
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float xf = DT<T>::to_f32(xv.data[j]);
        float dyf = DT<T>::to_f32(dyv.data[j]);
        float wf = DT<T>::to_f32(wv.data[j]);
        float extra = dextra ? DT<T>::to_f32(ev.data[j]) : 0.f;
        const float dxf = inv * wf * dyf - xf * c + extra;
        o.data[j] = DT<T>::from_f32(dxf);
        if (ddelta) {
          float mf = 1.f;
          if (p_drop > 0.f) {
            const long long idx = (long long)row * D + i + j;
            mf = (hash_u01(seed, idx) >= p_drop) ? keep_inv : 0.f;
          }
          od.data[j] = DT<T>::from_f32(dxf * mf);
        }
        dwacc[slot * VEC + j] += dyf * xf * inv;
      }
      vstore<T, VEC>(dxr + i, o);
      if (ddelta) vstore<T, VEC>(ddelta + (long)row * D + i, od);
    }
    __syncthreads();
  }
  int slot = 0;
  float* out = dw_part + (long)blockIdx.x * D;
  for (int i = threadIdx.x * VEC; i < D; i += blockDim.x * VEC, ++slot) {
    VecT<float, VEC> o;
#pragma unroll
    for (int j = 0; j < VEC; ++j) o.data[j] = dwacc[slot * VEC + j];
    vstore<float, VEC>(out + i, o);
  }
}

// column sum of the <=256 per-block partials, emitted directly in the weight
// Single-pass bwd for D <= threads*VEC (T5-base D=768): each thread owns ONE
// vector chunk of the row, so x/dy/w stay in REGISTERS across the dot
// reduction instead of being re-read from HBM — the generic kernel's second
// load pass was the latency bottleneck at B=2 (19 us/call, ROADMAP r1 #5).
// Rows are processed in PAIRS (r, r+rows/2): at the headline shape the grid
// is ~1 wave/SIMD with nothing to hide the load->reduce->store chain, so
// issuing both rows' loads before the first block reduce overlaps the two
// HBM round trips inside one block.
template <typename T, int VEC>
__global__ void rmsnorm_bwd_1p_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                                      const T* __restrict__ w, const float* __restrict__ inv_rms,
                                      const T* __restrict__ dextra,
                                      T* __restrict__ dx, float* __restrict__ dw_part,
                                      int rows, int D,
                                      T* __restrict__ ddelta,
                                      float p_drop, const long* __restrict__ seed_ptr,
                                      long salt) {
  __shared__ float red[16];
  const long long seed = (seed_ptr ? *seed_ptr : 0) ^ salt;
  const float keep_inv = p_drop > 0.f ? 1.f / (1.f - p_drop) : 1.f;
  const int i = threadIdx.x * VEC;
  const bool act = i < D;
  float dwacc[VEC];
#pragma unroll
  for (int j = 0; j < VEC; ++j) dwacc[j] = 0.f;
  VecT<T, VEC> wv;
  if (act) wv = vload<T, VEC>(w + i);

  const int half = (rows + 1) >> 1;
  for (int p = blockIdx.x; p < half; p += gridDim.x) {
    const int r0 = p, r1 = p + half;
    const bool has1 = r1 < rows;
    const float inv0 = inv_rms[r0];
    const float inv1 = has1 ? inv_rms[r1] : 0.f;
    VecT<T, VEC> xv0, dyv0, xv1, dyv1;
    float dot0 = 0.f, dot1 = 0.f;
    if (act) {
      xv0 = vload<T, VEC>(x + (long)r0 * D + i);
      dyv0 = vload<T, VEC>(dy + (long)r0 * D + i);
      if (has1) {
        xv1 = vload<T, VEC>(x + (long)r1 * D + i);
        dyv1 = vload<T, VEC>(dy + (long)r1 * D + i);
      }
#pragma unroll
      for (int j = 0; j < VEC; ++j)
        dot0 += DT<T>::to_f32(dyv0.data[j]) * DT<T>::to_f32(wv.data[j]) *
                DT<T>::to_f32(xv0.data[j]);
      if (has1) {
#pragma unroll
        for (int j = 0; j < VEC; ++j)
          dot1 += DT<T>::to_f32(dyv1.data[j]) * DT<T>::to_f32(wv.data[j]) *
                  DT<T>::to_f32(xv1.data[j]);
      }
    }
    dot0 = block_reduce_sum(dot0, red);
    dot1 = block_reduce_sum(dot1, red);
    if (!act) continue;
#pragma unroll
    for (int half_row = 0; half_row < 2; ++half_row) {
      if (half_row == 1 && !has1) break;
      const int row = half_row ? r1 : r0;
      const float inv = half_row ? inv1 : inv0;
      const float dot = half_row ? dot1 : dot0;
      const VecT<T, VEC>& xv = half_row ? xv1 : xv0;
      const VecT<T, VEC>& dyv = half_row ? dyv1 : dyv0;
      const float c = inv * inv * inv / D * dot;
      VecT<T, VEC> o, od, ev;
      if (dextra) ev = vload<T, VEC>(dextra + (long)row * D + i);
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        const float xf = DT<T>::to_f32(xv.data[j]);
        const float dyf = DT<T>::to_f32(dyv.data[j]);
        const float wf = DT<T>::to_f32(wv.data[j]);
        const float extra = dextra ? DT<T>::to_f32(ev.data[j]) : 0.f;
        const float dxf = inv * wf * dyf - xf * c + extra;
        o.data[j] = DT<T>::from_f32(dxf);
        if (ddelta) {
          float mf = 1.f;
          if (p_drop > 0.f) {
            const long long idx = (long long)row * D + i + j;
            mf = (hash_u01(seed, idx) >= p_drop) ? keep_inv : 0.f;
          }
          od.data[j] = DT<T>::from_f32(dxf * mf);
        }
        dwacc[j] += dyf * xf * inv;
      }
      vstore<T, VEC>(dx + (long)row * D + i, o);
      if (ddelta) vstore<T, VEC>(ddelta + (long)row * D + i, od);
    }
  }
  if (act) {
    VecT<float, VEC> o;
#pragma unroll
    for (int j = 0; j < VEC; ++j) o.data[j] = dwacc[j];
    vstore<float, VEC>(dw_part + (long)blockIdx.x * D + i, o);
  }
}

// dtype (no separate fp32->bf16 cast launch, no zero-fill of dw).
// block = 16 row-lanes x 16 columns so the row loop is 16-way parallel —
// a 1-thread-per-column version was latency-bound at 59 us/call.
template <typename OT, int RLANES>
__global__ void colsum_kernel(const float* __restrict__ part, OT* __restrict__ out,
                              int nparts, int D) {
  constexpr int CLANES = 256 / RLANES;
  __shared__ float red[RLANES][CLANES + 1];
  const int cl = threadIdx.x % CLANES;
  const int rl = threadIdx.x / CLANES;
  const int c = blockIdx.x * CLANES + cl;
  float s = 0.f;
  if (c < D)
    for (int r = rl; r < nparts; r += RLANES) s += part[(long)r * D + c];
  red[rl][cl] = s;
  __syncthreads();
  if (rl == 0 && c < D) {
    float t = 0.f;
#pragma unroll
    for (int i = 0; i < RLANES; ++i) t += red[i][cl];
    out[c] = DT<OT>::from_f32(t);
  }
}

template <typename T>
void launch_fwd(const torch::Tensor& x, const torch::Tensor& w, torch::Tensor& y,
                torch::Tensor& inv, float eps, hipStream_t stream) {
  const int rows = x.size(0), D = x.size(1);
  constexpr int VEC = sizeof(T) == 4 ? 4 : 8;
  TORCH_CHECK(D % VEC == 0, "rmsnorm: D must be a multiple of ", VEC);
  const int threads = (D / VEC >= 256) ? 256 : 128;
  int grid = std::min(rows, 2048);
  hipLaunchKernelGGL((rmsnorm_fwd_kernel<T, VEC>), dim3(grid), dim3(threads), 0, stream,
                     (const T*)x.data_ptr(), (const T*)w.data_ptr(), (T*)y.data_ptr(),
                     inv.data_ptr<float>(), rows, D, eps);
  HIP_CHECK_KERNEL();
}

template <typename T>
void launch_bwd(const torch::Tensor& dy, const torch::Tensor& x, const torch::Tensor& w,
                const torch::Tensor& inv, const torch::Tensor* dextra,
                torch::Tensor& dx, torch::Tensor& dw, torch::Tensor* ddelta,
                float p_drop, const torch::Tensor* seed, long salt,
                hipStream_t stream) {
  const int rows = x.size(0), D = x.size(1);
  constexpr int VEC = sizeof(T) == 4 ? 4 : 8;
  const int threads = (D / VEC >= 256) ? 256 : 128;
  TORCH_CHECK(D <= threads * VEC * 4, "rmsnorm bwd: D too large for dw accumulator");
  const bool onepass = D <= 256 * VEC;
  // big-batch rows want more blocks; small batches keep the partial slab
  // (and its colsum) tiny.  The one-pass kernel is latency-bound at small
  // batch, so give it 2 blocks/CU.
  const int grid = onepass ? std::min((rows + 1) / 2, 512)
                           : (rows <= 2048 ? std::min(rows, 256) : 1024);
  auto part = torch::empty({grid, D}, dw.options().dtype(torch::kFloat));
  if (onepass) {
    const int t1 = std::min(256, ((D + VEC - 1) / VEC + 63) / 64 * 64);
    hipLaunchKernelGGL((rmsnorm_bwd_1p_kernel<T, VEC>), dim3(grid), dim3(t1), 0, stream,
                       (const T*)dy.data_ptr(), (const T*)x.data_ptr(), (const T*)w.data_ptr(),
                       inv.data_ptr<float>(),
                       dextra ? (const T*)dextra->data_ptr() : nullptr,
                       (T*)dx.data_ptr(), part.data_ptr<float>(), rows, D,
                       ddelta ? (T*)ddelta->data_ptr() : nullptr, p_drop,
                       seed ? seed->data_ptr<long>() : nullptr, salt);
  } else {
    hipLaunchKernelGGL((rmsnorm_bwd_kernel<T, VEC, 4>), dim3(grid), dim3(threads), 0, stream,
                       (const T*)dy.data_ptr(), (const T*)x.data_ptr(), (const T*)w.data_ptr(),
                       inv.data_ptr<float>(),
                       dextra ? (const T*)dextra->data_ptr() : nullptr,
                       (T*)dx.data_ptr(), part.data_ptr<float>(), rows, D,
                       ddelta ? (T*)ddelta->data_ptr() : nullptr, p_drop,
                       seed ? seed->data_ptr<long>() : nullptr, salt);
  }
  HIP_CHECK_KERNEL();
  if (grid > 256)
    hipLaunchKernelGGL((colsum_kernel<T, 64>), dim3((D + 3) / 4), dim3(256), 0,
                       stream, part.data_ptr<float>(), (T*)dw.data_ptr(), grid, D);
  else
    hipLaunchKernelGGL((colsum_kernel<T, 16>), dim3((D + 15) / 16), dim3(256), 0,
                       stream, part.data_ptr<float>(), (T*)dw.data_ptr(), grid, D);
  HIP_CHECK_KERNEL();
}

}  // namespace

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w, double eps) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous());
  auto y = torch::empty_like(x);
  auto inv = torch::empty({x.size(0)}, x.options().dtype(torch::kFloat));
  auto stream = at::hip::getCurrentHIPStream().stream();
  if (x.scalar_type() == torch::kBFloat16)
    launch_fwd<__hip_bfloat16>(x, w, y, inv, (float)eps, stream);
  else if (x.scalar_type() == torch::kHalf)
    launch_fwd<__half>(x, w, y, inv, (float)eps, stream);
  else if (x.scalar_type() == torch::kFloat)
    launch_fwd<float>(x, w, y, inv, (float)eps, stream);
  else
    TORCH_CHECK(false, "rmsnorm: unsupported dtype");
  return {y, inv};
}

std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x, torch::Tensor w,
                                       torch::Tensor inv,
                                       c10::optional<torch::Tensor> dextra,
                                       double p_drop, c10::optional<torch::Tensor> seed,
                                       long salt) {
  auto dx = torch::empty_like(x);
  TORCH_CHECK(w.scalar_type() == x.scalar_type(), "rmsnorm bwd: w/x dtype mismatch");
  auto dw = torch::empty({x.size(1)}, w.options());
  auto stream = at::hip::getCurrentHIPStream().stream();
  const torch::Tensor* de = (dextra.has_value() && dextra->defined()) ? &*dextra : nullptr;
  const torch::Tensor* sp = (seed.has_value() && seed->defined()) ? &*seed : nullptr;
  torch::Tensor ddelta;
  torch::Tensor* dd = nullptr;
  if (p_drop > 0.0) {
    ddelta = torch::empty_like(x);
    dd = &ddelta;
  }
  if (x.scalar_type() == torch::kBFloat16)
    launch_bwd<__hip_bfloat16>(dy, x, w, inv, de, dx, dw, dd, (float)p_drop, sp, salt, stream);
  else if (x.scalar_type() == torch::kHalf)
    launch_bwd<__half>(dy, x, w, inv, de, dx, dw, dd, (float)p_drop, sp, salt, stream);
  else if (x.scalar_type() == torch::kFloat)
    launch_bwd<float>(dy, x, w, inv, de, dx, dw, dd, (float)p_drop, sp, salt, stream);
  else
    TORCH_CHECK(false, "rmsnorm: unsupported dtype");
  if (dd) return {dx, dw, ddelta};
  return {dx, dw};
}

std::vector<torch::Tensor> add_rmsnorm_fwd(torch::Tensor x, torch::Tensor r,
                                           torch::Tensor w, double eps,
                                           double p_drop, c10::optional<torch::Tensor> seed,
                                           long salt) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous() && r.is_contiguous());
  auto s_out = torch::empty_like(x);
  auto y = torch::empty_like(x);
  auto inv = torch::empty({x.size(0)}, x.options().dtype(torch::kFloat));
  auto stream = at::hip::getCurrentHIPStream().stream();
  const int rows = x.size(0), D = x.size(1);
#define ACASE(T, VEC)                                                                  \
  do {                                                                                 \
    const long* sp = (seed.has_value() && seed->defined())                             \
                         ? seed->data_ptr<long>() : nullptr;                           \
    if (D <= 256 * VEC) {                                                              \
      const int t1 = std::min(256, ((D + VEC - 1) / VEC + 63) / 64 * 64);              \
      hipLaunchKernelGGL((add_rmsnorm_fwd_1p_kernel<T, VEC>),                          \
                         dim3(std::min((rows + 1) / 2, 1024)), dim3(t1), 0, stream,    \
                         (const T*)x.data_ptr(), (const T*)r.data_ptr(),               \
                         (T*)s_out.data_ptr(), (T*)y.data_ptr(),                       \
                         (const T*)w.data_ptr(), inv.data_ptr<float>(), rows, D,       \
                         (float)eps, (float)p_drop, sp, salt);                         \
    } else {                                                                           \
      hipLaunchKernelGGL((add_rmsnorm_fwd_kernel<T, VEC>),                             \
                         dim3(std::min(rows, 2048)), dim3((D / VEC >= 256) ? 256 : 128),\
                         0, stream, (const T*)x.data_ptr(), (const T*)r.data_ptr(),    \
                         (T*)s_out.data_ptr(), (T*)y.data_ptr(), (const T*)w.data_ptr(),\
                         inv.data_ptr<float>(), rows, D, (float)eps, (float)p_drop,    \
                         sp, salt);                                                    \
    }                                                                                  \
  } while (0)
  if (x.scalar_type() == torch::kBFloat16) ACASE(__hip_bfloat16, 8);
  else if (x.scalar_type() == torch::kHalf) ACASE(__half, 8);
  else if (x.scalar_type() == torch::kFloat) ACASE(float, 4);
  else TORCH_CHECK(false, "add_rmsnorm: unsupported dtype");
#undef ACASE
  HIP_CHECK_KERNEL();
  return {s_out, y, inv};
}
