// LayerNorm forward/backward (SURVEY.md K1 partner kernel).
//
// Layout: x [R, H] row-major, H in {64..4096}. One 64-lane wavefront per row:
// vectorized 4-element (8 B) bf16/fp16 loads when H % 256 == 0, scalar
// fallback otherwise; fp32 statistics via 64-wide shuffle reduction
// (fp32-stats requirement: SURVEY.md §7 hard part 5).
//
// Backward splits into (a) wave-per-row dx kernel and (b) a column-parallel
// dw/db reduction over row chunks accumulated into fp32 scratch with one
// atomicAdd per (chunk, column) — 8-32 atomics per column instead of R.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

// ---------------- forward ----------------
template <typename T, int VEC>
__global__ void ln_fwd_kernel(const T* __restrict__ x, const T* __restrict__ w,
                              const T* __restrict__ b, T* __restrict__ y,
                              float* __restrict__ mean_out,
                              float* __restrict__ rstd_out,
                              int H, float eps) {
  const int row = blockIdx.x;
  const int lane = threadIdx.x & (WAVE - 1);
  const T* xr = x + (long)row * H;
  T* yr = y + (long)row * H;

  float sum = 0.f, sumsq = 0.f;
  if (VEC == 4) {
    for (int base = lane * 4; base < H; base += WAVE * 4) {
      const short4 raw = *reinterpret_cast<const short4*>(xr + base);
      const T* px = reinterpret_cast<const T*>(&raw);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float v = to_f32<T>(px[j]);
        sum += v;
        sumsq += v * v;
      }
    }
  } else {
    for (int c = lane; c < H; c += WAVE) {
      float v = to_f32<T>(xr[c]);
      sum += v;
      sumsq += v * v;
    }
  }
  sum = wave_sum(sum);
  sumsq = wave_sum(sumsq);
  const float mean = sum / H;
  const float var = sumsq / H - mean * mean;
  const float rstd = rsqrtf(var + eps);
  if (lane == 0) {
    mean_out[row] = mean;
    rstd_out[row] = rstd;
  }
  if (VEC == 4) {
    for (int base = lane * 4; base < H; base += WAVE * 4) {
      const short4 rx = *reinterpret_cast<const short4*>(xr + base);
      const short4 rw = *reinterpret_cast<const short4*>(w + base);
      const short4 rb = *reinterpret_cast<const short4*>(b + base);
      const T* px = reinterpret_cast<const T*>(&rx);
      const T* pw = reinterpret_cast<const T*>(&rw);
      const T* pb = reinterpret_cast<const T*>(&rb);
      short4 ry;
      T* py = reinterpret_cast<T*>(&ry);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float v = (to_f32<T>(px[j]) - mean) * rstd;
        py[j] = from_f32<T>(v * to_f32<T>(pw[j]) + to_f32<T>(pb[j]));
      }
      *reinterpret_cast<short4*>(yr + base) = ry;
    }
  } else {
    for (int c = lane; c < H; c += WAVE) {
      float v = (to_f32<T>(xr[c]) - mean) * rstd;
      yr[c] = from_f32<T>(v * to_f32<T>(w[c]) + to_f32<T>(b[c]));
    }
  }
}

// ---------------- backward: dx ----------------
template <typename T>
__global__ void ln_bwd_dx_kernel(const T* __restrict__ dy,
                                 const T* __restrict__ x,
                                 const T* __restrict__ w,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ rstd,
                                 T* __restrict__ dx, int H) {
  const int row = blockIdx.x;
  const int lane = threadIdx.x & (WAVE - 1);
  const T* dyr = dy + (long)row * H;
  const T* xr = x + (long)row * H;
  T* dxr = dx + (long)row * H;
  const float mu = mean[row], rs = rstd[row];

  float s1 = 0.f, s2 = 0.f;
  for (int c = lane; c < H; c += WAVE) {
    const float dyw = to_f32<T>(dyr[c]) * to_f32<T>(w[c]);
    const float xh = (to_f32<T>(xr[c]) - mu) * rs;
    s1 += dyw;
    s2 += dyw * xh;
  }
  s1 = wave_sum(s1) / H;
  s2 = wave_sum(s2) / H;
  for (int c = lane; c < H; c += WAVE) {
    const float dyw = to_f32<T>(dyr[c]) * to_f32<T>(w[c]);
    const float xh = (to_f32<T>(xr[c]) - mu) * rs;
    dxr[c] = from_f32<T>(rs * (dyw - s1 - xh * s2));
  }
}

// ---------------- backward: dw/db column reduce ----------------
template <typename T>
__global__ void ln_bwd_dwdb_kernel(const T* __restrict__ dy,
                                   const T* __restrict__ x,
                                   const float* __restrict__ mean,
                                   const float* __restrict__ rstd,
                                   float* __restrict__ dw32,
                                   float* __restrict__ db32, int R, int H,
                                   int rows_per_chunk) {
  const int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= H) return;
  const int r0 = blockIdx.y * rows_per_chunk;
  const int r1 = min(r0 + rows_per_chunk, R);
  float dw = 0.f, db = 0.f;
  for (int r = r0; r < r1; ++r) {
    const float d = to_f32<T>(dy[(long)r * H + col]);
    const float xh = (to_f32<T>(x[(long)r * H + col]) - mean[r]) * rstd[r];
    dw += d * xh;
    db += d;
  }
  atomicAdd(dw32 + col, dw);
  atomicAdd(db32 + col, db);
}

}  // namespace

std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor b, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  const int H = x.size(-1);
  const long R = x.numel() / H;
  auto y = torch::empty_like(x);
  auto mean = torch::empty({R}, x.options().dtype(torch::kFloat32));
  auto rstd = torch::empty({R}, x.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_FLOAT_TYPES(x.scalar_type(), "layernorm_fwd", [&] {
    const bool vec = !std::is_same<scalar_t, float>::value && (H % 256 == 0);
    if (vec) {
      hipLaunchKernelGGL((ln_fwd_kernel<scalar_t, 4>), dim3(R), dim3(WAVE), 0,
                         stream,
                         (const scalar_t*)x.data_ptr(),
                         (const scalar_t*)w.data_ptr(),
                         (const scalar_t*)b.data_ptr(),
                         (scalar_t*)y.data_ptr(), mean.data_ptr<float>(),
                         rstd.data_ptr<float>(), H, (float)eps);
    } else {
      hipLaunchKernelGGL((ln_fwd_kernel<scalar_t, 1>), dim3(R), dim3(WAVE), 0,
                         stream,
                         (const scalar_t*)x.data_ptr(),
                         (const scalar_t*)w.data_ptr(),
                         (const scalar_t*)b.data_ptr(),
                         (scalar_t*)y.data_ptr(), mean.data_ptr<float>(),
                         rstd.data_ptr<float>(), H, (float)eps);
    }
  });
  return {y, mean, rstd};
}

std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor w, torch::Tensor mean,
                                         torch::Tensor rstd) {
  const int H = x.size(-1);
  const long R = x.numel() / H;
  auto dx = torch::empty_like(x);
  auto dw32 = torch::zeros({H}, x.options().dtype(torch::kFloat32));
  auto db32 = torch::zeros({H}, x.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  // grid-fill: ~16 rows per chunk so (H/256)*chunks blocks cover 256 CUs
  const int rows_per_chunk = 16;
  const int chunks = (int)((R + rows_per_chunk - 1) / rows_per_chunk);
  DISPATCH_FLOAT_TYPES(x.scalar_type(), "layernorm_bwd", [&] {
    hipLaunchKernelGGL((ln_bwd_dx_kernel<scalar_t>), dim3(R), dim3(WAVE), 0,
                       stream,
                       (const scalar_t*)dy.data_ptr(),
                       (const scalar_t*)x.data_ptr(),
                       (const scalar_t*)w.data_ptr(), mean.data_ptr<float>(),
                       rstd.data_ptr<float>(), (scalar_t*)dx.data_ptr(), H);
    dim3 grid((H + 255) / 256, chunks);
    hipLaunchKernelGGL((ln_bwd_dwdb_kernel<scalar_t>), grid, dim3(256), 0,
                       stream,
                       (const scalar_t*)dy.data_ptr(),
                       (const scalar_t*)x.data_ptr(), mean.data_ptr<float>(),
                       rstd.data_ptr<float>(), dw32.data_ptr<float>(),
                       db32.data_ptr<float>(), (int)R, H, rows_per_chunk);
  });
  return {dx, dw32.to(w.scalar_type()), db32.to(w.scalar_type())};
}
