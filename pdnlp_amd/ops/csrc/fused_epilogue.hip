// Fused bias + dropout + residual + LayerNorm forward/backward
// (SURVEY.md K6/K8 epilogues + K16 dropout).
//
//   out = LN(dropout(y + bias) + residual)
//
// One wavefront per row. Dropout mask from a counter-based hash RNG
// (deterministic in (seed, element index), saved as u8 for backward —
// SURVEY.md K16). The pre-LN sum is saved (bf16) so backward avoids
// recomputing the dropout path. Five vendor-kernel passes in the reference
// (bias add, dropout, residual add, LN) collapse to one HBM round trip.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

// seed is read from DEVICE memory (seed_base) + a per-call-site salt so the
// dropout mask changes across hipGraph replays (the host updates *seed_base
// between replays; a kernel-arg seed would be frozen into the graph).
template <typename T, bool DROP>
__global__ void bdrl_fwd_kernel(const T* __restrict__ y,
                                const T* __restrict__ bias,
                                const T* __restrict__ res,
                                const T* __restrict__ lnw,
                                const T* __restrict__ lnb,
                                T* __restrict__ out, T* __restrict__ xsum,
                                unsigned char* __restrict__ mask_out,
                                float* __restrict__ mean_out,
                                float* __restrict__ rstd_out, int H, float p,
                                float eps,
                                const unsigned long long* __restrict__ seed_base,
                                unsigned long long salt) {
  const unsigned long long seed = (DROP ? *seed_base : 0ull) + salt;
  const long row = blockIdx.x;
  const int lane = threadIdx.x & (WAVE - 1);
  const T* yr = y + row * H;
  const T* rr = res + row * H;
  T* xr = xsum + row * H;
  T* outr = out + row * H;
  const float keep = 1.f - p;
  const float inv_keep = 1.f / keep;

  float sum = 0.f, sumsq = 0.f;
  for (int c = lane; c < H; c += WAVE) {
    float h = to_f32<T>(yr[c]) + to_f32<T>(bias[c]);
    if (DROP) {
      const unsigned int r = hash_rng(seed, row * (unsigned long long)H + c);
      const bool live = (r * 2.3283064365386963e-10f) >= p;
      mask_out[row * H + c] = live;
      h = live ? h * inv_keep : 0.f;
    }
    h += to_f32<T>(rr[c]);
    xr[c] = from_f32<T>(h);
    // recompute from the rounded value so stats match saved xsum exactly
    const float hs = to_f32<T>(xr[c]);
    sum += hs;
    sumsq += hs * hs;
  }
  sum = wave_sum(sum);
  sumsq = wave_sum(sumsq);
  const float mean = sum / H;
  const float rstd = rsqrtf(sumsq / H - mean * mean + eps);
  if (lane == 0) {
    mean_out[row] = mean;
    rstd_out[row] = rstd;
  }
  for (int c = lane; c < H; c += WAVE) {
    const float xh = (to_f32<T>(xr[c]) - mean) * rstd;
    outr[c] = from_f32<T>(xh * to_f32<T>(lnw[c]) + to_f32<T>(lnb[c]));
  }
}

template <typename T, bool DROP>
__global__ void bdrl_bwd_dx_kernel(const T* __restrict__ dout,
                                   const T* __restrict__ xsum,
                                   const unsigned char* __restrict__ mask,
                                   const T* __restrict__ lnw,
                                   const float* __restrict__ mean,
                                   const float* __restrict__ rstd,
                                   T* __restrict__ dy, T* __restrict__ dres,
                                   int H, float p) {
  const long row = blockIdx.x;
  const int lane = threadIdx.x & (WAVE - 1);
  const T* dor = dout + row * H;
  const T* xr = xsum + row * H;
  T* dyr = dy + row * H;
  T* drr = dres + row * H;
  const float mu = mean[row], rs = rstd[row];
  const float inv_keep = 1.f / (1.f - p);

  float s1 = 0.f, s2 = 0.f;
  for (int c = lane; c < H; c += WAVE) {
    const float dw = to_f32<T>(dor[c]) * to_f32<T>(lnw[c]);
    const float xh = (to_f32<T>(xr[c]) - mu) * rs;
    s1 += dw;
    s2 += dw * xh;
  }
  s1 = wave_sum(s1) / H;
  s2 = wave_sum(s2) / H;
  for (int c = lane; c < H; c += WAVE) {
    const float dw = to_f32<T>(dor[c]) * to_f32<T>(lnw[c]);
    const float xh = (to_f32<T>(xr[c]) - mu) * rs;
    const float dxs = rs * (dw - s1 - xh * s2);
    drr[c] = from_f32<T>(dxs);
    float g = dxs;
    if (DROP) g = mask[row * H + c] ? g * inv_keep : 0.f;
    dyr[c] = from_f32<T>(g);
  }
}

// also folds the projection-bias gradient (column sum of dy, the
// post-dropout grad produced by the dx kernel) into the same pass — saves a
// separate torch reduce launch per call.
template <typename T>
__global__ void bdrl_bwd_dwdb_kernel(const T* __restrict__ dout,
                                     const T* __restrict__ xsum,
                                     const T* __restrict__ dy,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ rstd,
                                     float* __restrict__ dw32,
                                     float* __restrict__ db32,
                                     float* __restrict__ dbias32, long R,
                                     int H, long rows_per_chunk) {
  const int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= H) return;
  const long r0 = blockIdx.y * rows_per_chunk;
  const long r1 = min(r0 + rows_per_chunk, R);
  float dw = 0.f, db = 0.f, dbias = 0.f;
  for (long r = r0; r < r1; ++r) {
    const float d = to_f32<T>(dout[r * H + col]);
    const float xh = (to_f32<T>(xsum[r * H + col]) - mean[r]) * rstd[r];
    dw += d * xh;
    db += d;
    dbias += to_f32<T>(dy[r * H + col]);
  }
  atomicAdd(dw32 + col, dw);
  atomicAdd(db32 + col, db);
  atomicAdd(dbias32 + col, dbias);
}

}  // namespace

std::vector<torch::Tensor> bias_dropout_residual_ln_fwd(
    torch::Tensor y, torch::Tensor bias, torch::Tensor res, torch::Tensor lnw,
    torch::Tensor lnb, double p, double eps, torch::Tensor seed_buf,
    long salt) {
  const int H = y.size(-1);
  const long R = y.numel() / H;
  auto out = torch::empty_like(y);
  auto xsum = torch::empty_like(y);
  const bool drop = p > 0.0;
  TORCH_CHECK(!drop || (seed_buf.defined() && seed_buf.numel() >= 1 &&
                        seed_buf.is_cuda() &&
                        seed_buf.scalar_type() == torch::kLong),
              "dropout needs a cuda int64 seed buffer");
  auto mask = drop
      ? torch::empty({R, (long)H}, y.options().dtype(torch::kUInt8))
      : torch::empty({0}, y.options().dtype(torch::kUInt8));
  auto mean = torch::empty({R}, y.options().dtype(torch::kFloat32));
  auto rstd = torch::empty({R}, y.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_FLOAT_TYPES(y.scalar_type(), "bdrl_fwd", [&] {
    if (drop) {
      hipLaunchKernelGGL((bdrl_fwd_kernel<scalar_t, true>), dim3(R),
                         dim3(WAVE), 0, stream,
                         (const scalar_t*)y.data_ptr(),
                         (const scalar_t*)bias.data_ptr(),
                         (const scalar_t*)res.data_ptr(),
                         (const scalar_t*)lnw.data_ptr(),
                         (const scalar_t*)lnb.data_ptr(),
                         (scalar_t*)out.data_ptr(),
                         (scalar_t*)xsum.data_ptr(),
                         mask.data_ptr<unsigned char>(),
                         mean.data_ptr<float>(), rstd.data_ptr<float>(), H,
                         (float)p, (float)eps,
                         (const unsigned long long*)seed_buf.data_ptr(),
                         (unsigned long long)salt);
    } else {
      hipLaunchKernelGGL((bdrl_fwd_kernel<scalar_t, false>), dim3(R),
                         dim3(WAVE), 0, stream,
                         (const scalar_t*)y.data_ptr(),
                         (const scalar_t*)bias.data_ptr(),
                         (const scalar_t*)res.data_ptr(),
                         (const scalar_t*)lnw.data_ptr(),
                         (const scalar_t*)lnb.data_ptr(),
                         (scalar_t*)out.data_ptr(),
                         (scalar_t*)xsum.data_ptr(), nullptr,
                         mean.data_ptr<float>(), rstd.data_ptr<float>(), H,
                         (float)p, (float)eps, nullptr, 0ull);
    }
  });
  return {out, xsum, mask, mean, rstd};
}

std::vector<torch::Tensor> bias_dropout_residual_ln_bwd(
    torch::Tensor dout, torch::Tensor xsum, torch::Tensor mask,
    torch::Tensor lnw, torch::Tensor mean, torch::Tensor rstd, double p) {
  const int H = xsum.size(-1);
  const long R = xsum.numel() / H;
  auto dy = torch::empty_like(xsum);
  auto dres = torch::empty_like(xsum);
  auto dw32 = torch::zeros({H}, xsum.options().dtype(torch::kFloat32));
  auto db32 = torch::zeros({H}, xsum.options().dtype(torch::kFloat32));
  auto dbias32 = torch::zeros({H}, xsum.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  const bool drop = p > 0.0 && mask.numel() > 0;
  const long rows_per_chunk = 16;
  const long chunks = (R + rows_per_chunk - 1) / rows_per_chunk;
  DISPATCH_FLOAT_TYPES(xsum.scalar_type(), "bdrl_bwd", [&] {
    if (drop) {
      hipLaunchKernelGGL((bdrl_bwd_dx_kernel<scalar_t, true>), dim3(R),
                         dim3(WAVE), 0, stream,
                         (const scalar_t*)dout.data_ptr(),
                         (const scalar_t*)xsum.data_ptr(),
                         mask.data_ptr<unsigned char>(),
                         (const scalar_t*)lnw.data_ptr(),
                         mean.data_ptr<float>(), rstd.data_ptr<float>(),
                         (scalar_t*)dy.data_ptr(), (scalar_t*)dres.data_ptr(),
                         H, (float)p);
    } else {
      hipLaunchKernelGGL((bdrl_bwd_dx_kernel<scalar_t, false>), dim3(R),
                         dim3(WAVE), 0, stream,
                         (const scalar_t*)dout.data_ptr(),
                         (const scalar_t*)xsum.data_ptr(), nullptr,
                         (const scalar_t*)lnw.data_ptr(),
                         mean.data_ptr<float>(), rstd.data_ptr<float>(),
                         (scalar_t*)dy.data_ptr(), (scalar_t*)dres.data_ptr(),
                         H, 0.f);
    }
    dim3 grid((H + 255) / 256, chunks);
    hipLaunchKernelGGL((bdrl_bwd_dwdb_kernel<scalar_t>), grid, dim3(256), 0,
                       stream,
                       (const scalar_t*)dout.data_ptr(),
                       (const scalar_t*)xsum.data_ptr(),
                       (const scalar_t*)dy.data_ptr(),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       dw32.data_ptr<float>(), db32.data_ptr<float>(),
                       dbias32.data_ptr<float>(), R, H, rows_per_chunk);
  });
  auto dt = lnw.scalar_type();
  return {dy, dbias32.to(dt), dres, dw32.to(dt), db32.to(dt)};
}
