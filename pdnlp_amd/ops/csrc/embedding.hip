// Fused embedding gather + add + LayerNorm and its backward
// (embedding scatter-add, SURVEY.md K11 note).
//
// Forward: one wavefront per token row; gathers word/position/token-type
// rows, sums in fp32, LayerNorm with fp32 stats — one kernel instead of the
// reference's three embedding gathers + add + LN (vendor kernels inside HF
// BERT, reference call site multi-gpu-distributed-cls.py:132-136).
//
// Backward: (a) wave-per-row kernel recomputes the summed embedding (gather
// is cheaper than saving [R,H]), applies the LN-input gradient formula and
// scatter-adds fp32 into per-table scratch (native fp32 atomics; vocab-sized
// scratch is nothing against 288 GB HBM3E); (b) column-parallel dLNw/dLNb
// reduce; (c) scratch is cast to the table dtype by the binding.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

template <typename T>
__global__ void emb_ln_fwd_kernel(
    const long* __restrict__ ids, const long* __restrict__ type_ids,
    const long* __restrict__ pos_ids, const T* __restrict__ word,
    const T* __restrict__ pos, const T* __restrict__ type_,
    const T* __restrict__ w, const T* __restrict__ b, T* __restrict__ y,
    float* __restrict__ mean_out, float* __restrict__ rstd_out, int H,
    float eps) {
  const int row = blockIdx.x;
  const int lane = threadIdx.x & (WAVE - 1);
  const T* wr = word + (long)ids[row] * H;
  const T* pr = pos + (long)pos_ids[row] * H;
  const T* tr = type_ + (long)type_ids[row] * H;
  T* yr = y + (long)row * H;

  float sum = 0.f, sumsq = 0.f;
  for (int c = lane; c < H; c += WAVE) {
    const float v = to_f32<T>(wr[c]) + to_f32<T>(pr[c]) + to_f32<T>(tr[c]);
    sum += v;
    sumsq += v * v;
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
    const float v = to_f32<T>(wr[c]) + to_f32<T>(pr[c]) + to_f32<T>(tr[c]);
    yr[c] = from_f32<T>((v - mean) * rstd * to_f32<T>(w[c]) + to_f32<T>(b[c]));
  }
}

template <typename T>
__global__ void emb_ln_bwd_scatter_kernel(
    const T* __restrict__ dy, const long* __restrict__ ids,
    const long* __restrict__ type_ids, const long* __restrict__ pos_ids,
    const T* __restrict__ word, const T* __restrict__ pos,
    const T* __restrict__ type_, const T* __restrict__ w,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    float* __restrict__ dword32, T* __restrict__ dsum_ws, int H) {
  // One wave per token row. The LN-input gradient dsum is the gradient of
  // ALL THREE tables at this row's indices; only the word table gets
  // scatter-atomics here (random ids => low contention). Position/type
  // rows are shared by many tokens (token_type 0 is shared by EVERY row:
  // a 4096-way atomic pileup measured at ~190 us) — those tables are
  // reduced from the dsum workspace by the column-parallel kernels below.
  const int row = blockIdx.x;
  const int lane = threadIdx.x & (WAVE - 1);
  const T* dyr = dy + (long)row * H;
  const T* wr = word + (long)ids[row] * H;
  const T* pr = pos + (long)pos_ids[row] * H;
  const T* tr = type_ + (long)type_ids[row] * H;
  const float mu = mean[row], rs = rstd[row];

  float s1 = 0.f, s2 = 0.f;
  for (int c = lane; c < H; c += WAVE) {
    const float dyw = to_f32<T>(dyr[c]) * to_f32<T>(w[c]);
    const float x = to_f32<T>(wr[c]) + to_f32<T>(pr[c]) + to_f32<T>(tr[c]);
    const float xh = (x - mu) * rs;
    s1 += dyw;
    s2 += dyw * xh;
  }
  s1 = wave_sum(s1) / H;
  s2 = wave_sum(s2) / H;
  float* dwr = dword32 + (long)ids[row] * H;
  for (int c = lane; c < H; c += WAVE) {
    const float dyw = to_f32<T>(dyr[c]) * to_f32<T>(w[c]);
    const float x = to_f32<T>(wr[c]) + to_f32<T>(pr[c]) + to_f32<T>(tr[c]);
    const float xh = (x - mu) * rs;
    const float dsum = rs * (dyw - s1 - xh * s2);
    atomicAdd(dwr + c, dsum);
    dsum_ws[(long)row * H + c] = from_f32<T>(dsum);
  }
}

// dtab[v][c] = sum over rows with index v of dsum_ws[row][c], for a SMALL
// table (position S<=512 rows, token-type 2 rows). Column-chunked: thread
// owns one column over a row chunk, register accumulator per table row is
// impossible generically, so accumulate per (row-index) via atomics into
// the table — but now each chunk contributes ONE atomic per touched index
// per column instead of one per token row.
template <typename T>
__global__ void emb_ln_bwd_small_table_kernel(
    const T* __restrict__ dsum_ws, const long* __restrict__ idx,
    float* __restrict__ dtab32, int R, int H, int rows_per_chunk,
    int tab_rows) {
  const int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= H) return;
  const int r0 = blockIdx.y * rows_per_chunk;
  const int r1 = min(r0 + rows_per_chunk, R);
  long cur = idx[r0];
  float acc = 0.f;
  for (int r = r0; r < r1; ++r) {
    const long v = idx[r];
    if (v != cur) {
      atomicAdd(dtab32 + cur * H + col, acc);
      cur = v;
      acc = 0.f;
    }
    acc += to_f32<T>(dsum_ws[(long)r * H + col]);
  }
  atomicAdd(dtab32 + cur * H + col, acc);
}

template <typename T>
__global__ void emb_ln_bwd_dwdb_kernel(
    const T* __restrict__ dy, const long* __restrict__ ids,
    const long* __restrict__ type_ids, const long* __restrict__ pos_ids,
    const T* __restrict__ word, const T* __restrict__ pos,
    const T* __restrict__ type_, const float* __restrict__ mean,
    const float* __restrict__ rstd, float* __restrict__ dlnw32,
    float* __restrict__ dlnb32, int R, int H, int rows_per_chunk) {
  const int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= H) return;
  const int r0 = blockIdx.y * rows_per_chunk;
  const int r1 = min(r0 + rows_per_chunk, R);
  float dw = 0.f, db = 0.f;
  for (int r = r0; r < r1; ++r) {
    const float d = to_f32<T>(dy[(long)r * H + col]);
    const float x = to_f32<T>(word[(long)ids[r] * H + col]) +
                    to_f32<T>(pos[(long)pos_ids[r] * H + col]) +
                    to_f32<T>(type_[(long)type_ids[r] * H + col]);
    const float xh = (x - mean[r]) * rstd[r];
    dw += d * xh;
    db += d;
  }
  atomicAdd(dlnw32 + col, dw);
  atomicAdd(dlnb32 + col, db);
}

}  // namespace

std::vector<torch::Tensor> embedding_ln_fwd(
    torch::Tensor ids, torch::Tensor type_ids, torch::Tensor pos_ids,
    torch::Tensor word, torch::Tensor pos, torch::Tensor type_,
    torch::Tensor w, torch::Tensor b, double eps) {
  TORCH_CHECK(ids.is_cuda() && ids.scalar_type() == torch::kLong);
  const int H = word.size(1);
  const long R = ids.numel();
  auto y = torch::empty({ids.size(0), ids.size(1), H}, word.options());
  auto mean = torch::empty({R}, word.options().dtype(torch::kFloat32));
  auto rstd = torch::empty({R}, word.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_FLOAT_TYPES(word.scalar_type(), "embedding_ln_fwd", [&] {
    hipLaunchKernelGGL((emb_ln_fwd_kernel<scalar_t>), dim3(R), dim3(WAVE), 0,
                       stream,
                       ids.data_ptr<long>(), type_ids.data_ptr<long>(),
                       pos_ids.data_ptr<long>(),
                       (const scalar_t*)word.data_ptr(),
                       (const scalar_t*)pos.data_ptr(),
                       (const scalar_t*)type_.data_ptr(),
                       (const scalar_t*)w.data_ptr(),
                       (const scalar_t*)b.data_ptr(), (scalar_t*)y.data_ptr(),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(), H,
                       (float)eps);
  });
  return {y, mean, rstd};
}

std::vector<torch::Tensor> embedding_ln_bwd(
    torch::Tensor dy, torch::Tensor ids, torch::Tensor type_ids,
    torch::Tensor pos_ids, torch::Tensor word, torch::Tensor pos,
    torch::Tensor type_, torch::Tensor w, torch::Tensor mean,
    torch::Tensor rstd) {
  const int H = word.size(1);
  const long R = ids.numel();
  auto opts32 = word.options().dtype(torch::kFloat32);
  auto dword32 = torch::zeros({word.size(0), H}, opts32);
  auto dpos32 = torch::zeros({pos.size(0), H}, opts32);
  auto dtype32 = torch::zeros({type_.size(0), H}, opts32);
  auto dlnw32 = torch::zeros({H}, opts32);
  auto dlnb32 = torch::zeros({H}, opts32);
  auto dsum_ws = torch::empty({R, (long)H}, word.options());
  auto stream = at::hip::getCurrentHIPStream();
  const int rows_per_chunk = 16;
  const int chunks = (int)((R + rows_per_chunk - 1) / rows_per_chunk);
  DISPATCH_FLOAT_TYPES(word.scalar_type(), "embedding_ln_bwd", [&] {
    hipLaunchKernelGGL((emb_ln_bwd_scatter_kernel<scalar_t>), dim3(R),
                       dim3(WAVE), 0, stream,
                       (const scalar_t*)dy.data_ptr(), ids.data_ptr<long>(),
                       type_ids.data_ptr<long>(), pos_ids.data_ptr<long>(),
                       (const scalar_t*)word.data_ptr(),
                       (const scalar_t*)pos.data_ptr(),
                       (const scalar_t*)type_.data_ptr(),
                       (const scalar_t*)w.data_ptr(), mean.data_ptr<float>(),
                       rstd.data_ptr<float>(), dword32.data_ptr<float>(),
                       (scalar_t*)dsum_ws.data_ptr(), H);
    dim3 gridc((H + 255) / 256, chunks);
    hipLaunchKernelGGL((emb_ln_bwd_small_table_kernel<scalar_t>), gridc,
                       dim3(256), 0, stream,
                       (const scalar_t*)dsum_ws.data_ptr(),
                       pos_ids.data_ptr<long>(), dpos32.data_ptr<float>(),
                       (int)R, H, rows_per_chunk, (int)pos.size(0));
    hipLaunchKernelGGL((emb_ln_bwd_small_table_kernel<scalar_t>), gridc,
                       dim3(256), 0, stream,
                       (const scalar_t*)dsum_ws.data_ptr(),
                       type_ids.data_ptr<long>(), dtype32.data_ptr<float>(),
                       (int)R, H, rows_per_chunk, (int)type_.size(0));
    hipLaunchKernelGGL((emb_ln_bwd_dwdb_kernel<scalar_t>), gridc, dim3(256),
                       0, stream,
                       (const scalar_t*)dy.data_ptr(), ids.data_ptr<long>(),
                       type_ids.data_ptr<long>(), pos_ids.data_ptr<long>(),
                       (const scalar_t*)word.data_ptr(),
                       (const scalar_t*)pos.data_ptr(),
                       (const scalar_t*)type_.data_ptr(),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       dlnw32.data_ptr<float>(), dlnb32.data_ptr<float>(),
                       (int)R, H, rows_per_chunk);
  });
  auto dt = word.scalar_type();
  return {dword32.to(dt), dpos32.to(dt), dtype32.to(dt), dlnw32.to(dt),
          dlnb32.to(dt)};
}
