// Fused cross-entropy forward/backward: log-softmax + NLL
// on [B, C] logits (C = 6 for the reference task), one wavefront per row,
// fp32 throughout. Forward returns per-row losses (binding takes the mean)
// and saves log-probs for a one-pass backward.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

__global__ void ce_fwd_kernel(const float* __restrict__ logits,
                              const long* __restrict__ labels,
                              float* __restrict__ losses,
                              float* __restrict__ logprobs, int C) {
  const long row = blockIdx.x;
  const int lane = threadIdx.x & (WAVE - 1);
  const float* lr = logits + row * C;
  float* lpr = logprobs + row * C;
  float m = -3.4e38f;
  for (int c = lane; c < C; c += WAVE) m = fmaxf(m, lr[c]);
  m = wave_max(m);
  float sum = 0.f;
  for (int c = lane; c < C; c += WAVE) sum += __expf(lr[c] - m);
  sum = wave_sum(sum);
  const float lse = m + __logf(sum);
  for (int c = lane; c < C; c += WAVE) lpr[c] = lr[c] - lse;
  if (lane == 0) losses[row] = lse - lr[labels[row]];
}

__global__ void ce_bwd_kernel(const float* __restrict__ dloss,
                              const float* __restrict__ logprobs,
                              const long* __restrict__ labels,
                              float* __restrict__ dlogits, int C, float invB) {
  const long row = blockIdx.x;
  const int lane = threadIdx.x & (WAVE - 1);
  const float d = dloss[0] * invB;
  for (int c = lane; c < C; c += WAVE) {
    const float p = __expf(logprobs[row * C + c]);
    dlogits[row * C + c] = d * (p - (c == labels[row] ? 1.f : 0.f));
  }
}

}  // namespace

std::vector<torch::Tensor> cross_entropy_fwd(torch::Tensor logits,
                                             torch::Tensor labels) {
  TORCH_CHECK(logits.scalar_type() == torch::kFloat32);
  const int C = logits.size(-1);
  const long B = logits.size(0);
  auto losses = torch::empty({B}, logits.options());
  auto logprobs = torch::empty_like(logits);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(ce_fwd_kernel, dim3(B), dim3(WAVE), 0, stream,
                     logits.data_ptr<float>(), labels.data_ptr<long>(),
                     losses.data_ptr<float>(), logprobs.data_ptr<float>(), C);
  return {losses.mean(), logprobs};
}

torch::Tensor cross_entropy_bwd(torch::Tensor dloss, torch::Tensor logprobs,
                                torch::Tensor labels) {
  const int C = logprobs.size(-1);
  const long B = logprobs.size(0);
  auto dlogits = torch::empty_like(logprobs);
  auto stream = at::hip::getCurrentHIPStream();
  auto d = dloss.to(torch::kFloat32).contiguous();
  hipLaunchKernelGGL(ce_bwd_kernel, dim3(B), dim3(WAVE), 0, stream,
                     d.data_ptr<float>(), logprobs.data_ptr<float>(),
                     labels.data_ptr<long>(), dlogits.data_ptr<float>(), C,
                     1.f / B);
  return dlogits;
}
