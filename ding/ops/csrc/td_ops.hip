// Fused n-step Q TD forward for the DQN family (CDNA4 / gfx950).
//
// One thread per batch row replaces the ~10-kernel eager chain
// (gather, gather, pow-weighted reward sum, masks, sub, square, mul, mean):
//   q_sa      = q[b, a_b]
//   boot      = next_n_q[b, a'_b]           (value-rescale: h^-1(boot))
//   ret       = sum_{t<n} gamma^t r[t,b] + G_n * boot * (1 - done_b)
//               where G_n = value_gamma[b] if given else gamma^n
//               (value-rescale: ret = h(ret))
//   td        = q_sa - ret
// out[b] = {td, ret, q_sa}. The loss/backward stay in torch: the gradient
// only touches q[b, a_b], one scatter.
#include "common.h"

__device__ __forceinline__ float h_tf(float x) {
    // value_transform: sign(x)(sqrt(|x|+1)-1) + eps x, eps=1e-2
    float s = x >= 0.f ? 1.f : -1.f;
    return s * (sqrtf(fabsf(x) + 1.f) - 1.f) + 1e-2f * x;
}

__device__ __forceinline__ float h_inv(float x) {
    const float eps = 1e-2f;
    float s = x >= 0.f ? 1.f : -1.f;
    float t = (sqrtf(1.f + 4.f * eps * (fabsf(x) + 1.f + eps)) - 1.f) / (2.f * eps);
    return s * (t * t - 1.f);
}

__global__ void q_nstep_fwd_kernel(
    const float* __restrict__ q,             // [B, N]
    const float* __restrict__ next_n_q,      // [B, N]
    const int64_t* __restrict__ action,      // [B]
    const int64_t* __restrict__ next_action, // [B]
    const float* __restrict__ reward,        // [nstep, B]
    const float* __restrict__ done,          // [B]
    const float* __restrict__ value_gamma,   // [B] or nullptr
    float* __restrict__ out,                 // [B, 3]: td, ret, q_sa
    int B,
    int N,
    int nstep,
    float gamma,
    int rescale
) {
    int b = blockIdx.x * blockDim.x + threadIdx.x;
    if (b >= B) return;
    float q_sa = q[(int64_t)b * N + action[b]];
    float boot = next_n_q[(int64_t)b * N + next_action[b]];
    if (rescale) boot = h_inv(boot);
    float acc = 0.f;
    float g = 1.f;
    for (int t = 0; t < nstep; ++t) {
        acc += g * reward[(int64_t)t * B + b];
        g *= gamma;
    }
    float gn = value_gamma ? value_gamma[b] : g;
    float ret = acc + gn * boot * (1.f - done[b]);
    if (rescale) ret = h_tf(ret);
    float* o = out + (int64_t)b * 3;
    o[0] = q_sa - ret;
    o[1] = ret;
    o[2] = q_sa;
}

std::vector<torch::Tensor> q_nstep_fwd(
    torch::Tensor q, torch::Tensor next_n_q, torch::Tensor action, torch::Tensor next_action,
    torch::Tensor reward, torch::Tensor done, torch::Tensor value_gamma, double gamma, int64_t nstep,
    int64_t rescale
) {
    CHECK_INPUT(q);
    CHECK_INPUT(next_n_q);
    int B = q.size(0), N = q.size(1);
    auto out = torch::empty({B, 3}, q.options());
    const float* vg = value_gamma.defined() && value_gamma.numel() > 0 ? value_gamma.data_ptr<float>() : nullptr;
    int block = 256;
    hipLaunchKernelGGL(
        q_nstep_fwd_kernel, dim3(cdiv(B, block)), dim3(block), 0, ding_current_stream(),
        q.data_ptr<float>(), next_n_q.data_ptr<float>(), action.data_ptr<int64_t>(),
        next_action.data_ptr<int64_t>(), reward.data_ptr<float>(), done.data_ptr<float>(), vg,
        out.data_ptr<float>(), B, N, (int)nstep, (float)gamma, (int)rescale
    );
    HIP_CHECK_LAST();
    return {out};
}
