// Reverse-scan kernels over the time dimension (GAE / v-trace / TD-lambda /
// UPGO returns). Replaces DI-hpc CUDA kernels #1/#8/#10.
//
// Layout: inputs are [T, N] row-major (N = flattened batch columns). One
// thread owns one column and walks T serially; consecutive lanes read
// consecutive addresses at each step, so every time-step is one coalesced
// 64-lane load per wave. The win vs the PyTorch lane is algorithmic: the
// eager loop launches T kernels (T=1024 for the DI-hpc harness shape), this
// is ONE launch with N-way parallelism.
#include "common.h"

__global__ void reverse_scan_kernel(
    const float* __restrict__ delta,
    const float* __restrict__ factor,
    float* __restrict__ out,
    int T,
    int N
) {
    int n = blockIdx.x * blockDim.x + threadIdx.x;
    if (n >= N) return;
    float acc = 0.f;
    for (int t = T - 1; t >= 0; --t) {
        int64_t idx = (int64_t)t * N + n;
        acc = delta[idx] + factor[idx] * acc;
        out[idx] = acc;
    }
}

// result[T-1] = r[T-1] + (1-done)*g[T-1]*V[T-1(bootstrap idx)]
// result[t]   = r[t] + (1-done[t])*(g*l*result[t+1] + g*(1-l)*V[t])
__global__ void mfv_kernel(
    const float* __restrict__ bootstrap,  // [T, N] = V at steps 1..T
    const float* __restrict__ rewards,
    const float* __restrict__ gammas,
    const float* __restrict__ lambdas,
    const float* __restrict__ done,
    float* __restrict__ out,
    int T,
    int N
) {
    int n = blockIdx.x * blockDim.x + threadIdx.x;
    if (n >= N) return;
    int64_t last = (int64_t)(T - 1) * N + n;
    float acc = rewards[last] + (1.f - done[last]) * gammas[last] * bootstrap[last];
    out[last] = acc;
    for (int t = T - 2; t >= 0; --t) {
        int64_t idx = (int64_t)t * N + n;
        float g = gammas[idx];
        float gl = g * lambdas[idx];
        acc = rewards[idx] + (1.f - done[idx]) * (gl * acc + (g - gl) * bootstrap[idx]);
        out[idx] = acc;
    }
}

torch::Tensor reverse_scan(torch::Tensor delta, torch::Tensor factor) {
    CHECK_INPUT(delta);
    CHECK_INPUT(factor);
    TORCH_CHECK(delta.dim() == 2 && factor.sizes() == delta.sizes(), "expect matching [T, N]");
    auto out = torch::empty_like(delta);
    int T = delta.size(0), N = delta.size(1);
    int block = 256;
    int grid = cdiv(N, block);
    hipLaunchKernelGGL(
        reverse_scan_kernel, dim3(grid), dim3(block), 0, ding_current_stream(),
        delta.data_ptr<float>(), factor.data_ptr<float>(), out.data_ptr<float>(), T, N
    );
    HIP_CHECK_LAST();
    return out;
}

torch::Tensor multistep_forward_view(
    torch::Tensor bootstrap, torch::Tensor rewards, torch::Tensor gammas, torch::Tensor lambdas, torch::Tensor done
) {
    CHECK_INPUT(bootstrap);
    CHECK_INPUT(rewards);
    CHECK_INPUT(gammas);
    CHECK_INPUT(lambdas);
    CHECK_INPUT(done);
    TORCH_CHECK(rewards.dim() == 2, "rewards must be [T, N]");
    auto out = torch::empty_like(rewards);
    int T = rewards.size(0), N = rewards.size(1);
    int block = 256;
    int grid = cdiv(N, block);
    hipLaunchKernelGGL(
        mfv_kernel, dim3(grid), dim3(block), 0, ding_current_stream(),
        bootstrap.data_ptr<float>(), rewards.data_ptr<float>(), gammas.data_ptr<float>(),
        lambdas.data_ptr<float>(), done.data_ptr<float>(), out.data_ptr<float>(), T, N
    );
    HIP_CHECK_LAST();
    return out;
}
