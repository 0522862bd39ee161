// Categorical-projection (C51) and scatter-connection kernels.
// Replaces DI-hpc CUDA kernels #2 and #7.
#include "common.h"

// One block per batch sample; the projected row is accumulated in LDS
// (atomic adds on 32-bank LDS, conflicts only when two atoms project to the
// same support cell), then written out coalesced.
__global__ void c51_project_kernel(
    const float* __restrict__ next_dist,  // [B, N, A]
    const int64_t* __restrict__ next_act, // [B]
    const float* __restrict__ reward,     // [B]
    const float* __restrict__ done,       // [B]
    float* __restrict__ proj,             // [B, A]
    int B,
    int N,
    int A,
    float v_min,
    float v_max,
    float gamma_n
) {
    extern __shared__ float row[];  // [A]
    int b = blockIdx.x;
    if (b >= B) return;
    float delta_z = (v_max - v_min) / (A - 1);
    const float* p = next_dist + ((int64_t)b * N + next_act[b]) * A;
    float r = reward[b];
    float nd = 1.f - done[b];
    for (int j = threadIdx.x; j < A; j += blockDim.x) row[j] = 0.f;
    __syncthreads();
    for (int j = threadIdx.x; j < A; j += blockDim.x) {
        float support = v_min + delta_z * j;
        float tz = r + nd * gamma_n * support;
        tz = fminf(fmaxf(tz, v_min), v_max);
        float pos = (tz - v_min) / delta_z;
        int l = (int)floorf(pos);
        int u = (int)ceilf(pos);
        // keep mass when l == u (pos integral)
        if (l == u) {
            if (u > 0) l = u - 1;
            else u = l + 1;
        }
        float pj = p[j];
        atomicAdd(&row[l], pj * (u - pos));
        atomicAdd(&row[u], pj * (pos - l));
    }
    __syncthreads();
    float* out = proj + (int64_t)b * A;
    for (int j = threadIdx.x; j < A; j += blockDim.x) out[j] = row[j];
}

torch::Tensor c51_project(
    torch::Tensor next_dist, torch::Tensor next_act, torch::Tensor reward, torch::Tensor done,
    double v_min, double v_max, double gamma_n
) {
    CHECK_INPUT(next_dist);
    CHECK_INPUT(next_act);
    CHECK_INPUT(reward);
    CHECK_INPUT(done);
    TORCH_CHECK(next_dist.dim() == 3, "next_dist must be [B, N, A]");
    int B = next_dist.size(0), N = next_dist.size(1), A = next_dist.size(2);
    auto proj = torch::empty({B, A}, next_dist.options());
    int block = 64;  // one wave per sample; A (51) fits one pass + remainder
    hipLaunchKernelGGL(
        c51_project_kernel, dim3(B), dim3(block), A * sizeof(float), ding_current_stream(),
        next_dist.data_ptr<float>(), next_act.data_ptr<int64_t>(), reward.data_ptr<float>(),
        done.data_ptr<float>(), proj.data_ptr<float>(), B, N, A, (float)v_min, (float)v_max, (float)gamma_n
    );
    HIP_CHECK_LAST();
    return proj;
}

// ---------------------------------------------------------------- scatter
// x [B, M, N] entity features scattered onto spatial positions index [B, M]
// (flat H*W). 'add': thread per (b, m, n-chunk) atomicAdd. 'cover': thread
// per (b, n) walks entities in order -> deterministic last-write-wins.
__global__ void scatter_add_kernel(
    const float* __restrict__ x, const int64_t* __restrict__ index, float* __restrict__ out,
    int B, int M, int N, int HW
) {
    int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t total = (int64_t)B * M * N;
    if (tid >= total) return;
    int n = tid % N;
    int m = (tid / N) % M;
    int b = tid / ((int64_t)N * M);
    int64_t pos = index[(int64_t)b * M + m];
    atomicAdd(&out[((int64_t)b * N + n) * HW + pos], x[tid]);
}

__global__ void scatter_cover_kernel(
    const float* __restrict__ x, const int64_t* __restrict__ index, float* __restrict__ out,
    int B, int M, int N, int HW
) {
    int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (tid >= (int64_t)B * N) return;
    int n = tid % N;
    int b = tid / N;
    for (int m = 0; m < M; ++m) {
        int64_t pos = index[(int64_t)b * M + m];
        out[((int64_t)b * N + n) * HW + pos] = x[((int64_t)b * M + m) * N + n];
    }
}

torch::Tensor scatter_connection(torch::Tensor x, torch::Tensor index, int64_t H, int64_t W, int64_t scatter_add) {
    CHECK_INPUT(x);
    CHECK_INPUT(index);
    TORCH_CHECK(x.dim() == 3, "x must be [B, M, N]");
    int B = x.size(0), M = x.size(1), N = x.size(2);
    int HW = (int)(H * W);
    auto out = torch::zeros({B, N, (int64_t)HW}, x.options());
    hipStream_t stream = ding_current_stream();
    if (scatter_add) {
        int64_t total = (int64_t)B * M * N;
        hipLaunchKernelGGL(
            scatter_add_kernel, dim3(cdiv(total, 256)), dim3(256), 0, stream,
            x.data_ptr<float>(), index.data_ptr<int64_t>(), out.data_ptr<float>(), B, M, N, HW
        );
    } else {
        hipLaunchKernelGGL(
            scatter_cover_kernel, dim3(cdiv((int64_t)B * N, 256)), dim3(256), 0, stream,
            x.data_ptr<float>(), index.data_ptr<int64_t>(), out.data_ptr<float>(), B, M, N, HW
        );
    }
    HIP_CHECK_LAST();
    return out.view({B, N, H, W});
}
