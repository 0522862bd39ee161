// Fully fused v-trace (IMPALA) loss, discrete actions, with analytic backward.
// Replaces DI-hpc CUDA kernel #10; parity with reference
// ding/rl_utils/vtrace.py:73 (vtrace_error_discrete_action).
//
// Eager lane at [T=32,B=128,N=6] issues ~35 kernels (two log_softmax, two
// gathers, clamps, the T-step python scan OR the scan kernel, cat, mse,
// entropy chain, three means); this is 2 forward launches + 1 backward
// launch + one 3-scalar reduction.
//
// Kernel 1 (rows, T*B-parallel): log-softmax over N for target & behaviour,
//   IS ratio, target log-prob of the action, entropy.
// Kernel 2 (columns, B-parallel): the serial reverse scan
//   vs_t = V_t + delta_t + gamma*lambda*c_t*(vs_{t+1} - V_{t+1}) and the
//   pg advantage, then the three per-row loss contributions.
// Backward (rows): d_logit from the pg + entropy terms, d_value from the
//   value term; upstream grad scales arrive as a device [3] tensor so the
//   launch is hipGraph-capture safe.
#include "common.h"

// row outputs: 0 = IS ratio, 1 = lp_a (target), 2 = entropy
__global__ void vtrace_row_kernel(
    const float* __restrict__ t_logit,   // [TB, N]
    const float* __restrict__ b_logit,   // [TB, N]
    const int64_t* __restrict__ action,  // [TB]
    float* __restrict__ row_out,         // [TB, 3]
    int TB,
    int N
) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= TB) return;
    const float* lt = t_logit + (int64_t)i * N;
    const float* lb = b_logit + (int64_t)i * N;
    int a = (int)action[i];

    float mx_t = -1e30f, mx_b = -1e30f;
    for (int j = 0; j < N; ++j) {
        mx_t = fmaxf(mx_t, lt[j]);
        mx_b = fmaxf(mx_b, lb[j]);
    }
    float se_t = 0.f, se_b = 0.f;
    for (int j = 0; j < N; ++j) {
        se_t += __expf(lt[j] - mx_t);
        se_b += __expf(lb[j] - mx_b);
    }
    float lse_t = __logf(se_t) + mx_t;
    float lse_b = __logf(se_b) + mx_b;
    float lp_a = lt[a] - lse_t;
    float entropy = 0.f;
    for (int j = 0; j < N; ++j) {
        float lpj = lt[j] - lse_t;
        entropy -= __expf(lpj) * lpj;
    }
    float* o = row_out + (int64_t)i * 3;
    o[0] = __expf(lp_a - (lb[a] - lse_b));
    o[1] = lp_a;
    o[2] = entropy;
}

// loss contribs: 0 = -lp_a*adv*w, 1 = (v-vs)^2*w, 2 = entropy*w
__global__ void vtrace_scan_kernel(
    const float* __restrict__ row_out,  // [T*B, 3]
    const float* __restrict__ value,    // [T+1, B]
    const float* __restrict__ reward,   // [T, B]
    const float* __restrict__ weight,   // [T, B] or nullptr
    float* __restrict__ ret,            // [T, B] (vs)
    float* __restrict__ adv,            // [T, B]
    float* __restrict__ contrib,        // [T*B, 3]
    int T,
    int B,
    float gamma,
    float lambda_,
    float rho_c,
    float c_c,
    float rho_pg_c
) {
    int b = blockIdx.x * blockDim.x + threadIdx.x;
    if (b >= B) return;
    // pass 1: vs via the reverse scan of the correction term
    float acc = 0.f;
    for (int t = T - 1; t >= 0; --t) {
        int64_t idx = (int64_t)t * B + b;
        float is = row_out[idx * 3 + 0];
        float rho = fminf(is, rho_c);
        float c = fminf(is, c_c);
        float delta = rho * (reward[idx] + gamma * value[idx + B] - value[idx]);
        acc = delta + gamma * lambda_ * c * acc;
        ret[idx] = value[idx] + acc;
    }
    // pass 2 (forward): advantage uses vs_{t+1} (bootstrap at t = T-1)
    for (int t = 0; t < T; ++t) {
        int64_t idx = (int64_t)t * B + b;
        float vs_next = (t + 1 < T) ? ret[idx + B] : value[(int64_t)T * B + b];
        float is = row_out[idx * 3 + 0];
        float pg_rho = fminf(is, rho_pg_c);
        float a = pg_rho * (reward[idx] + gamma * vs_next - value[idx]);
        adv[idx] = a;
        float w = weight ? weight[idx] : 1.f;
        float diff = value[idx] - ret[idx];
        contrib[idx * 3 + 0] = -row_out[idx * 3 + 1] * a * w;
        contrib[idx * 3 + 1] = diff * diff * w;
        contrib[idx * 3 + 2] = row_out[idx * 3 + 2] * w;
    }
}

// d(logit), d(value) for total = gs[0]*pg + gs[1]*value + gs[2]*entropy
__global__ void vtrace_bwd_kernel(
    const float* __restrict__ t_logit,   // [TB, N]
    const float* __restrict__ row_out,   // [TB, 3]
    const float* __restrict__ value,     // [T+1, B]
    const float* __restrict__ ret,       // [T, B]
    const float* __restrict__ adv,       // [T, B]
    const int64_t* __restrict__ action,  // [TB]
    const float* __restrict__ weight,    // [TB] or nullptr
    const float* __restrict__ gscales,   // [3] device
    float* __restrict__ d_logit,         // [TB, N]
    float* __restrict__ d_value,         // [T+1, B] (pre-zeroed; rows < T written)
    int TB,
    int N
) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= TB) return;
    float g_pg = gscales[0], g_v = gscales[1], g_ent = gscales[2];
    const float* lt = t_logit + (int64_t)i * N;
    float* dl = d_logit + (int64_t)i * N;
    int a = (int)action[i];
    float w = weight ? weight[i] : 1.f;
    float invTB = 1.f / TB;

    float mx = -1e30f;
    for (int j = 0; j < N; ++j) mx = fmaxf(mx, lt[j]);
    float se = 0.f;
    for (int j = 0; j < N; ++j) se += __expf(lt[j] - mx);
    float lse = __logf(se) + mx;

    float entropy = row_out[(int64_t)i * 3 + 2];
    float g_lp = g_pg * (-w * invTB) * adv[i];  // d(pg)/d(lp_a)
    for (int j = 0; j < N; ++j) {
        float lpj = lt[j] - lse;
        float pj = __expf(lpj);
        float d = g_lp * ((j == a ? 1.f : 0.f) - pj);
        d += g_ent * (w * invTB) * (-pj * (lpj + entropy));
        dl[j] = d;
    }
    d_value[i] = g_v * w * invTB * 2.f * (value[i] - ret[i]);
}

std::vector<torch::Tensor> vtrace_fwd(
    torch::Tensor t_logit, torch::Tensor b_logit, torch::Tensor action, torch::Tensor value,
    torch::Tensor reward, torch::Tensor weight, double gamma, double lambda_, double rho_c, double c_c,
    double rho_pg_c
) {
    CHECK_INPUT(t_logit);
    CHECK_INPUT(b_logit);
    CHECK_INPUT(action);
    CHECK_INPUT(value);
    CHECK_INPUT(reward);
    TORCH_CHECK(t_logit.dim() == 3, "t_logit must be [T, B, N]");
    int T = t_logit.size(0), B = t_logit.size(1), N = t_logit.size(2);
    TORCH_CHECK(value.size(0) == T + 1 && value.size(1) == B, "value must be [T+1, B]");
    int TB = T * B;
    auto row_out = torch::empty({TB, 3}, t_logit.options());
    auto ret = torch::empty({T, B}, t_logit.options());
    auto adv = torch::empty({T, B}, t_logit.options());
    auto contrib = torch::empty({TB, 3}, t_logit.options());
    const float* w_ptr = weight.defined() && weight.numel() > 0 ? weight.data_ptr<float>() : nullptr;
    int block = 256;
    hipLaunchKernelGGL(
        vtrace_row_kernel, dim3(cdiv(TB, block)), dim3(block), 0, ding_current_stream(),
        t_logit.data_ptr<float>(), b_logit.data_ptr<float>(), action.data_ptr<int64_t>(),
        row_out.data_ptr<float>(), TB, N
    );
    HIP_CHECK_LAST();
    hipLaunchKernelGGL(
        vtrace_scan_kernel, dim3(cdiv(B, 64)), dim3(64), 0, ding_current_stream(),
        row_out.data_ptr<float>(), value.data_ptr<float>(), reward.data_ptr<float>(), w_ptr,
        ret.data_ptr<float>(), adv.data_ptr<float>(), contrib.data_ptr<float>(), T, B, (float)gamma,
        (float)lambda_, (float)rho_c, (float)c_c, (float)rho_pg_c
    );
    HIP_CHECK_LAST();
    return {row_out, ret, adv, contrib};
}

std::vector<torch::Tensor> vtrace_bwd(
    torch::Tensor t_logit, torch::Tensor row_out, torch::Tensor value, torch::Tensor ret, torch::Tensor adv,
    torch::Tensor action, torch::Tensor weight, torch::Tensor gscales
) {
    CHECK_INPUT(t_logit);
    int T = t_logit.size(0), B = t_logit.size(1), N = t_logit.size(2);
    int TB = T * B;
    auto d_logit = torch::empty_like(t_logit);
    auto d_value = torch::zeros({T + 1, B}, t_logit.options());
    const float* w_ptr = weight.defined() && weight.numel() > 0 ? weight.data_ptr<float>() : nullptr;
    int block = 256;
    hipLaunchKernelGGL(
        vtrace_bwd_kernel, dim3(cdiv(TB, block)), dim3(block), 0, ding_current_stream(),
        t_logit.data_ptr<float>(), row_out.data_ptr<float>(), value.data_ptr<float>(), ret.data_ptr<float>(),
        adv.data_ptr<float>(), action.data_ptr<int64_t>(), w_ptr, gscales.data_ptr<float>(),
        d_logit.data_ptr<float>(), d_value.data_ptr<float>(), TB, N
    );
    HIP_CHECK_LAST();
    return {d_logit, d_value};
}
