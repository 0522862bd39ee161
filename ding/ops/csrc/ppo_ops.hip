// Fused PPO loss (discrete) with analytic backward.
// Replaces DI-hpc CUDA kernel #4: one kernel computes
// log-softmax, ratio, clipped surrogate, entropy, clipped value loss and the
// monitoring stats; one kernel produces d(logit_new) and d(value_new).
//
// Mapping: one thread per batch row; the action-dim loop runs in registers
// (N is small in DI workloads: 2-20 actions). The eager lane issues ~30
// separate kernels per call at [B<=~4096] shapes where launch+dispatch
// overhead dominates; the fused pair is 2 launches.
#include "common.h"

// per-row outputs packed for the host-side reduction:
// 0: -min(surr1, surr2)*w      (policy loss contrib)
// 1: 0.5*max(v1, v2)*w         (value loss contrib)
// 2: entropy*w                 (entropy contrib)
// 3: logp_old - logp_new       (approx_kl contrib)
// 4: clipped flag
// 5: ratio
// 6: use_surr1 / in-range indicator for backward
// 7: value branch indicator (1: unclipped grad, 0: clipped branch)
__global__ void ppo_fwd_kernel(
    const float* __restrict__ logit_new,   // [B, N]
    const float* __restrict__ logit_old,   // [B, N]
    const int64_t* __restrict__ action,    // [B]
    const float* __restrict__ value_new,   // [B]
    const float* __restrict__ value_old,   // [B]
    const float* __restrict__ adv,         // [B]
    const float* __restrict__ ret,         // [B]
    const float* __restrict__ weight,      // [B] (or nullptr)
    float* __restrict__ out,               // [B, 8]
    int B,
    int N,
    float clip_ratio,
    int use_value_clip
) {
    int b = blockIdx.x * blockDim.x + threadIdx.x;
    if (b >= B) return;
    const float* ln = logit_new + (int64_t)b * N;
    const float* lo = logit_old + (int64_t)b * N;
    int a = (int)action[b];
    float w = weight ? weight[b] : 1.f;

    // log-softmax new (max-subtracted) + entropy
    float mx_n = -1e30f, mx_o = -1e30f;
    for (int j = 0; j < N; ++j) {
        mx_n = fmaxf(mx_n, ln[j]);
        mx_o = fmaxf(mx_o, lo[j]);
    }
    float se_n = 0.f, se_o = 0.f;
    for (int j = 0; j < N; ++j) {
        se_n += __expf(ln[j] - mx_n);
        se_o += __expf(lo[j] - mx_o);
    }
    float lse_n = __logf(se_n) + mx_n;
    float lse_o = __logf(se_o) + mx_o;
    float lp_new = ln[a] - lse_n;
    float lp_old = lo[a] - lse_o;
    float entropy = 0.f;
    for (int j = 0; j < N; ++j) {
        float lpj = ln[j] - lse_n;
        entropy -= __expf(lpj) * lpj;
    }

    float r = __expf(lp_new - lp_old);
    float advb = adv[b];
    float surr1 = r * advb;
    float rc = fminf(fmaxf(r, 1.f - clip_ratio), 1.f + clip_ratio);
    float surr2 = rc * advb;
    float m = fminf(surr1, surr2);
    // gradient flows iff surr1 is the min, or r is inside the clip range
    float in_range = (r > 1.f - clip_ratio && r < 1.f + clip_ratio) ? 1.f : 0.f;
    float use1 = (surr1 <= surr2) ? 1.f : in_range;

    float v = value_new[b], vo = value_old[b], rt = ret[b];
    float v1 = (rt - v) * (rt - v);
    float vloss, vgrad_branch;
    if (use_value_clip) {
        float vclip = vo + fminf(fmaxf(v - vo, -clip_ratio), clip_ratio);
        float v2 = (rt - vclip) * (rt - vclip);
        if (v1 >= v2) {
            vloss = 0.5f * v1;
            vgrad_branch = 1.f;
        } else {
            vloss = 0.5f * v2;
            // grad only if vclip still tracks v
            vgrad_branch = (fabsf(v - vo) < clip_ratio) ? 2.f : 0.f;  // 2 marks clip branch
        }
    } else {
        vloss = 0.5f * v1;
        vgrad_branch = 1.f;
    }

    float* o = out + (int64_t)b * 8;
    o[0] = -m * w;
    o[1] = vloss * w;
    o[2] = entropy * w;
    o[3] = lp_old - lp_new;
    o[4] = (r > 1.f + clip_ratio || r < 1.f - clip_ratio) ? 1.f : 0.f;
    o[5] = r;
    o[6] = use1;
    o[7] = vgrad_branch;
}

// d(total)/d(logit_new), d(total)/d(value_new) where
// total = policy + wv * value - we * entropy   (means over B)
__global__ void ppo_bwd_kernel(
    const float* __restrict__ logit_new,
    const int64_t* __restrict__ action,
    const float* __restrict__ value_new,
    const float* __restrict__ value_old,
    const float* __restrict__ adv,
    const float* __restrict__ ret,
    const float* __restrict__ weight,
    const float* __restrict__ fwd_out,     // [B, 8]
    float* __restrict__ d_logit,           // [B, N]
    float* __restrict__ d_value,           // [B]
    int B,
    int N,
    float clip_ratio,
    const float* __restrict__ gscales      // [3] device: upstream grads of
                                           // (policy, value, entropy) losses —
                                           // device-resident so the launch is
                                           // hipGraph-capture safe (no sync)
) {
    int b = blockIdx.x * blockDim.x + threadIdx.x;
    if (b >= B) return;
    float grad_policy = gscales[0];
    float grad_value = gscales[1];
    float grad_entropy = gscales[2];
    const float* ln = logit_new + (int64_t)b * N;
    const float* o = fwd_out + (int64_t)b * 8;
    float* dl = d_logit + (int64_t)b * N;
    int a = (int)action[b];
    float w = weight ? weight[b] : 1.f;
    float invB = 1.f / B;

    float mx = -1e30f;
    for (int j = 0; j < N; ++j) mx = fmaxf(mx, ln[j]);
    float se = 0.f;
    for (int j = 0; j < N; ++j) se += __expf(ln[j] - mx);
    float lse = __logf(se) + mx;

    float r = o[5];
    float use1 = o[6];
    // policy: dL/dlp_new = -w/B * adv * r * use1 (per upstream grad)
    float g_lp = grad_policy * (-w * invB) * adv[b] * r * use1;
    // entropy term arrives with its own upstream grad (caller supplies sign)
    float entropy = o[2] / (w == 0.f ? 1.f : w);
    for (int j = 0; j < N; ++j) {
        float lpj = ln[j] - lse;
        float pj = __expf(lpj);
        float d = g_lp * ((j == a ? 1.f : 0.f) - pj);
        // dH/dz_j = -p_j (log p_j + H)
        d += grad_entropy * (w * invB) * (-pj * (lpj + entropy));
        dl[j] = d;
    }
    // value branch
    float branch = o[7];
    float v = value_new[b], vo = value_old[b], rt = ret[b];
    float dv = 0.f;
    if (branch == 1.f) {
        dv = -(rt - v);
    } else if (branch == 2.f) {
        float vclip = vo + fminf(fmaxf(v - vo, -clip_ratio), clip_ratio);
        dv = -(rt - vclip);
    }
    d_value[b] = grad_value * w * invB * dv;
}

std::vector<torch::Tensor> ppo_fwd(
    torch::Tensor logit_new, torch::Tensor logit_old, torch::Tensor action, torch::Tensor value_new,
    torch::Tensor value_old, torch::Tensor adv, torch::Tensor ret, torch::Tensor weight, double clip_ratio,
    int64_t use_value_clip
) {
    CHECK_INPUT(logit_new);
    CHECK_INPUT(logit_old);
    CHECK_INPUT(action);
    int B = logit_new.size(0), N = logit_new.size(1);
    auto out = torch::empty({B, 8}, logit_new.options());
    const float* w_ptr = weight.defined() && weight.numel() > 0 ? weight.data_ptr<float>() : nullptr;
    int block = 256;
    hipLaunchKernelGGL(
        ppo_fwd_kernel, dim3(cdiv(B, block)), dim3(block), 0, ding_current_stream(),
        logit_new.data_ptr<float>(), logit_old.data_ptr<float>(), action.data_ptr<int64_t>(),
        value_new.data_ptr<float>(), value_old.data_ptr<float>(), adv.data_ptr<float>(), ret.data_ptr<float>(),
        w_ptr, out.data_ptr<float>(), B, N, (float)clip_ratio, (int)use_value_clip
    );
    HIP_CHECK_LAST();
    return {out};
}

std::vector<torch::Tensor> ppo_bwd(
    torch::Tensor logit_new, torch::Tensor action, torch::Tensor value_new, torch::Tensor value_old,
    torch::Tensor adv, torch::Tensor ret, torch::Tensor weight, torch::Tensor fwd_out, double clip_ratio,
    torch::Tensor grad_scales
) {
    CHECK_INPUT(logit_new);
    int B = logit_new.size(0), N = logit_new.size(1);
    auto d_logit = torch::empty_like(logit_new);
    auto d_value = torch::empty_like(value_new);
    const float* w_ptr = weight.defined() && weight.numel() > 0 ? weight.data_ptr<float>() : nullptr;
    int block = 256;
    hipLaunchKernelGGL(
        ppo_bwd_kernel, dim3(cdiv(B, block)), dim3(block), 0, ding_current_stream(),
        logit_new.data_ptr<float>(), action.data_ptr<int64_t>(), value_new.data_ptr<float>(),
        value_old.data_ptr<float>(), adv.data_ptr<float>(), ret.data_ptr<float>(), w_ptr,
        fwd_out.data_ptr<float>(), d_logit.data_ptr<float>(), d_value.data_ptr<float>(), B, N,
        (float)clip_ratio, grad_scales.data_ptr<float>()
    );
    HIP_CHECK_LAST();
    return {d_logit, d_value};
}
