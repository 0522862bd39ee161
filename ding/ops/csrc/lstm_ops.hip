// Fused LayerNorm-LSTM pointwise cell for CDNA4 / gfx950.
//
// Replaces the ~13-kernel eager chain per LSTM timestep (LayerNorm on the
// hidden-side gates + gate arithmetic + state update) with:
//   fwd: one kernel  — LN(gh_raw) reduction + gates + (h', c')
//   bwd: one kernel  — gate backward + LN input-grad (two row reductions)
// The h @ W_h GEMM and the per-parameter (gamma/beta/bias) batch reductions
// stay in torch (hipBLASLt / single eager sums).
//
// One workgroup per batch row; 4H elements strided across 256 lanes; row
// statistics via LDS tree reduction.
#include "common.h"

__device__ __forceinline__ float block_reduce_sum(float v, float* tmp) {
    int tid = threadIdx.x;
    tmp[tid] = v;
    __syncthreads();
    for (int s = blockDim.x / 2; s > 0; s >>= 1) {
        if (tid < s) tmp[tid] += tmp[tid + s];
        __syncthreads();
    }
    float out = tmp[0];
    __syncthreads();
    return out;
}

__global__ void lstm_cell_fwd_kernel(
    const float* __restrict__ gxn,      // [B, 4H] input-side gates (already LN'd)
    const float* __restrict__ gh_raw,   // [B, 4H] h @ W_h (pre-LN)
    const float* __restrict__ gamma,    // [4H]
    const float* __restrict__ beta,     // [4H]
    const float* __restrict__ bias,     // [4H]
    const float* __restrict__ c_in,     // [B, H]
    float* __restrict__ h_out,          // [B, H]
    float* __restrict__ c_out,          // [B, H]
    float* __restrict__ xhat,           // [B, 4H] saved normalized gh
    float* __restrict__ acts,           // [B, 4H] saved activated gates i,f,o,u
    float* __restrict__ rstd_out,       // [B]
    int B, int H
) {
    __shared__ float tmp[256];
    int b = blockIdx.x;
    int tid = threadIdx.x;
    int G = 4 * H;
    const float* gh = gh_raw + (int64_t)b * G;

    float s = 0.f;
    for (int j = tid; j < G; j += blockDim.x) s += gh[j];
    float mean = block_reduce_sum(s, tmp) / G;
    float v = 0.f;
    for (int j = tid; j < G; j += blockDim.x) {
        float d = gh[j] - mean;
        v += d * d;
    }
    float rstd = rsqrtf(block_reduce_sum(v, tmp) / G + 1e-5f);
    if (tid == 0) rstd_out[b] = rstd;

    const float* gx = gxn + (int64_t)b * G;
    float* xh = xhat + (int64_t)b * G;
    float* ac = acts + (int64_t)b * G;
    // pass 1: gate pre-activations -> activations (i, f, o, u blocks)
    for (int j = tid; j < G; j += blockDim.x) {
        float hat = (gh[j] - mean) * rstd;
        xh[j] = hat;
        float g = gx[j] + gamma[j] * hat + beta[j] + bias[j];
        int block = j / H;  // 0:i 1:f 2:o 3:u
        ac[j] = block == 3 ? tanhf(g) : 1.f / (1.f + __expf(-g));
    }
    __syncthreads();
    // pass 2: state update (j < H)
    const float* ci = c_in + (int64_t)b * H;
    float* ho = h_out + (int64_t)b * H;
    float* co = c_out + (int64_t)b * H;
    for (int j = tid; j < H; j += blockDim.x) {
        float i_ = ac[j];
        float f_ = ac[H + j];
        float o_ = ac[2 * H + j];
        float u_ = ac[3 * H + j];
        float c = f_ * ci[j] + i_ * u_;
        co[j] = c;
        ho[j] = o_ * tanhf(c);
    }
}

__global__ void lstm_cell_bwd_kernel(
    const float* __restrict__ dh,       // [B, H]
    const float* __restrict__ dc_next,  // [B, H] (or nullptr)
    const float* __restrict__ acts,     // [B, 4H]
    const float* __restrict__ xhat,     // [B, 4H]
    const float* __restrict__ gamma,    // [4H]
    const float* __restrict__ c_in,     // [B, H]
    const float* __restrict__ c_out,    // [B, H]
    const float* __restrict__ rstd_in,  // [B]
    float* __restrict__ dg,             // [B, 4H] d(gate preact) == d(gxn)
    float* __restrict__ dgh,            // [B, 4H] d(gh_raw) through LN
    float* __restrict__ dc_in,          // [B, H]
    int B, int H
) {
    __shared__ float tmp[256];
    int b = blockIdx.x;
    int tid = threadIdx.x;
    int G = 4 * H;
    const float* ac = acts + (int64_t)b * G;
    const float* xh = xhat + (int64_t)b * G;
    const float* ci = c_in + (int64_t)b * H;
    const float* co = c_out + (int64_t)b * H;
    float* dgb = dg + (int64_t)b * G;
    float rstd = rstd_in[b];

    // gate backward (j < H covers all four gate slots)
    for (int j = tid; j < H; j += blockDim.x) {
        float i_ = ac[j], f_ = ac[H + j], o_ = ac[2 * H + j], u_ = ac[3 * H + j];
        float tc = tanhf(co[j]);
        float dh_ = dh[(int64_t)b * H + j];
        float dct = dh_ * o_ * (1.f - tc * tc);
        if (dc_next) dct += dc_next[(int64_t)b * H + j];
        dgb[j] = dct * u_ * i_ * (1.f - i_);                    // d i-preact
        dgb[H + j] = dct * ci[j] * f_ * (1.f - f_);             // d f-preact
        dgb[2 * H + j] = dh_ * tc * o_ * (1.f - o_);            // d o-preact
        dgb[3 * H + j] = dct * i_ * (1.f - u_ * u_);            // d u-preact
        dc_in[(int64_t)b * H + j] = dct * f_;
    }
    __syncthreads();
    // LN backward for the gh side: dxhat = dg * gamma
    float s1 = 0.f, s2 = 0.f;
    for (int j = tid; j < G; j += blockDim.x) {
        float dxh = dgb[j] * gamma[j];
        s1 += dxh;
        s2 += dxh * xh[j];
    }
    float m1 = block_reduce_sum(s1, tmp) / G;
    float m2 = block_reduce_sum(s2, tmp) / G;
    float* dghb = dgh + (int64_t)b * G;
    for (int j = tid; j < G; j += blockDim.x) {
        float dxh = dgb[j] * gamma[j];
        dghb[j] = rstd * (dxh - m1 - xh[j] * m2);
    }
}

std::vector<torch::Tensor> lstm_cell_fwd(
    torch::Tensor gxn, torch::Tensor gh_raw, torch::Tensor gamma, torch::Tensor beta, torch::Tensor bias,
    torch::Tensor c_in
) {
    CHECK_INPUT(gxn);
    CHECK_INPUT(gh_raw);
    int B = gxn.size(0), G = gxn.size(1), H = G / 4;
    auto h_out = torch::empty({B, H}, gxn.options());
    auto c_out = torch::empty({B, H}, gxn.options());
    auto xhat = torch::empty({B, G}, gxn.options());
    auto acts = torch::empty({B, G}, gxn.options());
    auto rstd = torch::empty({B}, gxn.options());
    hipLaunchKernelGGL(
        lstm_cell_fwd_kernel, dim3(B), dim3(256), 0, ding_current_stream(),
        gxn.data_ptr<float>(), gh_raw.data_ptr<float>(), gamma.data_ptr<float>(), beta.data_ptr<float>(),
        bias.data_ptr<float>(), c_in.data_ptr<float>(), h_out.data_ptr<float>(), c_out.data_ptr<float>(),
        xhat.data_ptr<float>(), acts.data_ptr<float>(), rstd.data_ptr<float>(), B, H
    );
    HIP_CHECK_LAST();
    return {h_out, c_out, xhat, acts, rstd};
}

std::vector<torch::Tensor> lstm_cell_bwd(
    torch::Tensor dh, torch::Tensor dc_next, torch::Tensor acts, torch::Tensor xhat, torch::Tensor gamma,
    torch::Tensor c_in, torch::Tensor c_out, torch::Tensor rstd
) {
    CHECK_INPUT(dh);
    int B = dh.size(0), H = dh.size(1), G = 4 * H;
    auto dg = torch::empty({B, G}, dh.options());
    auto dgh = torch::empty({B, G}, dh.options());
    auto dc_in = torch::empty({B, H}, dh.options());
    const float* dcn = dc_next.defined() && dc_next.numel() > 0 ? dc_next.data_ptr<float>() : nullptr;
    hipLaunchKernelGGL(
        lstm_cell_bwd_kernel, dim3(B), dim3(256), 0, ding_current_stream(),
        dh.data_ptr<float>(), dcn, acts.data_ptr<float>(), xhat.data_ptr<float>(), gamma.data_ptr<float>(),
        c_in.data_ptr<float>(), c_out.data_ptr<float>(), rstd.data_ptr<float>(), dg.data_ptr<float>(),
        dgh.data_ptr<float>(), dc_in.data_ptr<float>(), B, H
    );
    HIP_CHECK_LAST();
    return {dg, dgh, dc_in};
}
