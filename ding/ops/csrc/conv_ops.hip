// Direct 8x8-stride-4 "Atari stem" convolution for CDNA4 / gfx950.
//
// MIOpen's tuned choice for conv(4ch, 8x8, s4, 84x84) is a PER-SAMPLE
// im2col + GEMM loop: rocprof on the PPO bench shows 92,160 Im2d2Col
// launches (24% of GPU time) plus fp64-accum naive convs on untuned batch
// shapes. This kernel computes the conv directly:
//   - one workgroup per (sample, 4-output-row tile)
//   - the input patch (C x 20 x W, ~27 KB) staged in LDS; the 64 KB weight
//     tensor streams from global where every workgroup's identical reads
//     stay resident in the XCD's L2
//   - thread (oc, prow) accumulates W_out outputs in registers, FMA over
//     the 8x8xC receptive field
// Backward-weight (wrw): one workgroup per sample, register-accumulated
// partials + one atomicAdd per dW entry. dX is never needed: this is the
// input layer (observations are leaves).
#include "common.h"

#define STEM_K 8
#define STEM_S 4
#define PH_TILE 4
#define MAX_WOUT 32

template <int WOUT>
__global__ void stem_conv_fwd_kernel(
    const float* __restrict__ x,    // [B, C, H, W]
    const float* __restrict__ w,    // [O, C, 8, 8]
    const float* __restrict__ bias, // [O]
    float* __restrict__ y,          // [B, O, HO, WO]
    int B, int C, int H, int W, int O, int HO
) {
    // x tile in LDS (~27 KB); weights stream from global — every workgroup
    // reads the same 64 KB so they stay resident in the XCD's L2
    extern __shared__ float x_lds[];
    const int tile_h = (PH_TILE - 1) * STEM_S + STEM_K;

    int b = blockIdx.x;
    int ph0 = blockIdx.y * PH_TILE;
    int tid = threadIdx.x;
    int oc = tid % O;
    int prow = tid / O;          // 0..PH_TILE-1

    int h0 = ph0 * STEM_S;
    int xcount = C * tile_h * W;
    for (int i = tid; i < xcount; i += blockDim.x) {
        int c = i / (tile_h * W);
        int rem = i % (tile_h * W);
        int hh = rem / W;
        int ww = rem % W;
        int hsrc = h0 + hh;
        x_lds[i] = hsrc < H ? x[(((int64_t)b * C + c) * H + hsrc) * W + ww] : 0.f;
    }
    __syncthreads();

    int ph = ph0 + prow;
    if (ph >= HO || prow >= PH_TILE) return;

    // WOUT is a template constant: acc[] stays in registers and the pw loop
    // unrolls exactly (a runtime bound demotes it to scratch — measured 40x)
    float acc[WOUT];
    #pragma unroll
    for (int i = 0; i < WOUT; ++i) acc[i] = 0.f;

    const float* wo = w + (int64_t)oc * C * STEM_K * STEM_K;
    for (int c = 0; c < C; ++c) {
        const float* xc = x_lds + c * tile_h * W;
        for (int kh = 0; kh < STEM_K; ++kh) {
            const float* xrow = xc + (prow * STEM_S + kh) * W;
            const float* wrow = wo + (c * STEM_K + kh) * STEM_K;
            float wreg[STEM_K];
            #pragma unroll
            for (int kw = 0; kw < STEM_K; ++kw) wreg[kw] = wrow[kw];
            #pragma unroll
            for (int kw = 0; kw < STEM_K; ++kw) {
                #pragma unroll
                for (int pw = 0; pw < WOUT; ++pw) {
                    acc[pw] = fmaf(wreg[kw], xrow[pw * STEM_S + kw], acc[pw]);
                }
            }
        }
    }
    float bv = bias ? bias[oc] : 0.f;
    float* yrow = y + (((int64_t)b * O + oc) * HO + ph) * WOUT;
    #pragma unroll
    for (int pw = 0; pw < WOUT; ++pw) yrow[pw] = acc[pw] + bv;
}

// dW[o,c,kh,kw] = sum_{b,ph,pw} dY[b,o,ph,pw] * X[b,c,ph*4+kh,pw*4+kw]
// one workgroup per sample; LDS-resident partial dW, atomic flush.
__global__ void stem_conv_wrw_kernel(
    const float* __restrict__ x,   // [B, C, H, W]
    const float* __restrict__ dy,  // [B, O, HO, WO]
    float* __restrict__ dw,        // [O, C, 8, 8] (pre-zeroed)
    int B, int C, int H, int W, int O, int HO, int WO
) {
    const int wcount = O * C * STEM_K * STEM_K;
    int b = blockIdx.x;
    int tid = threadIdx.x;
    // each thread owns a strided subset of dW entries for this sample;
    // register accumulation + one atomicAdd per entry (dW is 64 KB: the
    // per-entry contention is one add per sample-block)
    for (int i = tid; i < wcount; i += blockDim.x) {
        int o = i / (C * STEM_K * STEM_K);
        int rem = i % (C * STEM_K * STEM_K);
        int c = rem / (STEM_K * STEM_K);
        int kh = (rem / STEM_K) % STEM_K;
        int kw = rem % STEM_K;
        const float* dyb = dy + ((int64_t)b * O + o) * HO * WO;
        const float* xb = x + ((int64_t)b * C + c) * H * W;
        float acc = 0.f;
        for (int ph = 0; ph < HO; ++ph) {
            const float* dyr = dyb + ph * WO;
            const float* xr = xb + (ph * STEM_S + kh) * W + kw;
            for (int pw = 0; pw < WO; ++pw) {
                acc = fmaf(dyr[pw], xr[pw * STEM_S], acc);
            }
        }
        atomicAdd(&dw[i], acc);
    }
}

torch::Tensor stem_conv_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor bias) {
    CHECK_INPUT(x);
    CHECK_INPUT(w);
    int B = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
    int O = w.size(0);
    TORCH_CHECK(w.size(2) == STEM_K && w.size(3) == STEM_K, "stem conv requires 8x8 kernel");
    int HO = (H - STEM_K) / STEM_S + 1;
    int WO = (W - STEM_K) / STEM_S + 1;
    auto y = torch::empty({B, O, HO, WO}, x.options());
    const float* bias_ptr = bias.defined() && bias.numel() > 0 ? bias.data_ptr<float>() : nullptr;
    int tile_h = (PH_TILE - 1) * STEM_S + STEM_K;
    size_t lds_bytes = (size_t)(C * tile_h * W) * sizeof(float);
    dim3 grid(B, cdiv(HO, PH_TILE));
    dim3 block(O * PH_TILE);
    auto stream = ding_current_stream();
    switch (WO) {
        case 20:
            hipLaunchKernelGGL((stem_conv_fwd_kernel<20>), grid, block, lds_bytes, stream,
                x.data_ptr<float>(), w.data_ptr<float>(), bias_ptr, y.data_ptr<float>(), B, C, H, W, O, HO);
            break;
        case 15:
            hipLaunchKernelGGL((stem_conv_fwd_kernel<15>), grid, block, lds_bytes, stream,
                x.data_ptr<float>(), w.data_ptr<float>(), bias_ptr, y.data_ptr<float>(), B, C, H, W, O, HO);
            break;
        default:
            TORCH_CHECK(false, "stem conv supports W_out in {15, 20}, got ", WO);
    }
    HIP_CHECK_LAST();
    return y;
}

torch::Tensor stem_conv_wrw(torch::Tensor x, torch::Tensor dy, int64_t O) {
    CHECK_INPUT(x);
    CHECK_INPUT(dy);
    int B = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
    int HO = dy.size(2), WO = dy.size(3);
    auto dw = torch::zeros({O, C, STEM_K, STEM_K}, x.options());
    hipLaunchKernelGGL(
        stem_conv_wrw_kernel, dim3(B), dim3(256), 0, ding_current_stream(),
        x.data_ptr<float>(), dy.data_ptr<float>(), dw.data_ptr<float>(), B, C, H, W, (int)O, HO, WO
    );
    HIP_CHECK_LAST();
    return dw;
}
