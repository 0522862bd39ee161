// NHWC conv2d backward-weights (wrw) for the Atari conv stack on CDNA4.
//
// MIOpen resolves NHWC fp32 wrw for these shapes to a naive fp64-accumulate
// solver (profiles/impala_nhwc_kernel_stats_r01.csv: ~0.7 ms/call, 20.5% of
// IMPALA kernel time). This kernel computes
//   dW[o, kh, kw, c] = sum_{b, ph, pw} dY[b, ph, pw, o] * X[b, ph*S+kh, pw*S+kw, c]
// directly in fp32:
//   - grid (n_chunk, kh, c_block); one workgroup owns one kh row of taps and
//     a C_sub channel slice, accumulating its dW tile in REGISTERS over its
//     share of (b, ph) pairs
//   - per (b, ph): the dY output row [WO, O] and the X input row
//     [W, C_sub] are staged in LDS once, then every (kw, oc, c) product
//     reuses them — dY is read from HBM once per kh (K times total) instead
//     of once per tap
//   - one atomicAdd per dW entry per workgroup at the end (dW tensors are
//     16-74 KB for these shapes: negligible contention)
// Accumulator arrays are compile-time sized (K x P) — runtime-bounded
// register arrays demote to scratch on gfx950 (measured 40x in round 1).
//
// Layouts are channels_last physical: X [B, H, W, C], dY [B, HO, WO, O],
// dW [O, K, K, C] (the memory layout of a channels_last conv weight).
#include "common.h"

template <int K, int P>
__global__ void conv_wrw_nhwc_kernel(
    const float* __restrict__ x,   // [B, H, W, C]
    const float* __restrict__ dy,  // [B, HO, WO, O]
    float* __restrict__ dw,        // [O, K, K, C]
    int B, int H, int W, int C,
    int HO, int WO, int O, int S,
    int c0, int C_sub               // channel slice [c0, c0 + C_sub)
) {
    extern __shared__ float lds[];
    float* dy_lds = lds;            // [WO * O]
    float* x_lds = lds + WO * O;    // [W * C_sub]

    const int kh = blockIdx.y;
    const int tid = threadIdx.x;

    // per-thread (oc, cc) ownership for each of the P register lanes
    int oc_of[P], cc_of[P];
    bool live[P];
#pragma unroll
    for (int j = 0; j < P; ++j) {
        int pair = tid + j * blockDim.x;
        live[j] = pair < O * C_sub;
        oc_of[j] = live[j] ? pair / C_sub : 0;
        cc_of[j] = live[j] ? pair % C_sub : 0;
    }

    float acc[K][P];
#pragma unroll
    for (int kw = 0; kw < K; ++kw)
#pragma unroll
        for (int j = 0; j < P; ++j) acc[kw][j] = 0.f;

    const int n_pairs = B * HO;
    for (int chunk = blockIdx.x; chunk < n_pairs; chunk += gridDim.x) {
        const int b = chunk / HO;
        const int ph = chunk % HO;
        const int h = ph * S + kh;
        if (h >= H) continue;
        // stage dY[b, ph, :, :] and X[b, h, :, c0:c0+C_sub]
        const float* dy_src = dy + (((int64_t)b * HO + ph) * WO) * O;
        for (int i = tid; i < WO * O; i += blockDim.x) dy_lds[i] = dy_src[i];
        const float* x_src = x + (((int64_t)b * H + h) * W) * C + c0;
        for (int i = tid; i < W * C_sub; i += blockDim.x) {
            int wq = i / C_sub, cq = i % C_sub;
            x_lds[i] = x_src[(int64_t)wq * C + cq];
        }
        __syncthreads();
        for (int pw = 0; pw < WO; ++pw) {
            const int base = pw * S;
#pragma unroll
            for (int j = 0; j < P; ++j) {
                const float dyv = dy_lds[pw * O + oc_of[j]];
#pragma unroll
                for (int kw = 0; kw < K; ++kw) {
                    acc[kw][j] = fmaf(dyv, x_lds[(base + kw) * C_sub + cc_of[j]], acc[kw][j]);
                }
            }
        }
        __syncthreads();
    }

#pragma unroll
    for (int kw = 0; kw < K; ++kw)
#pragma unroll
        for (int j = 0; j < P; ++j) {
            if (!live[j]) continue;
            int64_t idx = (((int64_t)oc_of[j] * K + kh) * K + kw) * C + c0 + cc_of[j];
            atomicAdd(&dw[idx], acc[kw][j]);
        }
}

// host-side dispatch: pick (K, P, c_split) for the shape
torch::Tensor conv_wrw_nhwc(torch::Tensor x, torch::Tensor dy, int64_t K_, int64_t S_) {
    TORCH_CHECK(x.is_cuda() && dy.is_cuda(), "conv_wrw_nhwc: GPU tensors required");
    // logical [B, C, H, W] in channels_last storage
    TORCH_CHECK(
        x.is_contiguous(at::MemoryFormat::ChannelsLast) || x.size(1) <= 4,
        "x must be channels_last"
    );
    auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
    auto dyc = dy.contiguous(at::MemoryFormat::ChannelsLast);
    int B = xc.size(0), C = xc.size(1), H = xc.size(2), W = xc.size(3);
    int O = dyc.size(1), HO = dyc.size(2), WO = dyc.size(3);
    int K = (int)K_, S = (int)S_;

    auto dw = torch::zeros({O, C, K, K}, x.options().memory_format(at::MemoryFormat::ChannelsLast));

    // choose the channel split so that P = O*C_sub/256 keeps K*P <= 96 regs
    int threads = 256;
    int c_split = 1;
    auto regs = [&](int cs) { return K * ((O * (C / cs) + threads - 1) / threads); };
    while (C % (c_split * 2) == 0 && regs(c_split) > 96) c_split *= 2;
    int C_sub = C / c_split;
    int P = (O * C_sub + threads - 1) / threads;

    int n_pairs = B * HO;
    int grid_x = std::min(n_pairs, std::max(2048 / (K * c_split), 1));
    dim3 grid(grid_x, K, c_split);
    size_t lds_bytes = (size_t)(WO * O + W * C_sub) * sizeof(float);
    TORCH_CHECK(lds_bytes <= 160 * 1024, "wrw LDS tile too large for this shape");

    bool launched = false;
#define WRW_CASE(KK, PP)                                                                           \
    if (!launched && K == KK && P == PP) {                                                         \
        for (int cz = 0; cz < c_split; ++cz) {                                                     \
            hipLaunchKernelGGL(                                                                    \
                (conv_wrw_nhwc_kernel<KK, PP>), dim3(grid_x, K, 1), dim3(threads), lds_bytes,      \
                ding_current_stream(), xc.data_ptr<float>(), dyc.data_ptr<float>(),                \
                dw.data_ptr<float>(), B, H, W, C, HO, WO, O, S, cz * C_sub, C_sub                  \
            );                                                                                     \
        }                                                                                          \
        launched = true;                                                                           \
    }
    WRW_CASE(8, 1)   // pong conv1: 4ch -> 64
    WRW_CASE(8, 2)   // impala conv1: 4ch -> 128
    WRW_CASE(4, 16)  // pong conv2: 64 -> 64
    WRW_CASE(4, 8)
    WRW_CASE(4, 32)  // impala conv2 (c_split 2): 128 -> 128
    WRW_CASE(3, 32)  // pong conv3: 64 -> 128
    WRW_CASE(3, 16)
    WRW_CASE(3, 8)
    WRW_CASE(8, 4)
    WRW_CASE(4, 4)
    WRW_CASE(3, 4)
    WRW_CASE(8, 8)
#undef WRW_CASE
    TORCH_CHECK(launched, "conv_wrw_nhwc: no kernel instance for K=", K, " P=", P);
    HIP_CHECK_LAST();
    return dw;
}
