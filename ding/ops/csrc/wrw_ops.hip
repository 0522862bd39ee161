// NHWC conv2d backward-weights (wrw) for the Atari conv stack on CDNA4.
//
// dW[o, kh, kw, c] = sum_{b, ph, pw} dY[b, ph, pw, o] * X[b, ph*S+kh, pw*S+kw, c]
//
// Design (fp32, no MFMA — CDNA4 matrix cores have no fp32 mode):
//   - grid (n_chunk, kh, c_block): a workgroup owns one kh row of taps and a
//     C_sub channel slice, accumulating its dW tile in REGISTERS over its
//     share of (b, ph) pairs
//   - per (b, ph): dY row [WO, O] and X row [W, C_sub] staged in LDS once;
//     the inner loop hoists the P dY values (j-varying) and the K X values
//     (kw-varying) so each FMA costs ~(P + K) / (P * K) LDS reads
//   - C_sub always divides the block size, so each thread's channel
//     cc = tid % C_sub is the same for every register lane j — X loads hoist
//     out of the j loop
//   - NO atomics: each workgroup writes its partial tile to
//     partial[chunk, O*K*K*C]; a single torch sum over dim 0 finishes the
//     reduction (the zero-filled buffer costs one hipMemset)
// Accumulators are compile-time sized (K x P): runtime-bounded register
// arrays demote to scratch on gfx950 (measured 40x in round 1).
//
// Physical layouts: X [B, H, W, C], dY [B, HO, WO, O] (channels_last storage
// of the logical NCHW tensors); dW partials are written physically
// [O, K, K, C] and returned as a permuted view of logical [O, C, K, K].
#include "common.h"

template <int K, int P>
__global__ void conv_wrw_nhwc_kernel(
    const float* __restrict__ x,        // [B, H, W, C]
    const float* __restrict__ dy,       // [B, HO, WO, O]
    float* __restrict__ partial,        // [grid_x, O*K*K*C]
    int B, int H, int W, int C,
    int HO, int WO, int O, int S,
    int c0, int C_sub
) {
    extern __shared__ float lds[];
    float* dy_lds = lds;                // [WO * O]
    float* x_lds = lds + WO * O;        // [W * C_sub]

    const int kh = blockIdx.y;
    const int tid = threadIdx.x;
    const int cc = tid % C_sub;         // j-invariant: blockDim % C_sub == 0
    int oc_of[P];
    bool live[P];
#pragma unroll
    for (int j = 0; j < P; ++j) {
        int pair = tid + j * blockDim.x;
        live[j] = pair < O * C_sub;
        oc_of[j] = live[j] ? pair / C_sub : 0;
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
        if (h < H) {
            const float* dy_src = dy + (((int64_t)b * HO + ph) * WO) * O;
            for (int i = tid; i < WO * O; i += blockDim.x) dy_lds[i] = dy_src[i];
            const float* x_src = x + (((int64_t)b * H + h) * W) * C + c0;
            for (int i = tid; i < W * C_sub; i += blockDim.x) {
                x_lds[i] = x_src[(int64_t)(i / C_sub) * C + (i % C_sub)];
            }
            __syncthreads();
            for (int pw = 0; pw < WO; ++pw) {
                float dyv[P];
#pragma unroll
                for (int j = 0; j < P; ++j) dyv[j] = dy_lds[pw * O + oc_of[j]];
                const int base = pw * S;
#pragma unroll
                for (int kw = 0; kw < K; ++kw) {
                    const float xv = x_lds[(base + kw) * C_sub + cc];
#pragma unroll
                    for (int j = 0; j < P; ++j) acc[kw][j] = fmaf(dyv[j], xv, acc[kw][j]);
                }
            }
            __syncthreads();
        }
    }

    float* out = partial + (int64_t)blockIdx.x * O * K * K * C;
#pragma unroll
    for (int kw = 0; kw < K; ++kw)
#pragma unroll
        for (int j = 0; j < P; ++j) {
            if (!live[j]) continue;
            out[(((int64_t)oc_of[j] * K + kh) * K + kw) * C + c0 + cc] = acc[kw][j];
        }
}

torch::Tensor conv_wrw_nhwc(torch::Tensor x, torch::Tensor dy, int64_t K_, int64_t S_) {
    TORCH_CHECK(x.is_cuda() && dy.is_cuda(), "conv_wrw_nhwc: GPU tensors required");
    auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
    auto dyc = dy.contiguous(at::MemoryFormat::ChannelsLast);
    int B = xc.size(0), C = xc.size(1), H = xc.size(2), W = xc.size(3);
    int O = dyc.size(1), HO = dyc.size(2), WO = dyc.size(3);
    int K = (int)K_, S = (int)S_;
    const int threads = 256;

    // channel split: P = O*C_sub/threads with K*P bounded (register budget)
    int c_split = 1;
    auto regs = [&](int cs) { return K * ((O * (C / cs) + threads - 1) / threads); };
    while (C % (c_split * 2) == 0 && regs(c_split) > 64) c_split *= 2;
    int C_sub = C / c_split;
    int P = (O * C_sub + threads - 1) / threads;
    TORCH_CHECK(threads % C_sub == 0 || C_sub > threads,
                "conv_wrw_nhwc: C_sub must divide the block size (C=", C, ")");

    int n_pairs = B * HO;
    // ~512 workgroups total fills 256 CUs without exploding the partials
    int grid_x = std::max(std::min(n_pairs, 512 / (K * c_split)), 1);
    size_t lds_bytes = (size_t)(WO * O + W * C_sub) * sizeof(float);
    TORCH_CHECK(lds_bytes <= 160 * 1024, "conv_wrw_nhwc: LDS tile too large");

    int64_t entries = (int64_t)O * K * K * C;
    auto partial = torch::zeros({grid_x, entries}, x.options());

    bool launched = false;
#define WRW_CASE(KK, PP)                                                                           \
    if (!launched && K == KK && P == PP) {                                                         \
        for (int cz = 0; cz < c_split; ++cz) {                                                     \
            hipLaunchKernelGGL(                                                                    \
                (conv_wrw_nhwc_kernel<KK, PP>), dim3(grid_x, K, 1), dim3(threads), lds_bytes,      \
                ding_current_stream(), xc.data_ptr<float>(), dyc.data_ptr<float>(),                \
                partial.data_ptr<float>(), B, H, W, C, HO, WO, O, S, cz * C_sub, C_sub             \
            );                                                                                     \
        }                                                                                          \
        launched = true;                                                                           \
    }
    WRW_CASE(8, 1)   // pong conv1: 4ch -> 64
    WRW_CASE(8, 2)   // impala conv1: 4ch -> 128
    WRW_CASE(4, 16)  // pong conv2: 64 -> 64 / impala conv2 split
    WRW_CASE(4, 8)
    WRW_CASE(4, 32)
    WRW_CASE(3, 32)  // pong conv3: 64 -> 128 / impala conv3 split
    WRW_CASE(3, 16)
    WRW_CASE(3, 8)
    WRW_CASE(8, 4)
    WRW_CASE(4, 4)
    WRW_CASE(3, 4)
    WRW_CASE(8, 8)
#undef WRW_CASE
    TORCH_CHECK(launched, "conv_wrw_nhwc: no kernel instance for K=", K, " P=", P);
    HIP_CHECK_LAST();
    // reduce partials, then view the physical [O, K, K, C] buffer as the
    // logical [O, C, K, K] weight grad (strides == channels_last weight)
    auto dw_phys = partial.sum(0).view({O, K, K, C});
    return dw_phys.permute({0, 3, 1, 2});
}
