// Backward kernels for the fused MeanPool layer (training path).
//
// Gradient flow (see ddls_amd/models/gnn.py MeanPoolLayer):
//   out[v] = mean over {self} u {in-edges} of relu(LN(m) @ Wr^T + br)
//   m_edge = [hn[src] || he],  m_self = [hn[v] || 0]
//   hn = relu(LN(z) @ Wn^T + bn),  he likewise.
//
// message_reduce_bwd: one wave per message (edge messages then self
// messages); recomputes the message LayerNorm from hn/he, uses the stored
// activations r for the relu mask, and accumulates
//   ghn (atomic scatter over src / self), ghe, and the reduce-module
//   parameter grads (per-wave private LDS slices, flushed once per block
//   with global atomics).
// row_mlp_bwd: one wave per row; produces gx and the node/edge-module
// parameter grads the same way.
//
// fp32 like the torch training path; parity asserted at 1e-3 in
// tests/test_models.py (gpu).

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <vector>

#define WAVE_B 64
#define WPB_B 4                   // waves per block
#define BLOCK_B (WAVE_B * WPB_B)
#define LN_EPS_B 1e-5f

__device__ __forceinline__ float wave_sum_b(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        v += __shfl_down(v, off, WAVE_B);
    return __shfl(v, 0, WAVE_B);
}

// ---------------------------------------------------------------------------
// message_reduce_bwd
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(BLOCK_B)
message_reduce_bwd_kernel(const float* __restrict__ hn,
                          const float* __restrict__ he,
                          const long* __restrict__ src,
                          const long* __restrict__ dst,
                          const long* __restrict__ indptr,  // in-deg CSR [N+1]
                          const float* __restrict__ ln_g,
                          const float* __restrict__ ln_b,
                          const float* __restrict__ Wr,     // [OUT][MSG]
                          const float* __restrict__ r_edge, // [E][OUT]
                          const float* __restrict__ r_self, // [N][OUT]
                          const float* __restrict__ gout,   // [N][OUT]
                          float* __restrict__ ghn,          // [N][half] (atomic)
                          float* __restrict__ ghe,          // [E][half]
                          float* __restrict__ gWr,          // [OUT][MSG] (atomic)
                          float* __restrict__ gbr,          // [OUT] (atomic)
                          float* __restrict__ gln_g,        // [MSG] (atomic)
                          float* __restrict__ gln_b,        // [MSG] (atomic)
                          int N, int E, int half, int OUT) {
    const int MSG = 2 * half;
    // dynamic LDS only (a static __shared__ would shift the dynamic base)
    extern __shared__ float dyn_s[];
    // dynamic LDS layout (per block):
    //   [0, OUT*MSG)                     : staged Wr
    //   + w*OUT*MSG .. per wave          : private gWr slice
    //   then per-wave staging: yln[MSG], mhat[MSG], ga[OUT], small grads
    float* WrS = dyn_s;
    float* gWrW = dyn_s + OUT * MSG + threadIdx.x / WAVE_B * (OUT * MSG);
    float* stage = dyn_s + OUT * MSG * (1 + WPB_B)
                 + (threadIdx.x / WAVE_B) * (3 * 64 + 4 * 64);
    float* ylnS = stage;            // [64]
    float* mhatS = stage + 64;      // [64]
    float* gaS = stage + 128;       // [64]
    float* gbrW = stage + 192;      // [64]
    float* glngW = stage + 256;     // [64]
    float* glnbW = stage + 320;     // [64]
    float* gylnS = stage + 384;     // [64]

    const int tid = threadIdx.x;
    const int lane = tid % WAVE_B;
    const int wave = tid / WAVE_B;
    for (int i = tid; i < OUT * MSG; i += BLOCK_B) WrS[i] = Wr[i];
    for (int i = lane; i < OUT * MSG; i += WAVE_B) gWrW[i] = 0.0f;
    if (lane < 64) {
        gbrW[lane] = 0.0f;
        glngW[lane] = 0.0f;
        glnbW[lane] = 0.0f;
    }
    __syncthreads();

    const long M = (long)E + N;
    const long m0 = (long)blockIdx.x * WPB_B + wave;
    const long mstride = (long)gridDim.x * WPB_B;
    for (long mid = m0; mid < M; mid += mstride) {
        long node, eid = -1;
        float mv;  // this lane's message element
        if (mid < E) {
            eid = mid;
            node = dst[eid];
            const long s = src[eid];
            if (lane < half) mv = hn[s * half + lane];
            else if (lane < MSG) mv = he[eid * half + (lane - half)];
            else mv = 0.0f;
        } else {
            node = mid - E;
            mv = (lane < half) ? hn[node * half + lane] : 0.0f;
        }
        const long deg = indptr[node + 1] - indptr[node];
        if (deg == 0) continue;  // zero-filled node: no gradient
        const float scale = 1.0f / (float)(deg + 1);

        // recompute LN stats
        const float inr = (lane < MSG) ? 1.0f : 0.0f;
        const float mean = wave_sum_b(mv * inr) / MSG;
        const float dv = (lane < MSG) ? (mv - mean) : 0.0f;
        const float var = wave_sum_b(dv * dv) / MSG;
        const float inv_sigma = rsqrtf(var + LN_EPS_B);
        const float mhat = dv * inv_sigma;
        if (lane < MSG) {
            mhatS[lane] = mhat;
            ylnS[lane] = mhat * ln_g[lane] + ln_b[lane];
        }
        __builtin_amdgcn_wave_barrier();

        // ga[j] = relu'(a) * gout[node][j] / (deg+1)
        float ga = 0.0f;
        if (lane < OUT) {
            const float r = (eid >= 0) ? r_edge[eid * OUT + lane]
                                       : r_self[node * OUT + lane];
            if (r > 0.0f)
                ga = gout[node * OUT + lane] * scale;
            gaS[lane] = ga;
            gbrW[lane] += ga;
            // private gWr row j accumulation
            float* row = gWrW + lane * MSG;
            for (int k = 0; k < MSG; ++k)
                row[k] = fmaf(ga, ylnS[k], row[k]);
        }
        __builtin_amdgcn_wave_barrier();

        // gyln[k] = sum_j ga[j] * Wr[j][k]
        float gyln = 0.0f;
        if (lane < MSG) {
            for (int j = 0; j < OUT; ++j)
                gyln = fmaf(gaS[j], WrS[j * MSG + lane], gyln);
            glngW[lane] += gyln * mhat;
            glnbW[lane] += gyln;
            gylnS[lane] = gyln * ln_g[lane];  // gmh
        }
        __builtin_amdgcn_wave_barrier();
        // gm = (gmh - mean(gmh) - mhat * mean(gmh*mhat)) * inv_sigma
        const float gmh = (lane < MSG) ? gylnS[lane] : 0.0f;
        const float mean_gmh = wave_sum_b(gmh) / MSG;
        const float mean_gmh_mhat = wave_sum_b(gmh * mhat) / MSG;
        const float gm = (gmh - mean_gmh - mhat * mean_gmh_mhat) * inv_sigma;

        if (mid < E) {
            if (lane < half)
                atomicAdd(&ghn[src[eid] * half + lane], gm);
            else if (lane < MSG)
                ghe[eid * half + (lane - half)] = gm;
        } else {
            if (lane < half)
                atomicAdd(&ghn[node * half + lane], gm);
            // grads into the zero padding are discarded
        }
        __builtin_amdgcn_wave_barrier();
    }

    // flush per-wave accumulators
    __syncthreads();
    for (int i = tid; i < OUT * MSG; i += BLOCK_B) {
        float acc = 0.0f;
        for (int w = 0; w < WPB_B; ++w)
            acc += dyn_s[OUT * MSG * (1 + w) + i];
        if (acc != 0.0f) atomicAdd(&gWr[i], acc);
    }
    if (tid < 64) {
        float b_acc = 0.0f, g_acc = 0.0f, bb_acc = 0.0f;
        for (int w = 0; w < WPB_B; ++w) {
            const float* st = dyn_s + OUT * MSG * (1 + WPB_B)
                            + w * (3 * 64 + 4 * 64);
            b_acc += st[192 + tid];
            g_acc += st[256 + tid];
            bb_acc += st[320 + tid];
        }
        if (tid < OUT && b_acc != 0.0f) atomicAdd(&gbr[tid], b_acc);
        if (tid < MSG) {
            if (g_acc != 0.0f) atomicAdd(&gln_g[tid], g_acc);
            if (bb_acc != 0.0f) atomicAdd(&gln_b[tid], bb_acc);
        }
    }
}

// ---------------------------------------------------------------------------
// row_mlp_bwd: y = relu(LN(x) @ W^T + b);  given gy and stored y -> gx + grads
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(BLOCK_B)
row_mlp_bwd_kernel(const float* __restrict__ x,     // [R][F]
                   const float* __restrict__ y,     // [R][H] stored activations
                   const float* __restrict__ gy,    // [R][H]
                   const float* __restrict__ ln_g,
                   const float* __restrict__ ln_b,
                   const float* __restrict__ W,     // [H][F]
                   float* __restrict__ gx,          // [R][F]
                   float* __restrict__ gW,          // [H][F] (atomic)
                   float* __restrict__ gb,          // [H] (atomic)
                   float* __restrict__ gln_g,       // [F] (atomic)
                   float* __restrict__ gln_b,       // [F] (atomic)
                   int R, int F, int H) {
    extern __shared__ float dyn_s[];
    // layout: staged W [H*F]; per-wave private gW [H*F]; per-wave staging
    float* WS = dyn_s;
    float* gWW = dyn_s + H * F + (threadIdx.x / WAVE_B) * (H * F);
    float* stage = dyn_s + H * F * (1 + WPB_B)
                 + (threadIdx.x / WAVE_B) * (5 * 64);
    float* preS = stage;          // [64]
    float* ylnS = stage + 64;     // [64]
    float* gbW = stage + 128;     // [64]
    float* glngW = stage + 192;   // [64]
    float* glnbW = stage + 256;   // [64]

    const int tid = threadIdx.x;
    const int lane = tid % WAVE_B;
    const int wave = tid / WAVE_B;
    for (int i = tid; i < H * F; i += BLOCK_B) WS[i] = W[i];
    for (int i = lane; i < H * F; i += WAVE_B) gWW[i] = 0.0f;
    if (lane < 64) {
        gbW[lane] = 0.0f;
        glngW[lane] = 0.0f;
        glnbW[lane] = 0.0f;
    }
    __syncthreads();

    const long r0 = (long)blockIdx.x * WPB_B + wave;
    const long rstride = (long)gridDim.x * WPB_B;
    for (long row = r0; row < R; row += rstride) {
        const float xv = (lane < F) ? x[row * F + lane] : 0.0f;
        const float inr = (lane < F) ? 1.0f : 0.0f;
        const float mean = wave_sum_b(xv * inr) / F;
        const float dv = (lane < F) ? (xv - mean) : 0.0f;
        const float var = wave_sum_b(dv * dv) / F;
        const float inv_sigma = rsqrtf(var + LN_EPS_B);
        const float xhat = dv * inv_sigma;
        if (lane < F)
            ylnS[lane] = xhat * ln_g[lane] + ln_b[lane];
        float pre = 0.0f;
        if (lane < H) {
            const float yv = y[row * H + lane];
            if (yv > 0.0f) pre = gy[row * H + lane];
            preS[lane] = pre;
            gbW[lane] += pre;
        } else if (lane < 64) {
            preS[lane] = 0.0f;
        }
        __builtin_amdgcn_wave_barrier();
        if (lane < H && pre != 0.0f) {
            float* rowW = gWW + lane * F;
            for (int k = 0; k < F; ++k)
                rowW[k] = fmaf(pre, ylnS[k], rowW[k]);
        }
        // gyln[k] = sum_j pre[j] W[j][k]
        float gyln = 0.0f;
        if (lane < F) {
            for (int j = 0; j < H; ++j)
                gyln = fmaf(preS[j], WS[j * F + lane], gyln);
            glngW[lane] += gyln * xhat;
            glnbW[lane] += gyln;
        }
        const float gxh = (lane < F) ? gyln * ln_g[lane] : 0.0f;
        const float mean_g = wave_sum_b(gxh) / F;
        const float mean_gx = wave_sum_b(gxh * xhat) / F;
        if (lane < F)
            gx[row * F + lane] = (gxh - mean_g - xhat * mean_gx) * inv_sigma;
        __builtin_amdgcn_wave_barrier();
    }

    __syncthreads();
    for (int i = tid; i < H * F; i += BLOCK_B) {
        float acc = 0.0f;
        for (int w = 0; w < WPB_B; ++w)
            acc += dyn_s[H * F * (1 + w) + i];
        if (acc != 0.0f) atomicAdd(&gW[i], acc);
    }
    if (tid < 64) {
        float b_acc = 0.0f, g_acc = 0.0f, bb_acc = 0.0f;
        for (int w = 0; w < WPB_B; ++w) {
            const float* st = dyn_s + H * F * (1 + WPB_B) + w * (5 * 64);
            b_acc += st[128 + tid];
            g_acc += st[192 + tid];
            bb_acc += st[256 + tid];
        }
        if (tid < H && b_acc != 0.0f) atomicAdd(&gb[tid], b_acc);
        if (tid < F) {
            if (g_acc != 0.0f) atomicAdd(&gln_g[tid], g_acc);
            if (bb_acc != 0.0f) atomicAdd(&gln_b[tid], bb_acc);
        }
    }
}

// ---------------------------------------------------------------------------
// host wrappers
// ---------------------------------------------------------------------------

std::vector<torch::Tensor> message_reduce_bwd(
        torch::Tensor hn, torch::Tensor he, torch::Tensor src,
        torch::Tensor dst, torch::Tensor indptr, torch::Tensor ln_g,
        torch::Tensor ln_b, torch::Tensor Wr, torch::Tensor r_edge,
        torch::Tensor r_self, torch::Tensor gout) {
    const int N = hn.size(0), half = hn.size(1), E = he.size(0);
    const int OUT = Wr.size(0), MSG = 2 * half;
    TORCH_CHECK(MSG <= 64 && OUT <= 64);
    auto ghn = torch::zeros_like(hn);
    auto ghe = torch::zeros_like(he);
    auto gWr = torch::zeros_like(Wr);
    auto gbr = torch::zeros({OUT}, Wr.options());
    auto gln_g = torch::zeros({MSG}, Wr.options());
    auto gln_b = torch::zeros({MSG}, Wr.options());
    const long M = (long)E + N;
    int blocks = std::min<long>((M + WPB_B - 1) / WPB_B, 512);
    if (blocks == 0) blocks = 1;
    size_t lds = sizeof(float) * (OUT * MSG * (1 + WPB_B)
                                  + WPB_B * (3 * 64 + 4 * 64));
    hipStream_t stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(message_reduce_bwd_kernel, dim3(blocks), dim3(BLOCK_B),
                       lds, stream,
                       hn.data_ptr<float>(), he.data_ptr<float>(),
                       src.data_ptr<long>(), dst.data_ptr<long>(),
                       indptr.data_ptr<long>(), ln_g.data_ptr<float>(),
                       ln_b.data_ptr<float>(), Wr.data_ptr<float>(),
                       r_edge.data_ptr<float>(), r_self.data_ptr<float>(),
                       gout.data_ptr<float>(), ghn.data_ptr<float>(),
                       ghe.data_ptr<float>(), gWr.data_ptr<float>(),
                       gbr.data_ptr<float>(), gln_g.data_ptr<float>(),
                       gln_b.data_ptr<float>(), N, E, half, OUT);
    return {ghn, ghe, gWr, gbr, gln_g, gln_b};
}

std::vector<torch::Tensor> row_mlp_bwd(torch::Tensor x, torch::Tensor y,
                                       torch::Tensor gy, torch::Tensor ln_g,
                                       torch::Tensor ln_b, torch::Tensor W) {
    const int R = x.size(0), F = x.size(1), H = W.size(0);
    TORCH_CHECK(F <= 64 && H <= 64);
    auto gx = torch::zeros_like(x);
    auto gW = torch::zeros_like(W);
    auto gb = torch::zeros({H}, W.options());
    auto gln_g = torch::zeros({F}, W.options());
    auto gln_b = torch::zeros({F}, W.options());
    int blocks = std::min((R + WPB_B - 1) / WPB_B, 512);
    if (blocks == 0) blocks = 1;
    size_t lds = sizeof(float) * (H * F * (1 + WPB_B) + WPB_B * (5 * 64));
    hipStream_t stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(row_mlp_bwd_kernel, dim3(blocks), dim3(BLOCK_B), lds,
                       stream, x.data_ptr<float>(), y.data_ptr<float>(),
                       gy.data_ptr<float>(), ln_g.data_ptr<float>(),
                       ln_b.data_ptr<float>(), W.data_ptr<float>(),
                       gx.data_ptr<float>(), gW.data_ptr<float>(),
                       gb.data_ptr<float>(), gln_g.data_ptr<float>(),
                       gln_b.data_ptr<float>(), R, F, H);
    return {gx, gW, gb, gln_g, gln_b};
}
