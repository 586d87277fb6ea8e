// Fused PPO loss forward + analytic backward for gfx950.
//
// Inside the hipGraph-captured SGD step the clipped-surrogate loss chain
// (log_softmax, gather, exp, min/clamp, four means, entropy, and all their
// autograd nodes) is ~100 tiny elementwise kernels per replay — more GPU
// time than the GNN itself.  This pair replaces the whole chain with one
// kernel each way; gradients are the closed forms of the RLlib-style loss
//   loss = -mean(surr) + kl_coef*mean(old - logp_a)
//          + vf_coef*mean(clamp((v-t)^2, 0, vf_clip)) - ent_coef*mean(H)
// with surr = min(ratio*A, clamp(ratio, 1-c, 1+c)*A).  d surr/d logp_a is
// ratio*A except when the CLAMPED branch is strictly smaller and ratio sits
// outside the clip window (then 0) — at ties torch.min splits the gradient
// but both branches coincide, so the closed form matches autograd.
// kl_coef is read from a device scalar at run time (capture-safe adaptive
// KL).
#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <vector>

#define WAVE 64
#define WAVES 4
#define BLOCK (WAVE * WAVES)

__device__ __forceinline__ float wave_max_l(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        v = fmaxf(v, __shfl_down(v, off, WAVE));
    return __shfl(v, 0, WAVE);
}

__device__ __forceinline__ float wave_sum_l(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        v += __shfl_down(v, off, WAVE);
    return __shfl(v, 0, WAVE);
}

// one wave per row; block accumulates the five loss terms.
// outs: loss[1], stats[5] = {policy_loss, vf_loss, kl, entropy, total}
__global__ void __launch_bounds__(BLOCK)
ppo_loss_fwd_kernel(const float* __restrict__ logits,
                    const float* __restrict__ values,
                    const long* __restrict__ actions,
                    const float* __restrict__ old_logp,
                    const float* __restrict__ adv,
                    const float* __restrict__ vtarg,
                    const float* __restrict__ kl_coef,
                    float* __restrict__ p_out,      // [B, A]
                    float* __restrict__ lp_out,     // [B, A]
                    float* __restrict__ coef_out,   // [B] d loss/d logp_a * B
                    float* __restrict__ h_out,      // [B] entropy per row
                    float* __restrict__ loss_out,   // [1]
                    float* __restrict__ stats_out,  // [5]
                    int B, int A, float clip, float vf_clip, float vf_coef,
                    float ent_coef) {
    __shared__ float acc[WAVES][4];  // surr, kl, vf, ent partial sums
    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    float s_surr = 0.0f, s_kl = 0.0f, s_vf = 0.0f, s_ent = 0.0f;
    for (int b = blockIdx.x * WAVES + wave; b < B;
         b += gridDim.x * WAVES) {
        const float x = (lane < A) ? logits[(long)b * A + lane] : -3.0e38f;
        const float mx = wave_max_l(x);
        const float ex = (lane < A) ? __expf(x - mx) : 0.0f;
        const float Z = wave_sum_l(ex);
        const float lse = mx + __logf(Z);
        const float lp = x - lse;          // log-prob (valid for lane < A)
        const float p = ex / Z;
        if (lane < A) {
            p_out[(long)b * A + lane] = p;
            lp_out[(long)b * A + lane] = lp;
        }
        const float ent = -wave_sum_l((lane < A) ? p * lp : 0.0f);
        const long a = actions[b];
        const float logp_a = __shfl(lp, (int)a, WAVE);
        const float ratio = __expf(logp_a - old_logp[b]);
        const float A_b = adv[b];
        const float r_cl = fminf(fmaxf(ratio, 1.0f - clip), 1.0f + clip);
        const float surr1 = ratio * A_b, surr2 = r_cl * A_b;
        const float surr = fminf(surr1, surr2);
        // d surr/d logp_a (see header)
        float ds = ratio * A_b;
        if (surr2 < surr1 && (ratio < 1.0f - clip || ratio > 1.0f + clip))
            ds = 0.0f;
        const float kl_b = old_logp[b] - logp_a;
        const float verr = values[b] - vtarg[b];
        const float v2 = verr * verr;
        const float vf_b = fminf(v2, vf_clip);
        if (lane == 0) {
            s_surr += surr;
            s_kl += kl_b;
            s_vf += vf_b;
            s_ent += ent;
            h_out[b] = ent;
            // d loss/d logp_a, times B (the means divide by B in backward)
            coef_out[b] = -ds - kl_coef[0];
        }
    }
    if (lane == 0) {
        acc[wave][0] = s_surr;
        acc[wave][1] = s_kl;
        acc[wave][2] = s_vf;
        acc[wave][3] = s_ent;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
        float t[4] = {0, 0, 0, 0};
        for (int w = 0; w < WAVES; ++w)
            for (int i = 0; i < 4; ++i) t[i] += acc[w][i];
        // raw sums; the finalize kernel turns them into means + total loss
        atomicAdd(&stats_out[0], -t[0]);
        atomicAdd(&stats_out[1], t[2]);
        atomicAdd(&stats_out[2], t[1]);
        atomicAdd(&stats_out[3], t[3]);
    }
}

__global__ void ppo_loss_finalize_kernel(const float* __restrict__ kl_coef,
                                         float* __restrict__ loss_out,
                                         float* __restrict__ stats_out,
                                         int B, float vf_coef,
                                         float ent_coef) {
    const float inv = 1.0f / B;
    const float pl = stats_out[0] * inv;
    const float vf = stats_out[1] * inv;
    const float kl = stats_out[2] * inv;
    const float ent = stats_out[3] * inv;
    const float loss = pl + kl_coef[0] * kl + vf_coef * vf - ent_coef * ent;
    stats_out[0] = pl;
    stats_out[1] = vf;
    stats_out[2] = kl;
    stats_out[3] = ent;
    stats_out[4] = loss;
    loss_out[0] = loss;
}

__global__ void __launch_bounds__(BLOCK)
ppo_loss_bwd_kernel(const float* __restrict__ p,
                    const float* __restrict__ lp,
                    const float* __restrict__ coef,
                    const float* __restrict__ h,
                    const long* __restrict__ actions,
                    const float* __restrict__ values,
                    const float* __restrict__ vtarg,
                    const float* __restrict__ gl,   // [1] upstream dloss
                    float* __restrict__ glogits,    // [B, A]
                    float* __restrict__ gvalues,    // [B]
                    int B, int A, float vf_clip, float vf_coef,
                    float ent_coef) {
    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const float g = gl[0];
    for (int b = blockIdx.x * WAVES + wave; b < B;
         b += gridDim.x * WAVES) {
        const long a = actions[b];
        const float c = coef[b] / B;
        const float hb = h[b];
        if (lane < A) {
            const float pj = p[(long)b * A + lane];
            const float lpj = lp[(long)b * A + lane];
            const float delta = (lane == (int)a) ? 1.0f : 0.0f;
            // d[-ec*H]/dlogits = ec/B * p*(lp + H)
            glogits[(long)b * A + lane] =
                g * (c * (delta - pj)
                     + (ent_coef / B) * pj * (lpj + hb));
        }
        if (lane == 0) {
            const float verr = values[b] - vtarg[b];
            const float mask = (verr * verr <= vf_clip) ? 1.0f : 0.0f;
            gvalues[b] = g * vf_coef * 2.0f * verr * mask / B;
        }
    }
}

std::vector<torch::Tensor> ppo_loss_fwd(
    torch::Tensor logits, torch::Tensor values, torch::Tensor actions,
    torch::Tensor old_logp, torch::Tensor adv, torch::Tensor vtarg,
    torch::Tensor kl_coef, double clip, double vf_clip, double vf_coef,
    double ent_coef) {
    TORCH_CHECK(logits.is_cuda() && logits.dtype() == torch::kFloat32);
    const int B = (int)logits.size(0), A = (int)logits.size(1);
    TORCH_CHECK(A <= 64, "ppo_loss_fwd needs A <= 64 (one wave per row)");
    auto opt = logits.options();
    auto p = torch::empty({B, A}, opt);
    auto lp = torch::empty({B, A}, opt);
    auto coef = torch::empty({B}, opt);
    auto h = torch::empty({B}, opt);
    auto loss = torch::zeros({1}, opt);
    auto stats = torch::zeros({5}, opt);
    hipStream_t stream = at::cuda::getCurrentCUDAStream();
    int blocks = (B + WAVES - 1) / WAVES;
    if (blocks > 128) blocks = 128;
    hipLaunchKernelGGL(ppo_loss_fwd_kernel, dim3(blocks), dim3(BLOCK), 0,
                       stream,
                       logits.data_ptr<float>(), values.data_ptr<float>(),
                       actions.data_ptr<long>(), old_logp.data_ptr<float>(),
                       adv.data_ptr<float>(), vtarg.data_ptr<float>(),
                       kl_coef.data_ptr<float>(), p.data_ptr<float>(),
                       lp.data_ptr<float>(), coef.data_ptr<float>(),
                       h.data_ptr<float>(), loss.data_ptr<float>(),
                       stats.data_ptr<float>(), B, A, (float)clip,
                       (float)vf_clip, (float)vf_coef, (float)ent_coef);
    hipLaunchKernelGGL(ppo_loss_finalize_kernel, dim3(1), dim3(1), 0, stream,
                       kl_coef.data_ptr<float>(), loss.data_ptr<float>(),
                       stats.data_ptr<float>(), B, (float)vf_coef,
                       (float)ent_coef);
    return {loss, stats, p, lp, coef, h};
}

std::vector<torch::Tensor> ppo_loss_bwd(
    torch::Tensor p, torch::Tensor lp, torch::Tensor coef, torch::Tensor h,
    torch::Tensor actions, torch::Tensor values, torch::Tensor vtarg,
    torch::Tensor gl, double vf_clip, double vf_coef, double ent_coef) {
    const int B = (int)p.size(0), A = (int)p.size(1);
    auto glogits = torch::empty({B, A}, p.options());
    auto gvalues = torch::empty({B}, p.options());
    hipStream_t stream = at::cuda::getCurrentCUDAStream();
    int blocks = (B + WAVES - 1) / WAVES;
    if (blocks > 1024) blocks = 1024;
    hipLaunchKernelGGL(ppo_loss_bwd_kernel, dim3(blocks), dim3(BLOCK), 0,
                       stream, p.data_ptr<float>(), lp.data_ptr<float>(),
                       coef.data_ptr<float>(), h.data_ptr<float>(),
                       actions.data_ptr<long>(), values.data_ptr<float>(),
                       vtarg.data_ptr<float>(), gl.data_ptr<float>(),
                       glogits.data_ptr<float>(), gvalues.data_ptr<float>(),
                       B, A, (float)vf_clip, (float)vf_coef, (float)ent_coef);
    return {glogits, gvalues};
}
