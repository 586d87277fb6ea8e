// Batched RAMP lookahead: the discrete-event tick loop of
// RampClusterEnvironment._run_lookahead as ONE persistent HIP kernel over a
// batch of mounted jobs — the north-star "step() conflict-resolution loop as
// batched CDNA4 kernels over vectorised envs" (SURVEY.md K3).
//
// One 256-thread workgroup per env/job.  Per tick (all barriers
// workgroup-wide, every thread of the group converges each iteration):
//   1. segmented argmax: highest-priority READY op per worker
//      (packed (prio+1)<<32|op u64 atomicMax into a per-env scratch table)
//   2. segmented min-reduce: shortest remaining time among priority ops
//   3. ready non-flow dep detection; else per-channel priority flow argmax +
//      min remaining comm time
//   4. masked decrement of deps (pre-op snapshot semantics) then priority
//      ops; completion detection feeds the DAG wavefront (ready-set update)
//   5. overhead + active-worker accounting
// until all ops and deps of the env's job are complete.
//
// All time arithmetic is fp64 with the exact operation order of the CPU
// reference (min/subtract only -> bitwise-identical results; the parity test
// asserts exact equality).

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#define LK_BLOCK 256

// all-int64 so the host can pack it as a [B, 11] int64 tensor
struct EnvDesc {
    long op_off;        // offset into op arrays
    long dep_off;       // offset into dep arrays
    long csr_off;       // offset into out_csr_edges
    long indptr_off;    // offset into out_indptr (n_ops+1 entries per env)
    long worker_off;    // offset into worker scratch
    long channel_off;   // offset into channel scratch
    long n_ops;
    long n_deps;
    long n_workers;     // local (remapped) worker count
    long n_channels;    // local (remapped) channel count
    long max_ticks;
};
#define ENV_DESC_LONGS 11

__device__ __forceinline__ double block_min_f64(double v, double* scratch) {
    scratch[threadIdx.x] = v;
    __syncthreads();
    for (int s = LK_BLOCK / 2; s > 0; s >>= 1) {
        if (threadIdx.x < s)
            scratch[threadIdx.x] = fmin(scratch[threadIdx.x],
                                        scratch[threadIdx.x + s]);
        __syncthreads();
    }
    double r = scratch[0];
    __syncthreads();
    return r;
}

__device__ __forceinline__ int block_sum_i32(int v, int* scratch) {
    scratch[threadIdx.x] = v;
    __syncthreads();
    for (int s = LK_BLOCK / 2; s > 0; s >>= 1) {
        if (threadIdx.x < s)
            scratch[threadIdx.x] += scratch[threadIdx.x + s];
        __syncthreads();
    }
    int r = scratch[0];
    __syncthreads();
    return r;
}

extern "C" __global__ void __launch_bounds__(LK_BLOCK)
lookahead_batch_kernel(const EnvDesc* __restrict__ descs,
                       // op arrays (concatenated over envs)
                       double* __restrict__ op_remaining,
                       const int* __restrict__ op_worker,
                       const int* __restrict__ op_priority,
                       const long* __restrict__ out_indptr,   // per env: n_ops+1, offset by op_off+env
                       const int* __restrict__ out_edges,     // csr_off-based
                       const int* __restrict__ true_parent_count,
                       int* __restrict__ parent_done,
                       unsigned char* __restrict__ op_ready,
                       unsigned char* __restrict__ op_completed,
                       // dep arrays
                       double* __restrict__ dep_remaining,
                       const unsigned char* __restrict__ dep_is_flow,
                       const int* __restrict__ dep_channel,
                       const int* __restrict__ dep_priority,
                       const int* __restrict__ dep_dst,
                       unsigned char* __restrict__ dep_ready,
                       unsigned char* __restrict__ dep_completed,
                       // scratch
                       unsigned long long* __restrict__ worker_best,
                       unsigned long long* __restrict__ channel_best,
                       // outputs per env
                       double* __restrict__ out_t,
                       double* __restrict__ out_comp_oh,
                       double* __restrict__ out_comm_oh,
                       double* __restrict__ out_active_sum,
                       int* __restrict__ out_ticks,
                       int* __restrict__ out_status) {
    const EnvDesc d = descs[blockIdx.x];
    const int tid = threadIdx.x;

    double* rem_op = op_remaining + d.op_off;
    const int* w_of = op_worker + d.op_off;
    const int* p_of = op_priority + d.op_off;
    const long* indptr = out_indptr + d.indptr_off;  // n_ops+1 entries
    const int* edges = out_edges + d.csr_off;
    const int* tpc = true_parent_count + d.op_off;
    int* pdone = parent_done + d.op_off;
    unsigned char* o_ready = op_ready + d.op_off;
    unsigned char* o_done = op_completed + d.op_off;

    double* rem_dep = dep_remaining + d.dep_off;
    const unsigned char* is_flow = dep_is_flow + d.dep_off;
    const int* ch_of = dep_channel + d.dep_off;
    const int* dp_of = dep_priority + d.dep_off;
    const int* dst_of = dep_dst + d.dep_off;
    unsigned char* e_ready = dep_ready + d.dep_off;
    unsigned char* e_done = dep_completed + d.dep_off;

    unsigned long long* wbest = worker_best + d.worker_off;
    unsigned long long* cbest = channel_best + d.channel_off;

    __shared__ double s_f64[LK_BLOCK];
    __shared__ int s_i32[LK_BLOCK];
    __shared__ int s_done_ops, s_done_deps;
    __shared__ double s_t, s_comp, s_comm, s_active;

    if (tid == 0) {
        s_done_ops = 0;
        s_done_deps = 0;
        s_t = 0.0;
        s_comp = 0.0;
        s_comm = 0.0;
        s_active = 0.0;
    }
    __syncthreads();

    const double INF = __builtin_inf();
    int tick_counter = 0;

    while (true) {
        // ---- phase 1: per-worker priority op ----
        for (int w = tid; w < d.n_workers; w += LK_BLOCK) wbest[w] = 0ull;
        __syncthreads();
        for (int i = tid; i < d.n_ops; i += LK_BLOCK) {
            if (o_ready[i]) {
                unsigned long long pk =
                    (((unsigned long long)(p_of[i] + 1)) << 32) | (unsigned)i;
                atomicMax(&wbest[w_of[i]], pk);
            }
        }
        __syncthreads();
        // ---- phase 2: min remaining over priority ops + active count ----
        double my_min = INF;
        int my_active = 0;
        for (int w = tid; w < d.n_workers; w += LK_BLOCK) {
            unsigned long long pk = wbest[w];
            if (pk != 0ull) {
                my_active += 1;
                my_min = fmin(my_min, rem_op[(int)(pk & 0xffffffffu)]);
            }
        }
        double shortest_op = block_min_f64(my_min, s_f64);
        int n_active = block_sum_i32(my_active, s_i32);

        // ---- phase 3: non-flow detection / channel priority flows ----
        int my_nonflow = 0;
        for (int e = tid; e < d.n_deps; e += LK_BLOCK)
            if (e_ready[e] && !is_flow[e]) my_nonflow = 1;
        int any_nonflow = block_sum_i32(my_nonflow, s_i32);

        double shortest_comm;
        if (any_nonflow > 0) {
            shortest_comm = 0.0;
        } else {
            for (int c = tid; c < d.n_channels; c += LK_BLOCK) cbest[c] = 0ull;
            __syncthreads();
            for (int e = tid; e < d.n_deps; e += LK_BLOCK) {
                if (e_ready[e]) {  // all ready deps are flows here
                    unsigned long long pk =
                        (((unsigned long long)(dp_of[e] + 1)) << 32) | (unsigned)e;
                    atomicMax(&cbest[ch_of[e]], pk);
                }
            }
            __syncthreads();
            double cmin = INF;
            for (int c = tid; c < d.n_channels; c += LK_BLOCK) {
                unsigned long long pk = cbest[c];
                if (pk != 0ull)
                    cmin = fmin(cmin, rem_dep[(int)(pk & 0xffffffffu)]);
            }
            shortest_comm = block_min_f64(cmin, s_f64);
        }

        const double tick = fmin(shortest_op, shortest_comm);
        if (isinf(tick)) {  // deadlock (should not happen on valid DAGs)
            if (tid == 0) out_status[blockIdx.x] = 2;
            return;
        }

        // ---- phase 4a: tick deps on the PRE-op-tick ready snapshot ----
        // (op completions below only ADD ready deps; ticking deps first is
        // equivalent to the reference's snapshot, reference :429)
        int my_dep_done = 0;
        int my_flow_ticked = 0;
        for (int e = tid; e < d.n_deps; e += LK_BLOCK) {
            if (!e_ready[e]) continue;
            if (any_nonflow > 0 && is_flow[e]) continue;  // only non-flows tick
            if (any_nonflow == 0) my_flow_ticked = 1;
            double r = rem_dep[e];
            r = r - fmin(tick, r);
            rem_dep[e] = r;
            if (r == 0.0 && !e_done[e]) {
                e_done[e] = 1;
                e_ready[e] = 0;
                my_dep_done += 1;
                const int child = dst_of[e];
                const int before = atomicAdd(&pdone[child], 1);
                if (before + 1 == tpc[child]) o_ready[child] = 1;
            }
        }
        int flows_ticked = block_sum_i32(my_flow_ticked, s_i32);
        int deps_done_now = block_sum_i32(my_dep_done, s_i32);

        // ---- phase 4b: tick priority ops, wavefront new-ready deps ----
        int my_op_done = 0;
        for (int w = tid; w < d.n_workers; w += LK_BLOCK) {
            unsigned long long pk = wbest[w];
            if (pk == 0ull) continue;
            const int op = (int)(pk & 0xffffffffu);
            double r = rem_op[op];
            r = r - fmin(tick, r);
            rem_op[op] = r;
            if (r == 0.0) {
                o_done[op] = 1;
                o_ready[op] = 0;
                my_op_done += 1;
                const long b = indptr[op], e2 = indptr[op + 1];
                for (long k = b; k < e2; ++k)
                    e_ready[edges[k]] = 1;
            }
        }
        int ops_done_now = block_sum_i32(my_op_done, s_i32);

        // ---- phase 5: accounting ----
        if (tid == 0) {
            if (n_active > 0) s_comp += tick;
            if (flows_ticked > 0) s_comm += tick;
            s_active += (double)n_active * tick;
            s_t += tick;
            s_done_ops += ops_done_now;
            s_done_deps += deps_done_now;
        }
        __syncthreads();
        ++tick_counter;

        if (s_done_ops == d.n_ops && s_done_deps == d.n_deps) {
            if (tid == 0) {
                out_t[blockIdx.x] = s_t;
                out_comp_oh[blockIdx.x] = s_comp;
                out_comm_oh[blockIdx.x] = s_comm;
                out_active_sum[blockIdx.x] = s_active;
                out_ticks[blockIdx.x] = tick_counter;
                out_status[blockIdx.x] = 0;
            }
            return;
        }
        if (tick_counter >= d.max_ticks) {
            if (tid == 0) out_status[blockIdx.x] = 1;  // tick budget exceeded
            return;
        }
        __syncthreads();
    }
}

// ---------------------------------------------------------------------------
std::vector<torch::Tensor> lookahead_batch(torch::Tensor descs,
                              torch::Tensor op_remaining,
                              torch::Tensor op_worker,
                              torch::Tensor op_priority,
                              torch::Tensor out_indptr,
                              torch::Tensor out_edges,
                              torch::Tensor true_parent_count,
                              torch::Tensor parent_done,
                              torch::Tensor op_ready,
                              torch::Tensor op_completed,
                              torch::Tensor dep_remaining,
                              torch::Tensor dep_is_flow,
                              torch::Tensor dep_channel,
                              torch::Tensor dep_priority,
                              torch::Tensor dep_dst,
                              torch::Tensor dep_ready,
                              torch::Tensor dep_completed,
                              torch::Tensor worker_best,
                              torch::Tensor channel_best) {
    TORCH_CHECK(descs.dtype() == torch::kInt64 && descs.is_cuda());
    TORCH_CHECK(descs.dim() == 2 && descs.size(1) == ENV_DESC_LONGS);
    TORCH_CHECK(descs.is_contiguous());
    const int B = descs.size(0);
    auto opts_f64 = op_remaining.options();
    auto t_out = torch::zeros({B}, opts_f64);
    auto comp_out = torch::zeros({B}, opts_f64);
    auto comm_out = torch::zeros({B}, opts_f64);
    auto active_out = torch::zeros({B}, opts_f64);
    auto i_opts = descs.options().dtype(torch::kInt32);
    auto ticks_out = torch::zeros({B}, i_opts);
    auto status_out = torch::full({B}, 3, i_opts);  // 3 = never finished
    hipStream_t stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(lookahead_batch_kernel, dim3(B), dim3(LK_BLOCK), 0,
                       stream,
                       reinterpret_cast<const EnvDesc*>(descs.data_ptr<long>()),
                       op_remaining.data_ptr<double>(),
                       op_worker.data_ptr<int>(),
                       op_priority.data_ptr<int>(),
                       out_indptr.data_ptr<long>(),
                       out_edges.data_ptr<int>(),
                       true_parent_count.data_ptr<int>(),
                       parent_done.data_ptr<int>(),
                       op_ready.data_ptr<unsigned char>(),
                       op_completed.data_ptr<unsigned char>(),
                       dep_remaining.data_ptr<double>(),
                       dep_is_flow.data_ptr<unsigned char>(),
                       dep_channel.data_ptr<int>(),
                       dep_priority.data_ptr<int>(),
                       dep_dst.data_ptr<int>(),
                       dep_ready.data_ptr<unsigned char>(),
                       dep_completed.data_ptr<unsigned char>(),
                       (unsigned long long*)worker_best.data_ptr<long>(),
                       (unsigned long long*)channel_best.data_ptr<long>(),
                       t_out.data_ptr<double>(),
                       comp_out.data_ptr<double>(),
                       comm_out.data_ptr<double>(),
                       active_out.data_ptr<double>(),
                       ticks_out.data_ptr<int>(),
                       status_out.data_ptr<int>());
    return {t_out, comp_out, comm_out, active_out, ticks_out, status_out};
}
