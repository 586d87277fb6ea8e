// Single pybind11 module for all ddls_amd HIP kernels.
#include <torch/extension.h>
#include <vector>

torch::Tensor row_mlp(torch::Tensor x, torch::Tensor ln_g, torch::Tensor ln_b,
                      torch::Tensor W, torch::Tensor b);
torch::Tensor message_reduce(torch::Tensor hn, torch::Tensor he,
                             torch::Tensor src, torch::Tensor edge_order,
                             torch::Tensor indptr, torch::Tensor ln_g,
                             torch::Tensor ln_b, torch::Tensor Wr,
                             torch::Tensor br);
torch::Tensor segment_mean(torch::Tensor x, torch::Tensor node_ptr, int64_t G);
std::vector<torch::Tensor> message_reduce_train(
    torch::Tensor hn, torch::Tensor he, torch::Tensor src,
    torch::Tensor edge_order, torch::Tensor indptr, torch::Tensor ln_g,
    torch::Tensor ln_b, torch::Tensor Wr, torch::Tensor br);
std::vector<torch::Tensor> message_reduce_bwd(
    torch::Tensor hn, torch::Tensor he, torch::Tensor src, torch::Tensor dst,
    torch::Tensor indptr, torch::Tensor ln_g, torch::Tensor ln_b,
    torch::Tensor Wr, torch::Tensor r_edge, torch::Tensor r_self,
    torch::Tensor gout);
std::vector<torch::Tensor> row_mlp_bwd(torch::Tensor x, torch::Tensor y,
                                       torch::Tensor gy, torch::Tensor ln_g,
                                       torch::Tensor ln_b, torch::Tensor W);
torch::Tensor row_mlp_mfma(torch::Tensor x, torch::Tensor ln_g,
                           torch::Tensor ln_b, torch::Tensor W,
                           torch::Tensor b);
std::vector<torch::Tensor> message_mlp_mfma(
    torch::Tensor hn, torch::Tensor he, torch::Tensor src, torch::Tensor ln_g,
    torch::Tensor ln_b, torch::Tensor Wr, torch::Tensor br);
torch::Tensor segment_combine(torch::Tensor r_edge, torch::Tensor r_self,
                              torch::Tensor edge_order, torch::Tensor indptr);
torch::Tensor segment_mean_bwd(torch::Tensor gout, torch::Tensor node_ptr,
                               int64_t N);
std::vector<torch::Tensor> message_reduce_bwd_mfma(
    torch::Tensor hn, torch::Tensor he, torch::Tensor src, torch::Tensor dst,
    torch::Tensor indptr, torch::Tensor ln_g, torch::Tensor ln_b,
    torch::Tensor Wr, torch::Tensor r_edge, torch::Tensor r_self,
    torch::Tensor gout);
void flat_adam(torch::Tensor p, torch::Tensor g, torch::Tensor m,
               torch::Tensor v, torch::Tensor step_t, torch::Tensor normsq,
               double clip, double lr, double b1, double b2, double eps);
void flat_sumsq(torch::Tensor x, torch::Tensor out);
std::vector<torch::Tensor> head_fwd(
    torch::Tensor gm, torch::Tensor gf, torch::Tensor mask,
    torch::Tensor ln_w, torch::Tensor ln_b, torch::Tensor Wg,
    torch::Tensor bg, torch::Tensor W1p, torch::Tensor b1p,
    torch::Tensor W2p, torch::Tensor b2p, torch::Tensor W1v,
    torch::Tensor b1v, torch::Tensor W2v, torch::Tensor b2v);
std::vector<torch::Tensor> head_bwd(
    torch::Tensor gf, torch::Tensor emb, torch::Tensor h1p, torch::Tensor h1v,
    torch::Tensor glogits, torch::Tensor gvalue, torch::Tensor ln_w,
    torch::Tensor ln_b, torch::Tensor Wg, torch::Tensor W1p,
    torch::Tensor W2p, torch::Tensor W1v, torch::Tensor W2v);
std::vector<torch::Tensor> ppo_loss_fwd(
    torch::Tensor logits, torch::Tensor values, torch::Tensor actions,
    torch::Tensor old_logp, torch::Tensor adv, torch::Tensor vtarg,
    torch::Tensor kl_coef, double clip, double vf_clip, double vf_coef,
    double ent_coef);
std::vector<torch::Tensor> ppo_loss_bwd(
    torch::Tensor p, torch::Tensor lp, torch::Tensor coef, torch::Tensor h,
    torch::Tensor actions, torch::Tensor values, torch::Tensor vtarg,
    torch::Tensor gl, double vf_clip, double vf_coef, double ent_coef);
void env_step_batch(std::vector<torch::Tensor> T, std::vector<double> fscal,
                    std::vector<int64_t> iscal);
void cached_step_fwd(std::vector<torch::Tensor> T, std::vector<double> fs);
void cached_step_bwd(std::vector<torch::Tensor> T, std::vector<double> fs);
std::vector<torch::Tensor> lookahead_batch(
    torch::Tensor descs, torch::Tensor op_remaining, torch::Tensor op_worker,
    torch::Tensor op_priority, torch::Tensor out_indptr, torch::Tensor out_edges,
    torch::Tensor true_parent_count, torch::Tensor parent_done,
    torch::Tensor op_ready, torch::Tensor op_completed,
    torch::Tensor dep_remaining, torch::Tensor dep_is_flow,
    torch::Tensor dep_channel, torch::Tensor dep_priority, torch::Tensor dep_dst,
    torch::Tensor dep_ready, torch::Tensor dep_completed,
    torch::Tensor worker_best, torch::Tensor channel_best);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("row_mlp", &row_mlp, "fused LN+Linear+ReLU over rows");
    m.def("message_reduce", &message_reduce,
          "fused gather + LN + GEMV + ReLU + segment-mean message passing");
    m.def("segment_mean", &segment_mean, "per-graph mean of node embeddings");
    m.def("lookahead_batch", &lookahead_batch,
          "batched RAMP lookahead discrete-event simulation");
    m.def("message_reduce_train", &message_reduce_train,
          "message_reduce storing per-message activations for backward");
    m.def("message_reduce_bwd", &message_reduce_bwd,
          "fused MeanPool message-passing backward");
    m.def("row_mlp_bwd", &row_mlp_bwd, "fused LN+Linear+ReLU backward");
    m.def("row_mlp_mfma", &row_mlp_mfma,
          "LN+Linear+ReLU on v_mfma_f32_16x16x4_f32 tiles");
    m.def("message_mlp_mfma", &message_mlp_mfma,
          "per-message reduce-MLP on the matrix core");
    m.def("segment_combine", &segment_combine,
          "CSR segment mean of stored message activations");
    m.def("ppo_loss_fwd", &ppo_loss_fwd, "fused PPO clipped-surrogate loss");
    m.def("ppo_loss_bwd", &ppo_loss_bwd, "analytic PPO loss backward");
    m.def("segment_mean_bwd", &segment_mean_bwd,
          "per-graph mean backward (broadcast/scale)");
    m.def("flat_adam", &flat_adam,
          "fused grad-clip + Adam over flat param/grad/m/v buffers");
    m.def("message_reduce_bwd_mfma", &message_reduce_bwd_mfma,
          "matrix-core message-passing backward (3-stage MFMA)");
    m.def("flat_sumsq", &flat_sumsq, "capture-safe sum of squares");
    m.def("head_fwd", &head_fwd,
          "fused policy/value head forward (LN+graphMLP+concat+2 branches)");
    m.def("head_bwd", &head_bwd, "fused policy/value head backward");
    m.def("cached_step_fwd", &cached_step_fwd,
          "fully-fused cached-models PPO minibatch forward (GNN+head+loss)");
    m.def("cached_step_bwd", &cached_step_bwd,
          "analytic whole-net backward into the flat grad buffer");
    m.def("env_step_batch", &env_step_batch,
          "batched RAMP env step over vectorised envs (placement search + "
          "memo hash probe + event loop + obs encode)");
}
