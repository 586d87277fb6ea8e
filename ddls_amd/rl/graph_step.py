"""hipGraph-captured PPO SGD minibatch step.

The SGD update is launch-bound on MI355X: one minibatch fwd+bwd+Adam of the
~1e5-param GNN policy is ~120 tiny kernels, each far cheaper than its host
dispatch.  This module captures the WHOLE minibatch step (zero-grad -> fused
HIP GNN forward -> PPO loss -> backward -> grad clip -> capturable Adam ->
device-side stats accumulation) in a single hipGraph (torch.cuda.CUDAGraph is
hipGraph on ROCm) and replays it per minibatch with one launch.

Static-shape strategy: graphs vary in node/edge count, so minibatches are
packed into fixed-capacity buffers with ONE trailing dummy graph (id B) that
absorbs every padded node and edge — padded edges point at the last padded
node, padded nodes carry graph id B, and all loss math slices [:B], so
garbage in the padded region can never reach a real gradient.  Capacities
grow geometrically with re-capture on overflow.

The reference has no equivalent (RLlib eager torch); this is
MI355X-native new functionality ("capture launch-bound inner loops in
hipGraphs").

Distributed: RCCL collectives are kept OUTSIDE the graph — two captured
segments (fwd+bwd | clip+step) with the eager fused all-reduce between them.
"""
from __future__ import annotations

import os
from typing import List, Optional

import numpy as np
import torch
import torch.nn.functional as F

from ..models.gnn import GraphBatch
from ..parallel import all_reduce_gradients, is_distributed
from .rollout import CompactObs


class _PPOLossFn(torch.autograd.Function):
    """Fused PPO loss (ops/hip/ppo_loss.hip): one kernel forward, one kernel
    analytic backward — replaces ~100 tiny elementwise kernels per replay."""

    @staticmethod
    def forward(ctx, logits, values, actions, old_logp, adv, vtarg,
                kl_coef_t, clip, vf_clip, vf_coef, ent_coef):
        from .. import ops as hip_ops
        ext = hip_ops.get_extension(required=True)
        loss, stats, p, lp, coef, h = ext.ppo_loss_fwd(
            logits.contiguous(), values.contiguous(), actions, old_logp,
            adv, vtarg, kl_coef_t.reshape(1), float(clip), float(vf_clip),
            float(vf_coef), float(ent_coef))
        ctx.save_for_backward(p, lp, coef, h, actions, values, vtarg)
        ctx.consts = (float(vf_clip), float(vf_coef), float(ent_coef))
        ctx.mark_non_differentiable(stats)
        return loss.reshape(()), stats

    @staticmethod
    def backward(ctx, gl, _gstats):
        from .. import ops as hip_ops
        ext = hip_ops.get_extension(required=True)
        p, lp, coef, h, actions, values, vtarg = ctx.saved_tensors
        vf_clip, vf_coef, ent_coef = ctx.consts
        glogits, gvalues = ext.ppo_loss_bwd(
            p, lp, coef, h, actions, values, vtarg,
            gl.reshape(1).contiguous(), vf_clip, vf_coef, ent_coef)
        return (glogits, gvalues) + (None,) * 9


class _FusedCachedEngine:
    """Host side of ops/hip/cached_step.hip: the ENTIRE cached-models PPO
    minibatch fwd+bwd as two kernels, weight grads written directly into the
    flat grad buffer.  Eligible for the tuned PAC-ML policy shape
    (module_depth 1, 2 rounds, one FC hidden, relu, masked)."""

    @staticmethod
    def eligible(policy, ext) -> bool:
        cfg = policy.config
        # the kernels are compiled for the tuned PAC-ML dims (cached_step.hip
        # K* constants); other shapes take the autograd cached path
        return (ext is not None and hasattr(ext, "cached_step_fwd")
                and cfg["module_depth"] == 1 and cfg["num_rounds"] == 2
                and cfg["fcnet_hiddens"] == [256]
                and cfg["aggregator_activation"] == "relu"
                and cfg["fcnet_activation"] == "relu"
                and cfg["apply_action_mask"]
                and cfg["in_features_node"] == 5
                and cfg["in_features_edge"] == 2
                and cfg["in_features_graph"] == 17
                and cfg["out_features_msg"] == 32
                and cfg["out_features_hidden"] == 64
                and cfg["out_features_node"] == 16
                and cfg["out_features_graph"] == 8
                and policy.num_actions == 17
                and os.environ.get("DDLS_AMD_DISABLE_FUSED_STEP", "0") != "1")

    # weight slots in the kernel's fixed semantic order (enum W_* in
    # cached_step.hip)
    _SLOTS = [
        "gnn.layers.0.node_module.0.weight", "gnn.layers.0.node_module.0.bias",
        "gnn.layers.0.node_module.1.weight", "gnn.layers.0.node_module.1.bias",
        "gnn.layers.0.edge_module.0.weight", "gnn.layers.0.edge_module.0.bias",
        "gnn.layers.0.edge_module.1.weight", "gnn.layers.0.edge_module.1.bias",
        "gnn.layers.0.reduce_module.0.weight",
        "gnn.layers.0.reduce_module.0.bias",
        "gnn.layers.0.reduce_module.1.weight",
        "gnn.layers.0.reduce_module.1.bias",
        "gnn.layers.1.node_module.0.weight", "gnn.layers.1.node_module.0.bias",
        "gnn.layers.1.node_module.1.weight", "gnn.layers.1.node_module.1.bias",
        "gnn.layers.1.edge_module.0.weight", "gnn.layers.1.edge_module.0.bias",
        "gnn.layers.1.edge_module.1.weight", "gnn.layers.1.edge_module.1.bias",
        "gnn.layers.1.reduce_module.0.weight",
        "gnn.layers.1.reduce_module.0.bias",
        "gnn.layers.1.reduce_module.1.weight",
        "gnn.layers.1.reduce_module.1.bias",
        "graph_module.0.weight", "graph_module.0.bias",
        "graph_module.1.weight", "graph_module.1.bias",
        "policy_branch.0.weight", "policy_branch.0.bias",
        "policy_branch.2.weight", "policy_branch.2.bias",
        "value_branch.0.weight", "value_branch.0.bias",
        "value_branch.2.weight", "value_branch.2.bias",
    ]

    # must equal the C_* enum length in ops/hip/cached_step.hip (C_NT);
    # pinned by tests/test_aux.py::test_fused_scratch_contract
    N_SCRATCH = 90

    def __init__(self, stepper, models_batch):
        self.st = stepper
        ext = stepper._ext
        policy = stepper.policy
        dev = stepper.device
        cfg = policy.config
        B, A = stepper.B, stepper.A
        mb = models_batch
        order, indptr = mb.csr_by_dst()
        nptr = mb.node_ptr()
        N, E = mb.z.shape[0], mb.src.shape[0]
        M = mb.num_graphs
        src_order = torch.argsort(mb.src, stable=True)
        src_counts = torch.bincount(mb.src, minlength=N)
        src_indptr = torch.zeros(N + 1, dtype=torch.int64, device=dev)
        torch.cumsum(src_counts, 0, out=src_indptr[1:])

        # flat offsets in the kernel's semantic slot order
        name_to_off = {}
        off = 0
        for name, p in policy.named_parameters():
            name_to_off[name] = off
            off += p.numel()
        offs = torch.tensor([name_to_off[n] for n in self._SLOTS],
                            dtype=torch.int64, device=dev)

        H = cfg["out_features_msg"] // 2
        HID = cfg["out_features_hidden"]
        OUT = cfg["out_features_node"]
        GEMB = cfg["out_features_graph"]
        FC = cfg["fcnet_hiddens"][0]
        F0 = mb.z.shape[1]
        FE = mb.e.shape[1]
        GFin = cfg["in_features_graph"] + A
        MSG = 2 * H

        z = lambda *shape: torch.zeros(shape, device=dev)
        self.scratch = [
            # statics
            mb.z.contiguous(), mb.e.contiguous(), mb.src, mb.dst,
            order, indptr, src_order, src_indptr,
            mb.graph_of_node, nptr,
            # staged inputs (the stepper's d-buffers, filled per minibatch)
            stepper.d["model_ids"], stepper.d["gf"], stepper.d["mask"],
            stepper.d["actions"], stepper.d["old_logp"], stepper.d["adv"],
            stepper.d["vtarg"], stepper.kl_coeff_t.reshape(1),
            # weights
            stepper.flat_p, stepper.flat_g, offs,
            # fwd activations
            z(N, H), z(E, H), z(E, HID), z(N, HID), z(N, HID),
            z(N, H), z(E, H), z(E, OUT), z(N, OUT), z(N, OUT), z(M, OUT),
            z(N, F0), z(E, FE), z(E, MSG), z(N, MSG), z(N), z(E), z(E), z(N),
            z(N, HID), z(E, FE), z(E, MSG), z(N, MSG), z(N), z(E), z(E), z(N),
            z(B, GFin), z(B, OUT + GEMB), z(B, FC), z(B, FC), z(B),
            z(B, A), z(B, A), z(B), z(B),
            stepper.stats_acc,
            # bwd scratch
            z(B, A), z(B), z(B, FC), z(B, FC), z(B, OUT + GEMB), z(B, GFin),
            z(M, OUT), z(N, OUT), z(E, MSG), z(N, MSG), z(N, H), z(E, H),
            z(N, HID),
            z(E, MSG), z(N, MSG), z(N, H), z(E, H),
            z(N, OUT), z(E, OUT), z(N, H), z(E, H),
            z(E, HID), z(N, HID),
            # LN-out grad rows (gu) stored by the data-bwd kernels
            z(E, MSG), z(N, MSG), z(N, HID), z(E, FE),
            z(E, MSG), z(N, MSG), z(N, F0), z(E, FE),
            # row-split wgrad partials: 6 jobs x WSPLIT(16) x WG_JSTRIDE
            # (2240) — mirrors cached_step.hip's WSPLIT/WG_JSTRIDE
            z(6 * 16 * 2240),
        ]
        assert len(self.scratch) == self.N_SCRATCH, len(self.scratch)
        c = stepper.cfg
        self.fscal = [float(c.clip_param), float(c.vf_clip_param),
                      float(c.vf_loss_coeff), float(c.entropy_coeff)]
        self.ext = ext

    def run(self):
        self.ext.cached_step_fwd(self.scratch, self.fscal)
        self.ext.cached_step_bwd(self.scratch, self.fscal)


class CapturedSGDStep:
    """Replayable hipGraph of one PPO SGD minibatch step.

    ``step()`` returns False (caller runs the eager path) when the minibatch
    is partial, capture is unsupported, or a capture attempt failed.
    """

    GROWTH = 1.3
    # dummy-node slots always kept free beyond the real-node capacity: padded
    # edges are spread across MANY dummy nodes — a single mega-segment would
    # serialize the wave-per-node message_reduce forward and the backward's
    # ghn atomics
    RESERVE = 256

    def __init__(self, policy, optimizer, config, device, models_batch=None):
        self.policy = policy
        self.optimizer = optimizer
        self.cfg = config
        self.device = device
        # cached-models mode: every sample's node/edge features are static
        # per workload model (vectorised-engine obs), so the GNN runs ONCE on
        # the M model graphs per minibatch and per-sample gradients reach it
        # summed through index_select's backward — exact by linearity, ~25x
        # less GNN work, and every captured shape is static (no capacity
        # management).
        self.models_batch = models_batch
        if models_batch is not None:
            models_batch.csr_by_dst()   # materialise before capture
            models_batch.node_ptr()
        self.B = config.sgd_minibatch_size
        self.A = policy.num_actions
        self.Fn = policy.config["in_features_node"]
        self.Fe = policy.config["in_features_edge"]
        self.Fg = policy.config["in_features_graph"] + self.A
        self.broken = not (device.type == "cuda" and torch.cuda.is_available()
                           and hasattr(torch.cuda, "CUDAGraph"))
        self.n_cap = 0
        self.e_cap = 0
        self.graph = None
        self.graph_opt = None   # second segment when distributed
        self.capture_count = 0
        self.last_error: Optional[str] = None
        try:
            from .. import ops as hip_ops
            ext = hip_ops.get_extension()
        except Exception:
            ext = None
        self._fused_loss = (
            ext is not None and hasattr(ext, "ppo_loss_fwd")
            and self.A <= 64
            and os.environ.get("DDLS_AMD_DISABLE_FUSED_LOSS", "0") != "1")
        # flat-buffer optimizer engine: every param and grad becomes a view
        # into one flat tensor each, and clip+Adam run as ONE fused kernel
        # (torch capturable Adam decomposes into ~2 kernels per param)
        self._flat_engine = (
            ext is not None and hasattr(ext, "flat_adam")
            and os.environ.get("DDLS_AMD_DISABLE_FLAT_ADAM", "0") != "1")
        self._ext = ext
        self.flat_p = None
        # [policy_loss, vf_loss, kl, entropy, total_loss] device accumulator:
        # read ONCE per update() so replays never host-sync
        self.stats_acc = torch.zeros(5, device=device)
        self.kl_coeff_t = torch.zeros((), device=device)

    # ------------------------------------------------------------------
    def set_kl_coeff(self, v: float):
        self.kl_coeff_t.fill_(float(v))

    def reset_stats(self):
        self.stats_acc.zero_()

    # ------------------------------------------------------------------
    def ensure_capacity(self, n_cap: int, e_cap: int) -> bool:
        """(Re-)capture if the buffers cannot hold n_cap nodes / e_cap edges.
        Call once per update with B x the rollout's max per-sample counts so
        the minibatch loop never recaptures."""
        if self.broken:
            return False
        if self.models_batch is not None:
            if self.graph is not None:
                return True
            n_cap = e_cap = 0       # fixed-shape staging, single capture
        if (self.graph is not None and n_cap <= self.n_cap - self.RESERVE
                and e_cap <= self.e_cap):
            return True
        try:
            self._capture(max(n_cap + self.RESERVE,
                              int(self.n_cap * self.GROWTH)),
                          max(e_cap, int(self.e_cap * self.GROWTH)))
            return True
        except Exception as exc:  # unsupported torch/ROCm combo -> eager
            self.broken = True
            self.last_error = f"{type(exc).__name__}: {exc}"
            # hand the flat Adam state back so eager training continues
            # exactly where the captured steps left off
            try:
                self.sync_state_to_optimizer()
            except Exception:
                pass
            # capturable eager Adam is slow; revert it for the fallback path
            for g in self.optimizer.param_groups:
                g["capturable"] = False
            for st in self.optimizer.state.values():
                if "step" in st and torch.is_tensor(st["step"]):
                    st["step"] = st["step"].cpu()
            return False

    def step(self, mb_obs: List[CompactObs], actions: np.ndarray,
             old_logp: np.ndarray, adv: np.ndarray,
             vtarg: np.ndarray) -> bool:
        if self.broken or len(mb_obs) != self.B:
            return False
        if self.models_batch is not None:
            if not self.ensure_capacity(0, 0):
                return False
        else:
            n = sum(len(o.node_features) for o in mb_obs)
            e = sum(len(o.edges_src) for o in mb_obs)
            if not self.ensure_capacity(n, e) or n > self.n_cap - self.RESERVE:
                return False
        self.commit(self.prepare(mb_obs, actions, old_logp, adv, vtarg))
        return True

    def prepare(self, mb_obs, actions, old_logp, adv, vtarg) -> int:
        """Stage one minibatch into the next pinned set (CPU work only —
        safe to run in a prefetch thread while the GPU replays the previous
        minibatch).  Alternates the two pinned sets; calls must be strictly
        sequential."""
        return self._fill(mb_obs, actions, old_logp, adv, vtarg)

    def commit(self, j: int):
        """Enqueue H2D copies from pinned set j + replay the graph."""
        if self.models_batch is not None:
            self._flat_f32.copy_(self.pin[j]["_flat_f32"], non_blocking=True)
            self._flat_i64.copy_(self.pin[j]["_flat_i64"], non_blocking=True)
        else:
            for k, dst_t in self.d.items():
                dst_t.copy_(self.pin[j][k], non_blocking=True)
        self.copy_events[j].record()
        self.graph.replay()
        if self.graph_opt is not None:
            if self.flat_p is not None:
                self._all_reduce_flat()
            else:
                all_reduce_gradients(self.policy.parameters())
            self.graph_opt.replay()

    # ------------------------------------------------------------------
    def _alloc(self, n_cap: int, e_cap: int):
        dev, B = self.device, self.B

        def dbuf(shape, dtype):
            return torch.zeros(shape, dtype=dtype, device=dev)

        def pbuf(shape, dtype):
            return torch.zeros(shape, dtype=dtype, pin_memory=True)

        if self.models_batch is not None:
            # cached-models mode: tiny fixed-shape staging, no dummy graph.
            # All staged tensors are VIEWS of two packed flats (one f32, one
            # i64) so each commit is TWO H2D copies instead of seven.
            self.n_cap, self.e_cap = 0, 0
            f32_specs = [("gf", (B, self.Fg)), ("mask", (B, self.A)),
                         ("old_logp", (B,)), ("adv", (B,)), ("vtarg", (B,))]
            i64_specs = [("model_ids", (B,)), ("actions", (B,))]
            nf = sum(int(np.prod(sh)) for _k, sh in f32_specs)
            ni = sum(int(np.prod(sh)) for _k, sh in i64_specs)
            self._flat_f32 = dbuf(nf, torch.float32)
            self._flat_i64 = dbuf(ni, torch.int64)
            self.d = {}
            off = 0
            for k, sh in f32_specs:
                n = int(np.prod(sh))
                self.d[k] = self._flat_f32[off:off + n].view(sh)
                off += n
            off = 0
            for k, sh in i64_specs:
                n = int(np.prod(sh))
                self.d[k] = self._flat_i64[off:off + n].view(sh)
                off += n
            self.pin, self.pin_np = [], []
            for _ in range(2):
                pf = pbuf((nf,), torch.float32)
                pi = pbuf((ni,), torch.int64)
                pset = {"_flat_f32": pf, "_flat_i64": pi}
                npset = {}
                off = 0
                for k, sh in f32_specs:
                    n = int(np.prod(sh))
                    npset[k] = pf[off:off + n].view(sh).numpy()
                    off += n
                off = 0
                for k, sh in i64_specs:
                    n = int(np.prod(sh))
                    npset[k] = pi[off:off + n].view(sh).numpy()
                    off += n
                self.pin.append(pset)
                self.pin_np.append(npset)
            # benign pre-first-commit content for capture warmup (all-ones
            # mask keeps log(mask) finite)
            self.d["mask"].fill_(1.0)
            for npset in self.pin_np:
                npset["mask"][:] = 1.0
            self.copy_events = [torch.cuda.Event(), torch.cuda.Event()]
            self._buf = 0
            self.batch = None
            return
        self.n_cap, self.e_cap = n_cap, e_cap
        self.d = {
            "z": dbuf((n_cap, self.Fn), torch.float32),
            "e": dbuf((e_cap, self.Fe), torch.float32),
            "src": dbuf(e_cap, torch.int64),
            "dst": dbuf(e_cap, torch.int64),
            "order": dbuf(e_cap, torch.int64),
            "indptr": dbuf(n_cap + 1, torch.int64),
            "gon": dbuf(n_cap, torch.int64),
            "nptr": dbuf(B + 2, torch.int64),
            "gf": dbuf((B + 1, self.Fg), torch.float32),
            "mask": torch.ones((B + 1, self.A), dtype=torch.float32, device=dev),
            "actions": dbuf(B, torch.int64),
            "old_logp": dbuf(B, torch.float32),
            "adv": dbuf(B, torch.float32),
            "vtarg": dbuf(B, torch.float32),
        }
        # double-buffered pinned staging: _fill for minibatch k+1 must not
        # overwrite host memory the async H2D copies of minibatch k are still
        # reading (no per-minibatch sync exists), so alternate two pinned sets
        # guarded by events recorded after each copy batch
        self.pin = []
        self.pin_np = []
        for _ in range(2):
            pset = {k: pbuf(tuple(v.shape), v.dtype) for k, v in self.d.items()}
            pset["mask"].fill_(1.0)
            self.pin.append(pset)
            self.pin_np.append({k: v.numpy() for k, v in pset.items()})
        self.copy_events = [torch.cuda.Event(), torch.cuda.Event()]
        self._buf = 0
        # one static flat graph: B real graphs + dummy graph B for padding
        self.batch = GraphBatch(
            z=self.d["z"], e=self.d["e"], src=self.d["src"], dst=self.d["dst"],
            graph_of_node=self.d["gon"], num_graphs=B + 1,
            _csr=(self.d["order"], self.d["indptr"]),
            _node_ptr=self.d["nptr"])

    def _fill(self, mb_obs, actions, old_logp, adv, vtarg):
        if self.models_batch is not None:
            j = self._buf = self._buf ^ 1
            self.copy_events[j].synchronize()
            p = self.pin_np[j]
            p["model_ids"][:] = [o.model_id for o in mb_obs]
            p["gf"][:] = np.stack([o.graph_features for o in mb_obs])
            p["mask"][:] = np.stack([o.action_mask for o in mb_obs])
            p["actions"][:] = actions
            p["old_logp"][:] = old_logp
            p["adv"][:] = adv
            p["vtarg"][:] = vtarg
            return j
        ns = np.array([len(o.node_features) for o in mb_obs], dtype=np.int64)
        offsets = np.concatenate([[0], np.cumsum(ns)[:-1]])
        n = int(ns.sum())
        z = np.concatenate([o.node_features for o in mb_obs])
        e = np.concatenate([o.edge_features for o in mb_obs])
        src = np.concatenate([o.edges_src + off
                              for o, off in zip(mb_obs, offsets)])
        dst = np.concatenate([o.edges_dst + off
                              for o, off in zip(mb_obs, offsets)])
        m = len(src)
        # per-sample CSR cached on the obs (reused across the 8 SGD epochs):
        # samples are node-offset-ordered, so concatenating sample-local
        # dst-sorted orders IS the global dst-sorted order
        orders, cnts = [], []
        eoff = 0
        for o in mb_obs:
            c = getattr(o, "_csr", None)
            if c is None:
                c = (np.argsort(o.edges_dst, kind="stable").astype(np.int64),
                     np.bincount(o.edges_dst,
                                 minlength=len(o.node_features)))
                o._csr = c
            orders.append(c[0] + eoff)
            cnts.append(c[1])
            eoff += len(c[0])
        j = self._buf = self._buf ^ 1
        # wait until the GPU finished the copies that last read this pinned set
        self.copy_events[j].synchronize()
        p = self.pin_np[j]
        p["z"][:n] = z
        p["e"][:m] = e
        p["src"][:m] = src
        p["dst"][:m] = dst
        # spread padded edges monotonically over the dummy nodes [n, n_cap) as
        # self-loops: short per-node segments, no serial mega-segment and no
        # atomic hot spot in the backward
        pad_e = self.e_cap - m
        pad_nodes = self.n_cap - n
        pad_dst = n + (np.arange(pad_e, dtype=np.int64) * pad_nodes) // max(pad_e, 1)
        p["src"][m:] = pad_dst
        p["dst"][m:] = pad_dst
        # padded edges sit at node ids above every real dst and are already
        # sorted, so only real edges need sorting (from the per-sample cache)
        p["order"][:m] = np.concatenate(orders) if orders else []
        p["order"][m:] = np.arange(m, self.e_cap)
        counts = np.zeros(self.n_cap, dtype=np.int64)
        cat_cnts = np.concatenate(cnts) if cnts else np.zeros(0, np.int64)
        counts[:len(cat_cnts)] = cat_cnts
        counts += np.bincount(pad_dst, minlength=self.n_cap)
        p["indptr"][0] = 0
        np.cumsum(counts, out=p["indptr"][1:])
        p["gon"][:n] = np.repeat(np.arange(self.B, dtype=np.int64), ns)
        p["gon"][n:] = self.B
        p["nptr"][0] = 0
        np.cumsum(ns, out=p["nptr"][1:self.B + 1])
        p["nptr"][self.B + 1] = self.n_cap  # dummy graph holds all padding
        p["gf"][:self.B] = np.stack([o.graph_features for o in mb_obs])
        p["mask"][:self.B] = np.stack([o.action_mask for o in mb_obs])
        # row B of gf stays zero and of mask stays all-ones (log(1)=0), so the
        # dummy padding graph has finite logits
        p["actions"][:] = actions
        p["old_logp"][:] = old_logp
        p["adv"][:] = adv
        p["vtarg"][:] = vtarg
        return j

    # ------------------------------------------------------------------
    _fused_cached = None

    def _body_fwd_bwd(self):
        cfg, B = self.cfg, self.B
        if self._fused_cached is not None:
            # fully-fused path: two kernels, grads straight into flat_g
            self._fused_cached.run()
            return
        if self.flat_p is None:
            # flat engine zeroes grads inside flat_adam instead
            grads = [pa.grad for pa in self.policy.parameters()
                     if pa.grad is not None]
            if grads:
                torch._foreach_zero_(grads)
        else:
            # pin every grad to its flat_g view at RECORD time: autograd's
            # AccumulateGrad may otherwise accumulate OUT-OF-PLACE (rebinding
            # p.grad to a capture-pool tensor), in which case replays never
            # write flat_g and training silently freezes
            base = self.flat_g.data_ptr()
            for p_, (o, np_) in zip(self._params, self._offs):
                if (p_.grad is None
                        or p_.grad.data_ptr() != base + 4 * o):
                    p_.grad = self.flat_g[o:o + np_].view(p_.shape)
        if self.models_batch is not None:
            # GNN once over the M model graphs; per-sample grads reach it
            # summed through index_select's backward (exact by linearity)
            from ..models.gnn import graph_mean
            node_emb = self.policy.gnn(self.models_batch)
            pooled = graph_mean(node_emb, self.models_batch)     # [M, out]
            emb = pooled.index_select(0, self.d["model_ids"])    # [B, out]
            graph_emb = self.policy.graph_module(self.d["gf"])
            final = torch.cat([emb, graph_emb], dim=-1)
            logits = self.policy.policy_branch(final)
            values = self.policy.value_branch(final).squeeze(-1)
            if self.policy.config["apply_action_mask"]:
                inf_mask = torch.clamp(torch.log(self.d["mask"]),
                                       min=torch.finfo(torch.float32).min)
                logits = logits + inf_mask
        else:
            logits, values = self.policy.forward_flat(
                self.batch, self.d["gf"], self.d["mask"])
            logits, values = logits[:B], values[:B]
        if self._fused_loss:
            loss, stats = _PPOLossFn.apply(
                logits, values, self.d["actions"], self.d["old_logp"],
                self.d["adv"], self.d["vtarg"], self.kl_coeff_t,
                cfg.clip_param, cfg.vf_clip_param, cfg.vf_loss_coeff,
                cfg.entropy_coeff)
            loss.backward()
            self._repair_grad_views()
            self.stats_acc += stats
            return
        # Categorical re-implemented with log_softmax: torch.distributions
        # argument validation host-syncs, which aborts stream capture
        logp_all = F.log_softmax(logits, dim=-1)
        logp = logp_all.gather(1, self.d["actions"].unsqueeze(1)).squeeze(1)
        ratio = torch.exp(logp - self.d["old_logp"])
        adv = self.d["adv"]
        surr = torch.min(ratio * adv,
                         torch.clamp(ratio, 1 - cfg.clip_param,
                                     1 + cfg.clip_param) * adv)
        policy_loss = -surr.mean()
        kl = (self.d["old_logp"] - logp).mean()
        vf_err = (values - self.d["vtarg"]) ** 2
        vf_loss = torch.clamp(vf_err, 0, cfg.vf_clip_param).mean()
        entropy = (-(logp_all.exp() * logp_all).sum(-1)).mean()
        loss = (policy_loss + self.kl_coeff_t * kl
                + cfg.vf_loss_coeff * vf_loss - cfg.entropy_coeff * entropy)
        loss.backward()
        self._repair_grad_views()
        self.stats_acc += torch.stack([policy_loss.detach(), vf_loss.detach(),
                                       kl.detach(), entropy.detach(),
                                       loss.detach()])

    repair_count = 0  # grads rebound off their views at last capture

    def _repair_grad_views(self):
        """If AccumulateGrad rebound any p.grad off its flat_g view during
        this (possibly capture-recorded) backward, record a copy from the
        rebound tensor into flat_g — pool tensor addresses are stable across
        replays, so the recorded copies keep flat_adam fed."""
        if self.flat_p is None:
            return
        base = self.flat_g.data_ptr()
        self.repair_count = 0
        for p_, (o, np_) in zip(self._params, self._offs):
            g = p_.grad
            if g is not None and g.data_ptr() != base + 4 * o:
                self.flat_g[o:o + np_].copy_(g.detach().reshape(-1))
                self.repair_count += 1

    def _body_opt(self):
        if self.flat_p is not None:
            # fused grad-clip + Adam over the flat buffers (the norm uses the
            # in-house sumsq kernel: torch reductions drift under replay)
            self._ext.flat_sumsq(self.flat_g, self.normsq)
            self._ext.flat_adam(self.flat_p, self.flat_g, self.flat_m,
                                self.flat_v, self.step_t, self.normsq,
                                float(self.cfg.grad_clip or 0.0),
                                float(self.cfg.lr), 0.9, 0.999, 1e-8)
            return
        if self.cfg.grad_clip is not None:
            torch.nn.utils.clip_grad_norm_(self.policy.parameters(),
                                           self.cfg.grad_clip)
        self.optimizer.step()

    # ------------------------------------------------------------------
    def _flatten_params(self):
        """Re-materialise every param and grad as a view into one flat
        tensor each (apex-style), enabling the fused flat_adam kernel and a
        copy-free distributed all-reduce of flat_g."""
        if self.flat_p is not None:
            return
        params = list(self.policy.parameters())
        numel = sum(p.numel() for p in params)
        dev = self.device
        self.flat_p = torch.zeros(numel, device=dev)
        self.flat_g = torch.zeros(numel, device=dev)
        self.flat_m = torch.zeros(numel, device=dev)
        self.flat_v = torch.zeros(numel, device=dev)
        self.step_t = torch.zeros(1, device=dev)
        self.normsq = torch.zeros(1, device=dev)
        self._params = params
        self._offs = []
        off = 0
        for p in params:
            n = p.numel()
            self.flat_p[off:off + n].copy_(p.data.reshape(-1))
            p.data = self.flat_p[off:off + n].view(p.shape)
            p.grad = self.flat_g[off:off + n].view(p.shape)
            self._offs.append((off, n))
            off += n
        # adopt any existing torch-Adam state (e.g. a loaded checkpoint)
        step_val = None
        for p, (o, n) in zip(params, self._offs):
            st = self.optimizer.state.get(p)
            if st and "exp_avg" in st:
                self.flat_m[o:o + n].copy_(st["exp_avg"].reshape(-1).to(dev))
                self.flat_v[o:o + n].copy_(
                    st["exp_avg_sq"].reshape(-1).to(dev))
                if "step" in st:
                    step_val = float(st["step"])
        if step_val is not None:
            self.step_t.fill_(step_val)

    def sync_state_to_optimizer(self):
        """Write the flat Adam state back into the torch optimizer (for
        checkpointing and for the eager fallback path)."""
        if self.flat_p is None:
            return
        step_val = float(self.step_t.item())
        for p, (o, n) in zip(self._params, self._offs):
            st = self.optimizer.state.setdefault(p, {})
            st["step"] = torch.tensor(step_val)
            st["exp_avg"] = self.flat_m[o:o + n].detach().clone().view(p.shape)
            st["exp_avg_sq"] = (self.flat_v[o:o + n].detach().clone()
                                .view(p.shape))

    def _all_reduce_flat(self):
        import torch.distributed as dist
        g = self.flat_g
        if dist.get_backend() == "gloo" and g.is_cuda:
            host = g.cpu()
            dist.all_reduce(host)
            g.copy_(host.to(g.device))
        else:
            dist.all_reduce(g)
        g.div_(dist.get_world_size())

    def _capture(self, n_cap: int, e_cap: int):
        # drop the old graph first: its recorded state holds references that
        # push AccumulateGrad off the in-place path during re-capture
        self.graph = None
        self.graph_opt = None
        self._alloc(n_cap, e_cap)
        split = is_distributed()

        def whole():
            self._body_fwd_bwd()
            if not split:
                self._body_opt()

        saved_stats = self.stats_acc.clone()
        if self._flat_engine:
            self._flatten_params()
            if (self.models_batch is not None and self._fused_cached is None
                    and _FusedCachedEngine.eligible(self.policy, self._ext)):
                self._fused_cached = _FusedCachedEngine(self,
                                                        self.models_batch)
            # warmup (required before capture) runs REAL steps; snapshot the
            # flat training state and restore it in place afterwards so the
            # captured graph sees the live tensors but training resumes
            # exactly where an uncaptured run would be
            snap = tuple(t.clone() for t in (self.flat_p, self.flat_g,
                                             self.flat_m, self.flat_v,
                                             self.step_t))
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(3):
                    whole()
                    if split:
                        self._body_opt()
            torch.cuda.current_stream().wait_stream(s)
            torch.cuda.synchronize()
            for dst, src in zip((self.flat_p, self.flat_g, self.flat_m,
                                 self.flat_v, self.step_t), snap):
                dst.copy_(src)
        else:
            # torch-Adam path: lr=0 during warmup so the optimizer steps
            # cannot move the parameters, and the Adam state is
            # snapshot/restored in place (same tensor addresses)
            saved_lr = [g["lr"] for g in self.optimizer.param_groups]
            for g in self.optimizer.param_groups:
                g["lr"] = 0.0
                # capturable Adam keeps `step` on-device so the whole update
                # is graph-safe; migrate state created by earlier eager steps
                g["capturable"] = True
            for st in self.optimizer.state.values():
                if "step" in st and torch.is_tensor(st["step"]):
                    st["step"] = st["step"].to(self.device)
            snap = {p_: {k: v.clone() if torch.is_tensor(v) else v
                         for k, v in st.items()}
                    for p_, st in self.optimizer.state.items()}
            try:
                s = torch.cuda.Stream()
                s.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(s):
                    for _ in range(3):
                        whole()
                        if split:
                            self._body_opt()
                torch.cuda.current_stream().wait_stream(s)
                torch.cuda.synchronize()
            finally:
                for g, lr in zip(self.optimizer.param_groups, saved_lr):
                    g["lr"] = lr
            # undo warmup's pollution of the optimizer state: restore
            # snapshotted entries, zero entries first created by warmup
            # (fresh-Adam semantics)
            for p_, st in self.optimizer.state.items():
                prev = snap.get(p_)
                for k, v in st.items():
                    if not torch.is_tensor(v):
                        continue
                    if prev is not None and k in prev:
                        v.copy_(prev[k])
                    else:
                        v.zero_()
        self.stats_acc.copy_(saved_stats)

        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            whole()
        if split:
            self.graph_opt = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self.graph_opt):
                self._body_opt()
        # discard warmup/capture side effects on the accumulator
        self.stats_acc.copy_(saved_stats)
        self.capture_count += 1
        torch.cuda.synchronize()
