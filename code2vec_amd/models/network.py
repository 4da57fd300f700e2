"""The code2vec engine: parameters + explicit forward/backward/step orchestration.

One model implementation (vs the reference's TF/Keras dual backends). The math
is the reference graph (tensorflow_model.py:197-265):

    ctx    = dropout(concat(tok[src], path[p], tok[tgt]))        # K1-K3
    comb   = tanh(ctx @ TRANSFORM)                               # K4
    code   = Σ_c softmax_c(comb·ATTENTION + log mask) · comb     # K5-K7
    logits = code @ TARGETS^T                                    # K8
    loss   = mean CE(logits, target)                             # K9
    Adam (dense for TRANSFORM/ATTENTION/TARGETS; lazy sparse-row
    for the token/path embedding tables)                         # K10

The backward pass is hand-orchestrated (no autograd): every op is an explicit
fwd/bwd pair dispatched to HIP kernels on GPU (ops/functional.py), fp32
master weights with bf16 compute shadows, and data-parallel gradient
communication hooks placed exactly where overlap is possible (the dense
target-table grad all-reduce is launched before the rest of backward runs).
"""

import math
import os
from typing import Dict, NamedTuple, Optional

import torch

from ..config import Config
from ..ops import functional as F


class ForwardState(NamedTuple):
    ctx: torch.Tensor        # (B*C, 3d) compute dtype
    comb: torch.Tensor       # (B*C, D) compute dtype
    code: torch.Tensor       # (B, D) fp32
    alpha: torch.Tensor      # (B, C) fp32
    seed: int
    seed_t: Optional[torch.Tensor] = None


class NullReducer:
    """Single-process stand-in for the DP reducer (parallel/ddp.py)."""
    world_size = 1

    def allreduce_dense(self, key: str, tensor: torch.Tensor):
        return tensor

    def wait(self, key: str):
        pass

    def allgather_sparse(self, ids: torch.Tensor, rows: torch.Tensor):
        return ids, rows

    def allreduce_mean_scalar(self, value: float) -> float:
        return value

    def all_continue(self, have_next: bool) -> bool:
        return have_next

    def start_vote(self, have_next: bool):
        return have_next

    def finish_vote(self, vote) -> bool:
        return bool(vote)


class Code2VecNetwork:
    """Parameter store + step engine. Not an nn.Module: parameters are plain
    fp32 tensors updated in place by fused Adam kernels; state_dict I/O is
    provided explicitly for checkpoint compatibility."""

    def __init__(self, config: Config, token_vocab_size: int,
                 path_vocab_size: int, target_vocab_size: int,
                 device: Optional[str] = None):
        self.config = config
        self.device = torch.device(device or config.resolve_device())
        self.compute_dtype = (torch.bfloat16 if config.COMPUTE_DTYPE == 'bf16'
                              else torch.float32)
        d = config.TOKEN_EMBEDDINGS_SIZE
        D = config.CODE_VECTOR_SIZE
        g = torch.Generator(device='cpu').manual_seed(20260911)

        def uniform(shape, limit):
            t = torch.empty(shape, dtype=torch.float32)
            t.uniform_(-limit, limit, generator=g)
            return t.to(self.device)

        # Initializers match the reference (tensorflow_model.py:205-220,:249-250):
        # variance_scaling(scale=1, fan_out, uniform) for the three tables
        # → U(±√(3/fan_out)); glorot uniform for TRANSFORM and ATTENTION.
        self.tok_table = uniform((token_vocab_size, d), math.sqrt(3.0 / d))
        self.path_table = uniform((path_vocab_size, d), math.sqrt(3.0 / d))
        self.target_table = uniform((target_vocab_size, D), math.sqrt(3.0 / D))
        self.w = uniform((D, D), math.sqrt(6.0 / (D + D)))       # (in, out)
        self.a = uniform((D,), math.sqrt(6.0 / (D + 1)))

        # Adam state (fp32), lazy for the embedding tables.
        self.adam_step = 0
        self._adam_m: Dict[str, torch.Tensor] = {}
        self._adam_v: Dict[str, torch.Tensor] = {}
        for name in self.param_names():
            p = self.get_param(name)
            self._adam_m[name] = torch.zeros_like(p)
            self._adam_v[name] = torch.zeros_like(p)

        # bf16 (compute-dtype) shadows for GEMM operands.
        self._refresh_shadows()
        self._seed_counter = torch.Generator(device='cpu').manual_seed(4242).initial_seed()
        self._step_ctr = 0
        # Device-resident RNG seed and Adam step counters: under hipGraph
        # capture these are advanced by in-graph tensor ops so every replay
        # sees fresh values (a host scalar would be baked into the capture).
        self._seed_t = None
        self._step_t = None
        self._side_stream = None
        self._hash_stream = None
        if self.device.type == 'cuda':
            self._seed_t = torch.zeros(1, dtype=torch.int64, device=self.device)
            self._step_t = torch.zeros(1, dtype=torch.int32, device=self.device)
            # side stream for the independent target-table chain
            # (d_target GEMM -> all-reduce -> Adam) overlapping main backward
            self._side_stream = torch.cuda.Stream()
            # second side stream: the sparse-grad hash build depends only on
            # the input ids, so it runs under the FORWARD pass
            self._hash_stream = torch.cuda.Stream()

    # ---- parameters ----

    @staticmethod
    def param_names():
        return ['tok_table', 'path_table', 'target_table', 'w', 'a']

    def get_param(self, name: str) -> torch.Tensor:
        return getattr(self, name)

    def num_trainable_params(self) -> int:
        return sum(self.get_param(n).numel() for n in self.param_names())

    def _refresh_shadows(self, only_w: bool = False):
        """Refresh the compute-dtype shadows IN PLACE. The buffers must keep
        their identity: a hipGraph-captured step records reads of these exact
        allocations, so rebinding fresh tensors here would leave every replay
        computing with stale capture-time weights (found the hard way — the
        graph-vs-eager test caught a 2e-3 parameter drift)."""
        cd = self.compute_dtype
        if getattr(self, 'w_oi', None) is None:
            self.w_oi = self.w.t().contiguous().to(cd)  # (out,in): fwd B^T
            self.w_io = self.w.contiguous().to(cd)      # (in,out): bwd B^T
            self.a_c = self.a.to(torch.float32)
        else:
            self.w_oi.copy_(self.w.t())
            self.w_io.copy_(self.w)
            self.a_c.copy_(self.a)
        if not only_w:
            if getattr(self, 'target_shadow', None) is None:
                self.target_shadow = self.target_table.to(cd)
            else:
                self.target_shadow.copy_(self.target_table)

    def state_dict(self) -> Dict[str, torch.Tensor]:
        sd = {n: self.get_param(n).cpu() for n in self.param_names()}
        sd['adam_step'] = torch.tensor(self.adam_step)
        # dropout stream position: resuming mid-training replays the exact
        # per-step seeds the uninterrupted run would have used
        sd['dropout_step'] = torch.tensor(self._step_ctr)
        for n in self.param_names():
            sd['adam_m.' + n] = self._adam_m[n].cpu()
            sd['adam_v.' + n] = self._adam_v[n].cpu()
        return sd

    def weights_state_dict(self) -> Dict[str, torch.Tensor]:
        """Release form: weights only, optimizer state stripped
        (reference release flow, tensorflow_model.py:129-136)."""
        return {n: self.get_param(n).cpu() for n in self.param_names()}

    def load_state_dict(self, sd: Dict[str, torch.Tensor]):
        for n in self.param_names():
            self.get_param(n).copy_(sd[n].to(self.device))
        if 'dropout_step' in sd:
            self._step_ctr = int(sd['dropout_step'])
        if 'adam_step' in sd:
            self.adam_step = int(sd['adam_step'])
            for n in self.param_names():
                if ('adam_m.' + n) in sd:
                    self._adam_m[n].copy_(sd['adam_m.' + n].to(self.device))
                    self._adam_v[n].copy_(sd['adam_v.' + n].to(self.device))
        self._refresh_shadows()
        if self._step_t is not None:
            self._step_t.fill_(self.adam_step)
        if self._seed_t is not None:
            # Restore the GPU dropout seed stream to where the uninterrupted
            # run would be: each training step advances _seed_t by 2654435761
            # (train_step pre-advances; forward advances when called outside
            # train_step), so after _step_ctr steps it holds
            # _step_ctr*2654435761 with int64 wraparound semantics.
            v = (self._step_ctr * 2654435761) & 0xFFFFFFFFFFFFFFFF
            if v >= 1 << 63:
                v -= 1 << 64
            self._seed_t.fill_(v)

    # ---- forward ----

    def forward(self, src_ids, path_ids, tgt_ids, valid_mask,
                training: bool, _seed_preadvanced: bool = False,
                _post_ctx_hook=None) -> ForwardState:
        B, C = src_ids.shape
        D = self.config.CODE_VECTOR_SIZE
        self._step_ctr += 1
        seed = (self._seed_counter + self._step_ctr * 2654435761) & 0x7FFFFFFFFFFFFFFF
        seed_t = None
        if training and self._seed_t is not None:
            if not _seed_preadvanced:
                self._seed_t.add_(2654435761)  # in-graph advance (capture-safe)
            seed_t = self._seed_t
            seed = 0
        keep = self.config.DROPOUT_KEEP_RATE if training else 1.0
        ctx = F.gather_concat_fwd(self.tok_table, self.path_table, src_ids,
                                  path_ids, tgt_ids, keep, seed, training,
                                  out_dtype=self.compute_dtype, seed_t=seed_t)
        if _post_ctx_hook is not None:
            _post_ctx_hook()
        comb = F.transform_tanh_fwd(ctx, self.w_oi)                  # (B*C, D)
        code, alpha = F.attention_fwd(comb.reshape(B, C, D), self.a_c, valid_mask)
        return ForwardState(ctx=ctx, comb=comb, code=code, alpha=alpha,
                            seed=seed, seed_t=seed_t)

    def logits(self, code: torch.Tensor) -> torch.Tensor:
        """code (B,D) fp32 → (B, V_tgt) compute dtype:
        256-tile MFMA kernel on big shapes, hipBLASLt otherwise."""
        return F.logits_gemm(code.to(self.compute_dtype), self.target_shadow)

    # ---- full training step ----

    def train_step(self, src_ids, path_ids, tgt_ids, valid_mask, labels,
                   reducer: Optional[NullReducer] = None) -> torch.Tensor:
        """One full optimizer step. Returns the (device-resident, unsynced)
        scalar mean loss so callers control when to pay the H2D sync."""
        reducer = reducer or NullReducer()
        cfg = self.config
        B, C = src_ids.shape
        D = cfg.CODE_VECTOR_SIZE
        dt = cfg.TOKEN_EMBEDDINGS_SIZE

        # Launch the sparse-grad hash build (claim/compact/lookup over the
        # token and path ids) on its own stream NOW: it has no gradient
        # dependency, so it hides under the forward GEMMs instead of
        # serializing in the backward tail (~0.45 ms/step on Zipf ids).
        tok_ids = torch.cat([src_ids.reshape(-1), tgt_ids.reshape(-1)])
        path_ids_flat = path_ids.reshape(-1)
        # Scheduling note (measured, counterintuitive): the seed add_ kernel
        # launched AFTER the hash build gets CU-starved under the claim
        # kernel's atomic storm (~360 us in gpurun_out/prof_r02samp), and the
        # forward's gather waits on it — but that implicit serialization is
        # FASTER than releasing the gather early: gather+claim thrash when
        # concurrent (394 vs 163 us gather), and a later claim start pushes
        # the hash tail into the sparse-Adam critical path. Same-box A/B
        # (gpurun_out/r02z_ab.log, r02z2_ab.log): pre-advancing the seed
        # costs ~45 us/step; post-gather hash placement costs ~150 us/step.
        # Both variants stay env-gated for re-measurement on new hardware.
        seed_pre = False
        if (self._seed_t is not None
                and os.environ.get('C2V_SEED_PRE', '0') == '1'):
            self._seed_t.add_(2654435761)
            seed_pre = True
        tok_state = path_state = None
        hash_done = None
        hash_box = {}

        def _enqueue_hash_build():
            ev = torch.cuda.Event()
            ev.record()
            with torch.cuda.stream(self._hash_stream):
                self._hash_stream.wait_event(ev)
                hash_box['tok'] = F.sparse_hash_build(tok_ids)
                hash_box['path'] = F.sparse_hash_build(path_ids_flat)
                done = torch.cuda.Event()
                done.record()
                hash_box['done'] = done
            if not torch.cuda.is_current_stream_capturing():
                tok_ids.record_stream(self._hash_stream)
                path_ids_flat.record_stream(self._hash_stream)

        hash_overlap = (self._hash_stream is not None
                        and os.environ.get('C2V_HASH_OVERLAP', '1') == '1')
        # Step-start placement measured fastest (see scheduling note above);
        # C2V_HASH_POS=postgather enqueues the build after the gather stage
        # via the forward hook instead, for re-measurement.
        hash_at_start = os.environ.get('C2V_HASH_POS', 'start') == 'start'
        post_ctx_hook = None
        if hash_overlap and hash_at_start:
            _enqueue_hash_build()
        elif hash_overlap:
            post_ctx_hook = _enqueue_hash_build

        st = self.forward(src_ids, path_ids, tgt_ids, valid_mask, training=True,
                          _seed_preadvanced=seed_pre,
                          _post_ctx_hook=post_ctx_hook)
        tok_state = hash_box.get('tok')
        path_state = hash_box.get('path')
        hash_done = hash_box.get('done')
        code_c = st.code.to(self.compute_dtype)
        S = int(cfg.SAMPLED_SOFTMAX_SIZE)
        V = self.target_table.shape[0]
        use_sampled = 0 < S < V

        if use_sampled:
            # Sampled-softmax path (BASELINE config 4): candidate set =
            # [B local labels | S shared log-uniform negatives]; the target
            # table gets SPARSE row grads (all-gathered under DP like the
            # embedding tables) instead of the dense 200 MB all-reduce.
            from ..ops.reference import sample_log_uniform
            sampled = sample_log_uniform(S, V, code_c.device)
            cand = torch.cat([labels, sampled])                  # (B+S,)
            # candidate logits/backward GEMMs run with the candidate-row
            # gather fused into the kernels' B staging — no w_cand
            # materialization, no torch GEMM in the sampled path; the
            # log-uniform corrections are computed inline in the CE kernels
            logits_cand = F.sampled_logits_gemm(code_c, self.target_shadow,
                                                cand)            # (B, B+S)
            loss_rows, lse = F.sampled_ce_fwd(logits_cand, labels, sampled, V)
            loss = loss_rows.float().mean()
            d_cand = F.sampled_ce_bwd(logits_cand, labels, sampled, V,
                                      lse, 1.0 / B)
            d_target_rows = F.sampled_bwd_target_rows(d_cand, code_c)
            if (reducer.world_size > 1
                    and os.environ.get('C2V_DP_DEDUP', '1') == '1'):
                # labels repeat across the candidate set: dedup before gather
                (cand_g, target_rows_g), = reducer.allgather_sparse_dedup(
                    [F.sparse_dedup_sum(cand, d_target_rows)]).wait()
            else:
                cand_g, target_rows_g = reducer.allgather_sparse(
                    cand, d_target_rows)
            d_code = F.sampled_bwd_code(d_cand, self.target_shadow,
                                        cand)                    # (B,D)
            d_target = None
        else:
            logits, loss_rows, lse = F.logits_ce_fused(code_c,
                                                       self.target_shadow,
                                                       labels)
            loss = loss_rows.float().mean()
            # C2V_FUSED_CEBWD=1: CE backward computed inside the consumers'
            # staging (gemm_nn_splitk_ce / gemm_tn_ce) — d_logits is never
            # materialized (saves a 535 MB write + two 535 MB reads). Kept
            # opt-in: measured net-neutral-to-slower at DP=1 because the tn
            # d_target kernel it requires is slower than hipBLASLt there
            # (profiles/r01_optimization_log.md).
            ce_scale = 1.0 / B
            ce_mode = F.ce_bwd_mode(logits)
            if ce_mode == 1:
                d_logits = None
                d_code = F.logits_bwd_code_ce(logits, self.target_shadow,
                                              lse, labels, ce_scale)
            elif ce_mode == 2:
                d_code, d_logits = F.logits_bwd_code_ce_write(
                    logits, self.target_shadow, lse, labels, ce_scale)
            else:
                d_logits = F.ce_bwd(logits, lse, labels, ce_scale)
                d_code = F.logits_bwd_code(d_logits, self.target_shadow)
            d_target = None  # see target-chain dispatch below

        side_done = None
        if not use_sampled:
            # lr_t must exist before the side chain runs its Adam step
            self.adam_step += 1
            t_pre, lr = self.adam_step, cfg.ADAM_LR
            b1, b2, eps = cfg.ADAM_BETA1, cfg.ADAM_BETA2, cfg.ADAM_EPS
            st_t_pre = None
            if self._step_t is not None:
                self._step_t.add_(1)
                st_t_pre = F.adam_lrt(self._step_t, lr, b1, b2)
            if self._side_stream is not None:
                # the whole target chain — the big GEMM, the (DP) all-reduce
                # and the 100M-param Adam — is independent of the rest of
                # backward; run it concurrently on the side stream
                ev = torch.cuda.Event()
                ev.record()
                with torch.cuda.stream(self._side_stream):
                    self._side_stream.wait_event(ev)
                    if ce_mode == 1:
                        d_target = F.logits_bwd_target_ce(
                            logits, code_c, lse, labels, ce_scale)
                    else:
                        d_target = F.logits_bwd_target(d_logits, code_c)
                    reducer.allreduce_dense('target_table', d_target)
                    reducer.wait('target_table')
                    F.adam_dense_step(self.target_table, d_target,
                                      self._adam_m['target_table'],
                                      self._adam_v['target_table'],
                                      t_pre, lr, cfg.ADAM_BETA1,
                                      cfg.ADAM_BETA2, cfg.ADAM_EPS,
                                      shadow=self.target_shadow,
                                      lrt_t=st_t_pre)
                    side_done = torch.cuda.Event()
                    side_done.record()
                if not torch.cuda.is_current_stream_capturing():
                    side_inputs = ((logits, lse, labels, code_c)
                                   if ce_mode == 1 else (d_logits, code_c))
                    for t in side_inputs:
                        t.record_stream(self._side_stream)
                    if st_t_pre is not None:
                        st_t_pre.record_stream(self._side_stream)
            else:
                if ce_mode == 1:
                    d_target = F.logits_bwd_target_ce(logits, code_c, lse,
                                                      labels, ce_scale)
                else:
                    d_target = F.logits_bwd_target(d_logits, code_c)
                reducer.allreduce_dense('target_table', d_target)

        # attention backward emits dL/dz directly (tanh' fused into its
        # d_comb write); the dX GEMM applies the dropout mask in its epilogue
        # — no separate tanh_bwd or dropout_bwd passes
        d_z3, d_a = F.attention_bwd(st.comb.reshape(B, C, D), self.a_c,
                                    st.alpha, d_code, fuse_tanh_bwd=True)
        d_z = d_z3.reshape(B * C, D)
        d_ctx, d_w = F.linear_bwd_dropout(d_z, self.w_io, st.ctx,
                                          cfg.DROPOUT_KEEP_RATE, st.seed,
                                          seed_t=st.seed_t, training=True)
        reducer.allreduce_dense('w', d_w)
        reducer.allreduce_dense('a', d_a)

        # Sparse embedding grads: (ids, rows) pairs; under DP these are
        # all-gathered (not dense-all-reduced) — SURVEY §2.4. Single-process
        # runs skip row materialization entirely (grads read straight out of
        # the d_ctx slices by the scatter-Adam kernel).
        ctx_direct = reducer.world_size == 1
        pending_gather = None
        if hash_done is not None:
            torch.cuda.current_stream().wait_event(hash_done)
            if not torch.cuda.is_current_stream_capturing():
                cur = torch.cuda.current_stream()
                for t in (tok_state or ()) + (path_state or ()):
                    t.record_stream(cur)
        if not ctx_direct:
            sparse_mode = os.environ.get('C2V_DP_SPARSE', 'dedup')
            if sparse_mode == 'owner':
                # owner-sharded reduce (opt-in): id%N all-to-all + disjoint
                # reduced-shard gather — ~half the ring bytes of the dedup
                # gather at java14m Zipf shapes (ddp.reduce_sparse_owner)
                if tok_state is not None:
                    tok_e = F.sparse_dedup_sum_ctx_pre(tok_state, d_ctx,
                                                       0, 2 * dt, 2, dt)
                    path_e = F.sparse_dedup_sum_ctx_pre(path_state, d_ctx,
                                                        dt, dt, 1, dt)
                else:
                    tok_e = F.sparse_dedup_sum_ctx(tok_ids, d_ctx,
                                                   0, 2 * dt, 2, dt)
                    path_e = F.sparse_dedup_sum_ctx(path_ids_flat, d_ctx,
                                                    dt, dt, 1, dt)
                (tok_ids, tok_rows), (path_ids_flat, path_rows) = \
                    reducer.reduce_sparse_owner([tok_e, path_e])
            elif os.environ.get('C2V_DP_DEDUP', '1') == '1':
                # rank-local dedup+sum before the gather: ships each unique
                # row once (3-5x fewer xGMI bytes on Zipf-shaped real ids);
                # the gather is launched async and waited just before the
                # sparse Adam so the dense w/a chain overlaps the comm
                if tok_state is not None:
                    tok_e = F.sparse_dedup_sum_ctx_pre(tok_state, d_ctx,
                                                       0, 2 * dt, 2, dt)
                    path_e = F.sparse_dedup_sum_ctx_pre(path_state, d_ctx,
                                                        dt, dt, 1, dt)
                else:
                    tok_e = F.sparse_dedup_sum_ctx(tok_ids, d_ctx,
                                                   0, 2 * dt, 2, dt)
                    path_e = F.sparse_dedup_sum_ctx(path_ids_flat, d_ctx,
                                                    dt, dt, 1, dt)
                pending_gather = reducer.allgather_sparse_dedup(
                    [tok_e, path_e])
            else:
                tok_rows = torch.cat([d_ctx[:, :dt], d_ctx[:, 2 * dt:]], dim=0)
                path_rows = d_ctx[:, dt:2 * dt]
                tok_ids, tok_rows = reducer.allgather_sparse(tok_ids, tok_rows)
                path_ids_flat, path_rows = reducer.allgather_sparse(
                    path_ids_flat, path_rows)

        # ---- optimizer (TF AdamOptimizer formulation) ----
        if use_sampled:
            self.adam_step += 1
            st_t = None
            if self._step_t is not None:
                self._step_t.add_(1)
                st_t = F.adam_lrt(self._step_t, cfg.ADAM_LR, cfg.ADAM_BETA1,
                                  cfg.ADAM_BETA2)
        else:
            st_t = st_t_pre  # counters already advanced pre-backward
        t, lr = self.adam_step, cfg.ADAM_LR
        b1, b2, eps = cfg.ADAM_BETA1, cfg.ADAM_BETA2, cfg.ADAM_EPS
        if ctx_direct:
            if tok_state is not None:
                F.adam_sparse_rows_from_ctx_pre(
                    self.tok_table, tok_state, d_ctx, 0, 2 * dt, 2, dt,
                    self._adam_m['tok_table'], self._adam_v['tok_table'],
                    t, lr, b1, b2, eps, lrt_t=st_t)
                F.adam_sparse_rows_from_ctx_pre(
                    self.path_table, path_state, d_ctx, dt, dt, 1, dt,
                    self._adam_m['path_table'], self._adam_v['path_table'],
                    t, lr, b1, b2, eps, lrt_t=st_t)
            else:
                F.adam_sparse_rows_from_ctx(
                    self.tok_table, tok_ids, d_ctx, 0, 2 * dt, 2, dt,
                    self._adam_m['tok_table'], self._adam_v['tok_table'],
                    t, lr, b1, b2, eps, lrt_t=st_t)
                F.adam_sparse_rows_from_ctx(
                    self.path_table, path_ids_flat, d_ctx, dt, dt, 1, dt,
                    self._adam_m['path_table'], self._adam_v['path_table'],
                    t, lr, b1, b2, eps, lrt_t=st_t)
        # dense w/a Adam before the sparse-table updates: under DP this work
        # (and its small all-reduce waits) overlaps the in-flight sparse
        # gather instead of queueing behind it
        reducer.wait('w')
        F.adam_dense_step(self.w, d_w, self._adam_m['w'], self._adam_v['w'],
                          t, lr, b1, b2, eps, lrt_t=st_t)
        reducer.wait('a')
        F.adam_dense_step(self.a, d_a, self._adam_m['a'], self._adam_v['a'],
                          t, lr, b1, b2, eps, lrt_t=st_t)
        self._refresh_shadows(only_w=True)
        if not ctx_direct:
            if pending_gather is not None:
                (tok_ids, tok_rows), (path_ids_flat, path_rows) = \
                    pending_gather.wait()
            F.adam_sparse_rows_step(self.tok_table, tok_ids, tok_rows,
                                    self._adam_m['tok_table'], self._adam_v['tok_table'],
                                    t, lr, b1, b2, eps, lrt_t=st_t)
            F.adam_sparse_rows_step(self.path_table, path_ids_flat, path_rows,
                                    self._adam_m['path_table'], self._adam_v['path_table'],
                                    t, lr, b1, b2, eps, lrt_t=st_t)
        if use_sampled:
            F.adam_sparse_rows_step(self.target_table, cand_g, target_rows_g,
                                    self._adam_m['target_table'],
                                    self._adam_v['target_table'],
                                    t, lr, b1, b2, eps,
                                    shadow=self.target_shadow, lrt_t=st_t)
        elif side_done is not None:
            # join the side-stream target chain
            torch.cuda.current_stream().wait_event(side_done)
        else:
            reducer.wait('target_table')
            F.adam_dense_step(self.target_table, d_target,
                              self._adam_m['target_table'],
                              self._adam_v['target_table'],
                              t, lr, b1, b2, eps, shadow=self.target_shadow,
                              lrt_t=st_t)
        return loss

    def adam_step_host_sync(self, delta: int = 0):
        """Keep the HOST adam_step mirror in sync when the device counter is
        advanced by a graph replay (the python train_step body didn't run)."""
        self.adam_step += delta

    # ---- hipGraph-captured training step ----

    def make_graph_step(self, batch_size: int):
        return GraphTrainStep(self, batch_size)

    # ---- evaluation / prediction forward ----

    @torch.no_grad()
    def eval_batch(self, src_ids, path_ids, tgt_ids, valid_mask, labels,
                   top_k: int):
        """predict_batch + per-row CE against `labels` (the reference Keras
        backend reports evaluation loss — keras_model.py:166-228). Returns
        (indices, scores, code, alpha, loss_sum) with loss_sum an unsynced
        device scalar (sum over the batch, fp32)."""
        st = self.forward(src_ids, path_ids, tgt_ids, valid_mask, training=False)
        logits = self.logits(st.code)
        loss_rows, _ = F.ce_fwd(logits, labels)
        scores, indices = F.topk(logits, k=min(top_k, logits.shape[1]))
        return indices, scores, st.code, st.alpha, loss_rows.float().sum()

    @torch.no_grad()
    def predict_batch(self, src_ids, path_ids, tgt_ids, valid_mask, top_k: int,
                      normalize_scores: bool = False):
        """Returns (topk_indices (B,k) int64, topk_scores (B,k) fp32,
        code (B,D) fp32, alpha (B,C) fp32)."""
        st = self.forward(src_ids, path_ids, tgt_ids, valid_mask, training=False)
        logits = self.logits(st.code)
        # k is clamped to the vocab size (reference: tensorflow_model.py:299-301)
        scores, indices = F.topk(logits, k=min(top_k, logits.shape[1]))
        if normalize_scores:
            scores = torch.softmax(scores, dim=1)
        return indices, scores, st.code, st.alpha


class GraphTrainStep:
    """hipGraph-captured training step (single-process path).

    Captures the ENTIRE train_step — forward, backward, Adam — into one HIP
    graph with static input buffers. All step-varying state (dropout seed,
    Adam bias-correction step) lives in device memory and is advanced by
    captured tensor ops, so each replay is a fresh, correct optimizer step
    with one graph launch instead of ~40 Python-driven kernel launches."""

    def __init__(self, net: Code2VecNetwork, batch_size: int, warmup: int = 2):
        assert net.device.type == 'cuda', 'graph capture needs a GPU'
        self.net = net
        C = net.config.MAX_CONTEXTS
        dev = net.device
        self.src = torch.zeros(batch_size, C, dtype=torch.int32, device=dev)
        self.pth = torch.zeros(batch_size, C, dtype=torch.int32, device=dev)
        self.tgt = torch.zeros(batch_size, C, dtype=torch.int32, device=dev)
        self.mask = torch.ones(batch_size, C, dtype=torch.float32, device=dev)
        self.labels = torch.ones(batch_size, dtype=torch.int64, device=dev)

        # The warmup steps are REAL optimizer steps on dummy batches (the
        # allocator must see the exact allocation pattern before capture);
        # snapshot all mutable state so they leave no trace on the model.
        snap_params = {n: net.get_param(n).detach().clone()
                       for n in net.param_names()}
        snap_m = {n: net._adam_m[n].detach().clone() for n in net.param_names()}
        snap_v = {n: net._adam_v[n].detach().clone() for n in net.param_names()}
        snap_adam_step, snap_step_ctr = net.adam_step, net._step_ctr
        snap_seed_t = net._seed_t.clone()
        snap_step_t = net._step_t.clone()

        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(warmup):  # real steps (python body runs: host and
                net.train_step(self.src, self.pth, self.tgt, self.mask,
                               self.labels)   # device counters both advance)
        torch.cuda.current_stream().wait_stream(side)

        self.graph = torch.cuda.CUDAGraph()
        # thread_local: background threads (reader workers pinning batches)
        # may touch the CUDA API while this thread captures
        with torch.cuda.graph(self.graph, capture_error_mode='thread_local'):
            self.loss = net.train_step(self.src, self.pth, self.tgt,
                                       self.mask, self.labels)
        # Capture RECORDS the device ops without executing them (only the
        # python-side counters moved). Roll every piece of state the warmup
        # touched back to the snapshot — in place, so the capture's recorded
        # reads still point at the live allocations.
        torch.cuda.synchronize()
        for n in net.param_names():
            net.get_param(n).copy_(snap_params[n])
            net._adam_m[n].copy_(snap_m[n])
            net._adam_v[n].copy_(snap_v[n])
        net.adam_step, net._step_ctr = snap_adam_step, snap_step_ctr
        net._seed_t.copy_(snap_seed_t)
        net._step_t.copy_(snap_step_t)
        net._refresh_shadows()

    def step(self, src, pth, tgt, mask, labels) -> torch.Tensor:
        self.src.copy_(src)
        self.pth.copy_(pth)
        self.tgt.copy_(tgt)
        self.mask.copy_(mask)
        self.labels.copy_(labels)
        self.graph.replay()
        self.net.adam_step_host_sync(+1)
        return self.loss
