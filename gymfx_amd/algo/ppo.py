"""PPO trainer: on-device rollout -> GAE -> minibatch clipped-PPO updates.

The whole loop is device-resident: env steps are the fused HIP kernels
(VecFxEnv native path), the policy is the MFMA MLP (models/mlp.py), GAE /
advantage-normalize / loss-backward / Adam are the ops/api kernels.  On CPU
the same code runs against the torch oracle implementations (tested by the
CPU suite, gloo multi-rank included).

Data parallelism (BASELINE config #3): one process per GPU; gradients live
in ONE flat bucket (model.grads) all-reduced per minibatch over RCCL/xGMI
(parallel/ddp.py) — gradient volume is ~0.4 MB so the collective is
latency-bound and a single fused bucket is optimal on this topology.
"""
from __future__ import annotations

import time
from dataclasses import dataclass
from typing import Any, Dict, List

import torch

from ..envs.vec_env import VecFxEnv
from ..models.lstm import ActorCriticLSTM
from ..models.mlp import ActorCriticMLP
from ..ops import api


@dataclass
class PPOConfig:
    rollout_steps: int = 128
    ppo_epochs: int = 4
    minibatches: int = 8
    gamma: float = 0.99
    gae_lambda: float = 0.95
    clip_eps: float = 0.2
    ent_coef: float = 0.01
    vf_coef: float = 0.5
    lr: float = 3e-4
    max_grad_norm: float = 0.5
    seed: int = 0
    hidden: int = 256
    normalize_adv: bool = True
    shuffle_rows: bool = True
    use_graphs: bool = True  # hipGraph-capture the update on GPU
    policy: str = "mlp"      # "mlp" | "lstm" (BASELINE configs #2 / #4)
    bptt_len: int = 16       # sequence-chunked BPTT length (lstm)
    fused_rollout: bool = False  # single-kernel MLP policy step (bitwise-
                                 # identical option; measured SLOWER: the
                                 # mega-kernel spills ~2.5k SGPRs and runs
                                 # at occupancy 1 — kept for the record)
    rollout_streams: int = 1  # >1: split rollout across HIP streams (GPU).
                             # Measured neutral-to-slightly-negative at
                             # N=4096 (per-kernel latency does not shrink
                             # with width) — kept for bigger fleets.
    fuse_sample: bool = True  # sample the action inside the env-step kernel
                              # (native engine): one fewer launch per rollout
                              # step; bitwise == the sample_head kernel.
    fuse_head: bool = False   # compute head = h @ W3 + b3 INSIDE the
                              # env-step kernel.  Measured SLOWER (rollout
                              # 6.4 -> 10.1 ms at N=4096): the env kernel
                              # runs only N/256 workgroups, so the per-lane
                              # 1024-FMA scalar dot serializes instead of
                              # the 6.6 us MFMA head GEMM it replaced —
                              # kept as a knob for much wider fleets
                              # (profiles/PERF_NOTES.md).
    overlap_gather: bool = False  # run minibatch gather on a side HIP
                                  # stream overlapped with the previous
                                  # minibatch's fwd/bwd (ping-pong slots).
                                  # Measured SLOWER at N=4096 (23.1 vs 22.0
                                  # ms/update): gather and the GEMMs are
                                  # both HBM-bound, so concurrency only
                                  # splits bandwidth and adds event-sync
                                  # overhead.  Kept with a bit-equality
                                  # GPU test, like fused_rollout.

    @classmethod
    def from_config(cls, cfg: Dict[str, Any]) -> "PPOConfig":
        out = cls()
        mapping = {
            "rollout_steps": "rollout_steps",
            "ppo_epochs": "ppo_epochs",
            "minibatches": "minibatches",
            "gamma": "gamma",
            "gae_lambda": "gae_lambda",
            "clip_eps": "clip_eps",
            "ent_coef": "ent_coef",
            "vf_coef": "vf_coef",
            "lr": "learning_rate",
            "max_grad_norm": "max_grad_norm",
            "hidden": "hidden_size",
            "normalize_adv": "normalize_adv",
            "shuffle_rows": "shuffle_rows",
            "use_graphs": "use_graphs",
            "policy": "policy_model",
            "bptt_len": "bptt_len",
            "rollout_streams": "rollout_streams",
            "fused_rollout": "fused_rollout",
            "fuse_sample": "fuse_sample",
            "fuse_head": "fuse_head",
            "overlap_gather": "overlap_gather",
        }
        from ..config.merger import convert_type

        for attr, key in mapping.items():
            if cfg.get(key) is not None:
                cur = getattr(out, attr)
                # bool("false") is True — route strings through the config
                # layer's typed coercion first
                val = convert_type(cfg[key]) if isinstance(cur, bool) else cfg[key]
                setattr(out, attr, type(cur)(val))
        if cfg.get("seed") is not None:
            out.seed = int(cfg["seed"])
        return out


class PPOTrainer:
    def __init__(self, env: VecFxEnv, cfg: PPOConfig, *, rank: int = 0,
                 world_size: int = 1, process_group=None):
        self.env = env
        self.cfg = cfg
        self.rank = rank
        self.world_size = world_size
        self.pg = process_group
        self.device = env.device
        if getattr(env.params, "action_space_mode", "discrete") != "discrete":
            raise ValueError(
                "PPO training uses the 3-way discrete head; set "
                "action_space_mode='discrete' (continuous mode is for "
                "external agents driving the env, reference "
                "env.py:343-360 coercion semantics)")

        N = env.n_envs
        T = cfg.rollout_steps
        D = env.obs_dim
        self.N, self.T, self.D = N, T, D
        n_actions = 3

        # model init is rank-INDEPENDENT (data-parallel replicas must start
        # identical); rollout sampling / shuffling seeds differ per rank.
        self.recurrent = cfg.policy == "lstm"
        if self.recurrent:
            if T % cfg.bptt_len != 0:
                raise ValueError("rollout_steps must divide by bptt_len")
            if not cfg.shuffle_rows:
                raise ValueError("recurrent PPO requires shuffle_rows")
            self.model = ActorCriticLSTM(
                D, n_actions, cfg.hidden, device=self.device, seed=cfg.seed
            )
        else:
            self.model = ActorCriticMLP(
                D, n_actions, cfg.hidden, device=self.device, seed=cfg.seed
            )

        dev = self.device
        self.obs_buf = torch.empty(T, N, D, dtype=torch.bfloat16, device=dev)
        self.act_buf = torch.empty(T, N, dtype=torch.int64, device=dev)
        self.logp_buf = torch.empty(T, N, dtype=torch.float32, device=dev)
        self.val_buf = torch.empty(T + 1, N, dtype=torch.float32, device=dev)
        self.rew_buf = torch.empty(T, N, dtype=torch.float32, device=dev)
        self.done_buf = torch.empty(T, N, dtype=torch.bool, device=dev)
        self.adv_buf = torch.empty(T, N, dtype=torch.float32, device=dev)
        self.ret_buf = torch.empty(T, N, dtype=torch.float32, device=dev)
        self._adv_part = torch.empty(512, dtype=torch.float32, device=dev)

        self.acts_rollout = self.model.alloc_acts(N)
        self.obs_bf16_step = torch.empty(N, D, dtype=torch.bfloat16, device=dev)

        # flat views (fixed addresses — required for hipGraph capture)
        TN = T * N
        self.obs_flat = self.obs_buf.view(TN, D)
        self.act_flat = self.act_buf.view(TN)
        self.logp_flat = self.logp_buf.view(TN)
        self.adv_flat = self.adv_buf.view(TN)
        self.ret_flat = self.ret_buf.view(TN)
        self.losses = torch.zeros(5, dtype=torch.float32, device=dev)

        if self.recurrent:
            H = cfg.hidden
            L = cfg.bptt_len
            self.n_chunks = T // L
            n_seq = self.n_chunks * N
            if n_seq % cfg.minibatches != 0:
                raise ValueError("(T/bptt_len)*n_envs must divide by minibatches")
            Mseq = n_seq // cfg.minibatches
            self.mseq = Mseq
            self.mb_rows = L * Mseq  # loss rows per minibatch
            # persistent rollout recurrent state + chunk-boundary snapshots
            self.rnn_state = self.model.alloc_state(N)
            self.h0_buf = torch.zeros(self.n_chunks, N, H, dtype=torch.float32, device=dev)
            self.c0_buf = torch.zeros(self.n_chunks, N, H, dtype=torch.float32, device=dev)
            # sequence minibatch buffers (time-major)
            self.obs_mb_seq = torch.empty(L, Mseq, D, dtype=torch.bfloat16, device=dev)
            self.act_mb = torch.empty(L, Mseq, dtype=torch.int64, device=dev)
            self.logp_mb = torch.empty(L, Mseq, dtype=torch.float32, device=dev)
            self.adv_mb = torch.empty(L, Mseq, dtype=torch.float32, device=dev)
            self.ret_mb = torch.empty(L, Mseq, dtype=torch.float32, device=dev)
            self.done_mb = torch.empty(L, Mseq, dtype=torch.bool, device=dev)
            self.h0_mb = torch.empty(Mseq, H, dtype=torch.bfloat16, device=dev)
            self.c0_mb = torch.empty(Mseq, H, dtype=torch.float32, device=dev)
            self.bptt = self.model.alloc_bptt(L, Mseq)
            self.dhead = torch.empty(L * Mseq, n_actions + 1,
                                     dtype=torch.bfloat16, device=dev)
            self._boot_state = self.model.alloc_state(N)
            self._overlap = False  # BPTT buffers are large; single slot
        else:
            M = TN // cfg.minibatches
            if M * cfg.minibatches != TN:
                raise ValueError("rollout_steps * n_envs must divide by minibatches")
            self.mb_rows = M
            self.acts_train = self.model.alloc_acts(M)
            self.scratch = self.model.alloc_scratch(M)
            self.dhead = torch.empty(M, n_actions + 1, dtype=torch.bfloat16, device=dev)
            # minibatch gather targets — 2 ping-pong slots when the gather
            # overlaps the previous minibatch's fwd/bwd on a side stream
            self._overlap = (self.device.type == "cuda" and cfg.overlap_gather
                             and cfg.shuffle_rows)
            ns = 2 if self._overlap else 1
            self.obs_mb_s = [torch.empty(M, D, dtype=torch.bfloat16, device=dev)
                             for _ in range(ns)]
            self.act_mb_s = [torch.empty(M, dtype=torch.int64, device=dev)
                             for _ in range(ns)]
            self.logp_mb_s = [torch.empty(M, dtype=torch.float32, device=dev)
                              for _ in range(ns)]
            self.adv_mb_s = [torch.empty(M, dtype=torch.float32, device=dev)
                             for _ in range(ns)]
            self.ret_mb_s = [torch.empty(M, dtype=torch.float32, device=dev)
                             for _ in range(ns)]
            # slot-0 aliases (noshuffle path, tests)
            self.obs_mb = self.obs_mb_s[0]
            self.act_mb = self.act_mb_s[0]
            self.logp_mb = self.logp_mb_s[0]
            self.adv_mb = self.adv_mb_s[0]
            self.ret_mb = self.ret_mb_s[0]

        # device counters (hipGraph-replayable RNG / schedule state)
        self.step_base = torch.zeros((), dtype=torch.int64, device=dev)
        self.mb_ctr = torch.zeros((), dtype=torch.int64, device=dev)
        self.sample_seed = cfg.seed * 1_000_003 + rank
        self.shuffle_seed = cfg.seed * 9973 + rank

        # gather+first-GEMM fusion (ROADMAP lever 3): MEASURED NEGATIVE at
        # the flagship shape — update 12.0 -> 18.0 ms: the per-lane Feistel
        # recompute in every staging loop of L1-fwd and W1-wgrad costs far
        # more than the one materialized 36 us gather pass it removes
        # (profiles/PERF_NOTES.md).  Opt-in via GYMFX_FUSE_GATHER=1; the
        # bitwise-equality GPU test keeps the path honest.
        import os as _os
        self._fuse_gather = (
            not self.recurrent and self.device.type == "cuda"
            and cfg.shuffle_rows and not self._overlap
            and _os.environ.get("GYMFX_FUSE_GATHER", "0") == "1")
        self._fm = {"seed": self.shuffle_seed, "minibatches": cfg.minibatches,
                    "ctr_off": -1, "step_base": self.step_base,
                    "mb_ctr": self.mb_ctr}

        self.global_step = 0   # host mirror of step_base (logging/ckpt)
        self.update_count = 0
        self.use_graphs = bool(cfg.use_graphs) and self.device.type == "cuda"
        # split rollout: the per-step kernel chain is latency-bound at
        # N=4096 (halving N barely changes kernel times), so running two
        # env halves on two HIP streams overlaps the chains.
        self._split = (self.device.type == "cuda" and cfg.rollout_streams > 1
                       and N % 2 == 0)
        self._fused = (self.device.type == "cuda" and cfg.fused_rollout
                       and not self.recurrent and cfg.hidden == 256)
        self._fuse_sample = (self.device.type == "cuda" and cfg.fuse_sample
                             and getattr(env, "_native", None) is not None
                             and env.params.action_space_mode != "continuous")
        self._fuse_head = (self._fuse_sample and cfg.fuse_head
                           and cfg.hidden % 8 == 0 and n_actions + 1 <= 8
                           and not cfg.fused_rollout)
        if self._split:
            self._s2 = torch.cuda.Stream()
            self.acts_half = [self.model.alloc_acts(N // 2),
                              self.model.alloc_acts(N // 2)]
        if self._overlap:
            self._gs = torch.cuda.Stream()
        self._graphs_ready = False
        self.g_rollout = None
        self.g_gather = None
        self.g_fwd_bwd = None
        self.g_opt = None
        from ..parallel.ddp import GradAllReducer
        self._reducer = GradAllReducer(world_size, process_group)

    # ------------------------------------------------------------------
    # graph-capturable bodies: no host syncs, no allocations, fixed
    # tensor addresses, all RNG/schedule state in device counters.
    # ------------------------------------------------------------------
    def _rollout_half(self, lo: int, hi: int, acts) -> None:
        """The T-step policy/env chain for env rows [lo, hi)."""
        env, model = self.env, self.model
        T = self.T
        rec = self.recurrent
        L = self.cfg.bptt_len if rec else 0
        state = None
        if rec:
            state = {"h": self.rnn_state["h"][lo:hi],
                     "c": self.rnn_state["c"][lo:hi]}
        for t in range(T):
            obs_t = self.obs_buf[t][lo:hi]
            h2 = None
            if rec:
                if t % L == 0:
                    # chunk-boundary recurrent state for BPTT (f32 snapshot)
                    ch = t // L
                    self.h0_buf[ch][lo:hi].copy_(state["h"])
                    self.c0_buf[ch][lo:hi].copy_(state["c"])
                if self._fuse_head:
                    # head computed inside the env kernel from the new h
                    model.step_forward(obs_t, state, acts, skip_head=True)
                    head, h2 = None, state["h"]
                else:
                    head = model.step_forward(obs_t, state, acts)
            elif self._fused:
                model.fused_step(
                    obs_t, self.act_buf[t][lo:hi], self.logp_buf[t][lo:hi],
                    self.val_buf[t][lo:hi], seed=self.sample_seed, step=t,
                    step_base=self.step_base, row_offset=lo,
                )
                head = None
            elif self._fuse_head:
                h2 = model.forward_hidden(obs_t, acts)
                head = None
            else:
                head = model.forward(obs_t, acts)
            fuse = self._fuse_sample and (head is not None or h2 is not None)
            if head is not None and not fuse:
                api.sample_head(
                    head, self.sample_seed, t,
                    self.act_buf[t][lo:hi], self.logp_buf[t][lo:hi],
                    self.val_buf[t][lo:hi], step_base=self.step_base,
                    row_offset=lo,
                )
            # fused env kernel writes reward/done into the rollout slabs and
            # the NEXT observation (bf16) into obs_buf[t+1] directly; with
            # fuse_sample it also samples the action from `head` itself.
            nxt = self.obs_buf[t + 1] if t + 1 < T else self.obs_bf16_step
            wname = "Wy" if rec else "W3"
            bname = "by" if rec else "b3"
            env.step(
                self.act_buf[t], reward_out=self.rew_buf[t],
                terminated_out=self.done_buf[t], obs_bf16_out=nxt,
                env_lo=lo, env_hi=hi,
                head=head if fuse else None,
                h2=h2 if fuse else None,
                w3t=model.wt(wname) if (fuse and h2 is not None) else None,
                b3=model.f32(bname) if (fuse and h2 is not None) else None,
                logp_out=self.logp_buf[t] if fuse else None,
                value_out=self.val_buf[t] if fuse else None,
                step_base=self.step_base if fuse else None,
                sample_seed=self.sample_seed, sample_step=t,
                # fresh episode -> zero recurrent state, fused into the
                # step kernel (saves a mask_reset launch per rollout step;
                # state["h"] ping-pongs per step and the recorded pointer
                # is the buffer the NEXT step reads — same tensor the old
                # mask_reset call saw at this point in the loop)
                rnn_h=state["h"] if rec else None,
                rnn_c=state["c"] if rec else None,
            )

    def _rollout_body(self) -> None:
        env, model = self.env, self.model
        T = self.T
        api.f32_to_bf16(env._obs, self.obs_buf[0])
        if self._split:
            half = self.N // 2
            ev_fork = torch.cuda.Event()
            ev_join = torch.cuda.Event()
            ev_fork.record()
            with torch.cuda.stream(self._s2):
                self._s2.wait_event(ev_fork)
                self._rollout_half(half, self.N, self.acts_half[1])
                ev_join.record()
            self._rollout_half(0, half, self.acts_half[0])
            torch.cuda.current_stream().wait_event(ev_join)
        else:
            self._rollout_half(0, self.N, self.acts_rollout)
        # bootstrap value (recurrent: peek one cell step WITHOUT mutating the
        # persistent state — use scratch state tensors)
        if self.recurrent:
            self._boot_state["h"].copy_(self.rnn_state["h"])
            self._boot_state["c"].copy_(self.rnn_state["c"])
            # fused=False: the fused step SWAPS the state's h tensor with
            # acts['h_tmp'] — a once-per-rollout swap would break the fixed
            # pointer cycle hipGraph replay requires
            head = model.step_forward(self.obs_bf16_step, self._boot_state,
                                      self.acts_rollout, fused=False)
        else:
            head = model.forward(self.obs_bf16_step, self.acts_rollout)
        self.val_buf[T].copy_(head[:, -1])
        api.increment_counter(self.step_base, T)

    def _gae_body(self) -> None:
        api.gae(
            self.rew_buf, self.val_buf, self.done_buf, self.adv_buf,
            self.ret_buf, self.cfg.gamma, self.cfg.gae_lambda,
        )
        if self.cfg.normalize_adv:
            api.adv_normalize(self.adv_buf.view(-1), self._adv_part)

    def _gather_body(self, slot: int = 0) -> None:
        """Feistel-shuffled minibatch gather (parameter-independent: it
        overlaps the previous minibatch's fwd/bwd on a side stream and, in
        data-parallel runs, the async gradient all-reduce)."""
        cfg = self.cfg
        if self.recurrent:
            api.mb_gather_seq(
                self.obs_buf, self.act_buf, self.logp_buf, self.adv_buf,
                self.ret_buf, self.done_buf, self.h0_buf, self.c0_buf,
                self.obs_mb_seq, self.act_mb, self.logp_mb, self.adv_mb,
                self.ret_mb, self.done_mb, self.h0_mb, self.c0_mb,
                L=cfg.bptt_len, seed=self.shuffle_seed,
                minibatches=cfg.minibatches, step_base=self.step_base,
                mb_ctr=self.mb_ctr,
            )
        else:
            api.mb_gather(
                self.obs_flat, self.act_flat, self.logp_flat, self.adv_flat,
                self.ret_flat, self.obs_mb_s[slot], self.act_mb_s[slot],
                self.logp_mb_s[slot], self.adv_mb_s[slot], self.ret_mb_s[slot],
                seed=self.shuffle_seed, minibatches=cfg.minibatches,
                step_base=self.step_base, mb_ctr=self.mb_ctr,
                skip_obs=self._fuse_gather,
            )
        api.increment_counter(self.mb_ctr, 1)

    def _fwd_bwd_body(self, slot: int = 0) -> None:
        """Forward + PPO loss backward + model backward on the gathered
        minibatch.  Gradients are fully overwritten by backward
        (deterministic split-M wgrad), so there is no zero_grad."""
        cfg, model = self.cfg, self.model
        if self.recurrent:
            head = model.bptt_forward(self.obs_mb_seq, self.done_mb,
                                      self.h0_mb, self.c0_mb, self.bptt)
            api.ppo_loss_bwd(
                head, self.act_mb.view(-1), self.logp_mb.view(-1),
                self.adv_mb.view(-1), self.ret_mb.view(-1), self.dhead,
                clip_eps=cfg.clip_eps, ent_coef=cfg.ent_coef,
                vf_coef=cfg.vf_coef, inv_count=1.0 / self.mb_rows,
                losses=self.losses,
            )
            model.bptt_backward(self.obs_mb_seq, self.done_mb, self.dhead,
                                self.bptt)
            return
        if self._fuse_gather:
            # obs rows come straight from the rollout slab through the
            # permutation (mb_ctr was already advanced by the gather of the
            # small fields: ctr_off=-1 in self._fm)
            head = model.forward(self.obs_flat, self.acts_train,
                                 a_feistel=self._fm)
        else:
            head = model.forward(self.obs_mb_s[slot], self.acts_train)
        api.ppo_loss_bwd(
            head, self.act_mb_s[slot], self.logp_mb_s[slot],
            self.adv_mb_s[slot], self.ret_mb_s[slot],
            self.dhead, clip_eps=cfg.clip_eps, ent_coef=cfg.ent_coef,
            vf_coef=cfg.vf_coef, inv_count=1.0 / self.mb_rows,
            losses=self.losses,
        )
        if self._fuse_gather:
            model.backward(self.obs_flat, self.acts_train, self.dhead,
                           self.scratch, a_feistel=self._fm)
        else:
            model.backward(self.obs_mb_s[slot], self.acts_train, self.dhead,
                           self.scratch)

    def _mb_body(self, slot: int = 0) -> None:
        self._gather_body(slot)
        self._fwd_bwd_body(slot)

    def _mb_body_noshuffle(self, epoch_mb: int) -> None:
        cfg, model = self.cfg, self.model
        M = self.mb_rows
        mb = epoch_mb % cfg.minibatches
        sl = slice(mb * M, (mb + 1) * M)
        self.obs_mb.copy_(self.obs_flat[sl])
        self.act_mb.copy_(self.act_flat[sl])
        self.logp_mb.copy_(self.logp_flat[sl])
        self.adv_mb.copy_(self.adv_flat[sl])
        self.ret_mb.copy_(self.ret_flat[sl])
        head = model.forward(self.obs_mb, self.acts_train)
        api.ppo_loss_bwd(
            head, self.act_mb, self.logp_mb, self.adv_mb, self.ret_mb,
            self.dhead, clip_eps=cfg.clip_eps, ent_coef=cfg.ent_coef,
            vf_coef=cfg.vf_coef, inv_count=1.0 / M, losses=self.losses,
        )
        model.backward(self.obs_mb, self.acts_train, self.dhead, self.scratch)

    def _opt_body(self) -> None:
        self.model.adam(self.cfg.lr, max_grad_norm=self.cfg.max_grad_norm)

    # ------------------------------------------------------------------
    def _snapshot(self) -> Dict[str, Any]:
        """Clone every tensor the warmup pass mutates (env + model +
        counters) so capture leaves training state untouched."""
        m = self.model
        snap = {
            "params": m.params.clone(), "m": m.m.clone(), "v": m.v.clone(),
            "params_bf16": m.params_bf16.clone(),
            "adam_ctr": m.adam_ctr.clone(), "adam_step": m.adam_step,
            "step_base": self.step_base.clone(), "mb_ctr": self.mb_ctr.clone(),
            "obs": self.env._obs.clone(),
            "st": {k: v.clone() for k, v in self.env.st.to_dict().items()},
        }
        if self.recurrent:
            snap["rnn_h"] = self.rnn_state["h"].clone()
            snap["rnn_c"] = self.rnn_state["c"].clone()
        return snap

    def _restore(self, snap: Dict[str, Any]) -> None:
        m = self.model
        m.params.copy_(snap["params"])
        m.m.copy_(snap["m"])
        m.v.copy_(snap["v"])
        m.params_bf16.copy_(snap["params_bf16"])
        m._refresh_wt()
        m.adam_ctr.copy_(snap["adam_ctr"])
        m.adam_step = snap["adam_step"]
        self.step_base.copy_(snap["step_base"])
        self.mb_ctr.copy_(snap["mb_ctr"])
        self.env._obs.copy_(snap["obs"])
        st = self.env.st.to_dict()
        for k, v in snap["st"].items():
            st[k].copy_(v)
        if self.recurrent:
            self.rnn_state["h"].copy_(snap["rnn_h"])
            self.rnn_state["c"].copy_(snap["rnn_c"])

    def _capture_graphs(self) -> None:
        """Capture rollout+GAE, minibatch fwd/bwd, and optimizer as three
        hipGraphs.  The gradient all-reduce (world_size > 1) stays eager
        between the mb and opt replays.  Device counters make every replay
        advance RNG / Adam step correctly.  The warmup pass initializes the
        RCCL communicator and lazy workspaces; training state is snapshotted
        and restored around it so graph mode is bit-equivalent to eager."""
        snap = self._snapshot()
        self._rollout_body()
        self._gae_body()
        self.mb_ctr.zero_()
        nslots = 2 if self._overlap else 1
        for s in range(nslots):
            self._mb_body(s)
        self._allreduce_grads()
        self._opt_body()
        torch.cuda.synchronize()
        self.g_rollout = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.g_rollout):
            self._rollout_body()
            self._gae_body()
        self.g_gather = []
        self.g_fwd_bwd = []
        for s in range(nslots):
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self._gather_body(s)
            self.g_gather.append(g)
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self._fwd_bwd_body(s)
            self.g_fwd_bwd.append(g)
        self.g_opt = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.g_opt):
            self._opt_body()
        torch.cuda.synchronize()
        self._restore(snap)
        torch.cuda.synchronize()
        self._graphs_ready = True

    # ------------------------------------------------------------------
    def collect_rollout(self) -> None:
        if self.use_graphs and self._graphs_ready:
            self.g_rollout.replay()
        else:
            self._rollout_body()
            self._gae_body()
        self.global_step += self.T

    def compute_advantages(self) -> None:
        # folded into collect_rollout (one graph); kept for API compat
        pass

    def update(self, with_stats: bool = True) -> Dict[str, float]:
        cfg = self.cfg
        self.losses.zero_()
        self.mb_ctr.zero_()
        graphs = self.use_graphs and self._graphs_ready
        n_mb = cfg.ppo_epochs * cfg.minibatches
        if not cfg.shuffle_rows:
            for i in range(n_mb):
                self._mb_body_noshuffle(i)
                self._allreduce_grads()
                self._opt_body()
            self.update_count += 1
            return self._stats(n_mb) if with_stats else {}
        def gather(slot: int) -> None:
            if graphs:
                self.g_gather[slot].replay()
            else:
                self._gather_body(slot)

        def fwd_bwd(slot: int) -> None:
            if graphs:
                self.g_fwd_bwd[slot].replay()
            else:
                self._fwd_bwd_body(slot)

        def opt() -> None:
            if graphs:
                self.g_opt.replay()
            else:
                self._opt_body()

        if not self._overlap:
            # pipelined loop: gather(i+1) overlaps the async all-reduce of
            # minibatch i's gradients (parallel/ddp.py)
            gather(0)
            for i in range(n_mb):
                fwd_bwd(0)
                self._reducer.start(self.model.grads)
                if i + 1 < n_mb:
                    gather(0)
                self._reducer.finish()
                opt()
            self.update_count += 1
            return self._stats(n_mb) if with_stats else {}
        # two-stream pipeline (ping-pong minibatch slots): gather(i+1) is
        # parameter-independent, so it runs on a side stream concurrently
        # with fwd/bwd(i) and the async gradient all-reduce.  Buffer slot
        # i&1 alternates; gather(i+1) into slot s is safe once fwd_bwd(i-1)
        # (the last reader of s) has been enqueued-before on the main
        # stream (ev_fork), and fwd_bwd(i+1) waits on its gather (ev_g).
        s2 = self._gs
        main = torch.cuda.current_stream()
        ev_fork = torch.cuda.Event()
        ev_g = [torch.cuda.Event(), torch.cuda.Event()]
        gather(0)
        for i in range(n_mb):
            slot = i & 1
            nxt = slot ^ 1
            if i + 1 < n_mb:
                ev_fork.record()
                with torch.cuda.stream(s2):
                    s2.wait_event(ev_fork)
                    gather(nxt)
                    ev_g[nxt].record()
            fwd_bwd(slot)
            self._reducer.start(self.model.grads)
            self._reducer.finish()
            opt()
            if i + 1 < n_mb:
                main.wait_event(ev_g[nxt])
        self.update_count += 1
        return self._stats(n_mb) if with_stats else {}

    def _stats(self, n_mb: int) -> Dict[str, float]:
        lv = (self.losses / n_mb).cpu()
        return {
            "pi_loss": float(lv[0]),
            "v_loss": float(lv[1]),
            "entropy": float(lv[2]),
            "approx_kl": float(lv[3]),
            "clipfrac": float(lv[4]),
        }

    def _allreduce_grads(self) -> None:
        if self.world_size <= 1:
            return
        import torch.distributed as dist

        dist.all_reduce(self.model.grads, op=dist.ReduceOp.SUM, group=self.pg)
        self.model.grads.mul_(1.0 / self.world_size)

    def train_update(self, with_stats: bool = True) -> Dict[str, float]:
        """One full PPO update (rollout + GAE + epochs)."""
        if self.use_graphs and not self._graphs_ready:
            self._capture_graphs()
        self.collect_rollout()
        return self.update(with_stats)


# ---------------------------------------------------------------------------
# config-driven entry (mode=training in the CLI runner)
# ---------------------------------------------------------------------------

def train_from_config(config: Dict[str, Any]) -> Dict[str, Any]:
    from .. import build_vec_environment

    cfg = PPOConfig.from_config(config)
    vec_cfg = dict(config)
    vec_cfg.setdefault("autoreset", True)
    vec_cfg.setdefault("env_start_mode", "spread")
    env = build_vec_environment(vec_cfg)
    env.reset(seed=cfg.seed)
    # torchrun-launched config training joins the data-parallel group the
    # same way bench.py does (one rank per GPU over RCCL; gloo on CPU) —
    # without this, WORLD_SIZE>1 would train unsynchronized replicas.
    from ..parallel.ddp import init_from_env

    rank, world_size, pg = init_from_env(env.device)
    trainer = PPOTrainer(env, cfg, rank=rank, world_size=world_size,
                         process_group=pg)
    ckpt_path = config.get("checkpoint_file")
    if rank != 0:
        ckpt_path = None  # only rank 0 writes checkpoints/results
    resumed = False
    if ckpt_path and config.get("resume"):
        from ..utils.checkpoint import load_checkpoint

        load_checkpoint(trainer, ckpt_path)
        resumed = True
    updates = int(config.get("train_updates", 10))
    # periodic save (atomic temp+rename): a crash mid-run resumes from the
    # last interval instead of zero.  0/None disables.
    ckpt_every = int(config.get("checkpoint_interval") or 0)
    trace_path = config.get("trace_file")
    tracer = writer = None
    if trace_path:
        from ..utils.trace import PhaseTimer, TraceWriter

        tracer = PhaseTimer(env.device)
        writer = TraceWriter(str(trace_path))
    t0 = time.perf_counter()
    history: List[Dict[str, float]] = []
    for u in range(updates):
        if tracer is None:
            stats = trainer.train_update()
        else:
            if trainer.use_graphs and not trainer._graphs_ready:
                trainer._capture_graphs()
            with tracer.phase("rollout"):
                trainer.collect_rollout()
            with tracer.phase("update"):
                stats = trainer.update()
            phases = tracer.drain()
            writer.write({"update": u, "global_step": trainer.global_step,
                          "phases_ms": phases, **stats})
        history.append(stats)
        if ckpt_path and ckpt_every and (u + 1) % ckpt_every == 0 and u + 1 < updates:
            from ..utils.checkpoint import save_checkpoint

            save_checkpoint(trainer, ckpt_path)
        if not config.get("quiet_mode") and (u % max(1, updates // 10) == 0):
            print(f"update {u}: {stats}")
    if writer is not None:
        writer.close()
    wall = time.perf_counter() - t0
    if ckpt_path:
        from ..utils.checkpoint import save_checkpoint

        save_checkpoint(trainer, ckpt_path,
                        extra={"config": {k: v for k, v in config.items()
                                          if isinstance(v, (int, float, str, bool, type(None)))}})
    steps = trainer.global_step * env.n_envs * world_size
    summary = {
        "mode": "training",
        "updates": updates,
        "resumed": resumed,
        "checkpoint_file": ckpt_path,
        "policy_model": cfg.policy,
        "rank": rank,
        "world_size": world_size,
        "env_steps": steps,
        "wall_seconds": wall,
        "env_steps_per_sec": steps / wall if wall > 0 else 0.0,
        "final_stats": history[-1] if history else {},
        "vec_summary": env.vec_summary(),
    }
    return summary
