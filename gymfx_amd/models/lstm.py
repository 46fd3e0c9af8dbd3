"""Recurrent actor-critic: LSTM(hidden) -> fused head [n_actions | value].

BASELINE.json config #4 (recurrent PPO LSTM actor-critic).  The reference is
agent-free (/root/reference/app/env.py:148-150); this model follows the same
MI355X-first design as models/mlp.py: one flat f32 master parameter buffer
(single RCCL all-reduce bucket), bf16 compute mirrors (+ transposed mirrors
so every GEMM stages its B operand contiguously), hand-written MFMA GEMMs
for x@Wx / h@Wh / head, and fused elementwise LSTM-cell kernels.

Training uses sequence-chunked BPTT: the x-projection of a
whole [L, M] sequence batch runs as ONE GEMM; only the recurrent h@Wh GEMM
and the cell kernel run per timestep; dWx/dWh/dWy each reduce over the full
[L*M] batch in one deterministic split-M wgrad.
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Dict, Optional, Tuple

import torch

from ..ops import api


@dataclass
class _ParamSlice:
    name: str
    shape: Tuple[int, ...]
    sl: slice
    is_weight: bool


class ActorCriticLSTM:
    recurrent = True

    def __init__(
        self,
        obs_dim: int,
        n_actions: int = 3,
        hidden: int = 256,
        *,
        device: torch.device,
        seed: int = 0,
    ):
        self.obs_dim = obs_dim
        self.n_actions = n_actions
        self.hidden = hidden
        self.head_dim = n_actions + 1
        self.device = device
        H = hidden

        dims = [
            ("Wx", (obs_dim, 4 * H), True),
            ("Wh", (H, 4 * H), True),
            ("b", (4 * H,), False),
            ("Wy", (H, self.head_dim), True),
            ("by", (self.head_dim,), False),
        ]
        self.slices: Dict[str, _ParamSlice] = {}
        off = 0
        for name, shape, is_w in dims:
            n = int(torch.tensor(shape).prod())
            self.slices[name] = _ParamSlice(name, shape, slice(off, off + n), is_w)
            off += n
        self.n_params = off

        g = torch.Generator().manual_seed(seed)
        flat = torch.empty(off, dtype=torch.float32)
        for name, shape, is_w in dims:
            s = self.slices[name]
            if is_w:
                bound = 1.0 / math.sqrt(shape[0])
                flat[s.sl] = (torch.rand(s.sl.stop - s.sl.start, generator=g) * 2 - 1) * bound
            else:
                flat[s.sl] = 0.0
        # Gate layout is INTERLEAVED along the 4H axis: column 4*k + g holds
        # gate g of hidden unit k (g: 0=i 1=f 2=g~ 3=o).  A 16-column GEMM
        # output tile then contains 4 COMPLETE hidden units, which is what
        # lets the recurrent GEMM compute the LSTM cell in its epilogue
        # (ops/csrc/ppo_kernels.hip lstm_gemm_cell_fwd) and makes every
        # cell-kernel gate access a contiguous 4-wide vector load instead of
        # four reads H apart.  Init is layout-invariant (iid uniform) except
        # the forget-gate bias:
        b = flat[self.slices["b"].sl].view(H, 4)
        b[:, 1] = 1.0  # forget gate
        self.params = flat.to(device)
        self.grads = torch.zeros_like(self.params)
        self.m = torch.zeros_like(self.params)
        self.v = torch.zeros_like(self.params)
        self.params_bf16 = self.params.to(torch.bfloat16)
        self._wt: Dict[str, torch.Tensor] = {}
        for name in ("Wx", "Wh", "Wy"):
            K_, N_ = self.slices[name].shape
            self._wt[name] = torch.empty(N_, K_, dtype=torch.bfloat16, device=device)
        self._refresh_wt()
        self.adam_step = 0
        self.adam_ctr = torch.zeros((), dtype=torch.int32, device=device)
        self._clip_part = torch.zeros(256, dtype=torch.float32, device=device)
        self._clip_scale = torch.ones(1, dtype=torch.float32, device=device)
        self._wg_ws: Dict[str, Tuple[torch.Tensor, Optional[torch.Tensor], int]] = {}
        self.wgrad_slabs = 64

    # -- param views ----------------------------------------------------
    def w(self, name: str) -> torch.Tensor:
        s = self.slices[name]
        return self.params_bf16[s.sl].view(*s.shape)

    def wt(self, name: str) -> torch.Tensor:
        return self._wt[name]

    def f32(self, name: str) -> torch.Tensor:
        s = self.slices[name]
        return self.params[s.sl].view(*s.shape)

    def grad(self, name: str) -> torch.Tensor:
        s = self.slices[name]
        return self.grads[s.sl].view(*s.shape)

    def _refresh_wt(self) -> None:
        for name, t in self._wt.items():
            api.transpose_bf16(self.w(name), t)

    def _wg_slabs(self, K: int, N: int) -> int:
        # (GYMFX_WGRAD_SLABS overrides — tuning knob, same as models/mlp.py)
        import os
        ov = os.environ.get("GYMFX_WGRAD_SLABS")
        if ov:
            return int(ov)
        tiles = ((K + 63) // 64) * ((N + 63) // 64)
        s = self.wgrad_slabs
        while tiles * s < 2048:
            s *= 2
        return s

    def _wg_workspace(self, name: str, K: int, N: int, want_db: bool):
        key = f"{name}:{K}x{N}"
        if key not in self._wg_ws:
            S = self._wg_slabs(K, N)
            dw = torch.empty(S, K, N, dtype=torch.float32, device=self.device)
            db = torch.empty(S, N, dtype=torch.float32, device=self.device) if want_db else None
            self._wg_ws[key] = (dw, db, S)
        return self._wg_ws[key]

    # -- rollout state / buffers ----------------------------------------
    def alloc_state(self, M: int) -> Dict[str, torch.Tensor]:
        dev, H = self.device, self.hidden
        return {
            "h": torch.zeros(M, H, dtype=torch.bfloat16, device=dev),
            "c": torch.zeros(M, H, dtype=torch.float32, device=dev),
        }

    def alloc_acts(self, M: int) -> Dict[str, torch.Tensor]:
        dev, H = self.device, self.hidden
        return {
            # gates_pre stored bf16 (like gates_h and h): the f32 original
            # cost 268 MB of write traffic per minibatch on the Wx GEMM and
            # doubled every cell read; the cell still sums gx+gh in f32
            "gates": torch.empty(M, 4 * H, dtype=torch.bfloat16, device=dev),
            "gates_h": torch.empty(M, 4 * H, dtype=torch.bfloat16, device=dev),
            "head": torch.empty(M, self.head_dim, dtype=torch.float32, device=dev),
            # fused-step h output (the fused GEMM+cell kernel must not write
            # h in place: other workgroups still read it as the A operand)
            "h_tmp": torch.empty(M, H, dtype=torch.bfloat16, device=dev),
        }

    # -- single step (rollout) -------------------------------------------
    def step_forward(self, obs_bf16: torch.Tensor, state: Dict[str, torch.Tensor],
                     acts: Dict[str, torch.Tensor], *,
                     fused: bool = True,
                     skip_head: bool = False) -> "Optional[torch.Tensor]":
        """One policy step: updates state['h']/state['c'] IN PLACE (the h
        TENSOR IDENTITY rotates with acts['h_tmp'] on the fused path —
        callers must read state['h'] through the dict), returns head
        [M, A+1] f32."""
        gates = acts["gates"]
        api.gemm(obs_bf16, self.wt("Wx"), self.f32("b"), gates, act=1, trans_b=True)
        if fused and "h_tmp" in acts:
            # recurrent GEMM + cell in one kernel; rollout never reads
            # gates_h back, so its global write is skipped on the GPU path
            api.lstm_step_fused(state["h"], self.wt("Wh"), gates,
                                acts["gates_h"], state["c"], state["c"],
                                acts["h_tmp"])  # rollout: no saved acts
            state["h"], acts["h_tmp"] = acts["h_tmp"], state["h"]
        else:
            api.gemm(state["h"], self.wt("Wh"), None, acts["gates_h"], act=1,
                     trans_b=True)
            api.lstm_cell_fwd(gates, acts["gates_h"], state["c"], state["c"],
                              state["h"])
        if skip_head:
            # head-in-step fusion: the env kernel computes h @ Wy + by
            return None
        api.gemm(state["h"], self.wt("Wy"), self.f32("by"), acts["head"], act=0, trans_b=True)
        return acts["head"]

    # -- BPTT training buffers -------------------------------------------
    def alloc_bptt(self, L: int, M: int) -> Dict[str, torch.Tensor]:
        dev, H = self.device, self.hidden
        return {
            # index l = input state of step l (masked); [L+1]: final unused as
            # input but written for uniformity
            "h_in": torch.empty(L + 1, M, H, dtype=torch.bfloat16, device=dev),
            "c_in": torch.empty(L + 1, M, H, dtype=torch.float32, device=dev),
            # raw outputs of step l (head input; cell-bwd c_new)
            "h_raw": torch.empty(L, M, H, dtype=torch.bfloat16, device=dev),
            "c_raw": torch.empty(L, M, H, dtype=torch.float32, device=dev),
            "gates": torch.empty(L, M, 4 * H, dtype=torch.bfloat16, device=dev),
            # saved gate ACTIVATIONS (i,f,g,o interleaved): what the
            # backward consumes — gates/gates_h are never re-read by bwd
            "acts": torch.empty(L, M, 4 * H, dtype=torch.bfloat16, device=dev),
            # scratch for the unfused fwd fallback's recurrent projection
            "gh_scratch": torch.empty(M, 4 * H, dtype=torch.bfloat16, device=dev),
            "head": torch.empty(L * M, self.head_dim, dtype=torch.float32, device=dev),
            "dgates": torch.empty(L, M, 4 * H, dtype=torch.bfloat16, device=dev),
            # head-dgrad slab bf16: halves the 2x67 MB/minibatch of
            # write+read traffic on the dh path (cell math still f32)
            "dh_all": torch.empty(L, M, H, dtype=torch.bfloat16, device=dev),
            # ping-pong recurrent-grad buffers (the fused bwd kernel writes
            # dh for step l-1 while reading this step's dh_next)
            "dh_na": torch.empty(M, H, dtype=torch.float32, device=dev),
            "dh_nb": torch.empty(M, H, dtype=torch.float32, device=dev),
            "dc_a": torch.empty(M, H, dtype=torch.float32, device=dev),
            "dc_b": torch.empty(M, H, dtype=torch.float32, device=dev),
        }

    def bptt_forward(self, obs_seq: torch.Tensor, done_seq: torch.Tensor,
                     h0: torch.Tensor, c0: torch.Tensor,
                     buf: Dict[str, torch.Tensor]) -> torch.Tensor:
        """obs_seq [L, M, D] bf16, done_seq [L, M] bool, h0 [M,H] bf16,
        c0 [M,H] f32 -> head [L*M, A+1].  Saves everything bwd needs."""
        L, M, D = obs_seq.shape
        H = self.hidden
        # x-projection for ALL timesteps in one GEMM
        gates_flat = buf["gates"].view(L * M, 4 * H)
        api.gemm(obs_seq.view(L * M, D), self.wt("Wx"), self.f32("b"),
                 gates_flat, act=1, trans_b=True)
        buf["h_in"][0].copy_(h0)
        buf["c_in"][0].copy_(c0)
        for l in range(L):
            # one fused kernel per sequential step: recurrent GEMM + cell +
            # reset-masked next-step state; the four gate activations are
            # saved (acts) for the backward
            api.lstm_step_fused(buf["h_in"][l], self.wt("Wh"),
                                buf["gates"][l], buf["gh_scratch"],
                                buf["c_in"][l], buf["c_raw"][l],
                                buf["h_raw"][l], done_seq[l],
                                buf["h_in"][l + 1], buf["c_in"][l + 1],
                                acts_out=buf["acts"][l])
        api.gemm(buf["h_raw"].view(L * M, H), self.wt("Wy"), self.f32("by"),
                 buf["head"], act=0, trans_b=True)
        return buf["head"]

    def bptt_backward(self, obs_seq: torch.Tensor, done_seq: torch.Tensor,
                      dhead: torch.Tensor, buf: Dict[str, torch.Tensor]) -> None:
        """dhead [L*M, A+1] bf16 -> accumulate grads (overwrites all slices)."""
        L, M, D = obs_seq.shape
        H = self.hidden
        h_raw_flat = buf["h_raw"].view(L * M, H)
        # head layer
        dw_p, db_p, slabs = self._wg_workspace("Wy", H, self.head_dim, True)
        api.wgrad(h_raw_flat, dhead, self.grad("Wy"), self.grad("by"),
                  workspace=(dw_p, db_p), slabs=slabs)
        dh_flat = buf["dh_all"].view(L * M, H)
        api.gemm(dhead, self.w("Wy"), None, dh_flat, act=1, trans_b=True)
        # backward through time: ONE fused kernel per step computes dgates
        # (cell bwd) AND the recurrent dgrad dh for step l-1
        dc_next: Optional[torch.Tensor] = None
        dh_next: Optional[torch.Tensor] = None
        dc_bufs = (buf["dc_a"], buf["dc_b"])
        dh_bufs = (buf["dh_na"], buf["dh_nb"])
        for l in range(L - 1, -1, -1):
            dc_out = dc_bufs[l & 1]
            dh_out = dh_bufs[l & 1] if l > 0 else None
            api.lstm_bwd_step(buf["acts"][l],
                              buf["c_in"][l], buf["c_raw"][l],
                              buf["dh_all"][l], dh_next, dc_next,
                              done_seq[l], self.w("Wh"), buf["dgates"][l],
                              dc_out, dh_out)
            dc_next = dc_out
            dh_next = dh_out
        # weight grads over the whole sequence batch
        dgates_flat = buf["dgates"].view(L * M, 4 * H)
        dw_p, db_p, slabs = self._wg_workspace("Wx", D, 4 * H, True)
        api.wgrad(obs_seq.view(L * M, D), dgates_flat, self.grad("Wx"),
                  self.grad("b"), workspace=(dw_p, db_p), slabs=slabs)
        dw_p, db_p, slabs = self._wg_workspace("Wh", H, 4 * H, False)
        api.wgrad(buf["h_in"][:L].reshape(L * M, H), dgates_flat,
                  self.grad("Wh"), None, workspace=(dw_p, db_p), slabs=slabs)

    # -- optimizer --------------------------------------------------------
    def adam(self, lr: float, *, beta1=0.9, beta2=0.999, eps=1e-8,
             max_grad_norm: float = 0.0) -> None:
        self.adam_step += 1
        clip = None
        if max_grad_norm and max_grad_norm > 0:
            # fused clipping: sumsq partials + per-block scale derivation
            # inside the adam launch (no separate clip_scale kernel)
            clip = (self._clip_part, max_grad_norm)
        api.adam(self.params, self.grads, self.m, self.v, self.params_bf16,
                 lr=lr, beta1=beta1, beta2=beta2, eps=eps, step=self.adam_step,
                 step_ctr=self.adam_ctr, clip=clip)
        api.increment_counter(self.adam_ctr, 1)
        self._refresh_wt()

    def zero_grad(self) -> None:
        self.grads.zero_()

    # -- checkpoint -------------------------------------------------------
    def state_dict(self) -> Dict[str, torch.Tensor]:
        return {
            "params": self.params.detach().cpu(),
            "m": self.m.detach().cpu(),
            "v": self.v.detach().cpu(),
            # the DEVICE counter is the source of truth: under hipGraph
            # replay the host mirror self.adam_step only advanced during
            # capture (checkpoint bug found by the GPU resume test)
            "adam_step": torch.tensor(int(self.adam_ctr.item())),
            "obs_dim": torch.tensor(self.obs_dim),
            "n_actions": torch.tensor(self.n_actions),
            "hidden": torch.tensor(self.hidden),
            "arch": "lstm",
        }

    def load_state_dict(self, sd: Dict[str, torch.Tensor]) -> None:
        self.params.copy_(sd["params"].to(self.device))
        self.m.copy_(sd["m"].to(self.device))
        self.v.copy_(sd["v"].to(self.device))
        self.adam_step = int(sd["adam_step"])
        self.adam_ctr.fill_(self.adam_step)
        self.params_bf16.copy_(self.params.to(torch.bfloat16))
        self._refresh_wt()
