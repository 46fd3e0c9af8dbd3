"""Actor-critic MLP(256,256) — bf16 MFMA compute, f32 master weights.

North-star model (BASELINE.json config #2): obs -> tanh(256) -> tanh(256)
-> fused head [n_actions logits | value].  Forward/backward run entirely on
the hand-written MFMA GEMM kernels (ops/api.gemm / wgrad); the optimizer is
the fused Adam kernel over one flat parameter buffer (single RCCL bucket in
data-parallel runs).
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
    is_weight: bool  # weights get a bf16 mirror; biases stay f32


class ActorCriticMLP:
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

        dims = [
            ("W1", (obs_dim, hidden), True),
            ("b1", (hidden,), False),
            ("W2", (hidden, hidden), True),
            ("b2", (hidden,), False),
            ("W3", (hidden, self.head_dim), True),
            ("b3", (self.head_dim,), False),
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
                fan_in = shape[0]
                bound = 1.0 / math.sqrt(fan_in)
                flat[s.sl] = (torch.rand(s.sl.stop - s.sl.start, generator=g) * 2 - 1) * bound
            else:
                flat[s.sl] = 0.0
        self.params = flat.to(device)
        self.grads = torch.zeros_like(self.params)
        self.m = torch.zeros_like(self.params)
        self.v = torch.zeros_like(self.params)
        # bf16 mirrors of the weight matrices (GEMM operands)
        self.params_bf16 = self.params.to(torch.bfloat16)
        # transposed [N,K] mirrors of the weight matrices: forward GEMMs use
        # the TRANS_B path (contiguous B staging); refreshed after every
        # Adam step by the transpose kernel.
        self._wt: Dict[str, torch.Tensor] = {}
        for name in ("W1", "W2", "W3"):
            s = self.slices[name]
            K_, N_ = s.shape
            self._wt[name] = torch.empty(N_, K_, dtype=torch.bfloat16, device=device)
        self._refresh_wt()
        self.adam_step = 0  # host mirror of adam_ctr (logging/checkpoint)
        # device step counter: Adam bias correction inside captured hipGraphs
        self.adam_ctr = torch.zeros((), dtype=torch.int32, device=device)
        # grad-clip workspace
        self._clip_part = torch.zeros(256, dtype=torch.float32, device=device)
        self._clip_scale = torch.ones(1, dtype=torch.float32, device=device)
        # wgrad slab workspaces (allocated lazily per layer shape)
        self._wg_ws: Dict[str, Tuple[torch.Tensor, Optional[torch.Tensor]]] = {}
        self.wgrad_slabs = 64

    # -- param views ----------------------------------------------------
    def w(self, name: str) -> torch.Tensor:
        s = self.slices[name]
        return self.params_bf16[s.sl].view(*s.shape)

    def wt(self, name: str) -> torch.Tensor:
        """Transposed [N,K] bf16 mirror (forward GEMM B operand)."""
        return self._wt[name]

    def _refresh_wt(self) -> None:
        for name, t in self._wt.items():
            api.transpose_bf16(self.w(name), t)

    def f32(self, name: str) -> torch.Tensor:
        s = self.slices[name]
        return self.params[s.sl].view(*s.shape)

    def grad(self, name: str) -> torch.Tensor:
        s = self.slices[name]
        return self.grads[s.sl].view(*s.shape)

    def _wg_slabs(self, K: int, N: int) -> int:
        # enough split-M slabs that tiles x slabs >= ~4 blocks per CU
        # (GYMFX_WGRAD_SLABS overrides — tuning knob)
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

    # -- forward ---------------------------------------------------------
    def alloc_acts(self, M: int) -> Dict[str, torch.Tensor]:
        dev = self.device
        return {
            "h1": torch.empty(M, self.hidden, dtype=torch.bfloat16, device=dev),
            "h2": torch.empty(M, self.hidden, dtype=torch.bfloat16, device=dev),
            "head": torch.empty(M, self.head_dim, dtype=torch.float32, device=dev),
        }

    def fused_step(self, obs_bf16: torch.Tensor, actions: torch.Tensor,
                   logp: torch.Tensor, value: torch.Tensor, *, seed: int,
                   step: int, step_base: torch.Tensor, row_offset: int = 0,
                   greedy: bool = False) -> None:
        """One fused policy step on GPU (forward + categorical sample in a
        single kernel; bitwise identical to forward()+sample_head)."""
        from ..ops import native
        native.require().mlp_policy_rollout(
            obs_bf16, self.wt("W1"), self.f32("b1"), self.wt("W2"),
            self.f32("b2"), self.wt("W3"), self.f32("b3"), actions, logp,
            value, seed, step, step_base, row_offset, greedy)

    def forward(self, obs_bf16: torch.Tensor, acts: Dict[str, torch.Tensor],
                a_feistel: "Optional[dict]" = None) -> torch.Tensor:
        """obs_bf16 [M, obs_dim] -> head f32 [M, A+1]; saves h1/h2 for bwd.

        B operands are the transposed mirrors (TRANS_B path): contiguous
        vector staging on gfx950 regardless of tile width.
        a_feistel: gather+first-GEMM fusion — obs_bf16 is the FULL rollout
        slab and L1 reads its rows through the epoch permutation."""
        api.gemm(obs_bf16, self.wt("W1"), self.f32("b1"), acts["h1"], act=2,
                 trans_b=True, a_perm=a_feistel)
        api.gemm(acts["h1"], self.wt("W2"), self.f32("b2"), acts["h2"], act=2, trans_b=True)
        api.gemm(acts["h2"], self.wt("W3"), self.f32("b3"), acts["head"], act=0, trans_b=True)
        return acts["head"]

    def forward_hidden(self, obs_bf16: torch.Tensor,
                       acts: Dict[str, torch.Tensor]) -> torch.Tensor:
        """Layers 1-2 only: the rollout's head-in-step fusion computes the
        tiny head GEMM inside the env kernel (h2 @ W3 + b3 per env row),
        removing one launch from the latency-bound per-step chain."""
        api.gemm(obs_bf16, self.wt("W1"), self.f32("b1"), acts["h1"], act=2, trans_b=True)
        api.gemm(acts["h1"], self.wt("W2"), self.f32("b2"), acts["h2"], act=2, trans_b=True)
        return acts["h2"]

    # -- backward (dhead [M, A+1] bf16 -> accumulate grads) --------------
    def backward(
        self,
        obs_bf16: torch.Tensor,
        acts: Dict[str, torch.Tensor],
        dhead: torch.Tensor,
        scratch: Dict[str, torch.Tensor],
        a_feistel: "Optional[dict]" = None,
    ) -> None:
        M = dhead.shape[0]
        dh2 = scratch["dh2"]
        dh1 = scratch["dh1"]
        # head layer
        dw_p, db_p, slabs = self._wg_workspace("W3", self.hidden, self.head_dim, True)
        api.wgrad(acts["h2"], dhead, self.grad("W3"), self.grad("b3"),
                  workspace=(dw_p, db_p), slabs=slabs)
        api.gemm(dhead, self.w("W3"), None, dh2, Yact=acts["h2"], trans_b=True,
                 act=1, dact_tanh=True)
        # layer 2
        dw_p, db_p, slabs = self._wg_workspace("W2", self.hidden, self.hidden, True)
        api.wgrad(acts["h1"], dh2, self.grad("W2"), self.grad("b2"),
                  workspace=(dw_p, db_p), slabs=slabs)
        api.gemm(dh2, self.w("W2"), None, dh1, Yact=acts["h1"], trans_b=True,
                 act=1, dact_tanh=True)
        # layer 1
        dw_p, db_p, slabs = self._wg_workspace("W1", self.obs_dim, self.hidden, True)
        api.wgrad(obs_bf16, dh1, self.grad("W1"), self.grad("b1"),
                  workspace=(dw_p, db_p), slabs=slabs, x_perm=a_feistel)

    def alloc_scratch(self, M: int) -> Dict[str, torch.Tensor]:
        dev = self.device
        return {
            "dh2": torch.empty(M, self.hidden, dtype=torch.bfloat16, device=dev),
            "dh1": torch.empty(M, self.hidden, dtype=torch.bfloat16, device=dev),
        }

    # -- optimizer --------------------------------------------------------
    def adam(self, lr: float, *, beta1=0.9, beta2=0.999, eps=1e-8,
             max_grad_norm: float = 0.0) -> None:
        self.adam_step += 1
        clip = None
        if max_grad_norm and max_grad_norm > 0:
            # fused clipping: sumsq partials + per-block scale derivation
            # inside the adam launch (no separate clip_scale kernel)
            clip = (self._clip_part, max_grad_norm)
        # bias correction reads the DEVICE counter (graph-replayable); the
        # host step is a fallback for paths that pass step_ctr=None.
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
        }

    def load_state_dict(self, sd: Dict[str, torch.Tensor]) -> None:
        self.params.copy_(sd["params"].to(self.device))
        self.m.copy_(sd["m"].to(self.device))
        self.v.copy_(sd["v"].to(self.device))
        self.adam_step = int(sd["adam_step"])
        self.adam_ctr.fill_(self.adam_step)
        self.params_bf16.copy_(self.params.to(torch.bfloat16))
        self._refresh_wt()
