"""Op dispatch: HIP kernels on GPU, torch reference implementations on CPU.

The torch implementations are the ORACLES — GPU tests assert the HIP kernel
matches them.  On a CUDA/ROCm device the native extension is required
(fail-loud, no silent eager fallback): see ops/native.py.
"""
from __future__ import annotations

from typing import Optional, Tuple

import os

import torch

from . import native

# A/B knob for the fused LSTM step kernels (GYMFX_LSTM_FUSED=0 forces the
# unfused gemm+cell pairs — bitwise-identical, for measurement)
_LSTM_FUSED = os.environ.get("GYMFX_LSTM_FUSED", "1") != "0"
# Fused BACKWARD step: measured SLOWER than the unfused pair in every
# structure tried (v1 row-slab 54.3 ms, v2 tile-parallel-recompute 73.8 ms,
# v2b register-prefetch 66.8 ms vs 39.4 ms unfused) — the cell's scattered
# input reads need high occupancy, the MFMA dgrad wants fat registers, and
# one kernel cannot have both.  Kept behind an opt-in knob with bitwise
# tests (profiles/PERF_NOTES.md).
_LSTM_BWD_FUSED = os.environ.get("GYMFX_LSTM_BWD_FUSED", "0") == "1"


def _use_native(t: torch.Tensor) -> bool:
    if t.is_cuda:
        native.require()
        return True
    return False


# ---------------------------------------------------------------------------
# GEMM: C = act(A @ B (+bias));  A bf16 [M,K], B bf16 [K,N] (or [N,K] when
# trans_b); act: 0 = f32 out, 1 = bf16 out, 2 = tanh bf16 out.
# dact_tanh: multiply the product by (1 - Yact^2) (fused tanh backward).
# ---------------------------------------------------------------------------

def gemm(
    A: torch.Tensor,
    B: torch.Tensor,
    bias: Optional[torch.Tensor],
    C: torch.Tensor,
    *,
    Yact: Optional[torch.Tensor] = None,
    trans_b: bool = False,
    act: int = 1,
    dact_tanh: bool = False,
    accum: bool = False,
    a_perm: "Optional[dict]" = None,
) -> torch.Tensor:
    """a_perm (gather+first-GEMM fusion): dict(seed, minibatches, ctr_off,
    step_base, mb_ctr) — A rows are read through the epoch Feistel
    permutation; A is the FULL rollout slab and C has minibatch rows."""
    if _use_native(A):
        if a_perm is not None:
            native.require().gemm(A, B, bias, C, Yact, trans_b, act,
                                  dact_tanh, accum,
                                  ap_seed=a_perm["seed"],
                                  ap_minibatches=a_perm["minibatches"],
                                  ap_ctr_off=a_perm.get("ctr_off", 0),
                                  ap_step_base=a_perm["step_base"],
                                  ap_mb_ctr=a_perm["mb_ctr"])
            return C
        native.require().gemm(A, B, bias, C, Yact, trans_b, act, dact_tanh,
                              accum)
        return C
    if a_perm is not None:
        A = A[_perm_rows(A.shape[0], C.shape[0], a_perm)]
    a = A.to(torch.float32)
    b = B.to(torch.float32)
    if trans_b:
        b = b.t()
    out = a @ b
    if bias is not None:
        out = out + bias
    if act == 2:
        out = torch.tanh(out)
    if dact_tanh:
        y = Yact.to(torch.float32)
        out = out * (1.0 - y * y)
    if act == 0:
        if accum:
            C.add_(out)
        else:
            C.copy_(out)
    else:
        C.copy_(out.to(torch.bfloat16))
    return C


def lstm_cell_fwd(
    gates_pre: torch.Tensor,
    gates_h: "Optional[torch.Tensor]",
    c_prev: torch.Tensor,
    c_new: torch.Tensor,
    h_new: torch.Tensor,
    done: "Optional[torch.Tensor]" = None,
    h_masked: "Optional[torch.Tensor]" = None,
    c_masked: "Optional[torch.Tensor]" = None,
    acts_out: "Optional[torch.Tensor]" = None,
) -> None:
    """gate pre-activations = gates_pre (bf16, x@Wx + b) [+ gates_h (bf16,
    h@Wh kept separate so the recurrent GEMM writes 2-byte outputs instead
    of read-modify-writing a fat buffer)] -> c_new f32, h_new bf16.
    With done/h_masked/c_masked the kernel also emits the reset-masked
    state that feeds the next BPTT step (fuses the masked_state launch).
    acts_out [M,4H] bf16: saves the four gate ACTIVATIONS (i,f,g,o
    interleaved) — the backward consumes these instead of re-reading and
    re-activating the pre-activation tensors."""
    if _use_native(gates_pre):
        native.require().lstm_cell_fwd(gates_pre, gates_h, c_prev, c_new,
                                       h_new, done, h_masked, c_masked,
                                       acts_out)
        return
    gp = gates_pre.to(torch.float32)
    if gates_h is not None:
        gp = gp + gates_h.to(torch.float32)
    M, H4 = gp.shape
    H = H4 // 4
    # interleaved gate layout: column 4*k + {0=i,1=f,2=g,3=o} (models/lstm.py)
    gv = gp.view(M, H, 4)
    i = torch.sigmoid(gv[..., 0])
    f = torch.sigmoid(gv[..., 1])
    g = torch.tanh(gv[..., 2])
    o = torch.sigmoid(gv[..., 3])
    if acts_out is not None:
        acts_out.copy_(torch.stack([i, f, g, o], dim=2).view(M, 4 * H)
                       .to(torch.bfloat16))
    c = f * c_prev + i * g
    c_new.copy_(c)
    h_new.copy_((o * torch.tanh(c)).to(torch.bfloat16))
    if done is not None:
        keep = (~done).unsqueeze(1)
        h_masked.copy_(torch.where(keep, h_new, torch.zeros_like(h_new)))
        c_masked.copy_(torch.where(keep, c_new, torch.zeros_like(c_new)))


def lstm_cell_bwd(
    acts: torch.Tensor,
    c_prev: torch.Tensor,
    c_new: torch.Tensor,
    dh_head: torch.Tensor,
    dh_next: Optional[torch.Tensor],
    dc_next: Optional[torch.Tensor],
    done: Optional[torch.Tensor],
    dgates: torch.Tensor,
    dc_prev: torch.Tensor,
) -> None:
    """BPTT cell backward from the SAVED bf16 activations (i,f,g,o
    interleaved, written by the forward cell) — no gate recompute, no
    pre-activation reads.  dh_next/dc_next (grads arriving from step l+1)
    are masked by `done` so nothing propagates across an episode reset."""
    if _use_native(acts):
        native.require().lstm_cell_bwd(acts, c_prev, c_new,
                                       dh_head, dh_next, dc_next, done,
                                       dgates, dc_prev)
        return
    M, H4 = acts.shape
    H = H4 // 4
    mask = None
    if done is not None:
        mask = (~done).to(torch.float32).unsqueeze(1)
    dh_head = dh_head.to(torch.float32)
    # interleaved gate layout: column 4*k + {0=i,1=f,2=g,3=o} (models/lstm.py)
    av = acts.to(torch.float32).view(M, H, 4)
    i = av[..., 0]
    f = av[..., 1]
    g = av[..., 2]
    o = av[..., 3]
    tc = torch.tanh(c_new)
    dh = dh_head.clone()
    if dh_next is not None:
        dh = dh + (mask * dh_next if mask is not None else dh_next)
    dc = dh * o * (1 - tc * tc)
    if dc_next is not None:
        dc = dc + (mask * dc_next if mask is not None else dc_next)
    d = torch.stack([
        dc * g * i * (1 - i),
        dc * c_prev * f * (1 - f),
        dc * i * (1 - g * g),
        dh * tc * o * (1 - o),
    ], dim=2).view(M, 4 * H)
    dgates.copy_(d.to(torch.bfloat16))
    dc_prev.copy_(dc * f)


def lstm_step_fused(
    h_in: torch.Tensor,
    wh_t: torch.Tensor,
    gates_pre: torch.Tensor,
    gates_h_buf: torch.Tensor,
    c_prev: torch.Tensor,
    c_new: torch.Tensor,
    h_new: torch.Tensor,
    done: "Optional[torch.Tensor]" = None,
    h_masked: "Optional[torch.Tensor]" = None,
    c_masked: "Optional[torch.Tensor]" = None,
    acts_out: "Optional[torch.Tensor]" = None,
) -> None:
    """One recurrent step: gates_h = h_in @ Wh^T then the LSTM cell, fused
    into a single kernel on GPU (cell in the GEMM epilogue — removes one
    launch from the sequential BPTT chain).  acts_out (BPTT) saves the
    four gate activations for the backward; the rollout passes None and
    skips that write entirely.  gates_h_buf is scratch for the unfused
    fallback (CPU / unsupported shapes), which is numerically identical."""
    if _LSTM_FUSED and _use_native(gates_pre):
        ok = native.require().lstm_gemm_cell_fwd(
            h_in, wh_t, gates_pre, acts_out,
            c_prev, c_new, h_new, done, h_masked, c_masked)
        if ok:
            return
    gemm(h_in, wh_t, None, gates_h_buf, act=1, trans_b=True)
    lstm_cell_fwd(gates_pre, gates_h_buf, c_prev, c_new, h_new, done,
                  h_masked, c_masked, acts_out)


def lstm_bwd_step(
    acts: torch.Tensor,
    c_prev: torch.Tensor,
    c_new: torch.Tensor,
    dh_head: torch.Tensor,
    dh_next: "Optional[torch.Tensor]",
    dc_next: "Optional[torch.Tensor]",
    done: "Optional[torch.Tensor]",
    wh: torch.Tensor,
    dgates: torch.Tensor,
    dc_prev: torch.Tensor,
    dh_prev: "Optional[torch.Tensor]" = None,
) -> None:
    """One BPTT backward step from saved activations: cell backward
    (dgates, dc_prev) plus the recurrent dgrad ``dh_prev = dgates @ Wh^T``
    for step l-1, fused into a single kernel when GYMFX_LSTM_BWD_FUSED=1.
    ``wh`` is the [H, 4H] weight (trans_b layout).  Default path is the
    unfused pair (measured faster, profiles/PERF_NOTES.md)."""
    if _LSTM_BWD_FUSED and _use_native(acts):
        ok = native.require().lstm_bwd_fused(
            acts, c_prev, c_new, dh_head, dh_next, dc_next,
            done, wh, dgates, dc_prev, dh_prev)
        if ok:
            return
    lstm_cell_bwd(acts, c_prev, c_new, dh_head, dh_next,
                  dc_next, done, dgates, dc_prev)
    if dh_prev is not None:
        gemm(dgates, wh, None, dh_prev, act=0, trans_b=True)


def mask_reset(h: torch.Tensor, c: torch.Tensor, done: torch.Tensor) -> None:
    """Zero the recurrent state of terminated envs (rollout autoreset)."""
    if _use_native(c):
        native.require().mask_reset(h, c, done)
        return
    h[done] = 0
    c[done] = 0


def masked_state(h_raw: torch.Tensor, c_raw: torch.Tensor,
                 done: torch.Tensor, h_in: torch.Tensor,
                 c_in: torch.Tensor) -> None:
    """h_in/c_in = raw state zeroed where done (BPTT step input)."""
    if _use_native(c_raw):
        native.require().masked_state(h_raw, c_raw, done, h_in, c_in)
        return
    keep = (~done).unsqueeze(1)
    h_in.copy_(torch.where(keep, h_raw, torch.zeros_like(h_raw)))
    c_in.copy_(torch.where(keep, c_raw, torch.zeros_like(c_raw)))


def wgrad(
    X: torch.Tensor,
    dY: torch.Tensor,
    dW: torch.Tensor,
    db: Optional[torch.Tensor],
    *,
    workspace: Optional[Tuple[torch.Tensor, Optional[torch.Tensor]]] = None,
    slabs: int = 64,
    x_perm: "Optional[dict]" = None,
) -> None:
    """dW[K,N] = X^T @ dY ; db[N] = colsum(dY). Deterministic split-M.
    x_perm: same contract as gemm's a_perm (X rows via the permutation)."""
    if _use_native(X):
        K, N = dW.shape
        if workspace is None:
            dW_part = torch.empty(slabs, K, N, dtype=torch.float32, device=X.device)
            db_part = (
                torch.empty(slabs, N, dtype=torch.float32, device=X.device)
                if db is not None
                else None
            )
        else:
            dW_part, db_part = workspace
        if x_perm is not None:
            native.require().wgrad(X, dY, dW_part, db_part, dW, db, slabs,
                                   ap_seed=x_perm["seed"],
                                   ap_minibatches=x_perm["minibatches"],
                                   ap_ctr_off=x_perm.get("ctr_off", 0),
                                   ap_step_base=x_perm["step_base"],
                                   ap_mb_ctr=x_perm["mb_ctr"])
            return
        native.require().wgrad(X, dY, dW_part, db_part, dW, db, slabs)
        return
    if x_perm is not None:
        X = X[_perm_rows(X.shape[0], dY.shape[0], x_perm)]
    x = X.to(torch.float32)
    dy = dY.to(torch.float32)
    dW.copy_(x.t() @ dy)
    if db is not None:
        db.copy_(dy.sum(dim=0))


def gae(
    rewards: torch.Tensor,
    values: torch.Tensor,
    dones: torch.Tensor,
    adv: torch.Tensor,
    ret: torch.Tensor,
    gamma: float,
    lam: float,
) -> None:
    if _use_native(rewards):
        native.require().gae(rewards, values, dones, adv, ret, gamma, lam)
        return
    T, N = rewards.shape
    running = torch.zeros(N, dtype=torch.float32)
    for t in range(T - 1, -1, -1):
        nonterm = (~dones[t]).to(torch.float32)
        delta = rewards[t] + gamma * values[t + 1] * nonterm - values[t]
        running = delta + gamma * lam * nonterm * running
        adv[t] = running
        ret[t] = running + values[t]


def adam(
    p: torch.Tensor,
    g: torch.Tensor,
    m: torch.Tensor,
    v: torch.Tensor,
    p_bf16: Optional[torch.Tensor],
    *,
    lr: float,
    beta1: float = 0.9,
    beta2: float = 0.999,
    eps: float = 1e-8,
    step: int = 1,
    gscale: Optional[torch.Tensor] = None,
    step_ctr: Optional[torch.Tensor] = None,
    clip: "Optional[Tuple[torch.Tensor, float]]" = None,
) -> None:
    """step_ctr (device int32 scalar): when given, bias correction uses
    ``*step_ctr + 1`` instead of the host ``step`` (hipGraph-replayable; pair
    with ``increment_counter(step_ctr, 1)`` after the call).
    clip=(partials, max_norm): fused grad clipping — the sumsq partials are
    computed and the scale derived inside the adam launch (one fewer kernel
    in the 32x-per-update optimizer chain)."""
    if _use_native(p):
        if clip is not None:
            part, max_norm = clip
            native.require().adam(p, g, m, v, p_bf16, lr, beta1, beta2, eps,
                                  step, None, step_ctr, part, float(max_norm))
        else:
            native.require().adam(p, g, m, v, p_bf16, lr, beta1, beta2, eps,
                                  step, gscale, step_ctr)
        return
    if step_ctr is not None:
        step = int(step_ctr.item()) + 1
    if clip is not None:
        _, max_norm = clip
        norm = g.norm()
        s = float(max_norm / norm) if (max_norm > 0 and norm > max_norm) else 1.0
    else:
        s = float(gscale.item()) if gscale is not None else 1.0
    geff = g * s
    m.mul_(beta1).add_(geff, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(geff, geff, value=1 - beta2)
    mhat = m / (1 - beta1 ** step)
    vhat = v / (1 - beta2 ** step)
    p.addcdiv_(mhat, vhat.sqrt() + eps, value=-lr)
    if p_bf16 is not None:
        p_bf16.copy_(p.to(torch.bfloat16))


def grad_clip_scale(
    g: torch.Tensor, max_norm: float, part: torch.Tensor, scale: torch.Tensor
) -> None:
    """scale = min(1, max_norm / ||g||) written into `scale` (device scalar)."""
    if _use_native(g):
        native.require().grad_clip(g, max_norm, part, scale)
        return
    norm = g.norm()
    s = max_norm / norm if (max_norm > 0 and norm > max_norm) else torch.ones(())
    scale.fill_(float(s))


def sample_head(
    head: torch.Tensor,
    seed: int,
    step: int,
    actions: torch.Tensor,
    logp: torch.Tensor,
    value: Optional[torch.Tensor] = None,
    entropy: Optional[torch.Tensor] = None,
    greedy: bool = False,
    step_base: Optional[torch.Tensor] = None,
    row_offset: int = 0,
) -> None:
    """head [M, A+1] f32 (logits | value) -> categorical sample + logp.

    step_base (device u64/i64 scalar): added to ``step`` inside the kernel so
    a captured hipGraph replays with fresh randomness.  row_offset: global
    row index of row 0 (split launches draw the same RNG stream as one
    full-width launch)."""
    if _use_native(head):
        native.require().sample_head(head, seed, step, actions, logp, value,
                                     entropy, greedy, step_base, row_offset)
        return
    if step_base is not None:
        step = step + int(step_base.item())
    M, W = head.shape
    A = W - 1
    logits = head[:, :A]
    logz = torch.logsumexp(logits, dim=1, keepdim=True)
    logpi = logits - logz
    pi = logpi.exp()
    if greedy:
        a = logits.argmax(dim=1)
    else:
        # same counter-based RNG as the kernel (splitmix64)
        u = _splitmix_uniform(seed, step, M, row_offset)
        cdf = pi.cumsum(dim=1)
        a = (u.unsqueeze(1) >= cdf).sum(dim=1).clamp(max=A - 1)
    actions.copy_(a)
    logp.copy_(logpi.gather(1, a.unsqueeze(1)).squeeze(1))
    if value is not None:
        value.copy_(head[:, A])
    if entropy is not None:
        entropy.copy_(-(pi * logpi).sum(dim=1))


def _splitmix_uniform(seed: int, step: int, M: int,
                      row_offset: int = 0) -> torch.Tensor:
    m = torch.arange(row_offset, row_offset + M, dtype=torch.int64)
    x = (seed ^ (step * 0x51E1F5 + m * 0x9E37)) & 0xFFFFFFFFFFFFFFFF

    def mix(x):
        mask = (1 << 64) - 1
        x = (x + 0x9E3779B97F4A7C15) & mask
        x = ((x ^ (x >> 30)) * 0xBF58476D1CE4E5B9) & mask
        x = ((x ^ (x >> 27)) * 0x94D049BB133111EB) & mask
        return x ^ (x >> 31)

    # python-int loop for exact uint64 semantics (CPU oracle only)
    vals = [mix(int(v)) for v in x.tolist()]
    u = torch.tensor([(v >> 11) * (1.0 / 9007199254740992.0) for v in vals],
                     dtype=torch.float32)
    return u.clamp(max=0.999999)


def ppo_loss_bwd(
    head: torch.Tensor,
    actions: torch.Tensor,
    old_logp: torch.Tensor,
    adv: torch.Tensor,
    ret: torch.Tensor,
    dhead: torch.Tensor,
    *,
    clip_eps: float,
    ent_coef: float,
    vf_coef: float,
    inv_count: float,
    losses: Optional[torch.Tensor] = None,
) -> None:
    """Clipped-PPO backward: writes dhead (bf16) and accumulates logging
    scalars [pi_loss, v_loss, entropy, approx_kl, clipfrac] into `losses`."""
    if _use_native(head):
        native.require().ppo_loss_bwd(
            head, actions, old_logp, adv, ret, dhead, clip_eps, ent_coef,
            vf_coef, inv_count, losses,
        )
        return
    M, W = head.shape
    A = W - 1
    logits = head[:, :A]
    logz = torch.logsumexp(logits, dim=1, keepdim=True)
    logpi = logits - logz
    pi = logpi.exp()
    lp = logpi.gather(1, actions.unsqueeze(1)).squeeze(1)
    ratio = (lp - old_logp).exp()
    surr1 = ratio * adv
    rclip = ratio.clamp(1 - clip_eps, 1 + clip_eps)
    surr2 = rclip * adv
    g_lp = torch.where(surr1 <= surr2, -adv * ratio, torch.zeros_like(ratio))
    onehot = torch.nn.functional.one_hot(actions, A).to(torch.float32)
    H = -(pi * logpi).sum(dim=1)
    dlogits = g_lp.unsqueeze(1) * (onehot - pi)
    dlogits = dlogits + ent_coef * pi * (logpi + H.unsqueeze(1))
    v = head[:, A]
    dv = vf_coef * (v - ret)
    d = torch.cat([dlogits, dv.unsqueeze(1)], dim=1) * inv_count
    dhead.copy_(d.to(torch.bfloat16))
    if losses is not None:
        losses[0] += float((-torch.minimum(surr1, surr2)).sum()) * inv_count
        losses[1] += float((0.5 * (v - ret) ** 2).sum()) * inv_count
        losses[2] += float(H.sum()) * inv_count
        losses[3] += float((old_logp - lp).sum()) * inv_count
        clipped = (ratio > 1 + clip_eps) | (ratio < 1 - clip_eps)
        losses[4] += float(clipped.to(torch.float32).sum()) * inv_count


def adv_normalize(adv: torch.Tensor, part: torch.Tensor) -> None:
    if _use_native(adv):
        native.require().adv_normalize(adv, part)
        return
    mean = adv.mean()
    # population std (matches kernel): sqrt(E[x^2] - mean^2)
    var = (adv * adv).mean() - mean * mean
    adv.sub_(mean).div_(var.clamp(min=0).sqrt() + 1e-8)


def f32_to_bf16(src: torch.Tensor, dst: torch.Tensor) -> None:
    if _use_native(src):
        native.require().f32_to_bf16(src, dst)
        return
    dst.view(-1).copy_(src.reshape(-1).to(torch.bfloat16))


# ---------------------------------------------------------------------------
# Feistel epoch permutation (stateless bijection) + fused minibatch gather.
# ---------------------------------------------------------------------------

_U64 = (1 << 64) - 1


def _splitmix64_int(x: int) -> int:
    x = (x + 0x9E3779B97F4A7C15) & _U64
    x = ((x ^ (x >> 30)) * 0xBF58476D1CE4E5B9) & _U64
    x = ((x ^ (x >> 27)) * 0x94D049BB133111EB) & _U64
    return x ^ (x >> 31)


def feistel_key(seed: int, step_base: int, epoch: int) -> int:
    """Key schedule — must match mb_gather_kernel exactly."""
    return _splitmix64_int(
        (seed ^ (step_base * 0x9E3779B97F4A7C15) ^ (epoch << 32)) & _U64
    )


def _perm_rows(n_rows: int, m_rows: int, perm: dict) -> torch.Tensor:
    """CPU fallback of the a_perm row selection (same indices the fused
    kernels compute on device)."""
    ctr = int(perm["mb_ctr"].item()) + int(perm.get("ctr_off", 0))
    minibatches = int(perm["minibatches"])
    epoch, mb = ctr // minibatches, ctr % minibatches
    key = feistel_key(int(perm["seed"]), int(perm["step_base"].item()), epoch)
    return feistel_perm(n_rows, key)[mb * m_rows:(mb + 1) * m_rows]


def feistel_perm(n: int, key: int) -> torch.Tensor:
    """dst -> src index map: 4-round balanced Feistel over the smallest
    even-bit power-of-two domain >= n, cycle-walked into [0, n).  CPU oracle
    of feistel_perm_idx in ops/csrc/ppo_kernels.hip (vectorized numpy)."""
    import numpy as np

    bits = 2
    while (1 << bits) < n:
        bits += 2
    half = bits // 2
    mask = np.uint64((1 << half) - 1)
    x = np.arange(n, dtype=np.uint64)
    active = np.ones(n, dtype=bool)
    while active.any():
        xa = x[active]
        a = xa & mask
        b = xa >> np.uint64(half)
        for r in range(4):
            h = (np.uint64(key) ^ np.uint64(r << 48) ^ b).astype(np.uint64)
            # splitmix64 vectorized (uint64 wraparound is numpy semantics)
            with np.errstate(over="ignore"):
                h = h + np.uint64(0x9E3779B97F4A7C15)
                h = (h ^ (h >> np.uint64(30))) * np.uint64(0xBF58476D1CE4E5B9)
                h = (h ^ (h >> np.uint64(27))) * np.uint64(0x94D049BB133111EB)
                h = h ^ (h >> np.uint64(31))
            f = h & mask
            a, b = b, a ^ f
        xa = (b << np.uint64(half)) | a
        x[active] = xa
        active = x >= n
    return torch.from_numpy(x.astype(np.int64))


def mb_gather(
    obs_src: torch.Tensor,
    act_src: torch.Tensor,
    logp_src: torch.Tensor,
    adv_src: torch.Tensor,
    ret_src: torch.Tensor,
    obs_mb: torch.Tensor,
    act_mb: torch.Tensor,
    logp_mb: torch.Tensor,
    adv_mb: torch.Tensor,
    ret_mb: torch.Tensor,
    *,
    seed: int,
    minibatches: int,
    step_base: torch.Tensor,
    mb_ctr: torch.Tensor,
    skip_obs: bool = False,
) -> None:
    """Gather minibatch ``mb_ctr % minibatches`` of epoch
    ``mb_ctr // minibatches`` under the Feistel permutation keyed by
    (seed, *step_base, epoch).  One fused kernel on GPU.  skip_obs: the
    gather+first-GEMM fusion reads obs through the permutation itself, so
    only the small per-row fields are copied."""
    if _use_native(obs_src):
        native.require().mb_gather(
            obs_src, act_src, logp_src, adv_src, ret_src, obs_mb, act_mb,
            logp_mb, adv_mb, ret_mb, seed, minibatches, step_base, mb_ctr,
            skip_obs,
        )
        return
    n = obs_src.shape[0]
    M = obs_mb.shape[0]
    ctr = int(mb_ctr.item())
    epoch, mb = ctr // minibatches, ctr % minibatches
    key = feistel_key(seed, int(step_base.item()), epoch)
    src = feistel_perm(n, key)[mb * M:(mb + 1) * M]
    if not skip_obs:
        obs_mb.copy_(obs_src[src])
    act_mb.copy_(act_src[src])
    logp_mb.copy_(logp_src[src])
    adv_mb.copy_(adv_src[src])
    ret_mb.copy_(ret_src[src])


def mb_gather_seq(
    obs_src: torch.Tensor,      # [T, N, D] bf16
    act_src: torch.Tensor,      # [T, N]
    logp_src: torch.Tensor,
    adv_src: torch.Tensor,
    ret_src: torch.Tensor,
    done_src: torch.Tensor,     # [T, N] bool
    h0_src: torch.Tensor,       # [n_chunks, N, H] f32
    c0_src: torch.Tensor,
    obs_mb: torch.Tensor,       # [L, Mseq, D] bf16
    act_mb: torch.Tensor,
    logp_mb: torch.Tensor,
    adv_mb: torch.Tensor,
    ret_mb: torch.Tensor,
    done_mb: torch.Tensor,      # [L, Mseq] bool
    h0_mb: torch.Tensor,        # [Mseq, H] bf16
    c0_mb: torch.Tensor,        # [Mseq, H] f32
    *,
    L: int,
    seed: int,
    minibatches: int,
    step_base: torch.Tensor,
    mb_ctr: torch.Tensor,
) -> None:
    """Recurrent-PPO sequence gather: minibatch of (chunk, env) sequences
    under the Feistel permutation, time-major output + chunk-boundary
    recurrent state."""
    n_chunks, N, H = h0_src.shape
    if _use_native(obs_src):
        native.require().mb_gather_seq(
            obs_src.view(-1, obs_src.shape[-1]), act_src.view(-1),
            logp_src.view(-1), adv_src.view(-1), ret_src.view(-1),
            done_src.view(-1),
            h0_src, c0_src, obs_mb.view(-1, obs_mb.shape[-1]),
            act_mb.view(-1), logp_mb.view(-1), adv_mb.view(-1),
            ret_mb.view(-1), done_mb.view(-1), h0_mb, c0_mb, L, N, seed,
            minibatches, step_base, mb_ctr,
        )
        return
    Mseq = c0_mb.shape[0]
    n_seq = n_chunks * N
    ctr = int(mb_ctr.item())
    epoch, mb = ctr // minibatches, ctr % minibatches
    key = feistel_key(seed, int(step_base.item()), epoch)
    src = feistel_perm(n_seq, key)[mb * Mseq:(mb + 1) * Mseq]
    chunk = src // N
    env = src % N
    h0_mb.copy_(h0_src[chunk, env].to(torch.bfloat16))
    c0_mb.copy_(c0_src[chunk, env])
    for li in range(L):
        t = chunk * L + li
        obs_mb[li].copy_(obs_src[t, env])
        act_mb[li].copy_(act_src[t, env])
        logp_mb[li].copy_(logp_src[t, env])
        adv_mb[li].copy_(adv_src[t, env])
        ret_mb[li].copy_(ret_src[t, env])
        done_mb[li].copy_(done_src[t, env])


def increment_counter(ctr: torch.Tensor, delta: int) -> None:
    if _use_native(ctr):
        native.require().increment_counter(ctr, delta)
        return
    ctr.add_(delta)


def transpose_bf16(src: torch.Tensor, dst: torch.Tensor) -> None:
    """dst[N,K] = src[K,N]^T (bf16). Keeps the model's transposed weight
    mirrors fresh so GEMM B operands always stage contiguously."""
    if _use_native(src):
        native.require().transpose_bf16(src, dst)
        return
    dst.copy_(src.t())
