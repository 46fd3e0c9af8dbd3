"""Recurrent PPO (LSTM actor-critic, BASELINE config #4), CPU oracle path:
cell fwd/bwd vs torch.autograd, BPTT gradient check vs an autograd replica,
reset masking, and end-to-end recurrent trainer determinism."""
import numpy as np
import pytest
import torch

from gymfx_amd import build_vec_environment
from gymfx_amd.algo.ppo import PPOConfig, PPOTrainer
from gymfx_amd.data.feed import synthetic_ohlcv
from gymfx_amd.models.lstm import ActorCriticLSTM
from gymfx_amd.ops import api


def test_lstm_cell_fwd_matches_torch():
    M, H = 32, 16
    g = torch.Generator().manual_seed(0)
    gates = torch.randn(M, 4 * H, generator=g)
    c_prev = torch.randn(M, H, generator=g)
    c_new = torch.empty(M, H)
    h_new = torch.empty(M, H, dtype=torch.bfloat16)
    api.lstm_cell_fwd(gates, None, c_prev, c_new, h_new)
    # interleaved gate layout: column 4*k + {0=i,1=f,2=g,3=o}
    i, f, gg, o = gates.view(gates.shape[0], -1, 4).unbind(2)
    c_ref = torch.sigmoid(f) * c_prev + torch.sigmoid(i) * torch.tanh(gg)
    h_ref = torch.sigmoid(o) * torch.tanh(c_ref)
    assert torch.allclose(c_new, c_ref, atol=1e-6)
    assert torch.allclose(h_new.float(), h_ref, atol=1e-2)


def test_lstm_cell_bwd_matches_autograd():
    M, H = 16, 8
    g = torch.Generator().manual_seed(1)
    gates = torch.randn(M, 4 * H, generator=g, requires_grad=True)
    c_prev = torch.randn(M, H, generator=g, requires_grad=True)
    # interleaved gate layout: column 4*k + {0=i,1=f,2=g,3=o}
    i, f, gg, o = gates.view(gates.shape[0], -1, 4).unbind(2)
    c_new = torch.sigmoid(f) * c_prev + torch.sigmoid(i) * torch.tanh(gg)
    h_new = torch.sigmoid(o) * torch.tanh(c_new)
    dh = torch.randn(M, H, generator=g)
    dc_next = torch.randn(M, H, generator=g)
    # autograd reference: total loss = sum(h*dh) + sum(c_new*dc_next)
    (h_new * dh).sum().backward(retain_graph=True)
    (c_new * dc_next).sum().backward()
    dgates_ref = gates.grad.clone()
    dc_prev_ref = c_prev.grad.clone()

    dgates = torch.empty(M, 4 * H, dtype=torch.bfloat16)
    dc_prev = torch.empty(M, H)
    # backward consumes the SAVED bf16 activations (fwd-cell output)
    acts = torch.stack([torch.sigmoid(i), torch.sigmoid(f), torch.tanh(gg),
                        torch.sigmoid(o)], dim=2).reshape(M, 4 * H) \
        .detach().to(torch.bfloat16)
    api.lstm_cell_bwd(acts, c_prev.detach(), c_new.detach(),
                      dh, None, dc_next, None, dgates, dc_prev)
    assert torch.allclose(dgates.float(), dgates_ref, atol=3e-2, rtol=3e-2)
    assert torch.allclose(dc_prev, dc_prev_ref, atol=2e-2, rtol=2e-2)


def test_lstm_cell_bwd_done_masks_recurrent_grads():
    M, H = 8, 4
    g = torch.Generator().manual_seed(2)
    acts = torch.rand(M, 4 * H, generator=g).to(torch.bfloat16)
    c_prev = torch.randn(M, H, generator=g)
    c_new = torch.randn(M, H, generator=g)
    dh_head = torch.randn(M, H, generator=g).to(torch.bfloat16)
    dh_next = torch.randn(M, H, generator=g)
    dc_next = torch.randn(M, H, generator=g)
    done = torch.zeros(M, dtype=torch.bool)
    done[::2] = True
    dgates_m = torch.empty(M, 4 * H, dtype=torch.bfloat16)
    dc_prev_m = torch.empty(M, H)
    api.lstm_cell_bwd(acts, c_prev, c_new, dh_head, dh_next, dc_next,
                      done, dgates_m, dc_prev_m)
    # for done rows the result must equal the no-next-grad case
    dgates_0 = torch.empty(M, 4 * H, dtype=torch.bfloat16)
    dc_prev_0 = torch.empty(M, H)
    api.lstm_cell_bwd(acts, c_prev, c_new, dh_head, None, None, None,
                      dgates_0, dc_prev_0)
    assert torch.equal(dgates_m[::2], dgates_0[::2])
    assert torch.equal(dc_prev_m[::2], dc_prev_0[::2])
    assert not torch.equal(dgates_m[1::2], dgates_0[1::2])


def test_bptt_matches_autograd_replica():
    """Full BPTT fwd+bwd vs a torch autograd replica of the same network
    (fp32; bf16 weight rounding is shared by feeding the bf16 weights in)."""
    torch.manual_seed(3)
    D, H, A, L, M = 12, 8, 3, 4, 6
    model = ActorCriticLSTM(D, A, H, device=torch.device("cpu"), seed=5)
    obs = (torch.randn(L, M, D) * 0.5).to(torch.bfloat16)
    done = torch.zeros(L, M, dtype=torch.bool)
    done[1, 2] = True  # reset mid-sequence
    h0 = torch.zeros(M, H, dtype=torch.bfloat16)
    c0 = torch.zeros(M, H)
    buf = model.alloc_bptt(L, M)
    head = model.bptt_forward(obs, done, h0, c0, buf).clone()
    dhead = torch.randn(L * M, A + 1) * 0.1
    model.bptt_backward(obs, done, dhead.to(torch.bfloat16), buf)

    # autograd replica
    Wx = model.f32("Wx").detach().clone().requires_grad_(True)
    Wh = model.f32("Wh").detach().clone().requires_grad_(True)
    b = model.f32("b").detach().clone().requires_grad_(True)
    Wy = model.f32("Wy").detach().clone().requires_grad_(True)
    by = model.f32("by").detach().clone().requires_grad_(True)
    # use the same bf16 compute operands the model used
    Wx_c = model.w("Wx").float() + (Wx - Wx.detach())
    Wh_c = model.w("Wh").float() + (Wh - Wh.detach())
    Wy_c = model.w("Wy").float() + (Wy - Wy.detach())
    h = h0.float()
    c = c0.clone()
    heads = []
    for l in range(L):
        gh = (h @ Wh_c)
        gh = gh.to(torch.bfloat16).float().detach() + (gh - gh.detach())  # bf16 fwd
        gx = obs[l].float() @ Wx_c + b
        gx = gx.to(torch.bfloat16).float().detach() + (gx - gx.detach())  # bf16 gates
        gates = gx + gh
        # interleaved gate layout: column 4*k + {0=i,1=f,2=g,3=o}
        i, f, gg, o = gates.view(gates.shape[0], -1, 4).unbind(2)
        c = torch.sigmoid(f) * c + torch.sigmoid(i) * torch.tanh(gg)
        h_raw = torch.sigmoid(o) * torch.tanh(c)
        heads.append(h_raw @ Wy_c + by)
        keep = (~done[l]).float().unsqueeze(1)
        # bf16 rounding of h between steps (matches the kernel dataflow)
        h = (h_raw.to(torch.bfloat16).float()) * keep
        c = c * keep
    head_ref = torch.cat(heads, dim=0)
    assert torch.allclose(head, head_ref, atol=5e-2, rtol=5e-2)
    (head_ref * dhead).sum().backward()
    for name, ref in [("Wx", Wx.grad), ("Wh", Wh.grad), ("b", b.grad),
                      ("Wy", Wy.grad), ("by", by.grad)]:
        got = model.grad(name)
        assert torch.allclose(got, ref, atol=8e-2, rtol=8e-2), name


def _make_rec_trainer(seed=21):
    md = synthetic_ohlcv(600, seed=5, vol=4e-4)
    cfg = {
        "n_envs": 16,
        "device": "cpu",
        "window_size": 8,
        "env_start_mode": "spread",
        "autoreset": True,
        "position_size": 1000.0,
        "seed": seed,
    }
    env = build_vec_environment(cfg, md)
    env.reset(seed=seed)
    pc = PPOConfig(rollout_steps=16, minibatches=4, ppo_epochs=2, seed=seed,
                   hidden=16, policy="lstm", bptt_len=4)
    return PPOTrainer(env, pc)


def test_recurrent_trainer_runs_and_deterministic():
    t1 = _make_rec_trainer()
    t2 = _make_rec_trainer()
    for _ in range(2):
        s1 = t1.train_update()
        s2 = t2.train_update()
    assert torch.equal(t1.model.params, t2.model.params)
    assert s1 == s2
    assert np.isfinite(list(s1.values())).all()
    assert 0.0 < s1["entropy"] <= np.log(3) + 1e-5
    # parameters actually moved
    t3 = _make_rec_trainer()
    assert not torch.equal(t1.model.params, t3.model.params)
