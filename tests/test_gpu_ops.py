"""GPU: PPO HIP kernels vs plain PyTorch fp32 references (MFMA GEMM fwd/bwd,
wgrad, GAE, Adam, sampling, PPO loss backward, advantage normalize)."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def _ext():
    from gymfx_amd.ops import native

    return native.require()


def _rand_bf16(*shape, seed=0, scale=1.0):
    g = torch.Generator().manual_seed(seed)
    return (torch.randn(*shape, generator=g) * scale).to(torch.bfloat16).cuda()


# ---------------------------------------------------------------------------
# GEMM
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("M,N,K", [(64, 64, 64), (256, 256, 96), (1000, 68, 132),
                                   (4096, 256, 324), (512, 4, 256),
                                   (16384, 256, 68), (20480, 224, 52)])
def test_gemm_identity_probe_and_random(M, N, K):
    ext = _ext()
    # asymmetric-B identity probe (guide rule G9: transpose-detecting)
    if M == K:
        A = torch.eye(M).to(torch.bfloat16).cuda()
        B = (torch.arange(K * N, dtype=torch.float32).reshape(K, N) % 37 / 37.0)
        Bg = B.to(torch.bfloat16).cuda()
        C = torch.empty(M, N, dtype=torch.float32, device="cuda")
        ext.gemm(A, Bg, None, C, None, False, 0, False)
        torch.cuda.synchronize()
        np.testing.assert_allclose(
            C.cpu().numpy(), Bg.float().cpu().numpy(), rtol=1e-5, atol=1e-5
        )
    A = _rand_bf16(M, K, seed=1)
    B = _rand_bf16(K, N, seed=2)
    bias = torch.randn(N, device="cuda")
    C = torch.empty(M, N, dtype=torch.float32, device="cuda")
    ext.gemm(A, B, bias, C, None, False, 0, False)
    torch.cuda.synchronize()
    ref = A.float() @ B.float() + bias
    np.testing.assert_allclose(C.cpu().numpy(), ref.cpu().numpy(), rtol=2e-2, atol=2e-2)


def test_gemm_tanh_bf16_out():
    ext = _ext()
    M, N, K = 512, 256, 128
    A = _rand_bf16(M, K, seed=3, scale=0.5)
    B = _rand_bf16(K, N, seed=4, scale=0.1)
    bias = torch.randn(N, device="cuda") * 0.1
    C = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
    ext.gemm(A, B, bias, C, None, False, 2, False)
    torch.cuda.synchronize()
    ref = torch.tanh(A.float() @ B.float() + bias)
    np.testing.assert_allclose(C.float().cpu().numpy(), ref.cpu().numpy(),
                               rtol=2e-2, atol=2e-2)


def test_gemm_trans_b_with_dact():
    ext = _ext()
    M, N, K = 512, 128, 256  # dX[M,N] = dY[M,K] @ W^T where W stored [N,K]
    dY = _rand_bf16(M, K, seed=5, scale=0.1)
    W = _rand_bf16(N, K, seed=6, scale=0.1)  # stored [N,K] -> trans_b
    Y = (_rand_bf16(M, N, seed=7, scale=1.0).float().tanh()).to(torch.bfloat16)
    C = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
    ext.gemm(dY, W, None, C, Y, True, 1, True)
    torch.cuda.synchronize()
    ref = (dY.float() @ W.float().t()) * (1 - Y.float() ** 2)
    np.testing.assert_allclose(C.float().cpu().numpy(), ref.cpu().numpy(),
                               rtol=3e-2, atol=3e-2)


@pytest.mark.parametrize("M,K,N", [(8192, 324, 256), (8192, 260, 1024),
                                   (4096, 256, 256), (2048, 68, 64)])
def test_wgrad_matches_and_deterministic(M, K, N):
    ext = _ext()
    X = _rand_bf16(M, K, seed=8, scale=0.3)
    dY = _rand_bf16(M, N, seed=9, scale=0.05)
    slabs = 64
    dW_part = torch.empty(slabs, K, N, dtype=torch.float32, device="cuda")
    db_part = torch.empty(slabs, N, dtype=torch.float32, device="cuda")
    dW = torch.empty(K, N, dtype=torch.float32, device="cuda")
    db = torch.empty(N, dtype=torch.float32, device="cuda")
    ext.wgrad(X, dY, dW_part, db_part, dW, db, slabs)
    torch.cuda.synchronize()
    ref_dW = X.float().t() @ dY.float()
    ref_db = dY.float().sum(0)
    np.testing.assert_allclose(dW.cpu().numpy(), ref_dW.cpu().numpy(), rtol=3e-2, atol=3e-2)
    np.testing.assert_allclose(db.cpu().numpy(), ref_db.cpu().numpy(), rtol=2e-2, atol=2e-1)
    dW2 = torch.empty_like(dW)
    ext.wgrad(X, dY, dW_part, db_part, dW2, db, slabs)
    torch.cuda.synchronize()
    np.testing.assert_array_equal(dW.cpu().numpy(), dW2.cpu().numpy())


def test_gae_matches_reference():
    ext = _ext()
    T, N = 64, 512
    g = torch.Generator().manual_seed(11)
    rewards = torch.randn(T, N, generator=g).cuda()
    values = torch.randn(T + 1, N, generator=g).cuda()
    dones = (torch.rand(T, N, generator=g) < 0.05).cuda()
    adv = torch.empty(T, N, device="cuda")
    ret = torch.empty(T, N, device="cuda")
    ext.gae(rewards, values, dones, adv, ret, 0.99, 0.95)
    torch.cuda.synchronize()
    adv_ref = torch.zeros(T, N)
    running = torch.zeros(N)
    r, v, d = rewards.cpu(), values.cpu(), dones.cpu()
    for t in range(T - 1, -1, -1):
        nonterm = (~d[t]).float()
        delta = r[t] + 0.99 * v[t + 1] * nonterm - v[t]
        running = delta + 0.99 * 0.95 * nonterm * running
        adv_ref[t] = running
    np.testing.assert_allclose(adv.cpu().numpy(), adv_ref.numpy(), rtol=1e-4, atol=1e-4)
    np.testing.assert_allclose(ret.cpu().numpy(), (adv_ref + v[:T]).numpy(),
                               rtol=1e-4, atol=1e-4)


def test_adam_matches_torch():
    ext = _ext()
    n = 10000
    g0 = torch.Generator().manual_seed(13)
    p = torch.randn(n, generator=g0).cuda()
    grad = torch.randn(n, generator=g0).cuda()
    m = torch.zeros(n, device="cuda")
    v = torch.zeros(n, device="cuda")
    pb = torch.empty(n, dtype=torch.bfloat16, device="cuda")
    p_ref = p.clone()
    ext.adam(p, grad, m, v, pb, 1e-3, 0.9, 0.999, 1e-8, 1, None)
    torch.cuda.synchronize()
    mr = 0.1 * grad
    vr = 0.001 * grad * grad
    mh = mr / (1 - 0.9)
    vh = vr / (1 - 0.999)
    ref = p_ref - 1e-3 * mh / (vh.sqrt() + 1e-8)
    np.testing.assert_allclose(p.cpu().numpy(), ref.cpu().numpy(), rtol=1e-5, atol=1e-6)
    np.testing.assert_allclose(pb.float().cpu().numpy(), ref.cpu().numpy(),
                               rtol=1e-2, atol=1e-2)


def test_grad_clip_scale():
    ext = _ext()
    g = torch.full((1000,), 2.0, device="cuda")
    part = torch.zeros(256, device="cuda")
    scale = torch.zeros(1, device="cuda")
    ext.grad_clip(g, 10.0, part, scale)
    torch.cuda.synchronize()
    norm = float(g.norm())
    assert scale.item() == pytest.approx(10.0 / norm, rel=1e-5)
    ext.grad_clip(g, 1e9, part, scale)
    torch.cuda.synchronize()
    assert scale.item() == 1.0


def test_sample_head_distribution_and_determinism():
    ext = _ext()
    M = 200_000
    head = torch.zeros(M, 4, device="cuda")
    head[:, 0] = 1.0  # logits [1,0,0] -> softmax ~ [0.576, 0.212, 0.212]
    actions = torch.empty(M, dtype=torch.int64, device="cuda")
    logp = torch.empty(M, device="cuda")
    value = torch.empty(M, device="cuda")
    ent = torch.empty(M, device="cuda")
    ext.sample_head(head, 42, 7, actions, logp, value, ent, False)
    torch.cuda.synchronize()
    freq = torch.bincount(actions.cpu(), minlength=3).float() / M
    pi = torch.softmax(torch.tensor([1.0, 0.0, 0.0]), dim=0)
    np.testing.assert_allclose(freq.numpy(), pi.numpy(), atol=0.01)
    # logp matches the sampled action's log-probability
    logpi = torch.log_softmax(torch.tensor([1.0, 0.0, 0.0]), 0)
    got = logp.cpu()
    want = logpi[actions.cpu()]
    np.testing.assert_allclose(got.numpy(), want.numpy(), rtol=1e-4, atol=1e-5)
    # determinism: same (seed, step) -> identical sample
    actions2 = torch.empty_like(actions)
    ext.sample_head(head, 42, 7, actions2, logp, value, ent, False)
    torch.cuda.synchronize()
    np.testing.assert_array_equal(actions.cpu().numpy(), actions2.cpu().numpy())
    # greedy picks argmax
    ext.sample_head(head, 0, 0, actions, logp, value, ent, True)
    torch.cuda.synchronize()
    assert int(actions.max()) == 0


def test_ppo_loss_bwd_matches_autograd():
    ext = _ext()
    M, A = 4096, 3
    g = torch.Generator().manual_seed(17)
    head = torch.randn(M, A + 1, generator=g).cuda()
    actions = torch.randint(0, A, (M,), generator=g).cuda()
    old_logp = (torch.log_softmax(head[:, :A], 1)
                .gather(1, actions.unsqueeze(1)).squeeze(1)
                + 0.1 * torch.randn(M, generator=g).cuda())
    adv = torch.randn(M, generator=g).cuda()
    ret = torch.randn(M, generator=g).cuda()
    dhead = torch.empty(M, A + 1, dtype=torch.bfloat16, device="cuda")
    losses = torch.zeros(5, device="cuda")
    clip_eps, ent_coef, vf_coef = 0.2, 0.01, 0.5
    ext.ppo_loss_bwd(head, actions, old_logp, adv, ret, dhead, clip_eps,
                     ent_coef, vf_coef, 1.0 / M, losses)
    torch.cuda.synchronize()

    h = head.detach().clone().requires_grad_(True)
    logits = h[:, :A]
    logpi = torch.log_softmax(logits, 1)
    lp = logpi.gather(1, actions.unsqueeze(1)).squeeze(1)
    ratio = (lp - old_logp).exp()
    surr = torch.minimum(ratio * adv, ratio.clamp(1 - clip_eps, 1 + clip_eps) * adv)
    H = -(logpi.exp() * logpi).sum(1)
    v = h[:, A]
    loss = (-surr - ent_coef * H + vf_coef * 0.5 * (v - ret) ** 2).mean()
    loss.backward()
    np.testing.assert_allclose(
        dhead.float().cpu().numpy(), h.grad.cpu().numpy(), rtol=5e-2, atol=2e-3
    )
    # logging scalars
    assert losses[0].item() == pytest.approx(float((-surr).mean()), rel=2e-3, abs=1e-4)
    assert losses[2].item() == pytest.approx(float(H.mean()), rel=2e-3, abs=1e-4)


def test_adv_normalize():
    ext = _ext()
    g = torch.Generator().manual_seed(19)
    adv = (torch.randn(100_000, generator=g) * 3 + 5).cuda()
    ref = adv.clone()
    part = torch.empty(512, device="cuda")
    ext.adv_normalize(adv, part)
    torch.cuda.synchronize()
    expected = (ref - ref.mean()) / (ref.std(unbiased=False) + 1e-8)
    np.testing.assert_allclose(adv.cpu().numpy(), expected.cpu().numpy(),
                               rtol=1e-3, atol=1e-3)
    assert abs(float(adv.mean())) < 1e-3
    assert float(adv.std()) == pytest.approx(1.0, abs=1e-2)


def test_model_fwd_bwd_matches_torch_oracle():
    """Whole MLP fwd+bwd on GPU kernels vs the CPU torch-oracle model path."""
    from gymfx_amd.models.mlp import ActorCriticMLP

    M, D = 1024, 100
    gm = ActorCriticMLP(D, 3, 256, device=torch.device("cuda"), seed=5)
    cm = ActorCriticMLP(D, 3, 256, device=torch.device("cpu"), seed=5)
    g = torch.Generator().manual_seed(23)
    obs = (torch.randn(M, D, generator=g) * 0.5).to(torch.bfloat16)
    acts_g = gm.alloc_acts(M)
    acts_c = cm.alloc_acts(M)
    hg = gm.forward(obs.cuda(), acts_g)
    hc = cm.forward(obs, acts_c)
    torch.cuda.synchronize()
    np.testing.assert_allclose(hg.cpu().numpy(), hc.numpy(), rtol=3e-2, atol=3e-2)
    dhead = (torch.randn(M, 4, generator=g) * 0.01).to(torch.bfloat16)
    gm.backward(obs.cuda(), acts_g, dhead.cuda(), gm.alloc_scratch(M))
    cm.backward(obs, acts_c, dhead, cm.alloc_scratch(M))
    torch.cuda.synchronize()
    for name in ("W1", "b1", "W2", "b2", "W3", "b3"):
        gg = gm.grad(name).cpu().numpy()
        cc = cm.grad(name).numpy()
        scale = max(np.abs(cc).max(), 1e-6)
        np.testing.assert_allclose(gg / scale, cc / scale, atol=5e-2,
                                   err_msg=f"grad {name}")


# ---------------------------------------------------------------------------
# Feistel minibatch gather + hipGraph device counters
# ---------------------------------------------------------------------------
def test_mb_gather_matches_cpu_oracle():
    from gymfx_amd.ops import api

    ext = _ext()
    n, mbs, D = 96 * 11, 8, 260  # non-pow2 n exercises cycle-walking
    M = n // mbs
    g = torch.Generator().manual_seed(3)
    obs_cpu = torch.randn(n, D, generator=g).to(torch.bfloat16)
    act_cpu = torch.randint(0, 3, (n,), generator=g)
    logp_cpu = torch.randn(n, generator=g)
    adv_cpu = torch.randn(n, generator=g)
    ret_cpu = torch.randn(n, generator=g)
    srcs = [t.cuda() for t in (obs_cpu, act_cpu, logp_cpu, adv_cpu, ret_cpu)]
    outs_g = [torch.empty(M, D, dtype=torch.bfloat16).cuda(),
              torch.empty(M, dtype=torch.int64).cuda(),
              torch.empty(M).cuda(), torch.empty(M).cuda(),
              torch.empty(M).cuda()]
    outs_c = [torch.empty(M, D, dtype=torch.bfloat16),
              torch.empty(M, dtype=torch.int64),
              torch.empty(M), torch.empty(M), torch.empty(M)]
    for ctr_val, sb_val in [(0, 0), (5, 128), (17, 999)]:
        sb_g = torch.tensor(sb_val, dtype=torch.int64).cuda()
        ctr_g = torch.tensor(ctr_val, dtype=torch.int64).cuda()
        ext.mb_gather(*srcs, *outs_g, 42, mbs, sb_g, ctr_g)
        api.mb_gather(obs_cpu, act_cpu, logp_cpu, adv_cpu, ret_cpu, *outs_c,
                      seed=42, minibatches=mbs,
                      step_base=torch.tensor(sb_val, dtype=torch.int64),
                      mb_ctr=torch.tensor(ctr_val, dtype=torch.int64))
        for gpu_t, cpu_t in zip(outs_g, outs_c):
            assert torch.equal(gpu_t.cpu(), cpu_t)


def test_sample_head_step_base_counter():
    ext = _ext()
    M = 512
    g = torch.Generator().manual_seed(1)
    head = torch.randn(M, 4, generator=g).cuda()
    a1 = torch.empty(M, dtype=torch.int64).cuda()
    a2 = torch.empty(M, dtype=torch.int64).cuda()
    lp = torch.empty(M).cuda()
    # step_base + step must equal host-summed step
    sb = torch.tensor(100, dtype=torch.int64).cuda()
    ext.sample_head(head, 7, 3, a1, lp, None, None, False, sb)
    ext.sample_head(head, 7, 103, a2, lp, None, None, False, None)
    assert torch.equal(a1, a2)
    # increment advances the counter on-device
    ext.increment_counter(sb, 28)
    ext.sample_head(head, 7, 3, a1, lp, None, None, False, sb)
    ext.sample_head(head, 7, 131, a2, lp, None, None, False, None)
    assert torch.equal(a1, a2)


def test_adam_device_step_ctr_matches_host_step():
    ext = _ext()
    n = 1000
    g = torch.Generator().manual_seed(2)

    def init():
        p = torch.randn(n, generator=g).cuda()
        return p, torch.randn(n, generator=g).cuda(), torch.zeros(n).cuda(), torch.zeros(n).cuda()

    p1, g1, m1, v1 = init()
    g.manual_seed(2)
    p2, g2, m2, v2 = init()
    ctr = torch.zeros((), dtype=torch.int32).cuda()
    for step in range(1, 4):
        ext.adam(p1, g1, m1, v1, None, 1e-3, 0.9, 0.999, 1e-8, step, None, None)
        ext.adam(p2, g2, m2, v2, None, 1e-3, 0.9, 0.999, 1e-8, 999, None, ctr)
        ext.increment_counter(ctr, 1)
    # host vs device powf for the bias correction differ in the last ulp
    assert torch.allclose(p1, p2, rtol=1e-6, atol=1e-7)
    assert int(ctr.item()) == 3


# ---------------------------------------------------------------------------
# LSTM kernels (recurrent PPO, BASELINE config #4)
# ---------------------------------------------------------------------------
def test_lstm_cell_kernels_match_cpu_oracle():
    from gymfx_amd.ops import api

    ext = _ext()
    M, H = 4096, 256
    g = torch.Generator().manual_seed(5)
    gates = torch.randn(M, 4 * H, generator=g).to(torch.bfloat16)
    c_prev = torch.randn(M, H, generator=g)
    dh_head = torch.randn(M, H, generator=g).to(torch.bfloat16)
    dh_next = torch.randn(M, H, generator=g)
    dc_next = torch.randn(M, H, generator=g)
    done = torch.rand(M, generator=g) < 0.1
    # fwd
    c_new_c = torch.empty(M, H)
    h_new_c = torch.empty(M, H, dtype=torch.bfloat16)
    gates_h = (torch.randn(M, 4 * H, generator=g) * 0.1).to(torch.bfloat16)
    api.lstm_cell_fwd(gates, gates_h, c_prev, c_new_c, h_new_c)
    c_new_g = torch.empty(M, H).cuda()
    h_new_g = torch.empty(M, H, dtype=torch.bfloat16).cuda()
    ext.lstm_cell_fwd(gates.cuda(), gates_h.cuda(), c_prev.cuda(), c_new_g,
                      h_new_g)
    torch.cuda.synchronize()
    np.testing.assert_allclose(c_new_g.cpu().numpy(), c_new_c.numpy(),
                               rtol=1e-4, atol=1e-5)
    np.testing.assert_allclose(h_new_g.float().cpu().numpy(),
                               h_new_c.float().numpy(), rtol=1e-2, atol=1e-2)
    # bwd (with masking) — consumes the saved bf16 activations
    acts = torch.rand(M, 4 * H, generator=g).to(torch.bfloat16)
    dg_c = torch.empty(M, 4 * H, dtype=torch.bfloat16)
    dcp_c = torch.empty(M, H)
    api.lstm_cell_bwd(acts, c_prev, c_new_c, dh_head, dh_next,
                      dc_next, done, dg_c, dcp_c)
    dg_g = torch.empty(M, 4 * H, dtype=torch.bfloat16).cuda()
    dcp_g = torch.empty(M, H).cuda()
    ext.lstm_cell_bwd(acts.cuda(), c_prev.cuda(), c_new_g,
                      dh_head.cuda(),
                      dh_next.cuda(), dc_next.cuda(), done.cuda(), dg_g, dcp_g)
    torch.cuda.synchronize()
    np.testing.assert_allclose(dg_g.float().cpu().numpy(),
                               dg_c.float().numpy(), rtol=2e-2, atol=2e-2)
    np.testing.assert_allclose(dcp_g.cpu().numpy(), dcp_c.numpy(),
                               rtol=1e-3, atol=1e-4)


def test_mask_ops_match_cpu():
    from gymfx_amd.ops import api

    ext = _ext()
    M, H = 1024, 64
    g = torch.Generator().manual_seed(6)
    h = torch.randn(M, H, generator=g).to(torch.bfloat16)
    c = torch.randn(M, H, generator=g)
    done = torch.rand(M, generator=g) < 0.3
    hg, cg = h.clone().cuda(), c.clone().cuda()
    ext.mask_reset(hg, cg, done.cuda())
    hc, cc = h.clone(), c.clone()
    api.mask_reset(hc, cc, done)
    torch.cuda.synchronize()
    assert torch.equal(hg.cpu(), hc) and torch.equal(cg.cpu(), cc)
    h_in_g = torch.empty_like(hg)
    c_in_g = torch.empty_like(cg)
    ext.masked_state(hg, cg, done.cuda(), h_in_g, c_in_g)
    h_in_c = torch.empty_like(hc)
    c_in_c = torch.empty_like(cc)
    api.masked_state(hc, cc, done, h_in_c, c_in_c)
    torch.cuda.synchronize()
    assert torch.equal(h_in_g.cpu(), h_in_c) and torch.equal(c_in_g.cpu(), c_in_c)


def test_mb_gather_seq_matches_cpu_oracle():
    from gymfx_amd.ops import api

    T, N, D, H, L, mbs = 16, 24, 20, 8, 4, 3
    n_chunks = T // L
    Mseq = n_chunks * N // mbs
    g = torch.Generator().manual_seed(7)
    obs = torch.randn(T, N, D, generator=g).to(torch.bfloat16)
    act = torch.randint(0, 3, (T, N), generator=g)
    logp = torch.randn(T, N, generator=g)
    adv = torch.randn(T, N, generator=g)
    ret = torch.randn(T, N, generator=g)
    done = torch.rand(T, N, generator=g) < 0.1
    h0 = torch.randn(n_chunks, N, H, generator=g)
    c0 = torch.randn(n_chunks, N, H, generator=g)

    def outs(dev):
        kw = {"device": dev}
        return [torch.empty(L, Mseq, D, dtype=torch.bfloat16, **kw),
                torch.empty(L, Mseq, dtype=torch.int64, **kw),
                torch.empty(L, Mseq, **kw), torch.empty(L, Mseq, **kw),
                torch.empty(L, Mseq, **kw),
                torch.empty(L, Mseq, dtype=torch.bool, **kw),
                torch.empty(Mseq, H, dtype=torch.bfloat16, **kw),
                torch.empty(Mseq, H, **kw)]

    for ctr, sb in [(0, 0), (4, 16)]:
        oc = outs("cpu")
        api.mb_gather_seq(obs, act, logp, adv, ret, done, h0, c0, *oc,
                          L=L, seed=3, minibatches=mbs,
                          step_base=torch.tensor(sb), mb_ctr=torch.tensor(ctr))
        og = outs("cuda")
        api.mb_gather_seq(obs.cuda(), act.cuda(), logp.cuda(), adv.cuda(),
                          ret.cuda(), done.cuda(), h0.cuda(), c0.cuda(), *og,
                          L=L, seed=3, minibatches=mbs,
                          step_base=torch.tensor(sb).cuda(),
                          mb_ctr=torch.tensor(ctr).cuda())
        torch.cuda.synchronize()
        for a, b in zip(oc, og):
            assert torch.equal(a, b.cpu())


def test_gemm_accum():
    ext = _ext()
    M, N, K = 512, 256, 64
    A = _rand_bf16(M, K, seed=30)
    B = _rand_bf16(N, K, seed=31)  # trans_b layout
    C = torch.randn(M, N, device="cuda")
    C0 = C.clone()
    ext.gemm(A, B, None, C, None, True, 0, False, True)
    torch.cuda.synchronize()
    ref = C0 + A.float() @ B.float().t()
    np.testing.assert_allclose(C.cpu().numpy(), ref.cpu().numpy(), rtol=2e-2,
                               atol=2e-2)
