"""Fused LSTM step kernels vs the unfused pairs — BITWISE equality.

The fused kernels (lstm_gemm_cell_fwd: recurrent MFMA GEMM with the cell
as epilogue; lstm_bwd_fused: cell backward + recurrent dgrad GEMM) must
reproduce the unfused gemm+cell pipelines bit-for-bit: same MFMA K-chunk
order, gates bf16-rounded at the same point (ops/csrc/ppo_kernels.hip).
"""
import pytest
import torch

from gymfx_amd.models.lstm import ActorCriticLSTM
from gymfx_amd.ops import api, native

pytestmark = pytest.mark.gpu


def _model(H=256, D=64):
    return ActorCriticLSTM(D, 3, H, device=torch.device("cuda"), seed=9)


@pytest.mark.parametrize("H", [64, 128, 256])
@pytest.mark.parametrize("with_done", [False, True])
def test_fused_fwd_step_bitwise(H, with_done):
    if not torch.cuda.is_available():
        pytest.skip("needs GPU")
    M, D = 512 + 16, 32  # non-multiple of 64 rows: exercises the tail path
    m = _model(H=H, D=D)
    g = torch.Generator(device="cuda").manual_seed(0)
    h = torch.randn(M, H, generator=g, device="cuda").to(torch.bfloat16)
    c = torch.randn(M, H, generator=g, device="cuda")
    gates = torch.randn(M, 4 * H, generator=g, device="cuda").to(torch.bfloat16)
    done = None
    hm_f = cm_f = hm_u = cm_u = None
    if with_done:
        done = torch.rand(M, generator=g, device="cuda") < 0.3
        hm_f = torch.empty(M, H, dtype=torch.bfloat16, device="cuda")
        cm_f = torch.empty(M, H, device="cuda")
        hm_u = torch.empty_like(hm_f)
        cm_u = torch.empty_like(cm_f)
    # unfused reference (saves the activations like the fused kernel)
    gh_u = torch.empty(M, 4 * H, dtype=torch.bfloat16, device="cuda")
    acts_u = torch.empty(M, 4 * H, dtype=torch.bfloat16, device="cuda")
    c_u = torch.empty(M, H, device="cuda")
    h_u = torch.empty(M, H, dtype=torch.bfloat16, device="cuda")
    api.gemm(h, m.wt("Wh"), None, gh_u, act=1, trans_b=True)
    api.lstm_cell_fwd(gates, gh_u, c, c_u, h_u, done, hm_u, cm_u, acts_u)
    # fused
    acts_f = torch.empty_like(acts_u)
    c_f = torch.empty_like(c_u)
    h_f = torch.empty_like(h_u)
    ok = native.require().lstm_gemm_cell_fwd(h, m.wt("Wh"), gates, acts_f, c,
                                             c_f, h_f, done, hm_f, cm_f)
    assert ok, "fused fwd kernel refused a supported shape"
    torch.cuda.synchronize()
    assert torch.equal(acts_f, acts_u)
    assert torch.equal(c_f, c_u)
    assert torch.equal(h_f, h_u)
    if with_done:
        assert torch.equal(hm_f, hm_u)
        assert torch.equal(cm_f, cm_u)


@pytest.mark.parametrize("H", [64, 256])
@pytest.mark.parametrize("last_step", [False, True])
def test_fused_bwd_step_bitwise(H, last_step):
    if not torch.cuda.is_available():
        pytest.skip("needs GPU")
    M, D = 512 + 16, 32
    m = _model(H=H, D=D)
    g = torch.Generator(device="cuda").manual_seed(1)
    acts = torch.rand(M, 4 * H, generator=g, device="cuda").to(torch.bfloat16)
    c_prev = torch.randn(M, H, generator=g, device="cuda")
    c_new = torch.randn(M, H, generator=g, device="cuda")
    dh_head = torch.randn(M, H, generator=g, device="cuda").to(torch.bfloat16)
    dh_next = None if last_step else torch.randn(M, H, generator=g, device="cuda")
    dc_next = None if last_step else torch.randn(M, H, generator=g, device="cuda")
    done = torch.rand(M, generator=g, device="cuda") < 0.25
    # unfused reference
    dg_u = torch.empty(M, 4 * H, dtype=torch.bfloat16, device="cuda")
    dcp_u = torch.empty(M, H, device="cuda")
    dhp_u = torch.empty(M, H, device="cuda")
    api.lstm_cell_bwd(acts, c_prev, c_new, dh_head, dh_next, dc_next,
                      done, dg_u, dcp_u)
    api.gemm(dg_u, m.w("Wh"), None, dhp_u, act=0, trans_b=True)
    # fused
    dg_f = torch.empty_like(dg_u)
    dcp_f = torch.empty_like(dcp_u)
    dhp_f = torch.empty_like(dhp_u)
    ok = native.require().lstm_bwd_fused(acts, c_prev, c_new, dh_head,
                                         dh_next, dc_next, done, m.w("Wh"),
                                         dg_f, dcp_f, dhp_f)
    assert ok, "fused bwd kernel refused a supported shape"
    torch.cuda.synchronize()
    assert torch.equal(dg_f, dg_u)
    assert torch.equal(dcp_f, dcp_u)
    assert torch.equal(dhp_f, dhp_u)


def test_fused_bwd_skips_dh_for_step0():
    if not torch.cuda.is_available():
        pytest.skip("needs GPU")
    H, M = 64, 128
    m = _model(H=H, D=16)
    g = torch.Generator(device="cuda").manual_seed(2)
    acts = torch.rand(M, 4 * H, generator=g, device="cuda").to(torch.bfloat16)
    c_prev = torch.randn(M, H, generator=g, device="cuda")
    c_new = torch.randn(M, H, generator=g, device="cuda")
    dh_head = torch.randn(M, H, generator=g, device="cuda").to(torch.bfloat16)
    dg = torch.empty(M, 4 * H, dtype=torch.bfloat16, device="cuda")
    dcp = torch.empty(M, H, device="cuda")
    ok = native.require().lstm_bwd_fused(acts, c_prev, c_new, dh_head,
                                         None, None, None, m.w("Wh"), dg,
                                         dcp, None)
    assert ok
    torch.cuda.synchronize()
    dg_u = torch.empty_like(dg)
    dcp_u = torch.empty_like(dcp)
    api.lstm_cell_bwd(acts, c_prev, c_new, dh_head, None, None, None,
                      dg_u, dcp_u)
    torch.cuda.synchronize()
    assert torch.equal(dg, dg_u) and torch.equal(dcp, dcp_u)


def test_fused_step_rollout_matches_unfused_model_path():
    """model.step_forward fused vs fused=False: identical head + state
    (the h tensor identity rotates on the fused path; values must agree)."""
    if not torch.cuda.is_available():
        pytest.skip("needs GPU")
    D, H, M = 40, 256, 256
    m = _model(H=H, D=D)
    g = torch.Generator(device="cuda").manual_seed(3)
    obs = [torch.randn(M, D, generator=g, device="cuda").to(torch.bfloat16)
           for _ in range(5)]

    def run(fused):
        state = m.alloc_state(M)
        acts = m.alloc_acts(M)
        heads = []
        for ob in obs:
            heads.append(m.step_forward(ob, state, acts, fused=fused).clone())
        return heads, state["h"].clone(), state["c"].clone()

    hf, shf, scf = run(True)
    hu, shu, scu = run(False)
    for a, b in zip(hf, hu):
        assert torch.equal(a, b)
    assert torch.equal(shf, shu)
    assert torch.equal(scf, scu)
