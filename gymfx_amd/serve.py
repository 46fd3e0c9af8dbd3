"""Policy serving: checkpoint -> batched actions over HTTP (mode=serve).

The reference has no serving story (its inference mode drives one env with
a scripted strategy, app/main.py:57-66).  This is the deployment-side
counterpart of mode=training for the MI355X stack: load a checkpoint, hold
the policy resident on the device, and answer batched observation requests
with greedy (or sampled) actions through the same MFMA forward kernels the
trainer uses — one process per GPU scales service capacity exactly like
training.

Endpoints
---------
GET  /health            -> {status, policy, obs_dim, n_actions, device}
GET  /metrics           -> Prometheus exposition (request/action counters,
                           batch-latency histogram, live session gauge)
POST /act               -> {"obs": [[f32 x obs_dim] x B], "greedy": bool,
                            "session": str|null}
                        -> {actions, logp, value}
POST /session/reset     -> {"session": str} zero a recurrent session state
Recurrent policies keep per-session (h, c) server-side, keyed by the
``session`` field; stateless MLP ignores it.

NOTE: no ``from __future__ import annotations`` here — PEP 563 string
annotations cannot resolve the closure-local pydantic models FastAPI needs
to see as real classes (they would silently become query params).
"""
import threading
from collections import OrderedDict
from typing import Any, Dict, List, Optional

import numpy as np
import torch

from .ops import api


class SessionBatchMismatch(ValueError):
    """A recurrent session was continued with a different batch size —
    silently reallocating would corrupt LSTM continuity, so refuse."""


class PolicyServer:
    """Device-resident policy with batched act(); framework-agnostic core
    (the FastAPI app below is a thin shell so this stays testable)."""

    def __init__(self, config: Dict[str, Any]):
        from . import build_vec_environment
        from .algo.ppo import PPOConfig

        cfg = dict(config)
        cfg.setdefault("autoreset", False)
        cfg.setdefault("env_start_mode", "zero")
        # a tiny template env provides obs_dim + device; the model shape
        # comes from the checkpoint (inference-only load: the training-time
        # n_envs does NOT have to match this template env)
        env = build_vec_environment(cfg)
        env.reset(seed=int(cfg.get("seed") or 0))
        pc = PPOConfig.from_config(cfg)
        ckpt = cfg.get("checkpoint_file")
        if ckpt:
            from .utils.checkpoint import load_model_for_inference

            model, meta = load_model_for_inference(ckpt, env.device)
            if meta["obs_dim"] != env.obs_dim:
                raise ValueError(
                    f"checkpoint obs_dim {meta['obs_dim']} != serving env "
                    f"obs_dim {env.obs_dim} (feature/window config mismatch)")
            self.policy = meta["policy"]
        else:
            if pc.policy == "lstm":
                from .models.lstm import ActorCriticLSTM

                model = ActorCriticLSTM(env.obs_dim, 3, pc.hidden,
                                        device=env.device, seed=pc.seed)
            else:
                from .models.mlp import ActorCriticMLP

                model = ActorCriticMLP(env.obs_dim, 3, pc.hidden,
                                       device=env.device, seed=pc.seed)
            self.policy = pc.policy
        self.model = model
        self.recurrent = self.policy == "lstm"
        self.device = env.device
        self.obs_dim = env.obs_dim
        self.n_actions = 3
        self.checkpoint = ckpt
        self.sample_seed = int(cfg.get("seed") or 0) * 7919 + 17
        self.max_sessions = int(cfg.get("serve_max_sessions") or 1024)
        self._step = 0
        self._sessions: "OrderedDict[str, Dict[str, torch.Tensor]]" = OrderedDict()
        # FastAPI serves sync endpoints from a threadpool: one lock keeps
        # the RNG step counter and per-session recurrent state consistent
        # (GPU throughput comes from batching within a request, not from
        # concurrent kernel submission).
        self._lock = threading.Lock()

    # -- recurrent session state ---------------------------------------
    def _session_state(self, session: str, batch: int) -> Dict[str, torch.Tensor]:
        st = self._sessions.get(session)
        if st is not None:
            if st["h"].shape[0] != batch:
                raise SessionBatchMismatch(
                    f"session {session!r} holds state for batch "
                    f"{st['h'].shape[0]}, request has batch {batch}; reset "
                    "the session or keep the batch size constant")
            self._sessions.move_to_end(session)  # LRU touch
            return st
        st = self.model.alloc_state(batch)
        self._sessions[session] = st
        # bounded session map: arbitrary client-supplied keys must not grow
        # device memory without limit — evict least-recently-used
        while len(self._sessions) > self.max_sessions:
            self._sessions.popitem(last=False)
        return st

    def reset_session(self, session: str) -> bool:
        with self._lock:
            return self._sessions.pop(session, None) is not None

    # -- inference ------------------------------------------------------
    def act(self, obs: np.ndarray, *, greedy: bool = True,
            session: Optional[str] = None) -> Dict[str, List[float]]:
        if obs.ndim != 2 or obs.shape[1] != self.obs_dim:
            raise ValueError(
                f"obs must be [batch, {self.obs_dim}], got {list(obs.shape)}")
        with self._lock:
            return self._act_locked(obs, greedy=greedy, session=session)

    def _act_locked(self, obs: np.ndarray, *, greedy: bool,
                    session: Optional[str]) -> Dict[str, List[float]]:
        B = obs.shape[0]
        obs_bf16 = torch.from_numpy(np.ascontiguousarray(obs, dtype=np.float32)) \
            .to(self.device).to(torch.bfloat16)
        acts_buf = self.model.alloc_acts(B)
        if self.recurrent:
            state = self._session_state(session or "default", B)
            head = self.model.step_forward(obs_bf16, state, acts_buf)
        else:
            head = self.model.forward(obs_bf16, acts_buf)
        actions = torch.empty(B, dtype=torch.int64, device=self.device)
        logp = torch.empty(B, dtype=torch.float32, device=self.device)
        value = torch.empty(B, dtype=torch.float32, device=self.device)
        api.sample_head(head, self.sample_seed, self._step, actions, logp,
                        value, greedy=greedy)
        self._step += 1
        return {
            "actions": actions.cpu().tolist(),
            "logp": [float(x) for x in logp.cpu().tolist()],
            "value": [float(x) for x in value.cpu().tolist()],
        }


def create_app(config: Dict[str, Any]):
    """FastAPI app around a PolicyServer (import-light: fastapi only
    needed when serving is actually used)."""
    from fastapi import FastAPI, HTTPException
    from pydantic import BaseModel

    server = PolicyServer(config)
    app = FastAPI(title="gymfx-amd policy server")
    app.state.server = server

    # Prometheus observability (production serving): a private registry so
    # repeated create_app calls (tests, multi-app processes) don't collide
    # in the global default registry.
    from prometheus_client import (CollectorRegistry, Counter, Gauge,
                                   Histogram, generate_latest,
                                   CONTENT_TYPE_LATEST)

    registry = CollectorRegistry()
    m_requests = Counter("gymfx_serve_requests_total",
                         "act requests served", ["outcome"],
                         registry=registry)
    m_actions = Counter("gymfx_serve_actions_total",
                        "actions produced", registry=registry)
    m_latency = Histogram(
        "gymfx_serve_batch_latency_seconds",
        "wall time of one batched act() call",
        buckets=(.0005, .001, .0025, .005, .01, .025, .05, .1, .25, 1.0),
        registry=registry)
    m_sessions = Gauge("gymfx_serve_live_sessions",
                       "recurrent sessions held server-side",
                       registry=registry)
    m_sessions.set_function(lambda: float(len(server._sessions)))

    class ActRequest(BaseModel):
        obs: List[List[float]]
        greedy: bool = True
        session: Optional[str] = None

    class SessionRequest(BaseModel):
        session: str

    @app.get("/health")
    def health():
        return {
            "status": "ok",
            "policy": server.policy,
            "obs_dim": server.obs_dim,
            "n_actions": server.n_actions,
            "device": str(server.device),
            "checkpoint": server.checkpoint,
        }

    @app.post("/act")
    def act(req: ActRequest):
        import time as _time

        t0 = _time.perf_counter()
        try:
            arr = np.asarray(req.obs, dtype=np.float32)
            out = server.act(arr, greedy=req.greedy, session=req.session)
        except SessionBatchMismatch as exc:
            m_requests.labels("conflict").inc()
            raise HTTPException(status_code=409, detail=str(exc))
        except ValueError as exc:
            m_requests.labels("invalid").inc()
            raise HTTPException(status_code=422, detail=str(exc))
        m_latency.observe(_time.perf_counter() - t0)
        m_requests.labels("ok").inc()
        m_actions.inc(len(out["actions"]))
        return out

    @app.get("/metrics")
    def metrics():
        from fastapi import Response

        return Response(generate_latest(registry),
                        media_type=CONTENT_TYPE_LATEST)

    @app.post("/session/reset")
    def session_reset(req: SessionRequest):
        return {"reset": server.reset_session(req.session)}

    return app


def serve_from_config(config: Dict[str, Any]) -> Dict[str, Any]:
    """mode=serve entry: run uvicorn until interrupted."""
    import uvicorn

    host = str(config.get("serve_host", "127.0.0.1"))
    port = int(config.get("serve_port", 8400))
    app = create_app(config)
    uvicorn.run(app, host=host, port=port, log_level="warning")
    return {"mode": "serve", "host": host, "port": port}
