"""Canonical configuration defaults.

Contract-compatible with the reference config surface (see
/root/reference/app/config.py:1-47): same key names, same defaults, same
precedence semantics (plugin-defaults < DEFAULT_VALUES < file < CLI-known
< CLI-unknown).  Extra keys below the marker are MI355X-native additions
(vectorized-env scale, PPO training, distributed) that the reference has no
counterpart for.
"""
from __future__ import annotations

DEFAULT_VALUES = {
    # execution
    "mode": "inference",  # training|optimization|inference|serve
    "driver_mode": "buy_hold",  # random|buy_hold|flat|replay
    "steps": 500,

    # plugin selection
    "data_feed_plugin": "default_data_feed",
    "broker_plugin": "default_broker",
    "strategy_plugin": "default_strategy",
    "preprocessor_plugin": "default_preprocessor",
    "reward_plugin": "pnl_reward",
    "metrics_plugin": "default_metrics",

    # data + symbol
    "input_data_file": "examples/data/eurusd.csv",
    "date_column": "DATE_TIME",
    "price_column": "CLOSE",
    "instrument": "EUR_USD",
    "timeframe": "M1",
    "headers": True,
    "max_rows": None,

    # env and execution settings
    "window_size": 32,
    "initial_cash": 10000.0,
    "position_size": 1.0,
    "simulation_engine": "vectorized",   # vectorized (native) — reference had backtrader|nautilus
    "execution_cost_profile": None,
    "commission": 0.0,
    "slippage": 0.0,

    # optional replay actions
    "replay_actions_file": None,

    # config I/O
    "remote_log": None,
    "remote_load_config": None,
    "remote_save_config": None,
    "username": None,
    "password": None,
    "load_config": None,
    "save_config": "./config_out.json",
    "save_log": "./debug_out.json",
    "results_file": "./results.json",
    "quiet_mode": False,

    # ------------------------------------------------------------------
    # MI355X-native additions (no reference counterpart)
    # ------------------------------------------------------------------
    "n_envs": 1,                 # vectorized env count (SoA on device)
    "device": "auto",            # auto|cpu|cuda
    "seed": None,
    "autoreset": False,          # auto-reset terminated envs (training)
    "env_start_mode": "zero",    # zero|spread|random — per-env episode start offsets
    "action_space_mode": "discrete",
    "continuous_action_threshold": 0.33,
    "min_equity": None,          # default: initial_cash * 0.01
    "financing_enabled": False,  # FX rollover interest at 22:00 UTC
    "rollover_rate_data": None,  # monthly central-bank rates (LOCATION/TIME/Value)
    "financing_rate_data_file": None,  # or a CSV path with the same columns
    "rollover_hour_utc": 22,
    "enforce_margin_preflight": False,  # deny fills lacking free margin

    # PPO training (mode=training; BASELINE configs #2-#4)
    "policy_model": "mlp",       # mlp | lstm (recurrent PPO)
    "hidden_size": 256,
    "rollout_steps": 128,
    "ppo_epochs": 4,
    "minibatches": 8,
    "bptt_len": 16,              # sequence-chunked BPTT length (lstm)
    "learning_rate": 3e-4,
    "gamma": 0.99,
    "gae_lambda": 0.95,
    "clip_eps": 0.2,
    "ent_coef": 0.01,
    "vf_coef": 0.5,
    "max_grad_norm": 0.5,
    "train_updates": 10,
    "use_graphs": True,          # hipGraph-capture the training loop (GPU)
    "fuse_sample": True,         # sample inside the env-step kernel (GPU)
    "overlap_gather": False,     # side-stream mb gather (measured slower)
    "checkpoint_file": None,     # save/resume path (mode=training)
    "checkpoint_interval": 0,    # save every N updates (0 = final only)
    "resume": False,
    "trace_file": None,          # per-update phase-timing JSONL (HIP events)

    # mode=serve (policy serving over HTTP; see serve.py)
    "serve_host": "127.0.0.1",
    "serve_port": 8400,

    # mode=optimization (built-in random search over hparam_schema)
    "optimization_trials": 16,
    "optimization_steps": 256,         # env steps scored per trial
    "optimization_refine_trials": 0,   # extra trials around the incumbent
}
