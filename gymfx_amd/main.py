#!/usr/bin/env python3
"""Runner (parity: /root/reference/app/main.py:34-96): parse args, merge
config with the five-tier precedence, instantiate the six plugins, build the
env, drive it with the configured driver, write results + config."""
from __future__ import annotations

import json
import os
from pathlib import Path
from typing import Any, Dict

from . import build_environment
from .cli import parse_args
from .config import DEFAULT_VALUES, load_config, merge_config, process_unknown_args, save_config
from .plugins import load_plugin


def _load_plugin_instance(group: str, name: str, config: Dict[str, Any]):
    klass, _ = load_plugin(group, name)
    instance = klass(config)
    instance.set_params(**config)
    return instance


def _collect_plugin_defaults(instances) -> Dict[str, Any]:
    merged: Dict[str, Any] = {}
    for instance in instances:
        merged.update(getattr(instance, "plugin_params", {}))
    return merged


def run_env(config: Dict[str, Any]) -> Dict[str, Any]:
    data_feed = _load_plugin_instance("data_feed.plugins", config["data_feed_plugin"], config)
    broker = _load_plugin_instance("broker.plugins", config["broker_plugin"], config)
    strategy = _load_plugin_instance("strategy.plugins", config["strategy_plugin"], config)
    preprocessor = _load_plugin_instance(
        "preprocessor.plugins", config["preprocessor_plugin"], config
    )
    reward = _load_plugin_instance("reward.plugins", config["reward_plugin"], config)
    metrics = _load_plugin_instance("metrics.plugins", config["metrics_plugin"], config)

    plugin_defaults = _collect_plugin_defaults(
        [data_feed, broker, strategy, preprocessor, reward, metrics]
    )
    config = merge_config(config, plugin_defaults, {}, {}, {}, {})

    env = build_environment(
        config=config,
        data_feed_plugin=data_feed,
        broker_plugin=broker,
        strategy_plugin=strategy,
        preprocessor_plugin=preprocessor,
        reward_plugin=reward,
        metrics_plugin=metrics,
    )
    try:
        obs, info = env.reset(seed=config.get("seed"))
        done = False
        steps = int(config.get("steps", 500))
        step_count = 0
        plugin_apply_errors = 0
        while not done and step_count < steps:
            try:
                action = strategy.decide_action(obs=obs, info=info,
                                                step=step_count)
            except Exception:
                # graceful degradation, never crash mid-rollout: a failing
                # (third-party) strategy plugin falls back to hold and is
                # counted, matching the reference's plugin_apply_errors
                # semantics (bt_bridge.py:191-201)
                plugin_apply_errors += 1
                action = 0
            obs, _, terminated, truncated, info = env.step(action)
            done = bool(terminated or truncated)
            step_count += 1
        summary = env.summary()
        if plugin_apply_errors:
            summary["plugin_apply_errors"] = plugin_apply_errors
        return summary
    finally:
        env.close()


def main(argv=None) -> None:
    args, unknown_args = parse_args(argv)
    cli_args = vars(args)

    config = DEFAULT_VALUES.copy()
    file_config = load_config(args.load_config) if args.load_config else {}
    unknown_args_dict = process_unknown_args(unknown_args)
    # remote-loaded config sits at the file tier (below CLI): an upgrade
    # over the reference, which defines remote config I/O
    # (app/config_handler.py:30-73) but never wires it into the runner.
    remote_url = (cli_args.get("remote_load_config")
                  or unknown_args_dict.get("remote_load_config")
                  or file_config.get("remote_load_config"))
    if remote_url:
        from .config import remote_load_config  # noqa: PLC0415

        remote_cfg = remote_load_config(
            remote_url,
            cli_args.get("username") or file_config.get("username"),
            cli_args.get("password") or file_config.get("password"),
        )
        if remote_cfg:
            file_config = {**remote_cfg, **file_config}
    config = merge_config(config, {}, {}, file_config, cli_args, unknown_args_dict)

    if config.get("mode") not in {"training", "optimization", "inference",
                                   "serve"}:
        raise ValueError(
            "mode must be one of training|optimization|inference|serve")

    if config.get("mode") == "training":
        from .algo.ppo import train_from_config  # noqa: PLC0415

        summary = train_from_config(config)
    elif config.get("mode") == "optimization":
        from .algo.optimize import optimize_from_config  # noqa: PLC0415

        summary = optimize_from_config(config)
    elif config.get("mode") == "serve":
        from .serve import serve_from_config  # noqa: PLC0415

        summary = serve_from_config(config)
    elif (config.get("mode") == "inference" and config.get("checkpoint_file")
          and int(config.get("n_envs", 1) or 1) > 1):
        # vectorized policy evaluation (checkpoint -> greedy rollout);
        # without a checkpoint the classic driver loop below runs instead
        from .algo.evaluate import evaluate_from_config  # noqa: PLC0415

        summary = evaluate_from_config(config)
    else:
        summary = run_env(config)

    # under torchrun only rank 0 writes results/logs (mode=training joins
    # the data-parallel group inside train_from_config)
    if int(os.environ.get("RANK", "0")) != 0:
        return

    results_file = Path(config.get("results_file", "results.json"))
    results_file.parent.mkdir(parents=True, exist_ok=True)
    with results_file.open("w", encoding="utf-8") as fh:
        json.dump(summary, fh, indent=2, default=str)

    if config.get("save_config"):
        save_config(config, config["save_config"])
    # debug log only when a path was explicitly given (the default
    # "./debug_out.json" would otherwise litter every run's cwd)
    explicit_save_log = (cli_args.get("save_log")
                         or unknown_args_dict.get("save_log")
                         or file_config.get("save_log"))
    if explicit_save_log:
        from .config import save_debug_info  # noqa: PLC0415

        save_debug_info(summary, str(explicit_save_log))
    if config.get("remote_save_config"):
        from .config import remote_save_config  # noqa: PLC0415

        remote_save_config(config, config["remote_save_config"],
                           config.get("username"), config.get("password"))
    if config.get("remote_log"):
        from .config import remote_log  # noqa: PLC0415

        remote_log(config, summary, config["remote_log"],
                   config.get("username"), config.get("password"))

    if not config.get("quiet_mode", False):
        print(json.dumps(summary, indent=2, default=str))


if __name__ == "__main__":
    main()
