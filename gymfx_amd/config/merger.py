"""Config precedence merge + typed coercion of unknown CLI args.

Behavioral contract (matches the reference's merge semantics,
/root/reference/app/config_merger.py:37-51, re-expressed here as a tier
fold): plugin defaults are the lowest tier, then DEFAULT_VALUES, then the
config file, then known CLI args (only when not None), then unknown
``--key value`` args with string->typed coercion.
"""
from __future__ import annotations

from typing import Any, Dict, List, Optional

_BOOL_WORDS = {"true": True, "false": False}
_NONE_WORDS = frozenset({"none", "null"})


def process_unknown_args(tokens: List[str]) -> Dict[str, Any]:
    """Pair up ``--key value`` tokens from argparse's unknown-args list.

    A ``--flag`` immediately followed by another ``--flag`` (or by the end
    of the list) is a valueless switch and maps to True; stray positional
    tokens with no preceding flag are dropped.
    """
    overrides: Dict[str, Any] = {}
    pending: Optional[str] = None
    for tok in tokens:
        if tok.startswith("--"):
            if pending is not None:
                overrides[pending] = True
            pending = tok.lstrip("-")
        elif pending is not None:
            overrides[pending] = tok
            pending = None
    if pending is not None:
        overrides[pending] = True
    return overrides


def convert_type(value: Any) -> Any:
    """Best-effort typed coercion of a CLI string: bool words, none words,
    int, float, else the string unchanged.  Non-strings pass through."""
    if not isinstance(value, str):
        return value
    word = value.strip().lower()
    if word in _BOOL_WORDS:
        return _BOOL_WORDS[word]
    if word in _NONE_WORDS:
        return None
    for parse in (int, float):
        try:
            return parse(value)
        except ValueError:
            continue
    return value


def merge_config(
    defaults: Optional[Dict[str, Any]],
    plugin_params1: Optional[Dict[str, Any]],
    plugin_params2: Optional[Dict[str, Any]],
    file_config: Optional[Dict[str, Any]],
    cli_args: Optional[Dict[str, Any]],
    unknown_args: Optional[Dict[str, Any]],
) -> Dict[str, Any]:
    """Fold the precedence tiers lowest-to-highest into one dict.

    Known CLI args participate only when set (not None); unknown CLI args
    win over everything and are coerced from their raw string form.
    """
    tiers = (
        plugin_params1,
        plugin_params2,
        defaults,
        file_config,
        {k: v for k, v in (cli_args or {}).items() if v is not None},
        {k: convert_type(v) for k, v in (unknown_args or {}).items()},
    )
    merged: Dict[str, Any] = {}
    for tier in tiers:
        if tier:
            merged.update(tier)
    return merged
