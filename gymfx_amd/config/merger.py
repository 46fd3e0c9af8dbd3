"""Config precedence merge + typed coercion of unknown CLI args.

Same precedence contract as the reference merge
(/root/reference/app/config_merger.py:37-51): plugin defaults are the lowest
tier, then DEFAULT_VALUES, then the config file, then known CLI args (only
when not None), then unknown ``--key value`` args with string->typed coercion
(/root/reference/app/config_merger.py:19-35).
"""
from __future__ import annotations

from typing import Any, Dict, List, Optional


def process_unknown_args(unknown_args: List[str]) -> Dict[str, Any]:
    parsed: Dict[str, Any] = {}
    i = 0
    while i < len(unknown_args):
        key = unknown_args[i]
        if not key.startswith("--"):
            i += 1
            continue
        if i + 1 < len(unknown_args) and not unknown_args[i + 1].startswith("--"):
            parsed[key.lstrip("-")] = unknown_args[i + 1]
            i += 2
        else:
            parsed[key.lstrip("-")] = True
            i += 1
    return parsed


def convert_type(value: Any) -> Any:
    if isinstance(value, bool):
        return value
    if not isinstance(value, str):
        return value
    lowered = value.strip().lower()
    if lowered in {"true", "false"}:
        return lowered == "true"
    if lowered in {"none", "null"}:
        return None
    try:
        return int(value)
    except ValueError:
        try:
            return float(value)
        except ValueError:
            return value


def merge_config(
    defaults: Optional[Dict[str, Any]],
    plugin_params1: Optional[Dict[str, Any]],
    plugin_params2: Optional[Dict[str, Any]],
    file_config: Optional[Dict[str, Any]],
    cli_args: Optional[Dict[str, Any]],
    unknown_args: Optional[Dict[str, Any]],
) -> Dict[str, Any]:
    merged: Dict[str, Any] = {}
    merged.update(plugin_params1 or {})
    merged.update(plugin_params2 or {})
    merged.update(defaults or {})
    merged.update(file_config or {})
    for key, value in (cli_args or {}).items():
        if value is not None:
            merged[key] = value
    for key, value in (unknown_args or {}).items():
        merged[key] = convert_type(value)
    return merged
