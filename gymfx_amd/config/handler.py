"""Config I/O: local JSON load/save (only non-default keys) + optional
HTTP remote load/save/log with basic auth.

Capability parity with the reference's config handler
(/root/reference/app/config_handler.py:6-73), re-expressed around one
shared HTTP helper: saving writes only keys that differ from
DEFAULT_VALUES; remote endpoints exchange ``json_config`` /
``json_result`` form fields (that field naming is the wire contract) and
degrade to stderr warnings on failure — a dead config server never kills
a training run.
"""
from __future__ import annotations

import json
import sys
from typing import Any, Dict, Optional, Tuple

from .defaults import DEFAULT_VALUES


def load_config(file_path: str) -> Dict[str, Any]:
    with open(file_path, "r", encoding="utf-8") as fh:
        return json.load(fh)


def compose_config(config: Dict[str, Any]) -> Dict[str, Any]:
    """The non-default subset: every key whose value differs from
    DEFAULT_VALUES (or is unknown to it)."""
    return {
        k: v for k, v in config.items()
        if DEFAULT_VALUES.get(k, _MISSING) != v
    }


_MISSING = object()  # sentinel: distinguishes "absent" from "None default"


def save_config(config: Dict[str, Any], path: str = "config_out.json"):
    with open(path, "w", encoding="utf-8") as fh:
        json.dump(compose_config(config), fh, indent=4, default=str)
    return config, path


def save_debug_info(debug_info: Dict[str, Any], path: str = "debug_out.json") -> None:
    with open(path, "w", encoding="utf-8") as fh:
        json.dump(debug_info, fh, indent=4, default=str)


# ---------------------------------------------------------------------------
# remote HTTP endpoints (optional capability — requests may be absent)
# ---------------------------------------------------------------------------

def _warn(what: str, detail: Any) -> None:
    print(f"[config] remote {what} skipped/failed: {detail}", file=sys.stderr)


def _post_form(what: str, url: str, auth: Optional[Tuple[str, str]],
               fields: Dict[str, str]) -> bool:
    try:
        import requests
    except ImportError:
        _warn(what, "requests not installed")
        return False
    try:
        resp = requests.post(url, auth=auth, data=fields)
        resp.raise_for_status()
        return True
    except requests.RequestException as exc:
        _warn(what, exc)
        return False


def remote_save_config(config, url, username, password) -> bool:
    body = {"json_config": json.dumps(compose_config(config))}
    return _post_form("save", url, (username, password), body)


def remote_load_config(url, username: Optional[str] = None,
                       password: Optional[str] = None):
    try:
        import requests
    except ImportError:
        _warn("load", "requests not installed")
        return None
    auth = (username, password) if username and password else None
    try:
        resp = requests.get(url, auth=auth)
        resp.raise_for_status()
        return resp.json()
    except requests.RequestException as exc:
        _warn("load", exc)
        return None


def remote_log(config, debug_info, url, username, password) -> bool:
    body = {
        "json_config": json.dumps(compose_config(config)),
        "json_result": json.dumps(debug_info, default=str),
    }
    return _post_form("log", url, (username, password), body)
