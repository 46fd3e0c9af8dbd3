"""Config I/O: local JSON load/save (only non-default keys) + optional
HTTP remote load/save/log with basic auth.

Capability parity with /root/reference/app/config_handler.py:6-73: saving
writes only keys that differ from DEFAULT_VALUES; remote endpoints post
``json_config`` / ``json_result`` form fields with basic auth and degrade to
stderr warnings on failure (never raise mid-run).
"""
from __future__ import annotations

import json
import sys
from typing import Any, Dict, Optional

from .defaults import DEFAULT_VALUES


def load_config(file_path: str) -> Dict[str, Any]:
    with open(file_path, "r", encoding="utf-8") as fh:
        return json.load(fh)


def compose_config(config: Dict[str, Any]) -> Dict[str, Any]:
    out: Dict[str, Any] = {}
    for k, v in config.items():
        if k not in DEFAULT_VALUES or v != DEFAULT_VALUES[k]:
            out[k] = v
    return out


def save_config(config: Dict[str, Any], path: str = "config_out.json"):
    config_to_save = compose_config(config)
    with open(path, "w", encoding="utf-8") as fh:
        json.dump(config_to_save, fh, indent=4, default=str)
    return config, path


def save_debug_info(debug_info: Dict[str, Any], path: str = "debug_out.json") -> None:
    with open(path, "w", encoding="utf-8") as fh:
        json.dump(debug_info, fh, indent=4, default=str)


def _requests():
    try:
        import requests  # noqa: PLC0415
        return requests
    except ImportError:
        return None


def remote_save_config(config, url, username, password) -> bool:
    requests = _requests()
    if requests is None:
        print("requests not available; remote save skipped", file=sys.stderr)
        return False
    config_to_save = compose_config(config)
    try:
        response = requests.post(
            url,
            auth=(username, password),
            data={"json_config": json.dumps(config_to_save)},
        )
        response.raise_for_status()
        return True
    except requests.RequestException as exc:
        print(f"Failed to save remote configuration: {exc}", file=sys.stderr)
        return False


def remote_load_config(url, username: Optional[str] = None, password: Optional[str] = None):
    requests = _requests()
    if requests is None:
        print("requests not available; remote load skipped", file=sys.stderr)
        return None
    try:
        if username and password:
            response = requests.get(url, auth=(username, password))
        else:
            response = requests.get(url)
        response.raise_for_status()
        return response.json()
    except requests.RequestException as exc:
        print(f"Failed to load remote configuration: {exc}", file=sys.stderr)
        return None


def remote_log(config, debug_info, url, username, password) -> bool:
    requests = _requests()
    if requests is None:
        print("requests not available; remote log skipped", file=sys.stderr)
        return False
    config_to_save = compose_config(config)
    try:
        data = {
            "json_config": json.dumps(config_to_save),
            "json_result": json.dumps(debug_info, default=str),
        }
        response = requests.post(url, auth=(username, password), data=data)
        response.raise_for_status()
        return True
    except requests.RequestException as exc:
        print(f"Failed to log remote information: {exc}", file=sys.stderr)
        return False
