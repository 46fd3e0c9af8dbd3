from .defaults import DEFAULT_VALUES
from .merger import merge_config, process_unknown_args, convert_type
from .handler import (
    load_config,
    save_config,
    compose_config,
    save_debug_info,
    remote_save_config,
    remote_load_config,
    remote_log,
)

__all__ = [
    "DEFAULT_VALUES",
    "merge_config",
    "process_unknown_args",
    "convert_type",
    "load_config",
    "save_config",
    "compose_config",
    "save_debug_info",
    "remote_save_config",
    "remote_load_config",
    "remote_log",
]
