"""YAML/JSON config loading and saving (reference: murmura/config/loader.py:11-66)."""

from __future__ import annotations

import json
from pathlib import Path
from typing import Any, Dict, Union

import yaml

from murmura_amd.config.schema import Config


def load_config(path: Union[str, Path]) -> Config:
    """Load a Config from a .yaml/.yml or .json file (dispatch by extension)."""
    path = Path(path)
    if not path.exists():
        raise FileNotFoundError(f"config file not found: {path}")
    text = path.read_text()
    if path.suffix in (".yaml", ".yml"):
        raw: Dict[str, Any] = yaml.safe_load(text) or {}
    elif path.suffix == ".json":
        raw = json.loads(text)
    else:
        raise ValueError(f"unsupported config extension: {path.suffix} (use .yaml/.json)")
    return Config(**raw)


def save_config(config: Config, path: Union[str, Path]) -> None:
    """Serialize a Config back to YAML or JSON."""
    path = Path(path)
    data = config.model_dump(exclude_none=True)
    if path.suffix in (".yaml", ".yml"):
        path.write_text(yaml.safe_dump(data, sort_keys=False))
    elif path.suffix == ".json":
        path.write_text(json.dumps(data, indent=2))
    else:
        raise ValueError(f"unsupported config extension: {path.suffix} (use .yaml/.json)")
