"""
dump/load — the on-disk model layout.

Layout spec (gordo/serializer/serializer.py:124-196) — a hard
compatibility requirement:

    <dir>/model.pkl        pickled model
    <dir>/metadata.json    full Machine dict incl. build metadata
    <dir>/info.json        {"checksum": ...}

``load_metadata``/``load_info`` search the dir and its parent.
``dumps``/``loads`` are raw pickle bytes.
"""
from __future__ import annotations

import hashlib
import json
import logging
import os
import pickle
from typing import Any, Dict, Optional

logger = logging.getLogger(__name__)

__all__ = [
    "dump",
    "load",
    "dumps",
    "loads",
    "load_metadata",
    "load_info",
    "metadata_path",
]


def dumps(model) -> bytes:
    """Serialize a model to pickle bytes.

    >>> from sklearn.preprocessing import MinMaxScaler
    >>> loads(dumps(MinMaxScaler())).__class__.__name__
    'MinMaxScaler'
    """
    return pickle.dumps(model)


def loads(bytes_object: bytes):
    return pickle.loads(bytes_object)


def dump(
    obj,
    dest_dir: str,
    metadata: Optional[Dict[str, Any]] = None,
    info: Optional[Dict[str, Any]] = None,
):
    """Serialize ``obj`` into ``dest_dir/model.pkl`` (+ metadata.json /
    info.json when given)."""
    os.makedirs(dest_dir, exist_ok=True)
    model_path = os.path.join(dest_dir, "model.pkl")
    with open(model_path, "wb") as f:
        pickle.dump(obj, f)
    if info is not None:
        if "checksum" not in info:
            info = dict(info)
            info["checksum"] = _file_checksum(model_path)
        with open(os.path.join(dest_dir, "info.json"), "w") as f:
            json.dump(info, f, default=str)
    if metadata is not None:
        with open(os.path.join(dest_dir, "metadata.json"), "w") as f:
            json.dump(metadata, f, default=str)


def load(source_dir: str):
    """Load the model pickled at ``source_dir/model.pkl``."""
    model_path = os.path.join(source_dir, "model.pkl")
    if not os.path.exists(model_path):
        raise FileNotFoundError(f"No model.pkl under {source_dir}")
    with open(model_path, "rb") as f:
        return pickle.load(f)


def _json_file_path(source_dir: str, name: str) -> Optional[str]:
    # search dir then parent (reference serializer.py:77-84)
    for d in (source_dir, os.path.dirname(os.path.normpath(source_dir))):
        candidate = os.path.join(d, name)
        if os.path.isfile(candidate):
            return candidate
    return None


def _load_json(source_dir: str, name: str) -> Optional[Dict[str, Any]]:
    path = _json_file_path(source_dir, name)
    if path is None:
        return None
    with open(path) as f:
        return json.load(f)


def load_metadata(source_dir: str) -> Dict[str, Any]:
    meta = _load_json(source_dir, "metadata.json")
    if meta is None:
        raise FileNotFoundError(f"No metadata.json under or beside {source_dir}")
    return meta


def load_info(source_dir: str) -> Optional[Dict[str, Any]]:
    return _load_json(source_dir, "info.json")


def metadata_path(source_dir: str) -> Optional[str]:
    return _json_file_path(source_dir, "metadata.json")


def _file_checksum(path: str) -> str:
    h = hashlib.md5()
    with open(path, "rb") as f:
        for chunk in iter(lambda: f.read(1 << 20), b""):
            h.update(chunk)
    return h.hexdigest()
