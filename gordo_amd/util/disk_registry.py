"""
File-per-key registry used as the model-build cache index.

Spec: gordo/util/disk_registry.py — key = cache key (sha3-512 hex),
value = model dir path, one file per key under the registry dir.
Unlike the reference (which documents concurrent same-key writes as
unsafe), writes here are atomic (tmp + rename) so concurrent per-GPU
build workers can share a registry.
"""
from __future__ import annotations

import logging
import os
import re
import tempfile
from typing import AnyStr, Optional, Union

logger = logging.getLogger(__name__)

_SAFE_KEY = re.compile(r"[^A-Za-z0-9_.\-]")


def _key_path(registry_dir: Union[os.PathLike, str], key: str) -> str:
    safe = _SAFE_KEY.sub("_", key)
    return os.path.join(os.fspath(registry_dir), safe)


def write_key(registry_dir: Union[os.PathLike, str], key: str, val: AnyStr):
    """Write ``val`` under ``key`` in the registry (atomic)."""
    os.makedirs(os.fspath(registry_dir), exist_ok=True)
    path = _key_path(registry_dir, key)
    data = val if isinstance(val, str) else val.decode()
    fd, tmp = tempfile.mkstemp(dir=os.fspath(registry_dir))
    try:
        with os.fdopen(fd, "w") as f:
            f.write(data)
        os.replace(tmp, path)
    except BaseException:
        if os.path.exists(tmp):
            os.unlink(tmp)
        raise
    logger.debug("Registry write %s -> %s", key, data)


def get_value(registry_dir: Union[os.PathLike, str], key: str) -> Optional[str]:
    """Return the value stored under ``key``, or None."""
    path = _key_path(registry_dir, key)
    if not os.path.isfile(path):
        return None
    with open(path) as f:
        return f.read()


def delete_value(registry_dir: Union[os.PathLike, str], key: str) -> bool:
    """Delete ``key``; True when it existed."""
    path = _key_path(registry_dir, key)
    if os.path.isfile(path):
        os.unlink(path)
        return True
    return False
