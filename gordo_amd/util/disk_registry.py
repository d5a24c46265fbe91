"""Tiny key→value file registry.

Parity: gordo_components/util/disk_registry.py — the builder registers each
built model dir under md5(config); a re-run with the same config hash reuses
the artifact instead of retraining (the fleet-level recovery unit).
"""
from __future__ import annotations

import logging
import os
import re
from typing import AnyStr, Optional, Union

logger = logging.getLogger(__name__)

_SAFE = re.compile(r"[^A-Za-z0-9_\-\.]")


def _key_path(registry_dir: Union[os.PathLike, str], key: str) -> str:
    safe = _SAFE.sub("_", key)
    return os.path.join(os.fspath(registry_dir), safe + ".md5")


def write_key(registry_dir: Union[os.PathLike, str], key: str, val: AnyStr):
    os.makedirs(os.fspath(registry_dir), exist_ok=True)
    path = _key_path(registry_dir, key)
    logger.debug("Registry write %s -> %r", path, val)
    with open(path, "w") as f:
        f.write(str(val))


def get_value(registry_dir: Union[os.PathLike, str], key: str) -> Optional[str]:
    path = _key_path(registry_dir, key)
    if not os.path.exists(path):
        return None
    with open(path) as f:
        return f.read()


def delete_value(registry_dir: Union[os.PathLike, str], key: str) -> bool:
    path = _key_path(registry_dir, key)
    if os.path.exists(path):
        os.remove(path)
        return True
    return False
