"""Sharded config bundles: split a large serialized Config into byte
chunks with an index file, and reassemble with integrity checking.

The reference stores its filter config in Kubernetes Secrets, which cap
object size at ~1 MiB, so large configs (many routes/backends) are split
across several Secrets described by an ``index.yaml``
(internal/filterapi/config_bundle.go, internal/controller/
filter_config_bundle.go). The standalone deployment has no Secret size
limit, but the same mechanism is useful for atomic multi-file config
delivery (e.g. configmap-style mounts, rsync'd part files): the index is
written last, carries a sha256 over the reassembled payload and a UUID,
and the watcher only swaps config when the UUID changes AND the checksum
verifies — a torn partial write of the parts can never be loaded.

Layout under a bundle directory::

    index.yaml          # version/uuid/checksum/parts list
    parts/000           # raw byte chunks of the YAML payload
    parts/001
    ...
"""

from __future__ import annotations

import hashlib
import os
import tempfile
from dataclasses import dataclass, field

import yaml

from aigw.filterapi.config import Config, ConfigError, dump_config_yaml, load_config

INDEX_FILE = "index.yaml"
DEFAULT_PART_SIZE = 700 * 1024
MAX_PARTS = 8


class BundleChecksumError(ConfigError):
    pass


@dataclass
class BundlePart:
    name: str
    path: str
    size_bytes: int = 0


@dataclass
class BundleIndex:
    version: str
    uuid: str
    checksum: str
    parts: list[BundlePart] = field(default_factory=list)


def _checksum(payload: bytes) -> str:
    return hashlib.sha256(payload).hexdigest()


def part_rel_path(i: int) -> str:
    return f"parts/{i:03d}"


def write_bundle(
    directory: str,
    config: Config,
    *,
    uuid: str,
    part_size: int = DEFAULT_PART_SIZE,
    version: str = "dev",
) -> BundleIndex:
    """Serialize ``config`` and write it as a sharded bundle. Parts are
    written before the index (readers gate on the index's checksum, so a
    reader racing the writer sees either the old complete bundle or the
    new one, never a torn mix)."""
    payload = dump_config_yaml(config).encode("utf-8")
    chunks = [payload[i : i + part_size] for i in range(0, len(payload), part_size)] or [b""]
    if len(chunks) > MAX_PARTS:
        raise ConfigError(
            f"config requires {len(chunks)} bundle parts, exceeds max {MAX_PARTS}"
        )
    os.makedirs(os.path.join(directory, "parts"), exist_ok=True)
    parts = []
    for i, chunk in enumerate(chunks):
        rel = part_rel_path(i)
        _atomic_write(os.path.join(directory, rel), chunk)
        parts.append(BundlePart(name=f"part-{i}", path=rel, size_bytes=len(chunk)))
    # remove stale higher-numbered parts from a previous, larger bundle
    for i in range(len(chunks), MAX_PARTS):
        try:
            os.unlink(os.path.join(directory, part_rel_path(i)))
        except FileNotFoundError:
            pass
    index = BundleIndex(version=version, uuid=uuid, checksum=_checksum(payload), parts=parts)
    _atomic_write(os.path.join(directory, INDEX_FILE), dump_index(index).encode())
    return index


def dump_index(index: BundleIndex) -> str:
    return yaml.safe_dump(
        {
            "version": index.version,
            "uuid": index.uuid,
            "checksum": index.checksum,
            "parts": [
                {"name": p.name, "path": p.path, "sizeBytes": p.size_bytes}
                for p in index.parts
            ],
        },
        sort_keys=False,
    )


def parse_index(raw: bytes) -> BundleIndex:
    doc = yaml.safe_load(raw)
    if not isinstance(doc, dict):
        raise ConfigError("bundle index is not a mapping")
    if not doc.get("checksum"):
        raise ConfigError("bundle index: empty checksum")
    parts = [
        BundlePart(
            name=p.get("name", ""),
            path=p["path"],
            size_bytes=int(p.get("sizeBytes", 0)),
        )
        for p in doc.get("parts") or []
    ]
    if not parts:
        raise ConfigError("bundle index: empty parts")
    return BundleIndex(
        version=str(doc.get("version", "")),
        uuid=str(doc.get("uuid", "")),
        checksum=str(doc["checksum"]).lower(),
        parts=parts,
    )


def read_bundle_payload(directory: str) -> tuple[bytes, BundleIndex]:
    with open(os.path.join(directory, INDEX_FILE), "rb") as f:
        index = parse_index(f.read())
    payload = bytearray()
    for p in index.parts:
        rel = os.path.normpath(p.path)
        if rel.startswith("..") or os.path.isabs(rel):
            raise ConfigError(f"bundle part escapes bundle dir: {p.path!r}")
        with open(os.path.join(directory, rel), "rb") as f:
            payload += f.read()
    if _checksum(bytes(payload)) != index.checksum:
        raise BundleChecksumError(
            f"bundle checksum mismatch (uuid {index.uuid}): parts do not match index"
        )
    return bytes(payload), index


def load_bundle(directory: str) -> Config:
    payload, _ = read_bundle_payload(directory)
    return load_config(yaml.safe_load(payload))


def is_bundle_dir(path: str) -> bool:
    return os.path.isdir(path) and os.path.exists(os.path.join(path, INDEX_FILE))


def _atomic_write(path: str, data: bytes) -> None:
    d = os.path.dirname(path) or "."
    fd, tmp = tempfile.mkstemp(dir=d, prefix=".bundle-")
    try:
        with os.fdopen(fd, "wb") as f:
            f.write(data)
        os.replace(tmp, path)
    except BaseException:
        try:
            os.unlink(tmp)
        except OSError:
            pass
        raise
