"""Sharded config bundle: split/reassemble integrity and watcher gating
(the standalone analogue of filterapi/config_bundle.go — the reference
splits large configs across k8s Secrets; here across part files)."""

import asyncio
import os

import pytest

from aigw.filterapi import bundle
import yaml as _yaml

from aigw.filterapi.config import ConfigError, load_config
from aigw.filterapi.watcher import ConfigWatcher

CFG = """\
uuid: first-uuid
routes:
  - name: r1
    backends:
      - name: b1
        schema: OpenAI
        upstream: {host: example.com, port: 443, tls: true}
"""


def _write(tmp_path, yaml_text, uuid, part_size=64):
    cfg = load_config(_yaml.safe_load(yaml_text))
    return bundle.write_bundle(str(tmp_path), cfg, uuid=uuid, part_size=part_size)


def test_roundtrip_multiple_parts(tmp_path):
    index = _write(tmp_path, CFG, "first-uuid")
    assert len(index.parts) > 1  # 64-byte parts force real sharding
    cfg = bundle.load_bundle(str(tmp_path))
    assert cfg.routes[0].name == "r1"
    assert cfg.routes[0].backends[0].upstream.host == "example.com"


def test_stale_parts_removed(tmp_path):
    _write(tmp_path, CFG, "u1", part_size=16)
    n_small = len(os.listdir(tmp_path / "parts"))
    _write(tmp_path, CFG, "u2", part_size=10_000)
    assert len(os.listdir(tmp_path / "parts")) == 1 < n_small
    assert bundle.load_bundle(str(tmp_path)).routes[0].name == "r1"


def test_checksum_mismatch_refused(tmp_path):
    _write(tmp_path, CFG, "u1")
    with open(tmp_path / "parts" / "000", "r+b") as f:
        f.write(b"#")  # corrupt one byte
    with pytest.raises(bundle.BundleChecksumError):
        bundle.load_bundle(str(tmp_path))


def test_part_path_traversal_refused(tmp_path):
    _write(tmp_path, CFG, "u1")
    idx_path = tmp_path / "index.yaml"
    txt = idx_path.read_text().replace("parts/000", "../outside")
    idx_path.write_text(txt)
    with pytest.raises(ConfigError):
        bundle.load_bundle(str(tmp_path))


def test_too_many_parts_refused(tmp_path):
    cfg = load_config(_yaml.safe_load(CFG))
    with pytest.raises(ConfigError):
        bundle.write_bundle(str(tmp_path), cfg, uuid="u", part_size=8)


def test_watcher_loads_and_reloads_bundle(tmp_path):
    _write(tmp_path, CFG, "first-uuid")
    seen = []
    w = ConfigWatcher(str(tmp_path), lambda rc: seen.append(rc), tick_s=0.01)
    rc = w.load_once()
    assert rc.config.routes[0].name == "r1"

    async def run():
        # same uuid -> no reload even though bytes could differ
        await w._check()
        assert not seen
        _write(tmp_path, CFG.replace("r1", "r2").replace("first-uuid", "second"), "second")
        await w._check()
        assert len(seen) == 1
        assert seen[0].config.routes[0].name == "r2"

    asyncio.run(run())


def test_watcher_keeps_config_on_torn_bundle(tmp_path):
    _write(tmp_path, CFG, "u1")
    w = ConfigWatcher(str(tmp_path), lambda rc: None, tick_s=0.01)
    w.load_once()
    # simulate a torn sync: new index (new uuid/checksum) but old parts
    idx = bundle.parse_index((tmp_path / "index.yaml").read_bytes())
    idx.uuid, idx.checksum = "new-uuid", "0" * 64
    (tmp_path / "index.yaml").write_text(bundle.dump_index(idx))

    async def run():
        with pytest.raises(bundle.BundleChecksumError):
            await w._check()

    asyncio.run(run())
