"""Foundation-layer tests: config, hashing, compression, security, errors."""
from __future__ import annotations

import dataclasses
import pathlib

import pytest

from infomesh_amd import compression
from infomesh_amd.config import Config, load_config, save_config, parse_toml
from infomesh_amd.errors import InfoMeshError, format_error
from infomesh_amd.hashing import content_hash, shard_of, short_hash, hash64
from infomesh_amd.security import is_url_safe, validate_url


def test_content_hash_stable():
    assert content_hash("abc") == content_hash(b"abc")
    assert len(content_hash("abc")) == 64
    assert short_hash("abc", 8) == content_hash("abc")[:8]


def test_shard_of_distribution():
    counts = [0] * 8
    for i in range(8000):
        counts[shard_of(f"https://example.com/{i}", 8)] += 1
    assert min(counts) > 700  # roughly uniform
    assert shard_of("x", 1) == 0


def test_hash64_range():
    assert 0 <= hash64("token") < 2**64


def test_zstd_roundtrip():
    data = b"The quick brown fox. " * 500
    for level in (3, 12, 19):
        z = compression.compress(data, level)
        assert compression.decompress(z) == data
        assert len(z) < len(data)


def test_zstd_bomb_guard():
    big = b"\x00" * 1_000_000
    z = compression.compress(big)
    with pytest.raises(ValueError):
        compression.Compressor(max_decompressed=1000).decompress(z)


def test_config_defaults():
    cfg = Config()
    assert cfg.crawl.max_concurrent == 5
    assert cfg.search.rrf_k == 60
    assert cfg.gpu.dtype == "bf16"


def test_config_env_override_and_clamp():
    cfg = load_config(pathlib.Path("/nonexistent"), env={
        "INFOMESH_SEARCH_MAX_RESULTS": "5000",      # clamps to 100
        "INFOMESH_GPU_DTYPE": "fp16",
        "INFOMESH_NODE_ROLE": "bogus",              # rejected, keeps default
    })
    assert cfg.search.max_results == 100
    assert cfg.gpu.dtype == "fp16"
    assert cfg.node.role == "full"


def test_config_toml_roundtrip(tmp_path):
    cfg = Config()
    cfg = dataclasses.replace(
        cfg, crawl=dataclasses.replace(cfg.crawl, max_concurrent=9),
        node=dataclasses.replace(cfg.node, name="testnode"))
    p = tmp_path / "config.toml"
    save_config(cfg, p)
    text = p.read_text()
    assert "max_concurrent = 9" in text
    assert "politeness_delay_s" not in text  # only non-defaults written
    loaded = load_config(p, env={})
    assert loaded.crawl.max_concurrent == 9
    assert loaded.node.name == "testnode"


def test_parse_toml_types():
    d = parse_toml('[a]\nx = 1\ny = 1.5\nz = "s"\nw = true\n# comment\n')
    assert d["a"] == {"x": 1, "y": 1.5, "z": "s", "w": True}


def test_ssrf_guard():
    assert is_url_safe("https://example.com")
    for bad in ("http://localhost/x", "http://127.0.0.1/", "http://10.0.0.1/",
                "http://192.168.1.1/", "http://169.254.169.254/latest",
                "file:///etc/passwd", "http://[::1]/", "http://user@evil.com/",
                "gopher://x.com"):
        assert not is_url_safe(bad), bad


def test_validate_url_raises_with_code():
    with pytest.raises(InfoMeshError) as ei:
        validate_url("http://127.0.0.1/")
    assert ei.value.code == "CRWL001"
    assert "CRWL001" in format_error(ei.value)


def test_format_error_unregistered():
    assert "unregistered" in format_error(ValueError("x"))
