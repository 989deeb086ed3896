"""Config tree tests — mirror reference pkg/config/config.go Default/Validate,
plus the file-loading capability the reference declares but never wires."""

import json

import pytest

from ggrmcp_amd.config import Config


def test_defaults_match_reference():
    cfg = Config.default()
    assert cfg.server.http_port == 50053
    assert cfg.grpc.port == 50051
    assert cfg.grpc.max_send_msg_bytes == 4 * 1024 * 1024
    assert cfg.server.max_body_bytes == 1024 * 1024
    assert cfg.server.max_response_bytes == 16 * 1024 * 1024
    assert cfg.session.max_sessions == 10_000
    assert cfg.session.ttl_s == 1800.0
    assert cfg.server.rate_limit_rps == 100.0
    assert "authorization" in cfg.header_forwarding.allowed_headers
    assert "cookie" in cfg.header_forwarding.blocked_headers
    cfg.validate()


def test_development_profile():
    cfg = Config.development()
    assert cfg.logging.level == "debug"
    assert cfg.logging.development
    assert not cfg.server.rate_limit_enabled


def test_validate_rejects_bad_values():
    cfg = Config.default()
    cfg.grpc.port = 0
    with pytest.raises(ValueError):
        cfg.validate()
    cfg = Config.default()
    cfg.descriptor_set.enabled = True
    with pytest.raises(ValueError):
        cfg.validate()
    cfg = Config.default()
    cfg.gpu.devices = 9
    with pytest.raises(ValueError):
        cfg.validate()
    cfg = Config.default()
    cfg.grpc.native_connections = 0
    with pytest.raises(ValueError):
        cfg.validate()


def test_from_dict_and_unknown_key():
    cfg = Config.from_dict({"grpc": {"host": "h", "port": 1234}, "gpu": {"devices": 4}})
    assert cfg.grpc.host == "h"
    assert cfg.grpc.port == 1234
    assert cfg.gpu.devices == 4
    with pytest.raises(ValueError):
        Config.from_dict({"nope": 1})


def test_from_json_file(tmp_path):
    p = tmp_path / "cfg.json"
    p.write_text(json.dumps({"server": {"http_port": 8081}}))
    cfg = Config.from_file(str(p))
    assert cfg.server.http_port == 8081


def test_from_yaml_file(tmp_path):
    p = tmp_path / "cfg.yaml"
    p.write_text("grpc:\n  host: backend\nextra_backends:\n  - host: b2\n    port: 50052\n")
    cfg = Config.from_file(str(p))
    assert cfg.grpc.host == "backend"
    assert len(cfg.extra_backends) == 1
    assert cfg.extra_backends[0].port == 50052
    assert len(cfg.all_backends()) == 2
