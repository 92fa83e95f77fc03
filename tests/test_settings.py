"""Settings schema: precedence, override grammar, client contract.

Covers the behaviors surveyed from the reference config layer
(reference settings.py precedence :7-11, grammar :17-32,
payload/sanitizer :1648/:1698)."""

import pytest

from selkies_amd.settings import load_settings


def test_defaults():
    s = load_settings(argv=[], env={})
    assert s.port == 8080
    assert s.mode == "websockets"
    assert s.encoder == "h264enc-striped"
    assert s.resolution_wh == (1920, 1080)


def test_precedence_cli_over_env():
    s = load_settings(argv=["--port", "9000"], env={"SELKIES_PORT": "7000"})
    assert s.port == 9000


def test_env_over_default_and_fallback_env():
    s = load_settings(argv=[], env={"SELKIES_PORT": "7000"})
    assert s.port == 7000
    s = load_settings(argv=[], env={"DISPLAY": ":42"})
    assert s.display == ":42"
    s = load_settings(argv=[], env={"DISPLAY": ":42", "SELKIES_DISPLAY": ":1"})
    assert s.display == ":1"


def test_bool_locked_grammar():
    s = load_settings(argv=[], env={"SELKIES_ENABLE_AUDIO": "false|locked"})
    assert s.enable_audio is False
    assert s.is_locked("enable_audio")
    with pytest.raises(PermissionError):
        s.set("enable_audio", True)


def test_enum_narrowing_grammar():
    s = load_settings(argv=[], env={"SELKIES_ENCODER": "jpeg,h264enc"})
    assert s.encoder == "jpeg"
    payload = s.build_client_settings_payload()
    assert payload["encoder"]["allowed"] == ["jpeg", "h264enc"]
    with pytest.raises(ValueError):
        s.set("encoder", "h264enc-striped")


def test_single_enum_locks():
    s = load_settings(argv=[], env={"SELKIES_ENCODER": "jpeg"})
    # plain single value (no comma) sets without locking
    assert s.encoder == "jpeg"
    assert not s.is_locked("encoder")


def test_range_grammar():
    s = load_settings(argv=[], env={"SELKIES_FRAMERATE": "30,15-60"})
    assert s.framerate == 30
    assert s.set("framerate", 90) == 60          # clamped to narrowed range
    s2 = load_settings(argv=[], env={"SELKIES_FRAMERATE": "30,30-30"})
    assert s2.is_locked("framerate")


def test_range_clamp_on_set():
    s = load_settings(argv=[], env={})
    assert s.set("video_crf", 99) == 51
    assert s.set("video_crf", -3) == 0


def test_client_payload_and_sanitize():
    s = load_settings(argv=[], env={})
    payload = s.build_client_settings_payload()
    assert "framerate" in payload and "encoder" in payload
    assert "port" not in payload                      # not client-exposed
    assert s.sanitize_client_setting("framerate", "30") == 30
    assert s.sanitize_client_setting("use_cpu", "true") is True
    with pytest.raises(KeyError):
        s.sanitize_client_setting("basic_auth_password", "x")


def test_unknown_cli_args_ignored():
    s = load_settings(argv=["--port", "8081", "--not-a-flag", "1"], env={})
    assert s.port == 8081
