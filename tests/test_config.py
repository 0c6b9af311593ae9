"""Env-config schema (llmlb_amd.config): defaults, overrides and
deprecated-name fallbacks mirror the reference's env-only config."""
import os

from llmlb_amd.config import Config, get_config


def test_defaults():
    for k in ("LLMLB_PORT", "LLMLB_QUEUE_MAX"):
        os.environ.pop(k, None)
    c = Config()
    assert c.port == 32768
    assert c.queue_max == 100
    assert c.tps_ema_alpha == 0.2
    assert c.hbm_admission_fraction == 0.92


def test_override_and_refresh(monkeypatch):
    monkeypatch.setenv("LLMLB_QUEUE_MAX", "7")
    monkeypatch.setenv("LLMLB_TPS_EMA_ALPHA", "0.5")
    c = get_config(refresh=True)
    assert c.queue_max == 7
    assert c.tps_ema_alpha == 0.5
    monkeypatch.delenv("LLMLB_QUEUE_MAX")
    assert get_config(refresh=True).queue_max == 100


def test_bad_values_fall_back(monkeypatch):
    monkeypatch.setenv("LLMLB_QUEUE_MAX", "not-a-number")
    assert get_config(refresh=True).queue_max == 100
