"""Environment-variable configuration for llmlb_amd.

Mirrors the knob surface of the reference's env-only config system
(/root/reference/llmlb/src/config.rs:28-127 — key names, defaults and
deprecated-name fallbacks), adapted for the MI355X build.  No config
files: env vars only, same as the reference.
"""
from __future__ import annotations

import os
from dataclasses import dataclass, field


def _env(name: str, default: str, *deprecated: str) -> str:
    """Read LLMLB_<name>, falling back to deprecated aliases then default."""
    for key in (name, *deprecated):
        val = os.environ.get(key)
        if val is not None:
            return val
    return default


def _env_int(name: str, default: int, *deprecated: str) -> int:
    raw = _env(name, str(default), *deprecated)
    try:
        return int(raw)
    except ValueError:
        return default


def _env_float(name: str, default: float, *deprecated: str) -> float:
    raw = _env(name, str(default), *deprecated)
    try:
        return float(raw)
    except ValueError:
        return default


@dataclass
class Config:
    # -- server ----------------------------------------------------------
    host: str = field(default_factory=lambda: _env("LLMLB_HOST", "0.0.0.0"))
    port: int = field(default_factory=lambda: _env_int("LLMLB_PORT", 32768))
    database_url: str = field(
        default_factory=lambda: _env("LLMLB_DATABASE_URL", "llmlb.sqlite3")
    )
    log_level: str = field(default_factory=lambda: _env("LLMLB_LOG_LEVEL", "info"))
    log_dir: str = field(default_factory=lambda: _env("LLMLB_LOG_DIR", "logs"))
    jwt_secret: str = field(default_factory=lambda: _env("LLMLB_JWT_SECRET", ""))
    admin_user: str = field(default_factory=lambda: _env("LLMLB_ADMIN_USER", "admin"))
    admin_password: str = field(
        default_factory=lambda: _env("LLMLB_ADMIN_PASSWORD", "")
    )
    body_limit_bytes: int = field(
        default_factory=lambda: _env_int("LLMLB_BODY_LIMIT_BYTES", 20 * 1024 * 1024)
    )

    # -- routing / queueing (reference: config.rs:87-99) ------------------
    queue_max: int = field(default_factory=lambda: _env_int("LLMLB_QUEUE_MAX", 100))
    queue_timeout_secs: int = field(
        default_factory=lambda: _env_int("LLMLB_QUEUE_TIMEOUT_SECS", 60)
    )
    load_balancer_mode: str = field(
        default_factory=lambda: _env("LLMLB_LOAD_BALANCER_MODE", "auto")
    )
    tps_ema_alpha: float = field(
        default_factory=lambda: _env_float("LLMLB_TPS_EMA_ALPHA", 0.2)
    )

    # -- health checking (reference: bootstrap.rs:106, endpoint_checker.rs) --
    health_check_interval_secs: int = field(
        default_factory=lambda: _env_int("LLMLB_HEALTH_CHECK_INTERVAL", 30)
    )
    health_check_timeout_secs: int = field(
        default_factory=lambda: _env_int("LLMLB_HEALTH_CHECK_TIMEOUT_SECS", 5)
    )
    consecutive_failures_for_offline: int = field(
        default_factory=lambda: _env_int("LLMLB_FAILURES_FOR_OFFLINE", 2)
    )
    auto_sync_models_interval_secs: int = field(
        default_factory=lambda: _env_int("LLMLB_AUTO_SYNC_MODELS_INTERVAL_SECS", 900)
    )

    # -- retention (reference: db/request_history.rs:21-24, logging.rs:22) --
    request_history_retention_days: int = field(
        default_factory=lambda: _env_int("LLMLB_REQUEST_HISTORY_RETENTION_DAYS", 7)
    )
    log_retention_days: int = field(
        default_factory=lambda: _env_int("LLMLB_LOG_RETENTION_DAYS", 7)
    )
    audit_archive_days: int = field(
        default_factory=lambda: _env_int("LLMLB_AUDIT_ARCHIVE_DAYS", 90)
    )

    # -- cloud providers --------------------------------------------------
    openai_api_key: str = field(default_factory=lambda: _env("OPENAI_API_KEY", ""))
    google_api_key: str = field(default_factory=lambda: _env("GOOGLE_API_KEY", ""))
    anthropic_api_key: str = field(
        default_factory=lambda: _env("ANTHROPIC_API_KEY", "")
    )

    # -- MI355X-native knobs (new; no reference counterpart) --------------
    # amdsmi poll period for per-GPU HBM occupancy / busy%.
    amdsmi_poll_interval_secs: float = field(
        default_factory=lambda: _env_float("LLMLB_AMDSMI_POLL_INTERVAL_SECS", 2.0)
    )
    # Admission control sized for 288 GB HBM3E per GPU: refuse placement
    # when projected HBM use exceeds this fraction.
    hbm_admission_fraction: float = field(
        default_factory=lambda: _env_float("LLMLB_HBM_ADMISSION_FRACTION", 0.92)
    )
    # Busy-fraction thresholds for delay/reject admission control
    # (reference: balancer/mod.rs:2255-2270 — 50% delay, 80% reject).
    admission_delay_fraction: float = field(
        default_factory=lambda: _env_float("LLMLB_ADMISSION_DELAY_FRACTION", 0.5)
    )
    admission_reject_fraction: float = field(
        default_factory=lambda: _env_float("LLMLB_ADMISSION_REJECT_FRACTION", 0.8)
    )


_config: Config | None = None


def get_config(refresh: bool = False) -> Config:
    global _config
    if _config is None or refresh:
        _config = Config()
    return _config
