// llmlb-gw: MI355X-native LLM gateway data plane.
//   llmlb-gw [--port N] [--host H] [--db PATH] [--workers N]
// Env knobs mirror the reference's config surface (config.rs):
//   LLMLB_HOST, LLMLB_PORT, LLMLB_DATABASE_URL, LLMLB_LOG_LEVEL,
//   LLMLB_QUEUE_MAX, LLMLB_HEALTH_CHECK_INTERVAL, LLMLB_ADMIN_TOKEN,
//   LLMLB_API_KEY, LLMLB_AUTO_SYNC_MODELS_INTERVAL_SECS,
//   LLMLB_{OPENAI,GOOGLE,ANTHROPIC}_BASE_URL, *_API_KEY,
//   LLMLB_CONTROL_PLANE_URL, LLMLB_HBM_ADMISSION_FRACTION
#include <csignal>

#include "gateway.h"

static gw::Gateway* g_gw = nullptr;

static void on_signal(int) {
  if (g_gw) g_gw->stop();
}

static std::string envs(const char* k, const std::string& dflt) {
  const char* v = getenv(k);
  return v && *v ? v : dflt;
}
static int envi(const char* k, int dflt) {
  const char* v = getenv(k);
  return v && *v ? atoi(v) : dflt;
}
static double envf(const char* k, double dflt) {
  const char* v = getenv(k);
  return v && *v ? atof(v) : dflt;
}

int main(int argc, char** argv) {
  gw::GatewayConfig cfg;
  cfg.host = envs("LLMLB_HOST", "0.0.0.0");
  cfg.port = envi("LLMLB_PORT", 32768);
  cfg.db_path = envs("LLMLB_DATABASE_URL", "llmlb.sqlite3");
  cfg.log_level = envs("LLMLB_LOG_LEVEL", "info");
  cfg.log_dir = envs("LLMLB_LOG_DIR", "");
  cfg.admin_token = envs("LLMLB_ADMIN_TOKEN", "");
  cfg.workers = envi("LLMLB_WORKERS", 0);
  cfg.health_interval_s = envi("LLMLB_HEALTH_CHECK_INTERVAL", 30);
  cfg.auto_sync_interval_s =
      envi("LLMLB_AUTO_SYNC_MODELS_INTERVAL_SECS", 900);
  cfg.admission.max_concurrent_per_endpoint = envi("LLMLB_QUEUE_MAX", 100);
  cfg.admission.hbm_admission_fraction =
      envf("LLMLB_HBM_ADMISSION_FRACTION", 0.92);
  cfg.retention_days = envi("LLMLB_REQUEST_HISTORY_RETENTION_DAYS", 7);
  cfg.audit_archive_days = envi("LLMLB_AUDIT_RETENTION_DAYS", 90);
  cfg.log_retention_days = envi("LLMLB_LOG_RETENTION_DAYS", 7);
  cfg.admission.round_robin =
      envs("LLMLB_LOAD_BALANCER_MODE", "tps") == std::string("round_robin");
  cfg.tps_ema_alpha = envf("LLMLB_TPS_EMA_ALPHA", 0.2);
  cfg.amdsmi_poll_interval_s = envf("LLMLB_AMDSMI_POLL_INTERVAL_SECS", 2.0);
  cfg.capture_bodies = envi("LLMLB_CAPTURE_BODIES", 0) != 0;
  cfg.openai_base = envs("LLMLB_OPENAI_BASE_URL", "");
  cfg.google_base = envs("LLMLB_GOOGLE_BASE_URL", "");
  cfg.anthropic_base = envs("LLMLB_ANTHROPIC_BASE_URL", "");
  cfg.openai_key = envs("OPENAI_API_KEY", "");
  cfg.google_key = envs("GOOGLE_API_KEY", "");
  cfg.anthropic_key = envs("ANTHROPIC_API_KEY", "");
  cfg.control_plane_base = envs("LLMLB_CONTROL_PLANE_URL", "");

  for (int i = 1; i < argc; ++i) {
    std::string a = argv[i];
    auto next = [&]() -> const char* {
      return i + 1 < argc ? argv[++i] : "";
    };
    if (a == "--port") cfg.port = atoi(next());
    else if (a == "--host") cfg.host = next();
    else if (a == "--db") cfg.db_path = next();
    else if (a == "--workers") cfg.workers = atoi(next());
    else if (a == "--log-level") cfg.log_level = next();
    else if (a == "--help") {
      printf("llmlb-gw [--host H] [--port N] [--db PATH] [--workers N] "
             "[--log-level L]\n");
      return 0;
    }
  }

  // heap-allocate (never freed: process-lifetime object).  On the
  // stack, TSAN conflated the HistoryWriter mutex address with prior
  // stack frames and cascaded false lock reports; heap placement also
  // matches the object's actual lifetime.
  auto* gateway_ptr = new gw::Gateway(cfg);
  gw::Gateway& gateway = *gateway_ptr;
  g_gw = &gateway;
  std::string err;
  if (!gateway.init(&err)) {
    fprintf(stderr, "init failed: %s\n", err.c_str());
    return 1;
  }
  signal(SIGINT, on_signal);
  signal(SIGTERM, on_signal);
  signal(SIGPIPE, SIG_IGN);
  gateway.run();
  return 0;
}
