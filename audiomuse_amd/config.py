"""Central configuration.

Mirrors the reference's single-module config surface
(/root/reference/config.py: env-read constants with baked defaults,
`refresh_config()` re-read, DB overrides layered from the app_config
table). Names intentionally match the reference vocabulary so that a
user of the reference finds the same knobs.

Env is read at import; `refresh_config()` re-applies env and any DB
overrides registered via `set_db_override_provider()`.
"""

from __future__ import annotations

import os
from typing import Callable, Dict, Optional

_DB_OVERRIDE_PROVIDER: Optional[Callable[[], Dict[str, str]]] = None


def _env(name: str, default: str) -> str:
    return os.environ.get(name, default)


def _env_int(name: str, default: int) -> int:
    try:
        return int(os.environ.get(name, default))
    except (TypeError, ValueError):
        return default


def _env_float(name: str, default: float) -> float:
    try:
        return float(os.environ.get(name, default))
    except (TypeError, ValueError):
        return default


def _env_bool(name: str, default: bool) -> bool:
    v = os.environ.get(name)
    if v is None:
        return default
    return v.strip().lower() in ("1", "true", "yes", "on")


# --------------------------------------------------------------------------
# Storage / control plane
# --------------------------------------------------------------------------
# The reference uses PostgreSQL as both store and queue (database.py,
# taskqueue/sql.py). DATABASE_URL switches backends: "postgresql://..."
# runs the deployment contract (first-party wire driver, SKIP LOCKED
# claims, advisory-lock liveness, LISTEN/NOTIFY — docs/POSTGRES.md);
# "sqlite:///path" is the zero-dependency single-box mode with the same
# schema and observable queue semantics.
DATABASE_URL = _env("DATABASE_URL", "sqlite:///" + os.path.join(
    os.environ.get("AUDIOMUSE_DATA_DIR", os.path.expanduser("~/.audiomuse-amd")),
    "audiomuse.db"))
DATA_DIR = _env("AUDIOMUSE_DATA_DIR", os.path.expanduser("~/.audiomuse-amd"))

# Queue semantics (reference: taskqueue/sql.py, maintenance.py)
QUEUE_MAX_ATTEMPTS = _env_int("QUEUE_MAX_ATTEMPTS", 3)
QUEUE_LEASE_SECONDS = _env_float("QUEUE_LEASE_SECONDS", 30.0)
QUEUE_HEARTBEAT_SECONDS = _env_float("QUEUE_HEARTBEAT_SECONDS", 5.0)
QUEUE_POLL_SECONDS = _env_float("QUEUE_POLL_SECONDS", 0.25)
MAX_QUEUED_ANALYSIS_JOBS = _env_int("MAX_QUEUED_ANALYSIS_JOBS", 30)
REBUILD_INDEX_BATCH_SIZE = _env_int("REBUILD_INDEX_BATCH_SIZE", 500)
WORKER_MAX_JOBS = _env_int("WORKER_MAX_JOBS", 50)

# Media-server HTTP behavior (reference: per-provider request helpers,
# navidrome.py:161-220 / jellyfin.py:310-359; centralised here in
# mediaserver/http.py)
MEDIASERVER_TIMEOUT_SECONDS = _env_float("MEDIASERVER_TIMEOUT_SECONDS", 30.0)
MEDIASERVER_DOWNLOAD_TIMEOUT_SECONDS = _env_float(
    "MEDIASERVER_DOWNLOAD_TIMEOUT_SECONDS", 300.0)
MEDIASERVER_RETRIES = _env_int("MEDIASERVER_RETRIES", 3)
MEDIASERVER_RETRY_BACKOFF_SECONDS = _env_float(
    "MEDIASERVER_RETRY_BACKOFF_SECONDS", 1.0)
MEDIASERVER_PAGE_SIZE = _env_int("MEDIASERVER_PAGE_SIZE", 500)

# --------------------------------------------------------------------------
# Audio front-end
# --------------------------------------------------------------------------
# CLAP mel (reference: clap_analyzer.py:396-430 + config.py:962-977)
CLAP_SAMPLE_RATE = 48000
CLAP_SEGMENT_SECONDS = 10.0
CLAP_SEGMENT_SAMPLES = 480000
CLAP_SEGMENT_HOP_SAMPLES = 240000
CLAP_AUDIO_N_MELS = _env_int("CLAP_AUDIO_N_MELS", 128)
CLAP_AUDIO_N_FFT = _env_int("CLAP_AUDIO_N_FFT", 2048)
CLAP_AUDIO_HOP_LENGTH = _env_int("CLAP_AUDIO_HOP_LENGTH", 480)
CLAP_AUDIO_FMIN = _env_float("CLAP_AUDIO_FMIN", 0.0)
CLAP_AUDIO_FMAX = _env_float("CLAP_AUDIO_FMAX", 14000.0)
CLAP_EMBEDDING_DIMENSION = _env_int("CLAP_EMBEDDING_DIMENSION", 512)
CLAP_ENABLED = _env_bool("CLAP_ENABLED", True)
LYRICS_ENABLED = _env_bool("LYRICS_ENABLED", False)
LYRICS_ASR_ENABLED = _env_bool("LYRICS_ASR_ENABLED", False)

# MusiCNN mel (reference: song.py:240-256)
MUSICNN_SAMPLE_RATE = 16000
MUSICNN_N_MELS = 96
MUSICNN_N_FFT = 512
MUSICNN_HOP = 256
MUSICNN_PATCH_FRAMES = 187
MUSICNN_BATCH_SIZE = _env_int("MUSICNN_BATCH_SIZE", 8)
EMBEDDING_DIMENSION = _env_int("EMBEDDING_DIMENSION", 200)

# Whisper log-mel (reference: lyrics/whisper_onnx.py:156-199)
WHISPER_SAMPLE_RATE = 16000
WHISPER_N_FFT = 400
WHISPER_HOP = 160
WHISPER_N_MELS = 80
WHISPER_CHUNK_SECONDS = 30

# Lyrics pipeline (reference: lyrics/lyrics_transcriber.py)
LYRICS_GTE_MAX_TOKENS = _env_int("LYRICS_GTE_MAX_TOKENS", 384)
LYRICS_MAX_AUDIO_SECONDS = _env_int("LYRICS_MAX_AUDIO_SECONDS", 240)
LYRICS_ASR_BEAM_SIZE = _env_int("LYRICS_ASR_BEAM_SIZE", 1)
LYRICS_EMBEDDING_DIMENSION = 768
LYRICS_AXIS_TEMPERATURE = _env_float("LYRICS_AXIS_TEMPERATURE", 0.1)

# --------------------------------------------------------------------------
# Labels (reference: config.py:752-818, 1195)
# --------------------------------------------------------------------------
MOOD_LABELS = [
    "rock", "pop", "alternative", "indie", "electronic", "female vocalists",
    "dance", "00s", "alternative rock", "jazz", "beautiful", "metal",
    "chillout", "male vocalists", "classic rock", "soul", "indie rock",
    "mellow", "electronica", "80s", "folk", "90s", "chill", "instrumental",
    "punk", "oldies", "blues", "hard rock", "ambient", "acoustic",
    "experimental", "female vocalist", "guitar", "hip-hop", "70s", "party",
    "country", "easy listening", "sexy", "catchy", "funk", "electro",
    "heavy metal", "progressive rock", "60s", "rnb", "indie pop",
    "sad", "house", "happy",
]
OTHER_FEATURE_LABELS = ["danceable", "aggressive", "happy", "party", "relaxed", "sad"]

LYRICS_AXES = [
    "love", "heartbreak", "party", "sadness", "joy", "anger", "hope",
    "nostalgia", "freedom", "faith", "money", "violence", "family",
    "friendship", "loneliness", "night", "summer", "city", "nature",
    "dance", "dreams", "death", "rebellion", "travel", "home", "work",
    "growing up",
]

# --------------------------------------------------------------------------
# IVF index engine (reference: config.py:1038-1043, paged_ivf.py:1412)
# --------------------------------------------------------------------------
IVF_NPROBE = _env_int("IVF_NPROBE", 1024)
IVF_NLIST_MAX = _env_int("IVF_NLIST_MAX", 8192)
IVF_STORAGE_DTYPE = _env("IVF_STORAGE_DTYPE", "i8")  # i8 | f16 | f32
IVF_TRAIN_POINTS_PER_CELL = _env_int("IVF_TRAIN_POINTS_PER_CELL", 256)
IVF_RERANK_OVERFETCH = _env_int("IVF_RERANK_OVERFETCH", 4)
# incremental refresh: full rebuild when more than this fraction changed
IVF_REFRESH_MAX_DRIFT = _env_float("IVF_REFRESH_MAX_DRIFT", 0.25)
# cron queue-guard retry cadence (reference: ALGORITHM.md 16.2-16.3)
CRON_RETRY_INTERVAL_MINUTES = _env_float("CRON_RETRY_INTERVAL_MINUTES", 5.0)
CRON_RETRY_MAX_MINUTES = _env_float("CRON_RETRY_MAX_MINUTES", 120.0)
# text-search model warm-up countdown (reference: clap_text_search.py:99)
CLAP_TEXT_SEARCH_WARMUP_DURATION = _env_float("CLAP_TEXT_SEARCH_WARMUP_DURATION", 300.0)
IVF_KMEANS_ITERS = _env_int("IVF_KMEANS_ITERS", 25)
IVF_MAX_PART_SIZE_MB = _env_int("IVF_MAX_PART_SIZE_MB", 32)

# Similarity / query behavior (reference: ivf_manager.py)
DUPLICATE_DISTANCE_CHECK_LOOKBACK = _env_int("DUPLICATE_DISTANCE_CHECK_LOOKBACK", 5)
DUPLICATE_DISTANCE_THRESHOLD_COSINE = _env_float("DUPLICATE_DISTANCE_THRESHOLD_COSINE", 0.01)
MAX_SONGS_PER_ARTIST = _env_int("MAX_SONGS_PER_ARTIST", 0)  # 0 = unlimited

# Simhash catalogue identity (reference: simhash.py)
SIMHASH_BITS = 200
SIMHASH_BANDS = 25
SIMHASH_CONFIRM_COSINE = _env_float("SIMHASH_CONFIRM_COSINE", 0.02)
SIMHASH_CONFIRM_DURATION_SECONDS = _env_float("SIMHASH_CONFIRM_DURATION_SECONDS", 5.0)

# --------------------------------------------------------------------------
# Clustering (reference: config.py clustering section)
# --------------------------------------------------------------------------
CLUSTERING_SUBSET_SONGS = _env_int("CLUSTERING_SUBSET_SONGS", 5000)
ITERATIONS_PER_BATCH_JOB = _env_int("ITERATIONS_PER_BATCH_JOB", 20)
MAX_CONCURRENT_BATCH_JOBS = _env_int("MAX_CONCURRENT_BATCH_JOBS", 4)
CLUSTERING_MAX_FAILED_BATCHES = _env_int("CLUSTERING_MAX_FAILED_BATCHES", 3)
CLUSTER_ALGORITHM = _env("CLUSTER_ALGORITHM", "kmeans")  # kmeans|dbscan|gmm|spectral
CLUSTERING_RUNS = _env_int("CLUSTERING_RUNS", 200)
TOP_N_PLAYLISTS = _env_int("TOP_N_PLAYLISTS", 10)

# --------------------------------------------------------------------------
# SemGrove fused lyrics+audio index (reference: sem_grove_manager.py)
# --------------------------------------------------------------------------
SEM_GROVE_LYRICS_WEIGHT = _env_float("SEM_GROVE_LYRICS_WEIGHT", 0.75)
SEM_GROVE_AUDIO_WEIGHT = _env_float("SEM_GROVE_AUDIO_WEIGHT", 0.25)

# Artist similarity (reference: artist_gmm_manager.py)
ARTIST_GMM_MIN_COMPONENTS = 2
ARTIST_GMM_MAX_COMPONENTS = 10
INDEX_BUILD_WORKERS = _env_int("INDEX_BUILD_WORKERS", 4)

# Sonic fingerprint (reference: sonic_fingerprint_manager.py)
SONIC_FINGERPRINT_HALF_LIFE_DAYS = _env_float("SONIC_FINGERPRINT_HALF_LIFE_DAYS", 30.0)
SONIC_FINGERPRINT_TOP_PLAYED = _env_int("SONIC_FINGERPRINT_TOP_PLAYED", 100)

# Hyperbolic explorer (reference: hyperbolic_manager.py / hyperbolic_geometry.py)
HYPERBOLIC_SCALE_PERCENTILE = _env_float("HYPERBOLIC_SCALE_PERCENTILE", 95.0)

# Radius walk (reference: radius_walk_helper.py)
RADIUS_WALK_BUCKETS = _env_int("RADIUS_WALK_BUCKETS", 10)
RADIUS_INSTRUMENTATION = _env_bool("RADIUS_INSTRUMENTATION", False)

# --------------------------------------------------------------------------
# GPU / compute substrate (new: MI355X-native)
# --------------------------------------------------------------------------
GPU_DTYPE = _env("GPU_DTYPE", "bf16")
CLAP_GPU_BATCH = _env_int("CLAP_GPU_BATCH", 256)
# opt-in fp8 e4m3 serving path for encoder GEMMs (ops/fp8.py); the
# headline bench stays bf16 regardless of this flag unless --fp8 is passed
CLAP_FP8_SERVING = _env_bool("AUDIOMUSE_FP8_SERVING", False)
FP8_HIDDEN_ENABLE = _env_bool("AUDIOMUSE_FP8_HIDDEN", False)
HIP_REQUIRE_NATIVE = _env_bool("HIP_REQUIRE_NATIVE", True)  # fail loudly on GPU without .so
RCCL_BUCKET_CAP_MB = _env_int("RCCL_BUCKET_CAP_MB", 64)

# AI instant playlist (reference: tasks/ai/)
AI_PROVIDER = _env("AI_PROVIDER", "none")  # none|openai|gemini|mistral
AI_MODEL_NAME = _env("AI_MODEL_NAME", "")
AI_BASE_URL = _env("AI_BASE_URL", "")      # override vendor default base
AI_API_KEY = _env("AI_API_KEY", "")
AI_MAX_TOOL_CALLS = _env_int("AI_MAX_TOOL_CALLS", 4)

# Web
JWT_SECRET = _env("AUDIOMUSE_JWT_SECRET", "")
API_TOKEN = _env("AUDIOMUSE_API_TOKEN", "")
# Opt-in trust of X-Forwarded-Prefix/Proto (reference: proxy_prefix.py
# is mounted only under a known proxy deployment)
BEHIND_PROXY = _env_bool("AUDIOMUSE_BEHIND_PROXY", False)


def set_db_override_provider(provider: Optional[Callable[[], Dict[str, str]]]) -> None:
    """Register a callable returning {CONFIG_NAME: value} persisted overrides
    (reference: config._apply_db_overrides, config.py:1395)."""
    global _DB_OVERRIDE_PROVIDER
    _DB_OVERRIDE_PROVIDER = provider


def apply_db_overrides(overrides: Dict[str, str]) -> int:
    """Layer persisted overrides onto the module (reference:
    config._apply_db_overrides :1395). Non-destructive: only keys that
    already exist change; types coerce to the current value's type."""
    import sys

    module = sys.modules[__name__]
    applied = 0
    for key, raw in (overrides or {}).items():
        if not key.isupper() or not hasattr(module, key):
            continue
        current = getattr(module, key)
        try:
            if isinstance(current, bool):
                value = str(raw).strip().lower() in ("1", "true", "yes", "on")
            elif isinstance(current, int):
                value = int(raw)
            elif isinstance(current, float):
                value = float(raw)
            elif isinstance(current, str):
                value = str(raw)
            else:
                continue  # lists/complex values are code-owned
        except (TypeError, ValueError):
            continue
        setattr(module, key, value)
        applied += 1
    return applied


def refresh_config() -> None:
    """Re-apply the registered DB override provider
    (reference: config.refresh_config :1389)."""
    if _DB_OVERRIDE_PROVIDER is not None:
        apply_db_overrides(_DB_OVERRIDE_PROVIDER() or {})
