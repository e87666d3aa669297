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
# Discrete POSTGRES_* variables (the reference's env-only group,
# docs/ALGORITHM.md:129-135) assemble a postgresql:// URL when
# POSTGRES_HOST is set; an explicit DATABASE_URL always wins.
POSTGRES_USER = _env("POSTGRES_USER", "audiomuse")
POSTGRES_PASSWORD = _env("POSTGRES_PASSWORD", "")
POSTGRES_DB = _env("POSTGRES_DB", "audiomuse")
POSTGRES_HOST = _env("POSTGRES_HOST", "")
POSTGRES_PORT = _env_int("POSTGRES_PORT", 5432)

_default_db = "sqlite:///" + os.path.join(
    os.environ.get("AUDIOMUSE_DATA_DIR", os.path.expanduser("~/.audiomuse-amd")),
    "audiomuse.db")
if POSTGRES_HOST:
    _default_db = (f"postgresql://{POSTGRES_USER}:{POSTGRES_PASSWORD}"
                   f"@{POSTGRES_HOST}:{POSTGRES_PORT}/{POSTGRES_DB}")
DATABASE_URL = _env("DATABASE_URL", _default_db)
DATA_DIR = _env("AUDIOMUSE_DATA_DIR", os.path.expanduser("~/.audiomuse-amd"))
TZ = _env("TZ", "UTC")
# apply the zone process-wide (reference: TZ is env-only and applied at
# boot, docs/ALGORITHM.md:129-135)
os.environ.setdefault("TZ", TZ)
try:
    import time as _time
    _time.tzset()
except Exception:   # pragma: no cover — tzset is POSIX-only
    pass

# Queue semantics (reference: taskqueue/sql.py, maintenance.py)
QUEUE_MAX_ATTEMPTS = _env_int("QUEUE_MAX_ATTEMPTS", 3)
QUEUE_LEASE_SECONDS = _env_float("QUEUE_LEASE_SECONDS", 30.0)
QUEUE_HEARTBEAT_SECONDS = _env_float("QUEUE_HEARTBEAT_SECONDS", 5.0)
QUEUE_POLL_SECONDS = _env_float("QUEUE_POLL_SECONDS", 0.25)
MAX_QUEUED_ANALYSIS_JOBS = _env_int("MAX_QUEUED_ANALYSIS_JOBS", 30)
REBUILD_INDEX_BATCH_SIZE = _env_int("REBUILD_INDEX_BATCH_SIZE", 500)
WORKER_MAX_JOBS = _env_int("WORKER_MAX_JOBS", 50)

# Queue control plane (reference: taskqueue/control.py)
CONTROL_WINDOW_SECONDS = _env_float("CONTROL_WINDOW_SECONDS", 60.0)
CRON_ENABLED = _env_bool("CRON_ENABLED", True)

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
MUSICSERVER_LYRICS_TIMEOUT = _env_float("MUSICSERVER_LYRICS_TIMEOUT", 2.5)

# Default media-server connection (reference: docs/PARAMETERS.md
# "Mediaserver General"; the music_servers registry rows override these
# per server — these are the single-server/legacy-env path)
MEDIASERVER_TYPE = _env("MEDIASERVER_TYPE", "synthetic")
NAVIDROME_URL = _env("NAVIDROME_URL", "")
NAVIDROME_USER = _env("NAVIDROME_USER", "")
NAVIDROME_PASSWORD = _env("NAVIDROME_PASSWORD", "")
NAVIDROME_API_KEY = _env("NAVIDROME_API_KEY", "")
JELLYFIN_URL = _env("JELLYFIN_URL", "")
JELLYFIN_USER_ID = _env("JELLYFIN_USER_ID", "")
JELLYFIN_TOKEN = _env("JELLYFIN_TOKEN", "")
EMBY_URL = _env("EMBY_URL", "")
EMBY_USER_ID = _env("EMBY_USER_ID", "")
EMBY_TOKEN = _env("EMBY_TOKEN", "")
LYRION_URL = _env("LYRION_URL", "")
PLEX_URL = _env("PLEX_URL", "")
PLEX_TOKEN = _env("PLEX_TOKEN", "")
# comma-separated library/section/folder ids to scope scans to
MUSIC_LIBRARIES = _env("MUSIC_LIBRARIES", "")

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

# Analysis orchestration (reference: tasks/analysis/ + config.py)
NUM_RECENT_ALBUMS = _env_int("NUM_RECENT_ALBUMS", 0)   # 0 = whole library
TOP_N_MOODS = _env_int("TOP_N_MOODS", 5)
ANALYSIS_MONITOR_DB_INTERVAL = _env_float("ANALYSIS_MONITOR_DB_INTERVAL", 2.0)
PER_SONG_MODEL_RELOAD = _env_bool("PER_SONG_MODEL_RELOAD", False)
CATALOGUE_ID_SCHEME_VERSION = _env_int("CATALOGUE_ID_SCHEME_VERSION", 4)
LYRICS_MUSICNN_SKIP = _env_bool("LYRICS_MUSICNN_SKIP", False)

# Chromaprint gates (reference: config.py CHROMAPRINT_* +
# tasks/chromaprint.py compare params)
CHROMAPRINT_COLLECTION_ENABLED = _env_bool("CHROMAPRINT_COLLECTION_ENABLED",
                                           True)
CHROMAPRINT_GATE_ENABLED = _env_bool("CHROMAPRINT_GATE_ENABLED", True)
CHROMAPRINT_BACKFILL_ALBUMS_PER_RUN = _env_int(
    "CHROMAPRINT_BACKFILL_ALBUMS_PER_RUN", 50)
CHROMAPRINT_MATCH_THRESHOLD = _env_float("CHROMAPRINT_MATCH_THRESHOLD", 0.85)
CHROMAPRINT_ALIGN_RANGE = _env_int("CHROMAPRINT_ALIGN_RANGE", 80)
CHROMAPRINT_MIN_OVERLAP = _env_int("CHROMAPRINT_MIN_OVERLAP", 120)

# Cleaning / multi-server sweep (reference: tasks/cleaning.py,
# multiserver_sync.py)
CLEANING_SAFETY_LIMIT = _env_int("CLEANING_SAFETY_LIMIT", 100)
CLEANING_CATALOGUE = _env_bool("CLEANING_CATALOGUE", False)
SWEEP_PRUNE_MIN_FETCH_RATIO = _env_float("SWEEP_PRUNE_MIN_FETCH_RATIO", 0.8)

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

# Lyrics pipeline (reference: lyrics/lyrics_transcriber.py; stage gates
# documented in docs/PARAMETERS.md "Lyrics")
LYRICS_GTE_MAX_TOKENS = _env_int("LYRICS_GTE_MAX_TOKENS", 384)
LYRICS_MAX_AUDIO_SECONDS = _env_int("LYRICS_MAX_AUDIO_SECONDS", 240)
LYRICS_ASR_BEAM_SIZE = _env_int("LYRICS_ASR_BEAM_SIZE", 1)
LYRICS_EMBEDDING_DIMENSION = 768
LYRICS_AXIS_TEMPERATURE = _env_float("LYRICS_AXIS_TEMPERATURE", 0.1)
VAD_VOICE_RECOGNITION = _env_bool("VAD_VOICE_RECOGNITION", True)
LYRICS_ASR_MIN_AVG_LOGPROB = _env_float("LYRICS_ASR_MIN_AVG_LOGPROB", -1.0)
LYRICS_ASR_NON_ENGLISH_MIN_LOGPROB = _env_float(
    "LYRICS_ASR_NON_ENGLISH_MIN_LOGPROB", -0.7)
LYRICS_TEXT_MAX_COMPRESSION_RATIO = _env_float(
    "LYRICS_TEXT_MAX_COMPRESSION_RATIO", 15.0)
LYRICS_MIN_CHARS_FOR_EMBEDDING = _env_int("LYRICS_MIN_CHARS_FOR_EMBEDDING",
                                          250)
LYRICS_LANG_CONFIDENCE_MIN = _env_float("LYRICS_LANG_CONFIDENCE_MIN", 0.7)
LYRICS_CJK_SCRIPT_MIN_RATIO = _env_float("LYRICS_CJK_SCRIPT_MIN_RATIO", 0.1)
LYRICS_GTE_WARMUP_DURATION = _env_float("LYRICS_GTE_WARMUP_DURATION", 300.0)
LYRICS_MAX_WORDS = _env_int("LYRICS_MAX_WORDS", 300)

# External lyrics APIs, tried before ASR (reference:
# lyrics_transcriber.py stages 1-2 + LYRICS_API_{1,2}_* in PARAMETERS.md).
# URL templates take {artist}/{title} placeholders.
LYRICS_API_ENABLE = _env_bool("LYRICS_API_ENABLE", False)
LYRICS_API_1_URL_TEMPLATE = _env(
    "LYRICS_API_1_URL_TEMPLATE", "https://lrclib.net/api/get")
LYRICS_API_1_ARTIST_PARAM = _env("LYRICS_API_1_ARTIST_PARAM", "artist_name")
LYRICS_API_1_TITLE_PARAM = _env("LYRICS_API_1_TITLE_PARAM", "track_name")
LYRICS_API_1_LYRICS_FIELD = _env("LYRICS_API_1_LYRICS_FIELD", "plainLyrics")
LYRICS_API_1_APIKEY_PARAM = _env("LYRICS_API_1_APIKEY_PARAM", "")
LYRICS_API_1_APIKEY_VALUE = _env("LYRICS_API_1_APIKEY_VALUE", "")
LYRICS_API_1_TIMEOUT = _env_float("LYRICS_API_1_TIMEOUT", 5.0)
LYRICS_API_2_URL_TEMPLATE = _env("LYRICS_API_2_URL_TEMPLATE", "")
LYRICS_API_2_ARTIST_PARAM = _env("LYRICS_API_2_ARTIST_PARAM", "artist")
LYRICS_API_2_TITLE_PARAM = _env("LYRICS_API_2_TITLE_PARAM", "title")
LYRICS_API_2_LYRICS_FIELD = _env("LYRICS_API_2_LYRICS_FIELD", "lyrics")
LYRICS_API_2_APIKEY_PARAM = _env("LYRICS_API_2_APIKEY_PARAM", "")
LYRICS_API_2_APIKEY_VALUE = _env("LYRICS_API_2_APIKEY_VALUE", "")
LYRICS_API_2_TIMEOUT = _env_float("LYRICS_API_2_TIMEOUT", 5.0)

# Published-vocabulary interop (models/text.py TrainedTokenizer):
# path to a HuggingFace tokenizer.json; empty = hashed stand-in
TOKENIZER_JSON = _env("AUDIOMUSE_TOKENIZER_JSON", "")

# Whisper decode behavior (models/whisper.py decoder loop)
WHISPER_MAX_NEW_TOKENS = _env_int("WHISPER_MAX_NEW_TOKENS", 224)
WHISPER_NO_REPEAT_NGRAM = _env_int("WHISPER_NO_REPEAT_NGRAM", 3)
WHISPER_REPETITION_PENALTY = _env_float("WHISPER_REPETITION_PENALTY", 1.2)

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
IVF_METRIC = _env("IVF_METRIC", "angular")  # angular | euclidean | dot
IVF_KMEANS_BATCH = _env_int("IVF_KMEANS_BATCH", 10000)
# cells larger than this split with a sub-kmeans (reference:
# paged_ivf.py:1337 oversized-cell split)
IVF_MAX_CELL_ROWS = _env_int("IVF_MAX_CELL_ROWS", 65536)
# similarity-result TTL cache (reference: ivf_manager._ResultCache :73)
IVF_RESULT_CACHE_SECONDS = _env_float("IVF_RESULT_CACHE_SECONDS", 300.0)
IVF_RESULT_CACHE_MAX = _env_int("IVF_RESULT_CACHE_MAX", 512)
# radius/max-distance queries probe wider (reference: config.py
# IVF_MAX_DISTANCE_NPROBE)
IVF_MAX_DISTANCE_NPROBE = _env_int("IVF_MAX_DISTANCE_NPROBE", 2048)

# Similarity / query behavior (reference: ivf_manager.py)
DUPLICATE_DISTANCE_CHECK_LOOKBACK = _env_int("DUPLICATE_DISTANCE_CHECK_LOOKBACK", 5)
DUPLICATE_DISTANCE_THRESHOLD_COSINE = _env_float("DUPLICATE_DISTANCE_THRESHOLD_COSINE", 0.01)
DUPLICATE_DISTANCE_THRESHOLD_EUCLIDEAN = _env_float(
    "DUPLICATE_DISTANCE_THRESHOLD_EUCLIDEAN", 0.15)
MAX_SONGS_PER_ARTIST = _env_int("MAX_SONGS_PER_ARTIST", 0)  # 0 = unlimited
SIMILARITY_ELIMINATE_DUPLICATES_DEFAULT = _env_bool(
    "SIMILARITY_ELIMINATE_DUPLICATES_DEFAULT", True)
SIMILARITY_RADIUS_DEFAULT = _env_bool("SIMILARITY_RADIUS_DEFAULT", False)
MOOD_SIMILARITY_ENABLE = _env_bool("MOOD_SIMILARITY_ENABLE", False)
MOOD_SIMILARITY_THRESHOLD = _env_float("MOOD_SIMILARITY_THRESHOLD", 0.2)
MOOD_SCORE_MATCH_THRESHOLD = _env_float("MOOD_SCORE_MATCH_THRESHOLD", 0.1)

# Song path (reference: path_manager.py + config PATH_*)
PATH_DISTANCE_METRIC = _env("PATH_DISTANCE_METRIC", "angular")
PATH_DEFAULT_LENGTH = _env_int("PATH_DEFAULT_LENGTH", 25)
PATH_FIX_SIZE = _env_bool("PATH_FIX_SIZE", True)

# Song alchemy (reference: song_alchemy.py + config ALCHEMY_*)
ALCHEMY_DEFAULT_N_RESULTS = _env_int("ALCHEMY_DEFAULT_N_RESULTS", 50)
ALCHEMY_MAX_N_RESULTS = _env_int("ALCHEMY_MAX_N_RESULTS", 200)
ALCHEMY_TEMPERATURE = _env_float("ALCHEMY_TEMPERATURE", 0.15)
ALCHEMY_SUBTRACT_RADIUS = _env_float("ALCHEMY_SUBTRACT_RADIUS", 0.25)

# Playlist ordering (reference: playlist_ordering.py)
PLAYLIST_ENERGY_ARC = _env_bool("PLAYLIST_ENERGY_ARC", False)
MAX_SONGS_PER_ARTIST_PLAYLIST = _env_int("MAX_SONGS_PER_ARTIST_PLAYLIST", 0)

# Simhash catalogue identity (reference: simhash.py)
SIMHASH_BITS = 200
SIMHASH_BANDS = 25
SIMHASH_CONFIRM_COSINE = _env_float("SIMHASH_CONFIRM_COSINE", 0.02)
SIMHASH_CONFIRM_DURATION_SECONDS = _env_float("SIMHASH_CONFIRM_DURATION_SECONDS", 5.0)

# --------------------------------------------------------------------------
# Clustering (reference: config.py clustering section + PARAMETERS.md)
# --------------------------------------------------------------------------
CLUSTERING_SUBSET_SONGS = _env_int("CLUSTERING_SUBSET_SONGS", 5000)
ITERATIONS_PER_BATCH_JOB = _env_int("ITERATIONS_PER_BATCH_JOB", 20)
MAX_CONCURRENT_BATCH_JOBS = _env_int("MAX_CONCURRENT_BATCH_JOBS", 4)
CLUSTERING_MAX_FAILED_BATCHES = _env_int("CLUSTERING_MAX_FAILED_BATCHES", 3)
CLUSTER_ALGORITHM = _env("CLUSTER_ALGORITHM", "kmeans")  # kmeans|dbscan|gmm|spectral
CLUSTERING_RUNS = _env_int("CLUSTERING_RUNS", 200)
TOP_N_PLAYLISTS = _env_int("TOP_N_PLAYLISTS", 10)
ENABLE_CLUSTERING_EMBEDDINGS = _env_bool("ENABLE_CLUSTERING_EMBEDDINGS", True)
MAX_SONGS_PER_CLUSTER = _env_int("MAX_SONGS_PER_CLUSTER", 0)
MIN_PLAYLIST_SIZE_FOR_TOP_N = _env_int("MIN_PLAYLIST_SIZE_FOR_TOP_N", 3)
CLUSTERING_MAX_PLAYLIST_SONGS = _env_int("CLUSTERING_MAX_PLAYLIST_SONGS", 100)
CLUSTERING_CLEANING = _env_bool("CLUSTERING_CLEANING", True)
USE_GPU_CLUSTERING = _env_bool("USE_GPU_CLUSTERING", True)
CLUSTERING_AUTO_CALIBRATION = _env_bool("CLUSTERING_AUTO_CALIBRATION", True)
CLUSTERING_CALIBRATION_MAX_TRIES = _env_int(
    "CLUSTERING_CALIBRATION_MAX_TRIES", 5)
CLUSTERING_EARLY_STOP_BATCHES = _env_int("CLUSTERING_EARLY_STOP_BATCHES", 30)
CLUSTERING_STALL_TIMEOUT_MINUTES = _env_float(
    "CLUSTERING_STALL_TIMEOUT_MINUTES", 30.0)
TOP_N_ELITES = _env_int("TOP_N_ELITES", 5)
EXPLOITATION_START_FRACTION = _env_float("EXPLOITATION_START_FRACTION", 0.2)
EXPLOITATION_PROBABILITY_CONFIG = _env_float(
    "EXPLOITATION_PROBABILITY_CONFIG", 0.6)
MUTATION_INT_ABS_DELTA = _env_int("MUTATION_INT_ABS_DELTA", 3)
MUTATION_FLOAT_ABS_DELTA = _env_float("MUTATION_FLOAT_ABS_DELTA", 0.25)
MUTATION_KMEANS_COORD_FRACTION = _env_float(
    "MUTATION_KMEANS_COORD_FRACTION", 0.05)
SAMPLING_PERCENTAGE_CHANGE_PER_RUN = _env_float(
    "SAMPLING_PERCENTAGE_CHANGE_PER_RUN", 0.2)
MIN_SONGS_PER_GENRE_FOR_STRATIFICATION = _env_int(
    "MIN_SONGS_PER_GENRE_FOR_STRATIFICATION", 100)
STRATIFIED_SAMPLING_TARGET_PERCENTILE = _env_float(
    "STRATIFIED_SAMPLING_TARGET_PERCENTILE", 50.0)
TOP_K_MOODS_FOR_PURITY_CALCULATION = _env_int(
    "TOP_K_MOODS_FOR_PURITY_CALCULATION", 3)
CLUSTER_NAMING_AI_HISTORY = _env_bool("CLUSTER_NAMING_AI_HISTORY", True)
PLAYLIST_NAME_HISTORY_ROUNDS = _env_int("PLAYLIST_NAME_HISTORY_ROUNDS", 3)

# parameter-space bounds for the evolutionary search (reference:
# clustering_helper.py:426-596 explore ranges)
NUM_CLUSTERS_MIN = _env_int("NUM_CLUSTERS_MIN", 2)
NUM_CLUSTERS_MAX = _env_int("NUM_CLUSTERS_MAX", 40)
DBSCAN_EPS_MIN = _env_float("DBSCAN_EPS_MIN", 0.2)
DBSCAN_EPS_MAX = _env_float("DBSCAN_EPS_MAX", 2.5)
DBSCAN_MIN_SAMPLES_MIN = _env_int("DBSCAN_MIN_SAMPLES_MIN", 3)
DBSCAN_MIN_SAMPLES_MAX = _env_int("DBSCAN_MIN_SAMPLES_MAX", 15)
GMM_N_COMPONENTS_MIN = _env_int("GMM_N_COMPONENTS_MIN", 2)
GMM_N_COMPONENTS_MAX = _env_int("GMM_N_COMPONENTS_MAX", 30)
GMM_COVARIANCE_TYPE = _env("GMM_COVARIANCE_TYPE", "diag")
SPECTRAL_N_CLUSTERS_MIN = _env_int("SPECTRAL_N_CLUSTERS_MIN", 2)
SPECTRAL_N_CLUSTERS_MAX = _env_int("SPECTRAL_N_CLUSTERS_MAX", 30)
SPECTRAL_N_NEIGHBORS = _env_int("SPECTRAL_N_NEIGHBORS", 10)
PCA_COMPONENTS_MIN = _env_int("PCA_COMPONENTS_MIN", 0)
PCA_COMPONENTS_MAX = _env_int("PCA_COMPONENTS_MAX", 32)

# 7-metric fitness weights (reference: clustering_helper.py:689
# _format_and_score_iteration_result; SCORE_WEIGHT_* in PARAMETERS.md)
SCORE_WEIGHT_DIVERSITY = _env_float("SCORE_WEIGHT_DIVERSITY", 2.0)
SCORE_WEIGHT_PURITY = _env_float("SCORE_WEIGHT_PURITY", 1.0)
SCORE_WEIGHT_OTHER_FEATURE_DIVERSITY = _env_float(
    "SCORE_WEIGHT_OTHER_FEATURE_DIVERSITY", 0.0)
SCORE_WEIGHT_OTHER_FEATURE_PURITY = _env_float(
    "SCORE_WEIGHT_OTHER_FEATURE_PURITY", 0.0)
SCORE_WEIGHT_SILHOUETTE = _env_float("SCORE_WEIGHT_SILHOUETTE", 0.0)
SCORE_WEIGHT_DAVIES_BOULDIN = _env_float("SCORE_WEIGHT_DAVIES_BOULDIN", 0.0)
SCORE_WEIGHT_CALINSKI_HARABASZ = _env_float(
    "SCORE_WEIGHT_CALINSKI_HARABASZ", 0.0)

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
SONIC_FINGERPRINT_TOP_N_SONGS = _env_int("SONIC_FINGERPRINT_TOP_N_SONGS", 50)
SONIC_FINGERPRINT_NEIGHBORS = _env_int("SONIC_FINGERPRINT_NEIGHBORS", 3)
SONIC_FINGERPRINT_MAX_SONGS_PER_ALBUM = _env_int(
    "SONIC_FINGERPRINT_MAX_SONGS_PER_ALBUM", 0)
SONIC_FINGERPRINT_CRON_PLAYLIST_NAME = _env(
    "SONIC_FINGERPRINT_CRON_PLAYLIST_NAME", "Sonic Fingerprint_automatic")

# Hyperbolic explorer (reference: hyperbolic_manager.py /
# hyperbolic_geometry.py; tree knobs in PARAMETERS.md "Hyperbolic")
HYPERBOLIC_SCALE_PERCENTILE = _env_float("HYPERBOLIC_SCALE_PERCENTILE", 95.0)
HYPERBOLIC_DEFAULT_LIMIT = _env_int("HYPERBOLIC_DEFAULT_LIMIT", 50)
HYPERBOLIC_MAX_LIMIT = _env_int("HYPERBOLIC_MAX_LIMIT", 500)
HYPERBOLIC_RADIAL_SPREAD = _env_float("HYPERBOLIC_RADIAL_SPREAD", 0.8)
HYPERBOLIC_CANDIDATE_OVERFETCH = _env_int("HYPERBOLIC_CANDIDATE_OVERFETCH", 4)
HYPERBOLIC_RADIUS_SCALE = _env_float("HYPERBOLIC_RADIUS_SCALE", 1.0)
HYPERBOLIC_RADIUS_PERCENTILE = _env_float("HYPERBOLIC_RADIUS_PERCENTILE", 95.0)
HYPERBOLIC_TARGET_LEAF_SIZE = _env_int("HYPERBOLIC_TARGET_LEAF_SIZE", 200)
HYPERBOLIC_MIN_CLUSTER_SIZE = _env_int("HYPERBOLIC_MIN_CLUSTER_SIZE", 10)
HYPERBOLIC_TREE_WARMUP_DURATION = _env_float(
    "HYPERBOLIC_TREE_WARMUP_DURATION", 300.0)

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
# fp8-ingest attention experiment: measured NEGATIVE on hardware (kernel
# 0.51 vs 0.45 ms s1 / 0.11 vs 0.09 ms s3, bench 10 275 vs 10 861 —
# profiles/r2_fp8_attn_negative.md): the window kernel is gather-LATENCY
# bound, and the e4m3->bf16 converts sit on the load->MFMA critical path
# while the halved bytes buy nothing. OFF by default.
FP8_ATTN_ENABLE = _env_bool("AUDIOMUSE_FP8_ATTN", False)
HIP_REQUIRE_NATIVE = _env_bool("HIP_REQUIRE_NATIVE", True)  # fail loudly on GPU without .so
RCCL_BUCKET_CAP_MB = _env_int("RCCL_BUCKET_CAP_MB", 64)

# AI instant playlist (reference: tasks/ai/ + PARAMETERS.md "AI")
AI_PROVIDER = _env("AI_MODEL_PROVIDER", _env("AI_PROVIDER", "none"))
AI_MODEL_NAME = _env("AI_MODEL_NAME", "")
AI_BASE_URL = _env("AI_BASE_URL", "")      # override vendor default base
AI_API_KEY = _env("AI_API_KEY", "")
AI_MAX_TOOL_CALLS = _env_int("AI_MAX_TOOL_CALLS", 4)
AI_REQUEST_TIMEOUT_SECONDS = _env_float("AI_REQUEST_TIMEOUT_SECONDS", 60.0)
MAX_SONGS_IN_AI_PROMPT = _env_int("MAX_SONGS_IN_AI_PROMPT", 100)
AI_TOOLCALL_TEMPERATURE = _env_float("AI_TOOLCALL_TEMPERATURE", 0.2)
AI_TOOLCALL_TOP_P = _env_float("AI_TOOLCALL_TOP_P", 0.9)
AI_TOOLCALL_TOP_K = _env_int("AI_TOOLCALL_TOP_K", 40)
AI_TOOLCALL_MIN_P = _env_float("AI_TOOLCALL_MIN_P", 0.0)
AI_TOOLCALL_NUM_PREDICT = _env_int("AI_TOOLCALL_NUM_PREDICT", 1024)
# per-vendor connection details (reference: providers/{openai,gemini,
# mistral}.py; OLLAMA rides the openai-compatible client)
OPENAI_API_KEY = _env("OPENAI_API_KEY", "no-key-needed")
OPENAI_MODEL_NAME = _env("OPENAI_MODEL_NAME", "gpt-4o-mini")
OPENAI_SERVER_URL = _env("OPENAI_SERVER_URL", "https://api.openai.com/v1")
OLLAMA_SERVER_URL = _env("OLLAMA_SERVER_URL", "http://127.0.0.1:11434/v1")
OLLAMA_MODEL_NAME = _env("OLLAMA_MODEL_NAME", "llama3.1")
GEMINI_API_KEY = _env("GEMINI_API_KEY", "")
GEMINI_MODEL_NAME = _env("GEMINI_MODEL_NAME", "gemini-2.0-flash")
MISTRAL_API_KEY = _env("MISTRAL_API_KEY", "")
MISTRAL_MODEL_NAME = _env("MISTRAL_MODEL_NAME", "mistral-small-latest")

# Plugins (reference: plugin/ + PARAMETERS.md "Plugins"; pip-install of
# plugin requirements is deliberately unsupported in this build, so
# PLUGIN_ALLOW_PIP defaults False and is refused when set)
PLUGINS_ENABLED = _env_bool("PLUGINS_ENABLED", True)
PLUGIN_MAX_DOWNLOAD_MB = _env_int("PLUGIN_MAX_DOWNLOAD_MB", 50)
PLUGIN_ALLOW_PIP = _env_bool("PLUGIN_ALLOW_PIP", False)

# Web / dashboard
JWT_SECRET = _env("AUDIOMUSE_JWT_SECRET", _env("JWT_SECRET", ""))
API_TOKEN = _env("AUDIOMUSE_API_TOKEN", _env("API_TOKEN", ""))
AUTH_ENABLED = _env_bool("AUTH_ENABLED", True)
AUDIOMUSE_USER = _env("AUDIOMUSE_USER", "")
AUDIOMUSE_PASSWORD = _env("AUDIOMUSE_PASSWORD", "")
# Opt-in trust of X-Forwarded-Prefix/Proto (reference name:
# ENABLE_PROXY_FIX; mounted only under a known reverse proxy)
BEHIND_PROXY = _env_bool("ENABLE_PROXY_FIX",
                         _env_bool("AUDIOMUSE_BEHIND_PROXY", False))
DASHBOARD_BROWSE_PAGE_SIZE = _env_int("DASHBOARD_BROWSE_PAGE_SIZE", 100)
DASHBOARD_BROWSE_MAX_OFFSET = _env_int("DASHBOARD_BROWSE_MAX_OFFSET", 10000)
DASHBOARD_REFRESH_SECONDS = _env_float("DASHBOARD_REFRESH_SECONDS", 60.0)


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


# Pristine env-resolved defaults, captured at import BEFORE any DB
# override is layered on (reference: /api/config/defaults — the setup
# UI shows "reset to default" values from here).
_DEFAULTS_SNAPSHOT: Dict[str, object] = {
    _k: _v for _k, _v in list(globals().items())
    if _k.isupper() and isinstance(_v, (bool, int, float, str))
}


def defaults() -> Dict[str, object]:
    return dict(_DEFAULTS_SNAPSHOT)
