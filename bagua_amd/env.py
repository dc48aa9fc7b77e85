"""Environment-variable configuration accessors.

MI355X-native re-design of the reference's env layer
(reference: bagua/torch_api/env.py:15-124). Same knob names where behavior
is equivalent so launcher scripts port unchanged; new knobs are prefixed
BAGUA_ as well.

Defaults are sized for one 8xMI355X node: the default fusion bucket is
32 MiB (vs the reference's 10 MiB) because xGMI ring collectives on 7
point-to-point links amortize launch latency better with larger buckets,
and 288 GB HBM3E per GPU makes the extra staging memory irrelevant.
"""

import os


def get_rank() -> int:
    """Global rank of the current process (default 0)."""
    return int(os.environ.get("RANK", 0))


def get_world_size() -> int:
    """Number of processes in the job (default 1)."""
    return int(os.environ.get("WORLD_SIZE", 1))


def get_local_rank() -> int:
    """Rank within the local node; also the GPU index (default 0)."""
    return int(os.environ.get("LOCAL_RANK", 0))


def get_local_size() -> int:
    """Number of processes on this node (default 1)."""
    return int(os.environ.get("LOCAL_WORLD_SIZE", 1))


def get_node_rank() -> int:
    """Rank of this node. torchelastic exports GROUP_RANK; NODE_RANK wins
    if set explicitly (reference: env.py:52-68)."""
    return int(os.environ.get("NODE_RANK", os.environ.get("GROUP_RANK", 0)))


def get_master_addr() -> str:
    return os.environ.get("MASTER_ADDR", "127.0.0.1")


def get_master_port() -> int:
    return int(os.environ.get("MASTER_PORT", 29500))


# ---------------------------------------------------------------------------
# Bagua-specific knobs
# ---------------------------------------------------------------------------

def get_default_bucket_size() -> int:
    """Gradient fusion bucket size in bytes (default 32 MiB on MI355X)."""
    return int(os.environ.get("BAGUA_DEFAULT_BUCKET_SIZE", 32 * 1024 * 1024))


def get_autotune_level() -> int:
    """0 = off, 1 = bucket-size/hierarchy search (reference: env.py:98-106)."""
    return int(os.environ.get("BAGUA_AUTOTUNE", 0))


def get_autotune_max_samples() -> int:
    return int(os.environ.get("BAGUA_AUTOTUNE_MAX_SAMPLES", 60))


def get_autotune_sampling_confidence_time_s() -> float:
    return float(os.environ.get("BAGUA_AUTOTUNE_SAMPLING_CONFIDENCE_TIME_S", 5.0))


def get_autotune_warmup_time_s() -> float:
    return float(os.environ.get("BAGUA_AUTOTUNE_WARMUP_TIME_S", 30.0))

def is_report_metrics_enabled() -> bool:
    return os.environ.get("BAGUA_REPORT_METRICS", "0") == "1"


def get_autotune_server_addr() -> str:
    return os.environ.get("AUTO_TUNE_SERVER_ADDR", "")


def get_bagua_service_port() -> int:
    return int(os.environ.get("BAGUA_SERVICE_PORT", -1))


def get_autotune_server_wait_time_s() -> float:
    return float(os.environ.get("BAGUA_AUTOTUNE_SERVER_WAIT_TIME_S", 60.0))


def is_output_autotune_log() -> bool:
    return os.environ.get("BAGUA_IS_OUTPUT_AUTOTUNE_LOG", "0") == "1"


def find_free_network_port() -> int:
    """Bind an ephemeral port on 127.0.0.1 and return it."""
    import socket

    s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port
