"""Deployment configuration knobs.

Parity target: /root/reference/metaflow/metaflow_config.py from_conf pattern
(env ``MFX_<NAME>`` > JSON profile ``~/.mfxconfig/config.json`` > default).
"""

import json
import os

_CONFIG_CACHE = None


def _load_profile():
    global _CONFIG_CACHE
    if _CONFIG_CACHE is None:
        _CONFIG_CACHE = {}
        profile = os.environ.get("MFX_PROFILE", "")
        suffix = "_%s" % profile if profile else ""
        path = os.path.expanduser("~/.mfxconfig/config%s.json" % suffix)
        if os.path.exists(path):
            try:
                with open(path) as f:
                    _CONFIG_CACHE = json.load(f)
            except Exception:
                _CONFIG_CACHE = {}
    return _CONFIG_CACHE


def from_conf(name, default=None, coerce=None):
    env_key = "MFX_%s" % name
    if env_key in os.environ:
        value = os.environ[env_key]
    else:
        value = _load_profile().get(name, default)
    if coerce is not None and value is not None:
        try:
            value = coerce(value)
        except (TypeError, ValueError):
            value = default
    return value


# --- scheduler envelope (reference: runtime.py:64-67) -----------------------
MAX_WORKERS = from_conf("MAX_WORKERS", 16, int)
MAX_NUM_SPLITS = from_conf("MAX_NUM_SPLITS", 100, int)
MAX_LOG_SIZE = from_conf("MAX_LOG_SIZE", 1024 * 1024, int)
POLL_TIMEOUT_MS = from_conf("POLL_TIMEOUT_MS", 1000, int)
MAX_ATTEMPTS = from_conf("MAX_ATTEMPTS", 4, int)
# scheduler liveness: kill a task whose wall-clock exceeds
# TASK_STALL_TIMEOUT seconds (0 = disabled; @timeout is the opt-in
# per-step variant INSIDE the task — this one catches tasks that can't
# even run their own signal handler, e.g. a wedged RCCL rendezvous), or
# whose heartbeat sidecar has gone silent for HEARTBEAT_TIMEOUT seconds
# (only enforced once a first heartbeat was seen; 0 = disabled).
TASK_STALL_TIMEOUT = from_conf("TASK_STALL_TIMEOUT", 0, float)
HEARTBEAT_TIMEOUT = from_conf("HEARTBEAT_TIMEOUT", 600, float)

# --- datastore ---------------------------------------------------------------
DATASTORE_LOCAL_DIR = from_conf("DATASTORE_LOCAL_DIR", ".mfx")
DEFAULT_DATASTORE = from_conf("DEFAULT_DATASTORE", "local")
DEFAULT_METADATA = from_conf("DEFAULT_METADATA", "local")
# CAS compression policy: blobs >= this size skip gzip (tensor shards etc.
# are incompressible and HBM-sized; see SURVEY §2.2 MI355X note)
CAS_COMPRESS_MAX_SIZE = from_conf("CAS_COMPRESS_MAX_SIZE", 1 << 22, int)
CAS_GZIP_LEVEL = from_conf("CAS_GZIP_LEVEL", 3, int)
# use the native C++ CAS engine when available
CAS_NATIVE = from_conf("CAS_NATIVE", 1, int)

# --- gang scheduler / GPU ----------------------------------------------------
GANG_MASTER_ADDR = from_conf("GANG_MASTER_ADDR", "127.0.0.1")
# exit code contract (reference: runtime.py:43): a task exiting with this
# code is never retried
EXIT_DISALLOW_RETRY = from_conf("EXIT_DISALLOW_RETRY", 202, int)

DEFAULT_NAMESPACE_USER = os.environ.get("USER", "user")
