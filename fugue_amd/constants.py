"""Global configuration keys and defaults (reference parity:
``fugue/constants.py``) plus the MI355X-specific ``fugue.hip.*`` keys."""
import threading
from typing import Any, Dict

KEYWORD_ROWCOUNT = "ROWCOUNT"
KEYWORD_CORECOUNT = "CONCURRENCY"
KEYWORD_PARALLELISM = "CONCURRENCY"

FUGUE_CONF_WORKFLOW_CONCURRENCY = "fugue.workflow.concurrency"
FUGUE_CONF_WORKFLOW_CHECKPOINT_PATH = "fugue.workflow.checkpoint.path"
FUGUE_CONF_WORKFLOW_AUTO_PERSIST = "fugue.workflow.auto_persist"
FUGUE_CONF_WORKFLOW_AUTO_PERSIST_VALUE = "fugue.workflow.auto_persist_value"
FUGUE_CONF_WORKFLOW_EXCEPTION_HIDE = "fugue.workflow.exception.hide"
FUGUE_CONF_WORKFLOW_EXCEPTION_INJECT = "fugue.workflow.exception.inject"
FUGUE_CONF_WORKFLOW_EXCEPTION_OPTIMIZE = "fugue.workflow.exception.optimize"
FUGUE_CONF_SQL_IGNORE_CASE = "fugue.sql.compile.ignore_case"
FUGUE_CONF_SQL_DIALECT = "fugue.sql.compile.dialect"
FUGUE_CONF_DEFAULT_PARTITIONS = "fugue.default.partitions"
FUGUE_CONF_CACHE_PATH = "fugue.workflow.cache.path"

# MI355X engine keys
FUGUE_HIP_CONF_DEVICE = "fugue.hip.device"
FUGUE_HIP_CONF_DEFAULT_PARTITIONS = "fugue.hip.default_partitions"
FUGUE_HIP_CONF_SPILL_THRESHOLD = "fugue.hip.spill_threshold_bytes"
FUGUE_HIP_CONF_SHUFFLE_CHUNK = "fugue.hip.shuffle_chunk_bytes"

FUGUE_ENTRYPOINT = "fugue.plugins"

FUGUE_COMPILE_TIME_CONFIGS = {
    FUGUE_CONF_WORKFLOW_EXCEPTION_HIDE,
    FUGUE_CONF_WORKFLOW_EXCEPTION_INJECT,
    FUGUE_CONF_WORKFLOW_EXCEPTION_OPTIMIZE,
    FUGUE_CONF_WORKFLOW_AUTO_PERSIST,
    FUGUE_CONF_WORKFLOW_AUTO_PERSIST_VALUE,
    FUGUE_CONF_SQL_IGNORE_CASE,
    FUGUE_CONF_SQL_DIALECT,
}

_FUGUE_FRAMEWORK_MODULE_PREFIXES = ("fugue_amd",)

_DEFAULT_GLOBAL_CONF: Dict[str, Any] = {
    FUGUE_CONF_WORKFLOW_CONCURRENCY: 1,
    FUGUE_CONF_WORKFLOW_AUTO_PERSIST: False,
    FUGUE_CONF_WORKFLOW_EXCEPTION_HIDE: "fugue_amd.,adagio.,six,pandas,numpy",
    FUGUE_CONF_WORKFLOW_EXCEPTION_INJECT: 3,
    FUGUE_CONF_WORKFLOW_EXCEPTION_OPTIMIZE: True,
    FUGUE_CONF_SQL_IGNORE_CASE: False,
    FUGUE_CONF_SQL_DIALECT: "spark",
}

_GLOBAL_CONF_LOCK = threading.RLock()
_FUGUE_GLOBAL_CONF: Dict[str, Any] = dict(_DEFAULT_GLOBAL_CONF)


def register_global_conf(conf: Dict[str, Any], on_dup: str = "overwrite") -> None:
    with _GLOBAL_CONF_LOCK:
        for k, v in conf.items():
            if k in _FUGUE_GLOBAL_CONF and on_dup == "throw":
                raise KeyError(f"global conf {k} already set")
            if k in _FUGUE_GLOBAL_CONF and on_dup == "ignore":
                continue
            _FUGUE_GLOBAL_CONF[k] = v


def get_global_conf() -> Dict[str, Any]:
    with _GLOBAL_CONF_LOCK:
        return dict(_FUGUE_GLOBAL_CONF)
