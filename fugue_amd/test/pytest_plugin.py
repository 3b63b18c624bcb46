"""pytest plugin hooks: enable with ``pytest_plugins =
["fugue_amd.test.pytest_plugin"]`` in conftest (or install the package
with a ``pytest11`` entry point).

Reference parity: ``fugue/test/plugins.py`` ini-driven backend conf —
lines in the ``fugue_test_conf`` ini section become engine conf; keys
prefixed with a backend name apply only to that backend::

    [pytest]
    fugue_test_conf =
        fugue.workflow.concurrency = 4
        hip.fugue.hip.broadcast_threshold_bytes = 1024
"""
from typing import Any


def _parse_value(v: str) -> Any:
    low = v.strip()
    if low.lower() in ("true", "false"):
        return low.lower() == "true"
    for cast in (int, float):
        try:
            return cast(low)
        except ValueError:
            pass
    return low


def pytest_addoption(parser: Any) -> None:
    parser.addini(
        "fugue_test_conf",
        "engine configuration lines for fugue test backends",
        type="linelist",
        default=[],
    )


def pytest_configure(config: Any) -> None:
    from fugue_amd.test.plugins import set_global_test_conf

    conf = {}
    for line in config.getini("fugue_test_conf"):
        if "=" in line:
            k, v = line.split("=", 1)
            conf[k.strip()] = _parse_value(v)
    set_global_test_conf(conf)
