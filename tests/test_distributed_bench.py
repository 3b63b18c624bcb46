"""World-4/8 gloo runs of the DRIVER'S exact launch shape:
``python -m torch.distributed.run --nnodes=1 --nproc-per-node N
--master-addr 127.0.0.1 bench.py/q3_bench.py`` on tiny synthetic data.
This exercises the whole distributed pipeline (shuffles, partial-merge
aggregation, co-shuffled joins, FugueSQL planning per rank) the way the
round-end scaling bench will, with no GPU."""
import json
import os
import socket
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _torchrun(nproc: int, script: str, *args: str, timeout: int = 420) -> str:
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    env.pop("LOCAL_RANK", None)
    env.pop("MASTER_ADDR", None)
    env.pop("MASTER_PORT", None)
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", f"--nproc-per-node={nproc}",
        "--master-addr", "127.0.0.1",
        "--master-port", str(_free_port()),
        os.path.join(REPO, script), *args,
    ]
    last = None
    for attempt in range(2):  # gloo rendezvous can flake under load
        res = subprocess.run(
            cmd,
            cwd=REPO, env=env, capture_output=True, text=True,
            timeout=timeout,
        )
        if res.returncode == 0:
            return res.stdout
        last = res
        cmd[cmd.index("--master-port") + 1] = str(_free_port())
    assert last is not None and last.returncode == 0, (
        f"torchrun failed\nSTDOUT:\n{last.stdout[-4000:]}\n"
        f"STDERR:\n{last.stderr[-4000:]}"
    )
    return last.stdout


def _last_json(out: str) -> dict:
    for line in reversed(out.strip().splitlines()):
        line = line.strip()
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{out[-2000:]}")


@pytest.mark.parametrize("nproc", [4, 8])
def test_bench_pipeline_world_n(nproc):
    out = _torchrun(
        nproc, "bench.py", "--rows", "20000", "--steps", "2", "--warmup", "1"
    )
    rec = _last_json(out)
    assert rec["n_gpus"] == nproc
    assert rec["config"]["global_rows"] == 20000 * nproc
    # 1M-group space over tiny rows: every row is ~unique, join keeps
    # only keys < DIM_ROWS; sanity: some rows survive
    assert rec["config"]["out_rows"] > 0


@pytest.mark.parametrize("nproc", [4])
def test_q3_pipeline_world_n(nproc):
    out = _torchrun(
        nproc, os.path.join("benchmarks", "q3_bench.py"),
        "--sf", "0.01", "--steps", "2", "--warmup", "1",
    )
    rec = _last_json(out)
    assert rec["n_gpus"] == nproc
    assert rec["value"] > 0
