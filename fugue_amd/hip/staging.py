"""Device↔host staging for the UDF boundary.

North-star requirement: ``map_engine.map_dataframe`` runs user
pandas/pyarrow UDFs via pinned ``hipMemcpyAsync`` device↔host staging on a
side stream, overlapped with the next batch's transfer — a 2-deep
double-buffered pipeline (copy batch k+1 while the UDF chews batch k).

The fast path applies to numeric/bool/datetime columns (zero-copy
numpy→pandas on the host side); frames containing strings or validity
masks take the arrow conversion path (correct, not overlapped).
"""
from typing import Dict, Iterator, List, Optional, Tuple

import numpy as np
import pandas as pd
import torch

from fugue_amd.hip.frame import HipDataFrame, StringDeviceColumn
from fugue_amd.schema import Schema


def can_fast_stage(df: HipDataFrame) -> bool:
    for c in df.columns_map.values():
        if isinstance(c, StringDeviceColumn) or c.valid is not None:
            return False
    return True


def staged_pandas_batches(
    df: HipDataFrame,
    bounds: List[int],
    target_batch_rows: int = 4_000_000,
) -> Iterator[Tuple[int, int, pd.DataFrame]]:
    """Yield (start_group_idx, end_group_idx, pandas_batch) where the
    batch holds the rows of groups [start, end) — copied D2H on a side
    stream one batch ahead of consumption.

    ``bounds`` is the group start-offsets list with a trailing total-row
    sentinel (len = n_groups + 1).
    """
    n_groups = len(bounds) - 1
    if n_groups <= 0:
        return
    use_cuda = df.device.startswith("cuda") and torch.cuda.is_available()
    # split groups into row-bounded batches
    batches: List[Tuple[int, int]] = []
    g = 0
    while g < n_groups:
        h = g + 1
        while h < n_groups and bounds[h + 1] - bounds[g] <= target_batch_rows:
            h += 1
        batches.append((g, h))
        g = h
    names = list(df.columns_map.keys())
    cols = [df.col(n) for n in names]
    if not use_cuda:
        for g0, g1 in batches:
            lo, hi = bounds[g0], bounds[g1]
            data = {n: c.data[lo:hi].numpy() for n, c in zip(names, cols)}
            yield g0, g1, _to_pandas(data, df.schema)
        return

    side = torch.cuda.Stream()

    def start_copy(b: int):
        g0, g1 = batches[b]
        lo, hi = bounds[g0], bounds[g1]
        host: Dict[str, torch.Tensor] = {}
        with torch.cuda.stream(side):
            for n, c in zip(names, cols):
                pinned = torch.empty(
                    hi - lo, dtype=c.data.dtype, pin_memory=True
                )
                pinned.copy_(c.data[lo:hi], non_blocking=True)
                host[n] = pinned
            ev = torch.cuda.Event()
            ev.record(side)
        return host, ev

    pending = start_copy(0)
    for b in range(len(batches)):
        host, ev = pending
        if b + 1 < len(batches):
            nxt = start_copy(b + 1)
        else:
            nxt = None
        ev.synchronize()
        data = {n: t.numpy() for n, t in host.items()}
        g0, g1 = batches[b]
        yield g0, g1, _to_pandas(data, df.schema)
        if nxt is None:
            return
        pending = nxt


def _to_pandas(data: Dict[str, np.ndarray], schema: Schema) -> pd.DataFrame:
    import pyarrow as pa

    out = {}
    for f in schema.fields:
        arr = data[f.name]
        if pa.types.is_timestamp(f.type):
            arr = arr.astype("datetime64[us]")
        elif pa.types.is_date(f.type):
            arr = arr.astype("datetime64[D]")
        out[f.name] = arr
    return pd.DataFrame(out, columns=schema.names)
