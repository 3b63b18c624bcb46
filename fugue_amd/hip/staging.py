"""Device↔host staging for the UDF boundary.

North-star requirement: ``map_engine.map_dataframe`` runs user
pandas/pyarrow UDFs via pinned ``hipMemcpyAsync`` device↔host staging on a
side stream, overlapped with the next batch's transfer — a 2-deep
double-buffered pipeline (copy batch k+1 while the UDF chews batch k).

The fast path covers numeric/bool/datetime columns, validity-masked
numeric columns (the mask is staged alongside the data) and string
columns (offset + byte buffers staged; host-side reassembly is a
zero-copy arrow ``from_buffers``).  Nested/decimal columns take the
arrow conversion path (correct, not overlapped).
"""
from typing import Any, Dict, Iterator, List, Optional, Tuple

import numpy as np
import pandas as pd
import pyarrow as pa
import torch

from fugue_amd.hip.frame import DeviceColumn, HipDataFrame, StringDeviceColumn
from fugue_amd.schema import Schema


def can_fast_stage(df: HipDataFrame) -> bool:
    # flat data (+ optional mask) and offsets+bytes (string/binary)
    # layouts are stageable; nested/decimal columns take the arrow path
    from fugue_amd.hip.frame import supported_device_type

    return all(supported_device_type(f.type) for f in df.schema.fields)


class _ColStager:
    """Per-column staging: issues async pinned D2H copies for a row
    range and reassembles the host pandas column."""

    def __init__(self, c: DeviceColumn, pa_type: pa.DataType):
        self.c = c
        self.pa_type = pa_type
        self.is_str = isinstance(c, StringDeviceColumn)
        if self.is_str:
            # offsets fetched once up front: byte ranges per batch are
            # then known host-side without extra syncs
            self.offsets_np = c.offsets.cpu().numpy()

    def _pinned_like(self, src: torch.Tensor) -> torch.Tensor:
        return torch.empty(src.shape, dtype=src.dtype, pin_memory=True)

    def start(self, lo: int, hi: int, use_cuda: bool) -> Dict[str, Any]:
        out: Dict[str, Any] = {}

        def cp(src: torch.Tensor) -> torch.Tensor:
            if not use_cuda:
                return src
            dst = self._pinned_like(src)
            dst.copy_(src, non_blocking=True)
            return dst

        if self.is_str:
            b0 = int(self.offsets_np[lo])
            b1 = int(self.offsets_np[hi])
            out["bytes"] = cp(self.c.bytes[b0:b1])
            out["off"] = self.offsets_np[lo : hi + 1] - b0
        else:
            out["data"] = cp(self.c.data[lo:hi])
        if self.c.valid is not None:
            out["valid"] = cp(self.c.valid[lo:hi])
        return out

    def finish(self, staged: Dict[str, Any]) -> pd.Series:
        valid_np = (
            staged["valid"].numpy() if "valid" in staged else None
        )
        if self.is_str:
            n = len(staged["off"]) - 1
            is_bin = pa.types.is_binary(self.pa_type) or pa.types.is_large_binary(
                self.pa_type
            )
            big = pa.large_binary() if is_bin else pa.large_string()
            bufs = [
                None if valid_np is None else pa.array(valid_np).buffers()[1],
                pa.py_buffer(np.ascontiguousarray(staged["off"]).tobytes()),
                pa.py_buffer(staged["bytes"].numpy().tobytes()),
            ]
            arr = pa.Array.from_buffers(big, n, bufs)
            return arr.to_pandas()
        data_np = staged["data"].numpy()
        if pa.types.is_timestamp(self.pa_type):
            data_np = data_np.astype("datetime64[us]")
        elif pa.types.is_date(self.pa_type):
            data_np = data_np.astype("datetime64[D]")
        if valid_np is None:
            return pd.Series(data_np)
        arr = pa.array(data_np, mask=~valid_np)
        return arr.to_pandas()


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
    names = [f.name for f in df.schema.fields]
    stagers = [
        _ColStager(df.col(f.name), f.type) for f in df.schema.fields
    ]

    def assemble(staged_cols: List[Dict[str, Any]]) -> pd.DataFrame:
        out = {
            n: st.finish(sc)
            for n, st, sc in zip(names, stagers, staged_cols)
        }
        return pd.DataFrame(out, columns=names)

    if not use_cuda:
        for g0, g1 in batches:
            lo, hi = bounds[g0], bounds[g1]
            staged = [st.start(lo, hi, False) for st in stagers]
            yield g0, g1, assemble(staged)
        return

    side = torch.cuda.Stream()

    def start_copy(b: int):
        g0, g1 = batches[b]
        lo, hi = bounds[g0], bounds[g1]
        with torch.cuda.stream(side):
            staged = [st.start(lo, hi, True) for st in stagers]
            ev = torch.cuda.Event()
            ev.record(side)
        return staged, ev

    pending = start_copy(0)
    for b in range(len(batches)):
        staged, ev = pending
        nxt = start_copy(b + 1) if b + 1 < len(batches) else None
        ev.synchronize()
        g0, g1 = batches[b]
        yield g0, g1, assemble(staged)
        if nxt is None:
            return
        pending = nxt
