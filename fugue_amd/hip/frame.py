"""Device-resident columnar frame: Arrow-schema'd columns in HBM as torch
tensors.

The MI355X analog of the reference's backend-native frames (SURVEY.md §2.2
``fugue_ray``'s Arrow-batch blocks are the closest reference): each column
is a contiguous device buffer + optional validity mask; strings are
(offsets, bytes) pairs.  288 GB HBM3E per GPU means frames are kept
resident; host round-trips happen only at UDF boundaries and IO.
"""
from typing import Any, Dict, Iterable, List, Optional, Tuple

import numpy as np
import pandas as pd
import pyarrow as pa
import torch

from fugue_amd.dataframe.dataframe import DataFrame, LocalBoundedDataFrame
from fugue_amd.dataframe.pandas_dataframe import PandasDataFrame
from fugue_amd.exceptions import (
    FugueDataFrameEmptyError,
    FugueDataFrameInitError,
    FugueDataFrameOperationError,
)
from fugue_amd.schema import Schema

_PA_TO_TORCH = {
    pa.int8(): torch.int8,
    pa.int16(): torch.int16,
    pa.int32(): torch.int32,
    pa.int64(): torch.int64,
    pa.float32(): torch.float32,
    pa.float64(): torch.float64,
    pa.bool_(): torch.bool,
}


def _torch_dtype_for(tp: pa.DataType) -> torch.dtype:
    if tp in _PA_TO_TORCH:
        return _PA_TO_TORCH[tp]
    if pa.types.is_timestamp(tp) or pa.types.is_date(tp):
        return torch.int64
    raise FugueDataFrameInitError(f"unsupported device column type {tp}")


def supported_device_type(tp: pa.DataType) -> bool:
    return (
        tp in _PA_TO_TORCH
        or pa.types.is_timestamp(tp)
        or pa.types.is_date(tp)
        or pa.types.is_string(tp)
        or pa.types.is_large_string(tp)
        or pa.types.is_binary(tp)
        or pa.types.is_large_binary(tp)
    )


class DeviceColumn:
    """One column in HBM: ``data`` tensor + optional validity mask
    (True = valid)."""

    def __init__(
        self,
        data: torch.Tensor,
        valid: Optional[torch.Tensor],
        pa_type: pa.DataType,
    ):
        self.data = data
        self.valid = valid
        self.pa_type = pa_type

    def __len__(self) -> int:
        return int(self.data.numel())

    @property
    def has_nulls(self) -> bool:
        return self.valid is not None

    @property
    def is_string(self) -> bool:
        return False

    def gather(self, idx: torch.Tensor) -> "DeviceColumn":
        return DeviceColumn(
            self.data.index_select(0, idx),
            None if self.valid is None else self.valid.index_select(0, idx),
            self.pa_type,
        )

    def slice(self, start: int, length: int) -> "DeviceColumn":
        return DeviceColumn(
            self.data[start : start + length],
            None if self.valid is None else self.valid[start : start + length],
            self.pa_type,
        )

    def concat_with(self, others: List["DeviceColumn"]) -> "DeviceColumn":
        cols = [self] + others
        data = torch.cat([c.data for c in cols])
        if any(c.valid is not None for c in cols):
            valid = torch.cat(
                [
                    c.valid
                    if c.valid is not None
                    else torch.ones(len(c), dtype=torch.bool, device=c.data.device)
                    for c in cols
                ]
            )
        else:
            valid = None
        return DeviceColumn(data, valid, self.pa_type)

    # --- conversion ---------------------------------------------------- #
    @staticmethod
    def from_arrow(arr: pa.ChunkedArray, tp: pa.DataType, device: str) -> "DeviceColumn":
        if isinstance(arr, pa.ChunkedArray):
            arr = arr.combine_chunks()
        if pa.types.is_string(tp) or pa.types.is_large_string(tp):
            return StringDeviceColumn.from_arrow_strings(arr, device)
        if pa.types.is_binary(tp) or pa.types.is_large_binary(tp):
            return StringDeviceColumn.from_arrow_strings(
                arr, device, pa_type=pa.binary()
            )
        np_arr = arr.to_numpy(zero_copy_only=False)
        valid_t: Optional[torch.Tensor] = None
        if arr.null_count > 0:
            valid_np = ~np.asarray(arr.is_null())
            valid_t = torch.from_numpy(valid_np).to(device)
            if pa.types.is_floating(tp):
                np_arr = np.nan_to_num(np_arr, nan=0.0)
            elif np_arr.dtype == np.dtype("object") or np_arr.dtype.kind == "f":
                # ints with nulls come back as float; fill and cast
                np_arr = np.nan_to_num(np_arr.astype("float64"), nan=0.0)
                np_arr = np_arr.astype(_np_dtype_for(tp))
        if pa.types.is_timestamp(tp):
            np_arr = np_arr.astype("datetime64[us]").astype("int64")
        elif pa.types.is_date(tp):
            np_arr = np_arr.astype("datetime64[D]").astype("int64")
        elif np_arr.dtype != _np_dtype_for(tp):
            np_arr = np_arr.astype(_np_dtype_for(tp))
        np_arr = np.ascontiguousarray(np_arr)
        if not np_arr.flags.writeable:
            np_arr = np_arr.copy()
        data = torch.from_numpy(np_arr).to(device)
        return DeviceColumn(data, valid_t, tp)

    def to_arrow(self) -> pa.Array:
        data_np = self.data.cpu().numpy()
        mask = None
        if self.valid is not None:
            mask = ~self.valid.cpu().numpy()
        if pa.types.is_timestamp(self.pa_type):
            return pa.Array.from_pandas(
                data_np.astype("datetime64[us]"), mask=mask, type=self.pa_type
            )
        if pa.types.is_date(self.pa_type):
            return pa.Array.from_pandas(
                data_np.astype("datetime64[D]"), mask=mask, type=self.pa_type
            )
        return pa.Array.from_pandas(data_np, mask=mask, type=self.pa_type)


def _np_dtype_for(tp: pa.DataType):
    return np.dtype(tp.to_pandas_dtype()) if tp not in (pa.bool_(),) else np.dtype("bool")


class StringDeviceColumn(DeviceColumn):
    """UTF-8 strings in HBM: int64 offsets [n+1] + uint8 bytes."""

    def __init__(
        self,
        offsets: torch.Tensor,
        bytes_: torch.Tensor,
        valid: Optional[torch.Tensor],
        pa_type: Optional[pa.DataType] = None,
    ):
        # ``data`` property is the offsets tensor (n+1 elements);
        # pa_type selects string (default) vs binary payloads — the
        # HBM layout (offsets + byte buffer) is identical
        super().__init__(offsets, valid, pa_type or pa.string())
        self.offsets = offsets
        self.bytes = bytes_

    def __len__(self) -> int:
        return int(self.offsets.numel()) - 1

    @property
    def is_string(self) -> bool:
        return True

    def gather(self, idx: torch.Tensor) -> "StringDeviceColumn":
        lengths = self.offsets[1:] - self.offsets[:-1]
        new_len = lengths.index_select(0, idx)
        new_offsets = torch.zeros(
            idx.numel() + 1, dtype=torch.int64, device=idx.device
        )
        torch.cumsum(new_len, 0, out=new_offsets[1:])
        # byte gather: source ranges → flat index
        starts = self.offsets.index_select(0, idx)
        total = int(new_offsets[-1].item()) if idx.numel() > 0 else 0
        if total > 0:
            seq = torch.arange(total, device=idx.device)
            row = torch.searchsorted(new_offsets[1:], seq, right=True)
            src_idx = starts.index_select(0, row) + (
                seq - new_offsets.index_select(0, row)
            )
            new_bytes = self.bytes.index_select(0, src_idx)
        else:
            new_bytes = torch.empty(0, dtype=torch.uint8, device=idx.device)
        return StringDeviceColumn(
            new_offsets,
            new_bytes,
            None if self.valid is None else self.valid.index_select(0, idx),
            pa_type=self.pa_type,
        )

    def slice(self, start: int, length: int) -> "StringDeviceColumn":
        offs = self.offsets[start : start + length + 1]
        b0 = int(offs[0].item())
        b1 = int(offs[-1].item())
        return StringDeviceColumn(
            offs - b0,
            self.bytes[b0:b1],
            None if self.valid is None else self.valid[start : start + length],
            pa_type=self.pa_type,
        )

    def concat_with(self, others: List["DeviceColumn"]) -> "StringDeviceColumn":
        cols: List[StringDeviceColumn] = [self] + others  # type: ignore
        device = self.offsets.device
        bytes_ = torch.cat([c.bytes for c in cols])
        lengths = torch.cat([c.offsets[1:] - c.offsets[:-1] for c in cols])
        offsets = torch.zeros(
            int(lengths.numel()) + 1, dtype=torch.int64, device=device
        )
        torch.cumsum(lengths, 0, out=offsets[1:])
        if any(c.valid is not None for c in cols):
            valid = torch.cat(
                [
                    c.valid
                    if c.valid is not None
                    else torch.ones(len(c), dtype=torch.bool, device=device)
                    for c in cols
                ]
            )
        else:
            valid = None
        return StringDeviceColumn(offsets, bytes_, valid)

    @staticmethod
    def from_arrow_strings(
        arr: pa.Array, device: str, pa_type: Optional[pa.DataType] = None
    ) -> "StringDeviceColumn":
        is_bin = pa_type is not None and pa.types.is_binary(pa_type)
        arr = arr.cast(pa.large_binary() if is_bin else pa.large_string())
        buffers = arr.buffers()
        # buffers: [validity, offsets(int64), data]
        offsets_np = np.frombuffer(
            buffers[1], dtype=np.int64, count=len(arr) + 1 + arr.offset
        )[arr.offset :].copy()
        base = offsets_np[0]
        offsets_np = offsets_np - base
        if buffers[2] is not None and len(buffers[2]) > 0:
            data_np = np.frombuffer(buffers[2], dtype=np.uint8)[
                base : base + offsets_np[-1]
            ].copy()
        else:
            data_np = np.empty(0, dtype=np.uint8)
        valid_t: Optional[torch.Tensor] = None
        if arr.null_count > 0:
            valid_np = ~np.asarray(arr.is_null())
            valid_t = torch.from_numpy(valid_np).to(device)
        return StringDeviceColumn(
            torch.from_numpy(offsets_np).to(device),
            torch.from_numpy(data_np).to(device),
            valid_t,
            pa_type=pa_type,
        )

    def to_arrow(self) -> pa.Array:
        is_bin = pa.types.is_binary(self.pa_type) or pa.types.is_large_binary(
            self.pa_type
        )
        big = pa.large_binary() if is_bin else pa.large_string()
        final = pa.binary() if is_bin else pa.string()
        offsets_np = self.offsets.cpu().numpy()
        bytes_np = self.bytes.cpu().numpy()
        arr = pa.Array.from_buffers(
            big,
            len(self),
            [
                None,
                pa.py_buffer(offsets_np.tobytes()),
                pa.py_buffer(bytes_np.tobytes()),
            ],
        )
        arr = arr.cast(final)
        if self.valid is not None:
            mask = self.valid.cpu().numpy()
            # rebuild with nulls
            py = arr.to_pylist()
            py = [v if m else None for v, m in zip(py, mask)]
            arr = pa.array(py, type=final)
        return arr


class SpilledDataFrame(LocalBoundedDataFrame):
    """A persisted shard spilled from HBM to (pinned) host DRAM.

    ``restore()`` re-uploads with ``hipMemcpyAsync`` (pinned source) to
    rebuild the device frame; the engine's ``to_df`` restores
    transparently when an op touches the frame."""

    def __init__(self, src: "HipDataFrame"):
        self._device = src.device
        self._host_cols: Dict[str, DeviceColumn] = {}
        pin = torch.cuda.is_available()

        def _to_host(t: torch.Tensor) -> torch.Tensor:
            if t.device.type == "cpu":
                return t
            host = torch.empty(t.shape, dtype=t.dtype, pin_memory=pin)
            host.copy_(t, non_blocking=False)
            return host

        for name, c in src.columns_map.items():
            valid = None if c.valid is None else _to_host(c.valid)
            if isinstance(c, StringDeviceColumn):
                self._host_cols[name] = StringDeviceColumn(
                    _to_host(c.offsets), _to_host(c.bytes), valid,
                    pa_type=c.pa_type,
                )
            else:
                self._host_cols[name] = DeviceColumn(
                    _to_host(c.data), valid, c.pa_type
                )
        super().__init__(src.schema)

    def restore(self) -> "HipDataFrame":
        dev = self._device

        def _up(t: torch.Tensor) -> torch.Tensor:
            return t.to(dev, non_blocking=True)

        cols: Dict[str, DeviceColumn] = {}
        for name, c in self._host_cols.items():
            valid = None if c.valid is None else _up(c.valid)
            if isinstance(c, StringDeviceColumn):
                cols[name] = StringDeviceColumn(
                    _up(c.offsets), _up(c.bytes), valid, pa_type=c.pa_type
                )
            else:
                cols[name] = DeviceColumn(_up(c.data), valid, c.pa_type)
        if dev.startswith("cuda") and torch.cuda.is_available():
            torch.cuda.synchronize()
        return HipDataFrame.from_columns(cols, self.schema, dev)

    # --- DataFrame interface (delegates to a temporary restore-less view) --
    @property
    def native(self) -> Dict[str, DeviceColumn]:
        return self._host_cols

    def native_as_df(self) -> "SpilledDataFrame":
        return self

    @property
    def empty(self) -> bool:
        return self.count() == 0

    def count(self) -> int:
        if len(self._host_cols) == 0:
            return 0
        return len(next(iter(self._host_cols.values())))

    def _as_host_frame(self) -> "HipDataFrame":
        return HipDataFrame.from_columns(self._host_cols, self.schema, "cpu")

    def peek_array(self) -> List[Any]:
        return self._as_host_frame().peek_array()

    def as_arrow(self, type_safe: bool = False) -> pa.Table:
        return self._as_host_frame().as_arrow()

    def as_pandas(self) -> pd.DataFrame:
        return self.as_arrow().to_pandas()

    def as_local_bounded(self) -> LocalBoundedDataFrame:
        from fugue_amd.dataframe.arrow_dataframe import ArrowDataFrame

        res = ArrowDataFrame(self.as_arrow())
        if self.has_metadata:
            res.reset_metadata(self.metadata)
        return res

    def as_array(self, columns=None, type_safe: bool = False) -> List[Any]:
        return self._as_host_frame().as_array(columns, type_safe=type_safe)

    def as_array_iterable(self, columns=None, type_safe: bool = False):
        yield from self.as_array(columns, type_safe=type_safe)

    def _drop_cols(self, cols: List[str]) -> DataFrame:
        return self._as_host_frame()._drop_cols(cols)

    def _select_cols(self, cols: List[Any]) -> DataFrame:
        return self._as_host_frame()._select_cols(cols)

    def rename(self, columns: Dict[str, str]) -> DataFrame:
        return self._as_host_frame().rename(columns)

    def alter_columns(self, columns: Any) -> DataFrame:
        return self._as_host_frame().alter_columns(columns)

    def head(self, n: int, columns=None) -> LocalBoundedDataFrame:
        return self._as_host_frame().head(n, columns)


class HipDataFrame(LocalBoundedDataFrame):
    """A device-resident bounded frame (one shard; the distributed engine
    holds one HipDataFrame per rank)."""

    def __init__(
        self,
        df: Any = None,
        schema: Any = None,
        device: Optional[str] = None,
        _columns: Optional[Dict[str, DeviceColumn]] = None,
    ):
        self._device = device or "cuda:0"
        if _columns is not None:
            self._cols = _columns
            super().__init__(schema)
            return
        try:
            if df is None:
                schema = Schema(schema).assert_not_empty()
                table = schema.create_empty_arrow()
            elif isinstance(df, pa.Table):
                table = df
                if schema is None:
                    schema = Schema(df.schema)
                else:
                    schema = Schema(schema)
                    if df.schema != schema.pa_schema:
                        table = df.select(schema.names).cast(schema.pa_schema)
            elif isinstance(df, pd.DataFrame):
                schema = Schema(schema) if schema is not None else None
                if schema is None:
                    table = pa.Table.from_pandas(
                        df.reset_index(drop=True), preserve_index=False
                    )
                    schema = Schema(table.schema)
                else:
                    from fugue_amd.utils.pandas_like import cast_pandas

                    table = pa.Table.from_pandas(
                        cast_pandas(df.reset_index(drop=True), schema),
                        schema=schema.pa_schema,
                        preserve_index=False,
                    )
            elif isinstance(df, DataFrame):
                schema = df.schema if schema is None else Schema(schema)
                table = df.as_arrow()
            elif isinstance(df, Iterable):
                schema = Schema(schema).assert_not_empty()
                rows = [
                    {c: row[i] for i, c in enumerate(schema.names)} for row in df
                ]
                table = pa.Table.from_pylist(rows, schema=schema.pa_schema)
            else:
                raise ValueError(f"{type(df)} is incompatible with HipDataFrame")
        except FugueDataFrameInitError:
            raise
        except Exception as e:
            raise FugueDataFrameInitError(str(e)) from e
        self._cols = {
            f.name: DeviceColumn.from_arrow(table.column(f.name), f.type, self._device)
            for f in Schema(schema).fields
        }
        super().__init__(schema)

    # --- construction helpers ------------------------------------------- #
    @staticmethod
    def from_columns(
        columns: Dict[str, DeviceColumn], schema: Schema, device: str
    ) -> "HipDataFrame":
        return HipDataFrame(schema=schema, device=device, _columns=columns)

    @property
    def device(self) -> str:
        return self._device

    @property
    def columns_map(self) -> Dict[str, DeviceColumn]:
        return self._cols

    def col(self, name: str) -> DeviceColumn:
        return self._cols[name]

    @property
    def native(self) -> Dict[str, DeviceColumn]:
        return self._cols

    def native_as_df(self) -> "HipDataFrame":
        return self

    @property
    def empty(self) -> bool:
        return self.count() == 0

    def count(self) -> int:
        if len(self._cols) == 0:
            return 0
        return len(next(iter(self._cols.values())))

    def num_bytes(self) -> int:
        total = 0
        for c in self._cols.values():
            total += c.data.numel() * c.data.element_size()
            if isinstance(c, StringDeviceColumn):
                total += c.bytes.numel()
            if c.valid is not None:
                total += c.valid.numel()
        return total

    def peek_array(self) -> List[Any]:
        if self.empty:
            raise FugueDataFrameEmptyError("dataframe is empty")
        head = self.gather_rows(
            torch.zeros(1, dtype=torch.int64, device=self._device)
        )
        return list(head.as_arrow().to_pylist()[0].values())

    # --- device ops ------------------------------------------------------ #
    def gather_rows(self, idx: torch.Tensor) -> "HipDataFrame":
        cols = self._gather_cols_fused(idx)
        return HipDataFrame.from_columns(cols, self.schema, self._device)

    def _gather_cols_fused(self, idx: torch.Tensor) -> Dict[str, "DeviceColumn"]:
        """Row gather across all columns: flat columns (data + validity
        tensors) go through one fused HIP kernel per element width
        (``gather_cols_kernel``); strings keep their offsets-aware
        path."""
        flat_items = [
            (n, c)
            for n, c in self._cols.items()
            if not isinstance(c, StringDeviceColumn)
        ]
        if idx.device.type != "cuda" or len(flat_items) <= 1:
            return {n: c.gather(idx) for n, c in self._cols.items()}
        from fugue_amd.hip.ext import get_ext

        ext = get_ext()
        tensors: List[torch.Tensor] = []
        slots: List[Tuple[str, str]] = []  # (col name, "data"|"valid")
        for n, c in flat_items:
            tensors.append(c.data.contiguous())
            slots.append((n, "data"))
            if c.valid is not None:
                tensors.append(c.valid.contiguous())
                slots.append((n, "valid"))
        gathered: Dict[Tuple[str, str], torch.Tensor] = {}
        for i in range(0, len(tensors), 16):
            outs = ext.gather_columns(idx, tensors[i : i + 16])
            for slot, out in zip(slots[i : i + 16], outs):
                gathered[slot] = out
        cols: Dict[str, DeviceColumn] = {}
        for n, c in self._cols.items():
            if isinstance(c, StringDeviceColumn):
                cols[n] = c.gather(idx)
            else:
                cols[n] = DeviceColumn(
                    gathered[(n, "data")],
                    gathered.get((n, "valid")),
                    c.pa_type,
                )
        return cols

    def slice_rows(self, start: int, length: int) -> "HipDataFrame":
        if start == 0 and length >= self.count():
            return self  # preserve column tensor identity (sizing memos)
        cols = {n: c.slice(start, length) for n, c in self._cols.items()}
        return HipDataFrame.from_columns(cols, self.schema, self._device)

    def concat_with(self, others: List["HipDataFrame"]) -> "HipDataFrame":
        others = [o for o in others if o.count() > 0]
        if len(others) == 0:
            return self
        if self.count() == 0 and len(others) == 1:
            return others[0]
        cols = {
            n: c.concat_with([o.col(n) for o in others])
            for n, c in self._cols.items()
        }
        return HipDataFrame.from_columns(cols, self.schema, self._device)

    # --- conversion ------------------------------------------------------ #
    def as_arrow(self, type_safe: bool = False) -> pa.Table:
        arrays = [self._cols[f.name].to_arrow() for f in self.schema.fields]
        return pa.Table.from_arrays(arrays, schema=self.schema.pa_schema)

    def as_pandas(self) -> pd.DataFrame:
        return self.as_arrow().to_pandas()

    def as_local_bounded(self) -> LocalBoundedDataFrame:
        from fugue_amd.dataframe.arrow_dataframe import ArrowDataFrame

        res = ArrowDataFrame(self.as_arrow())
        if self.has_metadata:
            res.reset_metadata(self.metadata)
        return res

    def as_array(
        self, columns: Optional[List[str]] = None, type_safe: bool = False
    ) -> List[Any]:
        tbl = self.as_arrow()
        if columns is not None:
            tbl = tbl.select(columns)
        return [list(d.values()) for d in tbl.to_pylist()]

    def as_array_iterable(
        self, columns: Optional[List[str]] = None, type_safe: bool = False
    ) -> Iterable[Any]:
        yield from self.as_array(columns, type_safe=type_safe)

    # --- frame ops -------------------------------------------------------- #
    def _drop_cols(self, cols: List[str]) -> "HipDataFrame":
        schema = self.schema.exclude(cols)
        return HipDataFrame.from_columns(
            {n: self._cols[n] for n in schema.names}, schema, self._device
        )

    def _select_cols(self, cols: List[Any]) -> "HipDataFrame":
        schema = self.schema.extract(cols)
        return HipDataFrame.from_columns(
            {n: self._cols[n] for n in schema.names}, schema, self._device
        )

    def rename(self, columns: Dict[str, str]) -> "HipDataFrame":
        try:
            schema = self.schema.rename(columns)
        except Exception as e:
            raise FugueDataFrameOperationError(str(e)) from e
        new_cols = {columns.get(n, n): c for n, c in self._cols.items()}
        return HipDataFrame.from_columns(
            {n: new_cols[n] for n in schema.names}, schema, self._device
        )

    def alter_columns(self, columns: Any) -> "HipDataFrame":
        schema = self._get_altered_schema(columns)
        if schema == self.schema:
            return self
        # cast through the host frame (pandas casting rules: str(datetime)
        # formatting, case-insensitive str->bool — reference triad
        # semantics); device-native casts later
        from fugue_amd.dataframe.arrow_dataframe import ArrowDataFrame

        altered = ArrowDataFrame(self.as_arrow()).alter_columns(columns)
        return HipDataFrame(
            altered.as_arrow(), altered.schema, device=self._device
        )

    def head(
        self, n: int, columns: Optional[List[str]] = None
    ) -> LocalBoundedDataFrame:
        sub = self if columns is None else self._select_cols(columns)
        n = min(n, sub.count())
        from fugue_amd.dataframe.arrow_dataframe import ArrowDataFrame

        return ArrowDataFrame(sub.slice_rows(0, n).as_arrow())
