"""Column-expression evaluation on device frames (torch elementwise ops —
each lowers to a HIP kernel; fusion of the scalar pipeline comes from
torch's elementwise fuser, while the relational ops use the hand-written
kernels in ``csrc/relational.hip``)."""
from typing import Any, Dict, List, Optional, Tuple

import pyarrow as pa
import torch

from fugue_amd.column.expressions import (
    ColumnExpr,
    _BinaryOpExpr,
    _FuncExpr,
    _LiteralColumnExpr,
    _NamedColumnExpr,
    _NotOpExpr,
    _UnaryAggFuncExpr,
    _UnaryOpExpr,
)
from fugue_amd.hip.frame import DeviceColumn, HipDataFrame, StringDeviceColumn


class DeviceExprError(NotImplementedError):
    """Expression can't be evaluated on device → caller falls back."""


def eval_device_expr(
    expr: ColumnExpr, df: HipDataFrame
) -> Tuple[torch.Tensor, Optional[torch.Tensor]]:
    """Evaluate a non-aggregate expression; returns (data, valid_or_None)."""
    if isinstance(expr, _BinaryOpExpr):
        fused = try_fused_value(expr, df)
        if fused is not None:
            return fused
    if isinstance(expr, _LiteralColumnExpr):
        n = df.count()
        v = expr.value
        device = df.device
        if v is None:
            return (
                torch.zeros(n, dtype=torch.float64, device=device),
                torch.zeros(n, dtype=torch.bool, device=device),
            )
        if isinstance(v, bool):
            return torch.full((n,), v, dtype=torch.bool, device=device), None
        if isinstance(v, int):
            return torch.full((n,), v, dtype=torch.int64, device=device), None
        if isinstance(v, float):
            return torch.full((n,), v, dtype=torch.float64, device=device), None
        raise DeviceExprError(f"literal {v!r} not supported on device")
    if isinstance(expr, _NamedColumnExpr):
        c = df.col(expr.name)
        if isinstance(c, StringDeviceColumn):
            raise DeviceExprError("string expressions not supported on device")
        data = c.data
        if expr.as_type is not None:
            data = _cast(data, expr.as_type)
        return data, c.valid
    if isinstance(expr, _BinaryOpExpr) and expr.op in ("==", "!="):
        res = _try_string_equality(expr, df)
        if res is not None:
            return res
    if isinstance(expr, _NotOpExpr):
        d, v = eval_device_expr(expr.col, df)
        return ~d.to(torch.bool), v
    if isinstance(expr, _UnaryOpExpr):
        d, v = eval_device_expr(expr.col, df)
        if expr.op == "IS_NULL":
            if v is None:
                return torch.zeros_like(d, dtype=torch.bool), None
            return ~v, None
        if expr.op == "NOT_NULL":
            if v is None:
                return torch.ones_like(d, dtype=torch.bool), None
            return v, None
        if expr.op == "-":
            return -d, v
        raise DeviceExprError(f"unary {expr.op}")
    if isinstance(expr, _UnaryAggFuncExpr):
        raise DeviceExprError("aggregate in scalar context")
    if isinstance(expr, _BinaryOpExpr):
        ld, lv = eval_device_expr(expr.left, df)
        rd, rv = eval_device_expr(expr.right, df)
        valid = _merge_valid(lv, rv)
        op = expr.op
        if op == "&":
            # SQL three-valued logic approximated: null treated as False
            res = _as_bool(ld, lv) & _as_bool(rd, rv)
            return res, None
        if op == "|":
            res = _as_bool(ld, lv) | _as_bool(rd, rv)
            return res, None
        if op in ("+", "-", "*", "/"):
            if op == "/":
                ld = ld.to(torch.float64)
                rd = rd.to(torch.float64)
            res = {
                "+": torch.add,
                "-": torch.sub,
                "*": torch.mul,
                "/": torch.div,
            }[op](ld, rd)
            out = res
            if expr.as_type is not None:
                out = _cast(out, expr.as_type)
            return out, valid
        cmp = {
            "==": torch.eq,
            "!=": torch.ne,
            "<": torch.lt,
            "<=": torch.le,
            ">": torch.gt,
            ">=": torch.ge,
        }.get(op)
        if cmp is None:
            raise DeviceExprError(f"binary {op}")
        if ld.dtype != rd.dtype:
            common = torch.promote_types(ld.dtype, rd.dtype)
            ld = ld.to(common)
            rd = rd.to(common)
        res = cmp(ld, rd)
        if valid is not None:
            res = res & valid  # null comparison → false
        return res, None
    if isinstance(expr, _FuncExpr):
        fname = expr.func.upper()
        if fname == "COALESCE":
            out_d: Optional[torch.Tensor] = None
            out_v: Optional[torch.Tensor] = None
            for a in expr.args:
                d, v = eval_device_expr(a, df)
                if out_d is None:
                    out_d, out_v = d, v
                    continue
                if out_v is None:
                    break
                d = d.to(out_d.dtype)
                out_d = torch.where(out_v, out_d, d)
                out_v = out_v | (v if v is not None else torch.ones_like(out_v))
            if out_v is not None and bool(out_v.all().item()):
                out_v = None
            return out_d, out_v
        if fname == "LIKE":
            return _eval_like(expr, df)
        if fname == "CASE_WHEN":
            # args = [c1, v1, c2, v2, ..., else]; first match wins →
            # apply branches in reverse over torch.where
            args = expr.args
            out_d, out_v = eval_device_expr(args[-1], df)
            for i in range(len(args) - 2, 0, -2):
                cd, cv = eval_device_expr(args[i - 1], df)
                cond = _as_bool(cd, cv)
                vd, vv = eval_device_expr(args[i], df)
                if vd.dtype != out_d.dtype:
                    common = torch.promote_types(vd.dtype, out_d.dtype)
                    vd = vd.to(common)
                    out_d = out_d.to(common)
                out_d = torch.where(cond, vd, out_d)
                if out_v is None and vv is None:
                    new_v = None
                else:
                    ones = torch.ones_like(cond)
                    new_v = torch.where(
                        cond,
                        vv if vv is not None else ones,
                        out_v if out_v is not None else ones,
                    )
                out_v = new_v
            return out_d, out_v
        raise DeviceExprError(f"function {expr.func}")
    raise DeviceExprError(f"can't evaluate {expr} on device")


def _eval_like(
    expr: "_FuncExpr", df: HipDataFrame
) -> Tuple[torch.Tensor, Optional[torch.Tensor]]:
    """Device LIKE over the string column's flat byte buffer.

    Supported shapes (pattern ``p`` with no ``_``, no interior ``%``):
    exact, ``p%`` prefix, ``%p`` suffix, ``%p%`` contains.  Everything
    else falls back (DeviceExprError → pandas interpreter)."""
    col_e = expr.args[0]
    pat_e = expr.args[1]
    if not (
        isinstance(col_e, _NamedColumnExpr)
        and isinstance(pat_e, _LiteralColumnExpr)
        and isinstance(pat_e.value, str)
    ):
        raise DeviceExprError("LIKE shape not supported on device")
    c = df.col(col_e.name)
    if not isinstance(c, StringDeviceColumn):
        raise DeviceExprError("LIKE on non-string column")
    pat = pat_e.value
    prefix_any = pat.startswith("%")
    suffix_any = pat.endswith("%")
    core = pat.strip("%")
    if "_" in pat or "%" in core:
        # general pattern: segment matcher (greedy earliest-occurrence)
        return _eval_like_general(c, pat, torch.device(df.device))
    device = torch.device(df.device)
    n = len(c)
    lengths = c.offsets[1:] - c.offsets[:-1]
    kb = core.encode("utf-8")
    k = len(kb)
    if k == 0:
        # '%' / '%%' matches everything; '' matches only empty strings
        res = (
            torch.ones(n, dtype=torch.bool, device=device)
            if prefix_any or suffix_any
            else lengths == 0
        )
        return res, c.valid
    if int(c.bytes.numel()) == 0:
        # every row is the empty string; a non-empty literal never matches
        return torch.zeros(n, dtype=torch.bool, device=device), c.valid
    patt = torch.tensor(list(kb), dtype=torch.uint8, device=device)
    long_enough = lengths >= k
    if not prefix_any:  # 'core%' or exact: compare k bytes at row start
        starts = c.offsets[:-1]
        res = long_enough.clone()
        for j in range(k):
            idx = torch.clamp(
                starts + j, max=max(int(c.bytes.numel()) - 1, 0)
            )
            res = res & (c.bytes.index_select(0, idx) == patt[j])
        if not suffix_any:
            res = res & (lengths == k)
    elif not suffix_any:  # '%core': compare k bytes at row end
        ends = c.offsets[1:]
        res = long_enough.clone()
        for j in range(k):
            idx = torch.clamp(ends - k + j, min=0)
            res = res & (c.bytes.index_select(0, idx) == patt[j])
    else:  # '%core%': substring search over the flat byte buffer
        total = int(c.bytes.numel())
        if total < k:
            res = torch.zeros(n, dtype=torch.bool, device=device)
        else:
            w = total - k + 1
            m = c.bytes[0:w] == patt[0]
            for j in range(1, k):
                m = m & (c.bytes[j : w + j] == patt[j])
            pos = m.nonzero(as_tuple=True)[0]
            # map byte position → row, require the window inside the row
            row = torch.searchsorted(c.offsets[1:], pos, right=True)
            ok = (pos + k) <= c.offsets.index_select(0, row + 1)
            res = torch.zeros(n, dtype=torch.bool, device=device)
            rows = row[ok]
            if rows.numel() > 0:
                res.index_put_(
                    (rows,),
                    torch.ones(
                        rows.numel(), dtype=torch.bool, device=device
                    ),
                )
    return res, c.valid


def _eval_like_general(
    c: "StringDeviceColumn", pat: str, device: torch.device
) -> Tuple[torch.Tensor, Optional[torch.Tensor]]:
    """Full LIKE semantics on device tensors: the pattern is split on
    ``%`` into segments; the first/last segments are anchored at row
    start/end, middle segments are located greedily (earliest match ≥
    the running cursor, which is optimal for glob matching).  ``_``
    matches one byte, so patterns containing ``_`` fall back when the
    column holds non-ASCII data (multi-byte characters)."""
    n = len(c)
    total = int(c.bytes.numel())
    if total == 0:
        # every row is the empty string: the pattern matches iff it has
        # no literal bytes and no ``_`` (i.e. it is all ``%`` / empty)
        empty_ok = pat.replace("%", "") == ""
        res = torch.full((n,), empty_ok, dtype=torch.bool, device=device)
        return res, c.valid
    if "_" in pat and int(c.bytes.max().item()) >= 0x80:
        raise DeviceExprError("LIKE _ on non-ASCII data")
    starts = c.offsets[:-1]
    ends = c.offsets[1:]
    lengths = ends - starts

    def seg_bytes(seg: str) -> List[Optional[int]]:
        return [None if ch == "_" else b for ch, b in
                zip(seg, seg.encode("utf-8"))] if seg.isascii() else _multi(seg)

    def _multi(seg: str) -> List[Optional[int]]:
        out: List[Optional[int]] = []
        for ch in seg:
            if ch == "_":
                out.append(None)
            else:
                out.extend(ch.encode("utf-8"))
        return out

    def window_mask(sb: List[Optional[int]]) -> torch.Tensor:
        """bool over byte positions [0, total-k+1): segment matches."""
        k = len(sb)
        if total < k:
            return torch.zeros(0, dtype=torch.bool, device=device)
        w = total - k + 1
        m = torch.ones(w, dtype=torch.bool, device=device)
        for j, b in enumerate(sb):
            if b is None:
                continue
            m = m & (c.bytes[j : w + j] == b)
        return m

    def match_at(sb: List[Optional[int]], pos: torch.Tensor) -> torch.Tensor:
        """bool per row: segment matches at byte position pos (caller
        guarantees pos+k is inside the row when the row is eligible)."""
        k = len(sb)
        res = torch.ones(n, dtype=torch.bool, device=device)
        cap = max(total - 1, 0)
        for j, b in enumerate(sb):
            if b is None:
                continue
            idx = torch.clamp(pos + j, min=0, max=cap)
            res = res & (c.bytes.index_select(0, idx) == b)
        return res

    segs = pat.split("%")
    first, last, middles = segs[0], segs[-1] if len(segs) > 1 else "", [
        s for s in segs[1:-1] if s != ""
    ]
    if len(segs) == 1:
        # no %: fixed-length pattern with _ wildcards
        sb = seg_bytes(first)
        res = (lengths == len(sb)) & match_at(sb, starts)
        return res, c.valid
    res = torch.ones(n, dtype=torch.bool, device=device)
    cur = starts.clone()
    if first != "":
        sb = seg_bytes(first)
        res = res & (lengths >= len(sb)) & match_at(sb, starts)
        cur = starts + len(sb)
    for seg in middles:
        sb = seg_bytes(seg)
        k = len(sb)
        m = window_mask(sb)
        pos = m.nonzero(as_tuple=True)[0]
        # keep only matches fully inside a row
        row = torch.searchsorted(ends, pos, right=True)
        ok = (pos + k) <= ends.index_select(0, torch.clamp(row, max=n - 1))
        pos, row = pos[ok], row[ok]
        if pos.numel() == 0:
            res = torch.zeros(n, dtype=torch.bool, device=device)
            break
        # earliest match position >= cur, required to be in the same row
        idx = torch.searchsorted(pos, cur)
        found = idx < pos.numel()
        idx_c = torch.clamp(idx, max=pos.numel() - 1)
        p = pos.index_select(0, idx_c)
        r = row.index_select(0, idx_c)
        ok_row = found & (r == torch.arange(n, device=device))
        res = res & ok_row
        cur = torch.where(ok_row, p + k, cur)
    if last != "":
        sb = seg_bytes(last)
        k = len(sb)
        at = ends - k
        res = res & (at >= cur) & match_at(sb, at)
    return res, c.valid


def _try_string_equality(
    expr: "_BinaryOpExpr", df: HipDataFrame
) -> Optional[Tuple[torch.Tensor, Optional[torch.Tensor]]]:
    """``col == 'literal'`` on a device string column: compared via the
    128-bit row hash of the column vs the literal's hash (undetectable
    mismatch probability ~2^-128)."""
    sides = [expr.left, expr.right]
    col_e = lit_e = None
    for a, b in (sides, sides[::-1]):
        if (
            isinstance(a, _NamedColumnExpr)
            and a.name in df.schema._index
            and isinstance(df.col(a.name), StringDeviceColumn)
            and isinstance(b, _LiteralColumnExpr)
            and isinstance(b.value, str)
        ):
            col_e, lit_e = a, b
            break
    if col_e is None:
        return None
    from fugue_amd.hip import ops as dops
    import pyarrow as pa

    c = df.col(col_e.name)
    h1 = dops.hash_rows([c])
    h2 = dops.hash_rows([c], seed=dops._H2_SEED)
    lit_col = StringDeviceColumn.from_arrow_strings(
        pa.array([lit_e.value], type=pa.string()), df.device
    )
    l1 = dops.hash_rows([lit_col])
    l2 = dops.hash_rows([lit_col], seed=dops._H2_SEED)
    if h1.is_cuda:
        from fugue_amd.hip.ext import get_ext

        return (
            get_ext().eq2_mask(
                h1, h2, l1, l2, c.valid, expr.op == "!="
            ),
            None,
        )
    eq = (h1 == l1[0]) & (h2 == l2[0])
    if c.valid is not None:
        eq = eq & c.valid
    if expr.op == "!=":
        res = ~eq
        if c.valid is not None:
            res = res & c.valid
        return res, None
    return eq, None


def _merge_valid(
    a: Optional[torch.Tensor], b: Optional[torch.Tensor]
) -> Optional[torch.Tensor]:
    if a is None:
        return b
    if b is None:
        return a
    return a & b


def _as_bool(d: torch.Tensor, v: Optional[torch.Tensor]) -> torch.Tensor:
    b = d.to(torch.bool)
    if v is not None:
        b = b & v
    return b


def _cast(data: torch.Tensor, tp: pa.DataType) -> torch.Tensor:
    import pyarrow as pa

    if pa.types.is_floating(tp):
        return data.to(torch.float64 if tp == pa.float64() else torch.float32)
    if pa.types.is_integer(tp):
        m = {
            pa.int64(): torch.int64,
            pa.int32(): torch.int32,
            pa.int16(): torch.int16,
            pa.int8(): torch.int8,
        }
        return data.to(m.get(tp, torch.int64))
    if pa.types.is_boolean(tp):
        return data.to(torch.bool)
    raise DeviceExprError(f"cast to {tp} not supported on device")


# ------------------------------------------------------------------ #
# fused filter programs: compile a predicate tree to a postfix program
# executed by the one-pass device interpreter (``expr_filter_kernel``)
# instead of a chain of elementwise torch launches
# ------------------------------------------------------------------ #
_XOP = dict(
    COL=0, LIT_D=1, LIT_I=2, ADD=3, SUB=4, MUL=5, DIV=6,
    LT=7, LE=8, GT=9, GE=10, EQ=11, NE=12, AND=13, OR=14, NOT=15,
    ISNULL=16, NOTNULL=17, NEG=18,
)
_XDT = {
    torch.float64: 0, torch.float32: 1, torch.int64: 2, torch.int32: 3,
    torch.int16: 4, torch.int8: 5, torch.bool: 6,
}
_X_BINOPS = {
    "+": "ADD", "-": "SUB", "*": "MUL", "/": "DIV",
    "<": "LT", "<=": "LE", ">": "GT", ">=": "GE", "==": "EQ", "!=": "NE",
    "&": "AND", "|": "OR",
}


class _ProgBuilder:
    def __init__(self, df: HipDataFrame):
        self.df = df
        self.ops: List[int] = []
        self.aux: List[int] = []
        self.imm: List[int] = []
        self.cols: List[Any] = []
        self.col_idx: Dict[str, int] = {}
        self.depth = 0
        self.max_depth = 0
        self.tags: List[int] = []
        self.int64_seen = False

    def _push(self, op: str, aux: int = 0, dstack: int = 1) -> None:
        self.ops.append(_XOP[op])
        self.aux.append(aux)
        self.depth += dstack
        self.max_depth = max(self.max_depth, self.depth)
        # static result-tag simulation (mirrors the kernel's promotion)
        t = self.tags
        if op == "COL":
            dt = self.cols[aux].data.dtype
            t.append(
                0 if dt in (torch.float64, torch.float32)
                else (2 if dt == torch.bool else 1)
            )
            if dt == torch.int64:
                self.int64_seen = True
        elif op == "LIT_D":
            t.append(0)
        elif op == "LIT_I":
            t.append(1)
            self.int64_seen = True
        elif op in ("NOT", "ISNULL", "NOTNULL"):
            t[-1] = 2
        elif op == "NEG":
            pass
        elif op in ("AND", "OR"):
            t.pop()
            t[-1] = 2
        elif op in ("ADD", "SUB", "MUL"):
            b = t.pop()
            a = t[-1]
            t[-1] = 1 if (a != 0 and b != 0) else 0
        elif op == "DIV":
            t.pop()
            t[-1] = 0
        else:  # comparisons
            t.pop()
            t[-1] = 2

    def emit(self, e: ColumnExpr) -> None:
        if isinstance(e, _LiteralColumnExpr):
            v = e.value
            if v is None or isinstance(v, str):
                raise DeviceExprError("literal not fusable")
            if isinstance(v, bool) or isinstance(v, int):
                bits = int(v)
                self.imm.append(bits)
                self._push("LIT_I", len(self.imm) - 1)
            elif isinstance(v, float):
                import struct as _struct

                bits = _struct.unpack("<q", _struct.pack("<d", v))[0]
                self.imm.append(bits)
                self._push("LIT_D", len(self.imm) - 1)
            else:
                raise DeviceExprError("literal not fusable")
            return
        if isinstance(e, _NamedColumnExpr):
            if e.as_type is not None:
                raise DeviceExprError("cast not fusable")
            c = self.df.col(e.name)
            if isinstance(c, StringDeviceColumn) or c.data.dtype not in _XDT:
                raise DeviceExprError("column not fusable")
            if e.name not in self.col_idx:
                self.col_idx[e.name] = len(self.cols)
                self.cols.append(c)
            self._push("COL", self.col_idx[e.name])
            return
        if isinstance(e, _NotOpExpr):
            self.emit(e.col)
            self._push("NOT", dstack=0)
            return
        if isinstance(e, _UnaryOpExpr):
            self.emit(e.col)
            if e.op == "IS_NULL":
                self._push("ISNULL", dstack=0)
            elif e.op == "NOT_NULL":
                self._push("NOTNULL", dstack=0)
            elif e.op == "-":
                self._push("NEG", dstack=0)
            else:
                raise DeviceExprError(f"unary {e.op} not fusable")
            return
        if isinstance(e, _BinaryOpExpr):
            name = _X_BINOPS.get(e.op)
            if name is None or e.as_type is not None:
                raise DeviceExprError(f"binary {e.op} not fusable")
            self.emit(e.left)
            self.emit(e.right)
            self._push(name, dstack=-1)
            return
        raise DeviceExprError(f"{type(e).__name__} not fusable")


def _run_program(
    b: "_ProgBuilder", df: HipDataFrame, mode: str
) -> Any:
    from fugue_amd.hip.ext import get_ext

    ext = get_ext()
    empty = torch.empty(0)
    args = (
        torch.tensor(b.ops, dtype=torch.uint8),
        torch.tensor(b.aux, dtype=torch.int8),
        torch.tensor(b.imm, dtype=torch.int64),
        [c.data.contiguous() for c in b.cols],
        [
            c.valid.contiguous() if c.valid is not None else empty
            for c in b.cols
        ],
        torch.tensor([_XDT[c.data.dtype] for c in b.cols], dtype=torch.uint8),
        df.count(),
    )
    if mode == "filter":
        return ext.expr_filter(*args)
    return ext.expr_value(*args, 1 if mode == "int" else 0)


def _fusable(b: "_ProgBuilder") -> bool:
    return (
        len(b.ops) <= 48
        and 0 < len(b.cols) <= 12
        and len(b.imm) <= 12
        and b.max_depth <= 12
    )


def try_fused_value(
    expr: ColumnExpr, df: HipDataFrame
) -> Optional[Tuple[torch.Tensor, Optional[torch.Tensor]]]:
    """One-pass evaluation of a compound arithmetic expression; returns
    (data, valid) like eval_device_expr, or None when not fusable /
    not worth fusing (fewer than 3 fused launches saved)."""
    import os as _os

    if _os.environ.get("FUGUE_EXPR_FUSE", "1") == "0":
        return None
    if not df.device.startswith("cuda"):
        return None
    if not isinstance(expr, _BinaryOpExpr) or expr.op not in "+-*/":
        return None
    if expr.as_type is not None:
        return None
    b = _ProgBuilder(df)
    try:
        b.emit(expr)
    except DeviceExprError:
        return None
    n_arith = sum(1 for o in b.ops if _XOP["ADD"] <= o <= _XOP["NE"])
    if n_arith < 2 or not _fusable(b) or b.tags[-1] == 2:
        return None
    out_int = b.tags[-1] == 1
    if out_int and not b.int64_seen:
        # torch promotion would keep a narrower int dtype; skip fusion
        return None
    vals, valid = _run_program(b, df, "int" if out_int else "value")
    if all(c.valid is None for c in b.cols):
        valid = None
    return vals, valid


_CMP_OP_CODE = {"==": 0, "!=": 1, "<": 2, "<=": 3, ">": 4, ">=": 5}
_CMP_DTYPES = (
    torch.int64, torch.int32, torch.int16, torch.float64, torch.float32
)


def _try_simple_cmp(
    expr: ColumnExpr, df: HipDataFrame
) -> Optional[torch.Tensor]:
    """Dedicated one-pass kernels for the dominant WHERE shapes
    (``col <op> literal`` / ``col <op> col``): the general interpreter
    keeps its value stack in scratch memory and runs ~10x off SOL on
    these (profiles/NOTES.md r02c)."""
    if not isinstance(expr, _BinaryOpExpr) or expr.as_type is not None:
        return None
    code = _CMP_OP_CODE.get(expr.op)
    if code is None:
        return None
    left, right = expr.left, expr.right
    swap = False
    if isinstance(left, _LiteralColumnExpr) and isinstance(
        right, _NamedColumnExpr
    ):
        left, right = right, left
        swap = True
        code = {0: 0, 1: 1, 2: 4, 3: 5, 4: 2, 5: 3}[code]  # mirror op
    if not isinstance(left, _NamedColumnExpr) or left.as_type is not None:
        return None
    if left.name not in df.schema._index:
        return None
    a = df.col(left.name)
    if isinstance(a, StringDeviceColumn) or a.data.dtype not in _CMP_DTYPES:
        return None
    from fugue_amd.hip.ext import get_ext

    if isinstance(right, _LiteralColumnExpr):
        v = right.value
        if isinstance(v, bool) or v is None:
            return None
        if isinstance(v, int):
            return get_ext().cmp_imm(a.data, a.valid, v, float(v), True, code)
        if isinstance(v, float):
            return get_ext().cmp_imm(a.data, a.valid, 0, v, False, code)
        return None
    if isinstance(right, _NamedColumnExpr) and right.as_type is None:
        if right.name not in df.schema._index:
            return None
        b = df.col(right.name)
        if (
            isinstance(b, StringDeviceColumn)
            or b.data.dtype != a.data.dtype
        ):
            return None
        return get_ext().cmp_col(a.data, a.valid, b.data, b.valid, code)
    return None


def try_fused_filter(
    expr: ColumnExpr, df: HipDataFrame
) -> Optional[torch.Tensor]:
    """One-pass device evaluation of a filter predicate; None when the
    expression (or frame location) isn't fusable."""
    if not df.device.startswith("cuda"):
        return None
    simple = _try_simple_cmp(expr, df)
    if simple is not None:
        return simple
    b = _ProgBuilder(df)
    try:
        b.emit(expr)
    except DeviceExprError:
        return None
    if not _fusable(b):
        return None
    return _run_program(b, df, "filter")


def filter_mask(expr: ColumnExpr, df: HipDataFrame) -> torch.Tensor:
    fused = try_fused_filter(expr, df)
    if fused is not None:
        return fused
    d, v = eval_device_expr(expr, df)
    return _as_bool(d, v)
