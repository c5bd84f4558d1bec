"""Fused expression compilation: a projection/predicate list compiles to
one postfix program executed by ONE HIP kernel (csrc/fusedexpr.hip),
replacing a launch per expression node.

The reference evaluates whole expression lists per batch
(/root/reference/src/daft-recordbatch/src/lib.rs:1623 eval_expression_list);
on MI355X the per-node torch kernels were ~half of remaining GPU time
(profiles/tpch_sf100_kernel_stats.md), so the list becomes a single
interpreter launch: each thread walks the program with a register stack
of (f64 value, valid) pairs.

Compilation is conservative: any node outside the numeric core (strings,
decimals wider than 18, UDFs, aggs, dict columns, broadcasts) bails out
and that expression falls back to per-node evaluation.
"""
from __future__ import annotations

import datetime as _dt
from typing import Dict, List, Optional, Tuple

import torch

from ..expressions.expressions import (Alias, Between, BinaryOp, Cast,
                                       ColumnRef, ExprNode, FillNull, IfElse,
                                       IsIn, IsNull, Literal, Not)
from ..schema import DataType, TypeKind

OP_COL, OP_LIT = 0, 1
OP_ADD, OP_SUB, OP_MUL, OP_DIV = 2, 3, 4, 5
OP_EQ, OP_NE, OP_LT, OP_LE, OP_GT, OP_GE = 6, 7, 8, 9, 10, 11
OP_AND, OP_OR, OP_NOT, OP_NEG = 12, 13, 14, 15
OP_ISNULL, OP_NOTNULL, OP_FILLNULL, OP_SELECT = 16, 17, 18, 19
OP_STORE = 20

_BIN = {"add": OP_ADD, "sub": OP_SUB, "mul": OP_MUL, "div": OP_DIV,
        "eq": OP_EQ, "ne": OP_NE, "lt": OP_LT, "le": OP_LE,
        "gt": OP_GT, "ge": OP_GE, "and": OP_AND, "or": OP_OR}

_IN_CODES = {torch.float64: 0, torch.float32: 1, torch.int64: 2,
             torch.int32: 3, torch.int16: 4, torch.int8: 5,
             torch.bool: 6, torch.uint8: 6, torch.uint32: 7,
             torch.uint64: 8}
_OUT_CODES = {torch.float64: 0, torch.float32: 1, torch.int64: 2,
              torch.int32: 3, torch.bool: 6}

_MAX_STACK = 8  # must match VSTACK in csrc/fusedexpr.hip
_EPOCH = _dt.date(1970, 1, 1)


class _Bail(Exception):
    pass


class _Compiler:
    def __init__(self, batch):
        self.batch = batch
        self.n = len(batch)
        self.ins: List[Tuple[int, int]] = []
        self.lits: List[float] = []
        self.cols: List = []          # Series
        self.col_ix: Dict[str, int] = {}
        self.depth = 0
        self.max_depth = 0
        self.any_valid = False        # any pushed slot may be invalid

    def _push(self, k: int = 1):
        self.depth += k
        if self.depth > _MAX_STACK:
            raise _Bail("stack too deep")
        self.max_depth = max(self.max_depth, self.depth)

    def _pop(self, k: int = 1):
        self.depth -= k

    def emit(self, op: int, arg: int = 0):
        self.ins.append((op, arg))

    def lit(self, v: float) -> int:
        self.lits.append(float(v))
        return len(self.lits) - 1

    # -- literal conversion ------------------------------------------------
    def _lit_value(self, e: Literal) -> float:
        v = e.value
        dt = e.dtype
        if v is None:
            raise _Bail("null literal")
        if isinstance(v, bool):
            return 1.0 if v else 0.0
        if isinstance(v, _dt.datetime):
            us = int((v - _dt.datetime(1970, 1, 1)).total_seconds() * 1e6)
            if dt.kind == TypeKind.TIMESTAMP and dt.timeunit == "ms":
                return us / 1000.0
            if dt.kind == TypeKind.TIMESTAMP and dt.timeunit == "ns":
                return us * 1000.0
            return float(us)
        if isinstance(v, _dt.date):
            return float((v - _EPOCH).days)
        if isinstance(v, (int, float)):
            if isinstance(v, int) and abs(v) >= (1 << 53):
                raise _Bail("literal beyond f64 exact range")
            if dt is not None and dt.is_decimal():
                return float(v)        # semantic value; columns descale
            return float(v)
        raise _Bail(f"literal {type(v).__name__}")

    # -- column loading ----------------------------------------------------
    def col(self, name: str):
        if name in self.col_ix:
            ix = self.col_ix[name]
            s = self.cols[ix]
        else:
            s = self.batch.column(name)
            if s.pyobjs is not None or s.data is None or s.is_dict() or \
                    s.children:
                raise _Bail(f"column {name} not fusable")
            if len(s) != self.n:
                raise _Bail("broadcast column")
            dt = s.dtype
            if not (dt.is_numeric() or dt.is_boolean() or dt.is_temporal()
                    or dt.is_decimal()):
                raise _Bail(f"dtype {dt}")
            if dt.is_decimal() and s.data.dtype != torch.int64:
                raise _Bail("wide decimal")
            if s.data.dtype not in _IN_CODES:
                raise _Bail(f"torch dtype {s.data.dtype}")
            ix = len(self.cols)
            self.cols.append(s)
            self.col_ix[name] = ix
        if s.validity is not None:
            self.any_valid = True
        self.emit(OP_COL, ix)
        self._push()

    # -- tree walk ---------------------------------------------------------
    def compile(self, e: ExprNode):
        if isinstance(e, Alias):
            return self.compile(e.child)
        if isinstance(e, ColumnRef):
            return self.col(e.name)
        if isinstance(e, Literal):
            self.emit(OP_LIT, self.lit(self._lit_value(e)))
            self._push()
            return
        if isinstance(e, BinaryOp):
            op = _BIN.get(e.op)
            if op is None:
                raise _Bail(f"binop {e.op}")
            # temporal raw values are unit-dependent (days vs us): both
            # sides must agree on kind+unit or the f64 compare is wrong
            try:
                lt = e.left.to_field(self.batch.schema).dtype
                rt = e.right.to_field(self.batch.schema).dtype
            except Exception:
                raise _Bail("untyped operand")
            if lt.is_temporal() or rt.is_temporal():
                if lt.kind != rt.kind or \
                        getattr(lt, "timeunit", None) != \
                        getattr(rt, "timeunit", None):
                    raise _Bail("temporal unit mismatch")
                if e.op not in ("eq", "ne", "lt", "le", "gt", "ge", "sub"):
                    raise _Bail("temporal arithmetic")
            self.compile(e.left)
            self.compile(e.right)
            self.emit(op)
            self._pop()
            return
        if isinstance(e, Not):
            self.compile(e.child)
            self.emit(OP_NOT)
            return
        if isinstance(e, IsNull):
            self.compile(e.child)
            self.emit(OP_NOTNULL if getattr(e, "negate", False)
                      else OP_ISNULL)
            return
        if isinstance(e, FillNull):
            self.compile(e.child)
            self.compile(e.fill)
            self.emit(OP_FILLNULL)
            self._pop()
            return
        if isinstance(e, IfElse):
            self.compile(e.pred)
            self.compile(e.truthy)
            self.compile(e.falsy)
            self.emit(OP_SELECT)
            self._pop(2)
            return
        if isinstance(e, Between):
            self.compile(e.child)
            # dup via re-compilation (columns re-load from cache; cheap)
            self.compile(e.lo)
            self.emit(OP_GE)
            self._pop()
            self.compile(e.child)
            self.compile(e.hi)
            self.emit(OP_LE)
            self._pop()
            self.emit(OP_AND)
            self._pop()
            return
        if isinstance(e, IsIn):
            vals = getattr(e, "values", None)
            if vals is None or len(vals) == 0 or len(vals) > 8:
                raise _Bail("is_in size")
            first = True
            for v in vals:
                if not isinstance(v, (int, float, bool)) or \
                        (isinstance(v, int) and abs(v) >= (1 << 53)):
                    raise _Bail("is_in value")
                self.compile(e.child)
                self.emit(OP_LIT, self.lit(float(v)))
                self._push()
                self.emit(OP_EQ)
                self._pop()
                if not first:
                    self.emit(OP_OR)
                    self._pop()
                first = False
            return
        if isinstance(e, Cast):
            src = e.child.to_field(self.batch.schema).dtype
            dst = e.dtype
            ok = (src.is_numeric() or src.is_boolean() or
                  src.is_temporal()) and \
                (dst.is_numeric() and not dst.is_decimal())
            if not ok:
                raise _Bail(f"cast {src}->{dst}")
            self.compile(e.child)
            if dst.is_integer():
                # the kernel stores through llrint; mid-tree int casts of
                # non-integral values would diverge from torch truncation
                if not (src.is_integer() or src.is_boolean() or
                        src.is_temporal()):
                    raise _Bail("float->int cast")
            return
        raise _Bail(type(e).__name__)


class FusedPlan:
    __slots__ = ("prog", "lits", "cols", "out_meta", "n")

    def __init__(self, prog, lits, cols, out_meta, n):
        self.prog = prog
        self.lits = lits
        self.cols = cols
        self.out_meta = out_meta
        self.n = n


def try_fuse(exprs: List[ExprNode], batch) -> Optional[List]:
    """Compile as many of `exprs` as possible into one kernel; returns the
    full result list (fused + fallback per-node eval) or None when fusion
    is not worthwhile (fewer than 4 fused instructions)."""
    import os
    if os.environ.get("DAFT_AMD_DISABLE_FUSED"):
        return None
    if batch.device.type != "cuda" or len(batch) == 0:
        return None
    from . import load_native
    native = load_native()
    if native is None:
        return None

    jit = try_fuse_jit(exprs, batch)
    if jit is not None:
        return jit

    comp = _Compiler(batch)
    fused_ix: List[int] = []
    out_meta = []
    results: List = [None] * len(exprs)
    for i, e in enumerate(exprs):
        base = e
        while isinstance(base, Alias):
            base = base.child
        if isinstance(base, (ColumnRef, Literal)):
            continue      # zero-copy via normal eval; fusing would copy
        mark = (len(comp.ins), len(comp.lits), len(comp.cols),
                dict(comp.col_ix), comp.any_valid)
        try:
            f = e.to_field(batch.schema)
            out_dt = f.dtype
            tdt = out_dt.to_torch() if out_dt.is_fixed_width() else None
            if tdt not in _OUT_CODES:
                raise _Bail(f"out dtype {out_dt}")
            if out_dt.is_decimal():
                raise _Bail("decimal out")
            av_before = comp.any_valid
            comp.any_valid = False
            comp.compile(e)
            produces_null = comp.any_valid or _may_null(e)
            comp.any_valid = comp.any_valid or av_before
            comp.emit(OP_STORE, len(out_meta))
            comp._pop()
            out_meta.append((i, f.name, out_dt, _OUT_CODES[tdt],
                             produces_null))
            fused_ix.append(i)
        except _Bail:
            comp.ins = comp.ins[:mark[0]]
            comp.lits = comp.lits[:mark[1]]
            comp.cols = comp.cols[:mark[2]]
            comp.col_ix = mark[3]
            comp.any_valid = mark[4]
            comp.depth = 0
    if not fused_ix or len(comp.ins) < 4:
        return None

    import itertools
    prog = torch.tensor(list(itertools.chain.from_iterable(comp.ins)),
                        dtype=torch.int32)
    lits = torch.tensor(comp.lits, dtype=torch.float64)
    datas, valids, codes, scales = [], [], [], []
    for s in comp.cols:
        datas.append(s.data)
        valids.append(s.validity)
        codes.append(_IN_CODES[s.data.dtype])
        if s.dtype.is_decimal():
            scales.append(10.0 ** (-s.dtype.scale))
        else:
            scales.append(1.0)
    out_codes = [m[3] for m in out_meta]
    out_scales = [1.0] * len(out_meta)
    out_need_valid = [1 if m[4] else 0 for m in out_meta]
    n = len(batch)
    flat = native.fused_eval(prog, lits, datas, valids, codes, scales,
                             out_codes, out_scales, out_need_valid, n)
    from ..series import Series
    for k, (i, name, out_dt, code, pn) in enumerate(out_meta):
        data = flat[2 * k]
        valid = flat[2 * k + 1] if pn else None
        if valid is not None and bool(valid.all().item()):
            valid = None
        if out_dt.to_torch() != data.dtype:
            data = data.view(out_dt.to_torch()) \
                if data.element_size() == out_dt.to_torch().itemsize \
                else data.to(out_dt.to_torch())
        results[i] = Series(name, out_dt, data=data, validity=valid)
    for i, e in enumerate(exprs):
        if results[i] is None:
            results[i] = e.evaluate(batch)
    return results


def _may_null(e: ExprNode) -> bool:
    """Static: can this expression introduce nulls beyond input validity?"""
    if isinstance(e, (IsNull,)):
        return False
    return any(_may_null(c) for c in e.children())


# ---------------------------------------------------------------------------
# hipRTC codegen: straight-line kernel per (expression list, schema,
# validity pattern), compiled once per process and cached by source
# ---------------------------------------------------------------------------

_C_LOAD = {0: "((const double*)C{i}d)[i]",
           1: "(double)((const float*)C{i}d)[i]",
           2: "(double)((const long long*)C{i}d)[i]",
           3: "(double)((const int*)C{i}d)[i]",
           4: "(double)((const short*)C{i}d)[i]",
           5: "(double)((const signed char*)C{i}d)[i]",
           6: "(double)((const unsigned char*)C{i}d)[i]",
           7: "(double)((const unsigned int*)C{i}d)[i]",
           8: "(double)(long long)((const unsigned long long*)C{i}d)[i]"}


class _CodeGen:
    """Symbolic twin of _Compiler: walks the same node set but emits C
    statements (value temp, validity temp) instead of opcodes."""

    def __init__(self, comp: "_Compiler"):
        self.comp = comp
        self.lines: List[str] = []
        self.tmp = 0
        self.load_cache: Dict[int, Tuple[str, Optional[str]]] = {}
        self.expr_cache: Dict[str, Tuple[str, Optional[str]]] = {}

    def t(self) -> str:
        self.tmp += 1
        return f"t{self.tmp}"

    def emit(self, typ: str, name: str, expr: str):
        self.lines.append(f"      {typ} {name} = {expr};")

    def gen(self, e: ExprNode) -> Tuple[str, Optional[str]]:
        """Returns (value C expr, validity C expr or None=always valid)."""
        if isinstance(e, Alias):
            return self.gen(e.child)
        if e.children():
            key = repr(e)
            hit = self.expr_cache.get(key)
            if hit is not None:
                return hit
            out = self._gen_inner(e)
            self.expr_cache[key] = out
            return out
        return self._gen_inner(e)

    def _gen_inner(self, e: ExprNode) -> Tuple[str, Optional[str]]:
        if isinstance(e, ColumnRef):
            ix = self.comp.col_ix[e.name]
            hit = self.load_cache.get(ix)
            if hit is not None:
                return hit
            s = self.comp.cols[ix]
            code = _IN_CODES[s.data.dtype]
            v = self.t()
            load = _C_LOAD[code].format(i=ix)
            if s.dtype.is_decimal():
                load = f"({load}) * {10.0 ** (-s.dtype.scale)!r}"
            self.emit("const double", v, load)
            vv = None
            if s.validity is not None:
                vv = self.t()
                self.emit("const bool", vv, f"C{ix}v[i]")
            self.load_cache[ix] = (v, vv)
            return v, vv
        if isinstance(e, Literal):
            return repr(self.comp._lit_value(e)), None
        if isinstance(e, BinaryOp):
            a, av = self.gen(e.left)
            b, bv = self.gen(e.right)
            out = self.t()
            cop = {"add": "+", "sub": "-", "mul": "*", "div": "/",
                   "eq": "==", "ne": "!=", "lt": "<", "le": "<=",
                   "gt": ">", "ge": ">="}.get(e.op)
            if cop is not None:
                if e.op in ("eq", "ne", "lt", "le", "gt", "ge"):
                    self.emit("const double", out,
                              f"(({a}) {cop} ({b})) ? 1.0 : 0.0")
                else:
                    self.emit("const double", out, f"({a}) {cop} ({b})")
                return out, self._and_valid(av, bv)
            if e.op in ("and", "or"):
                # Kleene three-valued logic (matches kernels.logical_op)
                ab = self.t()
                bb = self.t()
                self.emit("const bool", ab, f"({a}) != 0.0")
                self.emit("const bool", bb, f"({b}) != 0.0")
                op = "&&" if e.op == "and" else "||"
                self.emit("const double", out,
                          f"(({ab}) {op} ({bb})) ? 1.0 : 0.0")
                if av is None and bv is None:
                    return out, None
                avv = av or "true"
                bvv = bv or "true"
                vv = self.t()
                if e.op == "and":
                    self.emit("const bool", vv,
                              f"(({avv}) && ({bvv})) || (({avv}) && "
                              f"!({ab})) || (({bvv}) && !({bb}))")
                else:
                    self.emit("const bool", vv,
                              f"(({avv}) && ({bvv})) || (({avv}) && "
                              f"({ab})) || (({bvv}) && ({bb}))")
                return out, vv
            raise _Bail(f"binop {e.op}")
        if isinstance(e, Not):
            a, av = self.gen(e.child)
            out = self.t()
            self.emit("const double", out, f"(({a}) != 0.0) ? 0.0 : 1.0")
            return out, av
        if isinstance(e, IsNull):
            a, av = self.gen(e.child)
            out = self.t()
            neg = getattr(e, "negate", False)
            if av is None:
                self.emit("const double", out, "0.0" if not neg else "1.0")
            elif neg:
                self.emit("const double", out, f"({av}) ? 1.0 : 0.0")
            else:
                self.emit("const double", out, f"({av}) ? 0.0 : 1.0")
            return out, None
        if isinstance(e, FillNull):
            a, av = self.gen(e.child)
            b, bv = self.gen(e.fill)
            if av is None:
                return a, None
            out = self.t()
            self.emit("const double", out, f"({av}) ? ({a}) : ({b})")
            if bv is None:
                return out, None
            vv = self.t()
            self.emit("const bool", vv, f"({av}) || ({bv})")
            return out, vv
        if isinstance(e, IfElse):
            c, cv = self.gen(e.pred)
            t_, tv = self.gen(e.truthy)
            f_, fv = self.gen(e.falsy)
            m = self.t()
            cvv = f" && ({cv})" if cv is not None else ""
            self.emit("const bool", m, f"(({c}) != 0.0){cvv}")
            out = self.t()
            self.emit("const double", out, f"({m}) ? ({t_}) : ({f_})")
            if tv is None and fv is None:
                return out, None
            vv = self.t()
            self.emit("const bool", vv,
                      f"({m}) ? ({tv or 'true'}) : ({fv or 'true'})")
            return out, vv
        if isinstance(e, Between):
            a, av = self.gen(e.child)
            lo, lov = self.gen(e.lo)
            hi, hiv = self.gen(e.hi)
            out = self.t()
            self.emit("const double", out,
                      f"((({a}) >= ({lo})) && (({a}) <= ({hi}))) "
                      f"? 1.0 : 0.0")
            return out, self._and_valid(self._and_valid(av, lov), hiv)
        if isinstance(e, IsIn):
            a, av = self.gen(e.child)
            parts = []
            for v in e.values:
                parts.append(f"(({a}) == {float(v)!r})")
            out = self.t()
            self.emit("const double", out,
                      f"({' || '.join(parts)}) ? 1.0 : 0.0")
            return out, av
        if isinstance(e, Cast):
            return self.gen(e.child)
        raise _Bail(type(e).__name__)

    def _and_valid(self, a: Optional[str], b: Optional[str]) -> Optional[str]:
        if a is None:
            return b
        if b is None:
            return a
        vv = self.t()
        self.emit("const bool", vv, f"({a}) && ({b})")
        return vv


_C_STORE = {0: "((double*)O{k}d)[i] = {v};",
            1: "((float*)O{k}d)[i] = (float)({v});",
            2: "((long long*)O{k}d)[i] = (long long)llrint({v});",
            3: "((int*)O{k}d)[i] = (int)llrint({v});",
            6: "((bool*)O{k}d)[i] = ({v}) != 0.0;"}


def _gen_source(comp: "_Compiler", gens, out_meta) -> str:
    """Full kernel source: hoisted column pointers, grid-stride row loop,
    straight-line body."""
    head = [
        "extern \"C\" __global__ void fe(const long long* __restrict__ "
        "cols, const long long* __restrict__ outs, long long n) {",
    ]
    for ix, s in enumerate(comp.cols):
        head.append(f"  const void* C{ix}d = (const void*)cols[{ix} * 2];")
        if s.validity is not None:
            head.append(f"  const bool* C{ix}v = "
                        f"(const bool*)cols[{ix} * 2 + 1];")
    for k, m in enumerate(out_meta):
        head.append(f"  void* O{k}d = (void*)outs[{k} * 2];")
        if m[4]:
            head.append(f"  bool* O{k}v = (bool*)outs[{k} * 2 + 1];")
    head.append("  const long long stride = (long long)gridDim.x * "
                "blockDim.x;")
    head.append("  for (long long i = (long long)blockIdx.x * blockDim.x + "
                "threadIdx.x; i < n; i += stride) {")
    body: List[str] = []
    for k, (lines, val, vld) in enumerate(gens):
        body.extend(lines)
        code = out_meta[k][3]
        body.append("      " + _C_STORE[code].format(k=k, v=val))
        if out_meta[k][4]:
            body.append(f"      O{k}v[i] = {vld if vld else 'true'};")
    tail = ["  }", "}"]
    return "\n".join(head + body + tail) + "\n"


def try_fuse_jit(exprs: List[ExprNode], batch) -> Optional[List]:
    """hipRTC path: compile the list to a straight-line kernel.  Returns
    the full result list or None (caller falls back to the interpreter)."""
    import os
    if os.environ.get("DAFT_AMD_DISABLE_JIT"):
        return None
    from . import load_native
    native = load_native()
    if native is None or not hasattr(native, "fused_eval_jit"):
        return None

    comp = _Compiler(batch)
    cg = _CodeGen(comp)     # ONE generator: shared temps + cross-expr CSE
    gens = []
    out_meta = []
    results: List = [None] * len(exprs)
    for i, e in enumerate(exprs):
        base = e
        while isinstance(base, Alias):
            base = base.child
        if isinstance(base, (ColumnRef, Literal)):
            continue          # zero-copy via normal eval
        mark = (len(comp.cols), dict(comp.col_ix), len(cg.lines), cg.tmp,
                dict(cg.load_cache), dict(cg.expr_cache))
        try:
            f = e.to_field(batch.schema)
            out_dt = f.dtype
            tdt = out_dt.to_torch() if out_dt.is_fixed_width() else None
            if tdt not in _OUT_CODES or out_dt.is_decimal():
                raise _Bail("out dtype")
            _register_cols(comp, e)
            lines_before = len(cg.lines)
            val, vld = cg.gen(e)
            produces_null = vld is not None
            gens.append((cg.lines[lines_before:], val, vld))
            out_meta.append((i, f.name, out_dt, _OUT_CODES[tdt],
                             produces_null))
        except _Bail:
            comp.cols = comp.cols[:mark[0]]
            comp.col_ix = mark[1]
            cg.lines = cg.lines[:mark[2]]
            cg.tmp = mark[3]
            cg.load_cache = mark[4]
            cg.expr_cache = mark[5]
    if not out_meta or len(cg.lines) < 3:
        return None

    src = _gen_source(comp, gens, out_meta)
    datas = [s.data for s in comp.cols]
    valids = [s.validity for s in comp.cols]
    out_codes = [m[3] for m in out_meta]
    out_need_valid = [1 if m[4] else 0 for m in out_meta]
    n = len(batch)
    try:
        flat = native.fused_eval_jit(src, datas, valids, out_codes,
                                     out_need_valid, n)
    except RuntimeError:
        return None        # hiprtc unavailable/failed: interpreter path
    from ..series import Series
    for k, (i, name, out_dt, code, pn) in enumerate(out_meta):
        data = flat[2 * k]
        valid = flat[2 * k + 1] if pn else None
        if valid is not None and bool(valid.all().item()):
            valid = None
        if out_dt.to_torch() != data.dtype:
            data = data.to(out_dt.to_torch())
        results[i] = Series(name, out_dt, data=data, validity=valid)
    for i, e in enumerate(exprs):
        if results[i] is None:
            results[i] = e.evaluate(batch)
    return results


def _register_cols(comp: "_Compiler", e: ExprNode):
    """Validate + register every column referenced by e (raises _Bail on
    unfusable columns) without emitting opcodes."""
    if isinstance(e, ColumnRef):
        ins_mark = len(comp.ins)
        d_mark = comp.depth
        comp.col(e.name)
        comp.ins = comp.ins[:ins_mark]
        comp.depth = d_mark
        return
    if isinstance(e, Literal):
        comp._lit_value(e)      # validates the literal kind
        return
    if isinstance(e, BinaryOp):
        if e.op not in _BIN:
            raise _Bail(f"binop {e.op}")
        lt = e.left.to_field(comp.batch.schema).dtype
        rt = e.right.to_field(comp.batch.schema).dtype
        if lt.is_temporal() or rt.is_temporal():
            if lt.kind != rt.kind or \
                    getattr(lt, "timeunit", None) != \
                    getattr(rt, "timeunit", None):
                raise _Bail("temporal unit mismatch")
            if e.op not in ("eq", "ne", "lt", "le", "gt", "ge", "sub"):
                raise _Bail("temporal arithmetic")
    if isinstance(e, Cast):
        src = e.child.to_field(comp.batch.schema).dtype
        dst = e.dtype
        ok = (src.is_numeric() or src.is_boolean() or src.is_temporal()) \
            and (dst.is_numeric() and not dst.is_decimal())
        if not ok:
            raise _Bail(f"cast {src}->{dst}")
        if dst.is_integer() and not (src.is_integer() or src.is_boolean()
                                     or src.is_temporal()):
            raise _Bail("float->int cast")
    if isinstance(e, IsIn):
        vals = getattr(e, "values", None)
        if not vals or len(vals) > 8 or any(
                not isinstance(v, (int, float, bool)) or
                (isinstance(v, int) and abs(v) >= (1 << 53))
                for v in vals):
            raise _Bail("is_in")
    if not isinstance(e, (Alias, ColumnRef, Literal, BinaryOp, Not, IsNull,
                          FillNull, IfElse, Between, IsIn, Cast)):
        raise _Bail(type(e).__name__)
    for c in e.children():
        _register_cols(comp, c)
