"""Golden-case reference evaluator — TEST INFRASTRUCTURE ONLY.

Evaluates the normalized plans `tools/extract_golden.py` extracts from the
reference's own sql-tests golden files
(/root/reference/sql/core/src/test/resources/sql-tests/results/*.sql.out,
harness: SQLQueryTestSuite.scala:146-155) over the literal tables those
files define. This is the CPU restatement of the hot-path SQL semantics
for those cases:

- GROUP BY groups NULL keys together, one group per distinct key tuple
  (TungstenAggregationIterator.scala:206; UnsafeRow key equality treats
  null == null);
- COUNT(col) counts non-null rows, COUNT(*)/COUNT(1) counts rows
  (Count.scala); SUM/MIN/MAX/AVG skip NULLs and are NULL iff no non-null
  input (Sum.scala:113-180, Min.scala, Max.scala, Average.scala);
- SUM(integral) -> bigint, AVG(integral) -> double (Sum.scala resultType,
  Average.scala);
- global aggregate (no grouping) emits exactly one row, also on empty
  input (AggUtils.scala emptyInputAggBuffer path);
- inner equi-join: NULL keys never match (HashedRelation.scala /
  SortMergeJoinExec.scala null-key handling);
- ORDER BY: SortOrder defaults NULLS FIRST for ASC, NULLS LAST for DESC
  (SortOrder.scala:35-45); multi-key = lexicographic;
- WHERE col cmp literal: NULL operand -> row filtered out (three-valued
  logic collapses to false in a filter).

Pinning: the expected outputs in each fixture ARE reference-emitted
(committed .sql.out files), so every green assertion pins this evaluator —
and, through the exec-mirror tests, the GPU engine — to the reference's own
results. Only tests/ may import this module.
"""
from typing import Any, Dict, List, Optional, Tuple

NULL = None


class Frame:
    """columns: name -> list of python values (int/float/str/None)."""

    def __init__(self, cols: Dict[str, list]):
        self.cols = cols
        ns = {len(v) for v in cols.values()}
        assert len(ns) <= 1, "ragged frame"
        self.n = ns.pop() if ns else 0

    def resolve(self, ref: str) -> str:
        """resolve a possibly-unqualified column reference against
        qualified ('tbl.col') or bare column names."""
        if ref in self.cols:
            return ref
        hits = [c for c in self.cols if c.endswith("." + ref)]
        if len(hits) == 1:
            return hits[0]
        raise KeyError(f"unresolvable column {ref!r} in {list(self.cols)}")

    def col(self, ref: str) -> list:
        return self.cols[self.resolve(ref)]


def _load_table(tdef: dict, qualifier: str) -> Frame:
    cols = {f"{qualifier}.{c}": [] for c in tdef["cols"]}
    names = [f"{qualifier}.{c}" for c in tdef["cols"]]
    for row in tdef["rows"]:
        for name, v in zip(names, row):
            cols[name].append(v)
    return Frame(cols)


def _cmp(v, op: str, lit) -> bool:
    if v is NULL or lit is NULL:
        return False
    if op == "==":
        return v == lit
    if op == "!=":
        return v != lit
    if op == "<":
        return v < lit
    if op == "<=":
        return v <= lit
    if op == ">":
        return v > lit
    return v >= lit


def _agg_one(fn: str, vals: list):
    nn = [v for v in vals if v is not NULL]
    if fn == "count":
        return len(nn)
    if fn == "count*":
        return len(vals)
    if not nn:
        return NULL
    if fn == "sum":
        s = sum(nn)
        if all(isinstance(v, int) for v in nn):
            # bigint wrap-around (non-ansi Sum.scala on LongType)
            s = ((s + (1 << 63)) % (1 << 64)) - (1 << 63)
        return s
    if fn == "min":
        return min(nn)
    if fn == "max":
        return max(nn)
    if fn == "avg":
        return float(sum(nn)) / len(nn)
    raise ValueError(fn)


def _sort_key_factory(desc: bool, nulls_first: Optional[bool]):
    """list.sort(key=..., reverse=desc) is stable either way; NULL gets an
    extreme rank chosen so its OUTPUT position matches the SortOrder null
    placement (SortOrder.scala:35-45 defaults: NULLS FIRST for ASC, NULLS
    LAST for DESC)."""
    if nulls_first is None:
        nulls_first = not desc
    null_rank = 2 if (desc == nulls_first) else 0

    def key(v):
        if v is NULL:
            return (null_rank, 0)
        return (1, v)
    return key, desc


def evaluate(plan: dict, tables: Dict[str, dict]) -> Frame:
    op = plan["op"]
    if op == "scan":
        return _load_table(tables[plan["table"]], plan.get("alias") or plan["table"])
    if op == "filter":
        f = evaluate(plan["child"], tables)
        c = f.col(plan["col"])
        keep = [i for i in range(f.n) if _cmp(c[i], plan["cmp"], plan["lit"])]
        return Frame({k: [v[i] for i in keep] for k, v in f.cols.items()})
    if op == "join":
        # kinds (joins/*.scala JoinType dispatch): inner | left | right |
        # semi | anti. NULL keys never match; left/right preserve the
        # named side with NULLs on the other; semi/anti emit left rows
        # once by match existence (anti includes NULL-key left rows).
        lf = evaluate(plan["left"], tables)
        rf = evaluate(plan["right"], tables)
        lk, rk = lf.col(plan["lkey"]), rf.col(plan["rkey"])
        kind = plan.get("kind", "inner")
        from collections import defaultdict
        buckets = defaultdict(list)
        for j, v in enumerate(rk):
            if v is not NULL:
                buckets[v].append(j)
        li, ri = [], []
        for i, v in enumerate(lk):
            hits = buckets.get(v, ()) if v is not NULL else ()
            if kind == "semi":
                if hits:
                    li.append(i); ri.append(None)
                continue
            if kind == "anti":
                if not hits:
                    li.append(i); ri.append(None)
                continue
            if hits:
                for j in hits:
                    li.append(i); ri.append(j)
            elif kind in ("left", "full"):
                li.append(i); ri.append(None)
        if kind in ("right", "full"):
            matched_r = set(j for j in ri if j is not None)
            for j in range(rf.n):
                if j not in matched_r and rk[j] is not NULL:
                    li.append(None); ri.append(j)
                elif rk[j] is NULL:
                    li.append(None); ri.append(j)
        cols = {}
        for k, v in lf.cols.items():
            cols[k] = [v[i] if i is not None else NULL for i in li]
        if kind not in ("semi", "anti"):
            for k, v in rf.cols.items():
                # NATURAL/USING joins keep both key columns; the extractor
                # disambiguates references (bare key -> preserved side) and
                # expands * with the merged key once, matching Spark
                cols[k] = [v[j] if j is not None else NULL for j in ri]
        return Frame(cols)
    if op == "agg":
        f = evaluate(plan["child"], tables)
        keys = [f.resolve(k) for k in plan["keys"]]
        groups: Dict[Tuple, List[int]] = {}
        order: List[Tuple] = []
        if keys:
            kcols = [f.cols[k] for k in keys]
            for i in range(f.n):
                t = tuple(c[i] for c in kcols)
                if t not in groups:
                    groups[t] = []
                    order.append(t)
                groups[t].append(i)
        else:
            groups[()] = list(range(f.n))
            order.append(())
        out: Dict[str, list] = {k: [] for k in keys}
        for fn, c in plan["aggs"]:
            out[agg_name(fn, c)] = []
        for t in order:
            rows = groups[t]
            for k, v in zip(keys, t):
                out[k].append(v)
            for fn, c in plan["aggs"]:
                vals = ([f.col(c)[i] for i in rows] if fn != "count*"
                        else [1] * len(rows))
                out[agg_name(fn, c)].append(_agg_one(fn, vals))
        return Frame(out)
    if op == "sort":
        f = evaluate(plan["child"], tables)
        idx = list(range(f.n))
        for key_ref, desc, nf in reversed(plan["orders"]):
            c = f.col(key_ref)
            key, d = _sort_key_factory(desc, nf)
            idx.sort(key=lambda i: key(c[i]), reverse=d)
        return Frame({k: [v[i] for i in idx] for k, v in f.cols.items()})
    if op == "project":
        f = evaluate(plan["child"], tables)
        cols = {}
        for it in plan["items"]:
            name = it["as"]
            if it["kind"] == "col":
                cols[name] = list(f.col(it["ref"]))
            else:  # literal column
                cols[name] = [it["val"]] * f.n
        return Frame(cols)
    raise ValueError(f"bad plan op {op}")


def agg_name(fn: str, col: Optional[str]) -> str:
    if fn == "count*":
        return "count(1)"
    base = col.split(".")[-1] if col else col
    return f"{fn}({base})"
