"""GPU golden parity: the exec mirror runs the extracted reference plans
through the HIP engine and must reproduce the reference-emitted rows
(sql-tests/results/*.sql.out — see tools/extract_golden.py), in complete
mode AND through the partial -> final merge split (AggUtils.scala two-phase
contract).

String columns are dictionary-encoded to int64 ids in UTF8-binary order
(ids order exactly like Spark's string ordering), decoded before
comparison — the engine operates on int64/float64 columns, as a Scala
host layer would hand it dictionary-encoded vectors."""
import numpy as np
import pytest

from golden_sql_util import (assert_rows_match, build_dictionary,
                             expected_rows, load_cases, table_col_types)

torch = pytest.importorskip("torch")
pytestmark = pytest.mark.gpu

from spark_amd import exec as gx  # noqa: E402

CASES = load_cases()


def pack_bits(mask: np.ndarray) -> torch.Tensor:
    return torch.from_numpy(np.packbits(mask, bitorder="little")).cuda()


def unpack_bits(bits: torch.Tensor, n: int) -> np.ndarray:
    return np.unpackbits(bits.cpu().numpy(), count=n, bitorder="little") \
        .astype(bool)


def make_batch(tdef, qualifier, dct):
    cols, validity = {}, {}
    types = table_col_types(tdef)
    n = len(tdef["rows"])
    for ci, cname in enumerate(tdef["cols"]):
        vals = [row[ci] for row in tdef["rows"]]
        data = np.zeros(n, dtype=np.int64)
        mask = np.ones(n, dtype=bool)
        for i, v in enumerate(vals):
            if v is None:
                mask[i] = False
            elif types[ci] == "string":
                data[i] = dct[v]
            else:
                data[i] = v
        name = f"{qualifier}.{cname}"
        cols[name] = torch.from_numpy(data).cuda()
        if not mask.all():
            validity[name] = pack_bits(mask)
    return gx.ColumnarBatch(cols, validity=validity or None), types


def resolve(out_names, ref):
    """resolve a plan reference against an exec node's output list:
    exact, suffix, or fn(base) against fn(qualified)."""
    if ref in out_names:
        return ref
    hits = [c for c in out_names if c.endswith("." + ref)]
    if len(hits) == 1:
        return hits[0]
    import re
    m = re.fullmatch(r"(\w+)\((\w+)\)", ref)
    if m:
        fn, base = m.groups()
        hits = [c for c in out_names
                if re.fullmatch(rf"{re.escape(fn)}\(((\w+\.)?{re.escape(base)})\)", c)]
        if len(hits) == 1:
            return hits[0]
    raise KeyError(f"cannot resolve {ref!r} in {out_names}")


def compile_plan(plan, case, dct, agg_mode):
    op = plan["op"]
    if op == "scan":
        tdef = case["tables"][plan["table"]]
        batch, _ = make_batch(tdef, plan.get("alias") or plan["table"], dct)
        return gx.InputBatches([batch])
    if op == "filter":
        child = compile_plan(plan["child"], case, dct, agg_mode)
        col = resolve(child.output, plan["col"])
        lit = plan["lit"]
        if isinstance(lit, str):
            lit = dct.get(lit, -1)   # absent string matches nothing
        return gx.FilterExec(col, plan["cmp"], lit, child)
    if op == "join":
        left = compile_plan(plan["left"], case, dct, agg_mode)
        right = compile_plan(plan["right"], case, dct, agg_mode)
        lk = resolve(left.output, plan["lkey"])
        rk = resolve(right.output, plan["rkey"])
        kind = plan.get("kind", "inner")
        jt = {"inner": "inner", "left": "left_outer", "right": "right_outer",
              "semi": "left_semi", "anti": "left_anti"}[kind]
        build = "left" if kind == "right" else "right"
        return gx.ShuffledHashJoinExec(lk, rk, build, left, right,
                                       join_type=jt)
    if op == "agg":
        child = compile_plan(plan["child"], case, dct, agg_mode)
        keys = tuple(resolve(child.output, k) for k in plan["keys"])
        aggs = [(fn, resolve(child.output, c) if c else None)
                for fn, c in plan["aggs"]]
        gk = keys if len(keys) != 1 else keys[0]
        if not keys:
            gk = None
        if agg_mode == "split":
            partial = gx.HashAggregateExec(gk, aggs, "partial", child)
            return gx.HashAggregateExec(gk, aggs, "final", partial)
        return gx.HashAggregateExec(gk, aggs, "complete", child)
    if op == "sort":
        child = compile_plan(plan["child"], case, dct, agg_mode)
        orders = [gx.SortOrder(resolve(child.output, r), desc, nf)
                  for r, desc, nf in plan["orders"]]
        return gx.SortExec(orders, False, child)
    raise ValueError(op)


def run_case(case, agg_mode):
    dct, dlist = build_dictionary(case)
    top = case["plan"]
    assert top["op"] == "project"
    node = compile_plan(top["child"], case, dct, agg_mode)
    node = gx.GpuColumnarRule().pre_columnar_transitions(node)
    batches = list(node.execute_columnar())
    assert len(batches) == 1
    b = batches[0]
    n = b.num_rows()
    out_names = list(b.columns().keys())
    types = case["schema_types"]
    rows = []
    colvals = []
    for i, it in enumerate(top["items"]):
        typ = types[i]
        if it["kind"] == "lit":
            colvals.append([it["val"]] * n)
            continue
        name = resolve(out_names, it["ref"])
        t = b.column(name)
        vals = t.cpu().numpy()
        valid = (unpack_bits(b.validity(name), n)
                 if b.validity(name) is not None else np.ones(n, dtype=bool))
        out = []
        for j in range(n):
            if not valid[j]:
                out.append(None)
            elif typ == "string":
                out.append(dlist[int(vals[j])])
            elif typ in ("double", "float"):
                out.append(float(vals[j]))
            else:
                out.append(int(vals[j]))
        colvals.append(out)
    rows = list(zip(*colvals)) if colvals and n else []
    return rows


@pytest.mark.parametrize("agg_mode", ["complete", "split"])
@pytest.mark.parametrize("case", CASES, ids=[c["_id"] for c in CASES])
def test_gpu_reproduces_reference_output(case, agg_mode):
    got = run_case(case, agg_mode)
    assert_rows_match(got, expected_rows(case), case)
