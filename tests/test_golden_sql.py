"""Golden parity against the reference's own sql-tests outputs.

CPU half: the oracle golden evaluator (oracle/golden_eval.py) must
reproduce every committed reference-emitted result row
(sql-tests/results/*.sql.out via tools/extract_golden.py fixtures).
GPU half (tests/test_golden_sql_gpu.py): the exec mirror runs the same
plans through the HIP engine and must match the same rows — in complete
mode AND through the partial->final merge split.
"""
import pytest

from golden_sql_util import (assert_rows_match, expected_rows, load_cases)

CASES = load_cases()
assert CASES, "no golden fixtures — run tools/extract_golden.py"


@pytest.mark.parametrize("case", CASES, ids=[c["_id"] for c in CASES])
def test_oracle_reproduces_reference_output(case):
    from oracle import golden_eval
    frame = golden_eval.evaluate(case["plan"], case["tables"])
    names = [it["as"] for it in case["plan"]["items"]]
    got = list(zip(*[frame.cols[n] for n in names])) if frame.cols else []
    if frame.n == 0:
        got = []
    assert_rows_match(got, expected_rows(case), case)
