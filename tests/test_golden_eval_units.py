"""CPU unit tests for the golden evaluator's join-kind semantics (the
extracted fixtures cover inner/left/right/semi/anti; ON-based FULL joins
don't appear in the supported golden subset, so pin the evaluator here
with hand-built expectations — JoinType dispatch per joins/*.scala)."""
from oracle import golden_eval as ge

TABLES = {
    "l": {"cols": ["k", "a"], "rows": [[1, 10], [2, 20], [None, 30], [5, 50]]},
    "r": {"cols": ["k", "b"], "rows": [[1, 100], [1, 101], [3, 300], [None, 400]]},
}


def join(kind):
    plan = {"op": "join", "kind": kind,
            "left": {"op": "scan", "table": "l"},
            "right": {"op": "scan", "table": "r"},
            "lkey": "l.k", "rkey": "r.k", "merged": False}
    f = ge.evaluate(plan, TABLES)
    cols = ["l.k", "l.a", "r.k", "r.b"] if kind not in ("semi", "anti") \
        else ["l.k", "l.a"]
    return sorted(zip(*[f.cols[c] for c in cols]),
                  key=lambda t: tuple((x is None, x or 0) for x in t))


def test_inner():
    assert join("inner") == [(1, 10, 1, 100), (1, 10, 1, 101)]


def test_left():
    assert join("left") == [(1, 10, 1, 100), (1, 10, 1, 101),
                            (2, 20, None, None), (5, 50, None, None),
                            (None, 30, None, None)]


def test_right():
    assert join("right") == [(1, 10, 1, 100), (1, 10, 1, 101),
                             (None, None, 3, 300), (None, None, None, 400)]


def test_full():
    assert join("full") == [(1, 10, 1, 100), (1, 10, 1, 101),
                            (2, 20, None, None), (5, 50, None, None),
                            (None, 30, None, None),
                            (None, None, 3, 300), (None, None, None, 400)]


def test_semi_anti():
    assert join("semi") == [(1, 10)]
    # anti keeps unmatched left rows INCLUDING the NULL-key row
    # (the non-null-aware anti)
    assert join("anti") == [(2, 20), (5, 50), (None, 30)]
