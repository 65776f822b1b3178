"""CPU tests of the host operator mirror: the ColumnarRule rewrite
(SparkSessionExtensionSuite.scala:959-1000 pattern) and the no-CPU-fallback
contract. No GPU calls here."""
import pytest

from spark_amd import exec as gx


def make_cpu_plan():
    scan = gx.InputBatches.__new__(gx.InputBatches)  # leaf without data
    gx.SparkPlan.__init__(scan)
    scan._batches = []
    left = gx.ShuffleExchangeExec(("k",), 8, scan)
    right = gx.ShuffleExchangeExec(("k",), 8, scan)
    join = gx.ShuffledHashJoinExec("k", "k", "right", left, right)
    agg = gx.HashAggregateExec("k", [("sum", "v")], "complete", join)
    sort = gx.SortExec(gx.SortOrder("k"), True, agg)
    return sort


def test_rule_replaces_all_hot_path_nodes():
    plan = gx.GpuColumnarRule().pre_columnar_transitions(make_cpu_plan())
    assert isinstance(plan, gx.GpuSortExec)
    agg = plan.children[0]
    assert isinstance(agg, gx.GpuHashAggregateExec)
    join = agg.children[0]
    assert isinstance(join, gx.GpuShuffledHashJoinExec)
    assert all(isinstance(c, gx.GpuShuffleExchangeExec) for c in join.children)
    assert all(c.supports_columnar for c in [plan, agg, join] + join.children)


def test_cpu_placeholders_refuse_to_execute():
    plan = make_cpu_plan()
    with pytest.raises(RuntimeError, match="no CPU fallback"):
        next(plan.execute_columnar())


def test_required_distributions_mirror_reference():
    plan = gx.GpuColumnarRule().pre_columnar_transitions(make_cpu_plan())
    # global sort requires ordered distribution (SortExec.scala:54-55)
    assert plan.required_child_distribution()[0].kind == "ordered"
    join = plan.children[0].children[0]
    # shuffled join requires clustering on keys both sides (ShuffledJoin.scala:57-69)
    dists = join.required_child_distribution()
    assert [d.kind for d in dists] == ["clustered", "clustered"]


def test_sort_order_null_defaults():
    # SortOrder.scala:35-45: asc => nulls first, desc => nulls last
    assert gx.SortOrder("k", descending=False).nulls_first is True
    assert gx.SortOrder("k", descending=True).nulls_first is False


def test_batch_lifetime():
    import numpy as np
    import torch
    b = gx.ColumnarBatch({"k": torch.arange(4)})
    assert b.num_rows() == 4
    b.close()
    with pytest.raises(AssertionError):
        b.column("k")


def test_sort_merge_join_replaced_by_gpu_hash_join():
    # SMJ and SHJ fill the same equi-join plan slot; the
    # rule maps both to the GPU hash join. SMJ additionally advertises
    # outputOrdering on the streamed keys (SortMergeJoinExec.scala:83), so
    # the rule re-sorts the hash join's output to honor that contract —
    # the replacement's output_ordering must cover what SMJ declared.
    scan = gx.InputBatches.__new__(gx.InputBatches)
    gx.SparkPlan.__init__(scan)
    scan._batches = []
    smj = gx.SortMergeJoinExec("a", "b", scan, scan)
    declared = smj.output_ordering
    assert [o.key for o in declared] == ["a"]
    plan = gx.GpuColumnarRule().pre_columnar_transitions(smj)
    assert isinstance(plan, gx.GpuSortExec)
    assert [o.key for o in plan.output_ordering] == ["a"]
    join = plan.children[0]
    assert isinstance(join, gx.GpuShuffledHashJoinExec)
    assert join.left_key == "a" and join.right_key == "b"
    # opt-out for plans that provably don't rely on SMJ ordering
    plan2 = gx.GpuColumnarRule(preserve_smj_ordering=False) \
        .pre_columnar_transitions(gx.SortMergeJoinExec("a", "b", scan, scan))
    assert isinstance(plan2, gx.GpuShuffledHashJoinExec)


def test_broadcast_join_rule():
    scan = gx.InputBatches.__new__(gx.InputBatches)
    gx.SparkPlan.__init__(scan)
    scan._batches = []
    bx = gx.BroadcastExchangeExec(scan)
    bhj = gx.BroadcastHashJoinExec("a", "b", "right", scan, bx)
    plan = gx.GpuColumnarRule().pre_columnar_transitions(bhj)
    assert isinstance(plan, gx.GpuBroadcastHashJoinExec)
    assert isinstance(plan.children[1], gx.GpuBroadcastExchangeExec)
    dists = plan.required_child_distribution()
    assert dists[1].kind == "broadcast" and dists[0].kind == "unspecified"
