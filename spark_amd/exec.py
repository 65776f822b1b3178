"""Host-side mirror of Spark's columnar physical-operator contract.

In a Spark deployment this layer is Scala: exec-node subclasses registered by
a ColumnarRule through SparkSessionExtensions.injectColumnar
(SparkSessionExtensions.scala:116,168), each overriding
supportsColumnar=true (SparkPlan.scala:92) and
doExecuteColumnar(): RDD[ColumnarBatch] (SparkPlan.scala:359) and calling
libgpuq over JNI (see INTEGRATION.md for that binding). This Python mirror
keeps the same node names, the same required-distribution semantics and the
same batch lifetime contract so the C-ABI is exercised exactly as the Scala
layer would; it is what tests and the TPC-H-style pipelines drive.

No CPU fallback anywhere: executing these nodes without the HIP engine (or
without a GPU) raises.
"""
from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

import torch


class ColumnarBatch:
    """Device-resident batch (ColumnarBatch.java:61-123 access contract;
    Arrow-style columns: dense data + optional validity bitmap)."""

    def __init__(self, columns: Dict[str, torch.Tensor],
                 validity: Optional[Dict[str, torch.Tensor]] = None):
        self._cols = columns
        self._validity = validity or {}
        ns = {t.numel() for t in columns.values()}
        assert len(ns) == 1, "ragged batch"
        self._num_rows = ns.pop()
        self._closed = False

    def num_rows(self) -> int:
        return self._num_rows

    def column(self, name: str) -> torch.Tensor:
        assert not self._closed, "batch used after close()"
        return self._cols[name]

    def validity(self, name: str):
        return self._validity.get(name)

    def columns(self):
        return dict(self._cols)

    def close(self):
        # consumer closes (Columnar.scala:224-240 ColumnarToRowExec pattern);
        # device memory returns to the torch pool
        self._cols = {}
        self._validity = {}
        self._closed = True


@dataclass
class Distribution:
    """requiredChildDistribution analog (SparkPlan.scala:180 contract)."""
    kind: str                      # "unspecified" | "clustered" | "ordered"
    keys: Tuple[str, ...] = ()


class SparkPlan:
    """Physical operator (execution/SparkPlan.scala:65). Single-partition-
    per-rank execution model: executeColumnar() yields this rank's batches."""

    def __init__(self, *children: "SparkPlan"):
        self.children = list(children)

    @property
    def output(self) -> List[str]:
        raise NotImplementedError

    @property
    def supports_columnar(self) -> bool:
        return False

    def required_child_distribution(self) -> List[Distribution]:
        return [Distribution("unspecified") for _ in self.children]

    def execute_columnar(self):
        """doExecuteColumnar (SparkPlan.scala:359)."""
        raise NotImplementedError(
            f"{type(self).__name__} does not support columnar execution")


# ---------------- CPU placeholders the rule replaces ----------------
# These mirror the Catalyst-produced nodes; they cannot execute here (the
# JVM engine isn't present) — they exist so the ColumnarRule's rewrite is
# the same shape as in Spark (SparkSessionExtensionSuite.scala:959-1000).

@dataclass
class SortOrder:
    key: str
    descending: bool = False
    nulls_first: Optional[bool] = None   # SortOrder.scala:35-45 defaults

    def __post_init__(self):
        if self.nulls_first is None:
            self.nulls_first = not self.descending


class _CpuNode(SparkPlan):
    def execute_columnar(self):
        raise RuntimeError(f"{type(self).__name__} is a CPU placeholder; "
                           "apply the ColumnarRule first (no CPU fallback)")


class SortExec(_CpuNode):
    def __init__(self, sort_order: SortOrder, global_sort: bool, child):
        super().__init__(child)
        self.sort_order, self.global_sort = sort_order, global_sort

    @property
    def output(self):
        return self.children[0].output


class HashAggregateExec(_CpuNode):
    def __init__(self, group_key: str, aggs: List[Tuple[str, str]], mode: str, child,
                 capacity: Optional[int] = None):
        super().__init__(child)
        self.group_key, self.aggs, self.mode = group_key, aggs, mode
        self.capacity = capacity

    @property
    def output(self):
        return [self.group_key] + [f"{fn}({col})" for fn, col in self.aggs]


class ShuffledHashJoinExec(_CpuNode):
    def __init__(self, left_key: str, right_key: str, build_side: str,
                 left, right):
        super().__init__(left, right)
        self.left_key, self.right_key, self.build_side = left_key, right_key, build_side

    @property
    def output(self):
        return self.children[0].output + self.children[1].output


class SortMergeJoinExec(_CpuNode):
    """CPU placeholder (joins/SortMergeJoinExec.scala:135). The GPU plan
    replaces SMJ with the hash join — both fill the same equi-join plan slot
    with identical ClusteredDistribution requirements (ShuffledJoin.scala:
    57-69), and on GPU the hash build/probe dominates sort+merge for
    co-partitioned batches."""

    def __init__(self, left_key: str, right_key: str, left, right):
        super().__init__(left, right)
        self.left_key, self.right_key = left_key, right_key

    @property
    def output(self):
        return self.children[0].output + self.children[1].output


class ShuffleExchangeExec(_CpuNode):
    def __init__(self, keys: Tuple[str, ...], num_partitions: int, child):
        super().__init__(child)
        self.keys, self.num_partitions = keys, num_partitions

    @property
    def output(self):
        return self.children[0].output


class FilterExec(_CpuNode):
    def __init__(self, col: str, op: str, literal, child):
        super().__init__(child)
        self.col, self.op, self.literal = col, op, literal

    @property
    def output(self):
        return self.children[0].output


class ProjectExec(_CpuNode):
    """projections: list of (out_name, a, op, b_or_None, literal_or_None) plus
    pass-through column names."""

    def __init__(self, projections, child):
        super().__init__(child)
        self.projections = projections

    @property
    def output(self):
        return [p if isinstance(p, str) else p[0] for p in self.projections]


class RangeExec(_CpuNode):
    """CPU placeholder for RangeExec (basicPhysicalOperators.scala:630)."""

    def __init__(self, n: int, start: int = 0, step: int = 1, name: str = "id"):
        super().__init__()
        self.n, self.start, self.step, self.name = n, start, step, name

    @property
    def output(self):
        return [self.name]


class BroadcastHashJoinExec(_CpuNode):
    """CPU placeholder (joins/BroadcastHashJoinExec.scala:40): inner
    equi-join with a broadcast (small) build side."""

    def __init__(self, left_key: str, right_key: str, build_side: str,
                 left, right):
        super().__init__(left, right)
        assert build_side in ("left", "right")
        self.left_key, self.right_key, self.build_side = left_key, right_key, build_side

    @property
    def output(self):
        return self.children[0].output + self.children[1].output


class BroadcastExchangeExec(_CpuNode):
    def __init__(self, child):
        super().__init__(child)

    @property
    def output(self):
        return self.children[0].output


class InputBatches(SparkPlan):
    """Leaf: pre-materialized device batches (scan stand-in)."""

    def __init__(self, batches: List[ColumnarBatch]):
        super().__init__()
        self._batches = batches

    @property
    def output(self):
        return list(self._batches[0].columns().keys())

    @property
    def supports_columnar(self):
        return True

    def execute_columnar(self):
        yield from self._batches


# ---------------- GPU exec nodes ----------------

class GpuSortExec(SparkPlan):
    """Replaces SortExec (SortExec.scala:75-126): radix-eligible single-key
    sort (canUseRadixSort analog: int64/float64 key)."""

    def __init__(self, sort_order: SortOrder, global_sort: bool, child):
        super().__init__(child)
        self.sort_order, self.global_sort = sort_order, global_sort

    @property
    def output(self):
        return self.children[0].output

    @property
    def supports_columnar(self):
        return True

    def required_child_distribution(self):
        if self.global_sort:
            return [Distribution("ordered", (self.sort_order.key,))]
        return [Distribution("unspecified")]

    def execute_columnar(self):
        import torch.distributed as dist
        from . import gpuq
        o = self.sort_order
        for batch in self.children[0].execute_columnar():
            if (self.global_sort and dist.is_initialized()
                    and dist.get_world_size() > 1):
                # global ORDER BY: range exchange first (the EnsureRequirements
                # RangePartitioning insertion, exchange/EnsureRequirements.scala:296);
                # after the local sort below, rank-major order is global order
                from .exchange import range_exchange
                assert batch.validity(o.key) is None, (
                    "global ORDER BY with NULL keys across ranks: validity "
                    "does not travel through the exchange yet (round-2)")
                payload = {n_: t for n_, t in batch.columns().items()
                           if n_ != o.key}
                k, payload = range_exchange(batch.column(o.key), payload,
                                            desc=o.descending)
                cols = {o.key: k}
                cols.update(payload)
                batch.close()
                batch = ColumnarBatch(cols)
            keys = batch.column(o.key)
            perm, skeys = gpuq.sort_perm(keys, desc=o.descending,
                                         nulls_first=o.nulls_first,
                                         key_validity=batch.validity(o.key))
            cols = {o.key: skeys}
            for name, t in batch.columns().items():
                if name != o.key:
                    cols[name] = gpuq.gather(t, perm)
            batch.close()
            yield ColumnarBatch(cols)


class GpuHashAggregateExec(SparkPlan):
    """Replaces HashAggregateExec (HashAggregateExec.scala:99-151).
    mode: "partial" | "final" | "complete" (AggUtils.scala:126-195 split);
    "complete" = single-node partial+final in one table."""

    def __init__(self, group_key: str, aggs: List[Tuple[str, str]], mode: str,
                 child, capacity: Optional[int] = None):
        super().__init__(child)
        assert mode in ("partial", "final", "complete")
        self.group_key, self.aggs, self.mode = group_key, aggs, mode
        self.capacity = capacity

    @property
    def output(self):
        return [self.group_key] + [f"{fn}({col})" for fn, col in self.aggs]

    @property
    def supports_columnar(self):
        return True

    def required_child_distribution(self):
        if self.mode == "final":
            return [Distribution("clustered", (self.group_key,))]
        return [Distribution("unspecified")]

    def execute_columnar(self):
        from . import gpuq
        fns = {fn for fn, _ in self.aggs}
        assert fns <= {"sum", "count", "avg"}, f"unsupported aggs {fns}"
        cols_used = {col for _, col in self.aggs}
        if len(cols_used) > 1 or len(self.aggs) > 2:
            yield from self._execute_multi(gpuq)
            return
        val_col = next(col for fn, col in self.aggs)
        ops = 0
        if fns & {"sum", "avg"} or self.mode == "final":
            ops |= gpuq.AGG_SUM
        if fns & {"count", "avg"}:
            ops |= gpuq.AGG_COUNT
        for batch in self.children[0].execute_columnar():
            keys = batch.column(self.group_key)
            vals = batch.column(val_col)
            if vals.dtype == torch.int64:
                # COUNT over any column / final-mode merge of partial counts:
                # values ride as f64 (exact below 2^53); SUM(int64)'s int64
                # result type is a round-2 item (Sum.scala resultType)
                vals = gpuq.cast_i64_f64(vals)
            n = keys.numel()
            cap = self.capacity or (1 << max(10, int(n).bit_length()))
            out = gpuq.hash_agg(keys, vals, cap, ops=ops or gpuq.AGG_SUM,
                                key_validity=batch.validity(self.group_key),
                                val_validity=batch.validity(val_col))
            ok, okv, osum, osv, ocnt = out
            cols = {self.group_key: ok}
            for fn, col in self.aggs:
                if fn == "avg":
                    # Average.evaluateExpression: sum / cast(count)
                    # (catalyst/.../expressions/aggregate/Average.scala)
                    cols[f"{fn}({col})"] = gpuq.project_binop(
                        osum, "/", b=gpuq.cast_i64_f64(ocnt))
                else:
                    cols[f"{fn}({col})"] = osum if fn == "sum" else ocnt
            batch.close()
            yield ColumnarBatch(cols, validity={self.group_key: None})

    def _execute_multi(self, gpuq):
        """multi-accumulator path (one gpuq_hash_agg_multi pass): several
        aggregate expressions over multiple value columns (the Q1 shape)."""
        for batch in self.children[0].execute_columnar():
            keys = batch.column(self.group_key)
            specs = []
            slots = []  # (fn, col, acc indices)
            for fn, col in self.aggs:
                t = batch.column(col)
                if fn == "avg" and t.dtype == torch.int64:
                    t = gpuq.cast_i64_f64(t)   # AVG divides f64 sums
                if fn == "sum":
                    specs.append(("sum", t)); slots.append((fn, col, [len(specs) - 1]))
                elif fn == "count":
                    specs.append(("count", t, batch.validity(col)))
                    slots.append((fn, col, [len(specs) - 1]))
                else:  # avg = sum + count
                    specs.append(("sum", t))
                    specs.append(("count", t, batch.validity(col)))
                    slots.append((fn, col, [len(specs) - 2, len(specs) - 1]))
            n = keys.numel()
            cap = self.capacity or (1 << max(10, int(n).bit_length()))
            ok, okv, accs = gpuq.hash_agg_multi(
                keys, specs, cap, key_validity=batch.validity(self.group_key),
                max_groups=min(n, cap) + 2)
            cols = {self.group_key: ok}
            for fn, col, idx in slots:
                if fn == "avg":
                    cols[f"avg({col})"] = gpuq.project_binop(
                        accs[idx[0]], "/", b=gpuq.cast_i64_f64(accs[idx[1]]))
                else:
                    cols[f"{fn}({col})"] = accs[idx[0]]
            batch.close()
            yield ColumnarBatch(cols, validity={self.group_key: None})


class GpuShuffleExchangeExec(SparkPlan):
    """Replaces ShuffleExchangeExec (ShuffleExchangeExec.scala:277-470):
    on-device radix partition + RCCL all-to-all (spark_amd/exchange.py).
    Runs only under torch.distributed; num_partitions == world size."""

    def __init__(self, keys: Tuple[str, ...], child):
        super().__init__(child)
        assert len(keys) == 1, "round 1: single int64 partition key"
        self.keys = keys

    @property
    def output(self):
        return self.children[0].output

    @property
    def supports_columnar(self):
        return True

    # --- ShuffleExchangeLike mirror (ShuffleExchangeExec.scala:51-151) ---
    # AQE materializes each shuffle stage and reads its map-output stats
    # (AdaptiveSparkPlanExec.scala:784, QueryStageExec.materialize :69);
    # these fields are what a Scala GpuShuffleExchangeExec would return from
    # mapOutputStatisticsFuture/runtimeStatistics, with per-partition byte
    # sizes taken from gpuq_partition_perm's counts (no per-row CPU work).

    @property
    def num_mappers(self) -> int:
        import torch.distributed as dist
        return dist.get_world_size() if dist.is_initialized() else 1

    @property
    def num_partitions(self) -> int:
        return self.num_mappers

    def runtime_statistics(self):
        """(total bytes written, rows) of this rank's map output —
        ShuffleExchangeLike.runtimeStatistics (:146)."""
        return dict(self._stats) if hasattr(self, "_stats") else None

    def execute_columnar(self):
        import torch.distributed as dist
        from . import gpuq
        from .exchange import exchange_columns
        assert dist.is_initialized(), "GpuShuffleExchangeExec needs torch.distributed"
        key_name = self.keys[0]
        world = self.num_partitions
        for batch in self.children[0].execute_columnar():
            key = batch.column(key_name)
            perm, counts = gpuq.partition_perm(key, world,
                                               key_validity=batch.validity(key_name))
            cols = {key_name: gpuq.gather(key, perm)}
            row_bytes = 0
            for name, t in batch.columns().items():
                if name != key_name:
                    cols[name] = gpuq.gather(t, perm)
                row_bytes += t.element_size()
            in_splits = counts.cpu().tolist()
            self._stats = {"bytes_by_partition": [c * row_bytes for c in in_splits],
                           "rows_written": int(sum(in_splits))}
            out, _ = exchange_columns(cols, in_splits)
            batch.close()
            yield ColumnarBatch(out)


class GpuShuffledHashJoinExec(SparkPlan):
    """Replaces ShuffledHashJoinExec inner join
    (ShuffledHashJoinExec.scala:103-132). Both children must already be
    clustered on the join key (ShuffledJoin.scala:57-69) — i.e. fed by (our)
    exchanges, as EnsureRequirements would arrange."""

    def __init__(self, left_key: str, right_key: str, build_side: str,
                 left, right):
        super().__init__(left, right)
        assert build_side in ("left", "right")
        self.left_key, self.right_key, self.build_side = left_key, right_key, build_side

    @property
    def output(self):
        return self.children[0].output + self.children[1].output

    @property
    def supports_columnar(self):
        return True

    def required_child_distribution(self):
        return [Distribution("clustered", (self.left_key,)),
                Distribution("clustered", (self.right_key,))]

    def execute_columnar(self):
        from . import gpuq
        left_b = list(self.children[0].execute_columnar())
        right_b = list(self.children[1].execute_columnar())
        assert len(left_b) == 1 and len(right_b) == 1, "one batch per partition"
        lb, rb = left_b[0], right_b[0]
        build, probe = (lb, rb) if self.build_side == "left" else (rb, lb)
        bkey = self.left_key if self.build_side == "left" else self.right_key
        pkey = self.right_key if self.build_side == "left" else self.left_key
        bk = build.column(bkey)
        bn = bk.numel()
        cap = 1 << max(4, int(bn * 2 - 1).bit_length() if bn else 4)
        ws = gpuq.join_build(bk, cap, key_validity=build.validity(bkey))
        pk = probe.column(pkey)
        out_cap = max(int(probe.num_rows() * 2) + 64, 64)
        while True:
            op, ob, nm = gpuq.join_probe(pk, ws, cap, bn, out_cap,
                                         key_validity=probe.validity(pkey))
            if op is not None:
                break
            out_cap = nm + 64
        cols = {}
        for name, t in build.columns().items():
            cols[name] = gpuq.gather(t, ob)
        for name, t in probe.columns().items():
            if name in cols:
                name = f"{name}#probe"
            cols[name] = gpuq.gather(t, op)
        lb.close(), rb.close()
        yield ColumnarBatch(cols)


class GpuRangeExec(SparkPlan):
    """Replaces RangeExec: generates the id column directly in HBM (the
    scan-feed adjacency, SURVEY §8(f).1 — no RowToColumnar CPU tax)."""

    def __init__(self, n: int, start: int = 0, step: int = 1, name: str = "id"):
        super().__init__()
        self.n, self.start, self.step, self.name = n, start, step, name

    @property
    def output(self):
        return [self.name]

    @property
    def supports_columnar(self):
        return True

    def execute_columnar(self):
        from . import gpuq
        yield ColumnarBatch({self.name: gpuq.range_i64(self.n, self.start, self.step)})


class GpuFilterExec(SparkPlan):
    """Replaces FilterExec for col OP literal predicates:
    stable compaction on device, then payload gather by the passing-row
    permutation."""

    def __init__(self, col: str, op: str, literal, child):
        super().__init__(child)
        self.col, self.op, self.literal = col, op, literal

    @property
    def output(self):
        return self.children[0].output

    @property
    def supports_columnar(self):
        return True

    def execute_columnar(self):
        from . import gpuq
        for batch in self.children[0].execute_columnar():
            perm, cnt = gpuq.filter_cmp(batch.column(self.col), self.op,
                                        self.literal,
                                        validity=batch.validity(self.col))
            cols = {name: gpuq.gather(t, perm)
                    for name, t in batch.columns().items()}
            batch.close()
            yield ColumnarBatch(cols)


class GpuProjectExec(SparkPlan):
    """Replaces ProjectExec for elementwise binary
    arithmetic; pass-through entries keep their column."""

    def __init__(self, projections, child):
        super().__init__(child)
        self.projections = projections

    @property
    def output(self):
        return [p if isinstance(p, str) else p[0] for p in self.projections]

    @property
    def supports_columnar(self):
        return True

    def execute_columnar(self):
        from . import gpuq
        for batch in self.children[0].execute_columnar():
            cols = {}
            for p in self.projections:
                if isinstance(p, str):
                    cols[p] = batch.column(p)
                else:
                    out_name, a, op, b, lit = p
                    cols[out_name] = gpuq.project_binop(
                        batch.column(a), op,
                        b=batch.column(b) if b is not None else None,
                        literal=lit)
            batch.close()
            yield ColumnarBatch(cols)


class GpuBroadcastExchangeExec(SparkPlan):
    """Replaces BroadcastExchangeExec: RCCL all-gather of the build-side
    batch across ranks (single-rank: identity). Unlike the shuffled
    exchange there is no partitioning — every rank gets the whole
    relation."""

    def __init__(self, child):
        super().__init__(child)

    @property
    def output(self):
        return self.children[0].output

    @property
    def supports_columnar(self):
        return True

    def execute_columnar(self):
        import torch.distributed as dist
        from .exchange import broadcast_gather
        for batch in self.children[0].execute_columnar():
            if dist.is_initialized() and dist.get_world_size() > 1:
                cols = broadcast_gather(batch.columns())
                batch.close()
                yield ColumnarBatch(cols)
            else:
                yield batch


class GpuBroadcastHashJoinExec(GpuShuffledHashJoinExec):
    """Replaces BroadcastHashJoinExec: identical device-side build/probe
    kernels; the build child is expected to be a GpuBroadcastExchangeExec,
    and the PROBE side needs no distribution at all (BroadcastDistribution,
    joins/BroadcastHashJoinExec.scala:60-66)."""

    def required_child_distribution(self):
        dists = [Distribution("unspecified"), Distribution("unspecified")]
        dists[0 if self.build_side == "left" else 1] = Distribution("broadcast")
        return dists


class GpuColumnarRule:
    """The injected rule (ColumnarRule, Columnar.scala:36-50; injection via
    SparkSessionExtensions.injectColumnar:168; applied at
    QueryExecution.scala:798 / AdaptiveSparkPlanExec.scala:184-186).
    preColumnarTransitions swaps CPU nodes for GPU subclasses — the
    SparkSessionExtensionSuite.scala:959-1000 pattern."""

    def pre_columnar_transitions(self, plan: SparkPlan) -> SparkPlan:
        children = [self.pre_columnar_transitions(c) for c in plan.children]
        if isinstance(plan, SortExec):
            return GpuSortExec(plan.sort_order, plan.global_sort, *children)
        if isinstance(plan, HashAggregateExec):
            return GpuHashAggregateExec(plan.group_key, plan.aggs, plan.mode,
                                        *children, capacity=plan.capacity)
        if isinstance(plan, ShuffledHashJoinExec):
            return GpuShuffledHashJoinExec(plan.left_key, plan.right_key,
                                           plan.build_side, *children)
        if isinstance(plan, SortMergeJoinExec):
            # SMJ -> GPU hash join (same slot, same required distribution;
            # build on the right side as SHJ's default would choose)
            return GpuShuffledHashJoinExec(plan.left_key, plan.right_key,
                                           "right", *children)
        if isinstance(plan, ShuffleExchangeExec):
            return GpuShuffleExchangeExec(plan.keys, *children)
        if isinstance(plan, FilterExec):
            return GpuFilterExec(plan.col, plan.op, plan.literal, *children)
        if isinstance(plan, ProjectExec):
            return GpuProjectExec(plan.projections, *children)
        if isinstance(plan, RangeExec):
            return GpuRangeExec(plan.n, plan.start, plan.step, plan.name)
        if isinstance(plan, BroadcastExchangeExec):
            return GpuBroadcastExchangeExec(*children)
        if isinstance(plan, BroadcastHashJoinExec):
            return GpuBroadcastHashJoinExec(plan.left_key, plan.right_key,
                                            plan.build_side, *children)
        plan.children = children
        return plan

    def post_columnar_transitions(self, plan: SparkPlan) -> SparkPlan:
        # Spark inserts Row<->Columnar transitions here (Columnar.scala:564-614);
        # in this mirror every node is columnar, nothing to insert.
        return plan
