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

    @property
    def output_ordering(self) -> list:
        """outputOrdering (SparkPlan.scala:180 contract): the SortOrders
        this node's output satisfies. A replacement node must deliver AT
        LEAST what the node it replaces advertised — parents may have had
        their sorts elided against it (EnsureRequirements)."""
        return []

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
    """sort_order: one SortOrder or a list (multi-key ORDER BY,
    SortExec.scala sortOrder: Seq[SortOrder])."""

    def __init__(self, sort_order, global_sort: bool, child):
        super().__init__(child)
        self.sort_order, self.global_sort = sort_order, global_sort

    @property
    def sort_orders(self):
        return ([self.sort_order] if isinstance(self.sort_order, SortOrder)
                else list(self.sort_order))

    @property
    def output(self):
        return self.children[0].output

    @property
    def output_ordering(self):
        return self.sort_orders


class HashAggregateExec(_CpuNode):
    """group_key: one column name or a tuple of names (composite GROUP BY,
    the UnsafeRow key-tuple case)."""

    def __init__(self, group_key, aggs: List[Tuple[str, str]], mode: str, child,
                 capacity: Optional[int] = None):
        super().__init__(child)
        self.group_key, self.aggs, self.mode = group_key, aggs, mode
        self.capacity = capacity

    @property
    def group_keys(self):
        return (self.group_key,) if isinstance(self.group_key, str) else tuple(self.group_key)

    @property
    def output(self):
        return list(self.group_keys) + [f"{fn}({col})" for fn, col in self.aggs]


class ShuffledHashJoinExec(_CpuNode):
    """join_type: "inner" | "left_outer" | "right_outer" | "left_semi" |
    "left_anti" (ShuffledHashJoinExec.scala joinType; the preserved side
    must stream, so left-preserving types build on the right and
    vice versa)."""

    def __init__(self, left_key: str, right_key: str, build_side: str,
                 left, right, join_type: str = "inner"):
        super().__init__(left, right)
        self.left_key, self.right_key, self.build_side = left_key, right_key, build_side
        self.join_type = join_type

    @property
    def output(self):
        if self.join_type in ("left_semi", "left_anti"):
            return self.children[0].output
        return self.children[0].output + self.children[1].output


class SortMergeJoinExec(_CpuNode):
    """CPU placeholder (joins/SortMergeJoinExec.scala:135). The GPU plan
    replaces SMJ with the hash join — both fill the same equi-join plan slot
    with identical ClusteredDistribution requirements (ShuffledJoin.scala:
    57-69), and on GPU the hash build/probe dominates sort+merge for
    co-partitioned batches."""

    def __init__(self, left_key: str, right_key: str, left, right,
                 join_type: str = "inner"):
        super().__init__(left, right)
        self.left_key, self.right_key = left_key, right_key
        self.join_type = join_type

    @property
    def output(self):
        return self.children[0].output + self.children[1].output

    @property
    def output_ordering(self):
        # SMJ's output ordering follows the join type
        # (SortMergeJoinExec.scala outputOrdering via getKeyOrdering:
        # inner/left-ish -> left keys, RightOuter -> right keys,
        # FullOuter -> none)
        if self.join_type == "full_outer":
            return []
        if self.join_type == "right_outer":
            return [SortOrder(self.right_key)]
        return [SortOrder(self.left_key)]


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


class ParquetScanExec(_CpuNode):
    """CPU placeholder for FileSourceScanExec over a Parquet file
    (DataSourceScanExec.scala:735)."""

    def __init__(self, path: str, columns: Optional[List[str]] = None):
        super().__init__()
        self.path, self.columns = path, columns

    @property
    def output(self):
        if self.columns:
            return list(self.columns)
        import pyarrow.parquet as pq
        return [f.name for f in pq.ParquetFile(self.path).schema_arrow]


class GpuParquetScanExec(SparkPlan):
    """Replaces FileSourceScanExec's columnar Parquet path (SURVEY
    §8(f).1 full form; VectorizedParquetRecordReader.java:67): pyarrow
    decodes row groups on the host, columns normalize to the engine's
    types (decimal(<=18,s) -> scaled int64, date32 -> int64 days,
    dictionary strings -> int64 ids) and ride pinned-host -> HBM copies
    with their Arrow validity bitmaps. One ColumnarBatch per row group.
    Dictionaries for string columns are exposed on `dictionaries` after
    execution (host-side decode concern, not engine compute)."""

    def __init__(self, path: str, columns: Optional[List[str]] = None):
        super().__init__()
        self.path, self.columns = path, columns
        self.dictionaries: Dict[str, list] = {}

    @property
    def output(self):
        return ParquetScanExec(self.path, self.columns).output

    @property
    def supports_columnar(self):
        return True

    def execute_columnar(self):
        import pyarrow.parquet as pq
        from .parquet_io import read_row_group_to_device
        pf = pq.ParquetFile(self.path)
        for rg in range(pf.num_row_groups):
            cols, validity, dicts = read_row_group_to_device(
                pf, rg, columns=self.columns)
            self.dictionaries.update(dicts)
            yield ColumnarBatch(cols, validity=validity or None)


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

def concat_batches(batches):
    """Concatenate a partition's batches into one (whole-partition
    operators: sort, join build). Validity bitmaps concatenate through
    the u8 wire form (row counts are not byte-aligned)."""
    from . import gpuq
    if len(batches) == 1:
        return batches[0]
    names = list(batches[0].columns().keys())
    cols = {n: torch.cat([b.column(n) for b in batches]) for n in names}
    validity = {}
    for n in names:
        if any(b.validity(n) is not None for b in batches):
            parts = []
            for b in batches:
                v = b.validity(n)
                nn = b.num_rows()
                if v is None:
                    parts.append(torch.ones(nn, dtype=torch.uint8,
                                            device="cuda"))
                else:
                    parts.append(gpuq.bits_to_u8(v, nn))
            validity[n] = gpuq.u8_to_bits(torch.cat(parts))
    for b in batches:
        b.close()
    return ColumnarBatch(cols, validity=validity or None)


class GpuSortExec(SparkPlan):
    """Replaces SortExec (SortExec.scala:75-126): radix-eligible
    int64/float64 keys (canUseRadixSort analog). Multi-key ORDER BY
    composes as successive STABLE radix passes from the last sort key to
    the first — the stability the kernel guarantees makes the composition
    exactly the lexicographic order SortExec produces."""

    def __init__(self, sort_order, global_sort: bool, child):
        super().__init__(child)
        self.sort_order, self.global_sort = sort_order, global_sort

    @property
    def sort_orders(self):
        return ([self.sort_order] if isinstance(self.sort_order, SortOrder)
                else list(self.sort_order))

    @property
    def output(self):
        return self.children[0].output

    @property
    def output_ordering(self):
        return self.sort_orders

    @property
    def supports_columnar(self):
        return True

    def required_child_distribution(self):
        if self.global_sort:
            return [Distribution("ordered", tuple(o.key for o in self.sort_orders))]
        return [Distribution("unspecified")]

    @staticmethod
    def _sort_pass(batch, o):
        """one stable sort of the whole batch by o (keys + payload +
        validity bitmaps through the same permutation)."""
        from . import gpuq
        keys = batch.column(o.key)
        kvalid = batch.validity(o.key)
        perm, skeys = gpuq.sort_perm(keys, desc=o.descending,
                                     nulls_first=o.nulls_first,
                                     key_validity=kvalid)
        cols = {o.key: skeys}
        validity = {}
        if kvalid is not None:
            validity[o.key] = gpuq.gather_bits(kvalid, perm)
        for name, t in batch.columns().items():
            if name != o.key:
                cols[name] = gpuq.gather(t, perm)
                v = batch.validity(name)
                if v is not None:
                    validity[name] = gpuq.gather_bits(v, perm)
        batch.close()
        return ColumnarBatch(cols, validity=validity or None)

    def execute_columnar(self):
        import torch.distributed as dist
        from . import gpuq  # noqa: F401 (engine presence check)
        orders = self.sort_orders
        # SortExec sorts the WHOLE partition (one sorter per task): a
        # multi-batch child (e.g. Parquet row groups) concatenates first
        batches = list(self.children[0].execute_columnar())
        assert batches, "sort over a batchless child"
        for batch in [concat_batches(batches)]:
            if (self.global_sort and dist.is_initialized()
                    and dist.get_world_size() > 1):
                # global ORDER BY: range exchange on the PRIMARY key first
                # (the EnsureRequirements RangePartitioning insertion,
                # exchange/EnsureRequirements.scala:296); after the local
                # sort below, rank-major order is global order. Multi-key
                # global sort would need tie-aware range bounds — the
                # single-key contract is asserted.
                assert len(orders) == 1, \
                    "multi-key global ORDER BY across ranks unsupported"
                from .exchange import range_exchange
                o = orders[0]
                payload = {n_: t for n_, t in batch.columns().items()
                           if n_ != o.key}
                validity = {n_: batch.validity(n_) for n_ in batch.columns()
                            if batch.validity(n_) is not None}
                k, payload, validity = range_exchange(
                    batch.column(o.key), payload, desc=o.descending,
                    nulls_first=o.nulls_first,
                    key_validity=batch.validity(o.key),
                    validity={n_: v for n_, v in validity.items()
                              if n_ != o.key},
                    key_name=o.key)
                cols = {o.key: k}
                cols.update(payload)
                batch.close()
                batch = ColumnarBatch(cols, validity=validity or None)
            for o in reversed(orders):
                batch = self._sort_pass(batch, o)
            yield batch


class GpuHashAggregateExec(SparkPlan):
    """Replaces HashAggregateExec (HashAggregateExec.scala:99-151).
    mode: "partial" | "final" | "complete" (AggUtils.scala:126-195 split).

    Partial mode outputs the aggregation BUFFER columns (the
    aggBufferAttributes analog, e.g. Average.scala's (sum, count) pair;
    Sum/Min/Max carry a count companion for NULL-ness), which travel
    through the exchange; final mode merges them exactly: partial sums
    merge with the same-dtype SUM op (int64 counts merge as wrapping i64
    adds, never as f64 — Sum.scala mergeExpressions / Count.scala), and
    partial MIN/MAX merge with MIN/MAX over non-NULL partials. A result is
    NULL iff the merged count of non-null inputs is 0.

    group_key: one name or a tuple (composite GROUP BY via
    gpuq_hash_agg_keys; the reference's UnsafeRow key-tuple path,
    UnsafeFixedWidthAggregationMap.java:39)."""

    def __init__(self, group_key, aggs: List[Tuple[str, str]], mode: str,
                 child, capacity: Optional[int] = None):
        super().__init__(child)
        assert mode in ("partial", "final", "complete")
        self.group_key, self.aggs, self.mode = group_key, aggs, mode
        self.capacity = capacity
        fns = {fn for fn, _ in self.aggs}
        assert fns <= {"sum", "count", "count*", "avg", "min", "max"}, \
            f"unsupported aggs {fns}"

    @property
    def group_keys(self):
        if self.group_key is None:
            return ()
        return (self.group_key,) if isinstance(self.group_key, str) else tuple(self.group_key)

    @staticmethod
    def agg_out_name(fn: str, col) -> str:
        return "count(1)" if fn == "count*" else f"{fn}({col})"

    def _buffer_cols(self, fn: str, col: str) -> List[str]:
        """agg buffer column names (partial-mode output schema)."""
        if fn == "sum":
            return [f"sum({col})", f"count({col})"]
        if fn == "count":
            return [f"count({col})"]
        if fn == "count*":
            return ["count(1)"]
        if fn == "avg":
            return [f"avg_sum({col})", f"count({col})"]
        return [f"{fn}({col})", f"count({col})"]  # min/max

    @property
    def output(self):
        keys = list(self.group_keys)
        if self.mode == "partial":
            seen, bufs = set(), []
            for fn, col in self.aggs:
                for b in self._buffer_cols(fn, col):
                    if b not in seen:
                        seen.add(b); bufs.append(b)
            return keys + bufs
        return keys + [self.agg_out_name(fn, col) for fn, col in self.aggs]

    @property
    def supports_columnar(self):
        return True

    def required_child_distribution(self):
        if self.mode == "final":
            return [Distribution("clustered", self.group_keys)]
        return [Distribution("unspecified")]

    def execute_columnar(self):
        # HashAggregateExec aggregates the WHOLE partition (one
        # TungstenAggregationIterator per task), not one table per input
        # batch: all child batches accumulate into a single hash table
        # (the kernel's first_batch/finalize contract) and ONE result
        # batch emits per partition.
        batches = list(self.children[0].execute_columnar())
        assert batches, "aggregate over a batchless child"
        yield self._agg_batches(batches)

    # ---- spec construction ----

    def _input_specs(self, batch):
        """specs over RAW input rows (partial/complete) + per-buffer slots.
        Returns (specs, buffer_slot: name -> spec index)."""
        from . import gpuq
        specs, slot = [], {}

        def add(name, spec):
            if name not in slot:
                slot[name] = len(specs)
                specs.append(spec)

        for fn, col in self.aggs:
            if fn == "count*":
                add("count(1)", ("count*",))
                continue
            t = batch.column(col)
            v = batch.validity(col)
            if fn == "avg":
                ts = gpuq.cast_i64_f64(t) if t.dtype == torch.int64 else t
                add(f"avg_sum({col})", ("sum", ts, v))
                add(f"count({col})", ("count", t, v))
            elif fn == "count":
                add(f"count({col})", ("count", t, v))
            else:
                add(f"{fn}({col})", (fn, t, v))
                # count companion tracks NULL-ness (result NULL iff no
                # non-null input). In complete mode over a non-null column
                # every group is non-empty, so the companion is redundant.
                if v is not None or self.mode != "complete":
                    add(f"count({col})", ("count", t, v))
        if not specs:
            # pure GROUP BY (no aggregate expressions — the SELECT DISTINCT
            # shape): the kernel wants >= 1 spec; count rows and discard
            add("__rows__", ("count*",))
        return specs, slot

    def _merge_specs(self, batch):
        """specs over PARTIAL BUFFER columns (final mode): same-dtype SUM
        for sums/counts, MIN/MAX over non-NULL partials."""
        specs, slot = [], {}

        def add(name, kind):
            if name in slot:
                return
            t = batch.column(name)
            v = batch.validity(name)
            slot[name] = len(specs)
            specs.append((kind, t, v))

        for fn, col in self.aggs:
            for b in self._buffer_cols(fn, col):
                if b.startswith("count"):
                    add(b, "sum")     # SUM_I64: exact merged counts
                elif b.startswith("min"):
                    add(b, "min")
                elif b.startswith("max"):
                    add(b, "max")
                else:
                    add(b, "sum")
        if not specs:
            slot["__rows__"] = len(specs)
            specs.append(("count*",))
        return specs, slot

    def _agg_batches(self, batches):
        from . import gpuq
        keys = self.group_keys
        total = sum(b.num_rows() for b in batches)
        cap = self.capacity or (1 << max(10, int(total).bit_length()))
        if self.mode == "final":
            per_specs = [self._merge_specs(b) for b in batches]
        else:
            per_specs = [self._input_specs(b) for b in batches]
        slot = per_specs[0][1]
        mg = min(total, cap) + 2
        nspecs = len(per_specs[0][0])
        last_i = len(batches) - 1
        if len(keys) == 0:
            # global aggregate (empty grouping): one output row even on
            # empty input — COUNT 0, SUM/MIN/MAX/AVG NULL (AggUtils
            # emptyInputAggBuffer)
            if total == 0:
                cols, validity = {}, {}
                dev = "cuda"
                b0 = batches[0]

                def src_dtype(name):
                    base = name.split("(", 1)[1][:-1]
                    try:
                        return b0.column(base).dtype
                    except KeyError:
                        return torch.float64
                names = (list(slot) if self.mode == "partial"
                         else [self.agg_out_name(fn, col)
                               for fn, col in self.aggs])
                for name in names:
                    if name.startswith("count"):
                        cols[name] = torch.zeros(1, dtype=torch.int64, device=dev)
                    elif name.startswith(("min(", "max(", "sum(")):
                        cols[name] = torch.zeros(1, dtype=src_dtype(name),
                                                 device=dev)
                        validity[name] = torch.zeros(1, dtype=torch.uint8,
                                                     device=dev)
                    else:  # avg / avg_sum: f64
                        cols[name] = torch.zeros(1, dtype=torch.float64,
                                                 device=dev)
                        validity[name] = torch.zeros(1, dtype=torch.uint8,
                                                     device=dev)
                for b in batches:
                    b.close()
                return ColumnarBatch(cols, validity=validity or None)
            ws = gpuq.agg_multi_workspace(cap, nspecs)
            res = None
            for i, (b, (specs, _)) in enumerate(zip(batches, per_specs)):
                key0 = gpuq.range_i64(b.num_rows(), 0, 0)  # one group
                res = gpuq.hash_agg_multi(key0, specs, cap, workspace=ws,
                                          first_batch=(i == 0),
                                          finalize=(i == last_i),
                                          max_groups=mg)
            ok, okv, accs = res
            key_cols, key_valid = {}, {}
        elif len(keys) == 1:
            k = keys[0]
            ws = gpuq.agg_multi_workspace(cap, nspecs)
            res = None
            any_kv = any(b.validity(k) is not None for b in batches)
            for i, (b, (specs, _)) in enumerate(zip(batches, per_specs)):
                res = gpuq.hash_agg_multi(
                    b.column(k), specs, cap, workspace=ws,
                    key_validity=b.validity(k),
                    first_batch=(i == 0), finalize=(i == last_i),
                    max_groups=mg)
            ok, okv, accs = res
            key_cols = {k: ok}
            # NULL-key group: okv==0 row. Only materialize a bitmap when
            # some input batch's key was nullable.
            key_valid = {}
            if any_kv:
                key_valid[k] = gpuq.u8_to_bits(okv)
        else:
            kcs = [[b.column(k) for k in keys] for b in batches]
            kvs = [[b.validity(k) for k in keys] for b in batches]
            packed = None
            if (len(keys) == 2
                    and all(v[0] is None and v[1] is None for v in kvs)
                    and kcs[0][0].dtype == torch.int64
                    and kcs[0][1].dtype == torch.int64):
                # narrow-tuple pack rule: when the GLOBAL key ranges fit,
                # pack (k1,k2) into one int64 so the single-key path
                # (incl. its per-block LDS tables at low cardinality —
                # the TPC-H Q1 (returnflag, linestatus) shape) runs
                # instead of the verify-slot composite table
                mm = [(gpuq.minmax_i64(kc[0]), gpuq.minmax_i64(kc[1]))
                      for kc in kcs]
                if all(a[2] and b_[2] for a, b_ in mm):
                    mn0 = min(a[0] for a, _ in mm)
                    mx0 = max(a[1] for a, _ in mm)
                    mn1 = min(b_[0] for _, b_ in mm)
                    mx1 = max(b_[1] for _, b_ in mm)
                    shift = max(1, int(mx1 - mn1).bit_length())
                    if (mx0 - mn0) < (1 << (63 - shift)):
                        packed = [gpuq.pack2_i64(kc[0], kc[1], mn0, mn1,
                                                 shift) for kc in kcs]
            if packed is not None:
                ws = gpuq.agg_multi_workspace(cap, nspecs)
                res = None
                for i, (pk, (specs, _)) in enumerate(zip(packed, per_specs)):
                    res = gpuq.hash_agg_multi(pk, specs, cap, workspace=ws,
                                              first_batch=(i == 0),
                                              finalize=(i == last_i),
                                              max_groups=mg)
                ok, okv, accs = res
                k0, k1 = gpuq.unpack2_i64(ok, mn0, mn1, shift)
                key_cols = {keys[0]: k0, keys[1]: k1}
                key_valid = {}
            else:
                ws = torch.empty(
                    gpuq.lib().gpuq_hash_agg_keys_workspace_bytes(
                        cap, len(keys), nspecs),
                    dtype=torch.uint8, device="cuda")
                res = None
                for i, (kc, kv, (specs, _)) in enumerate(
                        zip(kcs, kvs, per_specs)):
                    res = gpuq.hash_agg_keys(
                        kc, specs, cap, key_validities=kv, workspace=ws,
                        first_batch=(i == 0), finalize=(i == last_i),
                        max_groups=mg)
                okeys, kmask, accs = res
                key_cols = dict(zip(keys, okeys))
                key_valid = {}
                for c, k in enumerate(keys):
                    if any(v[c] is not None for v in kvs):
                        key_valid[k] = gpuq.maskbit_to_bits(kmask, c)
        cols, validity = dict(key_cols), dict(key_valid)
        if self.mode == "partial":
            for name, j in slot.items():
                if name == "__rows__":
                    continue
                cols[name] = accs[j]
                # partial sum/min/max of an all-NULL group is NULL (its
                # count companion is 0); counts themselves are never NULL
                if not name.startswith("count"):
                    base = name.split("(", 1)[1][:-1]
                    validity[name] = gpuq.nonzero_to_bits(
                        accs[slot[f"count({base})"]])
        else:
            for fn, col in self.aggs:
                out = self.agg_out_name(fn, col)
                if fn == "count*":
                    cols[out] = accs[slot["count(1)"]]
                    continue
                if fn == "count":
                    cols[out] = accs[slot[f"count({col})"]]
                    continue
                cnt_slot = slot.get(f"count({col})")
                if fn == "avg":
                    # Average.evaluateExpression: sum / cast(count)
                    cols[out] = gpuq.project_binop(
                        accs[slot[f"avg_sum({col})"]], "/",
                        b=gpuq.cast_i64_f64(accs[cnt_slot]))
                else:
                    cols[out] = accs[slot[f"{fn}({col})"]]
                # SQL NULL iff no non-null input (Sum/Min/Max/Average.scala);
                # no companion => non-null complete-mode input, always valid
                if cnt_slot is not None:
                    validity[out] = gpuq.nonzero_to_bits(accs[cnt_slot])
        for b in batches:
            b.close()
        return ColumnarBatch(cols, validity=validity or None)


class GpuShuffleExchangeExec(SparkPlan):
    """Replaces ShuffleExchangeExec (ShuffleExchangeExec.scala:277-470):
    on-device radix partition + RCCL all-to-all (spark_amd/exchange.py).
    Runs only under torch.distributed; num_partitions == world size."""

    def __init__(self, keys: Tuple[str, ...], child):
        super().__init__(child)
        assert 1 <= len(keys) <= 4, "1..4 int64 partition key columns"
        self.keys = tuple(keys)

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

    def get_shuffle_partitions(self, specs):
        """AQE getShuffleRDD(partitionSpecs) analog
        (ShuffleExchangeExec.scala:141, ShuffledRowRDD.scala:33
        CoalescedPartitionSpec): serve coalesced partition ranges straight
        from this rank's partition-contiguous map output, no re-partition.
        specs: list of (start_partition, end_partition) half-open ranges.
        Returns one ColumnarBatch per spec (this rank's contribution)."""
        assert hasattr(self, "_map_outputs"), "execute_columnar first"
        from . import gpuq
        out = []
        for start, end in specs:
            parts = []
            for cols, validity, offsets in self._map_outputs:
                lo, hi = offsets[start], offsets[end]
                sl = {n_: t[lo:hi] for n_, t in cols.items()}
                va = {}
                for n_, u8 in validity.items():
                    if hi > lo:
                        va[n_] = gpuq.u8_to_bits(u8[lo:hi])
                parts.append(ColumnarBatch(sl, validity=va or None))
            out.append(concat_batches(parts) if len(parts) > 1 else parts[0])
        return out

    def execute_columnar(self):
        import torch.distributed as dist
        from . import gpuq
        from .exchange import exchange_columns
        assert dist.is_initialized(), "GpuShuffleExchangeExec needs torch.distributed"
        world = self.num_partitions
        for batch in self.children[0].execute_columnar():
            kcols = [batch.column(k) for k in self.keys]
            kvals = [batch.validity(k) for k in self.keys]
            if len(self.keys) == 1:
                perm, counts = gpuq.partition_perm(kcols[0], world,
                                                   key_validity=kvals[0])
            else:
                perm, counts = gpuq.partition_perm_multi(kcols, world,
                                                         key_validities=kvals)
            cols, vcols = {}, {}
            row_bytes = 0
            for name, t in batch.columns().items():
                cols[name] = gpuq.gather(t, perm)
                row_bytes += t.element_size()
                v = batch.validity(name)
                if v is not None:
                    # bitmaps travel as u8 columns: partition split points
                    # are not byte-aligned (repacked on receive)
                    pv = gpuq.gather_bits(v, perm)
                    vcols[name] = gpuq.bits_to_u8(pv, t.numel())
                    row_bytes += 1
            in_splits = counts.cpu().tolist()
            self._stats = {"bytes_by_partition": [c * row_bytes for c in in_splits],
                           "rows_written": int(sum(in_splits))}
            offsets = [0]
            for c in in_splits:
                offsets.append(offsets[-1] + c)
            if not hasattr(self, "_map_outputs"):
                self._map_outputs = []
            self._map_outputs.append((cols, vcols, offsets))
            wire = dict(cols)
            wire.update({f"__valid__{n}": u8 for n, u8 in vcols.items()})
            out, _ = exchange_columns(wire, in_splits)
            batch.close()
            validity = {}
            for name in list(out):
                if name.startswith("__valid__"):
                    u8 = out.pop(name)
                    validity[name[len("__valid__"):]] = gpuq.u8_to_bits(u8)
            yield ColumnarBatch(out, validity=validity or None)


class GpuShuffledHashJoinExec(SparkPlan):
    """Replaces ShuffledHashJoinExec inner join
    (ShuffledHashJoinExec.scala:103-132). Both children must already be
    clustered on the join key (ShuffledJoin.scala:57-69) — i.e. fed by (our)
    exchanges, as EnsureRequirements would arrange."""

    def __init__(self, left_key: str, right_key: str, build_side: str,
                 left, right, join_type: str = "inner",
                 null_aware_anti: bool = False):
        super().__init__(left, right)
        assert build_side in ("left", "right")
        assert join_type in ("inner", "left_outer", "right_outer",
                             "left_semi", "left_anti", "full_outer")
        # NOT IN rewrite (BroadcastHashJoinExec.scala:48,137
        # isNullAwareAntiJoin): empty build -> all probe rows; any NULL
        # build key -> empty result; NULL probe keys are filtered
        assert not null_aware_anti or join_type == "left_anti"
        self.null_aware_anti = null_aware_anti
        # the preserved/streamed side must be the PROBE side
        # (HashJoin.scala buildSide constraints; full outer preserves both
        # — either side may build)
        if join_type in ("left_outer", "left_semi", "left_anti"):
            assert build_side == "right", f"{join_type} builds on the right"
        if join_type == "right_outer":
            assert build_side == "left", "right_outer builds on the left"
        self.left_key, self.right_key, self.build_side = left_key, right_key, build_side
        self.join_type = join_type

    @property
    def output(self):
        if self.join_type in ("left_semi", "left_anti"):
            return self.children[0].output
        return self.children[0].output + self.children[1].output

    @property
    def supports_columnar(self):
        return True

    def required_child_distribution(self):
        return [Distribution("clustered", (self.left_key,)),
                Distribution("clustered", (self.right_key,))]

    def execute_columnar(self):
        from . import gpuq
        bi = 0 if self.build_side == "left" else 1
        build_b = list(self.children[bi].execute_columnar())
        assert build_b, "join build over a batchless child"
        # buildHashedRelation materializes the whole build side per
        # partition: multi-batch children concatenate
        build = concat_batches(build_b)
        bkey = self.left_key if self.build_side == "left" else self.right_key
        pkey = self.right_key if self.build_side == "left" else self.left_key
        jt = {"inner": gpuq.JOIN_INNER, "left_outer": gpuq.JOIN_OUTER,
              "right_outer": gpuq.JOIN_OUTER, "left_semi": gpuq.JOIN_SEMI,
              "left_anti": gpuq.JOIN_ANTI,
              "full_outer": gpuq.JOIN_FULL}[self.join_type]
        bk = build.column(bkey)
        bn = bk.numel()
        if self.null_aware_anti:
            if bn == 0:
                # NOT IN (empty) is true for every row, NULL included
                yield from self.children[1 - bi].execute_columnar()
                build.close()
                return
            _, _, valid_cnt = gpuq.minmax_i64(bk,
                                              validity=build.validity(bkey))
            if valid_cnt < bn:
                # any NULL in the build side: NOT IN is never true
                for probe in self.children[1 - bi].execute_columnar():
                    empty = {n_: t[:0] for n_, t in probe.columns().items()}
                    probe.close()
                    yield ColumnarBatch(empty)
                build.close()
                return
            jt = gpuq.JOIN_ANTI_NULLAWARE
        cap = 1 << max(4, int(bn * 2 - 1).bit_length() if bn else 4)
        ws = gpuq.join_build(bk, cap, key_validity=build.validity(bkey))
        semi = self.join_type in ("left_semi", "left_anti")
        outer = self.join_type in ("left_outer", "right_outer", "full_outer")
        full = self.join_type == "full_outer"
        # the probe (streamed) side is consumed batch-at-a-time
        # (ShuffledHashJoinExec.doExecute streams streamedIter)
        for probe in self.children[1 - bi].execute_columnar():
            pk = probe.column(pkey)
            out_cap = max(int(probe.num_rows() * 2) + 64, 64)
            while True:
                op, ob, nm = gpuq.join_probe(pk, ws, cap, bn, out_cap,
                                             key_validity=probe.validity(pkey),
                                             join_type=jt)
                if op is not None:
                    break
                out_cap = nm + 64
            cols, validity = {}, {}
            if not semi:
                for name, t in build.columns().items():
                    if outer:
                        # unmatched probe rows pair with NIL -> NULL build
                        cols[name], validity[name] = gpuq.gather_nullable(
                            t, ob, validity=build.validity(name))
                    else:
                        cols[name] = gpuq.gather(t, ob)
                        v = build.validity(name)
                        if v is not None:
                            validity[name] = gpuq.gather_bits(v, ob)
            pcols = probe.columns()
            for name, t in pcols.items():
                v = probe.validity(name)
                if name in cols:
                    name = f"{name}#probe"
                if full:
                    # build-side unmatched rows carry NIL probe rids
                    cols[name], validity[name] = gpuq.gather_nullable(
                        t, op, validity=v)
                else:
                    cols[name] = gpuq.gather(t, op)
                    if v is not None:
                        validity[name] = gpuq.gather_bits(v, op)
            probe.close()
            yield ColumnarBatch(cols, validity=validity or None)
        build.close()


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
    def output_ordering(self):
        # stable compaction preserves the child's order (FilterExec does)
        return self.children[0].output_ordering

    @property
    def supports_columnar(self):
        return True

    def execute_columnar(self):
        from . import gpuq
        for batch in self.children[0].execute_columnar():
            perm, cnt = gpuq.filter_cmp(batch.column(self.col), self.op,
                                        self.literal,
                                        validity=batch.validity(self.col))
            cols, validity = {}, {}
            for name, t in batch.columns().items():
                cols[name] = gpuq.gather(t, perm)
                v = batch.validity(name)
                if v is not None and cnt:
                    validity[name] = gpuq.gather_bits(v, perm)
            batch.close()
            yield ColumnarBatch(cols, validity=validity or None)


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
        from . import gpuq
        from .exchange import broadcast_gather, _unpack_validity_wire
        for batch in self.children[0].execute_columnar():
            if dist.is_initialized() and dist.get_world_size() > 1:
                wire = batch.columns()
                n = batch.num_rows()
                for name in list(wire):
                    v = batch.validity(name)
                    if v is not None:
                        wire[f"__valid__{name}"] = gpuq.bits_to_u8(v, n)
                cols = broadcast_gather(wire)
                validity = _unpack_validity_wire(gpuq, cols)
                batch.close()
                yield ColumnarBatch(cols, validity=validity or None)
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

    def __init__(self, preserve_smj_ordering: bool = True):
        self.preserve_smj_ordering = preserve_smj_ordering

    def pre_columnar_transitions(self, plan: SparkPlan) -> SparkPlan:
        children = [self.pre_columnar_transitions(c) for c in plan.children]
        if isinstance(plan, SortExec):
            return GpuSortExec(plan.sort_order, plan.global_sort, *children)
        if isinstance(plan, HashAggregateExec):
            return GpuHashAggregateExec(plan.group_key, plan.aggs, plan.mode,
                                        *children, capacity=plan.capacity)
        if isinstance(plan, ShuffledHashJoinExec):
            return GpuShuffledHashJoinExec(plan.left_key, plan.right_key,
                                           plan.build_side, *children,
                                           join_type=plan.join_type)
        if isinstance(plan, SortMergeJoinExec):
            # SMJ -> GPU hash join (same slot, same required distribution;
            # build on the right side as SHJ's default would choose). SMJ
            # advertises outputOrdering on the streamed keys — parents may
            # have had sorts elided against it — so re-sort the hash join's
            # output to honor the contract (one radix pass; the GPU sort
            # node's output_ordering then matches what SMJ declared).
            build = "left" if plan.join_type == "right_outer" else "right"
            j = GpuShuffledHashJoinExec(plan.left_key, plan.right_key,
                                        build, *children,
                                        join_type=plan.join_type)
            declared = plan.output_ordering
            if self.preserve_smj_ordering and declared:
                return GpuSortExec(declared, False, j)
            return j
        if isinstance(plan, ShuffleExchangeExec):
            return GpuShuffleExchangeExec(plan.keys, *children)
        if isinstance(plan, FilterExec):
            return GpuFilterExec(plan.col, plan.op, plan.literal, *children)
        if isinstance(plan, ProjectExec):
            return GpuProjectExec(plan.projections, *children)
        if isinstance(plan, RangeExec):
            return GpuRangeExec(plan.n, plan.start, plan.step, plan.name)
        if isinstance(plan, ParquetScanExec):
            return GpuParquetScanExec(plan.path, plan.columns)
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
