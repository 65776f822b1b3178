"""spark_amd — MI355X-native columnar execution engine for the Spark SQL
hot path (SortExec / HashAggregateExec / ShuffledHashJoinExec /
ShuffleExchangeExec), built from scratch for gfx950.

Layers:
  include/gpuq.h + spark_amd/csrc/    the C-ABI engine (hand-written HIP)
  spark_amd/gpuq.py                   ctypes binding (torch = device memory
                                      + streams + RCCL plumbing only)
  spark_amd/exec.py                   host-side mirror of the SparkPlan
                                      columnar operator contract
  spark_amd/exchange.py               RCCL all-to-all exchange
                                      (torch.distributed nccl == RCCL)

The Scala/JNI host layer a Spark deployment would use binds the same C-ABI;
see INTEGRATION.md.
"""
