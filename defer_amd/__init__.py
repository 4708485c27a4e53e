"""defer_amd — MI355X-native pipeline-partitioned DNN inference engine.

A from-scratch rebuild of the capabilities of ANRGUSC/DEFER (reference:
/root/reference/src/dispatcher.py:20-115, node.py:110-127) designed for a
single 8x MI355X node: the Keras-DAG partitioner becomes an FX-graph
partitioner (`defer_amd.parallel.partitioner`), compute nodes become GPUs
driven by per-rank stage workers over RCCL/xGMI
(`defer_amd.parallel.pipeline`), and the TensorFlow kernel substrate becomes
hand-written CDNA4 HIP kernels (`defer_amd/csrc`, bound via
`defer_amd.ops`). The inter-stage activation relay — ZFP + LZ4 in the
reference (dispatcher.py:81-84) — is a GPU codec (`defer_amd.ops.codec`).

Public API (shape-parity with the reference's `DEFER` class,
dispatcher.py:21,107):

    from defer_amd import DEFER
    engine = DEFER(compute_nodes)            # list of GPU ids / ranks
    engine.run_defer(model, partition_layers, input_stream, output_stream)
"""

__version__ = "0.1.0"

from defer_amd.config import PipelineConfig  # noqa: F401
from defer_amd.parallel.pipeline import DEFER, DistPipeline  # noqa: F401
from defer_amd.parallel.partitioner import partition_model, auto_partition  # noqa: F401
