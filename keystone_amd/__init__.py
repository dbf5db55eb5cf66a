"""keystone_amd — an MI355X-native end-to-end ML pipeline framework.

A from-scratch rebuild of the capabilities of KeystoneML (amplab/keystone):
the Pipeline/Transformer/Estimator composition API, a whole-pipeline
optimizer, a serialized-pipeline checkpoint format, and the full
featurization + solver operator library — running SPMD with one process per
AMD Instinct MI355X GPU, dataset shards resident in 288 GB HBM3E, hot paths
as hand-written CDNA4 HIP kernels (MFMA/LDS) and RCCL collectives over xGMI
replacing Spark's shuffle/treeReduce.  See SURVEY.md for the reference map.
"""

__version__ = "0.1.0"

from .parallel.runtime import Runtime, get_runtime, init_runtime, shutdown_runtime
from .parallel.dataset import Dataset, ItemDataset, TensorDataset, from_items, from_rows
from .workflow.pipeline import (BatchTransformer, Estimator, FunctionTransformer,
                                Identity, LabelEstimator, Pipeline,
                                PipelineDataset, PipelineDatum, Transformer,
                                gather)
from .workflow.cache import Cacher, Checkpointer
from .workflow.optimizer import (AutoCacheRule, ElementwiseFusionRule,
                                 ElementwiseTransformer, EquivalentNodeMergeRule,
                                 Optimizable, Optimizer, Profiler, Rule,
                                 UnusedBranchRemovalRule, default_optimizer)

__all__ = [
    "Runtime", "get_runtime", "init_runtime", "shutdown_runtime",
    "Dataset", "ItemDataset", "TensorDataset", "from_items", "from_rows",
    "Pipeline", "Transformer", "BatchTransformer", "FunctionTransformer",
    "Estimator", "LabelEstimator", "PipelineDataset", "PipelineDatum",
    "Identity", "gather", "Cacher", "Checkpointer",
    "Optimizer", "Rule", "Optimizable", "Profiler", "default_optimizer",
    "EquivalentNodeMergeRule", "UnusedBranchRemovalRule",
    "ElementwiseFusionRule", "ElementwiseTransformer", "AutoCacheRule",
]
