"""Optimizer on/off A/B: prove the whole-pipeline
optimizer earns its keep on a pipeline with shared work.

The fit-data splice (``featurize.and_then(est, data)``) duplicates the
featurization chain between the fit path and the transform path; the
optimizer's EquivalentNodeMergeRule (CSE) merges them so the executor's
memoization computes the features ONCE, and the profiled AutoCacheRule
pins the shared intermediate ordered by measured saved-seconds/byte.
With the optimizer disabled the duplicated chain runs twice.

Run:  python scripts/ab_optimizer.py            (CPU or GPU)
"""
import sys
import time

sys.path.insert(0, ".")

import torch

import keystone_amd as ka
from keystone_amd.ops.learning import BlockLeastSquaresEstimator
from keystone_amd.ops.stats import CosineRandomFeatures, StandardScaler
from keystone_amd.ops.util import (ClassLabelIndicatorsFromIntLabels,
                                   MaxClassifier)
from keystone_amd.workflow.optimizer import (Optimizer, default_optimizer,
                                             set_default_optimizer)
from keystone_amd.loaders.synthetic import synthetic_dense
from keystone_amd.parallel.runtime import init_runtime


def build_and_run(n, d, feats, classes):
    X, y = synthetic_dense(n, d, classes, seed=3)
    Y = ka.from_rows(
        ClassLabelIndicatorsFromIntLabels(classes).apply_batch(y.data).float())
    featurize = StandardScaler().with_data(X).and_then(
        CosineRandomFeatures(d, feats, gamma=1.0 / d, seed=1))
    pipe = featurize.and_then(
        BlockLeastSquaresEstimator(block_size=1024, num_iter=3, lam=1e-3),
        X, Y).and_then(MaxClassifier())
    t0 = time.perf_counter()
    pred = pipe.apply(X).get()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    acc = (pred.data.squeeze(-1).cpu() == y.data.cpu()).float().mean().item()
    return dt, acc


def main():
    init_runtime()
    n, d, feats, classes = (60000, 784, 4096, 10) \
        if torch.cuda.is_available() else (20000, 256, 2048, 10)
    # warm caches/JIT
    build_and_run(1000, d, 256, classes)

    set_default_optimizer(Optimizer([]))          # optimizer OFF
    t_off, acc_off = build_and_run(n, d, feats, classes)
    set_default_optimizer(None) if False else None
    import keystone_amd.workflow.optimizer as om
    om._DEFAULT = None                            # optimizer ON (default)
    t_on, acc_on = build_and_run(n, d, feats, classes)
    assert abs(acc_on - acc_off) < 1e-6, (acc_on, acc_off)
    print(f"optimizer OFF: {t_off:.3f}s   ON: {t_on:.3f}s   "
          f"speedup {t_off / t_on:.2f}x   (accuracy {acc_on:.3f}, "
          f"device={'cuda' if torch.cuda.is_available() else 'cpu'})")


if __name__ == "__main__":
    main()
