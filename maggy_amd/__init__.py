"""maggy_amd — an MI355X-native distribution-transparent experiment engine.

Brand-new framework with the capabilities of logicalclocks/maggy
(/root/reference, v1.1.2): ``experiment.lagom(train_fn, config)`` runs the
same oblivious training function as a single run, an asynchronous parallel
hyperparameter search (RandomSearch / GridSearch / ASHA / GP / TPE, with
Hyperband pruning and median early stop), an ablation study (LOCO), or
data-parallel training — scheduled over a single-node pool of MI355X GPUs
(one worker process per GPU) with a lock-free shared-memory reporter
channel, RCCL-over-xGMI collectives, and hand-written CDNA4 HIP kernels for
the fused optimizer and reductions.  No Spark, no Py4J, no TF, no CUDA
shims.
"""
from maggy_amd.searchspace import Searchspace  # noqa: F401
from maggy_amd.trial import Trial  # noqa: F401
from maggy_amd import experiment  # noqa: F401  (lagom entry point)

__version__ = "0.1.0"
__all__ = ["Searchspace", "Trial", "experiment"]
