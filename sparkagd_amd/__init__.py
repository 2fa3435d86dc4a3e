"""sparkagd_amd — an MI355X-native distributed accelerated-gradient-descent framework.

A from-scratch re-design of the capabilities of staple/spark-agd (an Apache Spark
MLlib ``Optimizer`` implementing the TFOCS Auslender–Teboulle accelerated proximal
gradient method; reference: ``AcceleratedGradientDescent.scala:151-339``) for a
single node of AMD Instinct MI355X GPUs:

* Spark RDD partitions        -> GPU-resident data shards (``sparkagd_amd.data``)
* treeAggregate tree-reduce   -> RCCL all-reduce over xGMI (``sparkagd_amd.parallel``)
* Breeze/netlib BLAS + the per-example ``Gradient.compute`` JVM loop
                              -> hand-written CDNA4 HIP kernels (``sparkagd_amd/csrc``)
* MLlib Gradient/Updater      -> batched plug-ins (``sparkagd_amd.models``)
* driver loop                 -> ``sparkagd_amd.optimizer`` (same math, Python host loop,
                                 all vector state device-resident)

This is not a port: there is no JVM, no Spark shim, no CUDA compatibility layer.
"""

from .config import AGDConfig
from .models.gradient import (
    Gradient,
    LogisticGradient,
    LeastSquaresGradient,
    HingeGradient,
    SmoothedHingeGradient,
    MultinomialLogisticGradient,
)
from .models.updater import (
    Updater,
    SimpleUpdater,
    L1Updater,
    SquaredL2Updater,
    ElasticNetUpdater,
)
from .data import (DenseShard, CSRShard, MixedShard, generate_logistic_data,
                   generate_dense_problem, generate_multiclass_problem)
from .models.trainers import (LogisticRegressionWithAGD, LinearRegressionWithAGD,
                              SVMWithAGD, LogisticRegressionWithSGD,
                              LinearRegressionWithSGD, SVMWithSGD,
                              SoftmaxRegressionWithAGD, LinearModel,
                              MultinomialModel, regularization_path)
from .optimizer import (AcceleratedGradientDescent, GradientDescent, run,
                        run_mini_batch, runMiniBatch)
from .gram import GramOperator, run_gram
from .parallel.comm import Communicator
from . import evaluation
from .streaming import HostStreamedDenseShard

__version__ = "0.2.0"

__all__ = [
    "AGDConfig",
    "GradientDescent",
    "Gradient",
    "LogisticGradient",
    "LeastSquaresGradient",
    "HingeGradient",
    "SmoothedHingeGradient",
    "MultinomialLogisticGradient",
    "Updater",
    "SimpleUpdater",
    "L1Updater",
    "SquaredL2Updater",
    "ElasticNetUpdater",
    "LogisticRegressionWithAGD",
    "LinearRegressionWithAGD",
    "SVMWithAGD",
    "LogisticRegressionWithSGD",
    "LinearRegressionWithSGD",
    "SVMWithSGD",
    "SoftmaxRegressionWithAGD",
    "LinearModel",
    "MultinomialModel",
    "regularization_path",
    "evaluation",
    "HostStreamedDenseShard",
    "DenseShard",
    "CSRShard",
    "MixedShard",
    "generate_logistic_data",
    "generate_dense_problem",
    "generate_multiclass_problem",
    "AcceleratedGradientDescent",
    "run",
    "run_mini_batch",
    "runMiniBatch",
    "Communicator",
    "GramOperator",
    "run_gram",
]
