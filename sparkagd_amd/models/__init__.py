"""Loss/prox plug-in layer and user-facing model trainers.

Mirrors the delegate architecture of the reference (``Gradient`` and
``Updater`` plug-ins constructed by the caller and handed to the optimizer,
``AcceleratedGradientDescent.scala:41,104-120``), because that delegate
surface is what the reference's tests exercise — but the implementations are
batched GPU kernels, not per-example JVM loops.
"""

from .gradient import (Gradient, LogisticGradient, LeastSquaresGradient,
                       HingeGradient, SmoothedHingeGradient,
                       MultinomialLogisticGradient)
from .updater import Updater, SimpleUpdater, L1Updater, SquaredL2Updater, ElasticNetUpdater
from .trainers import (LogisticRegressionWithAGD, LinearRegressionWithAGD, SVMWithAGD,
                       LogisticRegressionWithSGD, LinearRegressionWithSGD, SVMWithSGD,
                       SoftmaxRegressionWithAGD, LinearModel, MultinomialModel,
                       regularization_path)

__all__ = [
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
]
