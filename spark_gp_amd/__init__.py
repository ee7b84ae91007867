"""spark_gp_amd — an MI355X-native sparse Gaussian-Process engine.

A from-scratch re-design of the capabilities of akopich/spark-gp (Scala /
Apache Spark) for AMD Instinct MI355X: PyTorch-ROCm front end, batched
expert execution, hand-written CDNA4 HIP kernels for the hot ops, and RCCL
over xGMI (via torch.distributed) for the expert-parallel collectives.

Quick start::

    from spark_gp_amd import (GaussianProcessRegression, ARDRBFKernel,
                              EyeKernel, Scalar)
    gp = (GaussianProcessRegression()
          .setKernel(lambda: 1 * ARDRBFKernel(5) + Scalar(1).const * EyeKernel())
          .setDatasetSizeForExpert(100)
          .setActiveSetSize(1000)
          .setSigma2(1e-4))
    model = gp.fit(X, y)
    mean = model.predict(X_test)
"""

from .active_set import (ActiveSetProvider, GreedilyOptimizingActiveSetProvider,
                         KMeansActiveSetProvider, RandomActiveSetProvider)
from .kernels import (ARDRBFKernel, EyeKernel, Kernel, Matern32Kernel,
                      Matern52Kernel, RBFKernel, Scalar, SumOfKernels,
                      WhiteNoiseKernel)
from .likelihoods import (Likelihood, LogisticLikelihood,
                          PoissonLikelihood, ProbitLikelihood)
from .models import (GaussianProcessClassificationModel,
                     GaussianProcessClassifier,
                     GaussianProcessPoissonModel,
                     GaussianProcessPoissonRegression,
                     GaussianProcessRegression,
                     GaussianProcessRegressionModel, load_model, save_model)
from .parallel import Comm, get_comm, init_from_env
from .ppa import NotPositiveDefiniteError
from .utils import (Integrator, OneVsRest, StandardScaler, accuracy,
                    cross_validate, rmse, scale, train_validation_split)

__version__ = "0.2.0"

__all__ = [
    "GaussianProcessRegression", "GaussianProcessRegressionModel",
    "GaussianProcessClassifier", "GaussianProcessClassificationModel",
    "GaussianProcessPoissonRegression", "GaussianProcessPoissonModel",
    "Likelihood", "LogisticLikelihood", "PoissonLikelihood",
    "ProbitLikelihood",
    "Kernel", "RBFKernel", "ARDRBFKernel", "Matern32Kernel",
    "Matern52Kernel", "EyeKernel", "WhiteNoiseKernel",
    "SumOfKernels", "Scalar",
    "ActiveSetProvider", "RandomActiveSetProvider", "KMeansActiveSetProvider",
    "GreedilyOptimizingActiveSetProvider",
    "NotPositiveDefiniteError",
    "save_model", "load_model",
    "Comm", "get_comm", "init_from_env",
    "rmse", "accuracy", "cross_validate", "train_validation_split",
    "OneVsRest", "Integrator", "StandardScaler", "scale",
    "__version__",
]
