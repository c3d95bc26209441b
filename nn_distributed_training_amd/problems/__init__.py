from .dist_mnist_problem import DistMNISTProblem  # noqa: F401
from .dist_dense_problem import DistDensityProblem  # noqa: F401
from .dist_online_dense_problem import DistOnlineDensityProblem  # noqa: F401
