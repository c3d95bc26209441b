from .dist_ppo import DistPPOProblem  # noqa: F401
from .ppo_optimizers import DiNNOPPO, DSGDPPO, DSGTPPO  # noqa: F401
