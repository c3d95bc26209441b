from .dinno import DiNNO  # noqa: F401
from .dsgd import DSGD  # noqa: F401
from .dsgt import DSGT  # noqa: F401


def build_optimizer(problem, device, opt_conf):
    alg = opt_conf["alg_name"]
    if alg == "dinno":
        return DiNNO(problem, device, opt_conf)
    if alg == "dsgd":
        return DSGD(problem, device, opt_conf)
    if alg == "dsgt":
        return DSGT(problem, device, opt_conf)
    raise NameError("Unknown distributed opt algorithm.")
