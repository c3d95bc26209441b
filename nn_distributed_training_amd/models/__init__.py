from .mnist_conv import MNISTConvNet  # noqa: F401
from .fourier import FourierNet, SIRENLayer  # noqa: F401
from .mlp import FFReLUNet, FFSigmoidNet, FFTanhNet  # noqa: F401
from .spec import LayerSpec, model_spec, param_layout  # noqa: F401
