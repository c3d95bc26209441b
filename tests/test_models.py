import torch

from nn_distributed_training_amd.models import (
    FFReLUNet,
    FFSigmoidNet,
    FFTanhNet,
    FourierNet,
    MNISTConvNet,
    model_spec,
)


def test_mnist_convnet_param_count():
    # paper config F=3, k=5, W=64 -> n = 28,440
    m = MNISTConvNet(3, 5, 64)
    n = sum(p.numel() for p in m.parameters())
    assert n == 28440
    x = torch.randn(4, 1, 28, 28)
    y = m(x)
    assert y.shape == (4, 10)
    # log-softmax output: rows sum to 1 in prob space
    assert torch.allclose(y.exp().sum(dim=1), torch.ones(4))


def test_fourier_net_param_count():
    # paper config [2,256,64,64,64,1] -> n = 25,601
    m = FourierNet([2, 256, 64, 64, 64, 1], scale=0.05)
    n = sum(p.numel() for p in m.parameters())
    assert n == 25601
    y = m(torch.randn(16, 2))
    assert y.shape == (16, 1)
    assert (y >= 0).all() and (y <= 1).all()  # sigmoid head


def test_fourier_relu_after_siren():
    """Reference applies ReLU after the SIREN sin (fourier_nn.py:44-58)."""
    m = FourierNet([2, 8, 1], scale=1.0)
    x = torch.randn(32, 2)
    enc = torch.sin(m.scale * m.encode.linear(x))
    h = torch.relu(enc)
    expect = torch.sigmoid(m.hidden[0](h))
    assert torch.allclose(m(x), expect)


def test_mlp_families():
    for cls, rng in [
        (FFReLUNet, (-10, 10)),
        (FFTanhNet, (-1, 1)),
        (FFSigmoidNet, (0, 1)),
    ]:
        m = cls([3, 16, 16, 2])
        y = m(torch.randn(8, 3))
        assert y.shape == (8, 2)
        if cls is not FFReLUNet:  # bounded output activations
            assert (y >= rng[0]).all() and (y <= rng[1]).all()


def test_model_spec_layout():
    m = MNISTConvNet(3, 5, 64)
    spec = model_spec(m)
    assert spec.n == 28440
    kinds = [l.kind for l in spec.layers]
    assert kinds == ["conv_pool", "linear", "linear"]
    acts = [l.activation for l in spec.layers]
    assert acts == ["relu", "relu", "logsoftmax"]
    # offsets follow parameters_to_vector order
    assert spec.layers[0].w_off == 0
    assert spec.layers[0].b_off == 75
    assert spec.layers[1].w_off == 78
    assert spec.layers[2].w_off == 78 + 27648 + 64
    f = FourierNet([2, 256, 64, 64, 64, 1], scale=0.05)
    fs = model_spec(f)
    assert fs.layers[0].activation == "sin_relu"
    assert fs.layers[-1].activation == "sigmoid"
    assert fs.n == 25601
