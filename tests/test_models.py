import pytest
import torch

from sodkit.models import MODEL_NAMES, build_model


@pytest.mark.parametrize("name", ["minet_vgg16", "minet_res50"])
def test_minet_forward_shape(name):
    m = build_model(name)
    x = torch.randn(2, 3, 64, 64)
    y = m(x)
    assert y.shape == (2, 1, 64, 64)


def test_minet_res2net_forward_shape():
    m = build_model("minet_res2net101")
    x = torch.randn(1, 3, 64, 64)
    assert m(x).shape == (1, 1, 64, 64)


@pytest.mark.parametrize("name", ["hdfnet_res50", "hdfnet_vgg16"])
def test_hdfnet_forward_shape(name):
    m = build_model(name)
    rgb = torch.randn(1, 3, 64, 64)
    d = torch.randn(1, 1, 64, 64)
    assert m(rgb, d).shape == (1, 1, 64, 64)


def test_param_counts_sane():
    # MINet-R50 should land in the tens of millions (~40-50M)
    m = build_model("minet_res50")
    n = sum(p.numel() for p in m.parameters())
    assert 20e6 < n < 80e6, n


def test_backward_flows_everywhere():
    m = build_model("minet_vgg16")
    x = torch.randn(1, 3, 32, 32)
    y = m(x)
    y.sum().backward()
    missing = [n for n, p in m.named_parameters()
               if p.requires_grad and p.grad is None]
    assert not missing, missing


def test_registry_names():
    assert set(MODEL_NAMES) == {"minet_res50", "minet_vgg16",
                                "minet_res2net101", "hdfnet_res50",
                                "hdfnet_vgg16"}
    with pytest.raises(KeyError):
        build_model("nope")
