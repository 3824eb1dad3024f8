"""The conv dispatch gates encode MEASURED per-shape policies (see
sodkit/ops/conv.py comments and profiles/r2/ubench_conv_final.log); these
CPU tests pin them so a refactor cannot silently re-route a shape class
that was measured to lose."""

import torch

from sodkit.ops import conv as c


def test_8p_gate_is_launch_size():
    # wins (measured): l1_1x1e 968 wgs, l2_1x1e 484, l3_1x1e 244
    assert c._use_8p(968, 256)
    assert c._use_8p(242, 512)
    assert c._use_8p(61, 1024)
    # losses (measured): l3_3x3 61 wgs, l4_1x1e 128, l4_3x3 32
    assert not c._use_8p(61, 256)
    assert not c._use_8p(16, 2048)
    # env forcing for A/B
    prev = c._8P_ENV
    try:
        c._8P_ENV = "1"
        assert c._use_8p(1, 256)
        c._8P_ENV = "0"
        assert not c._use_8p(10**6, 2048)
    finally:
        c._8P_ENV = prev


def _w(cout, k, cin):
    return torch.empty(cout, k, k, cin)


def _x(w):
    return torch.empty(2, 64, w, w)


def test_wgrad_dir_gate():
    # wins: W in [64, 184], 3x3 s1 d1 p1, C % 64 == 0
    assert c._use_wgrad_dir(_w(64, 3, 64), _x(176), 1, 1, 1)
    assert c._use_wgrad_dir(_w(64, 3, 64), _x(88), 1, 1, 1)
    assert c._use_wgrad_dir(_w(128, 3, 128), _x(64), 1, 1, 1)
    # measured losses / ineligible
    assert not c._use_wgrad_dir(_w(64, 3, 64), _x(44), 1, 1, 1)   # wash at 44
    assert not c._use_wgrad_dir(_w(256, 3, 256), _x(22), 1, 1, 1)
    assert not c._use_wgrad_dir(_w(64, 3, 64), _x(88), 2, 1, 1)   # strided
    assert not c._use_wgrad_dir(_w(64, 3, 64), _x(88), 1, 1, 2)   # dilated
    assert not c._use_wgrad_dir(_w(64, 1, 64), _x(88), 1, 0, 1)   # 1x1
    assert not c._use_wgrad_dir(_w(64, 3, 32), _x(88), 1, 1, 1)   # C % 64


def test_conv_dir_gate():
    # wins: W >= 128, cin in {32, 64}, cout <= 64
    assert c._use_conv_dir(64, 64, True, _x(176), 1, 1, 1)
    assert c._use_conv_dir(32, 64, True, _x(176), 1, 1, 1)
    assert c._use_conv_dir(64, 32, True, _x(128), 1, 1, 1)
    assert c._use_conv_dir(64, 64, True, _x(88), 1, 1, 1)  # pooled ROWS=2 win
    # ineligible forms / measured losses
    assert not c._use_conv_dir(64, 64, True, _x(44), 1, 1, 1)
    assert not c._use_conv_dir(128, 64, True, _x(176), 1, 1, 1)
    assert not c._use_conv_dir(64, 128, True, _x(176), 1, 1, 1)
    assert not c._use_conv_dir(64, 64, False, _x(176), 1, 0, 1)   # not 3x3
    assert not c._use_conv_dir(64, 64, True, _x(176), 2, 1, 1)    # strided


def test_wgrad_k1_is_opt_in_only():
    # measured NEGATIVE result: must stay off unless explicitly forced
    w = _w(256, 1, 64)
    assert not c._use_wgrad_k1(w, 0, 1)
    prev = c._WK1_ENV
    try:
        c._WK1_ENV = "1"
        assert c._use_wgrad_k1(w, 0, 1)
    finally:
        c._WK1_ENV = prev
