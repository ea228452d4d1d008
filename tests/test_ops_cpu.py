"""Unit tier for op-layer CPU fallbacks and planner helpers: every GPU
op must behave like its torch reference when the extension path is not
in play (the loud-failure GPU dispatch is covered by the gpu suite).
"""
import torch

from ravnest_amd.ops import Conv1x1, MXLinear, mx_linear, softmax
from ravnest_amd.ops.pool import MaxPool2d, AvgPool2d, AdaptiveAvgPool2d
from ravnest_amd.utils import wire_cast


def test_conv1x1_matches_conv2d():
    torch.manual_seed(0)
    for stride in (1, 2):
        ref = torch.nn.Conv2d(8, 16, 1, stride=stride, bias=False)
        ours = Conv1x1(8, 16, stride=stride)
        ours.load_state_dict(ref.state_dict())
        x = torch.randn(2, 8, 10, 10)
        assert torch.allclose(ours(x), ref(x), atol=1e-6)


def test_mx_linear_cpu_fallback():
    torch.manual_seed(1)
    m = MXLinear(32, 16)
    x = torch.randn(4, 32)
    ref = torch.nn.functional.linear(x, m.weight, m.bias)
    assert torch.allclose(m(x), ref, atol=1e-6)
    # functional form + from_linear interop
    lin = torch.nn.Linear(32, 16)
    m2 = MXLinear.from_linear(lin)
    assert torch.allclose(m2(x), lin(x), atol=1e-6)
    assert torch.allclose(mx_linear(x, lin.weight, lin.bias), lin(x),
                          atol=1e-6)


def test_softmax_cpu_fallback():
    x = torch.randn(5, 7)
    assert torch.allclose(softmax(x, -1), torch.softmax(x, -1), atol=1e-7)
    assert torch.allclose(softmax(x, 0), torch.softmax(x, 0), atol=1e-7)


def test_pool_cpu_fallbacks():
    torch.manual_seed(2)
    x = torch.randn(2, 3, 9, 9)
    assert torch.allclose(MaxPool2d(3, 2, 1)(x),
                          torch.nn.functional.max_pool2d(x, 3, 2, 1))
    assert torch.allclose(AvgPool2d(3, 1, 1)(x),
                          torch.nn.functional.avg_pool2d(x, 3, 1, 1))
    assert torch.allclose(AdaptiveAvgPool2d((1, 1))(x),
                          torch.nn.functional.adaptive_avg_pool2d(x, (1, 1)))


def test_wire_cast():
    f = torch.randn(3, 3)
    assert wire_cast(f, torch.bfloat16).dtype == torch.bfloat16
    assert wire_cast(f, None) is f
    i = torch.randint(0, 10, (3,))
    assert wire_cast(i, torch.bfloat16) is i  # ints never cast
    b = f.to(torch.bfloat16)
    assert wire_cast(b, torch.bfloat16) is b  # no-op when already there
