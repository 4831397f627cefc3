"""CPU eager path of the fused BatchNorm(+ReLU,+residual) op: matches
torch's fp32 BN composition in train AND eval mode, and a Bottleneck
block trains end-to-end on CPU through it."""

import torch

from scaffold import ops


def _ref(x, res, w, b, rm, rv, mom, eps, relu, training):
    y = torch.nn.functional.batch_norm(
        x.float(), rm, rv, w, b, training, mom, eps)
    if res is not None:
        y = y + res.float()
    if relu:
        y = torch.nn.functional.relu(y)
    return y.to(x.dtype)


def test_bn_act_eager_train_and_eval():
    torch.manual_seed(5)
    N, C, H, W = 2, 8, 5, 5
    x = torch.randn(N, C, H, W)
    res = torch.randn(N, C, H, W)
    w = torch.rand(C) + 0.5
    b = torch.randn(C)
    for training in (True, False):
        for relu, use_res in [(True, True), (True, False), (False, False)]:
            rm1, rv1 = torch.zeros(C), torch.ones(C)
            rm2, rv2 = torch.zeros(C), torch.ones(C)
            got = ops.batch_norm_act(x, res if use_res else None, w, b,
                                     rm1, rv1, 0.1, 1e-5, relu, training)
            want = _ref(x, res if use_res else None, w, b, rm2, rv2,
                        0.1, 1e-5, relu, training)
            assert torch.allclose(got, want, atol=1e-5), (training, relu, use_res)
            assert torch.allclose(rm1, rm2) and torch.allclose(rv1, rv2)


def test_bottleneck_block_cpu_backward():
    from scaffold.models.resnet import Bottleneck

    torch.manual_seed(6)
    blk = Bottleneck(16, 4)  # in 16 -> width 4 -> out 16, no downsample
    x = torch.randn(2, 16, 8, 8, requires_grad=True)
    y = blk(x)
    y.sum().backward()
    assert torch.isfinite(y).all()
    assert x.grad is not None and torch.isfinite(x.grad).all()
    for p in blk.parameters():
        assert p.grad is not None
