"""Autograd integration of the fused conv3x3+ReLU+maxpool2x2 kernels.

`ConvReluPool2d` is a drop-in module computing
conv2d(x, W, b, padding=1) -> relu -> max_pool2d(2) in ONE gfx950 kernel
per direction (ops/csrc/smallcnn.hip) on device tensors, with an exact
torch-semantics backward (first-index pool tie-break, relu gating). On
CPU it falls back to the equivalent torch ops, which also serve as the
differential-test oracle.
"""

import torch
from torch import nn
from torch.nn import functional as F

from . import _C, is_available


class _ConvReluPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, need_input_grad):
        N, CIN, H, W = x.shape
        COUT = weight.shape[0]
        pooled = torch.empty(N, COUT, H // 2, W // 2, device=x.device, dtype=x.dtype)
        argmax = torch.empty(N, COUT, H // 2, W // 2, device=x.device, dtype=torch.uint8)
        _C.conv3x3_relu_pool_fwd(x, weight, bias, pooled, argmax)
        ctx.save_for_backward(x, weight, pooled, argmax)
        ctx.need_input_grad = need_input_grad
        return pooled

    @staticmethod
    def backward(ctx, dpooled):
        x, weight, pooled, argmax = ctx.saved_tensors
        dpooled = dpooled.contiguous()
        dw = torch.zeros_like(weight)
        db = torch.zeros(weight.shape[0], device=weight.device, dtype=weight.dtype)
        _C.conv3x3_relu_pool_bwd_weight(dpooled, argmax, pooled, x, dw, db)
        dx = None
        if ctx.need_input_grad:
            dx = torch.empty_like(x)
            _C.conv3x3_relu_pool_bwd_data(dpooled, argmax, pooled, weight, dx)
        return dx, dw, db, None


class ConvReluPool2d(nn.Module):
    """conv3x3(pad=1) + ReLU + maxpool2x2, fused on gfx950.

    first_layer=True skips the input gradient (e.g. the image layer).
    """

    def __init__(self, in_channels: int, out_channels: int, first_layer: bool = False):
        super().__init__()
        self.first_layer = first_layer
        conv = nn.Conv2d(in_channels, out_channels, 3, padding=1)  # init only
        self.weight = nn.Parameter(conv.weight.detach().clone())
        self.bias = nn.Parameter(conv.bias.detach().clone())

    def forward(self, x):
        if x.is_cuda and is_available():
            need_dx = (not self.first_layer) or x.requires_grad
            return _ConvReluPoolFn.apply(x, self.weight, self.bias, need_dx)
        # CPU fallback / oracle
        y = F.conv2d(x, self.weight, self.bias, padding=1)
        return F.max_pool2d(F.relu(y), 2)


class FusedMnistCNN(nn.Module):
    """The benchmark MNIST-CNN built from fused layers + nn.Linear.

    Architecture identical to models/mnist.mnist_cnn (reference
    examples/mnist.py:27-37); state_dict-compatible keys differ (fused
    modules), parameter count identical.
    """

    def __init__(self):
        super().__init__()
        self.layer1 = ConvReluPool2d(1, 16, first_layer=True)
        self.layer2 = ConvReluPool2d(16, 16)
        self.fc = nn.Linear(784, 10)

    def forward(self, x):
        x = self.layer1(x)
        x = self.layer2(x)
        return self.fc(x.flatten(1))
