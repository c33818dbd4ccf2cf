"""Benchmark model zoo (synthetic-data training configs from BASELINE.json)."""

from .gpt2 import GPT2, GPT2Config, gpt2_small, gpt2_tiny  # noqa: F401
from .mnist import SyntheticMnist, mnist_cnn  # noqa: F401
from .resnet import ResNet, resnet50  # noqa: F401
