"""Unit tests for the library-GEMM routing thresholds (pure logic, CPU).

The measured crossovers live in split_learning_amd/ops/functional.py
(_LIB_MM_THRESH / _LIB_LIN_THRESH, GPU sweep in profiles/SUMMARY.md
"library-GEMM routing"); these tests pin the routing DECISIONS for the model
zoo's shapes so a threshold or formula regression shows up on CPU."""

import torch

from split_learning_amd.ops import functional as hf


def test_use_lib_mm_model_shapes():
    # ViT attention: B*H=128, S=65, hd=32 -> 3.5e7 FLOPs -> library
    assert hf.use_lib_mm(128, 65, 65, 32)
    # KWT attention: B*H=32, S=99(ish), hd=64 -> 4e7 -> library
    assert hf.use_lib_mm(32, 98, 98, 64)
    # fused-attention numerics test shape: tiny -> native
    assert not hf.use_lib_mm(12, 7, 7, 64)
    # matmul autograd test shape: tiny -> native
    assert not hf.use_lib_mm(1, 48, 80, 96)


def test_linear_routing_decision():
    thresh = hf._LIB_LIN_THRESH

    def flops(rows, n, k):
        return 2.0 * rows * n * k

    # VGG16 classifier [32,512]x[512,512]: native (our kernel wins, measured)
    assert flops(32, 512, 512) < thresh
    # MobileNet fc [32,1024]x[1024,10]: native
    assert flops(32, 10, 1024) < thresh
    # ViT mlp [2080,128]x[128,256]: library
    assert flops(2080, 256, 128) >= thresh
    # BERT dense [4096,768]x[768,768]: library
    assert flops(4096, 768, 768) >= thresh
    # KWT out_proj [3136,64]x[64,64]: library (2.57e7)
    assert flops(3136, 64, 64) >= thresh


def test_linear_cpu_library_route_works():
    """Big shapes on CPU take F.linear (no native ext needed) and keep grads."""
    x = torch.randn(4096, 64, requires_grad=True)
    w = torch.randn(128, 64, requires_grad=True)
    y = hf.linear(x, w, None)
    assert y.shape == (4096, 128)
    y.sum().backward()
    assert x.grad is not None and w.grad is not None
