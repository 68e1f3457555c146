"""Auxiliary block tests."""

import torch

from esr_amd.models.aux_blocks import (ConvLayer1D, ConvLayer3D,
                                       DenseEdgeConv, DilatedBlock,
                                       InceptionBlock, SelfAttention, knn)


def test_self_attention():
    m = SelfAttention(8)
    x = torch.rand(2, 16, 8)
    y = m(x)
    assert y.shape == x.shape
    y.mean().backward()


def test_conv_1d_3d():
    assert ConvLayer1D(8, 16, 3, padding=1)(torch.rand(2, 8, 32)).shape == \
        (2, 16, 32)
    assert ConvLayer3D(2, 4)(torch.rand(1, 2, 4, 8, 8)).shape == \
        (1, 4, 4, 8, 8)


def test_inception_dilated():
    x = torch.rand(1, 4, 16, 16)
    assert InceptionBlock(4, 8)(x).shape == (1, 8, 16, 16)
    assert DilatedBlock(4, 8)(x).shape == (1, 8, 16, 16)


def test_knn_self_is_nearest():
    pos = torch.rand(2, 32, 3)
    idx = knn(pos, 4)
    assert idx.shape == (2, 32, 4)
    assert (idx[:, :, 0] == torch.arange(32)).all()  # self is 0-distance


def test_dense_edge_conv():
    m = DenseEdgeConv(8, growth=12, num_layers=3, k=4)
    y = m(torch.rand(2, 32, 8))
    assert y.shape == (2, 32, 2 * 8 + 3 * 12)
    y.mean().backward()


def test_feedback_block_recurrence():
    import torch
    from esr_amd.models.aux_blocks import FeedbackBlock
    fb = FeedbackBlock(8, num_groups=3, scale=2)
    x = torch.rand(1, 8, 16, 16)
    y1 = fb(x)
    y2 = fb(x)  # hidden feedback -> different output on same input
    assert y1.shape == (1, 8, 16, 16)
    assert not torch.allclose(y1, y2)
    fb.reset_state()
    y3 = fb(x)
    assert torch.allclose(y1, y3, atol=1e-6)
