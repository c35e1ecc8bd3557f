import math

import torch

from mine_amd.utils.embedder import get_embedder


def test_out_dim():
    enc, out_dim = get_embedder(10)
    assert out_dim == 21
    x = torch.rand(7, 1)
    y = enc(x)
    assert y.shape == (7, 21)


def test_values_match_reference_order():
    """Channel order: [x, sin(2^0 x), cos(2^0 x), sin(2 x), cos(2 x), ...]
    (ref utils.py:144-175 with include_input + log_sampling)."""
    enc, _ = get_embedder(4)
    x = torch.tensor([[0.3], [1.7]])
    y = enc(x)
    assert y.shape == (2, 9)
    for b in range(2):
        v = x[b, 0].item()
        expected = [v]
        for k in range(4):
            expected += [math.sin(2 ** k * v), math.cos(2 ** k * v)]
        torch.testing.assert_close(y[b], torch.tensor(expected), rtol=1e-5, atol=1e-6)
