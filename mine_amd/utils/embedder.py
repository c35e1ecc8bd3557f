"""NeRF positional encoding for the continuous disparity input.

Reference behavior (ref utils.py:144-193): for a scalar input x and
`multires` frequencies with log sampling, output
``[x, sin(2^0 x), cos(2^0 x), ..., sin(2^(L-1) x), cos(2^(L-1) x)]``
-> out_dim = 1 + 2*multires (21 for the default multires=10, ref
configs/params_default.yaml `model.pos_encoding_multires`).

Implemented as a single vectorized outer-product + interleave rather
than a list of lambdas (one kernel, no 21-way cat).
"""
from __future__ import annotations

from typing import Tuple

import torch


class PositionalEncoder:
    def __init__(self, num_freqs: int, input_dims: int = 1, include_input: bool = True,
                 log_sampling: bool = True):
        self.num_freqs = num_freqs
        self.input_dims = input_dims
        self.include_input = include_input
        if log_sampling:
            self.freq_bands = 2.0 ** torch.linspace(0.0, num_freqs - 1, steps=num_freqs)
        else:
            self.freq_bands = torch.linspace(1.0, 2.0 ** (num_freqs - 1), steps=num_freqs)
        self.out_dim = input_dims * (int(include_input) + 2 * num_freqs)

    def __call__(self, x: torch.Tensor) -> torch.Tensor:
        """x: (..., D) -> (..., D * (1 + 2*num_freqs)).

        Channel order matches the reference's embed_fns list:
        [x, sin(f0 x), cos(f0 x), sin(f1 x), cos(f1 x), ...] per input dim
        (for D==1 the orders coincide; D>1 interleaves per-frequency blocks
        identically to the reference's cat of per-fn outputs).
        """
        freqs = self.freq_bands.to(device=x.device, dtype=x.dtype)  # (F,)
        xf = x.unsqueeze(-1) * freqs  # (..., D, F)
        sin = torch.sin(xf)
        cos = torch.cos(xf)
        sc = torch.stack((sin, cos), dim=-1)  # (..., D, F, 2)
        # reference cat order: for each freq f: [sin(f*x) over D dims] then [cos(f*x)]
        sc = sc.permute(*range(x.dim() - 1), -2, -1, -3)  # (..., F, 2, D)
        flat = sc.reshape(*x.shape[:-1], 2 * self.num_freqs * self.input_dims)
        if self.include_input:
            return torch.cat((x, flat), dim=-1)
        return flat


def get_embedder(multires: int) -> Tuple[PositionalEncoder, int]:
    enc = PositionalEncoder(num_freqs=multires, input_dims=1, include_input=True,
                            log_sampling=True)
    return enc, enc.out_dim
