"""R2D2 value-function rescaling + burn-in helpers
(reference optimizer/burn_in.py)."""

from __future__ import annotations

import torch


def value_function_rescaling(x: torch.Tensor, eps: float = 1e-3) -> torch.Tensor:
    """h(x) = sign(x)(sqrt(|x|+1)-1) + eps*x (R2D2 paper table 2; reference
    burn_in.py:23-25)."""
    return torch.sign(x) * (torch.sqrt(torch.abs(x) + 1.0) - 1.0) + eps * x


def inverse_value_function_rescaling(x: torch.Tensor,
                                     eps: float = 1e-3) -> torch.Tensor:
    """h^{-1}(x) per 'Observe and Look Further' Prop. A.2 (reference
    burn_in.py:27-32)."""
    return torch.sign(x) * (
        ((torch.sqrt(1.0 + 4.0 * eps * (torch.abs(x) + 1.0 + eps)) - 1.0)
         / (2.0 * eps)).square() - 1.0)


def slice_in_burnin(size: int, tensor: torch.Tensor) -> torch.Tensor:
    """Drop the first ``size`` timesteps (reference burn_in.py:3-4)."""
    return tensor[:, size:]
