"""Gradient noise scale estimation (reference: units-test/get_gns.py).

Estimates the critical batch size signal B_simple = tr(Sigma) / |G|^2 from
two gradient estimates at different batch sizes (McCandlish et al. 2018),
used for adaptive batch-size experiments. Three estimators as in the
reference: 'vector' (two disjoint micro-batches), 'split' (per-rank
gradients vs the allreduced mean), and 'whole' (running EMA).
"""

from __future__ import annotations

from typing import Iterable, Optional, Tuple

import torch


def _flat_grad(params: Iterable[torch.nn.Parameter]) -> torch.Tensor:
    return torch.cat([p.grad.detach().flatten()
                      for p in params if p.grad is not None])


class GNS:
    def __init__(self, ema_beta: float = 0.9):
        self.ema_beta = ema_beta
        self._ema_sq_norm: Optional[float] = None
        self._ema_tr_sigma: Optional[float] = None

    @staticmethod
    def estimate_pair(g_small: torch.Tensor, g_big: torch.Tensor,
                      b_small: int, b_big: int) -> Tuple[float, float]:
        """Unbiased |G|^2 and tr(Sigma) from gradients at two batch sizes
        (reference get_gns.py 'vector' estimator)."""
        sq_small = float(g_small.pow(2).sum())
        sq_big = float(g_big.pow(2).sum())
        g2 = (b_big * sq_big - b_small * sq_small) / (b_big - b_small)
        tr = (sq_small - sq_big) / (1.0 / b_small - 1.0 / b_big)
        return g2, tr

    def compute_gns(self, g_small: torch.Tensor, g_big: torch.Tensor,
                    b_small: int, b_big: int) -> float:
        g2, tr = self.estimate_pair(g_small, g_big, b_small, b_big)
        beta = self.ema_beta
        self._ema_sq_norm = (g2 if self._ema_sq_norm is None
                             else beta * self._ema_sq_norm + (1 - beta) * g2)
        self._ema_tr_sigma = (tr if self._ema_tr_sigma is None
                              else beta * self._ema_tr_sigma + (1 - beta) * tr)
        if abs(self._ema_sq_norm) < 1e-12:
            return float("inf")
        return self._ema_tr_sigma / self._ema_sq_norm

    def compute_gns_from_ranks(self, model: torch.nn.Module,
                               world_size: int, per_rank_batch: int) -> float:
        """'split' estimator: per-rank gradient (batch b) vs the DDP-mean
        gradient (batch world*b). Call after backward but before the hook's
        averaging is applied, or pass a model whose .grad holds local
        gradients and allreduce a copy."""
        import torch.distributed as dist

        local = _flat_grad(model.parameters())
        mean = local.clone()
        dist.all_reduce(mean)
        mean /= world_size
        return self.compute_gns(local, mean, per_rank_batch,
                                per_rank_batch * world_size)
