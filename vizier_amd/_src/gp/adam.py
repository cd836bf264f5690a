"""Adam-based ARD optimizer (alternative to the batched L-BFGS).

Capability parity with vizier/_src/jax/optimizers/optax_wrappers.py
(OptaxTrain :38): first-order training of the GP hyperparameters,
normalizing the loss by the observation count so one learning rate
works across study sizes. All restarts advance as one batched tensor
program (MI355X-first, like gp/lbfgs.py); per-restart best parameters
are tracked over epochs.
"""

from __future__ import annotations

from typing import Callable, Tuple

import torch


def minimize_adam(
    loss_fn: Callable[[torch.Tensor], torch.Tensor],
    x0: torch.Tensor,
    *,
    epochs: int = 100,
    learning_rate: float = 5e-2,
    normalize_by: float = 1.0,
) -> Tuple[torch.Tensor, torch.Tensor]:
  """Minimizes loss_fn over a batch of R parameter vectors with Adam.

  Args:
    loss_fn: (R, P) params -> (R,) losses (autograd-able; +inf allowed).
    x0: (R, P) initial points.
    epochs: Adam steps.
    learning_rate: Adam learning rate (on the normalized loss).
    normalize_by: divisor for the loss (typically the observation count,
      see OptaxTrain's pydoc).

  Returns:
    (x_best, f_best): best UNNORMALIZED losses (and params) per restart.
  """
  x = x0.detach().clone().requires_grad_(True)
  opt = torch.optim.Adam([x], lr=learning_rate)
  inf = torch.tensor(float('inf'), dtype=x.dtype, device=x.device)
  f_best = torch.full((x.shape[0],), float('inf'), dtype=x.dtype,
                      device=x.device)
  x_best = x.detach().clone()
  for _ in range(epochs):
    opt.zero_grad(set_to_none=True)
    loss = loss_fn(x)
    finite = torch.isfinite(loss)
    improved = finite & (loss.detach() < f_best)
    f_best = torch.where(improved, loss.detach(), f_best)
    x_best = torch.where(improved.unsqueeze(-1), x.detach(), x_best)
    # Backward only through the finite rows (inf rows poison grads).
    safe = torch.where(finite, loss, torch.zeros_like(loss))
    (safe.sum() / normalize_by).backward()
    with torch.no_grad():
      x.grad = torch.where(
          torch.isfinite(x.grad), x.grad,
          torch.zeros_like(x.grad)) if x.grad is not None else None
    opt.step()
  # Final evaluation (the last step may have improved).
  with torch.no_grad():
    loss = loss_fn(x)
    finite = torch.isfinite(loss)
    improved = finite & (loss < f_best)
    f_best = torch.where(improved, loss, f_best)
    x_best = torch.where(improved.unsqueeze(-1), x.detach(), x_best)
  return x_best, f_best
