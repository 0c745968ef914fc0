"""Multinomial Naive Bayes on numeric feature vectors.

Replaces MLlib `NaiveBayes.train` as used by the classification template
(examples/scala-parallel-classification/.../NaiveBayesAlgorithm.scala:35-59:
LabeledPoints of numeric attrs, additive smoothing lambda). Torch tensor
math end to end — runs on MI355X or CPU identically (SURVEY.md §2.9 K6:
tiny; library kernels suffice, no custom HIP needed).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch


@dataclass
class NaiveBayesModel:
    labels: torch.Tensor   # (C,) class label values
    pi: torch.Tensor       # (C,) log priors
    theta: torch.Tensor    # (C, D) log feature likelihoods

    def predict(self, x: torch.Tensor) -> torch.Tensor:
        """x: (D,) or (B, D) → predicted label(s)."""
        single = x.dim() == 1
        if single:
            x = x.unsqueeze(0)
        scores = x.to(self.theta) @ self.theta.t() + self.pi  # (B, C)
        out = self.labels[scores.argmax(dim=1)]
        return out[0] if single else out

    def predict_scores(self, x: torch.Tensor) -> torch.Tensor:
        if x.dim() == 1:
            x = x.unsqueeze(0)
        return x.to(self.theta) @ self.theta.t() + self.pi

    def to(self, device) -> "NaiveBayesModel":
        return NaiveBayesModel(self.labels.to(device), self.pi.to(device),
                               self.theta.to(device))


def train_naive_bayes(X: torch.Tensor, y: torch.Tensor,
                      lambda_: float = 1.0) -> NaiveBayesModel:
    """Multinomial NB with additive smoothing (MLlib semantics):
    pi_c = log(n_c / n); theta_cj = log((S_cj + λ) / (S_c + λ D))
    where S_cj = sum of feature j over class-c samples."""
    if (X < 0).any():
        raise ValueError("multinomial NB requires nonnegative features")
    labels, y_idx = torch.unique(y, return_inverse=True)
    C, D = labels.numel(), X.shape[1]
    n_c = torch.zeros(C, dtype=torch.float64, device=X.device)
    n_c.scatter_add_(0, y_idx, torch.ones_like(y_idx, dtype=torch.float64))
    S = torch.zeros((C, D), dtype=torch.float64, device=X.device)
    S.index_add_(0, y_idx, X.double())
    pi = (n_c / n_c.sum()).log()
    theta = ((S + lambda_) / (S.sum(dim=1, keepdim=True) + lambda_ * D)).log()
    return NaiveBayesModel(labels, pi.float(), theta.float())
