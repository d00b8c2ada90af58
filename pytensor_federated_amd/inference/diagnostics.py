"""MCMC chain diagnostics: split-R-hat, effective sample size, summaries.

The reference's demo reports an arviz summary (demo_model.py:44); this
module provides the same numbers without the arviz dependency, following
the split-chain formulations of Vehtari et al. 2021 (rank-normalization
omitted; the plain split statistics arviz also reports).
"""
from __future__ import annotations

from typing import Dict, Sequence

import numpy as np

__all__ = ["split_rhat", "effective_sample_size", "summary"]


def _to_chains(samples) -> np.ndarray:
    """Normalize input to [n_chains, n_draws]; 1-d input becomes 1 chain."""
    arr = np.asarray(samples, dtype=np.float64)
    if arr.ndim == 1:
        arr = arr[None, :]
    if arr.ndim != 2:
        raise ValueError("samples must be [n_draws] or [n_chains, n_draws]")
    return arr


def split_rhat(samples) -> float:
    """Split-chain potential scale reduction factor (Gelman-Rubin R-hat)."""
    chains = _to_chains(samples)
    n = chains.shape[1] // 2
    if n < 2:
        return float("nan")
    halves = np.concatenate([chains[:, :n], chains[:, n : 2 * n]], axis=0)
    means = halves.mean(axis=1)
    w = halves.var(axis=1, ddof=1).mean()
    b = n * means.var(ddof=1)
    var_plus = (n - 1) / n * w + b / n
    if w <= 0:
        return float("nan")
    return float(np.sqrt(var_plus / w))


def effective_sample_size(samples) -> float:
    """Split-chain ESS via the Geyer initial-monotone-sequence estimator."""
    chains = _to_chains(samples)
    n = chains.shape[1] // 2
    if n < 4:
        return float("nan")
    halves = np.concatenate([chains[:, :n], chains[:, n : 2 * n]], axis=0)
    m = halves.shape[0]
    means = halves.mean(axis=1, keepdims=True)
    w = halves.var(axis=1, ddof=1).mean()
    b = n * halves.mean(axis=1).var(ddof=1) if m > 1 else 0.0
    var_plus = (n - 1) / n * w + (b / n if m > 1 else 0.0)
    if var_plus <= 0:
        return float("nan")

    centered = halves - means
    # per-chain autocovariance via FFT
    n_fft = 1 << (2 * n - 1).bit_length()
    f = np.fft.rfft(centered, n=n_fft, axis=1)
    acov = np.fft.irfft(f * np.conj(f), n=n_fft, axis=1)[:, :n].real
    acov /= n
    rho = 1.0 - (w - acov.mean(axis=0)) / var_plus  # combined autocorrelation
    # Geyer initial-monotone estimator, Stan/arviz pairing: sum pairs
    # (rho[0]+rho[1]), (rho[2]+rho[3]), ... while positive, with rho[0] the
    # computed lag-0 value (slightly below 1 for finite n), enforce monotone
    # non-increase, then tau = -1 + 2*sum(pairs).
    tau = -1.0
    prev_pair = float("inf")
    t = 0
    while t + 1 < n:
        pair = rho[t] + rho[t + 1]
        if pair < 0:
            break
        pair = min(pair, prev_pair)
        tau += 2.0 * pair
        prev_pair = pair
        t += 2
    ess = m * n / max(tau, 1e-12)
    return float(min(ess, m * n))


def summary(chains_by_name: Dict[str, Sequence]) -> str:
    """arviz-style text summary: mean, sd, central 94% interval, ESS, R-hat.

    (Equal-tailed 3%/97% quantiles -- arviz reports an HDI; for the
    near-symmetric posteriors these drivers target the two coincide.)"""
    rows = []
    header = f"{'param':>12} {'mean':>10} {'sd':>10} {'q3%':>10} {'q97%':>10} {'ess':>8} {'r_hat':>6}"
    rows.append(header)
    for name, samples in chains_by_name.items():
        arr = _to_chains(samples)
        flat = arr.reshape(-1)
        lo, hi = np.percentile(flat, [3, 97])
        rows.append(
            f"{name:>12} {flat.mean():>10.4f} {flat.std(ddof=1):>10.4f} "
            f"{lo:>10.4f} {hi:>10.4f} {effective_sample_size(arr):>8.0f} "
            f"{split_rhat(arr):>6.3f}"
        )
    return "\n".join(rows)
