"""Cyclic (DFT / MDS-style) gradient code — construction and host-side decode math.

Re-derivation of the Draco cyclic code (reference: /root/reference/src/coding.py:4-68,
/root/reference/src/master/cyclic_master.py:152-188, /root/reference/src/c_coding.cpp:15-84),
implemented from the underlying algebra rather than translated:

Let n = number of logical workers, s = tolerated Byzantine workers, s_hat = 2s+1.
Let F be the symmetric DFT matrix F[p,q] = exp(-2*pi*i*p*q/n) and C = F/sqrt(n) (unitary,
symmetric).  Split C = [C1 | C2] with C1 = first n-2s columns, C2 = last 2s columns.

* Encoding matrix W = C1 @ Q (n x n), where column-normalised Q (Q[0,:] = 1) is solved
  per-row by least squares so that W[i, j] = 0 outside the cyclic band
  j in {i, i+1, ..., i+2s mod n}.  Worker i computes the 2s+1 sub-batch gradients in its
  band and ships the single complex combination r_i = sum_j W[i,j] * g_j.
* Parity: W_perp = C2^H, and W_perp @ W = C2^H C1 Q = 0 because C is unitary.
* Decode: received R = W @ G + E, where E has at most s nonzero rows (the adversaries).
  - syndrome = W_perp @ (R @ z) for a random projection z: only the error term survives.
    syndrome_k = sum_{i in bad} eps_i * z_i^(n-2s+k) with z_i = exp(2*pi*i*I/n): classic
    power-sum syndromes, so the error-locator polynomial a(x) (monic, degree s) satisfies
    a Hankel system in the syndromes (the reference solves the same system with Eigen SVD,
    c_coding.cpp:75-81 — but rebuilds C on every call; we precompute everything once).
  - estimation[i] = a(z_i) vanishes exactly at adversarial i -> healthy set H.
  - recombination: solve x with C1[H[:n-2s], :]^T x = e1; then v scattered at H[:n-2s]
    satisfies v^T W = e1^T Q = 1^T (because Q[0,:] = 1), so v^T R = sum_j g_j exactly,
    untouched by the adversaries (v is supported on healthy rows only).

All construction is done once per run in complex128 (the reference re-solves search_w twice
per launch and rebuilds C in C++ per step — both fixed here).  Per-step work is:
one (n x d) projection GEMV, O(s^3 + n^2) host algebra on n-vectors, and one (n x d)
recombination GEMV — the two GEMVs are the HIP kernels (ops.cyclic_*) and are sharded
over ranks by the parallel layer.
"""
from __future__ import annotations

import functools
from dataclasses import dataclass, field

import numpy as np


@dataclass(eq=False)  # eq=False keeps the class hashable (id) for the lru_cache below
class CyclicCode:
    """Precomputed cyclic-code operators for (n workers, s adversaries)."""

    n: int
    s: int
    W: np.ndarray = field(repr=False)  # (n, n) complex128 encoding matrix (banded)
    support: np.ndarray = field(repr=False)  # (n, 2s+1) int64: columns of each row's band
    W_perp: np.ndarray = field(repr=False)  # (2s, n) complex128 parity-check
    C1: np.ndarray = field(repr=False)  # (n, n-2s) complex128
    Z: np.ndarray = field(repr=False)  # (n, s+1) complex128 locator evaluation matrix

    @property
    def s_hat(self) -> int:
        return 2 * self.s + 1

    # ---------------------------------------------------------------- decode
    def locate_errors(self, syndrome_proj: np.ndarray, known_bad=()) -> np.ndarray:
        """Return sorted healthy worker indices from a projected syndrome.

        syndrome_proj: (2s,) complex — W_perp @ (R @ z) for random projection z.

        Fast path: the reference's error-locator polynomial (cyclic_master.py:152-170,
        c_coding.cpp:75-81), but with a RELATIVE threshold — the reference's absolute
        1e-9 cutoff silently mislocates whenever syndrome magnitudes are far from O(1)
        (e.g. fp32 transport or large gradients).  The located set is then VERIFIED by
        fitting error magnitudes (syndrome = W_perp[:, bad] @ eps is exactly solvable
        for the true bad set, since any 2s columns of the Vandermonde-structured
        parity matrix are independent); on verification failure a subset search over
        all C(n, s) candidate sets (tiny for the s<=3 regime this code targets)
        guarantees the correct location — this also covers the <s-adversaries case
        where the degenerate Hankel system makes the locator unreliable.
        """
        s, n = self.s, self.n
        syn = np.asarray(syndrome_proj, dtype=np.complex128).reshape(-1)
        assert syn.shape == (2 * s,), syn.shape
        syn_norm = np.linalg.norm(syn)

        # Hankel system for monic locator a(x) = x^s - sum_{j<s} alpha_j x^j:
        # rows i: sum_j syn[s-1-i+j] * alpha_j = syn[2s-1-i]   (c_coding.cpp:75-79)
        A = np.zeros((s, s), dtype=np.complex128)
        b = np.zeros(s, dtype=np.complex128)
        for i in range(s):
            A[i] = syn[s - i - 1 : 2 * s - i - 1]
            b[i] = syn[2 * s - i - 1]
        try:
            alpha = np.linalg.lstsq(A, b, rcond=None)[0]
        except np.linalg.LinAlgError:
            alpha = np.zeros(s, dtype=np.complex128)
        poly = np.zeros(s + 1, dtype=np.complex128)
        poly[:s] = -alpha
        poly[s] = 1.0
        est = np.abs(self.Z @ poly)  # |a(z_i)|; ~0 at adversarial i
        bad = np.nonzero(est <= 1e-5 * max(est.max(), 1e-300))[0]
        known = np.asarray(sorted(known_bad), dtype=np.int64)
        if len(known):  # erasures (e.g. straggler timeouts) are errors at KNOWN rows
            bad = np.union1d(bad, known)
        if 0 < len(bad) <= s and self._verify_bad(bad, syn, syn_norm):
            pass
        else:
            bad = self._search_bad(syn, syn_norm, known)
        mask = np.ones(n, dtype=bool)
        mask[bad] = False
        return np.nonzero(mask)[0]

    def _verify_bad(self, bad: np.ndarray, syn: np.ndarray, syn_norm: float) -> bool:
        B = self.W_perp[:, bad]
        eps, *_ = np.linalg.lstsq(B, syn, rcond=None)
        resid = np.linalg.norm(B @ eps - syn)
        return resid <= 1e-5 * max(syn_norm, 1e-300)

    def _search_bad(self, syn: np.ndarray, syn_norm: float, known=()) -> np.ndarray:
        """Exhaustive sparse-recovery fallback: smallest-residual support of size <= s
        (always containing the known erasure locations)."""
        import itertools

        known = np.asarray(sorted(known), dtype=np.int64)
        rest = [i for i in range(self.n) if i not in set(known.tolist())]
        best, best_resid = known, syn_norm
        for k in range(0, self.s - len(known) + 1):
            if k == 0 and len(known) == 0:
                continue
            for subset in itertools.combinations(rest, k):
                idx = np.union1d(known, np.asarray(subset, dtype=np.int64)).astype(np.int64)
                B = self.W_perp[:, idx]
                eps, *_ = np.linalg.lstsq(B, syn, rcond=None)
                resid = np.linalg.norm(B @ eps - syn)
                if resid < best_resid:
                    best, best_resid = idx, resid
                    if resid <= 1e-9 * max(syn_norm, 1e-300):
                        return best
        return best

    @functools.lru_cache(maxsize=256)
    def _recover_cached(self, healthy_key: tuple) -> np.ndarray:
        healthy = np.asarray(healthy_key, dtype=np.int64)
        n, s = self.n, self.s
        use = healthy[: n - 2 * s]
        sub = self.C1[use, :]  # (n-2s, n-2s)
        x = np.linalg.lstsq(sub.T, _e1(n - 2 * s), rcond=None)[0]
        v = np.zeros(n, dtype=np.complex128)
        v[use] = x
        return v

    def recombination_vector(self, healthy: np.ndarray) -> np.ndarray:
        """(n,) complex v with v^T W = 1^T, supported on healthy rows."""
        if len(healthy) < self.n - 2 * self.s:
            raise RuntimeError(
                f"cyclic decode: only {len(healthy)} healthy rows located, "
                f"need {self.n - 2 * self.s} (more than s={self.s} adversaries?)"
            )
        return self._recover_cached(tuple(int(i) for i in healthy))

    # -------------------------------------------------------- numpy oracles
    def encode_oracle(self, worker: int, grads: np.ndarray) -> np.ndarray:
        """Reference encode for tests: grads (2s+1, d) real (band order) -> (d,) complex."""
        cols = self.support[worker]
        out = np.zeros(grads.shape[1], dtype=np.complex128)
        for k, j in enumerate(cols):
            out += self.W[worker, j] * grads[k]
        return out

    def decode_oracle(self, R: np.ndarray, rng: np.random.Generator | None = None) -> np.ndarray:
        """Full decode for tests: R (n, d) complex -> (d,) real = sum_j g_j."""
        rng = rng or np.random.default_rng(0)
        z = rng.normal(loc=1.0, size=R.shape[1])
        healthy = self.locate_errors(self.W_perp @ (R @ z))
        v = self.recombination_vector(healthy)
        return np.real(v @ R)


def _e1(k: int) -> np.ndarray:
    e = np.zeros(k, dtype=np.complex128)
    e[0] = 1.0
    return e


def _dft_c(n: int) -> np.ndarray:
    p = np.arange(n)
    return np.exp(-2j * np.pi * np.outer(p, p) / n) / np.sqrt(n)


def _band_support(n: int, s_hat: int) -> np.ndarray:
    """(n, s_hat) columns of each row's cyclic band {i..i+s_hat-1 mod n}."""
    i = np.arange(n)[:, None]
    k = np.arange(s_hat)[None, :]
    return (i + k) % n


def build_cyclic_code(n: int, s: int) -> CyclicCode:
    """Construct the code once (reference rebuilds per launch AND per step; we don't)."""
    if s < 1:
        raise ValueError("cyclic code needs s >= 1")
    s_hat = 2 * s + 1
    if n < s_hat + 1:
        raise ValueError(f"cyclic code needs n >= 2s+2 (got n={n}, s={s})")
    C = _dft_c(n)
    C1 = C[:, : n - 2 * s]
    C2 = C[:, n - 2 * s :]
    support = _band_support(n, s_hat)

    # Solve Q column-by-column: Q[0, :] = 1 fixed (this is what makes the recombination
    # v^T W = e1^T Q = 1^T work); remaining entries chosen so W = C1 @ Q vanishes off the
    # band pattern (reference: coding.py:54-68 via scipy lsq_linear; plain complex lstsq
    # is exact here and dependency-free).  Row w of W (worker w's code row) is supported
    # on columns {w..w+2s mod n}; equivalently column j of W is supported on rows
    # {j-2s..j mod n} — the constraint is applied per column, as in the reference.
    m = n - 2 * s
    Q = np.ones((m, n), dtype=np.complex128)
    in_band = np.zeros((n, n), dtype=bool)  # in_band[w, j]: worker w covers sub-batch j
    np.put_along_axis(in_band, support, True, axis=1)
    for j in range(n):
        zero_rows = np.nonzero(~in_band[:, j])[0]  # workers whose row must be 0 at col j
        A = C1[zero_rows, 1:]
        b = -C1[zero_rows, 0]
        q = np.linalg.lstsq(A, b, rcond=None)[0]
        Q[1:, j] = q
    W = np.ascontiguousarray(C1 @ Q)  # (n, n); W[worker, sub_batch]
    # clean numerically-zero off-band entries
    W[~in_band] = 0.0

    i = np.arange(n)
    Z = np.exp(2j * np.pi * np.outer(i, np.arange(s + 1)) / n)
    return CyclicCode(n=n, s=s, W=W, support=support, W_perp=C2.conj().T, C1=C1, Z=Z)
