"""Restarted Lanczos eigensolver for the smallest eigenpairs of a symmetric
sparse matrix.

Reference parity: raft/sparse/solver/lanczos.cuh (CuPy-style restarted Lanczos:
lanczos_aux tridiagonalization with full reorthogonalization via 2 gemvs,
lanczos_solve_ritz on the projected matrix, thick restart; lanczos_types.hpp
config) exposed as pylibraft.sparse.linalg.eigsh (lanczos.pyx:99).

Algorithm: thick-restart Lanczos (Wu & Simon). After each ncv-step cycle the
projected matrix T is diagonalized; the k lowest Ritz vectors are kept, the
restart couplings beta_last*s[last,:k] form an arrowhead row in the new T, and
the recurrence continues from index k. T is a small [ncv, ncv] dense matrix —
diagonalized on-device (rocSOLVER syevd via torch.linalg.eigh).

MI355X: the hot loop is SpMV (native wave64 CSR kernel, HBM-bound ~12 B/nnz)
+ two tall-skinny gemvs for reorthogonalization (rocBLAS); everything stays
device-resident across iterations.
"""
from __future__ import annotations

from dataclasses import dataclass

import torch

from raft_amd.random.rng import RngState, uniform
from .linear_operator import LinearOperator, csr_operator
from ..types import CSR, COO
from ..convert import coo_to_csr


@dataclass
class LanczosConfig:
    """lanczos_solver_config parity (lanczos_types.hpp)."""
    n_components: int = 6
    max_iterations: int = 100     # restart cycles
    ncv: int = 0                  # subspace size; 0 -> min(n, max(2k+1, 32))
    tolerance: float = 1e-9
    seed: int = 42


def lanczos_min_eigenpairs(a, k: int | None = None, config: LanczosConfig | None = None,
                           v0: torch.Tensor | None = None):
    """Smallest-k eigenpairs of symmetric A (CSR/COO/LinearOperator/dense).

    Returns (eigenvalues [k] ascending, eigenvectors [n, k]).
    """
    cfg = config or LanczosConfig()
    if k is not None:
        cfg.n_components = k
    k = cfg.n_components

    if isinstance(a, COO):
        a = coo_to_csr(a)
    if isinstance(a, CSR):
        op, n, device, dtype = csr_operator(a), a.n_rows, a.device, a.values.dtype
    elif isinstance(a, LinearOperator):
        op, n, device, dtype = a, a.shape[0], a.device, a.dtype
    else:
        t = a
        op = LinearOperator(tuple(t.shape), lambda x: t @ x, device=t.device, dtype=t.dtype)
        n, device, dtype = t.shape[0], t.device, t.dtype

    ncv = cfg.ncv if cfg.ncv > 0 else min(n, max(2 * k + 1, 32))
    assert k < ncv <= n, f"need k < ncv <= n (k={k}, ncv={ncv}, n={n})"

    state = RngState(seed=cfg.seed)
    if v0 is None:
        v0 = uniform((n,), -1.0, 1.0, state=state, device=device, dtype=dtype)

    v = torch.zeros((ncv, n), dtype=dtype, device=device)   # Lanczos/Ritz basis
    t_mat = torch.zeros((ncv, ncv), dtype=dtype, device=device)  # projected matrix
    v[0] = v0 / v0.norm()
    v_next = None   # the (ncv+1)-th basis vector carried into the restart
    beta_last = torch.zeros((), dtype=dtype, device=device)

    def _reorth(u, basis):
        # two-pass classical Gram-Schmidt against `basis` (2 gemvs per pass),
        # the reference's "full reorth" (lanczos.cuh:345-369)
        u = u - basis.t() @ (basis @ u)
        u = u - basis.t() @ (basis @ u)
        return u

    # fused-step scratch (GPU fp32): device-resident alpha/norm2 scalars and
    # the carried (v_next, beta_last) buffers — the eager loop is ~16 host
    # dispatches per step; the fused kernels cut it to ~8 (hipGraph capture
    # regressed on ROCm 7.2, so dispatch count is the lever)
    fused_ok = device.type == "cuda" and dtype == torch.float32
    cycle_ok = False
    if fused_ok:
        from raft_amd._ext import require_ext as _re
        _ext = _re()
        _alpha = torch.zeros(1, dtype=torch.float64, device=device)
        _norm2 = torch.zeros(1, dtype=torch.float64, device=device)
        _v_next_buf = torch.empty(n, dtype=dtype, device=device)
        _beta_buf = torch.zeros(1, dtype=dtype, device=device)
        if isinstance(a, CSR):
            # plain-CSR operator: the WHOLE extension cycle runs as one C++
            # call (ext.lanczos_cycle_) — measured ~1 ms/step of python
            # dispatch at 10M rows disappears
            _ci = a.indptr.to(torch.int32).contiguous()
            _cj = a.indices.to(torch.int32).contiguous()
            _cv = a.values.contiguous()
            _u_buf = torch.empty(n, dtype=dtype, device=device)
            _w_buf = torch.empty(ncv, dtype=dtype, device=device)
            cycle_ok = True

    def _extend(start: int, careful: bool = False):
        """Run the three-term recurrence from index `start` to ncv-1, filling
        t_mat tridiagonally below/right of `start` (lanczos_aux).

        The loop is host-sync-free: beta degeneracy (invariant subspace) is
        detected ONCE per cycle by the caller, which re-runs the cycle with
        careful=True (per-step checks) in that rare case.
        """
        nonlocal v_next, beta_last
        if fused_ok and not careful and cycle_ok:
            _ext.lanczos_cycle_(_ci, _cj, _cv, v, t_mat, _u_buf, _w_buf,
                                _v_next_buf, _beta_buf, _alpha, _norm2,
                                start, ncv)
            v_next = _v_next_buf
            beta_last = _beta_buf.reshape(())
            return
        if fused_ok and not careful:
            for i in range(start, ncv):
                u = op(v[i])
                if not u.is_contiguous():
                    u = u.contiguous()
                if i == start and start > 0:
                    # arrowhead couplings (once per cycle): eager
                    u = torch.addmv(u, v[:start].t(), t_mat[start, :start],
                                    alpha=-1.0)
                    _ext.lanczos_pre_(u, v[i], None, None, _alpha)
                elif i > start:
                    _ext.lanczos_pre_(u, v[i], v[i - 1], t_mat[i, i - 1:i],
                                      _alpha)
                else:
                    _ext.lanczos_pre_(u, v[i], None, None, _alpha)
                _ext.lanczos_sub_alpha_(u, v[i], _alpha, t_mat[i, i:i + 1])
                basis = v[: i + 1]
                for _pass in range(2):   # 2-pass CGS, sub fused into addmv
                    w = basis @ u
                    u = torch.addmv(u, basis.t(), w, alpha=-1.0)
                _ext.lanczos_norm2_(u, _norm2)
                if i + 1 < ncv:
                    _ext.lanczos_normalize_(u, v[i + 1], _norm2,
                                            t_mat[i, i + 1:i + 2],
                                            t_mat[i + 1, i:i + 1], None)
                else:
                    _ext.lanczos_normalize_(u, _v_next_buf, _norm2, None,
                                            None, _beta_buf)
                    v_next = _v_next_buf
                    beta_last = _beta_buf.reshape(())
            return
        for i in range(start, ncv):
            u = op(v[i])
            if i == start and start > 0:
                # subtract arrowhead couplings: u -= sum_j t[start, j] v_j
                u = u - v[:start].t() @ t_mat[start, :start]
            elif i > start:
                u = u - t_mat[i, i - 1] * v[i - 1]
            ai = torch.dot(v[i], u)
            t_mat[i, i] = ai
            u = u - ai * v[i]
            u = _reorth(u, v[: i + 1])
            b = u.norm()
            if careful and float(b) < 1e-30:
                # invariant subspace: fresh random orthogonal direction
                u = uniform((n,), -1.0, 1.0, state=state, device=device, dtype=dtype)
                u = _reorth(u, v[: i + 1])
                b = u.norm()
            if i + 1 < ncv:
                t_mat[i, i + 1] = b
                t_mat[i + 1, i] = b
                v[i + 1] = u / b.clamp_min(1e-300)
            else:
                beta_last = b
                v_next = u / b.clamp_min(1e-300)

    # EXPERIMENTAL hipGraph capture of the restart cycle (opt-in:
    # RAFT_AMD_LANCZOS_GRAPH=1). Measured on MI355X (round 2): capture of
    # the torch-op chain through hipGraph REGRESSED the 10M-row bench
    # (105 vs 153 steps/s — replays tripping the degeneracy redo) and hung
    # outright on a 100k-row case, so the default stays the eager sync-free
    # loop. Kept for future ROCm versions.
    import os
    use_graph = (device.type == "cuda"
                 and os.environ.get("RAFT_AMD_LANCZOS_GRAPH", "0") == "1")
    graph_state = {}

    def _extend_graphed(start: int) -> bool:
        nonlocal v_next, beta_last
        if not use_graph:
            return False
        try:
            if "graph" not in graph_state:
                side = torch.cuda.Stream()
                side.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(side):
                    _extend(start, careful=False)  # allocator warmup
                torch.cuda.current_stream().wait_stream(side)
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g):
                    _extend(start, careful=False)
                # capture does not execute: replay below produces the values.
                # v_next/beta_last now reference capture-owned tensors that
                # every replay refreshes in place.
                graph_state["graph"] = g
                graph_state["outs"] = (v_next, beta_last)
            graph_state["graph"].replay()
            v_next, beta_last = graph_state["outs"]
            return True
        except Exception:
            graph_state["graph"] = None
            return False

    def _extend_checked(start: int):
        """Sync-free extend + one degeneracy check per cycle (rare redo)."""
        if not (start == k and graph_state.get("graph", True) is not None
                and _extend_graphed(start)):
            _extend(start, careful=False)
        betas = torch.diagonal(t_mat, 1)[max(start - 1, 0):]
        if bool((betas.abs() < 1e-30).any()) or bool(beta_last.abs() < 1e-30):
            _extend(start, careful=True)

    _extend_checked(0)
    n_iter = 0
    for n_iter in range(1, cfg.max_iterations + 1):
        w, s = torch.linalg.eigh(t_mat)
        res = (beta_last * s[ncv - 1, :k]).abs()
        if float(res.max()) < cfg.tolerance * max(1.0, float(w[:k].abs().max())):
            break
        # thick restart: V[:k] <- Ritz vectors, arrowhead couplings into row k
        ritz = s[:, :k].t() @ v                    # [k, n]
        v[:k] = ritz
        v[k] = v_next
        t_mat.zero_()
        idx = torch.arange(k, device=device)
        t_mat[idx, idx] = w[:k]
        bk = beta_last * s[ncv - 1, :k]
        t_mat[k, :k] = bk
        t_mat[:k, k] = bk
        _extend_checked(k)
    w, s = torch.linalg.eigh(t_mat)
    eigvecs = (s[:, :k].t() @ v).t().contiguous()   # [n, k]
    return w[:k].clone(), eigvecs


def eigsh(a, k: int = 6, which: str = "SA", ncv: int = 0, maxiter: int = 100,
          tol: float = 1e-9, seed: int = 42, v0: torch.Tensor | None = None):
    """pylibraft.sparse.linalg.eigsh-compatible wrapper (smallest algebraic)."""
    assert which in ("SA",), "only smallest-algebraic supported (reference parity)"
    cfg = LanczosConfig(n_components=k, max_iterations=maxiter, ncv=ncv,
                        tolerance=tol, seed=seed)
    return lanczos_min_eigenpairs(a, k=k, config=cfg, v0=v0)
