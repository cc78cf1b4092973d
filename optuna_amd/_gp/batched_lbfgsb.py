"""Batch of independent L-BFGS-B chains with batched function evaluations.

The reference multiplexes scipy's optimizer through greenlets so that B chains'
function evaluations coalesce into one batched call
(reference ``optuna/_gp/batched_lbfgsb.py`` :34-86, sequential fallback :141-166).
This build uses greenlet multiplexing when greenlet is importable and falls back
to a sequential loop otherwise. On the MI355X roadmap the batched evaluation runs
the K5 acqf kernel over all chains at once, which is what makes the multiplexing
pay off.
"""
from __future__ import annotations

from typing import Any, Callable

import numpy as np

from optuna_amd._imports import try_import


with try_import() as _greenlet_imports:
    import greenlet


FuncAndGrad = Callable[..., tuple[np.ndarray, np.ndarray]]


def _lbfgsb_once(
    func_and_grad: FuncAndGrad,
    x0: np.ndarray,
    args: tuple[Any, ...],
    bounds: list[tuple[float, float]],
    pgtol: float,
    max_iters: int,
) -> tuple[np.ndarray, float, int]:
    import scipy.optimize

    n_evals = 0

    def f(x: np.ndarray) -> tuple[float, np.ndarray]:
        nonlocal n_evals
        n_evals += 1
        fvals, grads = func_and_grad(x[np.newaxis, :], *[[a] for a in args])
        return float(fvals[0]), grads[0]

    x_opt, fval, info = scipy.optimize.fmin_l_bfgs_b(
        f, x0, bounds=bounds, pgtol=pgtol, maxiter=max_iters
    )
    return x_opt, float(fval), int(info["nit"])


def _batched_lbfgsb_greenlet(
    func_and_grad: FuncAndGrad,
    x0_batched: np.ndarray,
    batched_args: tuple[list[Any], ...],
    bounds: list[tuple[float, float]],
    pgtol: float,
    max_iters: int,
) -> tuple[np.ndarray, np.ndarray, np.ndarray]:
    import scipy.optimize

    B = len(x0_batched)
    results_x = x0_batched.copy()
    results_f = np.empty(B)
    results_it = np.zeros(B, dtype=int)

    # Each chain runs scipy's optimizer inside a greenlet; whenever it needs an
    # objective value it switches back here, and we evaluate all pending points
    # in ONE batched call.
    def make_runner(i: int) -> "greenlet.greenlet":
        def run(*_start_args: Any) -> None:  # first switch passes the start message
            def f(x: np.ndarray) -> tuple[float, np.ndarray]:
                fval, grad = greenlet.getcurrent().parent.switch(("eval", i, x))
                return fval, grad

            x_opt, fval, info = scipy.optimize.fmin_l_bfgs_b(
                f, x0_batched[i], bounds=bounds, pgtol=pgtol, maxiter=max_iters
            )
            greenlet.getcurrent().parent.switch(("done", i, (x_opt, fval, int(info["nit"]))))

        return greenlet.greenlet(run)

    runners = [make_runner(i) for i in range(B)]
    pending: list[tuple[int, np.ndarray]] = []
    replies: dict[int, tuple[float, np.ndarray]] = {}
    active = set(range(B))

    # Round-robin: step every active chain to its next eval request, batch-evaluate,
    # then feed the results back.
    requests: dict[int, np.ndarray] = {}
    messages: dict[int, Any] = {i: None for i in range(B)}
    while active:
        requests.clear()
        for i in sorted(active):
            msg = runners[i].switch(messages[i])
            kind, idx, payload = msg
            if kind == "done":
                x_opt, fval, nit = payload
                results_x[idx] = x_opt
                results_f[idx] = fval
                results_it[idx] = nit
                active.discard(idx)
            else:
                requests[idx] = payload
        if requests:
            idxs = sorted(requests)
            xs = np.stack([requests[i] for i in idxs])
            args_for_batch = tuple(
                [col[i] for i in idxs] for col in batched_args
            )
            fvals, grads = func_and_grad(xs, *args_for_batch)
            for row, i in enumerate(idxs):
                messages[i] = (float(fvals[row]), grads[row])
    return results_x, results_f, results_it


def batched_lbfgsb(
    func_and_grad: FuncAndGrad,
    x0_batched: np.ndarray,
    batched_args: tuple[list[Any], ...] = (),
    bounds: list[tuple[float, float]] | None = None,
    pgtol: float = 1e-5,
    max_iters: int = 200,
) -> tuple[np.ndarray, np.ndarray, np.ndarray]:
    """Minimize B independent problems; returns (x_opt (B,d), f_opt (B,), n_iter (B,))."""
    assert x0_batched.ndim == 2
    bounds = bounds or [(-np.inf, np.inf)] * x0_batched.shape[1]

    if _greenlet_imports.is_successful():
        return _batched_lbfgsb_greenlet(
            func_and_grad, x0_batched, batched_args, bounds, pgtol, max_iters
        )

    B = len(x0_batched)
    xs = np.empty_like(x0_batched)
    fs = np.empty(B)
    its = np.empty(B, dtype=int)
    for i in range(B):
        args_i = tuple(col[i] for col in batched_args)
        xs[i], fs[i], its[i] = _lbfgsb_once(
            func_and_grad, x0_batched[i], args_i, bounds, pgtol, max_iters
        )
    return xs, fs, its
