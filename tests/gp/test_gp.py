"""GP regressor numerics + fit behavior."""
from __future__ import annotations

import numpy as np
import pytest
import torch

from optuna_amd._gp import gp, prior


def _make_gpr(X: np.ndarray, y: np.ndarray, fit: bool = True) -> gp.GPRegressor:
    if fit:
        return gp.fit_kernel_params(
            X=X,
            Y=y,
            is_categorical=np.zeros(X.shape[1], dtype=bool),
            log_prior=prior.default_log_prior,
            minimum_noise=prior.DEFAULT_MINIMUM_NOISE_VAR,
            deterministic_objective=False,
        )
    gpr = gp.GPRegressor(
        is_categorical=torch.zeros(X.shape[1], dtype=torch.bool),
        X_train=torch.from_numpy(X),
        y_train=torch.from_numpy(y),
        inverse_squared_lengthscales=torch.ones(X.shape[1], dtype=torch.float64),
        kernel_scale=torch.tensor(1.0, dtype=torch.float64),
        noise_var=torch.tensor(1e-4, dtype=torch.float64),
    )
    gpr._cache_matrix()
    return gpr


def test_matern52_matches_closed_form() -> None:
    d2 = torch.linspace(0, 9, 100, dtype=torch.float64)
    ours = gp.matern52_of_sqdist(d2)
    s = torch.sqrt(5 * d2)
    ref = (1 + s + s**2 / 3) * torch.exp(-s)
    torch.testing.assert_close(ours, ref, rtol=1e-12, atol=1e-12)


def test_matern52_gradient_finite_at_zero() -> None:
    d2 = torch.tensor([0.0, 1.0], dtype=torch.float64, requires_grad=True)
    out = gp.matern52_of_sqdist(d2).sum()
    out.backward()
    assert torch.isfinite(d2.grad).all()
    assert d2.grad[0] == pytest.approx(-5 / 6)


def test_posterior_interpolates_training_data() -> None:
    rng = np.random.RandomState(0)
    X = rng.rand(30, 3)
    y = np.sin(X.sum(axis=1) * 3)
    gpr = _make_gpr(X, y)
    mean, var = gpr.posterior(torch.from_numpy(X))
    np.testing.assert_allclose(mean.numpy(), y, atol=0.05)
    assert np.all(var.numpy() >= 0)


def test_posterior_uncertainty_grows_away_from_data() -> None:
    X = np.array([[0.5, 0.5]])
    y = np.array([0.0])
    gpr = _make_gpr(X, y, fit=False)
    _, var_near = gpr.posterior(torch.tensor([0.5, 0.5], dtype=torch.float64))
    _, var_far = gpr.posterior(torch.tensor([3.0, 3.0], dtype=torch.float64))
    assert var_far > var_near


def test_mll_closed_form_small_case() -> None:
    X = np.array([[0.0], [1.0]])
    y = np.array([0.3, -0.2])
    gpr = _make_gpr(X, y, fit=False)
    mll = gpr.marginal_log_likelihood().item()
    # Direct dense computation.
    K = gpr.kernel().detach().numpy() + 1e-4 * np.eye(2)
    expected = -0.5 * np.log(np.linalg.det(K)) - 0.5 * y @ np.linalg.solve(K, y)
    assert mll == pytest.approx(expected, rel=1e-9)


def test_extend_cholesky_equals_full_factorization() -> None:
    rng = np.random.RandomState(1)
    A = rng.rand(6, 6)
    K = A @ A.T + np.eye(6)
    K11 = torch.from_numpy(K[:4, :4])
    K21 = torch.from_numpy(K[4:, :4])
    K22 = torch.from_numpy(K[4:, 4:])
    L11 = torch.linalg.cholesky(K11)
    L = gp._extend_cholesky(L11, K21, K22)
    torch.testing.assert_close(L @ L.T, torch.from_numpy(K), rtol=1e-10, atol=1e-10)


def test_fit_improves_mll() -> None:
    rng = np.random.RandomState(2)
    X = rng.rand(40, 2)
    y = np.sin(5 * X[:, 0]) + 0.1 * rng.randn(40)
    y = (y - y.mean()) / y.std()
    fitted = _make_gpr(X, y)
    default = _make_gpr(X, y, fit=False)
    # Need comparable objects: compute MLL with each's params on same data.
    assert fitted.marginal_log_likelihood().item() >= default.marginal_log_likelihood().item() - 1e-6


def test_warn_and_convert_inf() -> None:
    vals = np.array([[1.0], [np.inf], [-np.inf], [2.0]])
    with pytest.warns(UserWarning):
        out = gp.warn_and_convert_inf(vals)
    assert out.max() == 2.0 and out.min() == 1.0


def test_numpy_loss_grad_matches_torch_autograd() -> None:
    """The closed-form fit loss/grad must equal the torch-autograd path."""
    import torch

    from optuna_amd._gp import gp as gp_mod
    from optuna_amd._gp import prior

    rng = np.random.RandomState(0)
    n, d = 40, 5
    X = rng.rand(n, d)
    Y = rng.randn(n)
    is_cat = np.zeros(d, dtype=bool)
    gpr = gp_mod.GPRegressor(
        is_categorical=torch.from_numpy(is_cat),
        X_train=torch.from_numpy(X),
        y_train=torch.from_numpy(Y),
        inverse_squared_lengthscales=torch.ones(d, dtype=torch.float64),
        kernel_scale=torch.tensor(1.0, dtype=torch.float64),
        noise_var=torch.tensor(1.0, dtype=torch.float64),
    )
    minimum_noise = 1e-6
    sqd = gpr._squared_X_diff.numpy()

    for det in (False, True):
        for trial_i in range(5):
            raw = rng.randn(d + 2) * 0.7
            loss_np, grad_np = gpr._loss_and_grad_numpy(raw, sqd, Y, minimum_noise, det)

            raw_t = torch.from_numpy(raw).requires_grad_(True)
            with torch.enable_grad():
                gpr.inverse_squared_lengthscales = torch.exp(raw_t[:d])
                gpr.kernel_scale = torch.exp(raw_t[d])
                gpr.noise_var = (
                    torch.tensor(minimum_noise, dtype=torch.float64)
                    if det
                    else torch.exp(raw_t[d + 1]) + minimum_noise
                )
                loss_t = -gpr.marginal_log_likelihood() - prior.default_log_prior(gpr)
                loss_t.backward()
            np.testing.assert_allclose(loss_np, loss_t.item(), rtol=1e-10)
            np.testing.assert_allclose(
                grad_np, raw_t.grad.detach().numpy(), rtol=1e-8, atol=1e-10
            )


def test_ard_sqdist_gemm_matches_broadcast() -> None:
    rng = np.random.RandomState(3)
    X1 = torch.from_numpy(rng.rand(37, 6))
    X2 = torch.from_numpy(rng.rand(53, 6))
    eta = torch.from_numpy(np.exp(rng.randn(6)))
    got = gp._ard_sqdist_gemm(X1, X2, eta)
    want = ((X1.unsqueeze(-2) - X2.unsqueeze(-3)).square() * eta).sum(-1)
    torch.testing.assert_close(got, want, rtol=1e-9, atol=1e-10)
    # coincident points: exactly clamped at 0, never negative
    same = gp._ard_sqdist_gemm(X1, X1.clone(), eta)
    assert (same.diagonal() >= 0).all()


def test_closed_form_torch_loss_matches_numpy() -> None:
    rng = np.random.RandomState(7)
    N, D = 64, 4
    X = rng.rand(N, D)
    y = rng.randn(N)
    gpr = _make_gpr(X, y, fit=False)
    raw = rng.randn(D + 2) * 0.3
    sqd = (X[:, None, :] - X[None, :, :]) ** 2
    loss_np, grad_np = gpr._loss_and_grad_numpy(raw, sqd, y, 1e-6, False)
    loss_t, grad_t = gpr._loss_and_grad_closed_form_torch(
        raw, torch.from_numpy(X), torch.from_numpy(y), 1e-6, False
    )
    assert loss_t == pytest.approx(loss_np, rel=1e-8)
    np.testing.assert_allclose(grad_t, grad_np, rtol=1e-6, atol=1e-8)


def test_posterior_with_explicit_inverse_matches_solves() -> None:
    rng = np.random.RandomState(11)
    X = rng.rand(40, 3)
    y = rng.randn(40)
    gpr = _make_gpr(X, y, fit=False)
    x_eval = torch.from_numpy(rng.rand(9, 3))
    mean_ref, var_ref = gpr.posterior(x_eval)
    gpr._cov_Y_Y_inv = torch.cholesky_inverse(gpr._cov_Y_Y_chol)
    mean_inv, var_inv = gpr.posterior(x_eval)
    torch.testing.assert_close(mean_inv, mean_ref, rtol=1e-8, atol=1e-10)
    torch.testing.assert_close(var_inv, var_ref, rtol=1e-6, atol=1e-9)


def test_large_history_skips_dense_sqdiff_and_still_fits() -> None:
    # Above _MAX_DENSE_SQDIFF_OBS (no categorical, no cuda) the regressor must
    # fit through the GEMM identity without the (N, N, D) tensor.
    rng = np.random.RandomState(13)
    N = gp.GPRegressor._MAX_DENSE_SQDIFF_OBS + 8
    X = rng.rand(N, 2)
    y = np.sin(3 * X[:, 0]) + rng.randn(N) * 0.1
    gpr = gp.fit_kernel_params(
        X=X,
        Y=y,
        is_categorical=np.zeros(2, dtype=bool),
        log_prior=prior.default_log_prior,
        minimum_noise=prior.DEFAULT_MINIMUM_NOISE_VAR,
        deterministic_objective=False,
    )
    assert gpr._squared_X_diff is None
    mean, var = gpr.posterior(torch.from_numpy(X[:5]).to(gpr.device))
    assert torch.isfinite(mean).all() and (var >= 0).all()


def test_update_data_matches_fresh_cache() -> None:
    """Incremental extend (new rows + re-standardized targets) must reproduce
    the from-scratch covariance state for the same hyperparameters."""
    rng = np.random.RandomState(17)
    N0, D = gp.GPRegressor._MAX_DENSE_SQDIFF_OBS + 4, 3
    X0 = rng.rand(N0, D)
    y0 = rng.randn(N0)
    gpr = gp.fit_kernel_params(
        X=X0,
        Y=y0,
        is_categorical=np.zeros(D, dtype=bool),
        log_prior=prior.default_log_prior,
        minimum_noise=prior.DEFAULT_MINIMUM_NOISE_VAR,
        deterministic_objective=False,
    )
    X_full = np.vstack([X0, rng.rand(7, D)])
    y_full = np.concatenate([y0, rng.randn(7)]) * 1.01 + 0.003  # drifted targets
    assert gpr.update_data(X_full, y_full)
    assert gpr._X_train.shape[0] == N0 + 7

    dev = gpr.device  # cuda on a GPU box, cpu otherwise
    ref = gp.GPRegressor(
        is_categorical=torch.zeros(D, dtype=torch.bool).to(dev),
        X_train=torch.from_numpy(X_full).to(dev),
        y_train=torch.from_numpy(y_full).to(dev),
        inverse_squared_lengthscales=gpr.inverse_squared_lengthscales.clone(),
        kernel_scale=gpr.kernel_scale.clone(),
        noise_var=gpr.noise_var.clone(),
    )
    ref._cache_matrix()
    x_eval = torch.from_numpy(rng.rand(11, D)).to(dev)
    mean_u, var_u = gpr.posterior(x_eval)
    mean_r, var_r = ref.posterior(x_eval)
    torch.testing.assert_close(mean_u, mean_r, rtol=1e-7, atol=1e-9)
    torch.testing.assert_close(var_u, var_r, rtol=1e-5, atol=1e-9)


def test_update_data_rejects_non_extension() -> None:
    rng = np.random.RandomState(19)
    N0, D = gp.GPRegressor._MAX_DENSE_SQDIFF_OBS + 4, 3
    X0 = rng.rand(N0, D)
    gpr = gp.fit_kernel_params(
        X=X0,
        Y=rng.randn(N0),
        is_categorical=np.zeros(D, dtype=bool),
        log_prior=prior.default_log_prior,
        minimum_noise=prior.DEFAULT_MINIMUM_NOISE_VAR,
        deterministic_objective=False,
    )
    X_bad = X0.copy()
    X_bad[5, 0] += 0.5  # mutated prefix — must force a refit
    assert not gpr.update_data(X_bad, rng.randn(N0))
    assert not gpr.update_data(X0[:-1], rng.randn(N0 - 1))  # shrunk


def test_cloned_with_running_leaves_cache_pristine() -> None:
    rng = np.random.RandomState(23)
    X = rng.rand(50, 3)
    y = rng.randn(50)
    gpr = _make_gpr(X, y, fit=False)
    chol_before = gpr._cov_Y_Y_chol.clone()
    X_run = torch.from_numpy(rng.rand(4, 3))
    clone = gpr.cloned_with_running(X_run, gpr.posterior(X_run)[0])
    assert clone is not gpr
    assert clone._X_all.shape[0] == 54
    assert gpr._X_all.shape[0] == 50
    torch.testing.assert_close(gpr._cov_Y_Y_chol, chol_before)


def test_sobol_uniform_matches_scipy_unscrambled_structure() -> None:
    """Our torch Sobol reproduces scipy's unscrambled sequence when the
    digital shift is removed (same Joe-Kuo direction numbers, Gray order)."""
    from scipy.stats import qmc as scipy_qmc

    from optuna_amd._gp import qmc

    dim, n = 5, 64
    u = qmc.sobol_uniform(dim, n, seed=0)
    shift = np.random.RandomState(0).randint(0, 1 << qmc._BITS, size=dim).astype(np.int64)
    x_int = (u.numpy() * (1 << qmc._BITS) - 0.5).round().astype(np.int64)
    unscrambled = (x_int ^ shift).astype(np.float64) / (1 << qmc._BITS)
    ref = scipy_qmc.Sobol(dim, scramble=False, bits=qmc._BITS).random(n)
    np.testing.assert_allclose(unscrambled, ref, atol=1e-9)


def test_sobol_normal_moments_and_determinism() -> None:
    from optuna_amd._gp import qmc

    a = qmc.sample_from_normal_sobol(8, 1024, seed=3)
    b = qmc.sample_from_normal_sobol(8, 1024, seed=3)
    torch.testing.assert_close(a, b)
    c = qmc.sample_from_normal_sobol(8, 1024, seed=4)
    assert not torch.equal(a, c)
    assert abs(a.mean().item()) < 0.02
    assert abs(a.std().item() - 1.0) < 0.02
    assert torch.isfinite(a).all()
