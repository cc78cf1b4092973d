"""Phase-level timing of one GPSampler suggest at N obs (run on the GPU box)."""
import os, sys
sys.path.insert(0, os.getcwd())
import time, warnings, sys
import numpy as np

warnings.simplefilter("ignore")
import optuna_amd
from optuna_amd._gp import gp as gp_mod, prior, optim_mixed
from optuna_amd._gp import search_space as gp_ss
from optuna_amd._gp import acqf as acqf_mod
from optuna_amd.distributions import FloatDistribution

N = int(sys.argv[1]) if len(sys.argv) > 1 else 5000
D = 20
rng = np.random.RandomState(0)
X = rng.rand(N, D)
Y = np.sum((X - 0.4) ** 2, axis=1)
Y = (Y - Y.mean()) / Y.std()
is_cat = np.zeros(D, dtype=bool)

import torch
print("cuda:", torch.cuda.is_available())

# count L-BFGS iterations in the fit
n_evals = [0]
orig = gp_mod.GPRegressor._loss_and_grad_closed_form_torch
def counted(self, *a, **k):
    n_evals[0] += 1
    return orig(self, *a, **k)
gp_mod.GPRegressor._loss_and_grad_closed_form_torch = counted

gpr_cache = None
for rep in range(3):
    n_evals[0] = 0
    t0 = time.perf_counter()
    gpr = gp_mod.fit_kernel_params(X, Y, is_cat, prior.default_log_prior, 1e-6, False, gpr_cache=gpr_cache)
    torch.cuda.synchronize()
    t1 = time.perf_counter()
    gpr_cache = gpr
    print(f"fit[{rep}]: {(t1-t0)*1e3:.1f} ms, {n_evals[0]} loss evals, device={gpr.device}")

space = gp_ss.SearchSpace({f"x{i}": FloatDistribution(0.0, 1.0) for i in range(D)})
acqf = acqf_mod.LogEI(gpr=gpr, search_space=space, threshold=float(Y.max()))

for rep in range(2):
    t0 = time.perf_counter()
    xs = space.sample_normalized_params(2048, rng=rng)
    t1 = time.perf_counter()
    fv = acqf.eval_acqf_no_grad(xs)
    torch.cuda.synchronize()
    t2 = time.perf_counter()
    print(f"presample: sobol {(t1-t0)*1e3:.1f} ms, eval2048 {(t2-t1)*1e3:.1f} ms")

# count batched eval calls in local search
n_calls = [0, 0]
orig_ng = acqf_mod.BaseAcquisitionFunc.eval_acqf_no_grad
orig_bg = acqf_mod.BaseAcquisitionFunc.eval_acqf_batched_with_grad
def spy_ng(self, x):
    n_calls[0] += 1
    return orig_ng(self, x)
def spy_bg(self, x):
    n_calls[1] += 1
    return orig_bg(self, x)
acqf_mod.BaseAcquisitionFunc.eval_acqf_no_grad = spy_ng
acqf_mod.BaseAcquisitionFunc.eval_acqf_batched_with_grad = spy_bg

t0 = time.perf_counter()
x_opt, f_opt = optim_mixed.optimize_acqf_mixed(acqf, rng=np.random.RandomState(1))
torch.cuda.synchronize()
t1 = time.perf_counter()
print(f"optimize_acqf_mixed: {(t1-t0)*1e3:.1f} ms, no_grad calls={n_calls[0]}, grad calls={n_calls[1]}")
