import os, sys
sys.path.insert(0, os.getcwd())
import cProfile, pstats, warnings
import numpy as np
warnings.simplefilter("ignore")
import optuna_amd
optuna_amd.logging.set_verbosity(optuna_amd.logging.WARNING)

D = 100
sampler = optuna_amd.samplers.CmaEsSampler(seed=0, n_startup_trials=1)
study = optuna_amd.create_study(
    sampler=sampler,
    pruner=optuna_amd.pruners.HyperbandPruner(min_resource=1, max_resource=8, reduction_factor=2),
)
rng = np.random.RandomState(0)

def one_step():
    t = study.ask()
    x = np.array([t.suggest_float(f"x{i}", -5, 5) for i in range(D)])
    v = float(np.sum(x ** 2))
    for step in range(8):
        t.report(v * (1 + 1/(step+1)), step)
        if t.should_prune():
            study.tell(t, state=optuna_amd.trial.TrialState.PRUNED)
            return
    study.tell(t, v)

for _ in range(60):
    one_step()
pr = cProfile.Profile(); pr.enable()
for _ in range(60):
    one_step()
pr.disable()
pstats.Stats(pr).sort_stats("cumulative").print_stats(22)
