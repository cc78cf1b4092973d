from optuna_amd.trial._base import BaseTrial
from optuna_amd.trial._fixed import FixedTrial
from optuna_amd.trial._frozen import FrozenTrial, create_trial
from optuna_amd.trial._state import TrialState
from optuna_amd.trial._trial import Trial


__all__ = ["BaseTrial", "FixedTrial", "FrozenTrial", "Trial", "TrialState", "create_trial"]
