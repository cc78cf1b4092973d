from optuna_amd.samplers.nsgaii._crossovers import (
    BaseCrossover,
    BLXAlphaCrossover,
    SBXCrossover,
    SPXCrossover,
    UNDXCrossover,
    UniformCrossover,
    VSBXCrossover,
)
from optuna_amd.samplers.nsgaii._mutations import BaseMutation, PolynomialMutation
from optuna_amd.samplers.nsgaii._sampler import NSGAIISampler
from optuna_amd.samplers.nsgaii._strategies import (
    NSGAIIAfterTrialStrategy,
    NSGAIIChildGenerationStrategy,
    NSGAIIElitePopulationSelectionStrategy,
)


__all__ = [
    "BaseCrossover",
    "BaseMutation",
    "BLXAlphaCrossover",
    "NSGAIIAfterTrialStrategy",
    "NSGAIIChildGenerationStrategy",
    "NSGAIIElitePopulationSelectionStrategy",
    "NSGAIISampler",
    "PolynomialMutation",
    "SBXCrossover",
    "SPXCrossover",
    "UNDXCrossover",
    "UniformCrossover",
    "VSBXCrossover",
]
