from sparse_coding_amd.config import (  # noqa: F401
    BaseArgs, TrainArgs, EnsembleArgs, SyntheticEnsembleArgs, ErasureArgs,
    ToyArgs, InterpArgs, InterpGraphArgs, InvestigateArgs,
)
