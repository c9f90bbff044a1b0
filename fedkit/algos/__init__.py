from .strategies import (
    Strategy, NoConsensus, FedAvg, FedProx, ConsensusADMM, BBConfig,
    STRATEGIES,
)

__all__ = ["Strategy", "NoConsensus", "FedAvg", "FedProx", "ConsensusADMM",
           "BBConfig", "STRATEGIES"]
