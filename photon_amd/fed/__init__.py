from .client import ClientState, FedClient
from .flat import FlatParams, trainable_param_names
from .runtime import Comm, assign_clients_to_ranks, init_distributed, sample_clients
from .server import FedServer, TooManyFailuresError, weighted_loss_avg
from .strategies import (
    FedAdam,
    FedAvgEfficient,
    FedMom,
    FedNesterov,
    FedYogi,
    Strategy,
    dispatch_strategy,
)

__all__ = [
    "ClientState",
    "FedClient",
    "FlatParams",
    "trainable_param_names",
    "Comm",
    "assign_clients_to_ranks",
    "init_distributed",
    "sample_clients",
    "FedServer",
    "TooManyFailuresError",
    "weighted_loss_avg",
    "Strategy",
    "dispatch_strategy",
    "FedAvgEfficient",
    "FedNesterov",
    "FedMom",
    "FedAdam",
    "FedYogi",
]
