from distributed_reinforcement_learning_amd.models.actor_critic import ActorCritic, VectorActorCritic
from distributed_reinforcement_learning_amd.models.impala_actor_critic import ImpalaActorCritic
from distributed_reinforcement_learning_amd.models.apex_value import ApexDuelingQ, VectorDuelingQ
from distributed_reinforcement_learning_amd.models.r2d2_lstm import R2D2LstmQ
from distributed_reinforcement_learning_amd.models.impala_resnet import ImpalaResNetActorCritic

__all__ = [
    "ActorCritic", "VectorActorCritic", "ImpalaActorCritic",
    "ApexDuelingQ", "VectorDuelingQ", "R2D2LstmQ",
    "ImpalaResNetActorCritic",
]
