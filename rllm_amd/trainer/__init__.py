from rllm_amd.trainer.agent_trainer import AgentTrainer
from rllm_amd.trainer.unified_trainer import TrainerConfig, TrainerState, UnifiedTrainer

__all__ = ["AgentTrainer", "TrainerConfig", "TrainerState", "UnifiedTrainer"]
