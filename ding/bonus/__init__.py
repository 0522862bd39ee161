from .agents import (
    BaseAgent, DQNAgent, PPOF, PPOOffPolicyAgent, A2CAgent, C51Agent, SACAgent, DDPGAgent, TD3Agent, SQLAgent,
    PGAgent, TrainingReturn, EvalReturn,
)
