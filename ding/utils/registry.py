"""Name -> class registries.

Parity: reference ding/utils/registry.py:11 (Registry) and
ding/utils/registry_factory.py:3-24 (the 22 global instances).
"""
import inspect
from typing import Any, Callable, Dict, Iterable, Optional


class Registry:
    """A string-keyed factory table with decorator registration."""

    def __init__(self, name: str):
        self._name = name
        self._table: Dict[str, Any] = {}
        self._aliases: Dict[str, str] = {}

    @property
    def name(self) -> str:
        return self._name

    def register(self, name: Optional[str] = None, force_overwrite: bool = False) -> Callable:

        def _do(cls):
            key = name if name is not None else cls.__name__
            if key in self._table and not force_overwrite:
                existing = self._table[key]
                if existing is not cls:
                    raise KeyError(f"duplicate key '{key}' in registry '{self._name}'")
            self._table[key] = cls
            return cls

        # allow bare usage: @REG.register
        if inspect.isclass(name) or inspect.isfunction(name):
            cls, name = name, None
            return _do(cls)
        return _do

    def get(self, key: str) -> Any:
        key = self._aliases.get(key, key)
        if key not in self._table:
            raise KeyError(
                f"'{key}' not found in registry '{self._name}'. Known: {sorted(self._table)}"
            )
        return self._table[key]

    def alias(self, key: str, alias: str) -> None:
        self._aliases[alias] = key

    def build(self, key: str, *args, **kwargs) -> Any:
        return self.get(key)(*args, **kwargs)

    def __contains__(self, key: str) -> bool:
        return key in self._table or key in self._aliases

    def keys(self) -> Iterable[str]:
        return self._table.keys()

    def items(self):
        return self._table.items()

    def query_details(self) -> Dict[str, str]:
        return {k: f"{v.__module__}.{v.__qualname__}" for k, v in self._table.items()}


# Global registries (reference registry_factory.py:3-24)
POLICY_REGISTRY = Registry("policy")
ENV_REGISTRY = Registry("env")
ENV_MANAGER_REGISTRY = Registry("env_manager")
ENV_WRAPPER_REGISTRY = Registry("env_wrapper")
MODEL_REGISTRY = Registry("model")
REWARD_MODEL_REGISTRY = Registry("reward_model")
WORLD_MODEL_REGISTRY = Registry("world_model")
BUFFER_REGISTRY = Registry("buffer")
DATASET_REGISTRY = Registry("dataset")
SERIAL_COLLECTOR_REGISTRY = Registry("serial_collector")
SERIAL_EVALUATOR_REGISTRY = Registry("serial_evaluator")
PARALLEL_COLLECTOR_REGISTRY = Registry("parallel_collector")
LEARNER_REGISTRY = Registry("learner")
COMM_LEARNER_REGISTRY = Registry("comm_learner")
COMM_COLLECTOR_REGISTRY = Registry("comm_collector")
COMMANDER_REGISTRY = Registry("commander")
LEAGUE_REGISTRY = Registry("league")
PLAYER_REGISTRY = Registry("player")
MQ_REGISTRY = Registry("message_queue")
STOP_VALUE_REGISTRY = Registry("stop_value")
AGENT_REGISTRY = Registry("agent")
HOOK_REGISTRY = Registry("hook")
STOCHASTIC_OPTIMIZER_REGISTRY = Registry("stochastic_optimizer")

REGISTRIES = {
    r.name: r
    for r in (
        POLICY_REGISTRY, ENV_REGISTRY, ENV_MANAGER_REGISTRY, ENV_WRAPPER_REGISTRY, MODEL_REGISTRY,
        REWARD_MODEL_REGISTRY, WORLD_MODEL_REGISTRY, BUFFER_REGISTRY, DATASET_REGISTRY,
        SERIAL_COLLECTOR_REGISTRY, SERIAL_EVALUATOR_REGISTRY, PARALLEL_COLLECTOR_REGISTRY,
        LEARNER_REGISTRY, COMM_LEARNER_REGISTRY, COMM_COLLECTOR_REGISTRY, COMMANDER_REGISTRY,
        LEAGUE_REGISTRY, PLAYER_REGISTRY, MQ_REGISTRY, STOP_VALUE_REGISTRY, AGENT_REGISTRY,
        HOOK_REGISTRY,
        STOCHASTIC_OPTIMIZER_REGISTRY,
    )
}
