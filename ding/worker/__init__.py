from .learner.base_learner import BaseLearner, create_learner, LoadCkptHook, SaveCkptHook, LogShowHook, Hook, build_learner_hook_by_cfg
from .collector.sample_serial_collector import SampleSerialCollector, EpisodeSerialCollector, create_serial_collector, get_serial_collector_cls, ISerialCollector
from .collector.interaction_serial_evaluator import InteractionSerialEvaluator, create_serial_evaluator
from .coordinator.base_serial_commander import BaseSerialCommander
from .replay_buffer.naive_buffer import NaiveReplayBuffer, AdvancedReplayBuffer, EpisodeReplayBuffer, create_buffer, get_buffer_cls, IBuffer
from .coordinator.coordinator import Coordinator, ResourceManager
from .adapter.learner_aggregator import LearnerAggregator
from .collector.battle_collector import BattleSampleSerialCollector, BattleEpisodeSerialCollector, BattleInteractionSerialEvaluator
from .collector.metric_serial_evaluator import MetricSerialEvaluator, IMetric
from .coordinator.parallel_commander import BaseCommander, NaiveCommander, SoloCommander, OneVsOneCommander, create_parallel_commander
from .comm import FlaskFileSystemLearner, FlaskFileSystemCollector
from .legacy_parity import (
    LearnerHook, register_learner_hook, add_learner_hook, merge_hooks, BaseCommLearner, BaseCommCollector,
    create_comm_learner, create_comm_collector, VectorEvalMonitor, to_tensor_transitions, SequenceReplayBuffer,
    BaseParallelCollector, ZerglingParallelCollector, MarineParallelCollector, NaiveCollector,
    create_parallel_collector, get_parallel_collector_cls, get_parallel_commander_cls,
)
from .collector.interaction_serial_evaluator import InteractionSerialEvaluator as ISerialEvaluator
