from .learner.base_learner import BaseLearner, create_learner, LoadCkptHook, SaveCkptHook, LogShowHook, Hook, build_learner_hook_by_cfg
from .collector.sample_serial_collector import SampleSerialCollector, EpisodeSerialCollector, create_serial_collector, get_serial_collector_cls, ISerialCollector
from .collector.interaction_serial_evaluator import InteractionSerialEvaluator, create_serial_evaluator
from .coordinator.base_serial_commander import BaseSerialCommander
from .replay_buffer.naive_buffer import NaiveReplayBuffer, AdvancedReplayBuffer, EpisodeReplayBuffer, create_buffer, get_buffer_cls, IBuffer
from .coordinator.coordinator import Coordinator, ResourceManager
from .adapter.learner_aggregator import LearnerAggregator
from .collector.battle_collector import BattleSampleSerialCollector, BattleEpisodeSerialCollector
from .collector.metric_serial_evaluator import MetricSerialEvaluator, IMetric
from .coordinator.parallel_commander import BaseCommander, NaiveCommander, SoloCommander, OneVsOneCommander, create_parallel_commander
