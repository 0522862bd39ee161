from .collector import TransitionList, inferencer, rolloutor
from .data_processor import (
    data_pusher, offpolicy_data_fetcher, offline_data_fetcher, offline_data_fetcher_from_mem, buffer_saver,
    sqil_data_pusher,
)
from .trainer import trainer, multistep_trainer
from .evaluator import interaction_evaluator, interaction_evaluator_ttorch, metric_evaluator, VectorEvalMonitor
from .explorer import eps_greedy_handler, eps_greedy_masker
from .advantage_estimator import gae_estimator, ppof_adv_estimator, montecarlo_return_estimator
from .enhancer import reward_estimator, her_data_enhancer, nstep_reward_enhancer
from .termination_checker import termination_checker, ddp_termination_checker, epoch_timer, final_ctx_saver
from .logger import online_logger, offline_logger, wandb_online_logger, wandb_offline_logger
from .priority import priority_calculator
from .data_processor import buffer_saver as offline_data_saver  # reference-name alias
from .data_processor import offline_data_fetcher_from_mem as OfflineMemoryDataFetcher  # reference-name alias
