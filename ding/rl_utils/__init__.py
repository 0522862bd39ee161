from .gae import gae, gae_data
from .td import (
    q_1step_td_data, q_1step_td_error, m_q_1step_td_data, m_q_1step_td_error, q_v_1step_td_data, q_v_1step_td_error,
    nstep_return_data, nstep_return, dist_1step_td_data, dist_1step_td_error, dist_nstep_td_data, dist_nstep_td_error,
    v_1step_td_data, v_1step_td_error, v_nstep_td_data, v_nstep_td_error, q_nstep_td_data, q_nstep_td_error,
    q_nstep_td_error_with_rescale, bdq_nstep_td_error, dqfd_nstep_td_data, dqfd_nstep_td_error,
    dqfd_nstep_td_error_with_rescale, qrdqn_nstep_td_data, qrdqn_nstep_td_error, q_nstep_sql_td_error,
    iqn_nstep_td_data, iqn_nstep_td_error, fqf_nstep_td_data, fqf_nstep_td_error, fqf_calculate_fraction_loss,
    evaluate_quantile_at_action, td_lambda_data, td_lambda_error, generalized_lambda_returns, multistep_forward_view,
    discount_cumsum, view_similar,
)
from .ppo import (
    ppo_data, ppo_data_continuous, ppo_policy_data, ppo_policy_data_continuous, ppo_value_data, ppo_loss,
    ppo_policy_loss, ppo_info, ppo_error, ppo_policy_error, ppo_value_error, ppo_error_continuous,
    ppo_policy_error_continuous, calculate_kl_div,
)
from .vtrace import (
    vtrace_data, vtrace_loss, vtrace_error_discrete_action, vtrace_error_continuous_action, vtrace_nstep_return,
    vtrace_advantage,
)
from .upgo import upgo_loss, upgo_returns, tb_cross_entropy
from .retrace import compute_q_retraces
from .a2c import a2c_data, a2c_loss, a2c_error, a2c_error_continuous
from .coma import coma_data, coma_loss, coma_error
from .acer import acer_policy_error, acer_value_error, acer_trust_region_update
from .ppg import ppg_data, ppg_joint_loss, ppg_joint_error
from .happo import (
    happo_data, happo_policy_data, happo_value_data, happo_loss, happo_policy_loss, happo_info, happo_error,
    happo_policy_error, happo_value_error, happo_error_continuous, happo_policy_error_continuous,
)
from .grpo_rloo import grpo_policy_data, grpo_policy_error, grpo_info, rloo_policy_data, rloo_policy_error, rloo_info
from .value_rescale import value_transform, value_inv_transform, symlog, inv_symlog
from .isw import compute_importance_weights
from .exploration import get_epsilon_greedy_fn, BaseNoise, GaussianNoise, OUNoise, create_noise_generator
from .adder import Adder, get_gae, get_gae_with_default_last_value, get_nstep_return_data, get_train_sample, list_split
from .beta_function import beta_function_map
from .sampler import (
    ArgmaxSampler, MultinomialSampler, MuSampler, ReparameterizationSampler, HybridStochasticSampler,
    HybridDeterminsticSampler,
)
from .log_prob_utils import naive_method, efficient_method, less_efficient_method
