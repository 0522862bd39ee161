"""Command-mode wrappers binding eps/lr schedules to every policy.

Parity: reference ding/policy/command_mode_policy_instance.py (~60
registrations '<name>_command'). Here the wrappers are generated
programmatically: eps-scheduled policies get EpsCommandModePolicy, the rest
DummyCommandModePolicy.
"""
from ding.rl_utils import get_epsilon_greedy_fn
from ding.utils import POLICY_REGISTRY
from .base_policy import CommandModePolicy


class EpsCommandModePolicy(CommandModePolicy):

    def _init_command(self) -> None:
        eps_cfg = self._cfg.other.eps
        self.epsilon_greedy = get_epsilon_greedy_fn(eps_cfg.start, eps_cfg.end, eps_cfg.decay, eps_cfg.type)

    def _get_setting_collect(self, command_info: dict) -> dict:
        learner_step = command_info.get('envstep', 0)
        return {'eps': self.epsilon_greedy(learner_step)}

    def _get_setting_learn(self, command_info: dict) -> dict:
        return {}

    def _get_setting_eval(self, command_info: dict) -> dict:
        return {}


class DummyCommandModePolicy(CommandModePolicy):

    def _init_command(self) -> None:
        pass

    def _get_setting_collect(self, command_info: dict) -> dict:
        return {}

    def _get_setting_learn(self, command_info: dict) -> dict:
        return {}

    def _get_setting_eval(self, command_info: dict) -> dict:
        return {}


_EPS_POLICIES = [
    'dqn', 'dqn_stdim', 'c51', 'qrdqn', 'iqn', 'fqf', 'rainbow', 'sql', 'mdqn', 'bdq', 'r2d2', 'r2d2_gtrxl',
    'r2d3', 'sqn', 'dqfd', 'ngu', 'pdqn', 'wqmix', 'qmix', 'collaq', 'coma', 'qtran', 'madqn', 'gtrxl_discrete_sac',
]


def _register_command_variants():
    """Create '<name>_command' classes for every registered base policy."""
    from ding.utils.registry import POLICY_REGISTRY as REG
    for name in list(REG.keys()):
        if name.endswith('_command'):
            continue
        cmd_name = name + '_command'
        if cmd_name in REG:
            continue
        base_cls = REG.get(name)
        mixin = EpsCommandModePolicy if name in _EPS_POLICIES else DummyCommandModePolicy
        cmd_cls = type(base_cls.__name__ + 'Command', (mixin, base_cls), {})
        REG.register(cmd_name)(cmd_cls)


_register_command_variants()
