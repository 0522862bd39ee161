"""Per-(algorithm, env) tuned preset configs for the ding.bonus agents
(reference ding/config/example/). Load via ding.config.example.get_example_config.
"""
from .loader import get_example_config, list_example_configs
