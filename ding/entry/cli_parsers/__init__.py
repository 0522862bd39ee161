from .slurm_parser import SlurmParser, slurm_parser
from .k8s_parser import K8SParser, k8s_parser

PLATFORM_PARSERS = {'slurm': slurm_parser, 'k8s': k8s_parser}
