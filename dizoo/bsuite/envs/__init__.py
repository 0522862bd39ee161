from .bsuite_env import BSuiteEnv
