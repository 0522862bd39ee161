from .stocks_env import StocksEnv
