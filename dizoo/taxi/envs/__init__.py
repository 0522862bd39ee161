from .taxi_env import TaxiEnv
