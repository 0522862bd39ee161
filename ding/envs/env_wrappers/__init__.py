from .env_wrappers import *
