from .spaces import Space, Discrete, Box, MultiDiscrete, Dict
