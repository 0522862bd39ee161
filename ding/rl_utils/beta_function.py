"""Risk-distortion beta functions for IQN.

Parity: reference ding/rl_utils/beta_function.py (cpw, CVaR, beta_function_map).
"""
import torch

beta_function_map = {}

beta_function_map['uniform'] = lambda x: x
# cumulative probability weighting (Tversky & Kahneman 1992)
beta_function_map['CPW'] = lambda x, eta=0.71: (x ** eta) / ((x ** eta + (1 - x) ** eta) ** (1 / eta))
# conditional value at risk (risk-averse)
beta_function_map['CVaR'] = lambda x, eta=0.71: x * eta
beta_function_map['Pow'] = lambda x, eta=0.0: x ** (1 / (1 + abs(eta))) if eta >= 0 \
    else 1 - (1 - x) ** (1 / (1 + abs(eta)))
