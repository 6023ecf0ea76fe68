from .stochastic_adamw import StochasticAdamW

__all__ = ["StochasticAdamW"]
