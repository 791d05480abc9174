from .assigner import Assigner, BITS_COST
from .profile import fit_cost_models

__all__ = ['Assigner', 'BITS_COST', 'fit_cost_models']
