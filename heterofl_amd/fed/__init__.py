from .federation import Federation, Axis, ParamSlice, FULL, PREFIX, GATHER
from .runner import FedRunner, sample_active_users
from .sequential import SequentialClientTrainer

__all__ = ['Federation', 'Axis', 'ParamSlice', 'FULL', 'PREFIX', 'GATHER',
           'FedRunner', 'sample_active_users', 'SequentialClientTrainer']
