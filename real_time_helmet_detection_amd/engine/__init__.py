from .checkpoint import save_checkpoint, load_checkpoint
from .trainer import distributed_worker, train_step, load_network
from .evaluator import single_device_evaluate, evaluate_step, Prediction

__all__ = ['save_checkpoint', 'load_checkpoint', 'distributed_worker',
           'train_step', 'load_network', 'single_device_evaluate',
           'evaluate_step', 'Prediction']
