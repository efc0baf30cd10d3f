from .metrics import accuracy
from .checkpoint import save_model_step, load_model_step, model_step_path

__all__ = ['accuracy', 'save_model_step', 'load_model_step', 'model_step_path']
