from .synthetic import SyntheticDataset, prepare_data, ResidentLoader

__all__ = ['SyntheticDataset', 'prepare_data', 'ResidentLoader']
