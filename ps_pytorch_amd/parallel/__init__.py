from .flat import FlatSpace, Bucket

__all__ = ['FlatSpace', 'Bucket']
