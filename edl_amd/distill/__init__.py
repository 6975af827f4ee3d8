from .reader import DistillReader
