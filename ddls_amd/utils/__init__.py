from .misc import Stopwatch, Sampler, seed_everything, flatten_list
from .registry import get_class_from_path

__all__ = ["Stopwatch", "Sampler", "seed_everything", "flatten_list",
           "get_class_from_path"]
