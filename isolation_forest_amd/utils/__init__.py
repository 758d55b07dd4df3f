from .math import avg_path_length, EULER_CONSTANT
from .params import ParamSpec, Params

__all__ = ["avg_path_length", "EULER_CONSTANT", "ParamSpec", "Params"]
