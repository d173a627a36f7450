from .split import annotate_split_points, split_into_equal_size  # noqa: F401
from .api import _compile_pp  # noqa: F401
