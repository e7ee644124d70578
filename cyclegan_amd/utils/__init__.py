from .summary import Summary, append_dict  # noqa: F401
from .plot import plot_cycle  # noqa: F401
