from .dp import DistContext, GradSync  # noqa: F401
from .flat import FlatParamGroup  # noqa: F401
