from . import params, textio  # noqa: F401
from .params import Params, ParamsError  # noqa: F401
