from . import constants  # noqa: F401
from .types import *  # noqa: F401,F403
