from .engine import Database, build_engine, run_migrations  # noqa: F401
from . import models  # noqa: F401
