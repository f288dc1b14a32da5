"""kray CLI (kubectl-plugin analog)."""
from .main import cli, main  # noqa: F401
