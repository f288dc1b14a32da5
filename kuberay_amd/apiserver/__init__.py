"""APIServer: v1 simplified CRUD + v2 restricted proxy (FastAPI)."""
from .app import create_app  # noqa: F401
