from .app import build_app, run_server
from .model_io import ModelRuntime

__all__ = ["build_app", "run_server", "ModelRuntime"]
