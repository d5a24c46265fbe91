from .server import EndpointStatuses, build_app, run_watchman

__all__ = ["EndpointStatuses", "build_app", "run_watchman"]
