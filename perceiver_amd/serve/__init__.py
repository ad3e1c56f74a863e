from perceiver_amd.serve.server import create_app

__all__ = ["create_app"]
