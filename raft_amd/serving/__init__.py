from raft_amd.serving.server import create_app  # noqa: F401
