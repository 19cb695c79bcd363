from raft_amd.models.raft import RAFT, RaftConfig  # noqa: F401
