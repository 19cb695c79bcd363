from raft_amd.parallel.ddp import BucketedDDP, init_distributed  # noqa: F401
