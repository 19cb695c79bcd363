"""npz checkpoint contract tests (SURVEY.md §5.4 key schema)."""
import numpy as np
import torch

from raft_amd import RAFT, RaftConfig
from raft_amd.utils import checkpoint as ckpt

THINGS_EXPECTED = [
    "fnet/conv1/W", "fnet/conv1/b",
    "fnet/layer1/0/conv1/W", "fnet/layer1/1/conv2/b",
    "fnet/layer2/0/downsample.0/W", "fnet/layer3/0/downsample.0/b",
    "fnet/conv2/W", "fnet/conv2/b",
    "cnet/norm1/gamma", "cnet/norm1/beta",
    "cnet/norm1/mean/EMA", "cnet/norm1/variance/EMA",
    "cnet/layer1/0/norm1/gamma", "cnet/layer2/0/downsample.1/variance/EMA",
    "update_block/encoder/convc1/W", "update_block/encoder/convc2/W",
    "update_block/encoder/convf1/W", "update_block/encoder/convf2/W",
    "update_block/encoder/conv/W",
    "update_block/gru/convz1/W", "update_block/gru/convr1/W",
    "update_block/gru/convq1/W", "update_block/gru/convz2/W",
    "update_block/gru/convr2/W", "update_block/gru/convq2/W",
    "update_block/flow_head/conv1/W", "update_block/flow_head/conv2/b",
    "update_block/mask/0/W", "update_block/mask/2/W",
]

SMALL_EXPECTED = [
    "fnet/conv1/W", "fnet/layer1/0/conv3/W", "fnet/layer2/0/downsample.0/W",
    "cnet/conv2/W",
    "update_block/encoder/convc1/W", "update_block/encoder/convf1/W",
    "update_block/encoder/convf2/W", "update_block/encoder/conv/W",
    "update_block/gru/convz/W", "update_block/gru/convr/b",
    "update_block/gru/convq/W", "update_block/flow_head/conv1/W",
]


def test_things_key_schema():
    keys = ckpt.expected_npz_keys(RAFT(RaftConfig(small=False)))
    missing = [k for k in THINGS_EXPECTED if k not in keys]
    assert not missing, missing
    # instance-norm fnet must carry NO norm params (center=False scale=False)
    assert not any("fnet" in k and ("gamma" in k or "beta" in k) for k in keys)
    # small-model-only keys must be absent
    assert "update_block/gru/convz/W" not in keys


def test_small_key_schema():
    keys = ckpt.expected_npz_keys(RAFT(RaftConfig(small=True)))
    missing = [k for k in SMALL_EXPECTED if k not in keys]
    assert not missing, missing
    # cnet norm 'none': no norm params anywhere
    assert not any("gamma" in k or "EMA" in k for k in keys)
    assert "update_block/mask/0/W" not in keys


def test_roundtrip_bitexact(tmp_path):
    for small in (False, True):
        m = RAFT(RaftConfig(small=small)).eval()
        p = str(tmp_path / f"w{int(small)}.npz")
        ckpt.save_npz(m, p)
        m2 = RAFT(RaftConfig(small=small)).eval()
        ckpt.load_npz(m2, p)
        for (k1, v1), (k2, v2) in zip(m.state_dict().items(),
                                      m2.state_dict().items()):
            assert k1 == k2
            assert torch.equal(v1, v2), k1
        x1 = torch.rand(1, 3, 64, 96)
        x2 = torch.rand(1, 3, 64, 96)
        with torch.no_grad():
            assert torch.equal(m(x1, x2, iters=2), m2(x1, x2, iters=2))


def test_hwio_layout_on_disk(tmp_path):
    """Conv weights on disk must be HWIO (TF layout)."""
    m = RAFT(RaftConfig(small=False)).eval()
    p = str(tmp_path / "w.npz")
    ckpt.save_npz(m, p)
    arr = np.load(p)
    w = arr["fnet/conv1/W"]
    assert w.shape == (7, 7, 3, 64)           # HWIO: 7x7, in 3, out 64
    assert arr["fnet/conv1/b"].shape == (64,)
    assert arr["update_block/mask/2/W"].shape == (1, 1, 256, 576)


def test_load_rejects_wrong_variant(tmp_path):
    m_small = RAFT(RaftConfig(small=True))
    p = str(tmp_path / "small.npz")
    ckpt.save_npz(m_small, p)
    m_things = RAFT(RaftConfig(small=False))
    try:
        ckpt.load_npz(m_things, p)
        assert False, "expected a loud mismatch error"
    except (KeyError, ValueError):
        pass
