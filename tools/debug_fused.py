"""GPU debug: stage-by-stage comparison of the fused NHWC path vs the
eager bf16 path on identical inputs. Run via gpurun."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from raft_amd import RAFT, RaftConfig
from raft_amd.models import fused
from raft_amd.ops import require_hip


def p(name, a, b):
    a = a.float()
    b = b.float()
    print(f"{name:28s} max {(a-b).abs().max().item():.5f} "
          f"mean {(a-b).abs().mean().item():.6f} "
          f"|ref| {b.abs().mean().item():.4f}")


def main():
    dev = torch.device("cuda:0")
    hip = require_hip()
    torch.manual_seed(11)
    m = RAFT(RaftConfig(small=False)).to(dev).to(torch.bfloat16).eval()
    B, H8, W8 = 1, 8, 12
    cfg = m.cfg
    ub = m.update_block

    net = (torch.randn(B, 128, H8, W8, device=dev) * 0.5).to(torch.bfloat16)
    inp = (torch.randn(B, 128, H8, W8, device=dev) * 0.5).to(torch.bfloat16)
    corr = (torch.randn(B, 324, H8, W8, device=dev)).to(torch.bfloat16)
    flow = (torch.randn(B, 2, H8, W8, device=dev)).to(torch.bfloat16)

    with torch.no_grad():
        ref_net, ref_mask, ref_df = ub(net, inp, corr, flow)
        # eager motion encoder internals
        cor_e = torch.relu(ub.encoder.convc1(corr))
        cor_e2 = torch.relu(ub.encoder.convc2(cor_e))
        flo_e = torch.relu(ub.encoder.convf1(flow))
        flo_e2 = torch.relu(ub.encoder.convf2(flo_e))
        mo_e = torch.relu(ub.encoder.conv(torch.cat([cor_e2, flo_e2], 1)))

    fu = fused.FusedBasicUpdate(ub, 328, cfg.context_dim)
    # physical NHWC inputs
    netp = net.permute(0, 2, 3, 1).contiguous()
    inpp = inp.permute(0, 2, 3, 1).contiguous()
    flowp = flow.permute(0, 2, 3, 1).contiguous()
    corrp = torch.zeros(B, H8, W8, 328, device=dev, dtype=torch.bfloat16)
    corrp[..., :324] = corr.permute(0, 2, 3, 1)

    with torch.no_grad():
        cor_f = fu.c1(hip, corrp)
        p("convc1", cor_f.permute(0, 3, 1, 2), cor_e)
        cor_f2 = fu.c2(hip, cor_f)
        p("convc2", cor_f2.permute(0, 3, 1, 2), cor_e2)
        fpad = torch.zeros(B, H8, W8, 8, device=dev, dtype=torch.bfloat16)
        fpad[..., 0:2] = flowp
        flo_f = hip.fconv_smallk(fpad, fu.f1.wp, fu.f1.bias, fu.f1.kh,
                                 fu.f1.kw, fused.ACT_RELU, 0, 2)
        p("convf1", flo_f.permute(0, 3, 1, 2), flo_e)
        flo_f2 = fu.f2(hip, flo_f)
        p("convf2", flo_f2.permute(0, 3, 1, 2), flo_e2)

        x_buf = torch.empty(B, H8, W8, 256, device=dev, dtype=torch.bfloat16)
        x_buf[..., :128] = inpp
        x_buf[..., 254:256] = flowp
        fu.cv(hip, cor_f2, flo_f2, fused.ACT_RELU, out=x_buf, n_off=128)
        p("motion.conv", x_buf[..., 128:254].permute(0, 3, 1, 2), mo_e)

        x2 = torch.empty_like(x_buf)
        x2[..., :128] = inpp
        x2[..., 254:256] = flowp    # the loop's lookup writes this slice
        coords = torch.randn(B, H8, W8, 2, device=dev)
        newnet, mask, cnew = fu(hip, netp, x2, corrp, coords)
        p("update.net", newnet.permute(0, 3, 1, 2), ref_net)
        p("update.mask", mask.permute(0, 3, 1, 2), ref_mask)
        p("update.dflow", (cnew - coords).permute(0, 3, 1, 2),
          ref_df.float())

    # full model comparison fused vs eager-bf16 (no fuse)
    x1 = torch.rand(1, 3, 64, 96, device=dev, dtype=torch.bfloat16)
    x2i = torch.rand(1, 3, 64, 96, device=dev, dtype=torch.bfloat16)
    with torch.no_grad():
        os.environ["RAFT_AMD_NO_FUSE"] = "1"
        ref = m(x1, x2i, iters=6)
        os.environ["RAFT_AMD_NO_FUSE"] = "0"
        out = m(x1, x2i, iters=6)
    p("full model (6 it, vs eager-bf16)", out, ref)

    for it in (1, 2, 4):
        with torch.no_grad():
            os.environ["RAFT_AMD_NO_FUSE"] = "1"
            r = m(x1, x2i, iters=it)
            os.environ["RAFT_AMD_NO_FUSE"] = "0"
            o = m(x1, x2i, iters=it)
        p(f"full model iters={it}", o, r)


if __name__ == "__main__":
    main()
