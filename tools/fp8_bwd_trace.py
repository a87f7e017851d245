"""Trace the fp8 layer backwards: per-call dH norms, state identity, and
amax values immediately after the relu_bwd recording."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import mpgcn_amd.ops.functional as F
from tools.fp8_conv_probe import run

orig = F._BDGCNLayerFp8Fn.backward


def dbg(ctx, dH, dY8=None):
    st = ctx.fp8_state
    pre = (dH.float().norm().item(), st["amax_y"].item(), st["scale_y"].item())
    out = orig(ctx, dH, dY8)
    torch.cuda.synchronize()
    print("  bwd st=%x dH %.3e | pre amax %.3e scale %.3e | post amax %.3e "
          "dXnorm %.3e" % (id(st) & 0xffff, pre[0], pre[1], pre[2],
                           st["amax_y"].item(), out[0].float().norm().item()))
    return out


F._BDGCNLayerFp8Fn.backward = staticmethod(dbg)
run("fp8", True, steps=3)
