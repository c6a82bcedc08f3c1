#!/usr/bin/env python3
"""Diagnose corr_lookup backward: (1) adjoint identity <F(x), g> == <x, F^T(g)>
for the HIP fwd/bwd pair per level; (2) per-level comparison vs the
grid_sample reference."""

import sys

import torch

sys.path.insert(0, ".")
import flowhip._C as C
from flowhip.ops import torch_ref

dev = torch.device("cuda:0")
torch.manual_seed(5)
B, H, W, R = 1, 12, 16, 4
P = H * W

l0 = torch.randn(B * P, 1, H, W, device=dev)
pyr = [p.detach().contiguous() for p in torch_ref.corr_pyramid(l0, 4)]
coords = (torch.rand(B, 2, H, W, device=dev) *
          torch.tensor([W, H], device=dev).view(1, 2, 1, 1)).contiguous()

out = C.corr_lookup_fwd(pyr, coords, R, False)
g = torch.randn_like(out)
grads = C.corr_lookup_bwd(g, coords, R, [list(p.shape) for p in pyr], False)

lhs = (out * g).sum().item()
rhs = sum((pyr[l] * grads[l]).sum().item() for l in range(4))
print(f"adjoint: <F(x),g>={lhs:.6f}  <x,F^T(g)>={rhs:.6f}  diff={lhs-rhs:.2e}")

# per-level comparison vs reference autograd
for l in range(4):
    leaves = [p.clone().requires_grad_(True) for p in pyr]
    ref = torch_ref.corr_lookup(leaves, coords, R)
    (dr,) = torch.autograd.grad((ref * g).sum(), leaves[l])
    d = (grads[l] - dr)
    print(f"level {l}: max|diff|={d.abs().max().item():.3e} "
          f"ref_norm={dr.norm().item():.3f} hip_norm={grads[l].norm().item():.3f}")

# also check the REF adjoint against itself (sanity)
leaves = [p.clone().requires_grad_(True) for p in pyr]
ref = torch_ref.corr_lookup(leaves, coords, R)
lhs_r = (ref * g).sum().item()
dr = torch.autograd.grad((ref * g).sum(), leaves)
rhs_r = sum((pyr[l] * dr[l]).sum().item() for l in range(4))
print(f"ref adjoint: {lhs_r:.6f} vs {rhs_r:.6f} diff={lhs_r-rhs_r:.2e}")
print(f"fwd vs ref fwd: max|diff|={(out-ref).abs().max().item():.3e}")
