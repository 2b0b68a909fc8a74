"""Debug the sphere-topk raytrace mismatch vs the CPU oracle."""
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from gcbfplus_amd import ops
from gcbfplus_amd.env import make_env
from gcbfplus_amd.env.utils import get_lidar, beam_dirs_3d

env = make_env("LinearDrone", num_agents=6, area_size=3.0, max_step=4, device="cuda")
rng = np.random.default_rng(70)
obs = env.sample_obstacles(4, rng)
pos = torch.rand(4, 6, 3) * 3.0
pos[0, 0] = obs.center[0, 0]
nb, tk, comm = env._params["n_rays"], env.N_HIT_RETURNS, env._params["comm_radius"]
hg = ops.raytrace_sphere_topk(pos.cuda(), obs.center.cuda(), obs.radius.cuda(),
                              nb, tk, comm).cpu()
hc = get_lidar(pos, obs, nb, comm, max_returns=tk)
hgf = hg.reshape(24, tk, 3)
hcf = hc.reshape(24, tk, 3)
d = torch.cdist(hgf, hcf)
scale = hcf.abs().amax(-1).clamp_min(1.0)
m1 = (d.min(dim=2).values / scale)
m2 = (d.min(dim=1).values / scale)
print("max gpu->cpu", m1.max().item(), "max cpu->gpu", m2.max().item())
bad = (m1.max(dim=1).values > 1e-3).nonzero().flatten()
print("bad rows:", bad.tolist()[:5])
for r in bad.tolist()[:2]:
    print(f"row {r}: agent pos {pos.reshape(24,3)[r].tolist()}")
    i = m1[r].argmax().item()
    print("  worst gpu hit:", hgf[r, i].tolist(), "rel err", m1[r, i].item())
    print("  gpu hits:", hgf[r, :4].tolist())
    print("  cpu hits:", hcf[r, :4].tolist())
    # alpha comparison: recompute CPU alphas for this agent
    dirs = beam_dirs_3d(nb)
    b_, n_ = r // 6, r % 6
    starts = pos[b_, n_][None].expand(dirs.shape[0], 3)
    ends = starts + comm * dirs
    al = obs.__class__(*[t[b_:b_+1] for t in obs]).raytrace(
        starts[None], ends[None])[0]
    srt = al.sort()
    print("  cpu alpha top8:", srt.values[:8].tolist())
    print("  cpu alpha idx8:", srt.indices[:8].tolist())
print("DONE")
